"""Coordinator settings: layered TOML + `XAYNET__`-prefixed environment
overrides + cross-field validation.

Mirrors the reference's settings surface and invariants
(rust/xaynet-server/src/settings/mod.rs:45-371, configs/config.toml):
sections [log] [api] [pet.{sum,update,sum2}] [mask] [model]
[metrics.influxdb] [restore] [redis] [s3] plus the local storage section
([storage] path). [redis] url selects the RESP-backed coordinator storage
(reference [redis] section); [storage] path the durable local store;
neither -> in-memory.

Env override syntax (settings/mod.rs:76-83): `XAYNET__` prefix, `__` as the
section separator, e.g. `XAYNET__API__BIND_ADDRESS=0.0.0.0:8081`,
`XAYNET__PET__SUM__PROB=0.3`.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Optional

import tomli

SUM_COUNT_MIN = 1  # reference message.rs:17-21
UPDATE_COUNT_MIN = 3

GROUP_TYPES = {"Integer": 0, "Prime": 1, "Power2": 2}
DATA_TYPES = {"F32": 0, "F64": 1, "I32": 2, "I64": 3}
BOUND_TYPES = {"B0": 0, "B2": 2, "B4": 4, "B6": 6, "Bmax": 255}
MODEL_TYPES = {"M3": 3, "M6": 6, "M9": 9, "M12": 12}


class SettingsError(ValueError):
    """Invalid or inconsistent settings."""


@dataclass
class CountRange:
    min: int = 1
    max: int = 100


@dataclass
class TimeRange:
    min: float = 0.0
    max: float = 3600.0


@dataclass
class PhaseSettings:
    prob: float = 0.5
    count: CountRange = field(default_factory=CountRange)
    time: TimeRange = field(default_factory=TimeRange)


@dataclass
class MaskSettings:
    group_type: str = "Prime"
    data_type: str = "F32"
    bound_type: str = "B0"
    model_type: str = "M3"


@dataclass
class ApiSettings:
    bind_address: str = "127.0.0.1:8081"
    workers: int = 4
    tls_certificate: Optional[str] = None
    tls_key: Optional[str] = None
    tls_client_auth: Optional[str] = None  # CA for mutual TLS


@dataclass
class MetricsSettings:
    enable: bool = False
    url: str = ""          # influxdb endpoint (http://host:port) or file: path
    db: str = "metrics"


@dataclass
class Settings:
    log_filter: str = "info"
    trace_file: Optional[str] = None    # [log] trace_file: span sink (logfmt lines)
    api: ApiSettings = field(default_factory=ApiSettings)
    sum: PhaseSettings = field(default_factory=lambda: PhaseSettings(0.5, CountRange(1, 100), TimeRange(5, 3600)))
    update: PhaseSettings = field(default_factory=lambda: PhaseSettings(0.9, CountRange(3, 10000), TimeRange(10, 3600)))
    sum2: PhaseSettings = field(default_factory=lambda: PhaseSettings(1.0, CountRange(1, 100), TimeRange(5, 3600)))
    mask: MaskSettings = field(default_factory=MaskSettings)
    model_length: int = 4
    metrics: MetricsSettings = field(default_factory=MetricsSettings)
    restore_enable: bool = False
    storage_path: Optional[str] = None  # None -> in-memory only
    redis_url: Optional[str] = None     # "redis://host:port" -> RESP backend
    s3_url: Optional[str] = None        # "http://host:port" -> S3 model store
    s3_bucket: str = "global-models"
    gpu: bool = False                   # aggregate on the MI355X data plane
    gpu_devices: int = 0                # 0 = all visible GPUs; N = use N devices

    # ------------------------------------------------------------ loading

    @classmethod
    def load(cls, path: Optional[str] = None, env: Optional[dict] = None) -> "Settings":
        raw: dict = {}
        if path is not None:
            with open(path, "rb") as f:
                raw = tomli.load(f)
        _apply_env_overrides(raw, os.environ if env is None else env)
        s = cls._from_dict(raw)
        s.validate()
        return s

    @classmethod
    def _from_dict(cls, raw: dict) -> "Settings":
        s = cls()
        log = raw.get("log", {})
        s.log_filter = str(log.get("filter", s.log_filter))
        s.trace_file = log.get("trace_file")
        api = raw.get("api", {})
        s.api.bind_address = str(api.get("bind_address", s.api.bind_address))
        s.api.workers = int(api.get("workers", s.api.workers))
        s.api.tls_certificate = api.get("tls_certificate")
        s.api.tls_key = api.get("tls_key")
        s.api.tls_client_auth = api.get("tls_client_auth")
        pet = raw.get("pet", {})
        for name in ("sum", "update", "sum2"):
            sec = pet.get(name, {})
            ph: PhaseSettings = getattr(s, name)
            if "prob" in sec:
                ph.prob = float(sec["prob"])
            cnt = sec.get("count", {})
            ph.count = CountRange(int(cnt.get("min", ph.count.min)), int(cnt.get("max", ph.count.max)))
            tim = sec.get("time", {})
            ph.time = TimeRange(float(tim.get("min", ph.time.min)), float(tim.get("max", ph.time.max)))
        mask = raw.get("mask", {})
        for k in ("group_type", "data_type", "bound_type", "model_type"):
            if k in mask:
                setattr(s.mask, k, str(mask[k]))
        model = raw.get("model", {})
        s.model_length = int(model.get("length", s.model_length))
        met = raw.get("metrics", {}).get("influxdb", {})
        if met:
            s.metrics.enable = True
            s.metrics.url = str(met.get("url", ""))
            s.metrics.db = str(met.get("db", "metrics"))
        restore = raw.get("restore", {})
        s.restore_enable = bool(restore.get("enable", False))
        storage = raw.get("storage", {})
        s.storage_path = storage.get("path")
        redis = raw.get("redis", {})
        s.redis_url = redis.get("url")
        s3 = raw.get("s3", {})
        s.s3_url = s3.get("url")
        s.s3_bucket = str(s3.get("bucket", s.s3_bucket))
        gpu = raw.get("gpu", {})
        s.gpu = bool(gpu.get("enable", False))
        s.gpu_devices = int(gpu.get("devices", 0))
        return s

    # ---------------------------------------------------------- validation

    def validate(self) -> None:
        """Cross-field invariants (reference settings/mod.rs:307-371)."""
        su, up, s2 = self.sum, self.update, self.sum2
        if not (
            SUM_COUNT_MIN <= su.count.min <= su.count.max
            and UPDATE_COUNT_MIN <= up.count.min <= up.count.max
            and SUM_COUNT_MIN <= s2.count.min <= s2.count.max
            and s2.count.min <= su.count.max
            and s2.count.max <= su.count.max
        ):
            raise SettingsError("invalid phase count range(s)")
        if not (
            su.time.min <= su.time.max
            and up.time.min <= up.time.max
            and s2.time.min <= s2.time.max
        ):
            raise SettingsError("invalid phase time range(s)")
        combined = su.prob + up.prob - su.prob * up.prob
        if not (0.0 < su.prob < 1.0 and 0.0 < up.prob <= 1.0 and 0.0 < combined <= 1.0):
            raise SettingsError("starvation: invalid sum/update probabilities")
        for k, table in (
            ("group_type", GROUP_TYPES),
            ("data_type", DATA_TYPES),
            ("bound_type", BOUND_TYPES),
            ("model_type", MODEL_TYPES),
        ):
            if getattr(self.mask, k) not in table:
                raise SettingsError(f"invalid mask {k}: {getattr(self.mask, k)!r}")
        if self.model_length < 1:
            raise SettingsError("model length must be >= 1")
        host, _, port = self.api.bind_address.rpartition(":")
        if not host or not port.isdigit() or not (0 <= int(port) < 65536):
            raise SettingsError(f"invalid api.bind_address {self.api.bind_address!r}")

    # ---------------------------------------------------------- conversion

    def mask_config_args(self):
        m = self.mask
        return (
            GROUP_TYPES[m.group_type],
            DATA_TYPES[m.data_type],
            BOUND_TYPES[m.bound_type],
            MODEL_TYPES[m.model_type],
        )

    def bind_host_port(self):
        host, _, port = self.api.bind_address.rpartition(":")
        return host, int(port)


def _apply_env_overrides(raw: dict, env) -> None:
    """`XAYNET__A__B__C=v` -> raw[a][b][c] = parsed(v)."""
    for key, val in env.items():
        if not key.startswith("XAYNET__"):
            continue
        parts = [p.lower() for p in key[len("XAYNET__"):].split("__") if p]
        if not parts:
            continue
        node = raw
        for p in parts[:-1]:
            node = node.setdefault(p, {})
            if not isinstance(node, dict):
                raise SettingsError(f"env override {key} conflicts with non-table value")
        node[parts[-1]] = _parse_env_value(val)


def _parse_env_value(v: str):
    low = v.strip().lower()
    if low in ("true", "false"):
        return low == "true"
    try:
        return int(v)
    except ValueError:
        pass
    try:
        return float(v)
    except ValueError:
        pass
    return v
