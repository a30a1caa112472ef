"""Native-participant shim: the module the reference's Python SDK calls
`xaynet_sdk.xaynet_sdk` (PyO3 module, bindings/python/src/python_ffi.rs:25-294).

Here it wraps the xaynet_amd C++ participant + native HTTP client. The public
surface is API-compatible: `Participant(url, scalar, state)`, `tick()`,
`set_model(list)`, `global_model()`, `made_progress()`, `should_set_model()`,
`new_global_model()`, `task()`, `save()`, `init_logging()`, and the exception
taxonomy.
"""
from __future__ import annotations

import logging
import os
import secrets
import threading
from fractions import Fraction
from typing import List, Optional
from urllib.parse import urlparse

import numpy as np

from xaynet_amd import _core

__all__ = [
    "Participant",
    "init_logging",
    "CryptoInit",
    "ParticipantInit",
    "ParticipantRestore",
    "UninitializedParticipant",
    "GlobalModelUnavailable",
    "GlobalModelDataTypeMisMatch",
    "LocalModelLengthMisMatch",
    "LocalModelDataTypeMisMatch",
]


class CryptoInit(Exception):
    """Initialization of the crypto library failed."""


class ParticipantInit(Exception):
    """The participant could not be initialized."""


class ParticipantRestore(Exception):
    """The serialized participant state is invalid."""


class UninitializedParticipant(Exception):
    """Method call on a consumed (stopped) participant."""


class GlobalModelUnavailable(Exception):
    """The coordinator could not be reached / returned no model."""


class GlobalModelDataTypeMisMatch(Exception):
    """The global model's data type differs from the round configuration."""


class LocalModelLengthMisMatch(Exception):
    """The local model's length differs from the round configuration."""


class LocalModelDataTypeMisMatch(Exception):
    """The local model's data type differs from the round configuration."""


_STATE_MAGIC = b"XAYP"
_DTYPES = {0: np.float32, 1: np.float64, 2: np.int32, 3: np.int64}


def init_logging():
    """Reference parity: initialize native-side logging (python_ffi.rs:283-294).
    The native core logs through Python's `logging` under 'xaynet_amd'."""
    logging.getLogger("xaynet_amd").addHandler(logging.NullHandler())


def _parse_url(url: str):
    if "//" not in url:
        url = "http://" + url
    u = urlparse(url)
    if not u.hostname:
        raise ParticipantInit(f"invalid coordinator url: {url!r}")
    return u.hostname, u.port or 80


def _scalar_fraction(scalar: float):
    if not (0.0 < scalar <= 1.0):
        raise ParticipantInit("scalar must be in (0, 1]")
    f = Fraction(scalar)
    if f.denominator >= 2**63:
        f = f.limit_denominator(2**62)
    return f.numerator, f.denominator


class Participant:
    """API-compatible wrapper of the native participant state machine."""

    def __init__(self, coordinator_url: str, scalar: float = 1.0, state=None):
        host, port = _parse_url(coordinator_url)
        self._client = _core.rest.HttpXaynetClient(host, port)
        num, den = _scalar_fraction(scalar)
        self._lock = threading.Lock()
        self._consumed = False
        if state is None:
            self._sign_seed = secrets.token_bytes(32)
            self._scalar = (num, den)
            self._inner = _core.sdk.Participant(
                self._sign_seed, num, den, self._client
            )
        else:
            raw = bytes(state)
            if raw[:4] != _STATE_MAGIC:
                # not the native envelope: a reference-client checkpoint
                # (xaynet-mobile SerializableState bincode, whose first 4
                # bytes are a u32 phase variant 0..7, never "XAYP")
                try:
                    self._inner = _core.sdk.Participant.restore_reference(
                        raw, self._client
                    )
                except RuntimeError as e:
                    raise ParticipantRestore(str(e)) from e
                self._sign_seed = b"\x00" * 32  # keys live inside the state
                self._scalar = (num, den)
                return
            if len(raw) < 4 + 32 + 16:
                raise ParticipantRestore("invalid serialized participant state")
            self._sign_seed = raw[4:36]
            snum = int.from_bytes(raw[36:44], "little")
            sden = int.from_bytes(raw[44:52], "little")
            self._scalar = (snum, sden)
            try:
                self._inner = _core.sdk.Participant.restore(
                    raw[52:], self._client, self._sign_seed, snum, sden
                )
            except RuntimeError as e:
                raise ParticipantRestore(str(e)) from e

    # -- internal helpers -------------------------------------------------

    def _check(self):
        if self._consumed:
            raise UninitializedParticipant("participant was stopped/consumed")

    # -- reference API ----------------------------------------------------

    def tick(self):
        self._check()
        self._inner.tick()

    def made_progress(self) -> bool:
        self._check()
        return self._inner.made_progress

    def should_set_model(self) -> bool:
        self._check()
        return self._inner.should_set_model

    def new_global_model(self) -> bool:
        self._check()
        return self._inner.new_global_model

    def task(self):
        self._check()
        return self._inner.task

    def set_model(self, local_model: list):
        self._check()
        dt = self._inner.model_data_type
        length = self._inner.model_length
        if dt < 0:
            # no round parameters yet: cache as f32 and let the native side
            # re-validate at send time
            dt = 0
        arr = np.asarray(local_model, dtype=_DTYPES[dt])
        if length and arr.shape != (length,):
            raise LocalModelLengthMisMatch(
                f"local model has {arr.size} weights, round expects {length}"
            )
        if not np.issubdtype(np.asarray(local_model).dtype, np.number):
            raise LocalModelDataTypeMisMatch("local model must be numeric")
        self._inner.set_model(arr)

    def global_model(self) -> Optional[list]:
        self._check()
        body = self._inner.global_model_bincode()
        if body is None:
            return None
        dt = self._inner.model_data_type
        if dt < 0:
            raise GlobalModelUnavailable("round parameters unknown")
        arr = _core.sdk.decode_model(body, dt)
        if arr is None:
            return None
        return arr.tolist()

    def save(self) -> List[int]:
        """Serialize and consume the participant (reference semantics)."""
        self._check()
        self._consumed = True
        num, den = self._scalar
        blob = (
            _STATE_MAGIC
            + self._sign_seed
            + num.to_bytes(8, "little")
            + den.to_bytes(8, "little")
            + self._inner.save()
        )
        return list(blob)
