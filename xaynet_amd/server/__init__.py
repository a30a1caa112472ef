"""Coordinator daemon: settings -> storage -> C++ coordinator + native REST
server (the reference's bin/main.rs equivalent: load settings, init metrics
and storage, restore if enabled, serve until interrupted).

Run: `python -m xaynet_amd.server -c configs/config.toml`
"""
from __future__ import annotations

import logging
import signal
import threading

from xaynet_amd import _core

from .settings import Settings, SettingsError  # noqa: F401

LOG = logging.getLogger("xaynet.server")

co = _core.coordinator


def build_coordinator(settings: Settings):
    """Construct (coordinator, store, models) from validated settings."""
    s = co.Settings()
    s.sum_prob = settings.sum.prob
    s.update_prob = settings.update.prob
    s.model_length = settings.model_length
    c = _core.mask.MaskConfig(*settings.mask_config_args())
    s.mask_cfg = _core.mask.MaskConfigPair(c, c)
    s.set_sum(settings.sum.count.min, settings.sum.count.max,
              settings.sum.time.min, settings.sum.time.max)
    s.set_update(settings.update.count.min, settings.update.count.max,
                 settings.update.time.min, settings.update.time.max)
    s.set_sum2(settings.sum2.count.min, settings.sum2.count.max,
               settings.sum2.time.min, settings.sum2.time.max)
    s.restore = settings.restore_enable

    if settings.redis_url:
        host, port = _parse_hostport(settings.redis_url, 6379)
        store = co.RedisStorage(host, port)
        models = co.RedisModels(host, port)
    elif settings.storage_path:
        store = co.FileStorage(settings.storage_path + "/coordinator")
        models = co.FileModels(settings.storage_path + "/global-models")
    else:
        store = co.InMemoryStorage()
        models = co.InMemoryModels()

    if settings.s3_url:
        host, port = _parse_hostport(settings.s3_url, 9000)
        models = co.S3Models(host, port, settings.s3_bucket)

    coordinator = co.Coordinator(s, store, models, settings.gpu)
    return coordinator, store, models


def _parse_hostport(url: str, default_port: int):
    """'redis://host:port' / 'http://host:port' / 'host:port' -> (host, port)."""
    rest = url.split("://", 1)[-1].rstrip("/")
    host, _, port = rest.partition(":")
    return host, int(port) if port else default_port


def serve(settings: Settings, ready_event: threading.Event | None = None,
          stop_event: threading.Event | None = None):
    """Run the coordinator + REST server until `stop_event` (or SIGINT)."""
    logging.basicConfig(level=getattr(logging, settings.log_filter.upper(), logging.INFO)
                        if settings.log_filter.upper() in ("DEBUG", "INFO", "WARNING", "ERROR")
                        else logging.INFO)

    if settings.trace_file:
        co.install_trace_file(settings.trace_file)
    if settings.metrics.enable and settings.metrics.url.startswith("file:"):
        co.install_metrics_file(settings.metrics.url[len("file:"):])
    elif settings.metrics.enable and settings.metrics.url.startswith("http"):
        host, port = _parse_hostport(settings.metrics.url, 8086)
        co.install_metrics_influxdb(host, port, settings.metrics.db)

    driver = None
    if settings.gpu:
        c = _core.mask.MaskConfig(*settings.mask_config_args())
        if c.bytes_per_number > 16:
            LOG.warning(
                "mask config %s has a group order wider than 128 bits; the GPU "
                "data plane covers orders up to 2^128 — using the CPU "
                "aggregation plane for this deployment", settings.mask.data_type)
            settings.gpu = False
    coordinator, store, models = build_coordinator(settings)
    if settings.gpu:
        c = _core.mask.MaskConfig(*settings.mask_config_args())
        n = settings.gpu_devices
        if n == 0:
            from xaynet_amd.ops import gpu_available

            from xaynet_amd import _hip

            n = _hip.device_count() if gpu_available() else 0
        if n > 1:
            # the serve plane itself is multi-GPU: one worker process per
            # device, sharded unmask over RCCL/xGMI
            from xaynet_amd.parallel.serve import MultiGpuServeDriver

            driver = MultiGpuServeDriver(coordinator, c, c, settings.model_length,
                                         n_workers=n, device_kind="cuda")
            LOG.info("multi-GPU serve plane: %d devices", n)
        else:
            from xaynet_amd.ops import make_coordinator_driver

            driver = make_coordinator_driver(coordinator, c, c, settings.model_length)
        driver.start()
    host, port = settings.bind_host_port()
    if settings.api.tls_certificate:
        server = _core.rest.RestServer(
            coordinator, host, port, settings.api.tls_certificate,
            settings.api.tls_key or "", settings.api.tls_client_auth or "")
    else:
        server = _core.rest.RestServer(coordinator, host, port, settings.api.workers)
    if not server.start():
        raise RuntimeError(f"failed to bind {host}:{port}")
    coordinator.start()
    LOG.info("coordinator serving on %s:%d", host, server.port)
    if ready_event is not None:
        ready_event.set()

    stop = stop_event or threading.Event()
    if stop_event is None:
        signal.signal(signal.SIGINT, lambda *_: stop.set())
        signal.signal(signal.SIGTERM, lambda *_: stop.set())
    try:
        while not stop.wait(0.2):
            pass
    finally:
        coordinator.stop()
        server.stop()
        if driver is not None:
            driver.stop()
        co.uninstall_metrics()
        co.uninstall_trace()
    return coordinator
