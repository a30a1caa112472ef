"""InfluxDB line-protocol HTTP metrics sink against a localhost stub:
asserts real line-protocol bodies on /write?db=... and the lossy-under-
pressure semantics of the bounded dispatch queue (reference
metrics/recorders/influxdb/service.rs:10-16)."""
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from xaynet_amd import _core

co = _core.coordinator


class _InfluxStub(BaseHTTPRequestHandler):
    def log_message(self, *a):
        pass

    def do_POST(self):
        n = int(self.headers.get("Content-Length", "0"))
        body = self.rfile.read(n)
        with self.server.lock:
            self.server.writes.append((self.path, body))
            delay = self.server.delay
        if delay:
            time.sleep(delay)
        self.send_response(204)
        self.send_header("Content-Length", "0")
        self.end_headers()


@pytest.fixture
def influx():
    srv = ThreadingHTTPServer(("127.0.0.1", 0), _InfluxStub)
    srv.writes = []
    srv.delay = 0.0
    srv.lock = threading.Lock()
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield srv
    srv.shutdown()
    co.uninstall_metrics()


def test_line_protocol_bodies(influx):
    co.install_metrics_influxdb("127.0.0.1", influx.server_port, db="metrics")
    # drive real metric emission through a coordinator phase
    s = co.Settings()
    s.model_length = 4
    c = _core.mask.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = _core.mask.MaskConfigPair(c, c)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    coord.run_one_phase()  # Idle: emits round_total_number + params + phase
    coord.stop()
    co.metrics_flush()
    time.sleep(0.3)
    with influx.lock:
        writes = list(influx.writes)
    assert writes, "no metrics reached the endpoint"
    paths = {p for p, _ in writes}
    assert paths == {"/write?db=metrics"}
    bodies = b"\n".join(b for _, b in writes).decode()
    # line protocol: measurement,tag=v value=x timestamp
    assert "round_total_number,round_id=1,phase=0 value=1" in bodies
    assert "round_param_sum,round_id=1,phase=0" in bodies
    for _, b in writes:
        parts = b.decode().split(" ")
        assert len(parts) == 3 and parts[1].startswith("value=") and parts[2].isdigit()


def test_lossy_under_pressure(influx):
    """A slow endpoint must shed metrics, never block the emitting thread."""
    influx.delay = 0.2
    co.install_metrics_influxdb("127.0.0.1", influx.server_port, db="m")
    s = co.Settings()
    s.model_length = 4
    c = _core.mask.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = _core.mask.MaskConfigPair(c, c)
    s.set_sum(1, 1, 0.0, 0.005)  # sum phase times out instantly -> Failure -> Idle
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    t0 = time.time()
    # each Idle/Sum/Failure cycle emits ~6 points against a ~5-write/s
    # endpoint; emission must never block the protocol thread
    # the queue cap is 4048 (reference Buffer<4048>): cycle until emission
    # clearly exceeds it so the shed path is exercised
    for _ in range(3000):
        coord.run_one_phase()
        if time.time() - t0 > 12:
            break
    emit_wall = time.time() - t0
    coord.stop()
    assert emit_wall < 20.0, "metric emission blocked on the slow endpoint"
    assert co.metrics_dropped() > 0, "expected load shedding on a slow endpoint"
