"""Coordinator restart with live participants: durable state restore +
round-freshness recovery (reference initializer.rs restore path +
xaynet-sdk phase.rs:160-200 freshness check), and the daemon CLI."""
import os
import signal
import subprocess
import sys
import time

import numpy as np

from xaynet_amd import _core
from xaynet_amd.server import Settings, build_coordinator

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask
rest = _core.rest


def _settings(tmp_path, port=0):
    s = Settings()
    s.api.bind_address = f"127.0.0.1:{port}"
    s.sum.prob, s.update.prob = 0.5, 0.999
    for ph in (s.sum, s.update, s.sum2):
        ph.time.min, ph.time.max = 0.05, 10.0
        ph.count.max = 100
    s.update.count.min = 3
    s.model_length = 16
    s.storage_path = str(tmp_path)
    s.validate()
    return s


def test_restart_mid_deployment_participants_recover(tmp_path):
    settings = _settings(tmp_path)
    coord1, store1, models1 = build_coordinator(settings)
    server1 = rest.RestServer(coord1, "127.0.0.1", 0, 4)
    assert server1.start()
    port = server1.port
    coord1.start()

    client = rest.HttpXaynetClient("127.0.0.1", port)
    rng = np.random.default_rng(55)
    ps = [sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
          for _ in range(10)]
    w = np.full(16, 0.5, dtype=np.float32)

    def drive_until_model(coord, deadline=30.0):
        t0 = time.time()
        while time.time() - t0 < deadline:
            for p in ps:
                p.tick()
                if p.should_set_model:
                    p.set_model(w)
            b = coord.fetch_model()
            if b and b[0] == 1:
                return True
            time.sleep(0.005)
        return False

    assert drive_until_model(coord1), "no model before restart"
    rid1 = coord1.round_id
    coord1.stop()
    server1.stop()

    # "crash" -> new process state: restore from the durable store, same port
    settings.restore_enable = True
    coord2, _, _ = build_coordinator(settings)
    assert coord2.round_id == rid1  # resumed, not reset
    server2 = rest.RestServer(coord2, "127.0.0.1", port, 4)
    assert server2.start()
    coord2.start()
    try:
        # the SAME participant objects (their HTTP clients reconnect, the
        # round-freshness check resets them into the new round)
        assert drive_until_model(coord2, 40.0), "no model after restart"
        assert coord2.round_id > rid1
    finally:
        coord2.stop()
        server2.stop()


def test_daemon_cli(tmp_path):
    import socket

    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    cfg = tmp_path / "c.toml"
    cfg.write_text(f"""
[api]
bind_address = "127.0.0.1:{port}"
[pet.sum]
prob = 0.5
count = {{ min = 1, max = 10 }}
time = {{ min = 1, max = 60 }}
[pet.update]
prob = 0.9
count = {{ min = 3, max = 10 }}
time = {{ min = 1, max = 60 }}
[pet.sum2]
count = {{ min = 1, max = 10 }}
time = {{ min = 1, max = 60 }}
[model]
length = 8
""")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    proc = subprocess.Popen(
        [sys.executable, "-m", "xaynet_amd.server", "-c", str(cfg)], env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        cl = rest.HttpClient("127.0.0.1", port, timeout_s=2.0)
        deadline = time.time() + 15
        status = None
        while time.time() < deadline:
            r = cl.request("GET", "/params")
            if r is not None and r[0] == 200 and len(r[1]) > 80:
                status = 200
                break
            time.sleep(0.2)
        assert status == 200, "daemon did not serve round params"
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
    assert proc.returncode is not None


def test_daemon_cli_rejects_bad_config(tmp_path):
    cfg = tmp_path / "bad.toml"
    cfg.write_text("[pet.sum]\nprob = 1.5\n")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    rc = subprocess.run(
        [sys.executable, "-m", "xaynet_amd.server", "-c", str(cfg)], env=env,
        capture_output=True, timeout=60).returncode
    assert rc == 2
