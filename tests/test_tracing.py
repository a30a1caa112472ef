"""Structured tracing spans (reference tracing: run_phase span per phase,
phase.rs:148; a span threaded through the request channel so protocol-thread
handling parents to its ingest span, requests.rs:120)."""
import re
import threading
import time

import numpy as np

from xaynet_amd import _core

co = _core.coordinator
mk = _core.mask
sdk = _core.sdk


def parse(line):
    d = {}
    for kv in line.split(" "):
        k, _, v = kv.partition("=")
        d[k] = v
    return d


def test_phase_and_request_spans_are_linked():
    lines = []
    lock = threading.Lock()
    co.install_trace_callback(lambda ln: (lock.acquire(), lines.append(ln), lock.release()))
    try:
        n, length = 8, 64
        s = co.Settings()
        s.sum_prob = 0.5
        s.update_prob = 1.0
        s.model_length = length
        c = mk.MaskConfig(1, 0, 0, 3)
        s.mask_cfg = mk.MaskConfigPair(c, c)
        s.set_sum(1, 100, 0.05, 10.0)
        s.set_update(3, 100, 0.05, 10.0)
        s.set_sum2(1, 100, 0.05, 10.0)
        coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
        client = sdk.InProcessClient(coord)
        rng = np.random.default_rng(3)
        ps = [sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
              for _ in range(n)]
        w = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]
        coord.start()
        t0 = time.time()
        model = None
        try:
            while time.time() - t0 < 60 and model is None:
                for i, p in enumerate(ps):
                    p.tick()
                    if p.should_set_model:
                        p.set_model(w[i])
                body = coord.fetch_model()
                if body and body[0] == 1:
                    model = True
                time.sleep(0.005)
        finally:
            coord.stop()
        assert model
    finally:
        co.uninstall_trace()

    with lock:
        spans = [parse(ln) for ln in lines]
    names = {s["name"] for s in spans}
    assert {"run_phase", "ingest", "handle_request"} <= names
    # every protocol phase ran under a run_phase span with a round tag
    phases = {s["phase"] for s in spans if s["name"] == "run_phase"}
    assert {"idle", "sum", "update", "sum2", "unmask"} <= phases
    for s in spans:
        assert re.fullmatch(r"\d+", s["dur_us"]) or s["dur_us"].isdigit()
    # request->phase linkage: handle_request spans are parented to LIVE
    # ingest span ids (the cross-thread edge through the mpsc channel)
    ingest_ids = {s["span"] for s in spans if s["name"] == "ingest"}
    handled = [s for s in spans if s["name"] == "handle_request"]
    assert handled
    linked = [s for s in handled if s["parent"] in ingest_ids]
    assert len(linked) == len(handled), "unparented handle_request spans"
    # and the handler spans carry phase/round context
    assert any(s["phase"] == "update" and s.get("result") == "0" for s in handled)
