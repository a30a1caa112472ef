"""REST API over real sockets: native HTTP server + native SDK HTTP client.

Reference parity (rust/xaynet-server/src/rest.rs, xaynet-sdk/src/client.rs):
POST /message always 200; GET bodies are bare bincode; 204 = None;
/seeds takes a percent-encoded base64 pk query.
"""
import base64
import time

import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
sdk = _core.sdk
rest = _core.rest
mk = _core.mask


def make_served_coordinator(model_length=32, n_expect=3):
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = model_length
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(n_expect, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    server = rest.RestServer(coord, "127.0.0.1", 0, 4)
    assert server.start()
    return coord, server


def test_routes_and_status_codes():
    coord, server = make_served_coordinator()
    coord.run_one_phase()  # Idle: publishes keys + round params
    try:
        cl = rest.HttpClient("127.0.0.1", server.port)
        # after Idle: params exists, model/sums None
        status, body = cl.request("GET", "/params")
        assert status == 200 and len(body) > 80
        assert cl.request("GET", "/model")[0] == 204
        assert cl.request("GET", "/sums")[0] == 204
        # bad pk -> 400
        assert cl.request("GET", "/seeds?pk=!!!")[0] == 400
        # valid-but-unknown pk -> 204
        pk = base64.b64encode(b"\x01" * 32).decode()
        pk_enc = pk.replace("+", "%2B").replace("/", "%2F").replace("=", "%3D")
        assert cl.request("GET", f"/seeds?pk={pk_enc}")[0] == 204
        # unknown route / wrong methods
        assert cl.request("GET", "/nope")[0] == 404
        assert cl.request("GET", "/message")[0] == 405
        assert cl.request("POST", "/params", b"x")[0] == 405
        # garbage POST /message still 200 (reference: errors logged, not surfaced)
        assert cl.request("POST", "/message", b"\x00" * 64)[0] == 200
        # keep-alive: the same connection served all of the above
    finally:
        server.stop()
        coord.stop()


def test_full_round_over_http():
    n, length = 12, 48
    coord, server = make_served_coordinator(model_length=length, n_expect=3)
    client = rest.HttpXaynetClient("127.0.0.1", server.port)
    rng = np.random.default_rng(11)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]

    coord.start()
    t0 = time.time()
    model = None
    raw = rest.HttpClient("127.0.0.1", server.port)
    seeds_body = None
    try:
        while time.time() - t0 < 30.0:
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            status, sums = raw.request("GET", "/sums")
            if status == 200:
                # first sum pk, query its seeds over HTTP
                npk = int.from_bytes(sums[:8], "little")
                if npk:
                    pk = sums[8:40]
                    enc = (
                        base64.b64encode(pk).decode()
                        .replace("+", "%2B").replace("/", "%2F").replace("=", "%3D")
                    )
                    st2, b2 = raw.request("GET", f"/seeds?pk={enc}")
                    if st2 == 200:
                        seeds_body = b2
            status, body = raw.request("GET", "/model")
            if status == 200:
                model = sdk.decode_model(b"\x01" + body, 0)
                break
            time.sleep(0.005)
    finally:
        coord.stop()
        server.stop()
    assert model is not None, "no global model published over HTTP within timeout"
    assert model.shape == (length,)
    assert seeds_body is not None  # seed dict was served over HTTP during the round

    # mean of accepted updaters: reconstruct accepted set from the seed dict
    nsd = int.from_bytes(seeds_body[:8], "little")
    accepted, off = set(), 8
    for _ in range(nsd):
        accepted.add(bytes(seeds_body[off : off + 32]))
        slen = int.from_bytes(seeds_body[off + 32 : off + 40], "little")
        assert slen == 80
        off += 40 + 80
    by_pk = {p.pk: i for i, p in enumerate(participants)}
    idx = [by_pk[pk] for pk in accepted]
    assert len(idx) >= 3
    expect = np.mean([weights[i].astype(np.float64) for i in idx], axis=0)
    assert np.abs(model.astype(np.float64) - expect).max() < 1e-4


def test_large_body_and_keepalive_reuse():
    """A large POST body round-trips through the server parser unharmed."""
    coord, server = make_served_coordinator()
    try:
        cl = rest.HttpClient("127.0.0.1", server.port)
        big = bytes(range(256)) * 4096  # 1 MiB garbage message
        assert cl.request("POST", "/message", big)[0] == 200
        # connection still usable
        assert cl.request("GET", "/params")[0] == 200
    finally:
        server.stop()
        coord.stop()
