"""TLS transport: server authentication, CA pinning, optional mutual TLS
(reference rest.rs `tls` feature + reqwest client TLS options)."""
import subprocess
import time

import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
sdk = _core.sdk
mk = _core.mask
rest = _core.rest


def make_cert(tmp_path, name, cn, ca=None):
    """Self-signed cert (ca=None) or CA-signed leaf."""
    key = tmp_path / f"{name}.key"
    crt = tmp_path / f"{name}.pem"
    if ca is None:
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes", "-keyout", str(key),
             "-out", str(crt), "-days", "1", "-subj", f"/CN={cn}",
             "-addext", "subjectAltName=IP:127.0.0.1,DNS:localhost"],
            check=True, capture_output=True)
    else:
        ca_key, ca_crt = ca
        csr = tmp_path / f"{name}.csr"
        subprocess.run(
            ["openssl", "req", "-newkey", "rsa:2048", "-nodes", "-keyout", str(key),
             "-out", str(csr), "-subj", f"/CN={cn}"],
            check=True, capture_output=True)
        subprocess.run(
            ["openssl", "x509", "-req", "-in", str(csr), "-CA", str(ca_crt), "-CAkey",
             str(ca_key), "-CAcreateserial", "-out", str(crt), "-days", "1"],
            check=True, capture_output=True)
    return key, crt


def serve_tls(tmp_path, model_length=16, client_ca=""):
    skey, scrt = make_cert(tmp_path, "server", "127.0.0.1")
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 1.0
    s.model_length = model_length
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 10.0)
    s.set_update(3, 100, 0.05, 10.0)
    s.set_sum2(1, 100, 0.05, 10.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    server = rest.RestServer(coord, "127.0.0.1", 0, str(scrt), str(skey), client_ca)
    assert server.start()
    return coord, server, scrt


def test_tls_round_trip_and_verification(tmp_path):
    coord, server, scrt = serve_tls(tmp_path)
    coord.run_one_phase()
    try:
        # CA-pinned client works
        cl = rest.TlsHttpClient("127.0.0.1", server.port, ca_file=str(scrt))
        status, body = cl.request("GET", "/params")
        assert status == 200 and len(body) > 80

        # verification ON without the CA -> handshake fails
        bad = rest.TlsHttpClient("127.0.0.1", server.port)
        assert bad.request("GET", "/params") is None

        # insecure mode skips verification
        ins = rest.TlsHttpClient("127.0.0.1", server.port, insecure=True)
        assert ins.request("GET", "/params")[0] == 200
    finally:
        server.stop()
        coord.stop()


def test_tls_full_round(tmp_path):
    length = 16
    coord, server, scrt = serve_tls(tmp_path, model_length=length)
    client = rest.TlsXaynetClient("127.0.0.1", server.port, ca_file=str(scrt))
    rng = np.random.default_rng(41)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(10)
    ]
    w = np.full(length, 0.25, dtype=np.float32)
    coord.start()
    raw = rest.TlsHttpClient("127.0.0.1", server.port, ca_file=str(scrt))
    model = None
    t0 = time.time()
    try:
        while time.time() - t0 < 30.0 and model is None:
            for p in participants:
                p.tick()
                if p.should_set_model:
                    p.set_model(w)
            r = raw.request("GET", "/model")
            if r is not None and r[0] == 200:
                model = sdk.decode_model(b"\x01" + r[1], 0)
            time.sleep(0.005)
    finally:
        coord.stop()
        server.stop()
    assert model is not None and np.allclose(model, 0.25, atol=1e-4)


def test_mutual_tls_client_auth(tmp_path):
    ca_key, ca_crt = make_cert(tmp_path, "ca", "xaynet-test-ca")
    ckey, ccrt = make_cert(tmp_path, "client", "participant-1", ca=(ca_key, ca_crt))
    coord, server, scrt = serve_tls(tmp_path, client_ca=str(ca_crt))
    coord.run_one_phase()
    try:
        # no client cert -> rejected at handshake
        anon = rest.TlsHttpClient("127.0.0.1", server.port, ca_file=str(scrt))
        assert anon.request("GET", "/params") is None
        # CA-signed client cert -> accepted
        auth = rest.TlsHttpClient("127.0.0.1", server.port, ca_file=str(scrt),
                                  cert_file=str(ccrt), key_file=str(ckey))
        assert auth.request("GET", "/params")[0] == 200
    finally:
        server.stop()
        coord.stop()


def test_tls_hostname_verification(tmp_path):
    """A valid cert for the WRONG host must be rejected even when its CA is
    trusted: chain verification alone would accept any pinned-CA cert for any
    domain (the reference's rustls client always checks hostnames)."""
    skey, scrt = make_cert(tmp_path, "wronghost", "evil.example.com")
    # rewrite SAN so the cert names a different host than the one we dial
    import subprocess as sp
    key = tmp_path / "wh.key"
    crt = tmp_path / "wh.pem"
    sp.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes", "-keyout", str(key),
         "-out", str(crt), "-days", "1", "-subj", "/CN=evil.example.com",
         "-addext", "subjectAltName=DNS:evil.example.com"],
        check=True, capture_output=True)
    s = co.Settings()
    s.model_length = 16
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    server = rest.RestServer(coord, "127.0.0.1", 0, str(crt), str(key), "")
    assert server.start()
    coord.run_one_phase()
    try:
        # CA pinned to the very cert the server presents — chain verifies,
        # but the peer identity (127.0.0.1) does not match DNS:evil.example.com
        cl = rest.TlsHttpClient("127.0.0.1", server.port, ca_file=str(crt))
        assert cl.request("GET", "/params") is None
        # insecure mode still connects (no verification at all)
        ins = rest.TlsHttpClient("127.0.0.1", server.port, insecure=True)
        assert ins.request("GET", "/params")[0] == 200
    finally:
        server.stop()
        coord.stop()
