"""Message-pipeline error taxonomy (reference ServiceError,
services/messages/error.rs + the tower pipeline stages): decrypt, parse,
signature, coordinator-pk, phase filter and task-eligibility failures all map
to their typed errors, and POST /message still returns 200 for every one of
them (errors are logged, not surfaced — rest.rs:93-101)."""
import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
mk = _core.mask
msgmod = _core.message
cr = _core.crypto
rest = _core.rest

E = co.PipelineError


def make_coord(sum_prob=0.5, update_prob=1.0):
    s = co.Settings()
    s.sum_prob = sum_prob
    s.update_prob = update_prob
    s.model_length = 8
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.05, 5.0)
    s.set_update(3, 100, 0.05, 5.0)
    s.set_sum2(1, 100, 0.05, 5.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    coord.run_one_phase()  # Idle -> publishes keys/params, enters Sum
    return coord


def coord_pk(coord) -> bytes:
    return bytes(coord.fetch_round_params()[:32])  # RoundParameters starts with pk


def seal_to(coord, wire: bytes) -> bytes:
    return cr.sealbox_seal(wire, coord_pk(coord))


def sum_wire(coord, seed=b"\x09" * 32, tag=None, cpk=None, corrupt_sig=False):
    payload = b"\x33" * 64 + b"\x44" * 32
    parts = msgmod.encode(
        tag if tag is not None else msgmod.TAG_SUM, payload, seed,
        cpk if cpk is not None else coord_pk(coord))
    wire = bytearray(parts[0])
    if corrupt_sig:
        wire[0] ^= 0xFF
    return bytes(wire)


def test_decrypt_failure():
    coord = make_coord()
    assert coord.handle_encrypted_message(b"\x01" * 64) == int(E.Decrypt)
    coord.stop()


def test_parse_failure():
    coord = make_coord()
    # too short / garbage after successful decrypt
    assert coord.handle_encrypted_message(seal_to(coord, b"\x00" * 16)) == int(E.Parsing)
    coord.stop()


def test_invalid_signature():
    coord = make_coord()
    wire = sum_wire(coord, corrupt_sig=True)
    assert coord.handle_message_bytes(wire) == int(E.InvalidMessageSignature)
    coord.stop()


def test_invalid_coordinator_pk():
    coord = make_coord()
    wire = sum_wire(coord, cpk=b"\x07" * 32)
    assert coord.handle_message_bytes(wire) == int(E.InvalidCoordinatorPublicKey)
    coord.stop()


def test_phase_filter_drops_wrong_tag():
    coord = make_coord()
    # coordinator is in Sum phase; a sum2 message is unexpected
    payload = b"\x11" * 64 + mk.derive_mask(b"\x01" * 32, 8, mk.MaskConfigPair(
        mk.MaskConfig(1, 0, 0, 3), mk.MaskConfig(1, 0, 0, 3))).serialize()
    parts = msgmod.encode(msgmod.TAG_SUM2, bytes(payload), b"\x09" * 32, coord_pk(coord))
    assert coord.handle_message_bytes(parts[0]) == int(E.UnexpectedMessage)
    coord.stop()


def test_sum_eligibility_rejection():
    # sum_prob tiny -> virtually no seed is sum-eligible
    coord = make_coord(sum_prob=1e-12)
    rng = np.random.default_rng(3)
    rejected = 0
    for i in range(8):
        seed = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        r = coord.handle_message_bytes(sum_wire(coord, seed=seed))
        if r == int(E.NotSumEligible):
            rejected += 1
    assert rejected >= 7, "sum-ineligible messages were not rejected"
    coord.stop()


def test_http_post_returns_200_for_all_errors():
    coord = make_coord()
    server = rest.RestServer(coord, "127.0.0.1", 0, 2)
    assert server.start()
    try:
        cl = rest.HttpClient("127.0.0.1", server.port)
        for bad in (b"", b"\x00" * 32, seal_to(coord, b"junk"),
                    sum_wire(coord, corrupt_sig=True)):
            assert cl.request("POST", "/message", bad)[0] == 200
    finally:
        server.stop()
        coord.stop()
