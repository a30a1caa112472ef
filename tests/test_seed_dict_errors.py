"""Seed-dict protocol errors via hand-crafted (signed, eligible) messages —
the reference's Redis-Lua error taxonomy (storage/traits.rs:260-311,
coordinator_storage/redis/mod.rs:208-267): length mismatch vs the sum dict,
unknown sum participant, duplicate update pk, and masked-model validation."""
import threading
import time

import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
mk = _core.mask
cr = _core.crypto
msgmod = _core.message

E = co.PipelineError
LEN = 8


def eligible_seed(rng, seed_round, want_sum: bool):
    """Find a signing seed whose task eligibility matches `want_sum`
    (sum_prob=0.5, update_prob=0.999)."""
    for _ in range(500):
        sgn = bytes(rng.integers(0, 256, 32, dtype=np.uint8))
        pk, sk = cr.sign_keypair_from_seed(sgn)
        sum_sig = cr.sign_detached(seed_round + b"sum", sk)
        upd_sig = cr.sign_detached(seed_round + b"update", sk)
        is_sum = cr.is_eligible(sum_sig, 0.5)
        if want_sum and is_sum:
            return sgn, sk, sum_sig, upd_sig
        if not want_sum and not is_sum and cr.is_eligible(upd_sig, 0.999):
            return sgn, sk, sum_sig, upd_sig
    raise AssertionError("no eligible seed found")


@pytest.fixture
def arena():
    s = co.Settings()
    s.sum_prob = 0.5
    s.update_prob = 0.999
    s.model_length = LEN
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 10, 5.0, 30.0)
    s.set_update(1, 10, 5.0, 30.0)
    s.set_sum2(1, 10, 5.0, 30.0)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    coord.run_one_phase()  # Idle -> Sum
    params = bytes(coord.fetch_round_params())
    yield coord, c, params[:32], params[48:80]
    coord.stop()


def drive_phase(coord):
    """Run one phase on a thread (phases block on count/time gates)."""
    t = threading.Thread(target=coord.run_one_phase, daemon=True)
    t.start()
    time.sleep(0.05)
    return t


def masked_object(c, length=LEN):
    w = np.zeros(length, np.float32)
    return mk.mask_model(b"\x01" * 32, mk.Scalar(1, 1), w, mk.MaskConfigPair(c, c))


def update_wire(c, sgn, sum_sig, upd_sig, cpk, entries, length=LEN):
    payload = bytes(sum_sig) + bytes(upd_sig) + bytes(masked_object(c, length).serialize())
    body = b"".join(pk + seed for pk, seed in entries)
    payload += (4 + len(body)).to_bytes(4, "big") + body  # INCLUSIVE length
    return bytes(msgmod.encode(msgmod.TAG_UPDATE, payload, sgn, cpk)[0])


def test_seed_dict_error_taxonomy(arena):
    coord, c, cpk, seed_round = arena
    rng = np.random.default_rng(3)

    # 1 sum participant joins
    s_sgn, s_sk, s_sum_sig, _ = eligible_seed(rng, seed_round, want_sum=True)
    ephm_pk, _ = cr.box_keypair()
    sum_payload = bytes(s_sum_sig) + ephm_pk
    sum_wire = bytes(msgmod.encode(msgmod.TAG_SUM, sum_payload, s_sgn, cpk)[0])
    t = drive_phase(coord)  # Sum phase
    assert coord.handle_message_bytes(sum_wire) == int(E.Ok)
    t.join(15)
    assert coord.phase == co.PhaseId.Update
    sum_pk = cr.sign_keypair_from_seed(s_sgn)[0]
    sealed_seed = cr.sealbox_seal(b"\x02" * 32, ephm_pk)

    t = drive_phase(coord)  # Update phase
    u = [eligible_seed(rng, seed_round, want_sum=False) for _ in range(4)]

    # (a) empty seed dict while the sum dict has 1 entry -> LengthMisMatch
    r = coord.handle_message_bytes(update_wire(c, u[0][0], u[0][2], u[0][3], cpk, []))
    assert r == int(E.MessageRejected)

    # (b) right length, unknown sum pk -> UnknownSumParticipant
    r = coord.handle_message_bytes(update_wire(
        c, u[1][0], u[1][2], u[1][3], cpk, [(b"\x09" * 32, sealed_seed)]))
    assert r == int(E.MessageRejected)

    # (c) well-formed update -> accepted
    good = update_wire(c, u[2][0], u[2][2], u[2][3], cpk, [(bytes(sum_pk), sealed_seed)])
    assert coord.handle_message_bytes(good) == int(E.Ok)

    # (d) same update pk again -> UpdatePkAlreadySubmitted
    assert coord.handle_message_bytes(good) == int(E.MessageRejected)

    # (e) masked model of the wrong length -> aggregation validation failure
    r = coord.handle_message_bytes(update_wire(
        c, u[3][0], u[3][2], u[3][3], cpk, [(bytes(sum_pk), sealed_seed)], length=4))
    assert r == int(E.AggregationFailed)

    t.join(15)
    assert coord.phase in (co.PhaseId.Sum2, co.PhaseId.Unmask, co.PhaseId.Idle,
                           co.PhaseId.Failure)


def test_mask_score_error_taxonomy(arena):
    """Sum2-phase errors (reference incr_mask_score Lua,
    redis/mod.rs:303-339): unknown sum participant and duplicate mask
    submission are rejected; a second distinct voter is accepted."""
    coord, c, cpk, seed_round = arena
    rng = np.random.default_rng(5)

    # two sum participants join
    summers = []
    t = drive_phase(coord)
    for _ in range(2):
        sgn, sk, sum_sig, _ = eligible_seed(rng, seed_round, want_sum=True)
        ephm_pk, _ = cr.box_keypair()
        wire = bytes(msgmod.encode(
            msgmod.TAG_SUM, bytes(sum_sig) + ephm_pk, sgn, cpk)[0])
        assert coord.handle_message_bytes(wire) == int(E.Ok)
        summers.append((sgn, sum_sig))
    t.join(15)
    assert coord.phase == co.PhaseId.Update

    # one valid update so the round can progress
    u_sgn, _, u_sum_sig, u_upd_sig = eligible_seed(rng, seed_round, want_sum=False)
    sum_pk0 = cr.sign_keypair_from_seed(summers[0][0])[0]
    sealed = cr.sealbox_seal(b"\x03" * 32, cr.box_keypair()[0])
    t = drive_phase(coord)
    entries = [(bytes(cr.sign_keypair_from_seed(sg)[0]), sealed) for sg, _ in summers]
    assert coord.handle_message_bytes(
        update_wire(c, u_sgn, u_sum_sig, u_upd_sig, cpk, entries)) == int(E.Ok)
    t.join(15)
    assert coord.phase == co.PhaseId.Sum2

    mask_bytes = bytes(masked_object(c).serialize())  # any valid MaskObject

    def sum2_wire(sgn, sum_sig):
        payload = bytes(sum_sig) + mask_bytes
        return bytes(msgmod.encode(msgmod.TAG_SUM2, payload, sgn, cpk)[0])

    t = drive_phase(coord)
    # (a) sum2 from a NON-summer (update-eligible participant cannot forge
    # sum eligibility -> NotSumEligible at task validation)
    r = coord.handle_message_bytes(sum2_wire(u_sgn, u_sum_sig))
    assert r == int(E.NotSumEligible)
    # (b) first vote from summer 0 -> accepted
    assert coord.handle_message_bytes(sum2_wire(*summers[0])) == int(E.Ok)
    # (c) duplicate vote from summer 0 -> MaskAlreadySubmitted -> rejected
    assert coord.handle_message_bytes(sum2_wire(*summers[0])) == int(E.MessageRejected)
    # (d) summer 1 votes -> accepted
    assert coord.handle_message_bytes(sum2_wire(*summers[1])) == int(E.Ok)
    t.join(15)


def test_wrong_length_winning_mask_fails_round_and_recovers(arena):
    """A sum2 mask of the wrong length is accepted at ingest (the reference's
    incr_mask_score stores raw bytes) but fails unmask validation
    (unmask.rs:118-130 -> UnmaskError) -> Failure phase -> fresh round."""
    coord, c, cpk, seed_round = arena
    rng = np.random.default_rng(7)

    s_sgn, _, s_sum_sig, _ = eligible_seed(rng, seed_round, want_sum=True)
    ephm_pk, _ = cr.box_keypair()
    t = drive_phase(coord)
    assert coord.handle_message_bytes(bytes(msgmod.encode(
        msgmod.TAG_SUM, bytes(s_sum_sig) + ephm_pk, s_sgn, cpk)[0])) == int(E.Ok)
    t.join(15)
    assert coord.phase == co.PhaseId.Update

    u_sgn, _, u_sum_sig, u_upd_sig = eligible_seed(rng, seed_round, want_sum=False)
    sum_pk = cr.sign_keypair_from_seed(s_sgn)[0]
    sealed = cr.sealbox_seal(b"\x05" * 32, ephm_pk)
    t = drive_phase(coord)
    assert coord.handle_message_bytes(update_wire(
        c, u_sgn, u_sum_sig, u_upd_sig, cpk, [(bytes(sum_pk), sealed)])) == int(E.Ok)
    t.join(15)
    assert coord.phase == co.PhaseId.Sum2

    # the summer votes with a syntactically valid mask of the WRONG length
    bad_mask = bytes(masked_object(c, length=4).serialize())
    t = drive_phase(coord)
    assert coord.handle_message_bytes(bytes(msgmod.encode(
        msgmod.TAG_SUM2, bytes(s_sum_sig) + bad_mask, s_sgn, cpk)[0])) == int(E.Ok)
    t.join(15)
    assert coord.phase == co.PhaseId.Unmask

    # unmask: length mismatch -> Failure, then Failure -> Idle (fresh round)
    assert coord.run_one_phase() == co.PhaseId.Failure
    rid = coord.round_id
    assert coord.run_one_phase() == co.PhaseId.Idle
    coord.run_one_phase()  # Idle -> Sum: new round, new keys
    assert coord.round_id == rid + 1
    assert coord.phase == co.PhaseId.Sum
