"""Reference-compatible participant checkpoints: the xaynet-sdk
`SerializableState` bincode layout (phase.rs:304-313).

Pins the byte layout with hand-derived fixtures (bincode 1.3 fixint rules:
u32 enum variant indices, u64 lengths, Option as 0/1 byte, raw fixed-size
key newtypes, length-prefixed Signature per its custom &[u8] serde,
num-bigint Vec<u32> digits), round-trips every phase, restores the
reference's Sending* variants to their `next` state, and proves a restored
participant completes a live round."""
import struct
import time

import numpy as np
import pytest

from xaynet_amd import _core

co = _core.coordinator
mk = _core.mask
cr = _core.crypto
sdk = _core.sdk


def sig_field(b: bytes) -> bytes:
    return struct.pack("<Q", len(b)) + b


def biguint(v: int) -> bytes:
    digits = []
    while v:
        digits.append(v & 0xFFFFFFFF)
        v >>= 32
    return struct.pack("<Q", len(digits)) + b"".join(struct.pack("<I", d) for d in digits)


def round_params_bytes(pk=b"\x00" * 32, sum_p=0.0, upd_p=0.0, seed=b"\x00" * 32,
                       cfg_idx=(1, 0, 0, 0), length=0) -> bytes:
    # MaskConfig serde = 4 u32 VARIANT INDICES (B0 -> 0, M3 -> 0 etc.)
    cfg = b"".join(struct.pack("<I", i) for i in cfg_idx)
    return (pk + struct.pack("<d", sum_p) + struct.pack("<d", upd_p) + seed
            + cfg + cfg + struct.pack("<Q", length))


def shared_bytes(sign_pk, sign_sk, numer=1, denom=1, msg_size=4096 - 136 - 48,
                 rp=None) -> bytes:
    out = sign_pk + sign_sk
    out += biguint(numer) + biguint(denom)
    out += b"\x01" + struct.pack("<Q", msg_size + 136 + 48)
    out += rp if rp is not None else round_params_bytes()
    return out


def make_keys(seed_byte=5):
    seed = bytes([seed_byte]) * 32
    pk, sk = cr.sign_keypair_from_seed(seed)
    return seed, bytes(pk), bytes(sk)


class _NullClient:
    pass


def null_client():
    # a PyTransport-less in-process client: bind to a throwaway coordinator
    s = co.Settings()
    s.model_length = 4
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    return sdk.InProcessClient(coord), coord


def test_newround_golden_layout():
    """A NewRound checkpoint is variant 0 + zero private bytes + shared."""
    _, pk, sk = make_keys()
    client, coord = null_client()
    p = sdk.Participant.restore_reference(
        struct.pack("<I", 0) + shared_bytes(pk, sk), client)
    assert p.phase_id == 0  # NewRound
    # saving immediately reproduces the exact fixture bytes
    out = bytes(p.save_reference())
    assert out == struct.pack("<I", 0) + shared_bytes(pk, sk)
    coord.stop()


def test_sum_state_round_trip():
    """Sum: ephm keypair (raw 32+32) + length-prefixed signature."""
    _, pk, sk = make_keys(7)
    ephm_pk, ephm_sk = b"\x21" * 32, b"\x22" * 32
    sum_sig = bytes(range(64))
    rp = round_params_bytes(pk=b"\x0c" * 32, sum_p=0.5, upd_p=0.9,
                            seed=b"\x0d" * 32, length=16)
    fixture = (struct.pack("<I", 2) + ephm_pk + ephm_sk + sig_field(sum_sig)
               + shared_bytes(pk, sk, numer=3, denom=4, rp=rp))
    client, coord = null_client()
    p = sdk.Participant.restore_reference(fixture, client)
    assert p.phase_id == 2  # Sum
    assert int(p.task) == 1  # Task.Sum
    assert bytes(p.save_reference()) == fixture
    coord.stop()


def test_update_and_sum2_round_trip():
    _, pk, sk = make_keys(9)
    sum_sig, upd_sig = b"\x31" * 64, b"\x32" * 64
    rp = round_params_bytes(length=8)
    upd = (struct.pack("<I", 3) + sig_field(sum_sig) + sig_field(upd_sig)
           + b"\x00" * 4 + shared_bytes(pk, sk, rp=rp))
    client, coord = null_client()
    p = sdk.Participant.restore_reference(upd, client)
    assert p.phase_id == 3 and int(p.task) == 2
    assert bytes(p.save_reference()) == upd

    s2 = (struct.pack("<I", 4) + b"\x41" * 32 + b"\x42" * 32 + sig_field(sum_sig)
          + b"\x00" * 3 + shared_bytes(pk, sk, rp=rp))
    p2 = sdk.Participant.restore_reference(s2, client)
    assert p2.phase_id == 4 and int(p2.task) == 1
    assert bytes(p2.save_reference()) == s2
    coord.stop()


def test_sending_variants_map_to_next_state():
    """SendingSum(5) -> Sum2; SendingUpdate(6)/SendingSum2(7) -> Awaiting.
    The embedded MessageEncoder (Simple and Multipart) is parsed and
    discarded — our sends are inline and the freshness check re-syncs."""
    _, pk, sk = make_keys(11)
    sum_sig = b"\x51" * 64
    shared = shared_bytes(pk, sk)
    # SendingSum { message: Simple(Some(payload)), failed: None, next: Sum2{..} }
    payload = b"\xaa" * 40
    sending_sum = (struct.pack("<I", 5)
                   + struct.pack("<I", 0) + b"\x01" + sig_field(payload)[:8] + payload
                   + b"\x00"
                   + b"\x61" * 32 + b"\x62" * 32 + sig_field(sum_sig) + b"\x00" * 3
                   + shared)
    client, coord = null_client()
    p = sdk.Participant.restore_reference(sending_sum, client)
    assert p.phase_id == 4  # Sum2

    # SendingUpdate { message: Multipart{keys, cpk, data, id, tag, size, mid},
    #                 failed: Some(chunk), next: Awaiting }
    data = b"\xbb" * 100
    mp = (struct.pack("<I", 1) + pk + sk + b"\x71" * 32
          + struct.pack("<Q", len(data)) + data
          + struct.pack("<H", 3)            # id: u16
          + struct.pack("<I", 1)            # tag enum index (Update)
          + struct.pack("<Q", 1000)         # payload_size
          + struct.pack("<H", 77))          # message_id: u16
    sending_update = (struct.pack("<I", 6) + mp
                      + b"\x01" + struct.pack("<Q", 5) + b"\xcc" * 5  # failed: Some
                      + shared)
    p2 = sdk.Participant.restore_reference(sending_update, client)
    assert p2.phase_id == 1  # Awaiting
    coord.stop()


def test_restored_participant_completes_round():
    """Save a live participant mid-protocol in the REFERENCE format, restore
    it against the same coordinator, and the round still completes."""
    length, n = 16, 8
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
    rng = np.random.default_rng(13)
    participants = [
        sdk.Participant(bytes(rng.integers(0, 256, 32, dtype=np.uint8)), 1, 1, client)
        for _ in range(n)
    ]
    weights = [rng.uniform(-1, 1, length).astype(np.float32) for _ in range(n)]
    coord.start()
    t0 = time.time()
    model = None
    swapped = False
    try:
        while time.time() - t0 < 60.0 and model is None:
            if not swapped and time.time() - t0 > 0.05:
                # checkpoint/restore every participant mid-round through the
                # reference format
                participants = [
                    sdk.Participant.restore_reference(p.save_reference(), client)
                    for p in participants
                ]
                swapped = True
            for i, p in enumerate(participants):
                p.tick()
                if p.should_set_model:
                    p.set_model(weights[i])
            body = coord.fetch_model()
            if swapped and body and body[0] == 1:
                model = np.asarray(sdk.decode_model(body, 0))
            time.sleep(0.005)
    finally:
        coord.stop()
    assert swapped
    assert model is not None and np.isfinite(model).all()
    assert np.abs(model).max() <= 1.0 + 1e-5
