"""Multipart buffer admission bounds (memory-exhaustion DoS guard).

Chunks are buffered before task validation, so an attacker with throwaway
keypairs could otherwise grow the reassembly map without bound for a whole
round. The coordinator bounds entries, per-pk bytes and global bytes; a chunk
past any bound is rejected (not buffered). Legit transfers within the bounds
still complete (test_multipart_e2e covers that path)."""
import numpy as np

from xaynet_amd import _core

co = _core.coordinator
mk = _core.mask
msgmod = _core.message


def make_coord(**overrides):
    s = co.Settings()
    s.sum_prob = 0.99
    s.update_prob = 0.99
    s.model_length = 8
    c = mk.MaskConfig(1, 0, 0, 3)
    s.mask_cfg = mk.MaskConfigPair(c, c)
    s.set_sum(1, 100, 0.3, 10.0)
    s.set_update(1, 100, 0.3, 10.0)
    s.set_sum2(1, 100, 0.3, 10.0)
    for k, v in overrides.items():
        setattr(s, k, v)
    coord = co.Coordinator(s, co.InMemoryStorage(), co.InMemoryModels(), False)
    coord.run_one_phase()  # Idle -> Sum
    return coord


def first_chunk(seed: bytes, cpk: bytes, message_id: int = 1):
    """A never-completing chunk set: only chunk 0 of 3 is delivered."""
    payload = b"\x33" * 64 + b"\x44" * 32  # sum payload shape
    parts = msgmod.encode(msgmod.TAG_SUM, payload, seed, cpk,
                          max_payload=40, message_id=message_id)
    assert len(parts) == 3
    return bytes(parts[0])


def test_entry_count_bound():
    coord = make_coord(multipart_max_entries=3)
    cpk = bytes(coord.fetch_round_params())[:32]
    E = co.PipelineError
    rng = np.random.default_rng(5)
    seeds = [bytes(rng.integers(0, 256, 32, dtype=np.uint8)) for _ in range(4)]
    for seed in seeds[:3]:
        assert coord.handle_message_bytes(first_chunk(seed, cpk)) == int(E.Ok)
    # 4th distinct (pk, message_id) entry exceeds the cap
    assert coord.handle_message_bytes(first_chunk(seeds[3], cpk)) == int(E.MessageRejected)
    # duplicates of an existing entry's chunk are still fine (overwrite)
    assert coord.handle_message_bytes(first_chunk(seeds[0], cpk)) == int(E.Ok)
    coord.stop()


def test_per_pk_byte_bound():
    coord = make_coord(multipart_max_per_pk_bytes=60)
    cpk = bytes(coord.fetch_round_params())[:32]
    E = co.PipelineError
    seed = b"\x07" * 32
    # chunk 0 carries 32 data bytes -> fits; a second in-flight message
    # from the same pk would reach 64 > 60 -> rejected
    assert coord.handle_message_bytes(first_chunk(seed, cpk, message_id=1)) == int(E.Ok)
    assert coord.handle_message_bytes(first_chunk(seed, cpk, message_id=2)) == int(E.MessageRejected)
    # another participant is unaffected (per-pk bound, not global)
    assert coord.handle_message_bytes(first_chunk(b"\x08" * 32, cpk)) == int(E.Ok)
    coord.stop()


def test_global_byte_bound():
    coord = make_coord(multipart_max_total_bytes=80)
    cpk = bytes(coord.fetch_round_params())[:32]
    E = co.PipelineError
    # each chunk carries 32 data bytes (max_payload 40 − 8 chunk overhead)
    assert coord.handle_message_bytes(first_chunk(b"\x01" * 32, cpk)) == int(E.Ok)
    assert coord.handle_message_bytes(first_chunk(b"\x02" * 32, cpk)) == int(E.Ok)
    # 3rd chunk would reach 96 > 80 regardless of pk
    assert coord.handle_message_bytes(first_chunk(b"\x03" * 32, cpk)) == int(E.MessageRejected)
    coord.stop()
