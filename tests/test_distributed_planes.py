"""Cross-rank aggregation semantics on CPU (gloo, world_size=2).

The multi-GPU design: RCCL all-reduce is an *integer* sum,
not a modular sum, so the aggregate is carried as u64-per-32-bit-digit
"digit planes" with deferred modular reduction — all_reduce(planes) then one
canonicalize pass is exactly the bench.py N>1 path. This test runs the same
arithmetic over torch.distributed/gloo with 2 processes and checks it against
the exact CPU oracle (_core.mask.Aggregation) on the union of both ranks'
masked updates."""
import os

import numpy as np
import pytest
import torch

from xaynet_amd import _core

mk = _core.mask

LENGTH = 257
PER_RANK = 5


def _mask_values(cfg, seed: bytes, length: int) -> np.ndarray:
    pair = mk.MaskConfigPair(cfg, cfg)
    obj = mk.derive_mask(seed, length, pair)
    return np.array([int(obj.element(i)) for i in range(length)], dtype=np.uint64)


def _to_planes(vals: np.ndarray, n_digits: int) -> torch.Tensor:
    planes = torch.zeros(n_digits, len(vals), dtype=torch.int64)
    for d in range(n_digits):
        planes[d] = torch.from_numpy(((vals >> np.uint64(32 * d)) & np.uint64(0xFFFFFFFF)).astype(np.int64))
    return planes

def _canonicalize(planes: torch.Tensor, order: int) -> np.ndarray:
    n_digits, length = planes.shape
    out = np.zeros(length, dtype=object)
    arr = planes.numpy()
    for d in reversed(range(n_digits)):
        out = out * (1 << 32) + arr[d].astype(object)
    return np.array([int(v) % order for v in out], dtype=np.uint64)


def _worker(rank: int, world: int, port: int):
    import torch.distributed as dist

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
    )
    cfg = mk.MaskConfig(1, 0, 0, 6)  # Prime/F32/B0/M6, bpn=7, order < 2^56
    order = int(cfg.order)
    n_digits = (cfg.bytes_per_number + 3) // 4

    # each rank owns PER_RANK masked updates (mask values stand in for
    # masked-model values: same group arithmetic)
    local = torch.zeros(n_digits, LENGTH, dtype=torch.int64)
    for k in range(PER_RANK):
        seed = bytes([rank * PER_RANK + k + 1]) * 32
        vals = _mask_values(cfg, seed, LENGTH)
        local += _to_planes(vals, n_digits)

    dist.all_reduce(local)  # plain int64 sum over ranks (the RCCL analog)
    got = _canonicalize(local, order)

    # exact oracle over the union of all ranks' updates
    pair = mk.MaskConfigPair(cfg, cfg)
    agg = mk.Aggregation(pair, LENGTH)
    for r in range(world):
        for k in range(PER_RANK):
            seed = bytes([r * PER_RANK + k + 1]) * 32
            agg.aggregate(mk.derive_mask(seed, LENGTH, pair))
    expect = np.array([int(agg.object.element(i)) for i in range(LENGTH)], dtype=np.uint64)

    assert (got == expect).all(), f"rank {rank}: plane reduction != modular oracle"
    dist.destroy_process_group()


def test_digit_plane_allreduce_matches_modular_sum():
    import socket

    import torch.multiprocessing as mp

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    world = 2
    ctx = mp.spawn(_worker, args=(world, port), nprocs=world, join=True)


def test_plane_headroom():
    """2^31 updates of 32-bit digits fit int64 planes without overflow (the
    deferred-reduction safety bound from kernels.hip)."""
    max_digit = (1 << 32) - 1
    n_updates = 1 << 31
    assert max_digit * n_updates < (1 << 63) - 1


def test_choose_reduce_strategy():
    from xaynet_amd.parallel import choose_reduce_strategy as pick

    order7 = 2**54 + 37  # ~7-byte order
    assert pick(1, 1000, order7) == "single"
    # 54 bits + 3 bits (world 8) <= 63 -> canonical-values RS
    assert pick(8, 1000 * 8, order7) == "values_rs"
    # uneven shard -> all-reduce fallback
    assert pick(8, 1001, order7) == "all_reduce"
    # forced fallback
    assert pick(8, 1000 * 8, order7, env={"XAYNET_ALLREDUCE": "1"}) == "all_reduce"
    # 62-bit order: 62 + 3 > 63 -> plane RS at world 8
    assert pick(8, 1000 * 8, 2**61 + 11) == "planes_rs"
    # but a 2-rank world still fits values RS (62 + 1 <= 63)
    assert pick(2, 1000 * 2, 2**61 + 11) == "values_rs"
