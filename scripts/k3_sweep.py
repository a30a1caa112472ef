#!/usr/bin/env python3
"""Time k3_aggregate EPT variants (register pressure vs load width) on MI355X.

Usage: python scripts/k3_sweep.py [--length 25000000] [--pool 300] [--bpn 7]
Prints ms/dispatch and effective read TB/s per variant.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from xaynet_amd import _core, _hip
from xaynet_amd.ops import GpuMaskedAggregator


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--length", type=int, default=25_000_000)
    ap.add_argument("--pool", type=int, default=300)
    ap.add_argument("--reps", type=int, default=3)
    ap.add_argument("--epts", type=str, default="0,4,8,16")
    ap.add_argument("--bpn", type=int, default=7, choices=[7, 10],
                    help="7 = Prime/F32/B0/M6 (headline); 10 = Prime/F64/B0/M3 "
                    "(wide u128 order)")
    args = ap.parse_args()

    mk = _core.mask
    cfg = (mk.MaskConfig(1, 0, 0, 6) if args.bpn == 7
           else mk.MaskConfig(1, 1, 0, 3))
    eng = GpuMaskedAggregator(cfg, cfg, args.length, device="cuda:0")
    pool = eng.alloc_update_pool(args.pool)
    if eng.wide:
        # K1/K5 cover u64 orders only; random bytes are fine for a
        # bandwidth sweep (digit-plane adds are value-independent)
        pool.copy_(torch.randint(0, 256, pool.shape, dtype=torch.uint8,
                                 device="cuda:0"))
    else:
        scratch = torch.empty(args.length, dtype=torch.int64, device="cuda:0")
        for p in range(args.pool):
            seed = (p + 1).to_bytes(32, "little")
            if p < 8:
                eng.derive_mask_values(seed, out=scratch)
            eng.synth_update(pool, p, scratch, participant=p, scalar=1.0 / args.pool)
    torch.cuda.synchronize()

    gb = args.pool * args.length * eng.bpn / 1e9
    for ept in (int(x) for x in args.epts.split(",")):
        eng.reset()
        _hip.aggregate_batch(eng.acc.data_ptr(), pool.data_ptr(), pool.stride(0),
                             args.pool, args.length, eng.bpn, ept)  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.reps):
            _hip.aggregate_batch(eng.acc.data_ptr(), pool.data_ptr(), pool.stride(0),
                                 args.pool, args.length, eng.bpn, ept)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / args.reps * 1000
        print(f"ept={ept:2d}: {ms:8.3f} ms/dispatch  {gb / ms:6.2f} TB/s effective")


if __name__ == "__main__":
    main()


def k1_bench():
    """Time mask expansion (K1) fused vs 3-pass (set XAYNET_K1_FUSED)."""
    import time as _t

    mk = _core.mask
    cfg = mk.MaskConfig(1, 0, 0, 6)
    eng = GpuMaskedAggregator(cfg, cfg, 25_000_000, device="cuda:0")
    out = torch.empty(25_000_000, dtype=torch.int64, device="cuda:0")
    for p in range(3):
        eng.derive_mask_values(bytes([p + 1]) * 32, out=out)  # warm
    torch.cuda.synchronize()
    t0 = _t.perf_counter()
    n = 20
    for p in range(n):
        eng.derive_mask_values(bytes([p + 1]) * 32, out=out)
    torch.cuda.synchronize()
    ms = (_t.perf_counter() - t0) / n * 1000
    mode = os.environ.get("XAYNET_K1_FUSED", "0")
    print(f"K1 expand 25M (fused={mode}): {ms:7.3f} ms/mask "
          f"({25e6 * 8 / ms / 1e6:6.1f} GB/s of accepted values)")
