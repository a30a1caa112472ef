"""Exercise the wide-order (u128, bpn>8) GPU kernel family at benchmark
size for rocprofv3 kernel stats: K3 digit-plane aggregation over 10-byte
limbs, k2_canonicalize_u128, and k4_unmask_u128.

Correctness of these kernels is pinned by tests/test_gpu_kernels.py (wide
roundtrip vs the exact-rational oracle at small size); this script only
generates load — update limbs are random bytes, which is fine for a
bandwidth profile (the digit-plane math is value-independent).

Run under rocprofv3:
  rocprofv3 --kernel-trace --stats -d gpurun_out/prof_wide -- \
      python scripts/wide_profile.py
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from xaynet_amd import _core
from xaynet_amd.ops.engine import GpuMaskedAggregator

mk = _core.mask


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--length", type=int, default=10_000_000)
    ap.add_argument("--pool", type=int, default=32)
    ap.add_argument("--iters", type=int, default=4)
    args = ap.parse_args()

    c = mk.MaskConfig(1, 1, 0, 3)  # Prime/F64/B0/M3 -> bpn=10, order > 2^64
    assert c.bytes_per_number == 10
    eng = GpuMaskedAggregator(c, c, args.length)
    assert eng.wide

    rng = np.random.default_rng(7)
    limbs = rng.integers(0, 256, args.length * c.bytes_per_number,
                         dtype=np.uint8).tobytes()
    mask_vals = torch.stack([
        # keep lo+hi*2^64 < order (order > 2^72 at bpn=10) so the mask
        # operand stays canonical for k4_unmask_u128
        torch.from_numpy(rng.integers(0, 1 << 62, args.length)).to("cuda"),
        torch.from_numpy(rng.integers(0, 1 << 6, args.length)).to("cuda"),
    ])

    pool = eng.alloc_update_pool(args.pool)
    for i in range(args.pool):
        eng.upload_update(pool, i, limbs)
    torch.cuda.synchronize()

    t0 = time.time()
    for _ in range(args.iters):
        eng.reset()
        eng.aggregate_pool(pool, args.pool, unit_sum=1)
        out = eng.unmask(mask_vals, 1)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / args.iters

    gb = args.pool * args.length * c.bytes_per_number / 1e9
    print(f"wide (bpn=10) pool={args.pool} length={args.length}: "
          f"{dt*1e3:.1f} ms/iter, K3 volume {gb:.1f} GB/iter, "
          f"out dtype {out.dtype}, mean {float(out.float().mean()):.3g}")


if __name__ == "__main__":
    main()
