#!/usr/bin/env python3
"""Round-2 kernel rates: wide (u128) K1 expansion, K5w weight masking, and
the u64 K1 reference point, at 25M elements."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from xaynet_amd import _core  # noqa: E402
from xaynet_amd.ops import GpuMaskedAggregator  # noqa: E402

mk = _core.mask
N = 25_000_000


def timed(fn, reps=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


for name, args in [("u64 f32-m6 (bpn7)", (1, 0, 0, 6)),
                   ("wide f64-m3 (bpn10, 3w/draw)", (1, 1, 0, 3)),
                   ("wide f64-b4-m9 (bpn14, 4w/draw)", (1, 1, 4, 9))]:
    c = mk.MaskConfig(*args)
    eng = GpuMaskedAggregator(c, c, N)
    shape = (2, N) if eng.wide else (N,)
    out = torch.empty(*shape, dtype=torch.int64, device="cuda")
    seed = b"\x05" * 32
    dt = timed(lambda: eng.derive_mask_values(seed, out=out))
    print(f"K1 {name}: {dt*1e3:.2f} ms/mask ({N/dt/1e6:.1f} Melt/s)")
    w = torch.rand(N, dtype=torch.float64 if args[1] == 1 else torch.float32,
                   device="cuda") * 2 - 1
    dt = timed(lambda: eng.mask_weights(seed, w, 1, 10))
    print(f"K1+K5w mask_weights {name}: {dt*1e3:.2f} ms/update ({N/dt/1e6:.1f} Melt/s)")
    # breakdown: derive (K1) vs quantize+mask+pack kernel vs D2H assembly
    mv = eng.derive_mask_values(seed)
    from xaynet_amd import _hip
    from xaynet_amd.ops.engine import _cfg_scalars
    vinfo = _cfg_scalars(c)
    outb = torch.empty(N * eng.bpn, dtype=torch.uint8, device="cuda")
    lo = mv[0].data_ptr() if eng.wide else mv.data_ptr()
    hi = mv[1].data_ptr() if eng.wide else mv.data_ptr()
    dt_map = {torch.float32: 0, torch.float64: 1}
    dk = timed(lambda: _hip.mask_weights(
        w.data_ptr(), dt_map[w.dtype], lo, hi, outb.data_ptr(), N, eng.bpn,
        eng.order, 0.1, vinfo["add_shift"], str(vinfo["exp_shift_u64"]), eng.wide))
    print(f"  K5w kernel alone: {dk*1e3:.2f} ms")
    del mv, outb
    del eng, out, w
    torch.cuda.empty_cache()
