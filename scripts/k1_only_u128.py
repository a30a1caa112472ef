#!/usr/bin/env python3
"""Wide-order K1-only loop for PMC counter runs (25M f64-m3, bpn 10)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from xaynet_amd import _core  # noqa: E402
from xaynet_amd.ops import GpuMaskedAggregator  # noqa: E402

c = _core.mask.MaskConfig(1, 1, 0, 3)  # Prime/F64/B0/M3 -> bpn 10
N = 25_000_000
eng = GpuMaskedAggregator(c, c, N)
out = torch.empty((2, N), dtype=torch.int64, device="cuda")
for i in range(10):
    eng.derive_mask_values(bytes([i + 1]) * 32, out=out)
torch.cuda.synchronize()
print("ok")
