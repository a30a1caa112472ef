#!/usr/bin/env python3
"""Phase timing for the config #5 stream step at 1B params (diagnostic)."""
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from xaynet_amd import _core  # noqa: E402
from xaynet_amd.ops import GpuMaskedAggregator  # noqa: E402

mk = _core.mask
N = 1_000_000_000
cfg = mk.MaskConfig(1, 0, 0, 6)
eng = GpuMaskedAggregator(cfg, cfg, N)
scratch = torch.empty(N, dtype=torch.int64, device="cuda")
pool = eng.alloc_update_pool(1)
mask_total = torch.zeros(N, dtype=torch.int64, device="cuda")
seed = (12345).to_bytes(32, "little")


def t(label, fn, reps=2):
    fn()  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/reps*1000:.1f} ms")


t("K1 derive_mask_values 1B", lambda: eng.derive_mask_values(seed, out=scratch))
t("K5 synth_update 1B", lambda: eng.synth_update(pool, 0, scratch, participant=3, scalar=0.25))
t("K3 aggregate_pool 1B", lambda: eng.aggregate_pool(pool, 1))
t("mod_add_values 1B", lambda: eng.mod_add_values(mask_total, scratch))


def unmask():
    eng.unit_acc = 1
    eng.nb_models = 4
    return eng.unmask(mask_total, 1, 4)


t("unmask 1B (K4 + D2H?)", unmask)
t("unit_draw", lambda: eng.unit_draw(seed), reps=10)
t("masked_unit_for", lambda: eng.masked_unit_for(seed, 1, 4), reps=10)
