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
    ap.add_argument("--mask-config", choices=["f32-m6", "i64-m6", "f32-m3", "f64-m3"],
                    default="f32-m6",
                    help="PET mask config (i64-m6 = BASELINE config #4 masking; "
                         "f64-m3 = wide u128 group order, single GPU)")
    ap.add_argument("--stream", action="store_true",
                    help="config #5 mode: streamed ChaCha20 mask-expand + pack + aggregate "
                         "per client (timed), no resident update pool")
    ap.add_argument("--h2d", action="store_true",
                    help="config #4 mode: updates staged in pinned host memory, "
                         "double-buffered H2D copies overlapped with aggregation")
    ap.add_argument("--ingest", action="store_true",
                    help="ingest-path mode: decrypt+validate+stage+H2D+aggregate "
                         "through the production serve plane (delegates to "
                         "scripts/ingest_bench.py; N=1 only)")
    ap.add_argument("--verify", action="store_true", help="small-scale correctness check first")
    args = ap.parse_args()

    if args.ingest:
        script = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                              "scripts", "ingest_bench.py")
        os.execv(sys.executable, [sys.executable, script,
                                  "--length", str(args.length),
                                  "--clients", str(args.clients_per_gpu),
                                  "--threads", "32"])

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

    # Prime/<dtype>/B0/<M> — M6 supports up to 1e6 models per round
    cfg_args = {"f32-m6": (1, 0, 0, 6), "i64-m6": (1, 3, 0, 6), "f32-m3": (1, 0, 0, 3),
                "f64-m3": (1, 1, 0, 3)}
    cfg = mk.MaskConfig(*cfg_args[args.mask_config])
    eng = GpuMaskedAggregator(cfg, cfg, args.length, device=device)
    if eng.wide and world > 1:
        raise SystemExit("wide (u128) configs run single-GPU; use --gpus 1")

    if args.verify and rank == 0:
        _verify(mk, cfg)

    if args.stream:
        return _run_stream(args, eng, cfg, rank, world, dist, device)

    clients = args.clients_per_gpu
    pool_n = min(args.pool, clients)
    cycles, rem = divmod(clients, pool_n)
    if rem:
        cycles += 0  # remainder handled per-round below

    # ---- untimed setup: synthesize the per-GPU update pool + global mask ----
    pool = eng.alloc_update_pool(pool_n)
    shape = (2, args.length) if eng.wide else (args.length,)
    mask_pool_sum = torch.zeros(*shape, dtype=torch.int64, device=device)
    scratch = torch.empty(*shape, dtype=torch.int64, device=device)
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
    mask_total = torch.zeros(*shape, dtype=torch.int64, device=device)
    for _ in range(clients // pool_n):
        eng.mod_add_values(mask_total, mask_pool_sum)
    # remainder clients: first `rem` pool masks again (approximate cycling);
    # keep clients divisible by pool for exactness
    unit_mask_total = sum(u for u, _ in unit_pool) * (clients // pool_n) % unit_order
    unit_masked_per_round = sum(mu for _, mu in unit_pool) * (clients // pool_n) % unit_order

    # cross-GPU composition: the SAME library module the coordinator serve
    # plane uses (xaynet_amd.parallel) — strategy chooser + sharded
    # reduce/unmask, so the N>1 path measured here is the production code
    from xaynet_amd.parallel import ShardedAggregation

    sharded = ShardedAggregation(eng, dist, rank, world)
    if world > 1:
        # global mask = modular sum across ranks
        sharded.allreduce_mask(mask_total)
        t = torch.tensor([unit_mask_total, unit_masked_per_round], dtype=torch.int64, device=device)
        dist.all_reduce(t)
        unit_mask_total = int(t[0].item()) % unit_order
        unit_masked_per_round = int(t[1].item()) % unit_order
    torch.cuda.synchronize()

    total_clients_per_round = world * (clients // pool_n) * pool_n

    if args.h2d:
        # config #4 mode: updates live in pinned HOST memory (as after network
        # ingest + decrypt); double-buffered async H2D on a copy stream
        # overlaps with K3 on the compute stream. PCIe-bound by design.
        chunk = max(1, pool_n // 5)
        host_pool = pool.cpu().pin_memory()
        del pool
        dbuf = [torch.empty(chunk, eng.row_stride(), dtype=torch.uint8, device=device)
                for _ in range(2)]
        copy_stream = torch.cuda.Stream(device=device)
        copy_done = [torch.cuda.Event() for _ in range(2)]
        comp_done = [torch.cuda.Event() for _ in range(2)]
        comp_done[0].record()
        comp_done[1].record()

        def aggregate_round():
            cur = 0
            for _ in range(clients // pool_n):
                for j in range(0, pool_n, chunk):
                    n = min(chunk, pool_n - j)
                    with torch.cuda.stream(copy_stream):
                        copy_stream.wait_event(comp_done[cur])  # buffer free?
                        dbuf[cur][:n].copy_(host_pool[j : j + n], non_blocking=True)
                        copy_done[cur].record(copy_stream)
                    torch.cuda.current_stream().wait_event(copy_done[cur])
                    eng.aggregate_pool(dbuf[cur], n)
                    comp_done[cur].record()
                    cur ^= 1
    else:
        def aggregate_round():
            done = 0
            while done < (clients // pool_n) * pool_n:
                eng.aggregate_pool(pool, pool_n)
                done += pool_n

    def round_once():
        eng.reset()
        aggregate_round()
        eng.unit_acc = unit_masked_per_round
        eng.nb_models = total_clients_per_round
        return sharded.unmask_global(mask_total, unit_mask_total, total_clients_per_round)

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

    mode = "h2d-overlap " if args.h2d else ""
    collective = {"values_rs": "rccl values-reduce-scatter/all-gather",
                  "planes_rs": "rccl reduce-scatter/all-gather",
                  "all_reduce": "rccl all-reduce",
                  "single": "rccl all-reduce"}[sharded.strategy]
    _emit(args, rank, world, updates_per_sec, ms_per_step, total_clients_per_round, sanity,
          parallelism=f"client-sharded dp{world} + {collective}"
                      + (" + pinned-h2d copy/compute overlap" if args.h2d else ""),
          mode=mode)
    if world > 1:
        dist.destroy_process_group()


def _emit(args, rank, world, updates_per_sec, ms_per_step, total_clients, sanity, parallelism,
          mode=""):
    mcfg = {
        "f32-m6": ("f32", "Prime/F32/B0/M6 (7-byte limbs)"),
        "i64-m6": ("i64", "Prime/I64/B0/M6 (7-byte limbs)"),
        "f32-m3": ("f32", "Prime/F32/B0/M3 (6-byte limbs)"),
    }[args.mask_config]
    if rank != 0:
        return
    mm = args.length // 1_000_000
    result = {
        "metric": f"masked updates aggregated/sec ({mm}M-param {mcfg[0]} model, "
                  f"PET round: {mode}aggregate+all-reduce+unmask)",
        "value": round(updates_per_sec, 1),
        "unit": "updates/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": mcfg[0],
        "data": "synthetic",
        "config": {
            "model": f"pet-masked-aggregation-{mm}M",
            "global_batch": total_clients,
            "seq_len": args.length,
            "parallelism": parallelism,
            "mask_config": mcfg[1],
            "sanity_mean_abs_weight": round(sanity, 6),
        },
    }
    print(json.dumps(result))


def _run_stream(args, eng, cfg, rank, world, dist, device):
    """Config #5 mode: no resident update pool — each client's update is
    produced ON-GPU inside the timed region (K1 ChaCha20 expand -> K5
    mask+pack -> K3 aggregate), sized for 1B-param vectors in 288 GB HBM."""
    import torch

    clients = args.clients_per_gpu
    unit_order = int(cfg.order)
    shape = (2, args.length) if eng.wide else (args.length,)
    scratch = torch.empty(*shape, dtype=torch.int64, device=device)
    pool = eng.alloc_update_pool(1)

    # untimed: global mask total (sum of the clients' masks) + unit sums
    mask_total = torch.zeros(*shape, dtype=torch.int64, device=device)
    unit_mask_total, unit_masked = 0, 0
    seeds = [(rank * 1_000_003 + p + 1).to_bytes(32, "little") for p in range(clients)]
    for p, seed in enumerate(seeds):
        eng.derive_mask_values(seed, out=scratch)
        eng.mod_add_values(mask_total, scratch)
        unit_mask_total = (unit_mask_total + eng.unit_draw(seed)) % unit_order
        unit_masked = (unit_masked + eng.masked_unit_for(seed, 1, world * clients)) % unit_order
    from xaynet_amd.parallel import ShardedAggregation

    sharded = ShardedAggregation(eng, dist, rank, world)
    if world > 1:
        sharded.allreduce_mask(mask_total)
        t = torch.tensor([unit_mask_total, unit_masked], dtype=torch.int64, device=device)
        dist.all_reduce(t)
        unit_mask_total = int(t[0].item()) % unit_order
        unit_masked = int(t[1].item()) % unit_order
    torch.cuda.synchronize()
    total_clients = world * clients

    def round_once():
        eng.reset()
        for p, seed in enumerate(seeds):
            eng.derive_mask_values(seed, out=scratch)  # K1 (the dominant cost)
            eng.synth_update(pool, 0, scratch, participant=rank * clients + p,
                             scalar=1.0 / total_clients)  # K5
            eng.aggregate_pool(pool, 1)  # K3
        eng.unit_acc = unit_masked
        eng.nb_models = total_clients
        return sharded.unmask_global(mask_total, unit_mask_total, total_clients)

    for _ in range(args.warmup):
        out = round_once()
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
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
    _emit(args, rank, world, total_clients * args.steps / elapsed,
          elapsed / args.steps * 1000.0, total_clients, sanity,
          parallelism=f"client-sharded dp{world} + rccl {sharded.strategy}",
          mode="streamed expand+")
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
