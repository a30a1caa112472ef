"""Multi-GPU composition of the masked-aggregation engine over RCCL/xGMI.

One process per GPU (`torch.distributed`, backend "nccl" = RCCL on ROCm).
Because the accumulator is carry-free u64 digit planes, the cross-GPU
reduction is a PLAIN int64 sum — exactly what RCCL provides — with the
modular correction applied once afterwards (docs/ARCHITECTURE.md decision
#2; SURVEY.md §2.6 collective mapping).

xGMI is point-to-point (7 links per GPU), so ring collectives are
per-link-bound: prefer reduce-scatter of the planes plus an all-gather of
the (4x smaller) unmasked output over a full all-reduce. When
`world * order < 2^64` the reduce-scatter can run on canonical u64 values
instead of digit planes, halving the bytes on the wire again.

`bench.py` and the multi-GPU serve plane both drive their timed N>1 path
through this module, so the benchmark measures the production code;
`tests/test_distributed_planes.py` pins the underlying math on CPU (gloo,
world_size 2).
"""
from __future__ import annotations

import os

import torch

__all__ = ["choose_reduce_strategy", "ShardedAggregation"]


def choose_reduce_strategy(world: int, length: int, order_int: int,
                           env: dict | None = None) -> str:
    """Pick the cross-GPU reduction for a round.

    Returns one of:
      "single"     — world == 1, no collective.
      "values_rs"  — reduce-scatter canonical u64 values (half the bytes of
                     the plane RS). Valid while the summed values cannot
                     overflow 63 bits: order_bits + ceil(log2(world)) <= 63.
      "planes_rs"  — reduce-scatter each int64 digit plane, unmask the local
                     shard, all-gather the output.
      "all_reduce" — full plane all-reduce (fallback when the length does
                     not shard evenly, or forced with XAYNET_ALLREDUCE=1).
    """
    env = os.environ if env is None else env
    if world <= 1:
        return "single"
    if length % world != 0 or env.get("XAYNET_ALLREDUCE", "0") == "1":
        return "all_reduce"
    if order_int.bit_length() + (world - 1).bit_length() <= 63:
        return "values_rs"
    return "planes_rs"


class ShardedAggregation:
    """Drive one rank's share of a distributed aggregation round.

    Each rank owns a GpuMaskedAggregator of the FULL model length and
    aggregates its own participants' updates locally; `unmask_global`
    performs the cross-rank reduction + unmask and returns the full
    unmasked model on every rank.

    `mask_total` must be the GLOBAL canonical mask values (same on every
    rank — use `allreduce_mask` to combine per-rank partial mask sums).
    u64-order configs only (wide orders use the single-GPU staged plane).
    """

    def __init__(self, eng, dist, rank: int, world: int):
        if eng.wide and world > 1:
            raise NotImplementedError("cross-rank sharded aggregation covers u64 "
                                      "orders; wide (u128) configs run single-GPU")
        self.eng = eng
        self.dist = dist
        self.rank = rank
        self.world = world
        self.strategy = choose_reduce_strategy(world, eng.length, eng.order_int)
        self._shard = eng.length // world if self.strategy.endswith("_rs") else eng.length
        self._buf = {}  # persistent scratch (steady-state rounds reuse them)

    def _scratch(self, name: str, *shape, dtype=torch.int64):
        t = self._buf.get(name)
        if t is None or t.shape != shape or t.dtype != dtype:
            t = torch.empty(*shape, dtype=dtype, device=self.eng.device)
            self._buf[name] = t
        return t

    def allreduce_mask(self, mask_partial: torch.Tensor) -> torch.Tensor:
        """Modular all-reduce of per-rank canonical mask sums: lift to digit
        planes (plain int64 sum is overflow-free), all-reduce, canonicalize."""
        eng = self.eng
        if self.world <= 1:
            return mask_partial
        planes = self._scratch("mask_planes", eng.n_digits, eng.length)
        planes.zero_()
        eng.values_to_planes(mask_partial, planes)
        self.dist.all_reduce(planes)
        eng.planes_to_values(planes, mask_partial)
        return mask_partial

    def unmask_global(self, mask_total: torch.Tensor, unit_mask_total: int,
                      nb_models: int) -> torch.Tensor:
        """Cross-rank reduce + unmask; returns the full model on every rank."""
        eng, dist = self.eng, self.dist
        if self.strategy == "single":
            return eng.unmask(mask_total, unit_mask_total, nb_models=nb_models)
        if self.strategy == "all_reduce":
            dist.all_reduce(eng.acc)
            return eng.unmask(mask_total, unit_mask_total, nb_models=nb_models)
        shard, lo = self._shard, self.rank * self._shard
        out_full = self._scratch("out_full", eng.length,
                                 dtype=eng._TORCH_DTYPES[eng.vect_cfg.dtype])
        if self.strategy == "values_rs":
            canon = eng.canonical(out=self._scratch("canon", eng.length))
            vals_shard = self._scratch("vals_shard", shard)
            dist.reduce_scatter_tensor(vals_shard, canon)
            out_shard = eng.unmask_values(vals_shard, mask_total[lo:lo + shard],
                                          unit_mask_total, nb_models)
        else:  # planes_rs
            shard_planes = self._scratch("shard_planes", eng.n_digits, shard)
            for d in range(eng.n_digits):
                dist.reduce_scatter_tensor(shard_planes[d], eng.acc[d])
            out_shard = eng.unmask_planes(shard_planes, mask_total[lo:lo + shard],
                                          unit_mask_total, nb_models)
        dist.all_gather_into_tensor(out_full, out_shard)
        return out_full
