"""GPU-accelerated sum2 task: aggregate the update participants' masks.

The reference's hottest client-side loop (the sum participant
derives `n_updaters x model_length` PRNG draws and modularly adds them,
xaynet-sdk phases/sum2.rs:170-190). On an MI355X-equipped sum participant
this runs as K1 ChaCha20 expansion + K2 modular adds and emits the exact
MaskObject wire bytes the coordinator expects — bit-identical to the CPU
oracle (the K1 compaction reproduces the reference's sequential rejection
stream).

All orders up to 2^128 (u64 fast path + split lo/hi u128 kernels); only
Bmax orders beyond 2^128 stay on the CPU path.
"""
from __future__ import annotations

import numpy as np
import torch

from xaynet_amd import _core

from .engine import GpuMaskedAggregator


def aggregate_masks(seeds, vect_cfg, unit_cfg, length: int, device: str = "cuda:0") -> bytes:
    """Derive and modularly aggregate one mask per 32-byte seed; returns the
    aggregated MaskObject wire bytes (MaskVect || MaskUnit)."""
    if not seeds:
        raise ValueError("no seeds")
    mk = _core.mask
    eng = GpuMaskedAggregator(vect_cfg, unit_cfg, length, device=device)
    shape = (2, length) if eng.wide else (length,)
    total = torch.zeros(*shape, dtype=torch.int64, device=eng.device)
    scratch = torch.empty(*shape, dtype=torch.int64, device=eng.device)
    unit_order = int(unit_cfg.order)
    unit_total = 0
    for seed in seeds:
        eng.derive_mask_values(seed, out=scratch)
        eng.mod_add_values(total, scratch)
        unit_total = (unit_total + eng.unit_draw(seed)) % unit_order

    limbs_dev = eng.pack_wire(total)
    nlimb = length * eng.bpn
    ubpn = unit_cfg.bytes_per_number
    wire = bytearray(8 + nlimb + 4 + ubpn)
    wire[0:4] = bytes(vect_cfg.to_bytes())
    wire[4:8] = length.to_bytes(4, "big")
    if getattr(eng, "_wire_pin", None) is None or eng._wire_pin.numel() < nlimb:
        eng._wire_pin = torch.empty(nlimb, dtype=torch.uint8, pin_memory=True)
    pin = eng._wire_pin[:nlimb]
    pin.copy_(limbs_dev)  # D2H at PCIe line rate (pinned)
    np.frombuffer(wire, dtype=np.uint8, count=nlimb, offset=8)[:] = pin.numpy()
    off = 8 + nlimb
    wire[off : off + 4] = bytes(unit_cfg.to_bytes())
    wire[off + 4 :] = unit_total.to_bytes(ubpn, "little")
    return bytes(wire)
