"""Participant-side GPU accelerator: wires the MI355X kernels into the SDK's
update-masking and sum2 mask-aggregation hot loops.

Usage:
    accel = ParticipantAccel(vect_cfg, unit_cfg, length)   # one per client
    accel.attach(participant)                              # sets both hooks

The hooks fall back to the CPU path (return None) on any error or shape
mismatch, so a GPU-less box or a mid-round config change degrades cleanly.
Replaces: Masker::mask (masking.rs:358-404) via K1+K5w and the sum2
derive+aggregate loop (xaynet-sdk sum2.rs:170-190) via K1+K2."""
from __future__ import annotations

import logging
import threading

import numpy as np

LOG = logging.getLogger("xaynet.accel")

_NP_DTYPES = {0: np.float32, 1: np.float64, 2: np.int32, 3: np.int64}


class ParticipantAccel:
    def __init__(self, vect_cfg, unit_cfg, length: int, device: str = "cuda:0",
                 scalar_num: int = 1, scalar_den: int = 1):
        import torch

        from .engine import GpuMaskedAggregator

        self.torch = torch
        self.eng = GpuMaskedAggregator(vect_cfg, unit_cfg, length, device=device)
        self.vect_cfg = vect_cfg
        self.unit_cfg = unit_cfg
        self.length = length
        self.scalar_num = scalar_num
        self.scalar_den = scalar_den
        # one accelerator may be shared by many participants (threads): the
        # engine + scratch are serialized under a lock
        self._mu = threading.Lock()
        # persistent sum2 scratch
        shape = (2, length) if self.eng.wide else (length,)
        self._total = torch.zeros(*shape, dtype=torch.int64, device=self.eng.device)
        self._scratch = torch.empty(*shape, dtype=torch.int64, device=self.eng.device)

    # ---- hooks ----

    def mask_model(self, seed: bytes, dtype: int, raw: bytes, n: int):
        with self._mu:
            return self._mask_model(seed, dtype, raw, n)

    def _mask_model(self, seed: bytes, dtype: int, raw: bytes, n: int):
        try:
            if n != self.length:
                return None
            w = np.frombuffer(raw, dtype=_NP_DTYPES[dtype])
            t = self.torch.from_numpy(w.copy())
            return self.eng.mask_weights(seed, t, self.scalar_num, self.scalar_den)
        except Exception:  # noqa: BLE001 — CPU fallback
            LOG.exception("GPU mask_model hook failed; falling back to CPU")
            return None

    def aggregate_masks(self, seeds):
        with self._mu:
            return self._aggregate_masks(seeds)

    def _aggregate_masks(self, seeds):
        try:
            from xaynet_amd import _core

            eng = self.eng
            self._total.zero_()
            unit_order = int(self.unit_cfg.order)
            unit_total = 0
            for seed in seeds:
                eng.derive_mask_values(bytes(seed), out=self._scratch)
                eng.mod_add_values(self._total, self._scratch)
                unit_total = (unit_total + eng.unit_draw(bytes(seed))) % unit_order
            limbs_dev = eng.pack_wire(self._total)
            nlimb = self.length * eng.bpn
            ubpn = self.unit_cfg.bytes_per_number
            wire = bytearray(8 + nlimb + 4 + ubpn)
            wire[0:4] = bytes(self.vect_cfg.to_bytes())
            wire[4:8] = self.length.to_bytes(4, "big")
            if getattr(eng, "_wire_pin", None) is None or eng._wire_pin.numel() < nlimb:
                eng._wire_pin = self.torch.empty(nlimb, dtype=self.torch.uint8,
                                                 pin_memory=True)
            pin = eng._wire_pin[:nlimb]
            pin.copy_(limbs_dev)  # D2H at PCIe line rate (pinned)
            np.frombuffer(wire, dtype=np.uint8, count=nlimb, offset=8)[:] = pin.numpy()
            off = 8 + nlimb
            wire[off : off + 4] = bytes(self.unit_cfg.to_bytes())
            wire[off + 4 :] = unit_total.to_bytes(ubpn, "little")
            return bytes(wire)
        except Exception:  # noqa: BLE001
            LOG.exception("GPU sum2 hook failed; falling back to CPU")
            return None

    def attach(self, participant) -> None:
        participant.set_mask_model_hook(self.mask_model)
        participant.set_sum2_hook(self.aggregate_masks)
