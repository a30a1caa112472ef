"""Participant-side GPU accelerator: wires the MI355X kernels into the SDK's
update-masking and sum2 mask-aggregation hot loops.

Usage:
    accel = ParticipantAccel()          # auto-configures per round
    accel.attach(participant)           # sets both hooks
    # or: xaynet_sdk.Participant(url, gpu=True) / spawn_participant(gpu=True)

The hooks receive the round's mask config + model length from the SDK and
lazily build (and cache) one engine per (config, length). They fall back to
the CPU path (return None) on any error, so a GPU-less box or an
out-of-range config degrades cleanly. Replaces: Masker::mask
(masking.rs:358-404) via K1+K5w and the sum2 derive+aggregate loop
(xaynet-sdk sum2.rs:170-190) via K1+K2."""
from __future__ import annotations

import logging
import threading

import numpy as np

LOG = logging.getLogger("xaynet.accel")

_NP_DTYPES = {0: np.float32, 1: np.float64, 2: np.int32, 3: np.int64}


class ParticipantAccel:
    def __init__(self, device: str = "cuda:0", scalar_num: int = 1, scalar_den: int = 1):
        import torch

        self.torch = torch
        self.device = device
        self.scalar_num = scalar_num
        self.scalar_den = scalar_den
        # one accelerator may be shared by many participants (threads): the
        # engines + scratch are serialized under a lock
        self._mu = threading.Lock()
        self._engines = {}  # (cfg8, length) -> (engine, total, scratch, cfgs)

    def _engine_for(self, cfg8, length: int):
        key = (tuple(cfg8), length)
        ent = self._engines.get(key)
        if ent is None:
            from xaynet_amd import _core

            from .engine import GpuMaskedAggregator

            mk = _core.mask
            vect_cfg = mk.MaskConfig(*cfg8[:4])
            unit_cfg = mk.MaskConfig(*cfg8[4:])
            eng = GpuMaskedAggregator(vect_cfg, unit_cfg, length, device=self.device)
            shape = (2, length) if eng.wide else (length,)
            total = self.torch.zeros(*shape, dtype=self.torch.int64, device=eng.device)
            scratch = self.torch.empty(*shape, dtype=self.torch.int64, device=eng.device)
            ent = (eng, total, scratch, vect_cfg, unit_cfg)
            self._engines[key] = ent
        return ent

    # ---- hooks ----

    def mask_model(self, seed: bytes, dtype: int, raw: bytes, n: int, cfg8):
        with self._mu:
            try:
                eng = self._engine_for(cfg8, n)[0]
                w = np.frombuffer(raw, dtype=_NP_DTYPES[dtype])
                t = self.torch.from_numpy(w.copy())
                return eng.mask_weights(seed, t, self.scalar_num, self.scalar_den)
            except Exception:  # noqa: BLE001 — CPU fallback
                LOG.exception("GPU mask_model hook failed; falling back to CPU")
                return None

    def aggregate_masks(self, seeds, length: int, cfg8):
        with self._mu:
            try:
                eng, total, scratch, vect_cfg, unit_cfg = self._engine_for(cfg8, length)
                total.zero_()
                unit_order = int(unit_cfg.order)
                unit_total = 0
                for seed in seeds:
                    eng.derive_mask_values(bytes(seed), out=scratch)
                    eng.mod_add_values(total, scratch)
                    unit_total = (unit_total + eng.unit_draw(bytes(seed))) % unit_order
                limbs_dev = eng.pack_wire(total)
                nlimb = length * eng.bpn
                ubpn = unit_cfg.bytes_per_number
                wire = bytearray(8 + nlimb + 4 + ubpn)
                wire[0:4] = bytes(vect_cfg.to_bytes())
                wire[4:8] = length.to_bytes(4, "big")
                if getattr(eng, "_wire_pin", None) is None or eng._wire_pin.numel() < nlimb:
                    eng._wire_pin = self.torch.empty(nlimb, dtype=self.torch.uint8,
                                                     pin_memory=True)
                pin = eng._wire_pin[:nlimb]
                pin.copy_(limbs_dev)  # D2H at PCIe line rate (pinned)
                np.frombuffer(wire, dtype=np.uint8, count=nlimb, offset=8)[:] = pin.numpy()
                off = 8 + nlimb
                wire[off : off + 4] = bytes(unit_cfg.to_bytes())
                wire[off + 4 :] = unit_total.to_bytes(ubpn, "little")
                return bytes(wire)
            except Exception:  # noqa: BLE001
                LOG.exception("GPU sum2 hook failed; falling back to CPU")
                return None

    def attach(self, participant) -> None:
        participant.set_mask_model_hook(self.mask_model)
        participant.set_sum2_hook(self.aggregate_masks)
