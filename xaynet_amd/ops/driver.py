"""GPU data-plane driver for the staged coordinator.

The C++ coordinator (AggregationPlane::Staged) validates and stages masked
updates as MaskObject wire bytes during the Update phase; this driver drains
them onto the MI355X, aggregates limbs in digit planes (K3), and at Unmask
derives the final model (K6 unpack + K4 modular finalize/unmask), returning
the bincode Option<Model> body the coordinator broadcasts and persists.

This replaces the reference's in-RAM `Aggregation::aggregate` hot loop
(rust/xaynet-core/src/mask/masking.rs:292-316) with the GPU engine while the
protocol thread stays untouched (SURVEY.md §2.6 K3/K4/K6 mapping).

All F32/F64/I32/I64 configs with group orders up to 2^128 are supported
(every BASELINE.json config, plus the F64 wide-order families via the u128
kernel set); only Bmax orders beyond 2^128 take the CPU oracle plane.
"""
from __future__ import annotations

import logging
import threading

import numpy as np
import torch

from xaynet_amd import _core

from .engine import GpuMaskedAggregator

LOG = logging.getLogger("xaynet.gpu_driver")


class GpuCoordinatorDriver(threading.Thread):
    def __init__(self, coordinator, vect_cfg, unit_cfg, length: int,
                 device: str = "cuda:0", pool_size: int = 512, poll_s: float = 0.002):
        super().__init__(daemon=True)
        self.coordinator = coordinator
        self.eng = GpuMaskedAggregator(vect_cfg, unit_cfg, length, device=device)
        self.length = length
        self.pool_size = pool_size
        self.pool = self.eng.alloc_update_pool(pool_size)
        self.poll_s = poll_s
        self._stop_event = threading.Event()
        self._row = 0
        self._unit_sum = 0
        self._supplied_round = -1
        self.rounds_unmasked = 0

        self._vect_bpn = self.eng.bpn
        self._unit_bpn = unit_cfg.bytes_per_number
        self._unit_order = int(unit_cfg.order)

    # ------------------------------------------------------------- wire

    def _split_mask_object(self, wire: bytes):
        """MaskObject = MaskVect(config 4B | count u32 BE | limbs) ||
        MaskUnit(config 4B | value limb) — reference
        mask/object/serialization/mod.rs:29-80, vect.rs, unit.rs."""
        count = int.from_bytes(wire[4:8], "big")
        if count != self.length:
            raise ValueError(f"masked vector length {count} != model length {self.length}")
        off = 8
        vect = wire[off : off + count * self._vect_bpn]
        off += count * self._vect_bpn
        off += 4  # unit config
        unit = int.from_bytes(wire[off : off + self._unit_bpn], "little")
        return vect, unit

    # ------------------------------------------------------------- ingest

    def _flush(self):
        if self._row:
            self.eng.aggregate_pool(self.pool, self._row, self._unit_sum)
            self._row = 0
            self._unit_sum = 0

    def _ingest(self, wire: bytes):
        vect, unit = self._split_mask_object(wire)
        self.eng.upload_update(self.pool, self._row, vect)
        self._unit_sum = (self._unit_sum + unit) % self._unit_order
        self._row += 1
        if self._row == self.pool_size:
            self._flush()

    # ------------------------------------------------------------- unmask

    def _finish(self, mask_bytes: bytes, nb_models: int):
        self._flush()
        torch.cuda.synchronize(self.eng.device)
        if self.eng.nb_models != nb_models:
            LOG.warning("driver aggregated %d updates, coordinator staged %d",
                        self.eng.nb_models, nb_models)
        mask_vect, mask_unit = self._split_mask_object(mask_bytes)
        t = torch.frombuffer(bytearray(mask_vect), dtype=torch.uint8).to(self.eng.device)
        mask_vals = self.eng.unpack_wire(t)
        out = self.eng.unmask(mask_vals, mask_unit, nb_models=nb_models)
        body = _core.sdk.encode_model(out.cpu().numpy())
        self.coordinator.supply_unmasked_model(body)
        self.eng.reset()
        self.rounds_unmasked += 1

    # ------------------------------------------------------------- loop

    def run(self):
        last_round = -1
        try:
            while not self._stop_event.is_set():
                work = False
                rid = self.coordinator.round_id
                if rid != last_round:
                    # new round (or a failed round restarted): drop any
                    # partially-ingested state so stale updates cannot leak
                    # into the next round's accumulator
                    if self._row or self.eng.nb_models:
                        self.eng.reset()
                        self._row = 0
                        self._unit_sum = 0
                    last_round = rid
                for wire in self.coordinator.drain_staged_updates():
                    self._ingest(bytes(wire))
                    work = True
                pu = self.coordinator.pending_unmask()
                if pu is not None and self.coordinator.round_id != self._supplied_round:
                    self._supplied_round = self.coordinator.round_id
                    self._finish(bytes(pu[0]), int(pu[1]))
                    work = True
                if not work:
                    self._stop_event.wait(self.poll_s)
        except Exception:  # noqa: BLE001 — surface in logs, release coordinator
            LOG.exception("GPU driver failed; coordinator round will fail over")

    def stop(self):
        self._stop_event.set()
        self.join(timeout=10)
