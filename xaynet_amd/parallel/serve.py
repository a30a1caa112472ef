"""Multi-GPU serve plane: per-device worker processes fed by a pinned
shared-memory ingest ring, composed with ShardedAggregation at unmask.

This makes `python -m xaynet_amd.server` itself multi-GPU (VERDICT r01 item
1): the coordinator's protocol thread stages validated masked updates (wire
bytes); the `MultiGpuServeDriver` thread drains them, writes the vector limbs
into per-worker shared-memory slots and round-robins batches across one
worker process per GPU. Each worker pins the shm mapping (hipHostRegister),
double-buffers async H2D copies on a copy stream overlapped with K3
aggregation on the compute stream (the productized ingest
pipeline — no synchronous pageable copies), and at Unmask the workers run
`ShardedAggregation.unmask_global` over RCCL/xGMI; rank 0 ships the unmasked
model back to the coordinator.

Replaces: the reference's single-threaded in-RAM accumulator
(rust/xaynet-server/src/state_machine/phases/update.rs:35-40) — here the
accumulator is sharded across all visible MI355X GPUs.

The same code runs on CPU boxes for CI (`device_kind="cpu"`: gloo backend +
CpuPlaneAggregator, plain memcpy instead of pinned H2D), so the distributed
serve path is proven by tests before any hardware run.
"""
from __future__ import annotations

import logging
import os
import queue
import sys
import threading
import traceback
from multiprocessing import get_context
from multiprocessing import shared_memory

LOG = logging.getLogger("xaynet.serve_plane")

_MP = get_context("spawn")  # CUDA-safe worker start


def _slot_bytes(length: int, bpn: int) -> int:
    return (length * bpn + 4095) // 4096 * 4096


# --------------------------------------------------------------- worker


def _worker_main(rank: int, world: int, device_kind: str, cfg_args, length: int,
                 slots: int, shm_name: str, port: int, cmd_conn, free_conn, res_q,
                 batch: int):
    """One rank: owns one GPU (or a CPU engine), aggregates its share of the
    staged updates, joins the collective unmask."""
    try:
        import numpy as np
        import torch

        from xaynet_amd import _core
        from xaynet_amd.parallel import ShardedAggregation

        mk = _core.mask
        vect_cfg = mk.MaskConfig(*cfg_args)
        unit_cfg = mk.MaskConfig(*cfg_args)
        sbytes = _slot_bytes(length, vect_cfg.bytes_per_number)

        shm = shared_memory.SharedMemory(name=shm_name)
        host = np.frombuffer(shm.buf, dtype=np.uint8)

        dist = None
        if world > 1:
            import torch.distributed as tdist

            backend = "nccl" if device_kind == "cuda" else "gloo"
            tdist.init_process_group(
                backend, init_method=f"tcp://127.0.0.1:{port}", rank=rank, world_size=world
            )
            dist = tdist

        if device_kind == "cuda":
            from xaynet_amd import _hip
            from xaynet_amd.ops import GpuMaskedAggregator

            torch.cuda.set_device(rank)
            eng = GpuMaskedAggregator(vect_cfg, unit_cfg, length, device=f"cuda:{rank}")
            # pin the whole ring once: H2D from it becomes true async DMA
            _hip.host_register(host.ctypes.data, host.nbytes)
            host_t = torch.from_numpy(host)
            copy_stream = torch.cuda.Stream(device=eng.device)
            copy_done = torch.cuda.Event()
        else:
            from xaynet_amd.ops.cpu_engine import CpuPlaneAggregator

            eng = CpuPlaneAggregator(vect_cfg, unit_cfg, length)
            host_t = torch.from_numpy(host)
            copy_stream = None

        sharded = ShardedAggregation(eng, dist, rank, world)
        received = 0
        pool = eng.alloc_update_pool(batch)
        row_bytes = length * eng.bpn
        mask_vals = None

        def stage_and_aggregate(slot_list):
            """Pinned-ring slots -> pool rows (async H2D) -> K3; frees slots."""
            nonlocal_free = []
            for start in range(0, len(slot_list), batch):
                group = slot_list[start : start + batch]
                if copy_stream is not None:
                    with torch.cuda.stream(copy_stream):
                        for i, slot in enumerate(group):
                            off = slot * sbytes
                            pool[i, :row_bytes].copy_(
                                host_t[off : off + row_bytes], non_blocking=True)
                        copy_done.record(copy_stream)
                    torch.cuda.current_stream().wait_event(copy_done)
                    eng.aggregate_pool(pool, len(group))
                    # free slots only once the DMA out of them has finished;
                    # K3 keeps running async on the compute stream meanwhile
                    copy_done.synchronize()
                else:
                    for i, slot in enumerate(group):
                        off = slot * sbytes
                        pool[i, :row_bytes].copy_(host_t[off : off + row_bytes])
                    eng.aggregate_pool(pool, len(group))
                for slot in group:
                    nonlocal_free.append(slot)
            for slot in nonlocal_free:
                free_conn.send(slot)

        while True:
            cmd = cmd_conn.recv()
            op = cmd[0]
            if op == "agg":
                received += len(cmd[1])
                stage_and_aggregate(cmd[1])
            elif op == "unmask":
                _, mask_slot, mask_unit, unit_acc, nb_models, round_id = cmd
                off = mask_slot * sbytes
                packed = host_t[off : off + row_bytes]
                if device_kind == "cuda":
                    packed = packed.to(eng.device, non_blocking=False)
                mask_vals = eng.unpack_wire(packed.contiguous())
                del packed  # drop the ring-slice view (shm must close cleanly)
                free_conn.send(mask_slot)
                eng.unit_acc = unit_acc
                out = sharded.unmask_global(mask_vals, mask_unit, nb_models)
                local_n = eng.nb_models
                eng.reset()
                received = 0
                if rank == 0:
                    res_q.put(("model", round_id, out.cpu().numpy(), local_n))
                else:
                    res_q.put(("done", round_id, None, local_n))
            elif op == "reset":
                eng.reset()
            elif op == "stop":
                break
        if dist is not None:
            dist.destroy_process_group()
        if device_kind == "cuda":
            _hip.host_unregister(host.ctypes.data)
        del host_t, host, pool, mask_vals  # release shm buffer exports
        import gc

        gc.collect()
        try:
            shm.close()
        except BufferError:
            pass  # a stray view (loop slice) still holds an export; the
            # mapping dies with the process either way
    except Exception:  # noqa: BLE001 — ship the traceback to the parent
        res_q.put(("error", rank, traceback.format_exc(), 0))


# --------------------------------------------------------------- parent


class ServePlane:
    """Parent-side handle: per-worker shm rings + queues + processes."""

    def __init__(self, cfg_args, length: int, n_workers: int, device_kind: str,
                 slots_per_worker: int = 32, batch: int = 16):
        import numpy as np

        from xaynet_amd import _core

        cfg = _core.mask.MaskConfig(*cfg_args)
        self.bpn = cfg.bytes_per_number
        self.length = length
        self.world = n_workers
        self.device_kind = device_kind
        self.sbytes = _slot_bytes(length, self.bpn)
        self.slots = slots_per_worker
        self._np = np

        port = _free_port()
        self.res_q = _MP.Queue()
        # per-worker transports are PIPES with synchronous sends: mp.Queue's
        # background feeder thread can be GIL-starved for seconds under
        # ingest load, leaving dispatched batches unsent while the driver
        # blocks on ring slots (observed; see test_serve_plane backpressure)
        self.shms, self.views, self.cmd_conns, self.free_conns, self.procs = [], [], [], [], []
        self._free_slots = []  # parent-side cache of free slot ids per worker
        for r in range(n_workers):
            shm = shared_memory.SharedMemory(create=True, size=self.sbytes * self.slots)
            self.shms.append(shm)
            self.views.append(np.frombuffer(shm.buf, dtype=np.uint8))
            # Pipe(duplex=False) -> (recv_end, send_end)
            cmd_recv, cmd_send = _MP.Pipe(duplex=False)
            free_recv, free_send = _MP.Pipe(duplex=False)
            self.cmd_conns.append(cmd_send)    # parent -> worker commands
            self.free_conns.append(free_recv)  # worker -> parent freed slots
            self._free_slots.append(list(range(self.slots)))
            p = _MP.Process(
                target=_worker_main,
                args=(r, n_workers, device_kind, tuple(cfg_args), length, self.slots,
                      shm.name, port, cmd_recv, free_send, self.res_q, batch),
                daemon=True,
            )
            p.start()
            cmd_recv.close()   # worker's read end, not ours
            free_send.close()  # worker's write end, not ours
            self.procs.append(p)
        self._rr = 0
        self._pending = [[] for _ in range(n_workers)]
        self.dispatched = 0
        # concurrency: slot accounting + pipe sends are tiny critical
        # sections; the big ring memcpys happen OUTSIDE the locks so
        # several driver drain threads can overlap them (the pop/copy
        # bindings release the GIL)
        self._rr_lock = threading.Lock()
        self._wlocks = [threading.Lock() for _ in range(n_workers)]

    # ---- ingest ----

    def _dispatch(self, r: int):
        # caller holds self._wlocks[r]
        if self._pending[r]:
            self.dispatched += len(self._pending[r])
            self.cmd_conns[r].send(("agg", self._pending[r]))
            self._pending[r] = []

    def _take_slot(self, r: int) -> int:
        """Free slot for worker r; backpressure when its ring is full."""
        fc = self.free_conns[r]
        try:
            while fc.poll(0):
                self._free_slots[r].append(fc.recv())
            if self._free_slots[r]:
                return self._free_slots[r].pop()
            # ring exhausted: make sure everything we hold is dispatched (the
            # worker can only free slots it has received), then wait
            self._dispatch(r)
            while True:
                if fc.poll(5.0):
                    return fc.recv()
                if not self.procs[r].is_alive():
                    raise RuntimeError(f"serve-plane worker {r} died")
        except (EOFError, BrokenPipeError, OSError) as e:
            raise RuntimeError(f"serve-plane worker {r} died") from e

    def put_update_direct(self, writer):
        """Zero-copy ingest: take a slot on the next worker and let
        `writer(ptr, capacity)` fill it directly (e.g. the coordinator's
        pop_staged_vect). A None result returns the slot unused; otherwise
        the slot is queued and the writer's result returned. Thread-safe:
        the slot bookkeeping is locked per worker, the (multi-hundred-MB)
        writer copy itself runs unlocked and overlaps across threads."""
        with self._rr_lock:
            r = self._rr
            self._rr = (self._rr + 1) % self.world
        with self._wlocks[r]:
            if len(self._pending[r]) >= max(1, self.slots // 4):
                self._dispatch(r)
            slot = self._take_slot(r)
        off = slot * self.sbytes
        try:
            res = writer(self.views[r].ctypes.data + off, self.sbytes)
        except BaseException:
            with self._wlocks[r]:
                self._free_slots[r].append(slot)
            raise
        with self._wlocks[r]:
            if res is None:
                self._free_slots[r].append(slot)
                return None
            self._pending[r].append(slot)
        return res

    def put_update(self, vect_bytes: bytes | memoryview):
        """Write one update's vector limbs into the next worker's ring."""
        with self._rr_lock:
            r = self._rr
            self._rr = (self._rr + 1) % self.world
        with self._wlocks[r]:
            if len(self._pending[r]) >= max(1, self.slots // 4):
                self._dispatch(r)
            slot = self._take_slot(r)
        off = slot * self.sbytes
        v = self.views[r]
        v[off : off + len(vect_bytes)] = self._np.frombuffer(vect_bytes, dtype=self._np.uint8)
        with self._wlocks[r]:
            self._pending[r].append(slot)

    def flush(self):
        for r in range(self.world):
            with self._wlocks[r]:
                self._dispatch(r)

    # ---- unmask ----

    def unmask(self, mask_vect: bytes, mask_unit: int, unit_acc: int, nb_models: int,
               round_id: int, timeout: float = 600.0):
        """Broadcast the winning mask, run the collective unmask, return the
        unmasked weights (numpy array) from rank 0."""
        self.flush()
        for r in range(self.world):
            with self._wlocks[r]:
                slot = self._take_slot(r)
                off = slot * self.sbytes
                self.views[r][off : off + len(mask_vect)] = self._np.frombuffer(
                    mask_vect, dtype=self._np.uint8)
                self.cmd_conns[r].send(
                    ("unmask", slot, mask_unit, unit_acc, nb_models, round_id))
        model = None
        total_n = 0
        for _ in range(self.world):
            kind, rid_or_rank, payload, local_n = self.res_q.get(timeout=timeout)
            if kind == "error":
                raise RuntimeError(f"serve-plane worker {rid_or_rank} failed:\n{payload}")
            total_n += local_n
            if kind == "model":
                model = payload
        if total_n != nb_models:
            LOG.warning("workers aggregated %d updates, coordinator staged %d "
                        "(parent dispatched %d)", total_n, nb_models, self.dispatched)
        return model

    def reset(self):
        """Drop partially-ingested round state (round restart)."""
        for r in range(self.world):
            with self._wlocks[r]:
                self._pending[r] = []
                self.cmd_conns[r].send(("reset",))

    def stop(self):
        for conn in self.cmd_conns:
            try:
                conn.send(("stop",))
            except Exception:  # noqa: BLE001
                pass
        for p in self.procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
        self.views = []  # release shm buffer exports before close()
        import gc

        gc.collect()
        for shm in self.shms:
            try:
                shm.close()
                shm.unlink()
            except (FileNotFoundError, BufferError):
                pass


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class MultiGpuServeDriver(threading.Thread):
    """Coordinator-side driver thread: drains staged updates into the
    ServePlane and completes the Unmask handoff (multi-GPU analog of
    ops.driver.GpuCoordinatorDriver)."""

    def __init__(self, coordinator, vect_cfg, unit_cfg, length: int,
                 n_workers: int, device_kind: str = "cuda",
                 slots_per_worker: int = 32, batch: int = 16, poll_s: float = 0.002):
        super().__init__(daemon=True)
        self.coordinator = coordinator
        self.length = length
        self.poll_s = poll_s
        self._vect_bpn = vect_cfg.bytes_per_number
        self._unit_bpn = unit_cfg.bytes_per_number
        self._unit_order = int(unit_cfg.order)
        self.plane = ServePlane(
            (vect_cfg.group, vect_cfg.dtype, vect_cfg.bound, vect_cfg.model),
            length, n_workers, device_kind,
            slots_per_worker=slots_per_worker, batch=batch)
        self._stop_event = threading.Event()
        self._acc_lock = threading.Lock()
        self._unit_acc = 0
        self._nb = 0
        self._supplied_round = -1
        self.rounds_unmasked = 0

    def _split_mask_object(self, wire: bytes):
        """MaskObject wire = MaskVect(config 4B | count u32 BE | limbs) ||
        MaskUnit(config 4B | limb) — reference serialization/mod.rs:29-80."""
        count = int.from_bytes(wire[4:8], "big")
        if count != self.length:
            raise ValueError(f"masked vector length {count} != model length {self.length}")
        off = 8
        vect = wire[off : off + count * self._vect_bpn]
        off += count * self._vect_bpn + 4
        unit = int.from_bytes(wire[off : off + self._unit_bpn], "little")
        return vect, unit

    def run(self):
        last_round = -1
        try:
            while not self._stop_event.is_set():
                work = False
                rid = self.coordinator.round_id
                if rid != last_round:
                    if self._nb:
                        self.plane.reset()
                        self._unit_acc = 0
                        self._nb = 0
                    last_round = rid
                if self._drain():
                    work = True
                pu = self.coordinator.pending_unmask()
                if pu is not None and self.coordinator.round_id != self._supplied_round:
                    self._supplied_round = self.coordinator.round_id
                    self._finish(bytes(pu[0]), int(pu[1]))
                    work = True
                if not work:
                    self._stop_event.wait(self.poll_s)
        except Exception:  # noqa: BLE001
            LOG.exception("serve-plane driver failed; coordinator round will fail over")

    def _drain_one(self) -> bool:
        """Pop one staged update into a ring slot; False when none left."""
        # zero-copy: the staged MaskObject's limbs are memcpy'd straight
        # into the pinned ring slot (one copy, vs serialize -> py bytes
        # -> slice -> ring = four copies of a 175 MB update)
        res = self.plane.put_update_direct(
            lambda ptr, cap: self.coordinator.pop_staged_vect(ptr, cap))
        if res is None:
            return False
        nbytes, unit_bytes = res
        expected = self.length * self._vect_bpn
        if nbytes != expected:
            raise RuntimeError(f"staged vect {nbytes} B != model {expected} B")
        unit = int.from_bytes(bytes(unit_bytes), "little")
        with self._acc_lock:
            self._unit_acc = (self._unit_acc + unit) % self._unit_order
            self._nb += 1
        return True

    def _drain(self) -> bool:
        work = self._drain_one()
        if work and self.coordinator.staged_count() > 1:
            # backlog: overlap the big ring memcpys across a few helper
            # threads (pop_staged_vect releases the GIL around its copy,
            # so this parallelizes the drain's memory bandwidth)
            errs = []

            def helper():
                try:
                    while self._drain_one():
                        pass
                except Exception as e:  # noqa: BLE001 — re-raised in driver
                    errs.append(e)

            n_help = min(3, max(1, self.plane.world))
            threads = [threading.Thread(target=helper, daemon=True)
                       for _ in range(n_help)]
            for t in threads:
                t.start()
            while self._drain_one():
                pass
            for t in threads:
                t.join()
            if errs:
                raise errs[0]
        if work:
            self.plane.flush()
        return work

    def _finish(self, mask_bytes: bytes, nb_models: int):
        from xaynet_amd import _core

        # a drain can stall mid-batch on ring backpressure while the protocol
        # thread races ahead to Unmask: collect every remaining staged update
        # BEFORE the collective (pending_unmask guarantees staging is done)
        while self._drain():
            pass
        mask_vect, mask_unit = self._split_mask_object(mask_bytes)
        weights = self.plane.unmask(
            mask_vect, mask_unit, self._unit_acc, nb_models,
            round_id=int(self.coordinator.round_id))
        body = _core.sdk.encode_model(weights)
        self.coordinator.supply_unmasked_model(body)
        self._unit_acc = 0
        self._nb = 0
        self.rounds_unmasked += 1

    def stop(self):
        self._stop_event.set()
        self.join(timeout=15)
        self.plane.stop()
