"""MI355X masked-aggregation engine (single GPU; multi-GPU composition in
xaynet_amd.parallel).

Data layout (see csrc/gpu/kernels.hip):
  - updates: wire-format packed limb rows, uint8 [n][stride] device tensor
  - accumulator: u64 digit planes, int64 [n_digits32][len] device tensor
    (plain integer sums; modular reduction deferred to finalize — this is
    what makes cross-GPU RCCL int64 all-reduce valid)
  - masks: canonical u64 values, int64 [len]

Memory is owned by torch (device tensors); kernels launch on the null
stream, which serializes with torch's default stream.

Fails loudly if the _hip extension or a GPU is unavailable — there is no
silent CPU fallback on a GPU box. Orders up to 2^128 are supported (wide
configs use split lo/hi u64 planes); only Bmax orders beyond 2^128 are
routed to the CPU oracle by the caller.
"""
import os

import torch

from xaynet_amd import _core

try:
    from xaynet_amd import _hip
except ImportError as e:  # pragma: no cover
    _hip = None
    _hip_err = e


def gpu_available() -> bool:
    return _hip is not None and _hip.device_count() > 0


def _require_gpu():
    if _hip is None:
        raise RuntimeError(f"xaynet_amd._hip extension not built: {_hip_err}")
    if _hip.device_count() == 0:
        raise RuntimeError("no AMD GPU visible (xaynet_amd._hip loaded, hipGetDeviceCount=0)")


class GpuMaskedAggregator:
    """Digit-plane aggregation of wire-format masked updates on one GPU."""

    def __init__(self, vect_cfg, unit_cfg, length: int, device: str = "cuda"):
        _require_gpu()
        if vect_cfg.bytes_per_number > 16:
            raise ValueError(
                f"GPU path requires group order <= 2^128 (got {vect_cfg.bytes_per_number} "
                "bytes/element); use the CPU oracle for this config"
            )
        self.vect_cfg = vect_cfg
        self.unit_cfg = unit_cfg
        self.length = length
        self.device = torch.device(device)
        self.bpn = vect_cfg.bytes_per_number
        # wide (u128-order) configs: canonical values are split lo/hi u64
        # planes; K1/K5 (mask synthesis) stay off-GPU for these
        self.wide = not vect_cfg.order_fits_u64
        self.n_digits = (self.bpn + 3) // 4
        self.order = vect_cfg.order  # decimal string
        self.order_int = int(vect_cfg.order)
        self.prng_nbytes = vect_cfg.prng_nbytes

        with torch.cuda.device(self.device):
            self.acc = torch.zeros(self.n_digits, length, dtype=torch.int64, device=self.device)
            self._expander = _hip.MaskExpander()
        self.nb_models = 0
        self.unit_acc = 0  # masked scalar sum (CPU, exact)

    # ---------------- update staging ----------------

    def alloc_update_pool(self, n: int) -> torch.Tensor:
        """uint8 [n, stride] rows of packed updates (stride padded/aligned)."""
        stride = self.row_stride()
        return torch.empty(n, stride, dtype=torch.uint8, device=self.device)

    def row_stride(self) -> int:
        # pad so the K3 tail reads stay in-row, and align rows to 16B
        raw = self.length * self.bpn + 16 * self.bpn  # EPT<=16 guard
        return (raw + 15) // 16 * 16

    def upload_update(self, pool: torch.Tensor, row: int, wire_limbs: bytes):
        t = torch.frombuffer(bytearray(wire_limbs), dtype=torch.uint8)
        # the source is pageable TEMPORARY memory: the copy must be
        # synchronous, or HIP may still be reading the bytearray after Python
        # frees it (observed as heap corruption in the staged-plane soak)
        pool[row, : t.numel()].copy_(t, non_blocking=False)

    # ---------------- mask expansion (K1) ----------------

    def derive_mask_values(self, seed: bytes, out: torch.Tensor | None = None) -> torch.Tensor:
        """Expand seed -> canonical mask values (vect part only; unit handled
        on CPU). u64 orders return [length] i64; wide orders a [2, length]
        lo/hi split. Bit-exact with _core.mask.derive_mask."""
        _, unit_words = _core.mask.unit_draw(seed, self.unit_cfg)
        if self.wide:
            if out is None:
                out = torch.empty(2, self.length, dtype=torch.int64, device=self.device)
            self._expander.expand_u128(
                seed, out[0].data_ptr(), out[1].data_ptr(), self.length, self.order,
                self.prng_nbytes, unit_words
            )
            return out
        if out is None:
            out = torch.empty(self.length, dtype=torch.int64, device=self.device)
        self._expander.expand(
            seed, out.data_ptr(), self.length, self.order, self.prng_nbytes, unit_words
        )
        return out

    def unit_draw(self, seed: bytes) -> int:
        v, _ = _core.mask.unit_draw(seed, self.unit_cfg)
        return int(v)

    # ---------------- aggregation (K3) ----------------

    def aggregate_pool(self, pool: torch.Tensor, n_updates: int, unit_sum: int = 0):
        """Accumulate n_updates packed rows into the digit planes."""
        _hip.aggregate_batch(
            self.acc.data_ptr(), pool.data_ptr(), pool.stride(0), n_updates, self.length,
            self.bpn, int(os.environ.get("XAYNET_K3_EPT", "0"))
        )
        self.nb_models += n_updates
        self.unit_acc = (self.unit_acc + unit_sum) % int(self.unit_cfg.order)

    # ---------------- finalize / unmask (K2, K4) ----------------

    def canonical(self, out: torch.Tensor | None = None) -> torch.Tensor:
        """Digit planes -> canonical values mod order. u64 configs return a
        [length] i64 tensor; wide configs a [2, length] lo/hi split."""
        if self.wide:
            if out is None:
                out = torch.empty(2, self.length, dtype=torch.int64, device=self.device)
            _hip.canonicalize_u128(self.acc.data_ptr(), out[0].data_ptr(), out[1].data_ptr(),
                                   self.length, self.n_digits, self.order)
            return out
        if out is None:
            out = torch.empty(self.length, dtype=torch.int64, device=self.device)
        _hip.canonicalize(self.acc.data_ptr(), out.data_ptr(), self.length, self.n_digits, self.order)
        return out

    def mod_add_values(self, a: torch.Tensor, b: torch.Tensor):
        """a = (a + b) mod order, canonical tensors (split lo/hi when wide)."""
        if self.wide:
            _hip.mod_add_u128(a[0].data_ptr(), a[1].data_ptr(), b[0].data_ptr(),
                              b[1].data_ptr(), self.length, self.order)
            return
        _hip.mod_add_u64(a.data_ptr(), b.data_ptr(), a.numel(), self.order)

    def add_values_to_planes(self, vals: torch.Tensor):
        _hip.add_u64_to_planes(self.acc.data_ptr(), vals.data_ptr(), self.length, self.n_digits)

    def values_to_planes(self, vals: torch.Tensor, planes: torch.Tensor):
        """Add canonical u64 values into an arbitrary [n_digits, n] plane tensor."""
        _hip.add_u64_to_planes(planes.data_ptr(), vals.data_ptr(), vals.numel(), self.n_digits)

    def planes_to_values(self, planes: torch.Tensor, out: torch.Tensor):
        """Canonicalize an arbitrary [n_digits, n] plane tensor mod order."""
        _hip.canonicalize(planes.data_ptr(), out.data_ptr(), out.numel(),
                          self.n_digits, self.order)

    _TORCH_DTYPES = {0: torch.float32, 1: torch.float64, 2: torch.int32, 3: torch.int64}

    def unmask(self, mask_values: torch.Tensor, mask_unit: int,
               nb_models: int | None = None, dtype: int | None = None) -> torch.Tensor:
        """Unmask the aggregate into model weights (reference unmask math,
        masking.rs:190-231). dtype follows mask::DataType (default: the vect
        config's data type); integers truncate toward zero."""
        nb = self.nb_models if nb_models is None else nb_models
        dt = self.vect_cfg.dtype if dtype is None else dtype
        info = _cfg_scalars(self.unit_cfg)
        # scalar_sum = n1/exp_1 - nb*add_1 (exact on CPU)
        n1 = (self.unit_acc + int(self.unit_cfg.order) - mask_unit) % int(self.unit_cfg.order)
        scalar_sum = n1 / info["exp_shift"] - nb * info["add_shift"]
        if scalar_sum == 0:
            raise ZeroDivisionError("scalar_sum is zero")
        vinfo = _cfg_scalars(self.vect_cfg)
        out = torch.empty(self.length, dtype=self._TORCH_DTYPES[dt], device=self.device)
        if self.wide:
            _hip.unmask_u128(
                self.acc.data_ptr(), mask_values[0].data_ptr(), mask_values[1].data_ptr(),
                out.data_ptr(), self.length, self.n_digits, self.order,
                str(vinfo["exp_shift_u64"]), nb * vinfo["add_shift"], scalar_sum, dt,
            )
            return out
        _hip.unmask(
            self.acc.data_ptr(), mask_values.data_ptr(), out.data_ptr(), self.length,
            self.n_digits, self.order, vinfo["exp_shift_u64"], nb * vinfo["add_shift"],
            scalar_sum, dt,
        )
        return out

    def unmask_f32(self, mask_values: torch.Tensor, mask_unit: int,
                   nb_models: int | None = None) -> torch.Tensor:
        return self.unmask(mask_values, mask_unit, nb_models=nb_models, dtype=0)

    def unmask_values(self, vals: torch.Tensor, mask_values: torch.Tensor, mask_unit: int,
                      nb_models: int, dtype: int | None = None) -> torch.Tensor:
        """Unmask a u64 tensor of (possibly cross-rank summed) canonical
        values — the reduce-scatter path. Valid while nb_ranks*order < 2^64."""
        if self.wide:
            raise NotImplementedError("values unmask covers u64 orders")
        n = vals.numel()
        dt = self.vect_cfg.dtype if dtype is None else dtype
        info = _cfg_scalars(self.unit_cfg)
        n1 = (self.unit_acc + int(self.unit_cfg.order) - mask_unit) % int(self.unit_cfg.order)
        scalar_sum = n1 / info["exp_shift"] - nb_models * info["add_shift"]
        if scalar_sum == 0:
            raise ZeroDivisionError("scalar_sum is zero")
        vinfo = _cfg_scalars(self.vect_cfg)
        out = torch.empty(n, dtype=self._TORCH_DTYPES[dt], device=vals.device)
        _hip.unmask_values(
            vals.data_ptr(), mask_values.data_ptr(), out.data_ptr(), n, self.order,
            vinfo["exp_shift_u64"], nb_models * vinfo["add_shift"], scalar_sum, dt,
        )
        return out

    def unmask_planes(self, planes: torch.Tensor, mask_values: torch.Tensor, mask_unit: int,
                      nb_models: int, dtype: int | None = None) -> torch.Tensor:
        """Unmask an arbitrary contiguous [n_digits, n] digit-plane tensor
        (e.g. this rank's reduce-scattered shard) against matching mask
        values. u64-order configs only."""
        if self.wide:
            raise NotImplementedError("shard unmask covers u64 orders")
        n = planes.shape[1]
        dt = self.vect_cfg.dtype if dtype is None else dtype
        info = _cfg_scalars(self.unit_cfg)
        n1 = (self.unit_acc + int(self.unit_cfg.order) - mask_unit) % int(self.unit_cfg.order)
        scalar_sum = n1 / info["exp_shift"] - nb_models * info["add_shift"]
        if scalar_sum == 0:
            raise ZeroDivisionError("scalar_sum is zero")
        vinfo = _cfg_scalars(self.vect_cfg)
        out = torch.empty(n, dtype=self._TORCH_DTYPES[dt], device=planes.device)
        _hip.unmask(
            planes.data_ptr(), mask_values.data_ptr(), out.data_ptr(), n,
            self.n_digits, self.order, vinfo["exp_shift_u64"],
            nb_models * vinfo["add_shift"], scalar_sum, dt,
        )
        return out

    # ---------------- synthetic updates (K5, bench/test-drive) ----------------

    def synth_update(self, pool: torch.Tensor, row: int, mask_values: torch.Tensor,
                     participant: int, scalar: float):
        vinfo = _cfg_scalars(self.vect_cfg)
        if self.wide:
            _hip.mask_pack_u128(
                mask_values[0].data_ptr(), mask_values[1].data_ptr(), pool[row].data_ptr(),
                self.length, self.bpn, self.order, participant, scalar, vinfo["add_shift"],
                vinfo["exp_shift"],
            )
            return
        _hip.mask_pack(
            mask_values.data_ptr(), pool[row].data_ptr(), self.length, self.bpn, self.order,
            participant, scalar, vinfo["add_shift"], vinfo["exp_shift"], vinfo["exp_shift_u64"],
        )

    def mask_weights(self, seed: bytes, weights: torch.Tensor,
                     scalar_num: int = 1, scalar_den: int = 1) -> bytes:
        """Mask a REAL weight tensor (the participant's update task): K1
        expand + K5w quantize+mask+pack; returns the full MaskObject wire
        bytes (MaskVect || MaskUnit). Replaces the CPU fast masker's hot
        loop (reference Masker::mask, masking.rs:358-404) for GPU clients.
        Quantization deviates from the exact-rational path by at most a few
        1/exp_shift quanta (double rounding before an exact mulshift)."""
        if weights.numel() != self.length:
            raise ValueError("weights length != model length")
        dt_map = {torch.float32: 0, torch.float64: 1, torch.int32: 2, torch.int64: 3}
        dt = dt_map[weights.dtype]
        w_dev = weights.to(self.device).contiguous()
        mask_vals = self.derive_mask_values(seed)
        vinfo = _cfg_scalars(self.vect_cfg)
        out = torch.empty(self.length * self.bpn, dtype=torch.uint8, device=self.device)
        if self.wide:
            lo, hi = mask_vals[0].data_ptr(), mask_vals[1].data_ptr()
        else:
            lo, hi = mask_vals.data_ptr(), mask_vals.data_ptr()
        _hip.mask_weights(
            w_dev.data_ptr(), dt, lo, hi, out.data_ptr(), self.length, self.bpn,
            self.order, scalar_num / scalar_den, vinfo["add_shift"],
            str(vinfo["exp_shift_u64"]), self.wide,
        )
        # masked unit (scalar clamp + quantize + unit mask), exact on CPU
        masked_unit = self.masked_unit_for(seed, scalar_num, scalar_den)
        # assemble the wire in ONE host buffer: a single D2H lands the limbs
        # directly in place (a 25M-param update is ~175 MB; intermediate
        # bytes/bytearray copies dominate otherwise)
        import numpy as np

        nlimb = self.length * self.bpn
        ubpn = self.unit_cfg.bytes_per_number
        wire = bytearray(8 + nlimb + 4 + ubpn)
        wire[0:4] = bytes(self.vect_cfg.to_bytes())
        wire[4:8] = self.length.to_bytes(4, "big")
        # stage through a cached PINNED buffer: a direct D2H into pageable
        # memory runs at ~2 GB/s in chunked staging copies (rocprof:
        # __amd_rocclr_copyBuffer dominated the wall time), pinned runs at
        # PCIe line rate and the host-side memcpy at memory speed
        if getattr(self, "_wire_pin", None) is None or self._wire_pin.numel() < nlimb:
            self._wire_pin = torch.empty(nlimb, dtype=torch.uint8, pin_memory=True)
        pin = self._wire_pin[:nlimb]
        pin.copy_(out)
        np.frombuffer(wire, dtype=np.uint8, count=nlimb, offset=8)[:] = pin.numpy()
        off = 8 + nlimb
        wire[off : off + 4] = bytes(self.unit_cfg.to_bytes())
        wire[off + 4 :] = int(masked_unit).to_bytes(ubpn, "little")
        return bytes(wire)

    def masked_unit_for(self, seed: bytes, scalar_num: int, scalar_den: int) -> int:
        """Masked scalar for a synthetic update (CPU, exact)."""
        info = _cfg_scalars(self.unit_cfg)
        unit_rand, _ = _core.mask.unit_draw(seed, self.unit_cfg)
        exp = int(info["exp_shift"])
        add = int(info["add_shift"])
        # trunc((clamp(scalar)+add)*exp): scalar = num/den <= add by protocol
        shifted = ((scalar_num + add * scalar_den) * exp) // scalar_den
        return (shifted + int(unit_rand)) % int(self.unit_cfg.order)

    def reset(self):
        self.acc.zero_()
        self.nb_models = 0
        self.unit_acc = 0

    def unpack_wire(self, packed: torch.Tensor, out: torch.Tensor | None = None) -> torch.Tensor:
        if self.wide:
            if out is None:
                out = torch.empty(2, self.length, dtype=torch.int64, device=self.device)
            _hip.unpack_u128(packed.data_ptr(), out[0].data_ptr(), out[1].data_ptr(),
                             self.length, self.bpn)
            return out
        if out is None:
            out = torch.empty(self.length, dtype=torch.int64, device=self.device)
        _hip.unpack_u64(packed.data_ptr(), out.data_ptr(), self.length, self.bpn)
        return out

    def pack_wire(self, values: torch.Tensor) -> torch.Tensor:
        out = torch.empty(self.length * self.bpn, dtype=torch.uint8, device=self.device)
        if self.wide:
            _hip.pack_u128(values[0].data_ptr(), values[1].data_ptr(), out.data_ptr(),
                           self.length, self.bpn)
            return out
        _hip.pack_u64(values.data_ptr(), out.data_ptr(), self.length, self.bpn)
        return out


def _cfg_scalars(cfg):
    """exp/add shifts as python scalars (u64-order configs only)."""
    exp_exp = {0: 10, 1: 20, 2: 10, 3: 10}[cfg.dtype]  # non-Bmax
    if cfg.bound == 255:
        raise ValueError("Bmax configs are not on the u64 GPU path")
    add = {0: 1.0, 2: 100.0, 4: 10_000.0, 6: 1_000_000.0}[cfg.bound]
    return {
        "exp_shift": float(10**exp_exp),
        "exp_shift_u64": 10**exp_exp,
        "add_shift": add,
    }
