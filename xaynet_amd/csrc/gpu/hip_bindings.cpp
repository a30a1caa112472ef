// pybind11 module `xaynet_amd._hip` — host orchestration for the MI355X
// kernels in kernels.hip. Compiled with hipcc (gfx950); no torch headers —
// callers pass raw device pointers (torch `tensor.data_ptr()`), and all
// launches go to the null stream, which serializes with PyTorch's default
// stream on the device.
#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <cmath>
#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

extern "C" {
hipError_t xhip_k1_candidates(const uint32_t*, uint64_t, uint64_t, uint64_t, int, int, uint64_t,
                              uint64_t*, uint8_t*, uint32_t*, int, uint32_t*);
hipError_t xhip_k1_scan(uint32_t*, uint32_t, uint64_t*);
hipError_t xhip_k1_scan_hier(uint32_t*, uint32_t, uint32_t*, uint64_t*);
hipError_t xhip_k1_scatter(const uint64_t*, const uint8_t*, const uint32_t*, uint64_t, int,
                           uint64_t, uint64_t*, uint64_t);
hipError_t xhip_k1_scatter_compact(const uint64_t*, const uint32_t*, const uint64_t*, uint32_t,
                                   int, uint64_t, uint64_t*, uint64_t);
int xhip_k1_use_reg(void);
hipError_t xhip_k1_expand_fused(const uint32_t*, uint64_t, uint64_t, uint64_t, int, int,
                                uint64_t, uint64_t, uint64_t*, uint64_t,
                                unsigned long long*, unsigned long long*, int, uint32_t*);
hipError_t xhip_k3_aggregate(uint64_t*, const uint8_t*, uint64_t, uint32_t, uint64_t, int, int);
hipError_t xhip_k4_unmask_f64(const uint64_t*, const uint64_t*, double*, uint64_t, int, uint64_t,
                              uint64_t, double, double);
hipError_t xhip_k4_unmask_i32(const uint64_t*, const uint64_t*, int32_t*, uint64_t, int, uint64_t,
                              uint64_t, double, double);
hipError_t xhip_k4_unmask_i64(const uint64_t*, const uint64_t*, int64_t*, uint64_t, int, uint64_t,
                              uint64_t, double, double);
hipError_t xhip_k4_unmask_f32(const uint64_t*, const uint64_t*, float*, uint64_t, int, uint64_t,
                              uint64_t, double, double);
hipError_t xhip_k2_canonicalize(const uint64_t*, uint64_t*, uint64_t, int, uint64_t);
hipError_t xhip_k2_mod_add_u64(uint64_t*, const uint64_t*, uint64_t, uint64_t);
hipError_t xhip_k5_mask_pack(const uint64_t*, uint8_t*, uint64_t, int, uint64_t, uint64_t, double,
                             double, double, uint64_t);
hipError_t xhip_k6_unpack_u64(const uint8_t*, uint64_t*, uint64_t, int);
hipError_t xhip_k6_pack_u64(const uint64_t*, uint8_t*, uint64_t, int);
hipError_t xhip_add_u64_to_planes(uint64_t*, const uint64_t*, uint64_t, int);
hipError_t xhip_k4_unmask_values_f32(const uint64_t*, const uint64_t*, float*, uint64_t,
                                     uint64_t, uint64_t, double, double);
hipError_t xhip_k4_unmask_values_f64(const uint64_t*, const uint64_t*, double*, uint64_t,
                                     uint64_t, uint64_t, double, double);
hipError_t xhip_k4_unmask_values_i32(const uint64_t*, const uint64_t*, int32_t*, uint64_t,
                                     uint64_t, uint64_t, double, double);
hipError_t xhip_k4_unmask_values_i64(const uint64_t*, const uint64_t*, int64_t*, uint64_t,
                                     uint64_t, uint64_t, double, double);
hipError_t xhip_k2_canonicalize_u128(const uint64_t*, uint64_t*, uint64_t*, uint64_t, int,
                                     uint64_t, uint64_t);
hipError_t xhip_k2_mod_add_u128(uint64_t*, uint64_t*, const uint64_t*, const uint64_t*, uint64_t,
                                uint64_t, uint64_t);
hipError_t xhip_k6_unpack_u128(const uint8_t*, uint64_t*, uint64_t*, uint64_t, int);
hipError_t xhip_k4_unmask_u128_f64(const uint64_t*, const uint64_t*, const uint64_t*, double*,
                                   uint64_t, int, uint64_t, uint64_t, uint64_t, uint64_t, double,
                                   double);
hipError_t xhip_k4_unmask_u128_f32(const uint64_t*, const uint64_t*, const uint64_t*, float*,
                                   uint64_t, int, uint64_t, uint64_t, uint64_t, uint64_t, double,
                                   double);
hipError_t xhip_k1_candidates_u128(const uint32_t*, uint64_t, uint64_t, uint64_t, int, int,
                                   uint64_t, uint64_t, uint64_t*, uint64_t*, uint32_t*, int,
                                   uint32_t*);
hipError_t xhip_k1_scatter_compact_u128(const uint64_t*, const uint64_t*, const uint32_t*,
                                        const uint64_t*, uint32_t, int, uint64_t, uint64_t*,
                                        uint64_t*, uint64_t);
hipError_t xhip_k5_mask_pack_u128(const uint64_t*, const uint64_t*, uint8_t*, uint64_t, int,
                                  uint64_t, uint64_t, uint64_t, double, double, double);
hipError_t xhip_k6_pack_u128(const uint64_t*, const uint64_t*, uint8_t*, uint64_t, int);
hipError_t xhip_k5_mask_weights(const void*, int, const uint64_t*, const uint64_t*, uint8_t*,
                                uint64_t, int, uint64_t, uint64_t, double, double, uint64_t,
                                uint64_t, int);
}

// decimal string -> u128 (orders/exp_shifts wider than u64)
static unsigned __int128 parse_u128(const std::string& dec) {
    unsigned __int128 v = 0;
    for (char c : dec) {
        if (c < '0' || c > '9') throw std::runtime_error("bad decimal");
        v = v * 10 + unsigned(c - '0');
    }
    return v;
}

static void check(hipError_t e, const char* what) {
    if (e != hipSuccess) {
        throw std::runtime_error(std::string(what) + ": " + hipGetErrorString(e));
    }
}

// ------------------------------------------------------------- MaskExpander
//
// Owns the rejection-sampling workspace. expand() reproduces the reference
// ChaCha20 draw stream exactly (see kernels.hip header comment): the caller
// provides start_word = keystream words consumed by the preceding unit draw
// (computed on CPU via _core.mask.unit_draw).
class MaskExpander {
  public:
    MaskExpander() = default;
    ~MaskExpander() { release(); }

    void release() {
        if (cand_) hipFree(cand_);
        if (accept_) hipFree(accept_);
        if (counts_) hipFree(counts_);
        if (state_) hipFree(state_);
        state_ = nullptr;
        if (total_dev_) hipFree(total_dev_);
        if (key_dev_) hipFree(key_dev_);
        cand_ = nullptr; accept_ = nullptr; counts_ = nullptr;
        total_dev_ = nullptr; key_dev_ = nullptr;
        cap_attempts_ = 0;
        if (scan_tmp_) hipFree(scan_tmp_);
        scan_tmp_ = nullptr;
        if (cand_lo_) hipFree(cand_lo_);
        if (cand_hi_) hipFree(cand_hi_);
        if (counts_w_) hipFree(counts_w_);
        if (scan_tmp_w_) hipFree(scan_tmp_w_);
        cand_lo_ = nullptr; cand_hi_ = nullptr; counts_w_ = nullptr; scan_tmp_w_ = nullptr;
        cap_attempts_w_ = 0;
    }

    void ensure_static() {
        if (!key_dev_) check(hipMalloc(&key_dev_, 32), "alloc key");
        if (!total_dev_) check(hipMalloc(&total_dev_, 8), "alloc total");
    }

    void reserve(uint64_t attempts) {
        if (attempts <= cap_attempts_) return;
        if (cand_) hipFree(cand_);
        if (accept_) hipFree(accept_);
        if (counts_) hipFree(counts_);
        if (state_) hipFree(state_);
        uint32_t max_wgs = uint32_t((attempts + 256 * 8 - 1) / (256 * 8)) + 2;
        if (fused()) {
            // single-pass path: only the per-block lookback state
            cand_ = nullptr;
            accept_ = nullptr;
            counts_ = nullptr;
            check(hipMalloc(&state_, sizeof(unsigned long long) * max_wgs), "alloc state");
        } else {
            check(hipMalloc(&cand_, attempts * 8), "alloc cand");
            // the compact default never touches accept[] (candidates writes
            // accepted draws in-order per wg segment) — only the
            // XAYNET_K1_REG=1 fallback needs the per-attempt flag array
            accept_ = nullptr;
            if (xhip_k1_use_reg()) check(hipMalloc(&accept_, attempts), "alloc accept");
            check(hipMalloc(&counts_, sizeof(uint32_t) * max_wgs), "alloc counts");
            if (scan_tmp_) hipFree(scan_tmp_);
            check(hipMalloc(&scan_tmp_, sizeof(uint32_t) * (max_wgs / 1024 + 2)),
                  "alloc scan tmp");
            state_ = nullptr;
        }
        cap_attempts_ = attempts;
    }

    static bool fused() {
        // default OFF: measured on MI355X, the single-pass lookback variant
        // loses to the 3-pass pipeline (3.8 vs 1.4 ms per 25M-element mask) —
        // the grid-wide inclusive-prefix propagation chain costs more than
        // the candidate buffer round-trip it saves (profiles/r01_k1_fused.md)
        const char* e = getenv("XAYNET_K1_FUSED");
        return e && e[0] == '1';
    }

    // Returns number of draw ATTEMPTS consumed (for stream-position tracking).
    uint64_t expand(const std::string& s, uintptr_t out_ptr, uint64_t len,
                    const std::string& order_dec, int prng_nbytes, uint64_t start_word) {
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        if (prng_nbytes > 8) throw std::runtime_error("expand: order > 2^64 not on GPU yet");
        uint64_t order = std::stoull(order_dec);

        int wpd = (prng_nbytes + 3) / 4;          // 1 or 2 words per attempt
        int dpt = 16 / wpd;                       // draws per thread (16 or 8)
        double p = double(order) * std::pow(2.0, -8.0 * prng_nbytes);  // acceptance
        uint64_t* out = reinterpret_cast<uint64_t*>(out_ptr);

        ensure_static();
        // env-toggled mode switch invalidates the workspace layout
        int f = fused() ? 1 : 0;
        if (f != last_mode_) {
            if (cand_) hipFree(cand_);
            if (accept_) hipFree(accept_);
            if (counts_) hipFree(counts_);
            if (state_) hipFree(state_);
            cand_ = nullptr; accept_ = nullptr; counts_ = nullptr; state_ = nullptr;
            cap_attempts_ = 0;
            last_mode_ = f;
        }
        check(hipMemcpy(key_dev_, s.data(), 32, hipMemcpyHostToDevice), "seed H2D");

        uint64_t filled = 0, attempt = 0;
        while (filled < len) {
            uint64_t remaining = len - filled;
            double exp_att = double(remaining) / p;
            uint64_t n_att = uint64_t(exp_att + 6.0 * std::sqrt(exp_att / p) + 1024.0);
            reserve(n_att);
            if (n_att > cap_attempts_) n_att = cap_attempts_;

            uint32_t n_wgs = 0;
            if (fused()) {
                check(xhip_k1_expand_fused(key_dev_, start_word, attempt, n_att, wpd,
                                           prng_nbytes, order, filled, out, len, state_,
                                           reinterpret_cast<unsigned long long*>(total_dev_),
                                           dpt, &n_wgs),
                      "k1_expand_fused");
            } else {
                check(xhip_k1_candidates(key_dev_, start_word, attempt, n_att, wpd, prng_nbytes,
                                         order, cand_, accept_, counts_, dpt, &n_wgs),
                      "k1_candidates");
                check(xhip_k1_scan_hier(counts_, n_wgs, scan_tmp_, total_dev_), "k1_scan");
                if (xhip_k1_use_reg())
                    check(xhip_k1_scatter(cand_, accept_, counts_, n_att, dpt, filled, out, len),
                          "k1_scatter");
                else
                    check(xhip_k1_scatter_compact(cand_, counts_, total_dev_, n_wgs, dpt,
                                                  filled, out, len),
                          "k1_scatter_compact");
            }
            uint64_t round_accepted = 0;
            check(hipMemcpy(&round_accepted, total_dev_, 8, hipMemcpyDeviceToHost), "total D2H");
            filled += round_accepted;  // may overshoot len; clamped below
            if (filled > len) filled = len;
            attempt += n_att;
            if (round_accepted == 0 && n_att > 0 && p <= 0.0) {
                throw std::runtime_error("expand: zero acceptance");
            }
        }
        // attempts consumed up to the len-th acceptance is NOT simply
        // `attempt` (the launch overshoots). The caller that needs the exact
        // stream position after `len` draws should use cpu-side accounting;
        // masks are always derived from a fresh seed so the tail position is
        // unused in the protocol.
        return attempt;
    }

    // Wide orders (2^64 < order <= 2^128): split lo/hi u64 candidate planes,
    // same candidates -> scan -> compact-scatter pipeline. Returns attempts
    // consumed (same caveat as expand()).
    uint64_t expand_u128(const std::string& s, uintptr_t out_lo_ptr, uintptr_t out_hi_ptr,
                         uint64_t len, const std::string& order_dec, int prng_nbytes,
                         uint64_t start_word) {
        if (s.size() != 32) throw std::runtime_error("seed must be 32 bytes");
        if (prng_nbytes <= 8 || prng_nbytes > 16)
            throw std::runtime_error("expand_u128: nbytes must be 9..16");
        unsigned __int128 order = parse_u128(order_dec);
        uint64_t order_lo = uint64_t(order);
        uint64_t order_hi = uint64_t(order >> 64);

        int wpd = (prng_nbytes + 3) / 4;  // 3 or 4
        int apt = 16 / wpd;               // attempts per thread: 5 or 4
        double p = (double(order_hi) * 18446744073709551616.0 + double(order_lo)) *
                   std::pow(2.0, -8.0 * prng_nbytes);
        uint64_t* out_lo = reinterpret_cast<uint64_t*>(out_lo_ptr);
        uint64_t* out_hi = reinterpret_cast<uint64_t*>(out_hi_ptr);

        ensure_static();
        check(hipMemcpy(key_dev_, s.data(), 32, hipMemcpyHostToDevice), "seed H2D");

        uint64_t filled = 0, attempt = 0;
        while (filled < len) {
            uint64_t remaining = len - filled;
            double exp_att = double(remaining) / p;
            uint64_t n_att = uint64_t(exp_att + 6.0 * std::sqrt(exp_att / p) + 1024.0);
            reserve_u128(n_att);
            if (n_att > cap_attempts_w_) n_att = cap_attempts_w_;

            uint32_t n_wgs = 0;
            check(xhip_k1_candidates_u128(key_dev_, start_word, attempt, n_att, wpd,
                                          prng_nbytes, order_lo, order_hi, cand_lo_, cand_hi_,
                                          counts_w_, apt, &n_wgs),
                  "k1_candidates_u128");
            check(xhip_k1_scan_hier(counts_w_, n_wgs, scan_tmp_w_, total_dev_), "k1_scan");
            check(xhip_k1_scatter_compact_u128(cand_lo_, cand_hi_, counts_w_, total_dev_, n_wgs,
                                               apt, filled, out_lo, out_hi, len),
                  "k1_scatter_compact_u128");
            uint64_t round_accepted = 0;
            check(hipMemcpy(&round_accepted, total_dev_, 8, hipMemcpyDeviceToHost), "total D2H");
            filled += round_accepted;
            if (filled > len) filled = len;
            attempt += n_att;
            if (round_accepted == 0 && n_att > 0 && p <= 0.0)
                throw std::runtime_error("expand_u128: zero acceptance");
        }
        return attempt;
    }

    void reserve_u128(uint64_t attempts) {
        if (attempts <= cap_attempts_w_) return;
        if (cand_lo_) hipFree(cand_lo_);
        if (cand_hi_) hipFree(cand_hi_);
        if (counts_w_) hipFree(counts_w_);
        uint32_t max_wgs = uint32_t((attempts + 256 * 4 - 1) / (256 * 4)) + 2;
        check(hipMalloc(&cand_lo_, attempts * 8), "alloc cand_lo");
        check(hipMalloc(&cand_hi_, attempts * 8), "alloc cand_hi");
        check(hipMalloc(&counts_w_, sizeof(uint32_t) * max_wgs), "alloc counts_w");
        if (scan_tmp_w_) hipFree(scan_tmp_w_);
        check(hipMalloc(&scan_tmp_w_, sizeof(uint32_t) * (max_wgs / 1024 + 2)),
              "alloc scan tmp w");
        cap_attempts_w_ = attempts;
    }

  private:
    uint64_t* cand_ = nullptr;
    uint8_t* accept_ = nullptr;
    uint32_t* counts_ = nullptr;
    uint32_t* scan_tmp_ = nullptr;
    unsigned long long* state_ = nullptr;
    int last_mode_ = -1;
    uint64_t* total_dev_ = nullptr;
    uint32_t* key_dev_ = nullptr;
    uint64_t cap_attempts_ = 0;
    // wide-path workspace
    uint64_t* cand_lo_ = nullptr;
    uint64_t* cand_hi_ = nullptr;
    uint32_t* counts_w_ = nullptr;
    uint32_t* scan_tmp_w_ = nullptr;
    uint64_t cap_attempts_w_ = 0;
};

PYBIND11_MODULE(_hip, m) {
    m.doc() = "xaynet_amd MI355X (gfx950) kernels: mask expand, aggregate, unmask";

    m.def("device_count", []() {
        int n = 0;
        hipError_t e = hipGetDeviceCount(&n);
        return e == hipSuccess ? n : 0;
    });
    m.def("synchronize", []() { check(hipDeviceSynchronize(), "sync"); });

    // Pin an existing host range (e.g. a POSIX shared-memory mapping) so
    // H2D copies from it can be asynchronous DMA — the serve plane's ingest
    // ring is a shm region shared with the coordinator process.
    m.def("host_register", [](uintptr_t ptr, size_t size) {
        check(hipHostRegister(reinterpret_cast<void*>(ptr), size, hipHostRegisterDefault),
              "hipHostRegister");
    });
    m.def("host_unregister", [](uintptr_t ptr) {
        check(hipHostUnregister(reinterpret_cast<void*>(ptr)), "hipHostUnregister");
    });

    py::class_<MaskExpander>(m, "MaskExpander")
        .def(py::init<>())
        .def("expand", &MaskExpander::expand, py::arg("seed"), py::arg("out_ptr"), py::arg("len"),
             py::arg("order"), py::arg("prng_nbytes"), py::arg("start_word"),
             py::call_guard<py::gil_scoped_release>())
        .def("expand_u128", &MaskExpander::expand_u128, py::arg("seed"), py::arg("out_lo_ptr"),
             py::arg("out_hi_ptr"), py::arg("len"), py::arg("order"), py::arg("prng_nbytes"),
             py::arg("start_word"), py::call_guard<py::gil_scoped_release>());

    m.def(
        "aggregate_batch",
        [](uintptr_t acc, uintptr_t updates, uint64_t stride, uint32_t n_updates, uint64_t len,
           int bpn, int ept) {
            check(xhip_k3_aggregate(reinterpret_cast<uint64_t*>(acc),
                                    reinterpret_cast<const uint8_t*>(updates), stride, n_updates,
                                    len, bpn, ept),
                  "k3_aggregate");
        },
        py::arg("acc"), py::arg("updates"), py::arg("stride"), py::arg("n_updates"),
        py::arg("len"), py::arg("bpn"), py::arg("ept") = 0,
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "unmask",
        [](uintptr_t acc, uintptr_t mask, uintptr_t out, uint64_t len, int n_digits,
           const std::string& order_dec, uint64_t exp_shift, double n_add_shift,
           double scalar_sum, int dtype) {
            const auto* a = reinterpret_cast<const uint64_t*>(acc);
            const auto* mk = reinterpret_cast<const uint64_t*>(mask);
            uint64_t ord = std::stoull(order_dec);
            hipError_t e;
            switch (dtype) {  // mask::DataType: 0=F32 1=F64 2=I32 3=I64
                case 0:
                    e = xhip_k4_unmask_f32(a, mk, reinterpret_cast<float*>(out), len, n_digits,
                                           ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                case 1:
                    e = xhip_k4_unmask_f64(a, mk, reinterpret_cast<double*>(out), len, n_digits,
                                           ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                case 2:
                    e = xhip_k4_unmask_i32(a, mk, reinterpret_cast<int32_t*>(out), len, n_digits,
                                           ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                case 3:
                    e = xhip_k4_unmask_i64(a, mk, reinterpret_cast<int64_t*>(out), len, n_digits,
                                           ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                default:
                    throw std::runtime_error("bad dtype");
            }
            check(e, "k4_unmask");
        },
        py::arg("acc"), py::arg("mask"), py::arg("out"), py::arg("len"), py::arg("n_digits"),
        py::arg("order"), py::arg("exp_shift"), py::arg("n_add_shift"),
        py::arg("scalar_sum"), py::arg("dtype") = 0,
        py::call_guard<py::gil_scoped_release>());
    // backwards-compatible alias
    m.def(
        "unmask_f32",
        [](uintptr_t acc, uintptr_t mask, uintptr_t out, uint64_t len, int n_digits,
           const std::string& order_dec, uint64_t exp_shift, double n_add_shift,
           double scalar_sum) {
            check(xhip_k4_unmask_f32(reinterpret_cast<const uint64_t*>(acc),
                                     reinterpret_cast<const uint64_t*>(mask),
                                     reinterpret_cast<float*>(out), len, n_digits,
                                     std::stoull(order_dec), exp_shift, n_add_shift,
                                     scalar_sum),
                  "k4_unmask");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "unmask_values",
        [](uintptr_t vals, uintptr_t mask, uintptr_t out, uint64_t len,
           const std::string& order_dec, uint64_t exp_shift, double n_add_shift,
           double scalar_sum, int dtype) {
            const auto* v = reinterpret_cast<const uint64_t*>(vals);
            const auto* mk = reinterpret_cast<const uint64_t*>(mask);
            uint64_t ord = std::stoull(order_dec);
            hipError_t e;
            switch (dtype) {
                case 0:
                    e = xhip_k4_unmask_values_f32(v, mk, reinterpret_cast<float*>(out), len, ord,
                                                  exp_shift, n_add_shift, scalar_sum);
                    break;
                case 1:
                    e = xhip_k4_unmask_values_f64(v, mk, reinterpret_cast<double*>(out), len,
                                                  ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                case 2:
                    e = xhip_k4_unmask_values_i32(v, mk, reinterpret_cast<int32_t*>(out), len,
                                                  ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                case 3:
                    e = xhip_k4_unmask_values_i64(v, mk, reinterpret_cast<int64_t*>(out), len,
                                                  ord, exp_shift, n_add_shift, scalar_sum);
                    break;
                default:
                    throw std::runtime_error("bad dtype");
            }
            check(e, "k4_unmask_values");
        },
        py::call_guard<py::gil_scoped_release>());

    // ---- u128-order variants (bpn 9..16: F64 families, narrow Bmax) ----
    m.def(
        "canonicalize_u128",
        [](uintptr_t acc, uintptr_t out_lo, uintptr_t out_hi, uint64_t len, int n_digits,
           const std::string& order_dec) {
            auto o = parse_u128(order_dec);
            check(xhip_k2_canonicalize_u128(reinterpret_cast<const uint64_t*>(acc),
                                            reinterpret_cast<uint64_t*>(out_lo),
                                            reinterpret_cast<uint64_t*>(out_hi), len, n_digits,
                                            uint64_t(o), uint64_t(o >> 64)),
                  "k2_canonicalize_u128");
        },
        py::call_guard<py::gil_scoped_release>());
    m.def(
        "mod_add_u128",
        [](uintptr_t a_lo, uintptr_t a_hi, uintptr_t b_lo, uintptr_t b_hi, uint64_t len,
           const std::string& order_dec) {
            auto o = parse_u128(order_dec);
            check(xhip_k2_mod_add_u128(reinterpret_cast<uint64_t*>(a_lo),
                                       reinterpret_cast<uint64_t*>(a_hi),
                                       reinterpret_cast<const uint64_t*>(b_lo),
                                       reinterpret_cast<const uint64_t*>(b_hi), len, uint64_t(o),
                                       uint64_t(o >> 64)),
                  "k2_mod_add_u128");
        },
        py::call_guard<py::gil_scoped_release>());
    m.def(
        "unpack_u128",
        [](uintptr_t in, uintptr_t out_lo, uintptr_t out_hi, uint64_t len, int bpn) {
            check(xhip_k6_unpack_u128(reinterpret_cast<const uint8_t*>(in),
                                      reinterpret_cast<uint64_t*>(out_lo),
                                      reinterpret_cast<uint64_t*>(out_hi), len, bpn),
                  "k6_unpack_u128");
        },
        py::call_guard<py::gil_scoped_release>());
    m.def(
        "unmask_u128",
        [](uintptr_t acc, uintptr_t mask_lo, uintptr_t mask_hi, uintptr_t out, uint64_t len,
           int n_digits, const std::string& order_dec, const std::string& exp_dec,
           double n_add_shift, double scalar_sum, int dtype) {
            auto o = parse_u128(order_dec);
            auto e = parse_u128(exp_dec);
            hipError_t rc;
            if (dtype == 1)
                rc = xhip_k4_unmask_u128_f64(
                    reinterpret_cast<const uint64_t*>(acc),
                    reinterpret_cast<const uint64_t*>(mask_lo),
                    reinterpret_cast<const uint64_t*>(mask_hi), reinterpret_cast<double*>(out),
                    len, n_digits, uint64_t(o), uint64_t(o >> 64), uint64_t(e),
                    uint64_t(e >> 64), n_add_shift, scalar_sum);
            else if (dtype == 0)
                rc = xhip_k4_unmask_u128_f32(
                    reinterpret_cast<const uint64_t*>(acc),
                    reinterpret_cast<const uint64_t*>(mask_lo),
                    reinterpret_cast<const uint64_t*>(mask_hi), reinterpret_cast<float*>(out),
                    len, n_digits, uint64_t(o), uint64_t(o >> 64), uint64_t(e),
                    uint64_t(e >> 64), n_add_shift, scalar_sum);
            else
                throw std::runtime_error("u128 unmask supports f32/f64 outputs");
            check(rc, "k4_unmask_u128");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "canonicalize",
        [](uintptr_t acc, uintptr_t out, uint64_t len, int n_digits, const std::string& order) {
            check(xhip_k2_canonicalize(reinterpret_cast<const uint64_t*>(acc),
                                       reinterpret_cast<uint64_t*>(out), len, n_digits,
                                       std::stoull(order)),
                  "k2_canonicalize");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "mod_add_u64",
        [](uintptr_t a, uintptr_t b, uint64_t len, const std::string& order) {
            check(xhip_k2_mod_add_u64(reinterpret_cast<uint64_t*>(a),
                                      reinterpret_cast<const uint64_t*>(b), len,
                                      std::stoull(order)),
                  "k2_mod_add");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "mask_pack",
        [](uintptr_t mask, uintptr_t out, uint64_t len, int bpn, const std::string& order,
           uint64_t participant, double scalar, double add_shift, double exp_shift_d,
           uint64_t exp_shift) {
            check(xhip_k5_mask_pack(reinterpret_cast<const uint64_t*>(mask),
                                    reinterpret_cast<uint8_t*>(out), len, bpn,
                                    std::stoull(order), participant, scalar, add_shift,
                                    exp_shift_d, exp_shift),
                  "k5_mask_pack");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "mask_pack_u128",
        [](uintptr_t mask_lo, uintptr_t mask_hi, uintptr_t out, uint64_t len, int bpn,
           const std::string& order, uint64_t participant, double scalar, double add_shift,
           double exp_shift_d) {
            unsigned __int128 o = parse_u128(order);
            check(xhip_k5_mask_pack_u128(reinterpret_cast<const uint64_t*>(mask_lo),
                                         reinterpret_cast<const uint64_t*>(mask_hi),
                                         reinterpret_cast<uint8_t*>(out), len, bpn, uint64_t(o),
                                         uint64_t(o >> 64), participant, scalar, add_shift,
                                         exp_shift_d),
                  "k5_mask_pack_u128");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "mask_weights",
        [](uintptr_t w, int dtype, uintptr_t mask_lo, uintptr_t mask_hi, uintptr_t out,
           uint64_t len, int bpn, const std::string& order, double scalar, double add_shift,
           const std::string& exp_shift, bool wide) {
            unsigned __int128 o = parse_u128(order);
            unsigned __int128 e = parse_u128(exp_shift);
            check(xhip_k5_mask_weights(reinterpret_cast<const void*>(w), dtype,
                                       reinterpret_cast<const uint64_t*>(mask_lo),
                                       reinterpret_cast<const uint64_t*>(mask_hi),
                                       reinterpret_cast<uint8_t*>(out), len, bpn, uint64_t(o),
                                       uint64_t(o >> 64), scalar, add_shift, uint64_t(e),
                                       uint64_t(e >> 64), wide ? 1 : 0),
                  "k5_mask_weights");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "pack_u128",
        [](uintptr_t in_lo, uintptr_t in_hi, uintptr_t out, uint64_t len, int bpn) {
            check(xhip_k6_pack_u128(reinterpret_cast<const uint64_t*>(in_lo),
                                    reinterpret_cast<const uint64_t*>(in_hi),
                                    reinterpret_cast<uint8_t*>(out), len, bpn),
                  "k6_pack_u128");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "unpack_u64",
        [](uintptr_t in, uintptr_t out, uint64_t len, int bpn) {
            check(xhip_k6_unpack_u64(reinterpret_cast<const uint8_t*>(in),
                                     reinterpret_cast<uint64_t*>(out), len, bpn),
                  "k6_unpack");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "pack_u64",
        [](uintptr_t in, uintptr_t out, uint64_t len, int bpn) {
            check(xhip_k6_pack_u64(reinterpret_cast<const uint64_t*>(in),
                                   reinterpret_cast<uint8_t*>(out), len, bpn),
                  "k6_pack");
        },
        py::call_guard<py::gil_scoped_release>());

    m.def(
        "add_u64_to_planes",
        [](uintptr_t acc, uintptr_t vals, uint64_t len, int n_digits) {
            check(xhip_add_u64_to_planes(reinterpret_cast<uint64_t*>(acc),
                                         reinterpret_cast<const uint64_t*>(vals), len, n_digits),
                  "add_u64_to_planes");
        },
        py::call_guard<py::gil_scoped_release>());
}
