#include "bincode.h"

#include <atomic>
#include <cmath>
#include <thread>

namespace xaynet::bincode {

using mask::MaskConfig;

void write_mask_config(Writer& w, const MaskConfig& c) {
    // serde C-like enums -> u32 LE variant INDEX (not the repr value!)
    auto bound_idx = [](mask::BoundType b) -> uint32_t {
        switch (b) {
            case mask::BoundType::B0: return 0;
            case mask::BoundType::B2: return 1;
            case mask::BoundType::B4: return 2;
            case mask::BoundType::B6: return 3;
            case mask::BoundType::Bmax: return 4;
        }
        return 0;
    };
    auto model_idx = [](mask::ModelType m) -> uint32_t {
        switch (m) {
            case mask::ModelType::M3: return 0;
            case mask::ModelType::M6: return 1;
            case mask::ModelType::M9: return 2;
            case mask::ModelType::M12: return 3;
        }
        return 0;
    };
    w.u32(uint32_t(c.group));
    w.u32(uint32_t(c.dtype));
    w.u32(bound_idx(c.bound));
    w.u32(model_idx(c.model));
}

bool read_mask_config(Reader& r, MaskConfig& c) {
    uint32_t g = r.u32(), d = r.u32(), b = r.u32(), m = r.u32();
    if (r.fail || g > 2 || d > 3 || b > 4 || m > 3) return false;
    static const mask::BoundType bounds[5] = {mask::BoundType::B0, mask::BoundType::B2,
                                              mask::BoundType::B4, mask::BoundType::B6,
                                              mask::BoundType::Bmax};
    static const mask::ModelType models[4] = {mask::ModelType::M3, mask::ModelType::M6,
                                              mask::ModelType::M9, mask::ModelType::M12};
    c.group = mask::GroupType(g);
    c.dtype = mask::DataType(d);
    c.bound = bounds[b];
    c.model = models[m];
    return true;
}

Bytes encode_round_parameters(const RoundParameters& rp) {
    Writer w;
    w.raw(rp.pk.data(), 32);
    w.f64(rp.sum);
    w.f64(rp.update);
    w.raw(rp.seed.data(), 32);
    write_mask_config(w, rp.mask_config.vect);
    write_mask_config(w, rp.mask_config.unit);
    w.u64(rp.model_length);
    return std::move(w.out);
}

std::optional<RoundParameters> decode_round_parameters(const uint8_t* p, size_t len,
                                                       size_t* consumed) {
    Reader r{p, len};
    RoundParameters rp;
    r.raw(rp.pk.data(), 32);
    rp.sum = r.f64();
    rp.update = r.f64();
    r.raw(rp.seed.data(), 32);
    if (!read_mask_config(r, rp.mask_config.vect)) return std::nullopt;
    if (!read_mask_config(r, rp.mask_config.unit)) return std::nullopt;
    rp.model_length = r.u64();
    if (r.fail) return std::nullopt;
    if (consumed) *consumed = r.off;
    return rp;
}

std::optional<RoundParameters> decode_round_parameters(const uint8_t* p, size_t len) {
    return decode_round_parameters(p, len, nullptr);
}

Bytes encode_option_sum_dict(const SumDict* d) {
    Writer w;
    if (!d) {
        w.u8(0);
        return std::move(w.out);
    }
    w.u8(1);
    w.u64(d->size());
    for (const auto& [k, v] : *d) {
        w.raw(k.data(), 32);
        w.raw(v.data(), 32);
    }
    return std::move(w.out);
}

std::optional<std::optional<SumDict>> decode_option_sum_dict(const uint8_t* p, size_t len) {
    Reader r{p, len};
    uint8_t has = r.u8();
    if (r.fail) return std::nullopt;
    if (!has) return std::optional<SumDict>();
    uint64_t n = r.u64();
    SumDict d;
    for (uint64_t i = 0; i < n; ++i) {
        msg::Key32 k, v;
        if (!r.raw(k.data(), 32) || !r.raw(v.data(), 32)) return std::nullopt;
        d.emplace(k, v);
    }
    return std::optional<SumDict>(std::move(d));
}

Bytes encode_option_update_seed_dict(const UpdateSeedDict* d) {
    Writer w;
    if (!d) {
        w.u8(0);
        return std::move(w.out);
    }
    w.u8(1);
    w.u64(d->size());
    for (const auto& [k, v] : *d) {
        w.raw(k.data(), 32);
        w.bytes_vec(v.data(), 80);  // EncryptedMaskSeed is Vec<u8>
    }
    return std::move(w.out);
}

std::optional<std::optional<UpdateSeedDict>> decode_option_update_seed_dict(const uint8_t* p,
                                                                            size_t len) {
    Reader r{p, len};
    uint8_t has = r.u8();
    if (r.fail) return std::nullopt;
    if (!has) return std::optional<UpdateSeedDict>();
    uint64_t n = r.u64();
    UpdateSeedDict d;
    for (uint64_t i = 0; i < n; ++i) {
        msg::Key32 k;
        if (!r.raw(k.data(), 32)) return std::nullopt;
        uint64_t sl = r.u64();
        if (sl != 80) return std::nullopt;
        msg::EncrSeed80 s;
        if (!r.raw(s.data(), 80)) return std::nullopt;
        d.emplace(k, s);
    }
    return std::optional<UpdateSeedDict>(std::move(d));
}

// BigUint -> Vec<u32> LE digits (num-bigint serde layout)
void write_biguint(Writer& w, const BigUint& v) {
    std::vector<uint32_t> digits;
    for (uint64_t limb : v.d) {
        digits.push_back(uint32_t(limb));
        digits.push_back(uint32_t(limb >> 32));
    }
    while (!digits.empty() && digits.back() == 0) digits.pop_back();
    w.u64(digits.size());
    for (uint32_t dg : digits) w.u32(dg);
}

bool read_biguint(Reader& r, BigUint& v) {
    uint64_t n = r.u64();
    if (r.fail || n > (1ull << 24)) return false;
    v.d.clear();
    v.d.resize((n + 1) / 2, 0);
    for (uint64_t i = 0; i < n; ++i) {
        uint32_t dg = r.u32();
        v.d[i / 2] |= uint64_t(dg) << (32 * (i % 2));
    }
    v.normalize();
    return !r.fail;
}

// BigInt -> (Sign{Minus=0,NoSign=1,Plus=2}, BigUint)
static void write_bigint(Writer& w, const BigInt& v) {
    w.u32(v.is_zero() ? 1u : (v.neg ? 0u : 2u));
    write_biguint(w, v.mag);
}

static bool read_bigint(Reader& r, BigInt& v) {
    uint32_t sign = r.u32();
    if (r.fail || sign > 2) return false;
    if (!read_biguint(r, v.mag)) return false;
    v.neg = (sign == 0) && !v.mag.is_zero();
    return true;
}

Bytes encode_option_model(const RationalModel* m) {
    Writer w;
    if (!m) {
        w.u8(0);
        return std::move(w.out);
    }
    w.u8(1);
    w.u64(m->size());
    for (const auto& rat : *m) {
        write_bigint(w, rat.numer);
        write_bigint(w, BigInt(rat.denom, false));
    }
    return std::move(w.out);
}

// ---- typed fast paths for Option<Model> ------------------------------
//
// Model elements sourced from primitives are dyadic rationals
// (±odd_mantissa / 2^k after reduction); emitting their bincode directly
// skips the per-element BigInt gcd (measured ~16 us/element -> ~40
// ns/element). Byte-identical to encode_option_model(model_from_*()).

static void write_dyadic(Writer& w, bool neg, uint64_t mant, int exp2) {
    // value = ±mant * 2^exp2, mant odd (or zero)
    if (mant == 0) {
        w.u32(1);  // NoSign
        w.u64(0);  // numer: no digits
        w.u32(2);  // Plus
        w.u64(1);
        w.u32(1);  // denom = 1
        return;
    }
    w.u32(neg ? 0u : 2u);
    if (exp2 >= 0) {
        // numer = mant << exp2, denom = 1
        int total_bits = 64 - __builtin_clzll(mant) + exp2;
        uint32_t ndig = uint32_t((total_bits + 31) / 32);
        w.u64(ndig);
        int word_shift = exp2 / 32, bit_shift = exp2 % 32;
        // mant occupies up to 3 u32 digits after shifting
        uint64_t lo = bit_shift ? (mant << bit_shift) : mant;
        uint64_t hi = bit_shift ? (mant >> (64 - bit_shift)) : 0;
        for (uint32_t i = 0; i < ndig; ++i) {
            int rel = int(i) - word_shift;
            uint32_t dg = 0;
            if (rel >= 0 && rel < 2) dg = uint32_t(lo >> (32 * rel));
            else if (rel == 2) dg = uint32_t(hi);
            w.u32(dg);
        }
        w.u32(2);
        w.u64(1);
        w.u32(1);
    } else {
        // numer = mant, denom = 2^(-exp2)
        uint32_t ndig = mant >> 32 ? 2 : 1;
        w.u64(ndig);
        w.u32(uint32_t(mant));
        if (ndig == 2) w.u32(uint32_t(mant >> 32));
        int k = -exp2;
        uint32_t ddig = uint32_t(k / 32) + 1;
        w.u32(2);
        w.u64(ddig);
        for (uint32_t i = 0; i + 1 < ddig; ++i) w.u32(0);
        w.u32(1u << (k % 32));
    }
}

static void write_double_elem(Writer& w, double v) {
    if (v == 0.0 || !std::isfinite(v)) {  // non-finite cannot occur post-validation
        write_dyadic(w, false, 0, 0);
        return;
    }
    int e;
    double m = std::frexp(std::fabs(v), &e);
    uint64_t mi = uint64_t(std::ldexp(m, 53));
    int exp2 = e - 53;
    int tz = __builtin_ctzll(mi);
    write_dyadic(w, std::signbit(v), mi >> tz, exp2 + tz);
}

static void encode_range_f32(Writer& w, const float* v, size_t b, size_t e) {
    for (size_t i = b; i < e; ++i) write_double_elem(w, double(v[i]));
}

static void encode_range_f64(Writer& w, const double* v, size_t b, size_t e) {
    for (size_t i = b; i < e; ++i) write_double_elem(w, v[i]);
}

static void encode_range_i64(Writer& w, const int64_t* v, size_t b, size_t e) {
    for (size_t i = b; i < e; ++i) {
        uint64_t mag = v[i] < 0 ? uint64_t(-(v[i] + 1)) + 1 : uint64_t(v[i]);
        if (mag == 0) {
            write_dyadic(w, false, 0, 0);
        } else {
            int tz = __builtin_ctzll(mag);
            write_dyadic(w, v[i] < 0, mag >> tz, tz);
        }
    }
}

static void encode_range_i32(Writer& w, const int32_t* v, size_t b, size_t e) {
    for (size_t i = b; i < e; ++i) {
        int64_t x = v[i];
        uint64_t mag = x < 0 ? uint64_t(-x) : uint64_t(x);
        if (mag == 0) {
            write_dyadic(w, false, 0, 0);
        } else {
            int tz = __builtin_ctzll(mag);
            write_dyadic(w, x < 0, mag >> tz, tz);
        }
    }
}

Bytes encode_option_model_f32(const float* v, size_t n) {
    Writer w;
    w.out.reserve(10 + n * 30);
    w.u8(1);
    w.u64(n);
    encode_range_f32(w, v, 0, n);
    return std::move(w.out);
}

Bytes encode_option_model_f64(const double* v, size_t n) {
    Writer w;
    w.out.reserve(10 + n * 34);
    w.u8(1);
    w.u64(n);
    encode_range_f64(w, v, 0, n);
    return std::move(w.out);
}

Bytes encode_option_model_i64(const int64_t* v, size_t n) {
    Writer w;
    w.out.reserve(10 + n * 30);
    w.u8(1);
    w.u64(n);
    encode_range_i64(w, v, 0, n);
    return std::move(w.out);
}

Bytes encode_option_model_i32(const int32_t* v, size_t n) {
    Writer w;
    w.out.reserve(10 + n * 26);
    w.u8(1);
    w.u64(n);
    encode_range_i32(w, v, 0, n);
    return std::move(w.out);
}

// ---- multi-thread chunked encode ---------------------------------------
// A 25M-element f32 model is an ~800 MB bincode body; the serial encoder
// costs seconds and sits on the serve plane's unmask tail and on every
// model PUT. Elements encode independently, so chunks are encoded on
// worker threads and concatenated — byte-identical to the serial path.

void EncodedModel::assemble(uint8_t* dst) const {
    std::vector<std::pair<const Bytes*, size_t>> segs;
    size_t off = head.size();
    std::memcpy(dst, head.data(), head.size());
    for (const auto& p : parts) {
        segs.emplace_back(&p, off);
        off += p.size();
    }
    std::vector<std::thread> th;
    for (auto& [p, o] : segs)
        th.emplace_back([dst, p = p, o = o] { std::memcpy(dst + o, p->data(), p->size()); });
    for (auto& t : th) t.join();
}

template <typename T, typename RangeFn>
static EncodedModel encode_model_mt(const T* v, size_t n, size_t est, RangeFn range) {
    EncodedModel out;
    Writer head;
    head.u8(1);
    head.u64(n);
    out.head = std::move(head.out);
    unsigned hw = std::thread::hardware_concurrency();
    unsigned T_ = n >= (size_t(1) << 19) ? std::min(16u, hw ? hw : 1u) : 1u;
    size_t chunk = (n + T_ - 1) / T_;
    out.parts.resize(T_);
    if (T_ == 1) {  // small model: no thread spawn
        Writer w;
        w.out.reserve(n * est);
        range(w, v, 0, n);
        out.parts[0] = std::move(w.out);
        return out;
    }
    std::vector<std::thread> th;
    std::atomic<bool> failed{false};  // an uncaught throw in std::thread terminates
    for (unsigned t = 0; t < T_; ++t) {
        th.emplace_back([&, t] {
            try {
                size_t b = size_t(t) * chunk, e = std::min(n, b + chunk);
                if (b >= e) return;
                Writer w;
                w.out.reserve((e - b) * est);
                range(w, v, b, e);
                out.parts[t] = std::move(w.out);
            } catch (...) {
                failed.store(true, std::memory_order_relaxed);
            }
        });
    }
    for (auto& x : th) x.join();
    if (failed.load()) throw std::bad_alloc();
    return out;
}

EncodedModel encode_option_model_mt_f32(const float* v, size_t n) {
    return encode_model_mt(v, n, 30, encode_range_f32);
}
EncodedModel encode_option_model_mt_f64(const double* v, size_t n) {
    return encode_model_mt(v, n, 34, encode_range_f64);
}
EncodedModel encode_option_model_mt_i32(const int32_t* v, size_t n) {
    return encode_model_mt(v, n, 26, encode_range_i32);
}
EncodedModel encode_option_model_mt_i64(const int64_t* v, size_t n) {
    return encode_model_mt(v, n, 30, encode_range_i64);
}

// fast f32/f64 decode: succeeds when every element is ±small/2^k (the
// shape every primitive-sourced model has); returns false -> caller uses
// the generic rational path
// decode `count` dyadic elements starting at byte `off0`; writes out[0..count).
// end_off receives the byte offset after the last element.
template <typename OUT>
static bool decode_range_dyadic(const uint8_t* p, size_t len, size_t off0, uint64_t count,
                                OUT* out, size_t* end_off) {
    Reader r{p, len};
    r.off = off0;
    for (uint64_t i = 0; i < count; ++i) {
        uint32_t nsign = r.u32();
        uint64_t nd = r.u64();
        if (r.fail || nsign > 2 || nd > (1u << 16)) return false;
        // numerator = (top 3 digits) * 2^(32*(nd-3)); primitive-sourced
        // values have <= 53 significant bits, so lower digits must be zero
        uint32_t d0 = 0, d1 = 0, d2 = 0;  // three highest digits (d2 top)
        for (uint64_t j = 0; j < nd; ++j) {
            uint32_t dg = r.u32();
            if (nd - j > 3) {
                if (dg != 0) return false;  // significant bits too wide
            }
            d0 = d1;
            d1 = d2;
            d2 = dg;
        }
        // reorder: after the loop d2 = last (highest) digit
        double nval = double(d2) * 4294967296.0 * 4294967296.0 +
                      double(d1) * 4294967296.0 + double(d0);
        int nexp = nd > 3 ? int(32 * (nd - 3)) : 0;
        if (nd == 1) { nval = double(d2); }
        else if (nd == 2) { nval = double(d2) * 4294967296.0 + double(d1); }
        double mant_d = nval;
        uint32_t dsign = r.u32();
        uint64_t dd = r.u64();
        if (r.fail || dsign != 2 || dd == 0 || dd > 64) return false;
        uint32_t dtop = 0;
        for (uint64_t j = 0; j < dd; ++j) {
            uint32_t dg = r.u32();
            if (j + 1 < dd && dg != 0) return false;  // denom must be a power of two
            if (j + 1 == dd) dtop = dg;
        }
        if (r.fail) return false;
        if ((dtop & (dtop - 1)) != 0 || dtop == 0) return false;
        int k = int((dd - 1) * 32) + __builtin_ctz(dtop);
        double value = std::ldexp(mant_d, nexp - k);
        out[i] = OUT(nsign == 0 ? -value : value);
    }
    *end_off = r.off;
    return !r.fail;
}

// length-only walk over `n` elements recording a chunk-start offset every
// `chunk` elements; validates the same structural bounds as the decoder
static bool scan_dyadic_boundaries(const uint8_t* p, size_t len, size_t head, uint64_t n,
                                   size_t chunk,
                                   std::vector<std::pair<size_t, uint64_t>>& starts) {
    const uint8_t* base = p;
    const uint8_t* q = p + head;
    const uint8_t* lim = p + len;
    for (uint64_t i = 0; i < n; ++i) {
        if (i % chunk == 0) starts.emplace_back(size_t(q - base), i);
        if (lim - q < 12) return false;
        uint64_t nd = load64_le(q + 4);  // numer: sign u32 | len u64 | digits
        if (nd > (1u << 16)) return false;
        q += 12 + 4 * nd;
        if (lim - q < 12) return false;
        uint64_t dd = load64_le(q + 4);  // denom: sign u32 | len u64 | digits
        if (dd == 0 || dd > 64) return false;
        q += 12 + 4 * dd;
        if (q > lim) return false;
    }
    return q == lim;
}

template <typename OUT>
static bool decode_model_dyadic(const uint8_t* p, size_t len, std::vector<OUT>& out) {
    Reader r{p, len};
    if (r.u8() != 1) return false;
    uint64_t n = r.u64();
    if (r.fail || n > (1ull << 32)) return false;
    size_t head = r.off;
    out.clear();
    unsigned hw = std::thread::hardware_concurrency();
    if (n >= (1u << 20) && hw > 1) {
        // multi-thread: serial length-only scan finds chunk boundaries,
        // then the chunks decode concurrently (an 800 MB 25M-element body
        // took ~2.3 s serially; participants decode it on every fetch)
        unsigned T = std::min(16u, hw);
        size_t chunk = (n + T - 1) / T;
        std::vector<std::pair<size_t, uint64_t>> starts;
        if (!scan_dyadic_boundaries(p, len, head, n, chunk, starts)) return false;
        out.resize(n);
        std::atomic<bool> ok{true};
        std::vector<std::thread> th;
        for (size_t s = 0; s < starts.size(); ++s) {
            th.emplace_back([&, s] {
                auto [off0, i0] = starts[s];
                uint64_t count = std::min<uint64_t>(chunk, n - i0);
                size_t end = 0;
                size_t expect = s + 1 < starts.size() ? starts[s + 1].first : len;
                if (!decode_range_dyadic(p, len, off0, count, out.data() + i0, &end) ||
                    end != expect)
                    ok.store(false, std::memory_order_relaxed);
            });
        }
        for (auto& t : th) t.join();
        return ok.load();
    }
    out.resize(n);
    size_t end = 0;
    if (!decode_range_dyadic(p, len, head, n, out.data(), &end)) return false;
    return end == len;
}

bool decode_option_model_f32_fast(const uint8_t* p, size_t len, std::vector<float>& out) {
    return decode_model_dyadic<float>(p, len, out);
}
bool decode_option_model_f64_fast(const uint8_t* p, size_t len, std::vector<double>& out) {
    return decode_model_dyadic<double>(p, len, out);
}

std::optional<std::optional<RationalModel>> decode_option_model(const uint8_t* p, size_t len) {
    Reader r{p, len};
    uint8_t has = r.u8();
    if (r.fail) return std::nullopt;
    if (!has) return std::optional<RationalModel>();
    uint64_t n = r.u64();
    if (r.fail || n > (1ull << 32)) return std::nullopt;
    RationalModel m;
    m.reserve(n);
    for (uint64_t i = 0; i < n; ++i) {
        BigInt numer, denom;
        if (!read_bigint(r, numer) || !read_bigint(r, denom)) return std::nullopt;
        if (denom.neg || denom.mag.is_zero()) return std::nullopt;
        Rational rat;
        rat.numer = std::move(numer);  // already reduced on the wire
        rat.denom = std::move(denom.mag);
        m.push_back(std::move(rat));
    }
    return std::optional<RationalModel>(std::move(m));
}

Bytes encode_seed_dict(const SeedDict& d) {
    Writer w;
    w.u64(d.size());
    for (const auto& [k, inner] : d) {
        w.raw(k.data(), 32);
        w.u64(inner.size());
        for (const auto& [uk, s] : inner) {
            w.raw(uk.data(), 32);
            w.bytes_vec(s.data(), 80);
        }
    }
    return std::move(w.out);
}

std::optional<SeedDict> decode_seed_dict(const uint8_t* p, size_t len) {
    Reader r{p, len};
    uint64_t n = r.u64();
    SeedDict d;
    for (uint64_t i = 0; i < n && !r.fail; ++i) {
        msg::Key32 k;
        if (!r.raw(k.data(), 32)) return std::nullopt;
        uint64_t m = r.u64();
        UpdateSeedDict inner;
        for (uint64_t j = 0; j < m; ++j) {
            msg::Key32 uk;
            if (!r.raw(uk.data(), 32)) return std::nullopt;
            if (r.u64() != 80) return std::nullopt;
            msg::EncrSeed80 s;
            if (!r.raw(s.data(), 80)) return std::nullopt;
            inner.emplace(uk, s);
        }
        d.emplace(k, std::move(inner));
    }
    if (r.fail) return std::nullopt;
    return d;
}

}  // namespace xaynet::bincode
