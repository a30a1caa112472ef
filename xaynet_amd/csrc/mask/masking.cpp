#include "masking.h"

#include <cmath>

namespace xaynet::mask {

// ------------------------------------------------------------------- PRNG

BigUint MaskPrng::generate_integer(const CfgInfo& ci) {
    if (ci.order.is_zero()) return BigUint();
    if (ci.prng_nbytes <= 8) return BigUint(generate_u64(ci));
    Bytes buf(ci.prng_nbytes);
    while (true) {
        rng_.fill_bytes(buf.data(), buf.size());
        BigUint v = BigUint::from_bytes_le(buf.data(), buf.size());
        if (v < ci.order) return v;
    }
}

uint64_t MaskPrng::generate_u64(const CfgInfo& ci) {
    const int nb = int(ci.prng_nbytes);
    while (true) {
        uint64_t v = rng_.draw_u64(nb);
        if (v < ci.order_u64) return v;
    }
}

unsigned __int128 MaskPrng::generate_u128(const CfgInfo& ci, unsigned __int128 order) {
    uint8_t buf[16] = {0};
    while (true) {
        rng_.fill_bytes(buf, ci.prng_nbytes);
        unsigned __int128 v = 0;
        for (size_t i = 16; i-- > 0;) v = (v << 8) | buf[i];
        if (v < order) return v;
    }
}

// ------------------------------------------------------------ derive_mask

MaskObject derive_mask(const uint8_t seed[32], size_t len, const MaskConfigPair& cfg) {
    const auto& ci_1 = cfg.unit.info();
    const auto& ci_n = cfg.vect.info();
    MaskPrng prng(seed);

    MaskObject out = MaskObject::zeros(cfg, len);
    out.unit.set_value(prng.generate_integer(ci_1));

    if (ci_n.prng_nbytes <= 8) {
        uint8_t* p = out.vect.data.data();
        size_t bpn = ci_n.bpn;
        for (size_t i = 0; i < len; ++i, p += bpn) {
            uint64_t v = prng.generate_u64(ci_n);
            for (size_t b = 0; b < bpn; ++b) p[b] = uint8_t(v >> (8 * b));
        }
    } else {
        for (size_t i = 0; i < len; ++i) out.vect.set_element(i, prng.generate_integer(ci_n));
    }
    return out;
}

// -------------------------------------------------------------- mask_model

static BigUint shift_quantize(const Rational& clamped, const Rational& add_shift,
                              const BigUint& exp_shift) {
    // trunc((clamped + add_shift) * exp_shift); non-negative by construction
    Rational shifted = (clamped + add_shift) * Rational::from_integer(BigInt(exp_shift, false));
    BigInt t = shifted.trunc();
    return t.mag;  // non-negative
}

// ---- fast exact quantizer (u64-order configs) -------------------------
//
// The reference masker quantizes with exact rationals; done naively that is
// ~20 us/weight (BigInt allocations + one big division per element) and is
// the reference's own client-side bottleneck (ROADMAP "optimising the
// client"). This path computes the SAME exact value with fixed-width
// integer arithmetic (<=256-bit intermediates, shift-subtract division) and
// falls back to the rational oracle for any element near a clamp/rounding
// boundary or outside the width guards — bit-equality is asserted in
// tests/test_masking.py::test_fast_masker_matches_oracle.

namespace {

struct U256 {
    uint64_t w[4] = {0, 0, 0, 0};

    static U256 from_u128(unsigned __int128 v) {
        U256 r;
        r.w[0] = uint64_t(v);
        r.w[1] = uint64_t(v >> 64);
        return r;
    }
    bool add(const U256& o) {  // returns carry (overflow)
        unsigned __int128 c = 0;
        for (int i = 0; i < 4; ++i) {
            unsigned __int128 s = (unsigned __int128)w[i] + o.w[i] + c;
            w[i] = uint64_t(s);
            c = s >> 64;
        }
        return c != 0;
    }
    bool sub(const U256& o) {  // this -= o; returns borrow (this < o)
        unsigned __int128 b = 0;
        for (int i = 0; i < 4; ++i) {
            unsigned __int128 d = (unsigned __int128)w[i] - o.w[i] - b;
            w[i] = uint64_t(d);
            b = (d >> 64) ? 1 : 0;
        }
        return b != 0;
    }
    int cmp(const U256& o) const {
        for (int i = 3; i >= 0; --i) {
            if (w[i] != o.w[i]) return w[i] < o.w[i] ? -1 : 1;
        }
        return 0;
    }
    bool mul_u64(uint64_t m) {  // this *= m; returns overflow
        unsigned __int128 c = 0;
        for (int i = 0; i < 4; ++i) {
            unsigned __int128 p = (unsigned __int128)w[i] * m + c;
            w[i] = uint64_t(p);
            c = p >> 64;
        }
        return c != 0;
    }
    int bits() const {
        for (int i = 3; i >= 0; --i)
            if (w[i]) return 64 * i + 64 - __builtin_clzll(w[i]);
        return 0;
    }
    void shl1() {
        for (int i = 3; i > 0; --i) w[i] = (w[i] << 1) | (w[i - 1] >> 63);
        w[0] <<= 1;
    }
    bool is_zero() const { return !(w[0] | w[1] | w[2] | w[3]); }
};

// q = n / d (shift-subtract); requires q < 2^128, d > 0
static unsigned __int128 u256_div_u256(const U256& n, const U256& d, bool& fits) {
    int shift = n.bits() - d.bits();
    if (shift < 0) {
        fits = true;
        return 0;
    }
    if (shift > 127) {  // quotient cannot fit u128
        fits = false;
        return 0;
    }
    U256 rem = n;
    U256 ds = d;
    for (int i = 0; i < shift; ++i) ds.shl1();
    unsigned __int128 q = 0;
    for (int i = shift; i >= 0; --i) {
        q <<= 1;
        if (rem.cmp(ds) >= 0) {
            rem.sub(ds);
            q |= 1;
        }
        // shift divisor right by one
        for (int j = 0; j < 3; ++j) ds.w[j] = (ds.w[j] >> 1) | (ds.w[j + 1] << 63);
        ds.w[3] >>= 1;
    }
    fits = true;
    return q;
}

struct FastMaskCtx {
    bool usable = false;
    unsigned __int128 order = 0;       // group order (u64 AND u128 families)
    uint64_t a = 0;                    // integer clamp bound (add_shift)
    unsigned __int128 E = 0;           // exp_shift (<= 10^20)
    unsigned __int128 aE = 0;          // a * E (< order by the order rule)
    uint64_t snum = 0, sden = 1;       // clamped scalar (num/den)
    double approx_scalar = 0, approx_E = 0;
};

static bool load_u128_le(const BigUint& v, unsigned __int128& out) {
    if (v.bits() > 128) return false;
    Bytes b = v.to_bytes_le();
    out = 0;
    for (size_t i = b.size(); i-- > 0;) out = (out << 8) | b[i];
    return true;
}

static bool cfg_fast_ctx(const CfgInfo& ci, uint64_t snum, uint64_t sden, FastMaskCtx& c) {
    // integer add_shift <= 2^32 (B0..B6; Bmax uses dtype maxima -> fallback)
    Rational a_r = ci.add_shift;
    BigInt a_t = a_r.trunc();
    if (a_t.neg) return false;
    if (!(Rational::cmp(a_r, Rational::from_integer(a_t)) == 0)) return false;
    if (a_t.mag.bits() > 32) return false;
    if (ci.exp_shift.bits() > 127) return false;
    c.a = a_t.mag.low_u64();
    c.E = 0;
    {
        // exp_shift as u128 (LE bytes)
        Bytes eb = ci.exp_shift.to_bytes_le();
        if (eb.size() > 16) return false;
        for (size_t i = eb.size(); i-- > 0;) c.E = (c.E << 8) | eb[i];
    }
    if (!load_u128_le(ci.order, c.order)) return false;  // u64 and u128 families
    unsigned __int128 aE = (unsigned __int128)c.a * c.E;
    if (aE >= c.order || c.order - aE <= aE) return false;  // need 2aE < order
    c.aE = aE;
    // snum/sden: the scalar ALREADY clamped against the unit config's bound
    // (reference masking.rs: scalar clamp precedes the vect loop)
    if (sden == 0) return false;
    c.snum = snum;
    c.sden = sden;
    c.approx_scalar = double(snum) / double(sden);
    c.approx_E = double(c.E);
    c.usable = true;
    return true;
}

// Exact trunc((clamp(scalar*w) + a) * E) for a finite double w.
// Returns false when the element must take the rational fallback.
static bool fast_quantize(double w, const FastMaskCtx& c, unsigned __int128& out) {
    if (!std::isfinite(w)) return false;
    if (w == 0.0) {
        out = c.aE;
        return true;
    }
    bool neg = std::signbit(w);
    double aw = std::fabs(w);
    double sc = aw * c.approx_scalar;  // approximate |scaled|
    if (c.snum == 0) {
        out = c.aE;
        return true;
    }
    // clamp decision with a safety margin; boundary -> exact fallback
    double ad = double(c.a);
    if (sc > ad * (1.0 - 1e-9)) {
        if (sc < ad * (1.0 + 1e-9)) return false;  // too close to the bound
        out = neg ? (unsigned __int128)0 : 2 * c.aE;  // clamped to -a / +a exactly
        return true;
    }
    // tiny values: |scaled|*E < 0.5 -> aE (positive) / aE-1 (negative)
    double scE = sc * c.approx_E;
    if (scE < 0.25) {
        out = neg ? c.aE - 1 : c.aE;
        return true;
    }
    if (scE < 1.0) return false;  // rounding boundary region

    // exact dyadic path: |w| = mant * 2^(e-53), mant in [2^52, 2^53)
    int e;
    double m = std::frexp(aw, &e);
    uint64_t mant = uint64_t(std::ldexp(m, 53));
    int K = 53 - e;  // |w| = mant / 2^K (K may be negative for |w| >= 2^53)

    // scaled = s / D with s = snum*mant*2^max(0,-K), D = sden*2^max(0,K)
    U256 S = U256::from_u128((unsigned __int128)c.snum * mant);
    U256 D = U256::from_u128(c.sden);
    if (K >= 0) {
        if ((64 - __builtin_clzll(c.sden | 1)) + K > 120) return false;
        for (int i = 0; i < K; ++i) D.shl1();
    } else {
        if (S.bits() - K > 200) return false;
        for (int i = 0; i < -K; ++i) S.shl1();
    }
    // N = a*D +/- s; always >= 0 when the clamp margin test passed
    U256 N = D;
    if (N.mul_u64(c.a)) return false;
    if (neg) {
        if (N.cmp(S) < 0) return false;  // boundary missed by the margin test
        N.sub(S);
    } else {
        if (N.add(S)) return false;
    }
    // P = N * E; E is 10^k — multiply in u64-sized factors
    U256 P = N;
    unsigned __int128 erem = c.E;
    while (erem > 1) {
        uint64_t f = erem > (unsigned __int128)10000000000ULL ? 10000000000ULL : uint64_t(erem);
        if (P.mul_u64(f)) return false;
        erem /= f;
    }
    unsigned __int128 q;
    if ((c.sden & (c.sden - 1)) == 0) {
        // power-of-two denominator (scalar 1/2^s, float-derived fractions):
        // the division is a plain shift
        int dshift = (K > 0 ? K : 0) + __builtin_ctzll(c.sden);
        U256 Q = P;
        int words = dshift / 64, rem = dshift % 64;
        if (words) {
            for (int i = 0; i + words < 4; ++i) Q.w[i] = Q.w[i + words];
            for (int i = 4 - words; i < 4; ++i) Q.w[i] = 0;
        }
        if (rem) {
            for (int i = 0; i < 3; ++i) Q.w[i] = (Q.w[i] >> rem) | (Q.w[i + 1] << (64 - rem));
            Q.w[3] >>= rem;
        }
        if (Q.w[2] | Q.w[3]) return false;
        q = ((unsigned __int128)Q.w[1] << 64) | Q.w[0];
    } else {
        bool fits;
        q = u256_div_u256(P, D, fits);
        if (!fits) return false;
    }
    out = q;
    return true;
}

}  // namespace

MaskObject mask_model(const uint8_t seed[32], const Scalar& scalar, const RationalModel& model,
                      const MaskConfigPair& cfg) {
    const auto& ci_1 = cfg.unit.info();
    const auto& ci_n = cfg.vect.info();
    MaskPrng prng(seed);

    MaskObject out = MaskObject::zeros(cfg, model.size());

    // clamp the scalar to [0, add_shift_1] (clamp_max; scalars are >= 0)
    Rational scalar_r = scalar.to_rational();
    if (Rational::cmp(scalar_r, ci_1.add_shift) > 0) scalar_r = ci_1.add_shift;

    // unit draw FIRST (stream order contract)
    BigUint rand_1 = prng.generate_integer(ci_1);

    Rational lo = Rational() - ci_n.add_shift;
    const Rational& hi = ci_n.add_shift;

    for (size_t i = 0; i < model.size(); ++i) {
        Rational scaled = scalar_r * model[i];
        if (Rational::cmp(scaled, lo) < 0) scaled = lo;
        else if (Rational::cmp(scaled, hi) > 0) scaled = hi;
        BigUint shifted = shift_quantize(scaled, ci_n.add_shift, ci_n.exp_shift);
        BigUint rand_n = prng.generate_integer(ci_n);
        out.vect.set_element(i, (shifted + rand_n) % ci_n.order);
    }

    BigUint shifted_1 = shift_quantize(scalar_r, ci_1.add_shift, ci_1.exp_shift);
    out.unit.set_value((shifted_1 + rand_1) % ci_1.order);
    return out;
}

// typed fast masker: exact fast path per element, rational oracle fallback
// for boundary/width cases, whole-call fallback for non-u64 configs.
template <typename T, typename ToRational>
static MaskObject mask_typed(const uint8_t seed[32], const Scalar& scalar, const T* w, size_t n,
                             const MaskConfigPair& cfg, ToRational to_rational,
                             MaskObject (*slow)(const uint8_t*, const Scalar&, const T*, size_t,
                                                const MaskConfigPair&)) {
    const auto& ci_1 = cfg.unit.info();
    const auto& ci_n = cfg.vect.info();

    // clamp the scalar against the unit bound (exact, once)
    Rational scalar_r = scalar.to_rational();
    if (scalar.numer.bits() > 63 || scalar.denom.bits() > 63)
        return slow(seed, scalar, w, n, cfg);
    uint64_t snum = scalar.numer.low_u64(), sden = scalar.denom.low_u64();
    if (Rational::cmp(scalar_r, ci_1.add_shift) > 0) {
        scalar_r = ci_1.add_shift;
        BigInt a1 = scalar_r.trunc();
        if (a1.neg || Rational::cmp(scalar_r, Rational::from_integer(a1)) != 0 ||
            a1.mag.bits() > 63)
            return slow(seed, scalar, w, n, cfg);
        snum = a1.mag.low_u64();
        sden = 1;
    }

    FastMaskCtx fc;
    if (!cfg_fast_ctx(ci_n, snum, sden, fc) || ci_n.prng_nbytes > 16)
        return slow(seed, scalar, w, n, cfg);

    MaskPrng prng(seed);
    MaskObject out = MaskObject::zeros(cfg, n);
    BigUint rand_1 = prng.generate_integer(ci_1);  // unit draw FIRST (stream order)

    const unsigned __int128 order = fc.order;
    const bool wide = ci_n.prng_nbytes > 8;
    const size_t bpn = ci_n.bpn;
    uint8_t* data = out.vect.data.data();
    Rational lo = Rational() - ci_n.add_shift;
    const Rational& hi = ci_n.add_shift;
    for (size_t i = 0; i < n; ++i) {
        unsigned __int128 rand_n =
            wide ? prng.generate_u128(ci_n, order) : (unsigned __int128)prng.generate_u64(ci_n);
        unsigned __int128 q;
        // i64 magnitudes beyond 2^53 are not exactly representable as double
        bool exact_as_double = true;
        if constexpr (std::is_same_v<T, int64_t>) {
            exact_as_double = w[i] > -(int64_t(1) << 53) && w[i] < (int64_t(1) << 53);
        }
        if (!exact_as_double || !fast_quantize(double(w[i]), fc, q)) {
            // exact rational fallback for this element only
            Rational scaled = scalar_r * to_rational(w[i]);
            if (Rational::cmp(scaled, lo) < 0) scaled = lo;
            else if (Rational::cmp(scaled, hi) > 0) scaled = hi;
            BigUint sq = shift_quantize(scaled, ci_n.add_shift, ci_n.exp_shift);
            unsigned __int128 qq = 0;
            if (!load_u128_le(sq, qq)) return slow(seed, scalar, w, n, cfg);
            q = qq;
        }
        // (q + rand) mod order; sum < 2*order <= 2^129, wrap-correct subtract
        unsigned __int128 v = q + rand_n;
        if (v < q || v >= order) v -= order;
        for (size_t b = 0; b < bpn; ++b) data[i * bpn + b] = uint8_t(v >> (8 * b));
    }

    BigUint shifted_1 = shift_quantize(scalar_r, ci_1.add_shift, ci_1.exp_shift);
    out.unit.set_value((shifted_1 + rand_1) % ci_1.order);
    return out;
}

static MaskObject mask_f32_slow(const uint8_t seed[32], const Scalar& scalar, const float* w,
                                size_t n, const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_f32(w, n), cfg);
}
static MaskObject mask_f64_slow(const uint8_t seed[32], const Scalar& scalar, const double* w,
                                size_t n, const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_f64(w, n), cfg);
}
static MaskObject mask_i32_slow(const uint8_t seed[32], const Scalar& scalar, const int32_t* w,
                                size_t n, const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_i32(w, n), cfg);
}
static MaskObject mask_i64_slow(const uint8_t seed[32], const Scalar& scalar, const int64_t* w,
                                size_t n, const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_i64(w, n), cfg);
}

MaskObject mask_f32(const uint8_t seed[32], const Scalar& scalar, const float* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_typed(seed, scalar, w, n, cfg,
                      [](float v) { return model_from_f32(&v, 1)[0]; }, mask_f32_slow);
}
MaskObject mask_f64(const uint8_t seed[32], const Scalar& scalar, const double* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_typed(seed, scalar, w, n, cfg,
                      [](double v) { return model_from_f64(&v, 1)[0]; }, mask_f64_slow);
}
MaskObject mask_i32(const uint8_t seed[32], const Scalar& scalar, const int32_t* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_typed(seed, scalar, w, n, cfg,
                      [](int32_t v) { return model_from_i32(&v, 1)[0]; }, mask_i32_slow);
}
MaskObject mask_i64(const uint8_t seed[32], const Scalar& scalar, const int64_t* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_typed(seed, scalar, w, n, cfg,
                      [](int64_t v) { return model_from_i64(&v, 1)[0]; }, mask_i64_slow);
}

// ------------------------------------------------------------- Aggregation

AggregationError Aggregation::validate_aggregation(const MaskObject& obj) const {
    if (object_.vect.cfg != obj.vect.cfg) return AggregationError::ModelMismatch;
    if (object_.unit.cfg != obj.unit.cfg) return AggregationError::ScalarMismatch;
    if (object_size_ != obj.vect.count) return AggregationError::ModelMismatch;
    if (nb_models_ >= object_.vect.cfg.info().max_nb_models) return AggregationError::TooManyModels;
    if (nb_models_ >= object_.unit.cfg.info().max_nb_models)
        return AggregationError::TooManyScalars;
    if (!obj.is_valid()) return AggregationError::InvalidObject;
    return AggregationError::Ok;
}

AggregationError Aggregation::validate_counts_only() const {
    if (nb_models_ >= object_.vect.cfg.info().max_nb_models) return AggregationError::TooManyModels;
    if (nb_models_ >= object_.unit.cfg.info().max_nb_models)
        return AggregationError::TooManyScalars;
    return AggregationError::Ok;
}

void Aggregation::aggregate(const MaskObject& obj) {
    if (nb_models_ == 0) {
        object_ = obj;
        nb_models_ = 1;
        return;
    }
    const auto& ci_n = object_.vect.cfg.info();
    if (ci_n.order_fits_u64) {
        uint8_t* a = object_.vect.data.data();
        const uint8_t* b = obj.vect.data.data();
        size_t bpn = ci_n.bpn;
        uint64_t order = ci_n.order_u64;
        for (size_t i = 0; i < object_size_; ++i) {
            uint64_t x = 0, y = 0;
            for (size_t k = 0; k < bpn; ++k) {
                x |= uint64_t(a[i * bpn + k]) << (8 * k);
                y |= uint64_t(b[i * bpn + k]) << (8 * k);
            }
            // both < order <= 2^63 (fits: order bytes <= 8); sum may exceed 64
            // bits only when order > 2^63 — use 128-bit to be safe
            unsigned __int128 s = (unsigned __int128)x + y;
            uint64_t r = uint64_t(s >= order ? s - order : s);
            for (size_t k = 0; k < bpn; ++k) a[i * bpn + k] = uint8_t(r >> (8 * k));
        }
    } else {
        const BigUint& order = ci_n.order;
        for (size_t i = 0; i < object_size_; ++i) {
            BigUint s = object_.vect.element(i) + obj.vect.element(i);
            if (s >= order) s = s - order;
            object_.vect.set_element(i, s);
        }
    }
    const auto& ci_1 = object_.unit.cfg.info();
    BigUint s1 = object_.unit.value() + obj.unit.value();
    if (s1 >= ci_1.order) s1 = s1 - ci_1.order;
    object_.unit.set_value(s1);
    nb_models_ += 1;
}

UnmaskingError Aggregation::validate_unmasking(const MaskObject& mask) const {
    if (nb_models_ == 0) return UnmaskingError::NoModel;
    if (nb_models_ > object_.vect.cfg.info().max_nb_models) return UnmaskingError::TooManyModels;
    if (nb_models_ > object_.unit.cfg.info().max_nb_models) return UnmaskingError::TooManyScalars;
    if (object_.vect.cfg != mask.vect.cfg || object_size_ != mask.vect.count)
        return UnmaskingError::MaskManyMismatch;
    if (object_.unit.cfg != mask.unit.cfg) return UnmaskingError::MaskOneMismatch;
    if (!mask.is_valid()) return UnmaskingError::InvalidMask;
    return UnmaskingError::Ok;
}

RationalModel Aggregation::unmask(const MaskObject& mask) const {
    const auto& ci_1 = object_.unit.cfg.info();
    const auto& ci_n = object_.vect.cfg.info();
    BigInt nb(BigUint(uint64_t(nb_models_)), false);

    // scalar sum
    Rational scaled_add_1 = ci_1.add_shift * Rational::from_integer(nb);
    BigUint n1 = (object_.unit.value() + ci_1.order - mask.unit.value()) % ci_1.order;
    Rational scalar_sum =
        Rational(BigInt(n1, false), ci_1.exp_shift) - scaled_add_1;

    // model
    Rational scaled_add_n = ci_n.add_shift * Rational::from_integer(nb);
    RationalModel out;
    out.reserve(object_size_);
    for (size_t i = 0; i < object_size_; ++i) {
        BigUint n = (object_.vect.element(i) + ci_n.order - mask.vect.element(i)) % ci_n.order;
        Rational unmasked = Rational(BigInt(n, false), ci_n.exp_shift) - scaled_add_n;
        out.push_back(unmasked / scalar_sum);
    }
    return out;
}

// ------------------------------------------------------------ conversions

static Rational rational_from_double_bounded(double f, double max_abs) {
    if (std::isnan(f)) return Rational();
    if (f > max_abs) f = max_abs;
    if (f < -max_abs) f = -max_abs;
    return Rational::from_double(f);
}

RationalModel model_from_f32(const float* w, size_t n) {
    RationalModel m;
    m.reserve(n);
    for (size_t i = 0; i < n; ++i)
        m.push_back(rational_from_double_bounded(double(w[i]), double(std::numeric_limits<float>::max())));
    return m;
}
RationalModel model_from_f64(const double* w, size_t n) {
    RationalModel m;
    m.reserve(n);
    for (size_t i = 0; i < n; ++i)
        m.push_back(rational_from_double_bounded(w[i], std::numeric_limits<double>::max()));
    return m;
}
RationalModel model_from_i32(const int32_t* w, size_t n) {
    RationalModel m;
    m.reserve(n);
    for (size_t i = 0; i < n; ++i) m.push_back(Rational::from_integer(BigInt(int64_t(w[i]))));
    return m;
}
RationalModel model_from_i64(const int64_t* w, size_t n) {
    RationalModel m;
    m.reserve(n);
    for (size_t i = 0; i < n; ++i) m.push_back(Rational::from_integer(BigInt(w[i])));
    return m;
}

double ratio_to_double(const Rational& r) { return r.to_double(); }

float ratio_to_float(const Rational& r) {
    // mirror the reference shift loop semantics approximately: compute in
    // double, clamp into f32 range by halving numer/denom never changes the
    // value, so equivalent to a double->float rounding with range check
    double v = r.to_double();
    if (v > double(std::numeric_limits<float>::max()) ||
        v < -double(std::numeric_limits<float>::max())) {
        // reference returns None -> ModelCastError; callers use tolerance
        // paths, we saturate
        return v > 0 ? std::numeric_limits<float>::max() : -std::numeric_limits<float>::max();
    }
    return float(v);
}

std::vector<float> model_to_f32(const RationalModel& m) {
    std::vector<float> out;
    out.reserve(m.size());
    for (const auto& r : m) out.push_back(ratio_to_float(r));
    return out;
}
std::vector<double> model_to_f64(const RationalModel& m) {
    std::vector<double> out;
    out.reserve(m.size());
    for (const auto& r : m) out.push_back(ratio_to_double(r));
    return out;
}
std::vector<int64_t> model_to_i64(const RationalModel& m) {
    std::vector<int64_t> out;
    out.reserve(m.size());
    for (const auto& r : m) {
        BigInt t = r.trunc();
        int64_t v = int64_t(t.mag.low_u64());
        out.push_back(t.neg ? -v : v);
    }
    return out;
}
std::vector<int32_t> model_to_i32(const RationalModel& m) {
    std::vector<int32_t> out;
    out.reserve(m.size());
    for (const auto& r : m) {
        BigInt t = r.trunc();
        int32_t v = int32_t(t.mag.low_u64());
        out.push_back(t.neg ? -v : v);
    }
    return out;
}

}  // namespace xaynet::mask
