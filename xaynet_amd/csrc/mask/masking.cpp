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
    uint8_t buf[8] = {0};
    while (true) {
        rng_.fill_bytes(buf, ci.prng_nbytes);
        uint64_t v = load64_le(buf);
        if (v < ci.order_u64) return v;
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

MaskObject mask_f32(const uint8_t seed[32], const Scalar& scalar, const float* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_f32(w, n), cfg);
}
MaskObject mask_f64(const uint8_t seed[32], const Scalar& scalar, const double* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_f64(w, n), cfg);
}
MaskObject mask_i32(const uint8_t seed[32], const Scalar& scalar, const int32_t* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_i32(w, n), cfg);
}
MaskObject mask_i64(const uint8_t seed[32], const Scalar& scalar, const int64_t* w, size_t n,
                    const MaskConfigPair& cfg) {
    return mask_model(seed, scalar, model_from_i64(w, n), cfg);
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
