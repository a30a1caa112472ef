// Masking, aggregation and unmasking (CPU reference path).
//
// Semantics mirror the reference exactly
// (rust/xaynet-core/src/mask/masking.rs):
//   mask:    clamp(scalar)*w -> clamp to [-add_shift, add_shift] ->
//            trunc((v + add_shift) * exp_shift) -> (+ prng) mod order
//   agg:     element-wise (a + b) mod order; first aggregate replaces
//   unmask:  (masked + order - mask) % order -> /exp_shift - n*add_shift
//            -> / scalar_sum   (exact rationals)
// PRNG stream: ONE unit draw (unit config) then `len` vect draws
// (rust/xaynet-core/src/mask/seed.rs:61-78); each draw consumes
// ceil(prng_nbytes/4) ChaCha20 keystream words per attempt (rejection
// sampling, rust/xaynet-core/src/crypto/prng.rs:16-27).
//
// This CPU path is the bit-exactness oracle for the HIP kernels (K1-K7) and
// the fallback for exotic configs; the production hot path is
// xaynet_amd/gpu (digit-plane aggregation on MI355X).
#pragma once

#include <vector>

#include "../crypto/chacha.h"
#include "object.h"

namespace xaynet::mask {

// Scalar = non-negative rational (reference mask/scalar.rs).
struct Scalar {
    BigUint numer;
    BigUint denom;
    Scalar() : numer(1), denom(1) {}
    Scalar(uint64_t n, uint64_t d) : numer(n), denom(d) {}
    Rational to_rational() const { return Rational(BigInt(numer, false), denom); }
};

// Model weights as exact rationals (reference Model = Vec<Ratio<BigInt>>).
using RationalModel = std::vector<Rational>;

// PRNG over a mask seed.
class MaskPrng {
  public:
    explicit MaskPrng(const uint8_t seed[32]) : rng_(seed) {}
    // one uniform draw in [0, order) — oracle path
    BigUint generate_integer(const CfgInfo& ci);
    uint64_t generate_u64(const CfgInfo& ci);  // fast path, requires prng_nbytes <= 8
    // fast path for u128 orders (prng_nbytes <= 16); order passed pre-loaded
    unsigned __int128 generate_u128(const CfgInfo& ci, unsigned __int128 order);
    uint64_t words_consumed() const { return rng_.words_consumed(); }
  private:
    crypto::ChaChaRng rng_;
};

// Expand a 32-byte seed into a full mask object (reference
// MaskSeed::derive_mask).
MaskObject derive_mask(const uint8_t seed[32], size_t len, const MaskConfigPair& cfg);

// Mask a rational model with a fresh or given seed. Returns the masked object;
// the caller provides the seed (32 bytes).
MaskObject mask_model(const uint8_t seed[32], const Scalar& scalar, const RationalModel& model,
                      const MaskConfigPair& cfg);

// Convenience: mask primitive weights (f32/f64/i32/i64 per cfg.vect dtype).
MaskObject mask_f32(const uint8_t seed[32], const Scalar& scalar, const float* w, size_t n,
                    const MaskConfigPair& cfg);
MaskObject mask_f64(const uint8_t seed[32], const Scalar& scalar, const double* w, size_t n,
                    const MaskConfigPair& cfg);
MaskObject mask_i32(const uint8_t seed[32], const Scalar& scalar, const int32_t* w, size_t n,
                    const MaskConfigPair& cfg);
MaskObject mask_i64(const uint8_t seed[32], const Scalar& scalar, const int64_t* w, size_t n,
                    const MaskConfigPair& cfg);

enum class AggregationError {
    Ok = 0,
    InvalidObject,
    TooManyModels,
    TooManyScalars,
    ModelMismatch,
    ScalarMismatch,
};

enum class UnmaskingError {
    Ok = 0,
    NoModel,
    TooManyModels,
    TooManyScalars,
    MaskManyMismatch,
    MaskOneMismatch,
    InvalidMask,
};

// CPU aggregator (oracle semantics; the GPU engine implements the same
// contract over digit planes).
class Aggregation {
  public:
    Aggregation(const MaskConfigPair& cfg, size_t object_size)
        : nb_models_(0), object_size_(object_size), object_(MaskObject::zeros(cfg, object_size)) {}
    explicit Aggregation(MaskObject obj)
        : nb_models_(1), object_size_(obj.vect.count), object_(std::move(obj)) {}

    size_t len() const { return object_size_; }
    size_t nb_models() const { return nb_models_; }
    const MaskObject& object() const { return object_; }
    MaskConfigPair config() const { return object_.config(); }

    AggregationError validate_aggregation(const MaskObject& obj) const;
    // round-state part only (model-count caps): for updates whose pure
    // checks (config/length/element range) ran on the ingest thread
    AggregationError validate_counts_only() const;
    void aggregate(const MaskObject& obj);

    UnmaskingError validate_unmasking(const MaskObject& mask) const;
    RationalModel unmask(const MaskObject& mask) const;

    // set state directly (restore paths / GPU readback)
    void set(MaskObject obj, size_t nb_models) {
        object_size_ = obj.vect.count;
        object_ = std::move(obj);
        nb_models_ = nb_models;
    }

  private:
    size_t nb_models_;
    size_t object_size_;
    MaskObject object_;
};

// Model <-> primitive conversions (reference mask/model.rs semantics).
RationalModel model_from_f32(const float* w, size_t n);   // bounded (NaN->0, inf->max)
RationalModel model_from_f64(const double* w, size_t n);
RationalModel model_from_i32(const int32_t* w, size_t n);
RationalModel model_from_i64(const int64_t* w, size_t n);
std::vector<float> model_to_f32(const RationalModel& m);
std::vector<double> model_to_f64(const RationalModel& m);
std::vector<int64_t> model_to_i64(const RationalModel& m);
std::vector<int32_t> model_to_i32(const RationalModel& m);

// ratio -> float with the reference's shift-until-representable loop
double ratio_to_double(const Rational& r);
float ratio_to_float(const Rational& r);

}  // namespace xaynet::mask
