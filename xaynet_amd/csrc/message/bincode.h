// bincode v1 (fixint LE, u64 lengths) codec for the coordinator's HTTP
// bodies — byte-compatible with the reference's serde types:
//   GET /params -> RoundParameters   (rust/xaynet-core/src/common.rs:8-27)
//   GET /sums   -> Option<SumDict>   (HashMap<SumParticipantPublicKey, EphemeralPk>)
//   GET /seeds  -> Option<UpdateSeedDict> (HashMap<UpdatePk, EncryptedMaskSeed(Vec<u8>)>)
//   GET /model  -> Option<Model>     (Vec<Ratio<BigInt>>)
// serde mappings: fixed arrays are raw bytes; Vec/HashMap get a u64 LE length;
// enums are u32 LE variant indices; Option is a 0/1 byte; BigUint is a
// Vec<u32> of LE digits; BigInt is (Sign enum, BigUint); Ratio is
// {numer, denom} in order.
#pragma once

#include <map>
#include <optional>
#include <vector>

#include "../bigint.h"
#include "../common.h"
#include "../mask/config.h"
#include "message.h"

namespace xaynet::bincode {

struct Writer {
    Bytes out;
    void u8(uint8_t v) { out.push_back(v); }
    void u32(uint32_t v) {
        size_t n = out.size();
        out.resize(n + 4);
        store32_le(out.data() + n, v);
    }
    void u64(uint64_t v) {
        size_t n = out.size();
        out.resize(n + 8);
        store64_le(out.data() + n, v);
    }
    void f64(double v) {
        uint64_t bits;
        std::memcpy(&bits, &v, 8);
        u64(bits);
    }
    void raw(const uint8_t* p, size_t n) { out.insert(out.end(), p, p + n); }
    void bytes_vec(const uint8_t* p, size_t n) {
        u64(n);
        raw(p, n);
    }
};

struct Reader {
    const uint8_t* p;
    size_t len;
    size_t off = 0;
    bool fail = false;

    bool need(size_t n) {
        if (off + n > len) {
            fail = true;
            return false;
        }
        return true;
    }
    uint8_t u8() {
        if (!need(1)) return 0;
        return p[off++];
    }
    uint32_t u32() {
        if (!need(4)) return 0;
        uint32_t v = load32_le(p + off);
        off += 4;
        return v;
    }
    uint64_t u64() {
        if (!need(8)) return 0;
        uint64_t v = load64_le(p + off);
        off += 8;
        return v;
    }
    double f64() {
        uint64_t bits = u64();
        double v;
        std::memcpy(&v, &bits, 8);
        return v;
    }
    bool raw(uint8_t* dst, size_t n) {
        if (!need(n)) return false;
        std::memcpy(dst, p + off, n);
        off += n;
        return true;
    }
    void skip(size_t n) {
        if (need(n)) off += n;
    }
};

// ---- protocol dict types ----

using SumDict = std::map<msg::Key32, msg::Key32>;               // sum_pk -> ephm_pk
using UpdateSeedDict = std::map<msg::Key32, msg::EncrSeed80>;   // update_pk -> seed
using SeedDict = std::map<msg::Key32, UpdateSeedDict>;          // sum_pk -> ...

struct RoundParameters {
    msg::Key32 pk{};  // coordinator X25519 pk
    double sum = 0;
    double update = 0;
    msg::Key32 seed{};  // round seed
    mask::MaskConfigPair mask_config;
    uint64_t model_length = 0;
    bool operator==(const RoundParameters& o) const {
        return pk == o.pk && sum == o.sum && update == o.update && seed == o.seed &&
               mask_config == o.mask_config && model_length == o.model_length;
    }
};

// exact-rational model for GET /model (reference Model = Vec<Ratio<BigInt>>)
using RationalModel = std::vector<Rational>;

void write_mask_config(Writer& w, const mask::MaskConfig& c);
bool read_mask_config(Reader& r, mask::MaskConfig& c);

// num-bigint serde: BigUint <-> Vec<u32> LE digits
void write_biguint(Writer& w, const BigUint& v);
bool read_biguint(Reader& r, BigUint& v);

Bytes encode_round_parameters(const RoundParameters& rp);
std::optional<RoundParameters> decode_round_parameters(const uint8_t* p, size_t len);
// `consumed` receives the number of bytes read (robust reader advance for
// embedded RoundParameters, e.g. coordinator state restore)
std::optional<RoundParameters> decode_round_parameters(const uint8_t* p, size_t len,
                                                       size_t* consumed);

Bytes encode_option_sum_dict(const SumDict* d);  // nullptr -> None
std::optional<std::optional<SumDict>> decode_option_sum_dict(const uint8_t* p, size_t len);

Bytes encode_option_update_seed_dict(const UpdateSeedDict* d);
std::optional<std::optional<UpdateSeedDict>> decode_option_update_seed_dict(const uint8_t* p,
                                                                            size_t len);

Bytes encode_option_model(const RationalModel* m);
std::optional<std::optional<RationalModel>> decode_option_model(const uint8_t* p, size_t len);

// Chunked multi-thread encode: head ‖ parts[0] ‖ parts[1] ‖ … is
// byte-identical to the serial typed encoders below. Lets the binding
// assemble straight into the final Python bytes object (no concat copy).
struct EncodedModel {
    Bytes head;
    std::vector<Bytes> parts;
    size_t total() const {
        size_t t = head.size();
        for (const auto& p : parts) t += p.size();
        return t;
    }
    // copy head ‖ parts into dst (parallel memcpy for multi-GB bodies)
    void assemble(uint8_t* dst) const;
};
EncodedModel encode_option_model_mt_f32(const float* v, size_t n);
EncodedModel encode_option_model_mt_f64(const double* v, size_t n);
EncodedModel encode_option_model_mt_i32(const int32_t* v, size_t n);
EncodedModel encode_option_model_mt_i64(const int64_t* v, size_t n);

// typed fast paths: byte-identical to encode_option_model(model_from_*())
// without per-element BigInt work; fast decode returns false (caller falls
// back to the rational path) when an element is not a dyadic rational
Bytes encode_option_model_f32(const float* v, size_t n);
Bytes encode_option_model_f64(const double* v, size_t n);
Bytes encode_option_model_i32(const int32_t* v, size_t n);
Bytes encode_option_model_i64(const int64_t* v, size_t n);
bool decode_option_model_f32_fast(const uint8_t* p, size_t len, std::vector<float>& out);
bool decode_option_model_f64_fast(const uint8_t* p, size_t len, std::vector<double>& out);

// (coordinator-internal) whole SeedDict, used for checkpointing
Bytes encode_seed_dict(const SeedDict& d);
std::optional<SeedDict> decode_seed_dict(const uint8_t* p, size_t len);

}  // namespace xaynet::bincode
