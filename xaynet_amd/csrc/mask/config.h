// Masking configuration (group order / data type / bounds / model count).
//
// Byte-compatible with the reference wire encoding
// (rust/xaynet-core/src/mask/config/serialization.rs: 4 bytes =
// group, data, bound, model) and the order catalogue
// (rust/xaynet-core/src/mask/config/mod.rs:234-640; generated here from
// scripts/gen_order_table.py, verified equal to all 240 literals).
#pragma once

#include <cstdint>
#include <optional>

#include "../bigint.h"

namespace xaynet::mask {

enum class GroupType : uint8_t { Integer = 0, Prime = 1, Power2 = 2 };
enum class DataType : uint8_t { F32 = 0, F64 = 1, I32 = 2, I64 = 3 };
enum class BoundType : uint8_t { B0 = 0, B2 = 2, B4 = 4, B6 = 6, Bmax = 255 };
enum class ModelType : uint8_t { M3 = 3, M6 = 6, M9 = 9, M12 = 12 };

struct MaskConfig;

// Derived quantities, computed once per distinct config and cached.
struct CfgInfo {
    BigUint order;
    size_t bpn;           // serialized bytes per element: ((order-1).bits()+7)/8
    size_t prng_nbytes;   // PRNG draw size: order.to_bytes_le().len()  (may differ from bpn!)
    size_t prng_words;    // ceil(prng_nbytes/4): keystream words consumed per draw
    Rational add_shift;   // clamp bound (exact, per masking math — not the catalogue approx)
    BigUint exp_shift;    // fixed-point scale 10^k
    unsigned exp_exponent;
    uint64_t max_nb_models;
    bool order_fits_u64;  // fast-path flag (single 64-bit limb element)
    uint64_t order_u64;   // valid iff order_fits_u64
    size_t n_digits32;    // ceil(bpn/4): 32-bit digits per element (GPU digit planes)
};

struct MaskConfig {
    GroupType group = GroupType::Prime;
    DataType dtype = DataType::F32;
    BoundType bound = BoundType::B0;
    ModelType model = ModelType::M3;

    bool operator==(const MaskConfig& o) const {
        return group == o.group && dtype == o.dtype && bound == o.bound && model == o.model;
    }
    bool operator!=(const MaskConfig& o) const { return !(*this == o); }

    const CfgInfo& info() const;

    void write_bytes(uint8_t out[4]) const {
        out[0] = uint8_t(group);
        out[1] = uint8_t(dtype);
        out[2] = uint8_t(bound);
        out[3] = uint8_t(model);
    }
    static std::optional<MaskConfig> from_bytes(const uint8_t in[4]);

    uint32_t key() const {
        return uint32_t(group) | (uint32_t(dtype) << 8) | (uint32_t(bound) << 16) |
               (uint32_t(model) << 24);
    }
};

// Separate configs for the masked vector and the masked scalar unit
// (reference MaskConfigPair, mask/config/mod.rs:642-655).
struct MaskConfigPair {
    MaskConfig vect;
    MaskConfig unit;
    bool operator==(const MaskConfigPair& o) const { return vect == o.vect && unit == o.unit; }
};

}  // namespace xaynet::mask
