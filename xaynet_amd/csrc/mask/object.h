// Mask object containers with GPU-friendly packed-limb storage.
//
// MI355X-first design decision: a masked vector is stored as the wire format
// itself — a dense (count x bytes_per_number) little-endian limb matrix
// (reference serialization: rust/xaynet-core/src/mask/object/serialization/
// vect.rs:172-246). Serialization is a memcpy, H2D upload needs no repacking,
// and the GPU kernels unpack limbs in-register.
#pragma once

#include <optional>

#include "../common.h"
#include "config.h"

namespace xaynet::mask {

struct MaskVect {
    MaskConfig cfg;
    size_t count = 0;
    Bytes data;  // count * cfg.info().bpn bytes, LE limbs per element

    static MaskVect zeros(const MaskConfig& c, size_t n) {
        MaskVect v;
        v.cfg = c;
        v.count = n;
        v.data.assign(n * c.info().bpn, 0);
        return v;
    }
    BigUint element(size_t i) const {
        size_t bpn = cfg.info().bpn;
        return BigUint::from_bytes_le(data.data() + i * bpn, bpn);
    }
    void set_element(size_t i, const BigUint& v) {
        size_t bpn = cfg.info().bpn;
        v.to_bytes_le_fixed(data.data() + i * bpn, bpn);
    }
    bool is_valid() const;

    // wire: config(4) || count(u32 BE) || limbs
    size_t byte_len() const { return 8 + data.size(); }
    void serialize(uint8_t* out) const;
    static std::optional<MaskVect> deserialize(const uint8_t* p, size_t len, size_t* consumed);
};

struct MaskUnit {
    MaskConfig cfg;
    Bytes data;  // bpn bytes

    static MaskUnit zero(const MaskConfig& c) {
        MaskUnit u;
        u.cfg = c;
        u.data.assign(c.info().bpn, 0);
        return u;
    }
    BigUint value() const { return BigUint::from_bytes_le(data.data(), data.size()); }
    void set_value(const BigUint& v) { v.to_bytes_le_fixed(data.data(), data.size()); }
    bool is_valid() const { return value() < cfg.info().order; }

    // wire: config(4) || one limb value (no count field)
    size_t byte_len() const { return 4 + data.size(); }
    void serialize(uint8_t* out) const;
    static std::optional<MaskUnit> deserialize(const uint8_t* p, size_t len, size_t* consumed);
};

struct MaskObject {
    MaskVect vect;
    MaskUnit unit;

    static MaskObject zeros(const MaskConfigPair& c, size_t n) {
        return MaskObject{MaskVect::zeros(c.vect, n), MaskUnit::zero(c.unit)};
    }
    MaskConfigPair config() const { return MaskConfigPair{vect.cfg, unit.cfg}; }
    bool is_valid() const { return vect.is_valid() && unit.is_valid(); }

    size_t byte_len() const { return vect.byte_len() + unit.byte_len(); }
    Bytes serialize() const;
    static std::optional<MaskObject> deserialize(const uint8_t* p, size_t len, size_t* consumed);
};

}  // namespace xaynet::mask
