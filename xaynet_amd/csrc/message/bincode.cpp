#include "bincode.h"

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

std::optional<RoundParameters> decode_round_parameters(const uint8_t* p, size_t len) {
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
    return rp;
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
static void write_biguint(Writer& w, const BigUint& v) {
    std::vector<uint32_t> digits;
    for (uint64_t limb : v.d) {
        digits.push_back(uint32_t(limb));
        digits.push_back(uint32_t(limb >> 32));
    }
    while (!digits.empty() && digits.back() == 0) digits.pop_back();
    w.u64(digits.size());
    for (uint32_t dg : digits) w.u32(dg);
}

static bool read_biguint(Reader& r, BigUint& v) {
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
