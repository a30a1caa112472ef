#include "config.h"

#include <map>
#include <mutex>

#include "order_table.h"

namespace xaynet::mask {

static int group_idx(GroupType g) { return int(g); }

static int dtype_idx(DataType d) { return int(d); }

static int bound_idx(BoundType b) {
    switch (b) {
        case BoundType::B0: return 0;
        case BoundType::B2: return 1;
        case BoundType::B4: return 2;
        case BoundType::B6: return 3;
        case BoundType::Bmax: return 4;
    }
    return 0;
}

static int model_idx(ModelType m) {
    switch (m) {
        case ModelType::M3: return 0;
        case ModelType::M6: return 1;
        case ModelType::M9: return 2;
        case ModelType::M12: return 3;
    }
    return 0;
}

// Exact clamp bounds per the masking math (reference MaskConfig::add_shift,
// mask/config/mod.rs:196-213): Bmax uses the exact dtype maxima.
static Rational compute_add_shift(DataType d, BoundType b) {
    switch (b) {
        case BoundType::B0: return Rational::from_integer(BigInt(int64_t(1)));
        case BoundType::B2: return Rational::from_integer(BigInt(int64_t(100)));
        case BoundType::B4: return Rational::from_integer(BigInt(int64_t(10000)));
        case BoundType::B6: return Rational::from_integer(BigInt(int64_t(1000000)));
        case BoundType::Bmax: break;
    }
    switch (d) {
        case DataType::F32:  // f32::MAX = (2^24-1) * 2^104
            return Rational::from_integer(BigInt((BigUint(0xffffff) << 104), false));
        case DataType::F64:  // f64::MAX = (2^53-1) * 2^971
            return Rational::from_integer(BigInt((BigUint(0x1fffffffffffffULL) << 971), false));
        case DataType::I32:  // -i32::MIN = 2^31
            return Rational::from_integer(BigInt(BigUint(uint64_t(1) << 31), false));
        case DataType::I64:  // -i64::MIN = 2^63
            return Rational::from_integer(BigInt(BigUint::pow2(63), false));
    }
    return Rational();
}

static unsigned compute_exp_exponent(DataType d, BoundType b) {
    switch (d) {
        case DataType::F32: return b == BoundType::Bmax ? 45 : 10;
        case DataType::F64: return b == BoundType::Bmax ? 324 : 20;
        case DataType::I32:
        case DataType::I64: return 10;
    }
    return 10;
}

const CfgInfo& MaskConfig::info() const {
    static std::map<uint32_t, CfgInfo> cache;
    static std::mutex mu;
    std::lock_guard<std::mutex> lock(mu);
    auto it = cache.find(key());
    if (it != cache.end()) return it->second;

    CfgInfo ci;
    const char* order_str =
        ORDER_TABLE[group_idx(group)][dtype_idx(dtype)][bound_idx(bound)][model_idx(model)];
    ci.order = BigUint::from_dec(order_str);
    BigUint max_number = ci.order - BigUint(1);
    ci.bpn = (max_number.bits() + 7) / 8;
    ci.prng_nbytes = ci.order.to_bytes_le().size();
    ci.prng_words = (ci.prng_nbytes + 3) / 4;
    ci.add_shift = compute_add_shift(dtype, bound);
    ci.exp_exponent = compute_exp_exponent(dtype, bound);
    ci.exp_shift = BigUint::pow10(ci.exp_exponent);
    ci.max_nb_models = 1;
    for (int i = 0; i < int(model); ++i) ci.max_nb_models *= 10;
    ci.order_fits_u64 = ci.order.d.size() <= 1;
    ci.order_u64 = ci.order.low_u64();
    ci.n_digits32 = (ci.bpn + 3) / 4;
    return cache.emplace(key(), std::move(ci)).first->second;
}

std::optional<MaskConfig> MaskConfig::from_bytes(const uint8_t in[4]) {
    MaskConfig c;
    switch (in[0]) {
        case 0: c.group = GroupType::Integer; break;
        case 1: c.group = GroupType::Prime; break;
        case 2: c.group = GroupType::Power2; break;
        default: return std::nullopt;
    }
    switch (in[1]) {
        case 0: c.dtype = DataType::F32; break;
        case 1: c.dtype = DataType::F64; break;
        case 2: c.dtype = DataType::I32; break;
        case 3: c.dtype = DataType::I64; break;
        default: return std::nullopt;
    }
    switch (in[2]) {
        case 0: c.bound = BoundType::B0; break;
        case 2: c.bound = BoundType::B2; break;
        case 4: c.bound = BoundType::B4; break;
        case 6: c.bound = BoundType::B6; break;
        case 255: c.bound = BoundType::Bmax; break;
        default: return std::nullopt;
    }
    switch (in[3]) {
        case 3: c.model = ModelType::M3; break;
        case 6: c.model = ModelType::M6; break;
        case 9: c.model = ModelType::M9; break;
        case 12: c.model = ModelType::M12; break;
        default: return std::nullopt;
    }
    return c;
}

}  // namespace xaynet::mask
