#include "object.h"

namespace xaynet::mask {

bool MaskVect::is_valid() const {
    const auto& ci = cfg.info();
    if (data.size() != count * ci.bpn) return false;
    // This check runs per accepted update on the protocol thread (reference
    // masking.rs:253-279 via object.is_valid), so it must stream at memory
    // speed: branchless u64 compares instead of per-byte lexicographic loops.
    if (ci.bpn <= 8 && count > 0) {
        uint64_t order = 0;
        Bytes ob(8, 0);
        ci.order.to_bytes_le_fixed(ob.data(), ci.bpn);
        std::memcpy(&order, ob.data(), 8);
        const uint64_t mask = ci.bpn == 8 ? ~0ull : ((1ull << (8 * ci.bpn)) - 1);
        const uint8_t* p = data.data();
        size_t n = count;
        bool ok = true;
        for (size_t i = 0; i + 1 < n; ++i) {  // all but last: safe 8B load
            uint64_t v;
            std::memcpy(&v, p + i * ci.bpn, 8);
            ok &= (v & mask) < order;
        }
        uint64_t last = 0;
        std::memcpy(&last, p + (n - 1) * ci.bpn, ci.bpn);
        return ok && last < order;
    }
    Bytes order_le(ci.bpn, 0);
    ci.order.to_bytes_le_fixed(order_le.data(), ci.bpn);
    for (size_t i = 0; i < count; ++i) {
        const uint8_t* e = data.data() + i * ci.bpn;
        // e < order ?
        bool less = false, geq = false;
        for (size_t b = ci.bpn; b-- > 0;) {
            if (e[b] != order_le[b]) {
                (e[b] < order_le[b] ? less : geq) = true;
                break;
            }
        }
        if (!less && !geq) return false;  // equal to order
        if (geq) return false;
    }
    return true;
}

void MaskVect::serialize(uint8_t* out) const {
    cfg.write_bytes(out);
    store32_be(out + 4, uint32_t(count));
    std::memcpy(out + 8, data.data(), data.size());
}

std::optional<MaskVect> MaskVect::deserialize(const uint8_t* p, size_t len, size_t* consumed) {
    if (len < 8) return std::nullopt;
    auto cfg = MaskConfig::from_bytes(p);
    if (!cfg) return std::nullopt;
    size_t count = load32_be(p + 4);
    size_t bpn = cfg->info().bpn;
    size_t total = 8 + count * bpn;
    if (len < total) return std::nullopt;
    MaskVect v;
    v.cfg = *cfg;
    v.count = count;
    v.data.assign(p + 8, p + total);
    if (consumed) *consumed = total;
    return v;
}

void MaskUnit::serialize(uint8_t* out) const {
    cfg.write_bytes(out);
    std::memcpy(out + 4, data.data(), data.size());
}

std::optional<MaskUnit> MaskUnit::deserialize(const uint8_t* p, size_t len, size_t* consumed) {
    if (len < 4) return std::nullopt;
    auto cfg = MaskConfig::from_bytes(p);
    if (!cfg) return std::nullopt;
    size_t bpn = cfg->info().bpn;
    if (len < 4 + bpn) return std::nullopt;
    MaskUnit u;
    u.cfg = *cfg;
    u.data.assign(p + 4, p + 4 + bpn);
    if (consumed) *consumed = 4 + bpn;
    return u;
}

Bytes MaskObject::serialize() const {
    Bytes out(byte_len());
    vect.serialize(out.data());
    unit.serialize(out.data() + vect.byte_len());
    return out;
}

std::optional<MaskObject> MaskObject::deserialize(const uint8_t* p, size_t len, size_t* consumed) {
    size_t c1 = 0, c2 = 0;
    auto v = MaskVect::deserialize(p, len, &c1);
    if (!v) return std::nullopt;
    auto u = MaskUnit::deserialize(p + c1, len - c1, &c2);
    if (!u) return std::nullopt;
    if (consumed) *consumed = c1 + c2;
    return MaskObject{std::move(*v), std::move(*u)};
}

}  // namespace xaynet::mask
