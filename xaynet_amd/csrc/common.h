// Common byte/int helpers for the xaynet_amd native core.
//
// The PET protocol is byte-format compatible with the reference
// (xaynetwork/xaynet); see SURVEY.md Appendix A for the wire contracts.
#pragma once

#include <cstdint>
#include <cstring>
#include <cstddef>
#include <string>
#include <vector>
#include <stdexcept>

namespace xaynet {

using Bytes = std::vector<uint8_t>;

inline uint32_t load32_le(const uint8_t* p) {
    uint32_t v;
    std::memcpy(&v, p, 4);
    return v;  // little-endian host
}

inline uint64_t load64_le(const uint8_t* p) {
    uint64_t v;
    std::memcpy(&v, p, 8);
    return v;
}

inline void store32_le(uint8_t* p, uint32_t v) { std::memcpy(p, &v, 4); }
inline void store64_le(uint8_t* p, uint64_t v) { std::memcpy(p, &v, 8); }

inline uint32_t load32_be(const uint8_t* p) {
    return (uint32_t(p[0]) << 24) | (uint32_t(p[1]) << 16) | (uint32_t(p[2]) << 8) | uint32_t(p[3]);
}
inline void store32_be(uint8_t* p, uint32_t v) {
    p[0] = uint8_t(v >> 24);
    p[1] = uint8_t(v >> 16);
    p[2] = uint8_t(v >> 8);
    p[3] = uint8_t(v);
}
inline uint64_t load64_be(const uint8_t* p) {
    return (uint64_t(load32_be(p)) << 32) | load32_be(p + 4);
}
inline void store64_be(uint8_t* p, uint64_t v) {
    store32_be(p, uint32_t(v >> 32));
    store32_be(p + 4, uint32_t(v));
}

inline uint32_t rotl32(uint32_t x, int n) { return (x << n) | (x >> (32 - n)); }
inline uint64_t rotr64(uint64_t x, int n) { return (x >> n) | (x << (64 - n)); }

// Constant-time byte comparison (crypto tags/keys).
inline bool ct_equal(const uint8_t* a, const uint8_t* b, size_t n) {
    uint8_t d = 0;
    for (size_t i = 0; i < n; ++i) d |= a[i] ^ b[i];
    return d == 0;
}

inline std::string to_hex(const uint8_t* p, size_t n) {
    static const char* k = "0123456789abcdef";
    std::string s;
    s.reserve(2 * n);
    for (size_t i = 0; i < n; ++i) {
        s.push_back(k[p[i] >> 4]);
        s.push_back(k[p[i] & 15]);
    }
    return s;
}

}  // namespace xaynet
