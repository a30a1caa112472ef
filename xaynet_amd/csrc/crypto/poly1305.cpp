#include "poly1305.h"

#include "../common.h"

namespace xaynet::crypto {

// 64-bit limb implementation (2x 44-bit style simplified to 3x u64 with
// unsigned __int128 products; r is clamped so products fit comfortably).
void poly1305_mac(uint8_t tag[16], const uint8_t* msg, size_t len, const uint8_t key[32]) {
    using u128 = unsigned __int128;

    // r clamped, split into 26-bit limbs for a classic 5-limb implementation.
    uint32_t r0 = load32_le(key + 0) & 0x3ffffff;
    uint32_t r1 = (load32_le(key + 3) >> 2) & 0x3ffff03;
    uint32_t r2 = (load32_le(key + 6) >> 4) & 0x3ffc0ff;
    uint32_t r3 = (load32_le(key + 9) >> 6) & 0x3f03fff;
    uint32_t r4 = (load32_le(key + 12) >> 8) & 0x00fffff;

    uint32_t s1 = r1 * 5, s2 = r2 * 5, s3 = r3 * 5, s4 = r4 * 5;

    uint32_t h0 = 0, h1 = 0, h2 = 0, h3 = 0, h4 = 0;

    while (len > 0) {
        uint8_t block[16] = {0};
        size_t take = len < 16 ? len : 16;
        std::memcpy(block, msg, take);
        uint32_t hibit = (take == 16) ? (1u << 24) : 0;
        if (take < 16) block[take] = 1;

        h0 += load32_le(block + 0) & 0x3ffffff;
        h1 += (load32_le(block + 3) >> 2) & 0x3ffffff;
        h2 += (load32_le(block + 6) >> 4) & 0x3ffffff;
        h3 += (load32_le(block + 9) >> 6) & 0x3ffffff;
        h4 += (load32_le(block + 12) >> 8) | hibit;

        uint64_t d0 = (uint64_t)h0 * r0 + (uint64_t)h1 * s4 + (uint64_t)h2 * s3 +
                      (uint64_t)h3 * s2 + (uint64_t)h4 * s1;
        uint64_t d1 = (uint64_t)h0 * r1 + (uint64_t)h1 * r0 + (uint64_t)h2 * s4 +
                      (uint64_t)h3 * s3 + (uint64_t)h4 * s2;
        uint64_t d2 = (uint64_t)h0 * r2 + (uint64_t)h1 * r1 + (uint64_t)h2 * r0 +
                      (uint64_t)h3 * s4 + (uint64_t)h4 * s3;
        uint64_t d3 = (uint64_t)h0 * r3 + (uint64_t)h1 * r2 + (uint64_t)h2 * r1 +
                      (uint64_t)h3 * r0 + (uint64_t)h4 * s4;
        uint64_t d4 = (uint64_t)h0 * r4 + (uint64_t)h1 * r3 + (uint64_t)h2 * r2 +
                      (uint64_t)h3 * r1 + (uint64_t)h4 * r0;

        uint64_t c;
        c = d0 >> 26; h0 = uint32_t(d0) & 0x3ffffff;
        d1 += c; c = d1 >> 26; h1 = uint32_t(d1) & 0x3ffffff;
        d2 += c; c = d2 >> 26; h2 = uint32_t(d2) & 0x3ffffff;
        d3 += c; c = d3 >> 26; h3 = uint32_t(d3) & 0x3ffffff;
        d4 += c; c = d4 >> 26; h4 = uint32_t(d4) & 0x3ffffff;
        h0 += uint32_t(c) * 5;
        c = h0 >> 26; h0 &= 0x3ffffff;
        h1 += uint32_t(c);

        msg += take;
        len -= take;
    }

    // full carry
    uint32_t c;
    c = h1 >> 26; h1 &= 0x3ffffff; h2 += c;
    c = h2 >> 26; h2 &= 0x3ffffff; h3 += c;
    c = h3 >> 26; h3 &= 0x3ffffff; h4 += c;
    c = h4 >> 26; h4 &= 0x3ffffff; h0 += c * 5;
    c = h0 >> 26; h0 &= 0x3ffffff; h1 += c;

    // compute h + -p
    uint32_t g0 = h0 + 5; c = g0 >> 26; g0 &= 0x3ffffff;
    uint32_t g1 = h1 + c; c = g1 >> 26; g1 &= 0x3ffffff;
    uint32_t g2 = h2 + c; c = g2 >> 26; g2 &= 0x3ffffff;
    uint32_t g3 = h3 + c; c = g3 >> 26; g3 &= 0x3ffffff;
    uint32_t g4 = h4 + c - (1u << 26);

    // select h if h < p, else g
    uint32_t mask = (g4 >> 31) - 1;  // all-ones if g4 >= 0 (i.e. h >= p)
    g0 &= mask; g1 &= mask; g2 &= mask; g3 &= mask; g4 &= mask;
    mask = ~mask;
    h0 = (h0 & mask) | g0;
    h1 = (h1 & mask) | g1;
    h2 = (h2 & mask) | g2;
    h3 = (h3 & mask) | g3;
    h4 = (h4 & mask) | g4;

    // h = h % 2^128, then add s
    uint64_t f0 = (uint64_t(h0) | (uint64_t(h1) << 26)) & 0xffffffffULL;
    uint64_t f1 = (uint64_t(h1 >> 6) | (uint64_t(h2) << 20)) & 0xffffffffULL;
    uint64_t f2 = (uint64_t(h2 >> 12) | (uint64_t(h3) << 14)) & 0xffffffffULL;
    uint64_t f3 = (uint64_t(h3 >> 18) | (uint64_t(h4) << 8)) & 0xffffffffULL;

    u128 acc = (u128)f0 + load32_le(key + 16);
    store32_le(tag + 0, uint32_t(acc));
    acc = (acc >> 32) + (u128)f1 + load32_le(key + 20);
    store32_le(tag + 4, uint32_t(acc));
    acc = (acc >> 32) + (u128)f2 + load32_le(key + 24);
    store32_le(tag + 8, uint32_t(acc));
    acc = (acc >> 32) + (u128)f3 + load32_le(key + 28);
    store32_le(tag + 12, uint32_t(acc));
}

}  // namespace xaynet::crypto
