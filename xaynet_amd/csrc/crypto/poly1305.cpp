#include "poly1305.h"

#include "../common.h"

namespace xaynet::crypto {

// Poly1305 MAC over 2^130-5, 3x44-bit limbs with unsigned __int128 products
// (the 64-bit-multiplier radix: 9 MULs per 16-byte block). key[0..16] is the
// clamped r, key[16..32] the final pad s. Full blocks are loaded directly
// (no staging memcpy); only the final partial block goes through a padded
// buffer. Replaces the round-1 26-bit 5-limb version: the serve-plane 25M
// ingest is bound by CPU sealed-box open, and this MAC is its hot pass
// (reference: xaynet sealed-box messages, crypto_box_seal / secretbox).
void poly1305_mac(uint8_t tag[16], const uint8_t* msg, size_t len, const uint8_t key[32]) {
    using u128 = unsigned __int128;
    constexpr uint64_t M44 = 0xfffffffffffULL;   // 2^44 - 1
    constexpr uint64_t M42 = 0x3ffffffffffULL;   // 2^42 - 1

    // r, clamped to the Poly1305 key shape, split 44/44/42.
    uint64_t t0 = load64_le(key + 0);
    uint64_t t1 = load64_le(key + 8);
    const uint64_t r0 = t0 & 0xffc0fffffffULL;
    const uint64_t r1 = ((t0 >> 44) | (t1 << 20)) & 0xfffffc0ffffULL;
    const uint64_t r2 = (t1 >> 24) & 0x00ffffffc0fULL;
    // 5*4*r: the 2^130-5 wraparound factor folded with the 2^2 radix gap
    // (limb 2 is 42 bits, so a wrap from limb 2 carries weight 2^-88 * 2^130
    // = 2^42 => multiply by 5 * 2^2 when folding into a 44-bit column).
    const uint64_t s1 = r1 * 20;
    const uint64_t s2 = r2 * 20;

    uint64_t h0 = 0, h1 = 0, h2 = 0;

    auto absorb = [&](uint64_t m0, uint64_t m1, uint64_t hibit) {
        h0 += m0 & M44;
        h1 += ((m0 >> 44) | (m1 << 20)) & M44;
        h2 += ((m1 >> 24) & M42) | hibit;
        u128 d0 = (u128)h0 * r0 + (u128)h1 * s2 + (u128)h2 * s1;
        u128 d1 = (u128)h0 * r1 + (u128)h1 * r0 + (u128)h2 * s2;
        u128 d2 = (u128)h0 * r2 + (u128)h1 * r1 + (u128)h2 * r0;
        uint64_t c = uint64_t(d0 >> 44);
        h0 = uint64_t(d0) & M44;
        d1 += c;
        c = uint64_t(d1 >> 44);
        h1 = uint64_t(d1) & M44;
        d2 += c;
        c = uint64_t(d2 >> 42);
        h2 = uint64_t(d2) & M42;
        h0 += c * 5;
        c = h0 >> 44;
        h0 &= M44;
        h1 += c;
    };

    while (len >= 16) {
        absorb(load64_le(msg), load64_le(msg + 8), 1ULL << 40);
        msg += 16;
        len -= 16;
    }
    if (len > 0) {
        uint8_t block[16] = {0};
        std::memcpy(block, msg, len);
        block[len] = 1;
        absorb(load64_le(block), load64_le(block + 8), 0);
    }

    // full carry propagation
    uint64_t c;
    c = h1 >> 44; h1 &= M44; h2 += c;
    c = h2 >> 42; h2 &= M42; h0 += c * 5;
    c = h0 >> 44; h0 &= M44; h1 += c;
    c = h1 >> 44; h1 &= M44; h2 += c;
    c = h2 >> 42; h2 &= M42; h0 += c * 5;
    c = h0 >> 44; h0 &= M44; h1 += c;

    // g = h + 5 - 2^130; select g when h >= p (no borrow out of the top)
    uint64_t g0 = h0 + 5; c = g0 >> 44; g0 &= M44;
    uint64_t g1 = h1 + c; c = g1 >> 44; g1 &= M44;
    uint64_t g2 = h2 + c - (1ULL << 42);
    uint64_t mask = (g2 >> 63) - 1;  // all-ones when h >= p
    h0 = (h0 & ~mask) | (g0 & mask);
    h1 = (h1 & ~mask) | (g1 & mask);
    h2 = (h2 & ~mask) | (g2 & mask);

    // tag = (h + s) mod 2^128
    uint64_t p0 = load64_le(key + 16);
    uint64_t p1 = load64_le(key + 24);
    u128 f = (u128)(h0 | (h1 << 44)) + p0;
    store64_le(tag + 0, uint64_t(f));
    f = (f >> 64) + (u128)((h1 >> 20) | (h2 << 24)) + p1;
    store64_le(tag + 8, uint64_t(f));
}

}  // namespace xaynet::crypto
