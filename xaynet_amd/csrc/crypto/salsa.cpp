#include "salsa.h"

#include "../common.h"

namespace xaynet::crypto {

// Salsa20 core on a 16-word state. If hsalsa, emit the HSalsa20 words
// (x0,x5,x10,x15,x6,x7,x8,x9) without the feed-forward; else standard
// Salsa20 block (x + input).
static void salsa20_core(uint32_t x[16], const uint32_t in[16], bool hsalsa) {
    for (int i = 0; i < 16; ++i) x[i] = in[i];
    for (int round = 0; round < 10; ++round) {
        // column round
        x[4] ^= rotl32(x[0] + x[12], 7);
        x[8] ^= rotl32(x[4] + x[0], 9);
        x[12] ^= rotl32(x[8] + x[4], 13);
        x[0] ^= rotl32(x[12] + x[8], 18);
        x[9] ^= rotl32(x[5] + x[1], 7);
        x[13] ^= rotl32(x[9] + x[5], 9);
        x[1] ^= rotl32(x[13] + x[9], 13);
        x[5] ^= rotl32(x[1] + x[13], 18);
        x[14] ^= rotl32(x[10] + x[6], 7);
        x[2] ^= rotl32(x[14] + x[10], 9);
        x[6] ^= rotl32(x[2] + x[14], 13);
        x[10] ^= rotl32(x[6] + x[2], 18);
        x[3] ^= rotl32(x[15] + x[11], 7);
        x[7] ^= rotl32(x[3] + x[15], 9);
        x[11] ^= rotl32(x[7] + x[3], 13);
        x[15] ^= rotl32(x[11] + x[7], 18);
        // row round
        x[1] ^= rotl32(x[0] + x[3], 7);
        x[2] ^= rotl32(x[1] + x[0], 9);
        x[3] ^= rotl32(x[2] + x[1], 13);
        x[0] ^= rotl32(x[3] + x[2], 18);
        x[6] ^= rotl32(x[5] + x[4], 7);
        x[7] ^= rotl32(x[6] + x[5], 9);
        x[4] ^= rotl32(x[7] + x[6], 13);
        x[5] ^= rotl32(x[4] + x[7], 18);
        x[11] ^= rotl32(x[10] + x[9], 7);
        x[8] ^= rotl32(x[11] + x[10], 9);
        x[9] ^= rotl32(x[8] + x[11], 13);
        x[10] ^= rotl32(x[9] + x[8], 18);
        x[12] ^= rotl32(x[15] + x[14], 7);
        x[13] ^= rotl32(x[12] + x[15], 9);
        x[14] ^= rotl32(x[13] + x[12], 13);
        x[15] ^= rotl32(x[14] + x[13], 18);
    }
    if (!hsalsa) {
        for (int i = 0; i < 16; ++i) x[i] += in[i];
    }
}

static const uint32_t SIGMA[4] = {0x61707865, 0x3320646e, 0x79622d32, 0x6b206574};

static void salsa20_state(uint32_t st[16], const uint8_t key[32], const uint8_t n[16]) {
    st[0] = SIGMA[0];
    st[5] = SIGMA[1];
    st[10] = SIGMA[2];
    st[15] = SIGMA[3];
    for (int i = 0; i < 4; ++i) st[1 + i] = load32_le(key + 4 * i);
    for (int i = 0; i < 4; ++i) st[11 + i] = load32_le(key + 16 + 4 * i);
    for (int i = 0; i < 4; ++i) st[6 + i] = load32_le(n + 4 * i);
}

void hsalsa20(uint8_t out[32], const uint8_t in[16], const uint8_t key[32]) {
    uint32_t st[16], x[16];
    salsa20_state(st, key, in);
    salsa20_core(x, st, true);
    store32_le(out + 0, x[0]);
    store32_le(out + 4, x[5]);
    store32_le(out + 8, x[10]);
    store32_le(out + 12, x[15]);
    store32_le(out + 16, x[6]);
    store32_le(out + 20, x[7]);
    store32_le(out + 24, x[8]);
    store32_le(out + 28, x[9]);
}

#if defined(__x86_64__)
#include <immintrin.h>

// AVX2 8-way Salsa20: eight consecutive counter blocks per pass in
// state-of-arrays form (16 ymm registers, lane b = block ctr+b). The
// counter is the only word that differs between lanes, so the stream is
// identical to the scalar walk — sealed-box decrypt of multi-MB update
// bodies is the serve plane's CPU bound (profiles/r02_ingest.md).
// 8x8 transpose of u32 lanes: out[b] holds lane b of each of r[0..7].
__attribute__((target("avx2"))) static inline void transpose8x8_u32(const __m256i* r,
                                                                    __m256i* out) {
    __m256i t0 = _mm256_unpacklo_epi32(r[0], r[1]);
    __m256i t1 = _mm256_unpackhi_epi32(r[0], r[1]);
    __m256i t2 = _mm256_unpacklo_epi32(r[2], r[3]);
    __m256i t3 = _mm256_unpackhi_epi32(r[2], r[3]);
    __m256i t4 = _mm256_unpacklo_epi32(r[4], r[5]);
    __m256i t5 = _mm256_unpackhi_epi32(r[4], r[5]);
    __m256i t6 = _mm256_unpacklo_epi32(r[6], r[7]);
    __m256i t7 = _mm256_unpackhi_epi32(r[6], r[7]);
    __m256i u0 = _mm256_unpacklo_epi64(t0, t2);
    __m256i u1 = _mm256_unpackhi_epi64(t0, t2);
    __m256i u2 = _mm256_unpacklo_epi64(t1, t3);
    __m256i u3 = _mm256_unpackhi_epi64(t1, t3);
    __m256i u4 = _mm256_unpacklo_epi64(t4, t6);
    __m256i u5 = _mm256_unpackhi_epi64(t4, t6);
    __m256i u6 = _mm256_unpacklo_epi64(t5, t7);
    __m256i u7 = _mm256_unpackhi_epi64(t5, t7);
    out[0] = _mm256_permute2x128_si256(u0, u4, 0x20);
    out[1] = _mm256_permute2x128_si256(u1, u5, 0x20);
    out[2] = _mm256_permute2x128_si256(u2, u6, 0x20);
    out[3] = _mm256_permute2x128_si256(u3, u7, 0x20);
    out[4] = _mm256_permute2x128_si256(u0, u4, 0x31);
    out[5] = _mm256_permute2x128_si256(u1, u5, 0x31);
    out[6] = _mm256_permute2x128_si256(u2, u6, 0x31);
    out[7] = _mm256_permute2x128_si256(u3, u7, 0x31);
}

__attribute__((target("avx2"))) static void salsa20_blocks8_avx2(
    uint8_t* c, const uint8_t* m, const uint32_t st0[16], uint64_t ctr) {
    __m256i x[16], in[16];
    for (int i = 0; i < 16; ++i) {
        if (i == 8) {  // counter low word
            in[i] = _mm256_add_epi32(
                _mm256_set1_epi32(int(uint32_t(ctr))),
                _mm256_setr_epi32(0, 1, 2, 3, 4, 5, 6, 7));
            // carry into the high word per lane
        } else if (i == 9) {
            uint32_t hi[8];
            for (int b = 0; b < 8; ++b) hi[b] = uint32_t((ctr + b) >> 32);
            in[i] = _mm256_setr_epi32(int(hi[0]), int(hi[1]), int(hi[2]), int(hi[3]),
                                      int(hi[4]), int(hi[5]), int(hi[6]), int(hi[7]));
        } else {
            in[i] = _mm256_set1_epi32(int(st0[i]));
        }
        x[i] = in[i];
    }
#define VROTL(v, r) _mm256_or_si256(_mm256_slli_epi32(v, r), _mm256_srli_epi32(v, 32 - (r)))
#define VQR(a, b, d, e)                                                       \
    x[a] = _mm256_xor_si256(x[a], VROTL(_mm256_add_epi32(x[b], x[d]), 7));    \
    x[e] = _mm256_xor_si256(x[e], VROTL(_mm256_add_epi32(x[a], x[b]), 9));    \
    x[d] = _mm256_xor_si256(x[d], VROTL(_mm256_add_epi32(x[e], x[a]), 13));   \
    x[b] = _mm256_xor_si256(x[b], VROTL(_mm256_add_epi32(x[d], x[e]), 18))
    for (int round = 0; round < 10; ++round) {
        // column round: (x4,x0,x12,x8), (x9,x5,x1,x13), (x14,x10,x6,x2), (x3,x15,x11,x7)
        VQR(4, 0, 12, 8);
        VQR(9, 5, 1, 13);
        VQR(14, 10, 6, 2);
        VQR(3, 15, 11, 7);
        // row round
        VQR(1, 0, 3, 2);
        VQR(6, 5, 4, 7);
        VQR(11, 10, 9, 8);
        VQR(12, 15, 14, 13);
    }
#undef VQR
#undef VROTL
    for (int i = 0; i < 16; ++i) x[i] = _mm256_add_epi32(x[i], in[i]);
    // De-interleave lanes -> 8 sequential 64-byte blocks entirely in
    // registers: two 8x8 u32 transposes (words 0-7 and 8-15), then each
    // block b is [rowA_b ‖ rowB_b]. The earlier scalar word-by-word
    // de-interleave cost more than the Salsa rounds themselves.
    __m256i rowA[8], rowB[8];
    transpose8x8_u32(x, rowA);
    transpose8x8_u32(x + 8, rowB);
    for (int b = 0; b < 8; ++b) {
        uint8_t* dst = c + size_t(b) * 64;
        __m256i lo = rowA[b], hi = rowB[b];
        if (m) {
            const uint8_t* src = m + size_t(b) * 64;
            lo = _mm256_xor_si256(
                lo, _mm256_loadu_si256(reinterpret_cast<const __m256i*>(src)));
            hi = _mm256_xor_si256(
                hi, _mm256_loadu_si256(reinterpret_cast<const __m256i*>(src + 32)));
        }
        _mm256_storeu_si256(reinterpret_cast<__m256i*>(dst), lo);
        _mm256_storeu_si256(reinterpret_cast<__m256i*>(dst + 32), hi);
    }
}

static bool have_avx2() {
    static const bool v = __builtin_cpu_supports("avx2");
    return v;
}
#endif  // __x86_64__

void xsalsa20_xor(uint8_t* c, const uint8_t* m, size_t len, const uint8_t nonce[24],
                  const uint8_t key[32], uint64_t ic) {
    // XSalsa20 = HSalsa20(key, nonce[0:16]) -> subkey; Salsa20(subkey, nonce[16:24])
    uint8_t subkey[32];
    hsalsa20(subkey, nonce, key);

    uint8_t n16[16];
    std::memcpy(n16, nonce + 16, 8);
    uint64_t ctr = ic;
    size_t off = 0;

#if defined(__x86_64__)
    if (len - off >= 512 && have_avx2()) {
        uint32_t st0[16];
        store32_le(n16 + 8, 0);
        store32_le(n16 + 12, 0);
        salsa20_state(st0, subkey, n16);
        while (len - off >= 512) {
            salsa20_blocks8_avx2(c + off, m ? m + off : nullptr, st0, ctr);
            off += 512;
            ctr += 8;
        }
    }
#endif

    uint32_t st[16], x[16];
    uint8_t block[64];
    while (off < len) {
        store32_le(n16 + 8, uint32_t(ctr));
        store32_le(n16 + 12, uint32_t(ctr >> 32));
        salsa20_state(st, subkey, n16);
        salsa20_core(x, st, false);
        for (int i = 0; i < 16; ++i) store32_le(block + 4 * i, x[i]);
        size_t take = len - off < 64 ? len - off : 64;
        if (m) {
            for (size_t i = 0; i < take; ++i) c[off + i] = m[off + i] ^ block[i];
        } else {
            std::memcpy(c + off, block, take);
        }
        off += take;
        ctr += 1;
    }
}

}  // namespace xaynet::crypto
