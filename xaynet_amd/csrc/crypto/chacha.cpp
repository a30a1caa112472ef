#include "chacha.h"

#include "../common.h"

namespace xaynet::crypto {

#define QR(a, b, c, d)                \
    a += b; d ^= a; d = rotl32(d, 16); \
    c += d; b ^= c; b = rotl32(b, 12); \
    a += b; d ^= a; d = rotl32(d, 8);  \
    c += d; b ^= c; b = rotl32(b, 7)

void chacha20_block(const uint8_t key[32], uint64_t counter, const uint8_t nonce[12],
                    uint8_t out[64]) {
    uint32_t s[16];
    s[0] = 0x61707865; s[1] = 0x3320646e; s[2] = 0x79622d32; s[3] = 0x6b206574;
    for (int i = 0; i < 8; ++i) s[4 + i] = load32_le(key + 4 * i);
    // 64-bit counter in words 12-13 (DJB variant); nonce words 14-15.
    // With nonce = 0 this is also the IETF layout for counters < 2^32.
    s[12] = uint32_t(counter);
    s[13] = uint32_t(counter >> 32) + (nonce ? load32_le(nonce) : 0);
    s[14] = nonce ? load32_le(nonce + 4) : 0;
    s[15] = nonce ? load32_le(nonce + 8) : 0;

    uint32_t x[16];
    for (int i = 0; i < 16; ++i) x[i] = s[i];
    for (int i = 0; i < 10; ++i) {
        QR(x[0], x[4], x[8], x[12]);
        QR(x[1], x[5], x[9], x[13]);
        QR(x[2], x[6], x[10], x[14]);
        QR(x[3], x[7], x[11], x[15]);
        QR(x[0], x[5], x[10], x[15]);
        QR(x[1], x[6], x[11], x[12]);
        QR(x[2], x[7], x[8], x[13]);
        QR(x[3], x[4], x[9], x[14]);
    }
    for (int i = 0; i < 16; ++i) store32_le(out + 4 * i, x[i] + s[i]);
}

ChaChaRng::ChaChaRng(const uint8_t seed[32]) { std::memcpy(key_, seed, 32); }

// 4 independent blocks with consecutive counters, interleaved state arrays
// so -O3 auto-vectorizes the quarter rounds 4-wide (SSE2 baseline).
static void chacha20_block4(const uint8_t key[32], uint64_t counter, uint8_t out[256]) {
    uint32_t s[16];
    s[0] = 0x61707865; s[1] = 0x3320646e; s[2] = 0x79622d32; s[3] = 0x6b206574;
    for (int i = 0; i < 8; ++i) s[4 + i] = load32_le(key + 4 * i);
    s[14] = 0;
    s[15] = 0;

    uint32_t x[16][4];
    uint32_t c0[4], c1[4];
    for (int b = 0; b < 4; ++b) {
        uint64_t ctr = counter + uint64_t(b);
        c0[b] = uint32_t(ctr);
        c1[b] = uint32_t(ctr >> 32);
    }
    for (int i = 0; i < 16; ++i)
        for (int b = 0; b < 4; ++b) x[i][b] = s[i];
    for (int b = 0; b < 4; ++b) {
        x[12][b] = c0[b];
        x[13][b] = c1[b];
    }

#define QR4(a, bq, c, d)                                                    for (int b = 0; b < 4; ++b) {                                               x[a][b] += x[bq][b]; x[d][b] ^= x[a][b]; x[d][b] = rotl32(x[d][b], 16);         x[c][b] += x[d][b]; x[bq][b] ^= x[c][b]; x[bq][b] = rotl32(x[bq][b], 12);         x[a][b] += x[bq][b]; x[d][b] ^= x[a][b]; x[d][b] = rotl32(x[d][b], 8);          x[c][b] += x[d][b]; x[bq][b] ^= x[c][b]; x[bq][b] = rotl32(x[bq][b], 7);     }
    for (int i = 0; i < 10; ++i) {
        QR4(0, 4, 8, 12);
        QR4(1, 5, 9, 13);
        QR4(2, 6, 10, 14);
        QR4(3, 7, 11, 15);
        QR4(0, 5, 10, 15);
        QR4(1, 6, 11, 12);
        QR4(2, 7, 8, 13);
        QR4(3, 4, 9, 14);
    }
#undef QR4

    for (int b = 0; b < 4; ++b) {
        uint8_t* o = out + 64 * b;
        for (int i = 0; i < 12; ++i) store32_le(o + 4 * i, x[i][b] + s[i]);
        store32_le(o + 48, x[12][b] + c0[b]);
        store32_le(o + 52, x[13][b] + c1[b]);
        store32_le(o + 56, x[14][b] + s[14]);
        store32_le(o + 60, x[15][b] + s[15]);
    }
}

void ChaChaRng::refill() {
    chacha20_block4(key_, block_idx_, buf_);
    block_idx_ += 4;
    block_off_ = 0;
}

uint64_t ChaChaRng::draw_u64_slow(int nbytes) {
    uint8_t tmp[8] = {0};
    fill_bytes(tmp, size_t(nbytes));
    uint64_t v = 0;
    std::memcpy(&v, tmp, 8);
    return v;
}

void ChaChaRng::fill_bytes(uint8_t* out, size_t n) {
    // Words needed for this fill: ceil(n/4); trailing bytes of the last word
    // are discarded (rand_core fill_via_u32_chunks).
    size_t need = n;
    while (need > 0) {
        if (block_off_ >= BUF) refill();
        size_t avail = BUF - block_off_;
        size_t take = need < avail ? need : avail;
        std::memcpy(out, buf_ + block_off_, take);
        out += take;
        need -= take;
        // advance by whole words
        size_t words = (take + 3) / 4;
        block_off_ += words * 4;
        word_pos_ += words;
    }
}

}  // namespace xaynet::crypto
