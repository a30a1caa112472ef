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

void xsalsa20_xor(uint8_t* c, const uint8_t* m, size_t len, const uint8_t nonce[24],
                  const uint8_t key[32], uint64_t ic) {
    // XSalsa20 = HSalsa20(key, nonce[0:16]) -> subkey; Salsa20(subkey, nonce[16:24])
    uint8_t subkey[32];
    hsalsa20(subkey, nonce, key);

    uint8_t n16[16];
    std::memcpy(n16, nonce + 16, 8);
    uint64_t ctr = ic;
    uint32_t st[16], x[16];
    uint8_t block[64];
    size_t off = 0;
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
