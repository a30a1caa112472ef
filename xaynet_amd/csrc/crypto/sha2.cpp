#include "sha2.h"

#include "../common.h"

namespace xaynet::crypto {

// ---------------------------------------------------------------- SHA-256

static const uint32_t K256[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1, 0x923f82a4, 0xab1c5ed5,
    0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3, 0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174,
    0xe49b69c1, 0xefbe4786, 0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147, 0x06ca6351, 0x14292967,
    0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13, 0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85,
    0xa2bfe8a1, 0xa81a664b, 0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a, 0x5b9cca4f, 0x682e6ff3,
    0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208, 0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2,
};

static inline uint32_t ror32(uint32_t x, int n) { return (x >> n) | (x << (32 - n)); }

static void sha256_block(uint32_t h[8], const uint8_t* p) {
    uint32_t w[64];
    for (int i = 0; i < 16; ++i) w[i] = load32_be(p + 4 * i);
    for (int i = 16; i < 64; ++i) {
        uint32_t s0 = ror32(w[i - 15], 7) ^ ror32(w[i - 15], 18) ^ (w[i - 15] >> 3);
        uint32_t s1 = ror32(w[i - 2], 17) ^ ror32(w[i - 2], 19) ^ (w[i - 2] >> 10);
        w[i] = w[i - 16] + s0 + w[i - 7] + s1;
    }
    uint32_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4], f = h[5], g = h[6], hh = h[7];
    // 8-round unroll with role rotation: removes the 8 register shuffles
    // per round of the rotating-variable loop
#define SHA256_RND(A, B, C, D, E, F, G, H, i)                                \
    {                                                                        \
        uint32_t S1 = ror32(E, 6) ^ ror32(E, 11) ^ ror32(E, 25);             \
        uint32_t t1 = H + S1 + ((E & F) ^ (~E & G)) + K256[i] + w[i];        \
        uint32_t S0 = ror32(A, 2) ^ ror32(A, 13) ^ ror32(A, 22);             \
        uint32_t t2 = S0 + ((A & B) ^ (A & C) ^ (B & C));                    \
        D += t1;                                                             \
        H = t1 + t2;                                                         \
    }
    for (int i = 0; i < 64; i += 8) {
        SHA256_RND(a, b, c, d, e, f, g, hh, i + 0);
        SHA256_RND(hh, a, b, c, d, e, f, g, i + 1);
        SHA256_RND(g, hh, a, b, c, d, e, f, i + 2);
        SHA256_RND(f, g, hh, a, b, c, d, e, i + 3);
        SHA256_RND(e, f, g, hh, a, b, c, d, i + 4);
        SHA256_RND(d, e, f, g, hh, a, b, c, i + 5);
        SHA256_RND(c, d, e, f, g, hh, a, b, i + 6);
        SHA256_RND(b, c, d, e, f, g, hh, a, i + 7);
    }
#undef SHA256_RND
    h[0] += a; h[1] += b; h[2] += c; h[3] += d;
    h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

Sha256::Sha256() {
    static const uint32_t init[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
                                     0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19};
    for (int i = 0; i < 8; ++i) h[i] = init[i];
}

void Sha256::update(const uint8_t* data, size_t len) {
    total += len;
    if (buflen) {
        size_t take = 64 - buflen;
        if (take > len) take = len;
        std::memcpy(buf + buflen, data, take);
        buflen += take;
        data += take;
        len -= take;
        if (buflen == 64) {
            sha256_block(h, buf);
            buflen = 0;
        }
    }
    while (len >= 64) {
        sha256_block(h, data);
        data += 64;
        len -= 64;
    }
    if (len) {
        std::memcpy(buf, data, len);
        buflen = len;
    }
}

void Sha256::final(uint8_t out[32]) {
    uint64_t bits = total * 8;
    uint8_t pad = 0x80;
    update(&pad, 1);
    uint8_t z = 0;
    while (buflen != 56) update(&z, 1);
    store64_be(buf + 56, bits);
    sha256_block(h, buf);
    for (int i = 0; i < 8; ++i) store32_be(out + 4 * i, h[i]);
}

std::array<uint8_t, 32> Sha256::hash(const uint8_t* data, size_t len) {
    Sha256 s;
    s.update(data, len);
    std::array<uint8_t, 32> out;
    s.final(out.data());
    return out;
}

// ---------------------------------------------------------------- SHA-512

static const uint64_t K512[80] = {
    0x428a2f98d728ae22ULL, 0x7137449123ef65cdULL, 0xb5c0fbcfec4d3b2fULL, 0xe9b5dba58189dbbcULL,
    0x3956c25bf348b538ULL, 0x59f111f1b605d019ULL, 0x923f82a4af194f9bULL, 0xab1c5ed5da6d8118ULL,
    0xd807aa98a3030242ULL, 0x12835b0145706fbeULL, 0x243185be4ee4b28cULL, 0x550c7dc3d5ffb4e2ULL,
    0x72be5d74f27b896fULL, 0x80deb1fe3b1696b1ULL, 0x9bdc06a725c71235ULL, 0xc19bf174cf692694ULL,
    0xe49b69c19ef14ad2ULL, 0xefbe4786384f25e3ULL, 0x0fc19dc68b8cd5b5ULL, 0x240ca1cc77ac9c65ULL,
    0x2de92c6f592b0275ULL, 0x4a7484aa6ea6e483ULL, 0x5cb0a9dcbd41fbd4ULL, 0x76f988da831153b5ULL,
    0x983e5152ee66dfabULL, 0xa831c66d2db43210ULL, 0xb00327c898fb213fULL, 0xbf597fc7beef0ee4ULL,
    0xc6e00bf33da88fc2ULL, 0xd5a79147930aa725ULL, 0x06ca6351e003826fULL, 0x142929670a0e6e70ULL,
    0x27b70a8546d22ffcULL, 0x2e1b21385c26c926ULL, 0x4d2c6dfc5ac42aedULL, 0x53380d139d95b3dfULL,
    0x650a73548baf63deULL, 0x766a0abb3c77b2a8ULL, 0x81c2c92e47edaee6ULL, 0x92722c851482353bULL,
    0xa2bfe8a14cf10364ULL, 0xa81a664bbc423001ULL, 0xc24b8b70d0f89791ULL, 0xc76c51a30654be30ULL,
    0xd192e819d6ef5218ULL, 0xd69906245565a910ULL, 0xf40e35855771202aULL, 0x106aa07032bbd1b8ULL,
    0x19a4c116b8d2d0c8ULL, 0x1e376c085141ab53ULL, 0x2748774cdf8eeb99ULL, 0x34b0bcb5e19b48a8ULL,
    0x391c0cb3c5c95a63ULL, 0x4ed8aa4ae3418acbULL, 0x5b9cca4f7763e373ULL, 0x682e6ff3d6b2b8a3ULL,
    0x748f82ee5defb2fcULL, 0x78a5636f43172f60ULL, 0x84c87814a1f0ab72ULL, 0x8cc702081a6439ecULL,
    0x90befffa23631e28ULL, 0xa4506cebde82bde9ULL, 0xbef9a3f7b2c67915ULL, 0xc67178f2e372532bULL,
    0xca273eceea26619cULL, 0xd186b8c721c0c207ULL, 0xeada7dd6cde0eb1eULL, 0xf57d4f7fee6ed178ULL,
    0x06f067aa72176fbaULL, 0x0a637dc5a2c898a6ULL, 0x113f9804bef90daeULL, 0x1b710b35131c471bULL,
    0x28db77f523047d84ULL, 0x32caab7b40c72493ULL, 0x3c9ebe0a15c9bebcULL, 0x431d67c49c100d4cULL,
    0x4cc5d4becb3e42b6ULL, 0x597f299cfc657e2aULL, 0x5fcb6fab3ad6faecULL, 0x6c44198c4a475817ULL,
};

static void sha512_block(uint64_t h[8], const uint8_t* p) {
    // separate schedule pass + unrolled rounds measured fastest here (a
    // fused 16-word ring raises live registers past x86's 16 and spills)
    uint64_t w[80];
    for (int i = 0; i < 16; ++i) w[i] = load64_be(p + 8 * i);
    for (int i = 16; i < 80; ++i) {
        uint64_t s0 = rotr64(w[i - 15], 1) ^ rotr64(w[i - 15], 8) ^ (w[i - 15] >> 7);
        uint64_t s1 = rotr64(w[i - 2], 19) ^ rotr64(w[i - 2], 61) ^ (w[i - 2] >> 6);
        w[i] = w[i - 16] + s0 + w[i - 7] + s1;
    }
    uint64_t a = h[0], b = h[1], c = h[2], d = h[3], e = h[4], f = h[5], g = h[6], hh = h[7];
    // 8-round unroll with role rotation (Ed25519 verify SHA-512s the whole
    // message body — at 175 MB updates this is the single largest ingest
    // CPU cost, so the compression loop matters)
#define SHA512_RND(A, B, C, D, E, F, G, H, i)                                \
    {                                                                        \
        uint64_t S1 = rotr64(E, 14) ^ rotr64(E, 18) ^ rotr64(E, 41);         \
        uint64_t t1 = H + S1 + ((E & F) ^ (~E & G)) + K512[i] + w[i];        \
        uint64_t S0 = rotr64(A, 28) ^ rotr64(A, 34) ^ rotr64(A, 39);         \
        uint64_t t2 = S0 + ((A & B) ^ (A & C) ^ (B & C));                    \
        D += t1;                                                             \
        H = t1 + t2;                                                         \
    }
    for (int i = 0; i < 80; i += 8) {
        SHA512_RND(a, b, c, d, e, f, g, hh, i + 0);
        SHA512_RND(hh, a, b, c, d, e, f, g, i + 1);
        SHA512_RND(g, hh, a, b, c, d, e, f, i + 2);
        SHA512_RND(f, g, hh, a, b, c, d, e, i + 3);
        SHA512_RND(e, f, g, hh, a, b, c, d, i + 4);
        SHA512_RND(d, e, f, g, hh, a, b, c, i + 5);
        SHA512_RND(c, d, e, f, g, hh, a, b, i + 6);
        SHA512_RND(b, c, d, e, f, g, hh, a, i + 7);
    }
#undef SHA512_RND
    h[0] += a; h[1] += b; h[2] += c; h[3] += d;
    h[4] += e; h[5] += f; h[6] += g; h[7] += hh;
}

Sha512::Sha512() {
    static const uint64_t init[8] = {0x6a09e667f3bcc908ULL, 0xbb67ae8584caa73bULL, 0x3c6ef372fe94f82bULL,
                                     0xa54ff53a5f1d36f1ULL, 0x510e527fade682d1ULL, 0x9b05688c2b3e6c1fULL,
                                     0x1f83d9abfb41bd6bULL, 0x5be0cd19137e2179ULL};
    for (int i = 0; i < 8; ++i) h[i] = init[i];
}

void Sha512::update(const uint8_t* data, size_t len) {
    total += len;
    if (buflen) {
        size_t take = 128 - buflen;
        if (take > len) take = len;
        std::memcpy(buf + buflen, data, take);
        buflen += take;
        data += take;
        len -= take;
        if (buflen == 128) {
            sha512_block(h, buf);
            buflen = 0;
        }
    }
    while (len >= 128) {
        sha512_block(h, data);
        data += 128;
        len -= 128;
    }
    if (len) {
        std::memcpy(buf, data, len);
        buflen = len;
    }
}

void Sha512::final(uint8_t out[64]) {
    uint64_t bits = total * 8;
    uint8_t pad = 0x80;
    update(&pad, 1);
    uint8_t z = 0;
    while (buflen != 112) update(&z, 1);
    // length is 128-bit big-endian; totals < 2^64 bits here
    std::memset(buf + 112, 0, 8);
    store64_be(buf + 120, bits);
    sha512_block(h, buf);
    for (int i = 0; i < 8; ++i) store64_be(out + 8 * i, h[i]);
}

std::array<uint8_t, 64> Sha512::hash(const uint8_t* data, size_t len) {
    Sha512 s;
    s.update(data, len);
    std::array<uint8_t, 64> out;
    s.final(out.data());
    return out;
}

}  // namespace xaynet::crypto
