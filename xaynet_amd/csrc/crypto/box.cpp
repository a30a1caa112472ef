#include "box.h"

#include <stdexcept>
#include <cstdio>

#include "blake2b.h"
#include "curve25519.h"
#include "poly1305.h"
#include "salsa.h"
#include "sha2.h"

namespace xaynet::crypto {

void randombytes(uint8_t* out, size_t n) {
    static thread_local FILE* urandom = nullptr;
    if (!urandom) {
        urandom = std::fopen("/dev/urandom", "rb");
        if (!urandom) throw std::runtime_error("cannot open /dev/urandom");
        setvbuf(urandom, nullptr, _IONBF, 0);
    }
    if (std::fread(out, 1, n, urandom) != n) throw std::runtime_error("urandom read failed");
}

void box_keypair(uint8_t pk[32], uint8_t sk[32]) {
    randombytes(sk, 32);
    x25519_base(pk, sk);
}

void box_seed_keypair(uint8_t pk[32], uint8_t sk[32], const uint8_t seed[32]) {
    auto h = Sha512::hash(seed, 32);
    std::memcpy(sk, h.data(), 32);
    x25519_base(pk, sk);
}

void box_beforenm(uint8_t k[32], const uint8_t pk[32], const uint8_t sk[32]) {
    uint8_t s[32];
    x25519(s, sk, pk);
    uint8_t zero16[16] = {0};
    hsalsa20(k, zero16, s);
}

void secretbox_seal(uint8_t* out, const uint8_t* m, size_t len, const uint8_t n[24],
                    const uint8_t k[32]) {
    // stream[0:32] = poly key; cipher = m XOR stream[32:]
    // NaCl layout: the first Salsa20 block covers poly key (32) + first 32
    // message bytes.
    uint8_t block0[64];
    xsalsa20_xor(block0, nullptr, 64, n, k, 0);

    uint8_t* tag = out;
    uint8_t* c = out + BOX_MAC_BYTES;
    size_t first = len < 32 ? len : 32;
    for (size_t i = 0; i < first; ++i) c[i] = m[i] ^ block0[32 + i];
    if (len > 32) xsalsa20_xor(c + 32, m + 32, len - 32, n, k, 1);

    poly1305_mac(tag, c, len, block0);
}

bool secretbox_open(uint8_t* m, const uint8_t* c, size_t clen, const uint8_t n[24],
                    const uint8_t k[32]) {
    if (clen < BOX_MAC_BYTES) return false;
    size_t len = clen - BOX_MAC_BYTES;
    const uint8_t* tag = c;
    const uint8_t* cm = c + BOX_MAC_BYTES;

    uint8_t block0[64];
    xsalsa20_xor(block0, nullptr, 64, n, k, 0);

    uint8_t expect[16];
    poly1305_mac(expect, cm, len, block0);
    if (!ct_equal(expect, tag, 16)) return false;

    size_t first = len < 32 ? len : 32;
    for (size_t i = 0; i < first; ++i) m[i] = cm[i] ^ block0[32 + i];
    if (len > 32) xsalsa20_xor(m + 32, cm + 32, len - 32, n, k, 1);
    return true;
}

static void seal_nonce(uint8_t nonce[24], const uint8_t epk[32], const uint8_t rpk[32]) {
    uint8_t buf[64];
    std::memcpy(buf, epk, 32);
    std::memcpy(buf + 32, rpk, 32);
    blake2b(nonce, 24, buf, 64);
}

Bytes sealbox_seal(const uint8_t* m, size_t len, const uint8_t pk[32]) {
    uint8_t epk[32], esk[32];
    box_keypair(epk, esk);

    uint8_t nonce[24];
    seal_nonce(nonce, epk, pk);

    uint8_t k[32];
    box_beforenm(k, pk, esk);

    Bytes out(SEAL_BYTES + len);
    std::memcpy(out.data(), epk, 32);
    secretbox_seal(out.data() + 32, m, len, nonce, k);
    return out;
}

void sealbox_seal_into(uint8_t* out, const uint8_t* m, size_t len, const uint8_t pk[32]) {
    uint8_t epk[32], esk[32];
    box_keypair(epk, esk);
    uint8_t nonce[24];
    seal_nonce(nonce, epk, pk);
    uint8_t k[32];
    box_beforenm(k, pk, esk);
    std::memcpy(out, epk, 32);
    secretbox_seal(out + 32, m, len, nonce, k);
}

bool sealbox_open(Bytes& out, const uint8_t* c, size_t clen, const uint8_t pk[32],
                  const uint8_t sk[32]) {
    if (clen < SEAL_BYTES) return false;
    const uint8_t* epk = c;

    uint8_t nonce[24];
    seal_nonce(nonce, epk, pk);

    uint8_t k[32];
    box_beforenm(k, epk, sk);

    out.resize(clen - SEAL_BYTES);
    return secretbox_open(out.data(), c + 32, clen - 32, nonce, k);
}

bool sealbox_open_into(uint8_t* out, const uint8_t* c, size_t clen, const uint8_t pk[32],
                       const uint8_t sk[32]) {
    if (clen < SEAL_BYTES) return false;
    const uint8_t* epk = c;
    uint8_t nonce[24];
    seal_nonce(nonce, epk, pk);
    uint8_t k[32];
    box_beforenm(k, epk, sk);
    return secretbox_open(out, c + 32, clen - 32, nonce, k);
}

}  // namespace xaynet::crypto
