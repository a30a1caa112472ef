// Salsa20 / HSalsa20 / XSalsa20 stream cipher (Bernstein), as used by the
// NaCl crypto_box / crypto_secretbox construction (libsodium-compatible).
#pragma once

#include <cstdint>
#include <cstddef>

namespace xaynet::crypto {

// HSalsa20: 32-byte key + 16-byte input -> 32-byte subkey.
void hsalsa20(uint8_t out[32], const uint8_t in[16], const uint8_t key[32]);

// XSalsa20 keystream XOR: c = m XOR stream(key, nonce24), starting at stream
// byte `ic*64`. m may equal c. If m == nullptr, writes the raw keystream.
void xsalsa20_xor(uint8_t* c, const uint8_t* m, size_t len, const uint8_t nonce[24],
                  const uint8_t key[32], uint64_t ic = 0);

}  // namespace xaynet::crypto
