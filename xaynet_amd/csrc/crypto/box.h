// libsodium-compatible crypto_box (X25519 + XSalsa20-Poly1305) and sealed
// boxes (crypto_box_seal / crypto_box_seal_open).
//
// PET usage (reference rust/xaynet-core/src/crypto/encrypt.rs):
//   - participants seal whole PET messages to the coordinator public key;
//   - update participants seal mask seeds to sum participants' ephemeral keys.
// SEALBYTES = 48 = 32 (ephemeral pk) + 16 (Poly1305 tag).
#pragma once

#include <cstdint>
#include <cstddef>

#include "../common.h"

namespace xaynet::crypto {

constexpr size_t BOX_PK_BYTES = 32;
constexpr size_t BOX_SK_BYTES = 32;
constexpr size_t BOX_SEED_BYTES = 32;
constexpr size_t BOX_NONCE_BYTES = 24;
constexpr size_t BOX_MAC_BYTES = 16;
constexpr size_t SEAL_BYTES = BOX_PK_BYTES + BOX_MAC_BYTES;  // 48

// pk = X25519 base mult of sk (sk used as-is; clamping inside the ladder).
void box_keypair(uint8_t pk[32], uint8_t sk[32]);                        // random sk
void box_seed_keypair(uint8_t pk[32], uint8_t sk[32], const uint8_t seed[32]);  // sha512(seed)[0:32]

// crypto_box_beforenm: shared key = HSalsa20(0, X25519(sk, pk))
void box_beforenm(uint8_t k[32], const uint8_t pk[32], const uint8_t sk[32]);

// secretbox: out = tag(16) || cipher(len)  (combined mode)
void secretbox_seal(uint8_t* out, const uint8_t* m, size_t len, const uint8_t n[24],
                    const uint8_t k[32]);
bool secretbox_open(uint8_t* m, const uint8_t* c, size_t clen, const uint8_t n[24],
                    const uint8_t k[32]);

// sealed box: out = epk(32) || tag(16) || cipher(len); nonce = blake2b24(epk||pk)
Bytes sealbox_seal(const uint8_t* m, size_t len, const uint8_t pk[32]);
// raw-destination variant: out must have room for len + SEAL_BYTES bytes
void sealbox_seal_into(uint8_t* out, const uint8_t* m, size_t len, const uint8_t pk[32]);
// raw-destination variant: out must have room for clen - SEAL_BYTES bytes
bool sealbox_open_into(uint8_t* out, const uint8_t* c, size_t clen, const uint8_t pk[32],
                       const uint8_t sk[32]);
bool sealbox_open(Bytes& out, const uint8_t* c, size_t clen, const uint8_t pk[32],
                  const uint8_t sk[32]);

void randombytes(uint8_t* out, size_t n);

}  // namespace xaynet::crypto
