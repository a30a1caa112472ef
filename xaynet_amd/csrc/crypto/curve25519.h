// Curve25519 field arithmetic (radix 2^51), X25519 (RFC 7748) and Ed25519
// (RFC 8032), libsodium-compatible behavior.
//
// These back the PET crypto layer: X25519 sealed boxes for message/seed
// encryption (reference rust/xaynet-core/src/crypto/encrypt.rs) and Ed25519
// detached signatures for messages and task eligibility
// (reference rust/xaynet-core/src/crypto/sign.rs).
#pragma once

#include <cstdint>
#include <cstddef>

namespace xaynet::crypto {

// ---- X25519 ----
// out = scalar mult of `scalar` (clamped) with u-coordinate `point`.
void x25519(uint8_t out[32], const uint8_t scalar[32], const uint8_t point[32]);
// out = scalar * basepoint(9)
void x25519_base(uint8_t out[32], const uint8_t scalar[32]);

// ---- Ed25519 ----
// pk (32) from 32-byte seed; also writes the 64-byte "secret key" layout
// libsodium uses (seed || pk) if sk64 != nullptr.
void ed25519_keypair_from_seed(uint8_t pk[32], uint8_t sk64[64], const uint8_t seed[32]);
// detached signature (64 bytes). sk64 = seed || pk (libsodium layout).
void ed25519_sign(uint8_t sig[64], const uint8_t* msg, size_t len, const uint8_t sk64[64]);
bool ed25519_verify(const uint8_t sig[64], const uint8_t* msg, size_t len, const uint8_t pk[32]);
// Convert an Ed25519 secret/public key to X25519 (libsodium
// crypto_sign_ed25519_sk_to_curve25519 / pk_to_curve25519). Returns false if
// the public key fails to decompress.
void ed25519_sk_to_x25519(uint8_t x_sk[32], const uint8_t sk64[64]);
bool ed25519_pk_to_x25519(uint8_t x_pk[32], const uint8_t pk[32]);

}  // namespace xaynet::crypto
