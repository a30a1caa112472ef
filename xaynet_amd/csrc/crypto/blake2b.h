// BLAKE2b (RFC 7693), unkeyed, variable digest length. Used for the sealed-box
// nonce: blake2b-24(ephemeral_pk || recipient_pk) (libsodium crypto_box_seal).
#pragma once

#include <cstdint>
#include <cstddef>

namespace xaynet::crypto {

void blake2b(uint8_t* out, size_t outlen, const uint8_t* in, size_t inlen);

}  // namespace xaynet::crypto
