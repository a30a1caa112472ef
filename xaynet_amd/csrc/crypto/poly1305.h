// Poly1305 one-time authenticator (RFC 8439), for crypto_secretbox.
#pragma once

#include <cstdint>
#include <cstddef>

namespace xaynet::crypto {

void poly1305_mac(uint8_t tag[16], const uint8_t* msg, size_t len, const uint8_t key[32]);

}  // namespace xaynet::crypto
