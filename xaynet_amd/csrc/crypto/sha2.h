// SHA-256 / SHA-512 (FIPS 180-4).
//
// Used for: the PET eligibility hash and round-seed derivation (SHA-256,
// reference rust/xaynet-core/src/crypto/hash.rs) and Ed25519 (SHA-512).
#pragma once

#include <array>
#include <cstdint>
#include <cstddef>

namespace xaynet::crypto {

struct Sha256 {
    static constexpr size_t DIGEST = 32;
    uint32_t h[8];
    uint8_t buf[64];
    uint64_t total = 0;
    size_t buflen = 0;

    Sha256();
    void update(const uint8_t* data, size_t len);
    void final(uint8_t out[32]);
    static std::array<uint8_t, 32> hash(const uint8_t* data, size_t len);
};

struct Sha512 {
    static constexpr size_t DIGEST = 64;
    uint64_t h[8];
    uint8_t buf[128];
    uint64_t total = 0;
    size_t buflen = 0;

    Sha512();
    void update(const uint8_t* data, size_t len);
    void final(uint8_t out[64]);
    static std::array<uint8_t, 64> hash(const uint8_t* data, size_t len);
};

}  // namespace xaynet::crypto
