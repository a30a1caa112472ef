// PET wire messages — byte-compatible with the reference formats.
//
// Layouts (verified against rust/xaynet-core/src/message/):
//   header (136 B): sig(64) | participant_pk(32) | coordinator_pk(32) |
//                   length(u32 BE, whole message) | tag(1) | flags(1) | reserved(2)
//     signature covers bytes [64, length)  (message.rs:336-358)
//   sum payload:    task_signature(64) | ephemeral_pk(32)
//   update payload: sum_signature(64) | update_signature(64) | MaskObject |
//                   LocalSeedDict (u32 BE INCLUSIVE length, then 112-B entries
//                   of sum_pk(32) | encrypted_seed(80))
//   sum2 payload:   sum_signature(64) | MaskObject
//   chunk payload:  id(u16 BE) | message_id(u16 BE) | flags(1: LAST=1) |
//                   reserved(3) | data
#pragma once

#include <array>
#include <optional>
#include <variant>
#include <vector>

#include "../common.h"
#include "../mask/object.h"

namespace xaynet::msg {

using Key32 = std::array<uint8_t, 32>;
using Sig64 = std::array<uint8_t, 64>;
using EncrSeed80 = std::array<uint8_t, 80>;

enum class Tag : uint8_t { Sum = 1, Update = 2, Sum2 = 3 };

constexpr size_t HEADER_LEN = 136;
constexpr uint8_t FLAG_MULTIPART = 1;
constexpr size_t SEED_ENTRY_LEN = 112;  // 32 + 80
constexpr size_t CHUNK_OVERHEAD = 8;

struct SumPayload {
    Sig64 sum_signature;
    Key32 ephm_pk;
    size_t byte_len() const { return 96; }
    void serialize(uint8_t* out) const;
    static std::optional<SumPayload> deserialize(const uint8_t* p, size_t len);
};

struct LocalSeedEntry {
    Key32 pk;
    EncrSeed80 seed;
};

struct UpdatePayload {
    Sig64 sum_signature;
    Sig64 update_signature;
    mask::MaskObject masked;
    std::vector<LocalSeedEntry> local_seed_dict;
    size_t byte_len() const {
        return 128 + masked.byte_len() + 4 + SEED_ENTRY_LEN * local_seed_dict.size();
    }
    void serialize(uint8_t* out) const;
    static std::optional<UpdatePayload> deserialize(const uint8_t* p, size_t len);
};

struct Sum2Payload {
    Sig64 sum_signature;
    mask::MaskObject mask;
    size_t byte_len() const { return 64 + mask.byte_len(); }
    void serialize(uint8_t* out) const;
    static std::optional<Sum2Payload> deserialize(const uint8_t* p, size_t len);
};

struct ChunkPayload {
    uint16_t id = 0;
    uint16_t message_id = 0;
    bool last = false;
    Bytes data;
    size_t byte_len() const { return CHUNK_OVERHEAD + data.size(); }
    void serialize(uint8_t* out) const;
    static std::optional<ChunkPayload> deserialize(const uint8_t* p, size_t len);
};

using Payload = std::variant<SumPayload, UpdatePayload, Sum2Payload, ChunkPayload>;

struct Message {
    Sig64 signature{};           // filled by to_bytes / parsed
    Key32 participant_pk{};      // Ed25519 signing pk
    Key32 coordinator_pk{};      // coordinator X25519 pk
    Tag tag = Tag::Sum;
    bool is_multipart = false;
    Payload payload;

    size_t byte_len() const;
    // Serialize + sign with the participant's Ed25519 secret key (64 B,
    // libsodium layout seed||pk).
    Bytes to_bytes(const uint8_t sk64[64]) const;
    // Parse; if verify, checks the Ed25519 signature over [64, length).
    static std::optional<Message> from_bytes(const uint8_t* p, size_t len, bool verify = true);
};

// SDK-side encoder: sign, and split payloads larger than max_payload_size
// into multipart chunk messages (reference xaynet-sdk message_encoder).
std::vector<Bytes> encode_message(const Message& m, const uint8_t sk64[64],
                                  size_t max_payload_size, uint16_t message_id);

}  // namespace xaynet::msg
