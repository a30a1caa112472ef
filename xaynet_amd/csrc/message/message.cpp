#include "message.h"

#include <cstring>

#include "../crypto/curve25519.h"

namespace xaynet::msg {

// ------------------------------------------------------------ payloads

void SumPayload::serialize(uint8_t* out) const {
    std::memcpy(out, sum_signature.data(), 64);
    std::memcpy(out + 64, ephm_pk.data(), 32);
}

std::optional<SumPayload> SumPayload::deserialize(const uint8_t* p, size_t len) {
    if (len < 96) return std::nullopt;
    SumPayload s;
    std::memcpy(s.sum_signature.data(), p, 64);
    std::memcpy(s.ephm_pk.data(), p + 64, 32);
    return s;
}

void UpdatePayload::serialize(uint8_t* out) const {
    std::memcpy(out, sum_signature.data(), 64);
    std::memcpy(out + 64, update_signature.data(), 64);
    size_t off = 128;
    Bytes mo = masked.serialize();
    std::memcpy(out + off, mo.data(), mo.size());
    off += mo.size();
    // LengthValue: u32 BE length INCLUSIVE of the 4 length bytes
    store32_be(out + off, uint32_t(4 + SEED_ENTRY_LEN * local_seed_dict.size()));
    off += 4;
    for (const auto& e : local_seed_dict) {
        std::memcpy(out + off, e.pk.data(), 32);
        std::memcpy(out + off + 32, e.seed.data(), 80);
        off += SEED_ENTRY_LEN;
    }
}

std::optional<UpdatePayload> UpdatePayload::deserialize(const uint8_t* p, size_t len) {
    if (len < 128 + 4) return std::nullopt;
    UpdatePayload u;
    std::memcpy(u.sum_signature.data(), p, 64);
    std::memcpy(u.update_signature.data(), p + 64, 64);
    size_t consumed = 0;
    auto mo = mask::MaskObject::deserialize(p + 128, len - 128, &consumed);
    if (!mo) return std::nullopt;
    u.masked = std::move(*mo);
    size_t off = 128 + consumed;
    if (len < off + 4) return std::nullopt;
    uint32_t total = load32_be(p + off);
    if (total < 4 || (total - 4) % SEED_ENTRY_LEN != 0) return std::nullopt;
    if (len < off + total) return std::nullopt;
    size_t n = (total - 4) / SEED_ENTRY_LEN;
    off += 4;
    u.local_seed_dict.resize(n);
    for (size_t i = 0; i < n; ++i) {
        std::memcpy(u.local_seed_dict[i].pk.data(), p + off, 32);
        std::memcpy(u.local_seed_dict[i].seed.data(), p + off + 32, 80);
        off += SEED_ENTRY_LEN;
    }
    // reject duplicate keys (reference traits.rs: decode error)
    for (size_t i = 0; i < n; ++i)
        for (size_t j = i + 1; j < n; ++j)
            if (u.local_seed_dict[i].pk == u.local_seed_dict[j].pk) return std::nullopt;
    return u;
}

void Sum2Payload::serialize(uint8_t* out) const {
    std::memcpy(out, sum_signature.data(), 64);
    Bytes mo = mask.serialize();
    std::memcpy(out + 64, mo.data(), mo.size());
}

std::optional<Sum2Payload> Sum2Payload::deserialize(const uint8_t* p, size_t len) {
    if (len < 64) return std::nullopt;
    Sum2Payload s;
    std::memcpy(s.sum_signature.data(), p, 64);
    auto mo = mask::MaskObject::deserialize(p + 64, len - 64, nullptr);
    if (!mo) return std::nullopt;
    s.mask = std::move(*mo);
    return s;
}

void ChunkPayload::serialize(uint8_t* out) const {
    store32_be(out, (uint32_t(id) << 16) | message_id);
    out[4] = last ? 1 : 0;
    out[5] = out[6] = out[7] = 0;
    std::memcpy(out + 8, data.data(), data.size());
}

std::optional<ChunkPayload> ChunkPayload::deserialize(const uint8_t* p, size_t len) {
    if (len < CHUNK_OVERHEAD) return std::nullopt;
    ChunkPayload c;
    c.id = (uint16_t(p[0]) << 8) | p[1];
    c.message_id = (uint16_t(p[2]) << 8) | p[3];
    c.last = (p[4] & 1) != 0;
    c.data.assign(p + 8, p + len);
    return c;
}

// ------------------------------------------------------------ message

size_t Message::byte_len() const {
    size_t pl = std::visit([](const auto& p) { return p.byte_len(); }, payload);
    return HEADER_LEN + pl;
}

Bytes Message::to_bytes(const uint8_t sk64[64]) const {
    Bytes out(byte_len());
    uint8_t* p = out.data();
    std::memcpy(p + 64, participant_pk.data(), 32);
    std::memcpy(p + 96, coordinator_pk.data(), 32);
    store32_be(p + 128, uint32_t(out.size()));
    p[132] = uint8_t(tag);
    p[133] = is_multipart ? FLAG_MULTIPART : 0;
    p[134] = p[135] = 0;
    std::visit([&](const auto& pl) { pl.serialize(p + HEADER_LEN); }, payload);
    if (sk64) {
        crypto::ed25519_sign(p, p + 64, out.size() - 64, sk64);
    } else {
        std::memcpy(p, signature.data(), 64);
    }
    return out;
}

std::optional<Message> Message::from_bytes(const uint8_t* p, size_t len, bool verify) {
    if (len < HEADER_LEN) return std::nullopt;
    uint32_t length = load32_be(p + 128);
    if (length < HEADER_LEN || length > len) return std::nullopt;

    Message m;
    std::memcpy(m.signature.data(), p, 64);
    std::memcpy(m.participant_pk.data(), p + 64, 32);
    std::memcpy(m.coordinator_pk.data(), p + 96, 32);
    uint8_t tag = p[132];
    m.is_multipart = (p[133] & FLAG_MULTIPART) != 0;

    if (verify &&
        !crypto::ed25519_verify(p, p + 64, length - 64, m.participant_pk.data())) {
        return std::nullopt;
    }

    const uint8_t* pl = p + HEADER_LEN;
    size_t pl_len = length - HEADER_LEN;
    switch (tag) {
        case 1: m.tag = Tag::Sum; break;
        case 2: m.tag = Tag::Update; break;
        case 3: m.tag = Tag::Sum2; break;
        default: return std::nullopt;
    }
    if (m.is_multipart) {
        auto c = ChunkPayload::deserialize(pl, pl_len);
        if (!c) return std::nullopt;
        m.payload = std::move(*c);
        return m;
    }
    switch (m.tag) {
        case Tag::Sum: {
            auto s = SumPayload::deserialize(pl, pl_len);
            if (!s) return std::nullopt;
            m.payload = std::move(*s);
            break;
        }
        case Tag::Update: {
            auto u = UpdatePayload::deserialize(pl, pl_len);
            if (!u) return std::nullopt;
            m.payload = std::move(*u);
            break;
        }
        case Tag::Sum2: {
            auto s = Sum2Payload::deserialize(pl, pl_len);
            if (!s) return std::nullopt;
            m.payload = std::move(*s);
            break;
        }
    }
    return m;
}

// ------------------------------------------------------------ encoder

std::vector<Bytes> encode_message(const Message& m, const uint8_t sk64[64],
                                  size_t max_payload_size, uint16_t message_id) {
    size_t pl_len = std::visit([](const auto& p) { return p.byte_len(); }, m.payload);
    if (max_payload_size == 0 || pl_len <= max_payload_size) {
        return {m.to_bytes(sk64)};
    }
    // serialize the payload once, then chunk it
    Bytes data(pl_len);
    std::visit([&](const auto& p) { p.serialize(data.data()); }, m.payload);
    size_t chunk_data = max_payload_size - CHUNK_OVERHEAD;
    size_t n_chunks = (data.size() + chunk_data - 1) / chunk_data;
    std::vector<Bytes> out;
    out.reserve(n_chunks);
    for (size_t i = 0; i < n_chunks; ++i) {
        ChunkPayload c;
        c.id = uint16_t(i);
        c.message_id = message_id;
        c.last = (i == n_chunks - 1);
        size_t begin = i * chunk_data;
        size_t end = begin + chunk_data < data.size() ? begin + chunk_data : data.size();
        c.data.assign(data.begin() + begin, data.begin() + end);
        Message cm;
        cm.participant_pk = m.participant_pk;
        cm.coordinator_pk = m.coordinator_pk;
        cm.tag = m.tag;
        cm.is_multipart = true;
        cm.payload = std::move(c);
        out.push_back(cm.to_bytes(sk64));
    }
    return out;
}

}  // namespace xaynet::msg
