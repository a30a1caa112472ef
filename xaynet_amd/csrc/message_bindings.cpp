// Low-level PET wire-message APIs: compose+sign(+chunk) a message from raw
// payload bytes, and parse headers — the tooling surface over
// message/message.cpp (reference xaynet-core/src/message/ +
// xaynet-sdk message_encoder).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "crypto/curve25519.h"
#include "message/message.h"

namespace py = pybind11;
using namespace xaynet;

static Bytes frompy_(py::bytes b) {
    std::string s = b;
    return Bytes(s.begin(), s.end());
}

static py::bytes pyb_(const Bytes& b) {
    return py::bytes(reinterpret_cast<const char*>(b.data()), b.size());
}

void bind_message(py::module_& m) {
    auto mm = m.def_submodule("message");
    mm.attr("HEADER_LEN") = msg::HEADER_LEN;
    mm.attr("CHUNK_OVERHEAD") = msg::CHUNK_OVERHEAD;
    mm.attr("FLAG_MULTIPART") = msg::FLAG_MULTIPART;
    mm.attr("TAG_SUM") = int(msg::Tag::Sum);
    mm.attr("TAG_UPDATE") = int(msg::Tag::Update);
    mm.attr("TAG_SUM2") = int(msg::Tag::Sum2);

    // Compose, sign and (when needed) chunk a message from raw payload bytes.
    // Returns the list of wire messages to POST (1 when it fits,
    // ceil(len/max_payload) chunked messages otherwise).
    mm.def(
        "encode",
        [](int tag, py::bytes payload, py::bytes sign_seed, py::bytes coordinator_pk,
           size_t max_payload, uint16_t message_id) {
            Bytes pl = frompy_(payload);
            Bytes seed = frompy_(sign_seed);
            Bytes cpk = frompy_(coordinator_pk);
            if (seed.size() != 32) throw std::runtime_error("sign_seed must be 32 bytes");
            if (cpk.size() != 32) throw std::runtime_error("coordinator_pk must be 32 bytes");
            msg::Message msgo;
            msgo.tag = msg::Tag(tag);
            switch (msg::Tag(tag)) {
                case msg::Tag::Sum: {
                    auto p = msg::SumPayload::deserialize(pl.data(), pl.size());
                    if (!p) throw std::runtime_error("invalid sum payload");
                    msgo.payload = *p;
                    break;
                }
                case msg::Tag::Update: {
                    auto p = msg::UpdatePayload::deserialize(pl.data(), pl.size());
                    if (!p) throw std::runtime_error("invalid update payload");
                    msgo.payload = std::move(*p);
                    break;
                }
                case msg::Tag::Sum2: {
                    auto p = msg::Sum2Payload::deserialize(pl.data(), pl.size());
                    if (!p) throw std::runtime_error("invalid sum2 payload");
                    msgo.payload = std::move(*p);
                    break;
                }
                default:
                    throw std::runtime_error("bad tag");
            }
            uint8_t pk[32], sk[64];
            crypto::ed25519_keypair_from_seed(pk, sk, seed.data());
            std::memcpy(msgo.participant_pk.data(), pk, 32);
            std::memcpy(msgo.coordinator_pk.data(), cpk.data(), 32);
            auto parts = msg::encode_message(msgo, sk, max_payload, message_id);
            py::list out;
            for (const auto& p : parts) out.append(pyb_(p));
            return out;
        },
        py::arg("tag"), py::arg("payload"), py::arg("sign_seed"), py::arg("coordinator_pk"),
        py::arg("max_payload") = 4096 - 136 - 48, py::arg("message_id") = 1);

    // Parse a wire message header (no signature verification).
    mm.def("parse_header", [](py::bytes data) {
        Bytes b = frompy_(data);
        if (b.size() < msg::HEADER_LEN) throw std::runtime_error("short message");
        py::dict d;
        d["signature"] = py::bytes(reinterpret_cast<const char*>(b.data()), 64);
        d["participant_pk"] = py::bytes(reinterpret_cast<const char*>(b.data() + 64), 32);
        d["coordinator_pk"] = py::bytes(reinterpret_cast<const char*>(b.data() + 96), 32);
        d["length"] = (uint32_t(b[128]) << 24) | (uint32_t(b[129]) << 16) |
                      (uint32_t(b[130]) << 8) | uint32_t(b[131]);
        d["tag"] = int(b[132]);
        d["flags"] = int(b[133]);
        return d;
    });

    // Verify a message's Ed25519 signature (bytes [64, length))
    mm.def("verify", [](py::bytes data) {
        Bytes b = frompy_(data);
        auto parsed = msg::Message::from_bytes(b.data(), b.size(), true);
        return parsed.has_value();
    });
}
