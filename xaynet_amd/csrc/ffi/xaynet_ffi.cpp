// C ABI implementation over the native SDK participant + HTTP client
// (reference xaynet-mobile ffi). The save envelope matches the Python shim's
// (xaynet_sdk/xaynet_sdk.py): "XAYP" | sign_seed(32) | scalar num u64 LE |
// scalar den u64 LE | native participant state — states are interchangeable
// between the C FFI and the Python SDK.
#include "xaynet_ffi.h"

#include <cmath>
#include <cstring>
#include <memory>
#include <random>
#include <string>

#include "../crypto/curve25519.h"
#include "../mask/masking.h"
#include "../rest/rest.h"
#include "../sdk/participant.h"

using namespace xaynet;

struct XaynetFfiSettings {
    std::string url;
    double scalar = 1.0;
    uint8_t sign_seed[32] = {};
    bool has_keys = false;
};

struct XaynetFfiParticipant {
    std::unique_ptr<sdk::Participant> p;
    uint8_t sign_seed[32];
    uint64_t scalar_num, scalar_den;
    bool consumed = false;
};

// exact double -> rational (the reference converts f64 scalars exactly via
// Ratio::from_float); clamps denominators beyond u64 range
static bool scalar_to_fraction(double s, uint64_t& num, uint64_t& den) {
    if (!(s > 0.0 && s <= 1.0)) return false;
    int exp;
    double mant = std::frexp(s, &exp);  // s = mant * 2^exp, mant in [0.5, 1)
    uint64_t m = uint64_t(std::ldexp(mant, 53));  // 53-bit integer mantissa
    int shift = 53 - exp;                         // s = m / 2^shift
    while ((m & 1) == 0 && shift > 0) {
        m >>= 1;
        shift -= 1;
    }
    if (shift >= 64) {  // denominator exceeds u64: round to the closest fit
        m >>= (shift - 63);
        shift = 63;
        if (m == 0) m = 1;
    }
    num = m;
    den = 1ULL << shift;
    return true;
}

static bool parse_url(const std::string& url, std::string& host, uint16_t& port) {
    std::string rest = url;
    auto pos = rest.find("//");
    if (pos != std::string::npos) rest = rest.substr(pos + 2);
    pos = rest.find('/');
    if (pos != std::string::npos) rest = rest.substr(0, pos);
    pos = rest.rfind(':');
    if (pos == std::string::npos) {
        host = rest;
        port = 80;
    } else {
        host = rest.substr(0, pos);
        port = uint16_t(atoi(rest.c_str() + pos + 1));
    }
    return !host.empty();
}

extern "C" {

XaynetFfiSettings* xaynet_ffi_settings_new(void) { return new XaynetFfiSettings(); }

int xaynet_ffi_settings_destroy(XaynetFfiSettings* s) {
    if (!s) return XAYNET_FFI_ERR_NULLPTR;
    delete s;
    return XAYNET_FFI_OK;
}

int xaynet_ffi_settings_set_url(XaynetFfiSettings* s, const char* url) {
    if (!s || !url) return XAYNET_FFI_ERR_NULLPTR;
    s->url = url;
    return XAYNET_FFI_OK;
}

int xaynet_ffi_settings_set_scalar(XaynetFfiSettings* s, double scalar) {
    if (!s) return XAYNET_FFI_ERR_NULLPTR;
    uint64_t n, d;
    if (!scalar_to_fraction(scalar, n, d)) return XAYNET_FFI_ERR_INVALID;
    s->scalar = scalar;
    return XAYNET_FFI_OK;
}

int xaynet_ffi_settings_set_keys(XaynetFfiSettings* s, const XaynetFfiKeyPair* keys) {
    if (!s || !keys) return XAYNET_FFI_ERR_NULLPTR;
    std::memcpy(s->sign_seed, keys->secret, 32);
    s->has_keys = true;
    return XAYNET_FFI_OK;
}

int xaynet_ffi_check_settings(const XaynetFfiSettings* s) {
    if (!s) return XAYNET_FFI_ERR_NULLPTR;
    std::string host;
    uint16_t port;
    if (s->url.empty() || !parse_url(s->url, host, port)) return XAYNET_FFI_ERR_INVALID;
    if (!s->has_keys) return XAYNET_FFI_ERR_INVALID;
    return XAYNET_FFI_OK;
}

const XaynetFfiKeyPair* xaynet_ffi_generate_key_pair(void) {
    auto* kp = new XaynetFfiKeyPair();
    std::random_device rd;
    for (int i = 0; i < 32; i += 4) {
        uint32_t r = rd();
        std::memcpy(kp->secret + i, &r, 4);
    }
    uint8_t sk64[64];
    crypto::ed25519_keypair_from_seed(kp->public_, sk64, kp->secret);
    return kp;
}

int xaynet_ffi_forget_key_pair(const XaynetFfiKeyPair* kp) {
    if (!kp) return XAYNET_FFI_ERR_NULLPTR;
    delete kp;
    return XAYNET_FFI_OK;
}

static XaynetFfiParticipant* make_participant(const std::string& url, const uint8_t seed[32],
                                              double scalar, const Bytes* native_state) {
    std::string host;
    uint16_t port;
    if (!parse_url(url, host, port)) return nullptr;
    uint64_t num, den;
    if (!scalar_to_fraction(scalar, num, den)) return nullptr;

    sdk::PetSettings st;
    uint8_t pk[32];
    crypto::ed25519_keypair_from_seed(pk, st.sign_sk, seed);
    std::memcpy(st.sign_pk.data(), pk, 32);
    st.scalar = mask::Scalar(num, den);

    auto client = std::make_shared<rest::HttpXaynetClient>(host, port);
    std::unique_ptr<sdk::Participant> p;
    if (native_state) {
        p = sdk::Participant::restore(*native_state, client, st);
        if (!p) return nullptr;
    } else {
        p = std::make_unique<sdk::Participant>(st, client);
    }
    auto* h = new XaynetFfiParticipant();
    h->p = std::move(p);
    std::memcpy(h->sign_seed, seed, 32);
    h->scalar_num = num;
    h->scalar_den = den;
    return h;
}

XaynetFfiParticipant* xaynet_ffi_participant_new(const XaynetFfiSettings* s) {
    if (!s || xaynet_ffi_check_settings(s) != XAYNET_FFI_OK) return nullptr;
    return make_participant(s->url, s->sign_seed, s->scalar, nullptr);
}

int xaynet_ffi_participant_destroy(XaynetFfiParticipant* p) {
    if (!p) return XAYNET_FFI_ERR_NULLPTR;
    delete p;
    return XAYNET_FFI_OK;
}

int xaynet_ffi_participant_tick(XaynetFfiParticipant* h) {
    if (!h || h->consumed) return XAYNET_FFI_ERR_NULLPTR;
    h->p->tick();
    int flags = 0;
    switch (h->p->task()) {
        case sdk::Task::None: flags |= XAYNET_FFI_PARTICIPANT_TASK_NONE; break;
        case sdk::Task::Sum: flags |= XAYNET_FFI_PARTICIPANT_TASK_SUM; break;
        case sdk::Task::Update: flags |= XAYNET_FFI_PARTICIPANT_TASK_UPDATE; break;
    }
    if (h->p->should_set_model()) flags |= XAYNET_FFI_PARTICIPANT_SHOULD_SET_MODEL;
    if (h->p->made_progress()) flags |= XAYNET_FFI_PARTICIPANT_MADE_PROGRESS;
    if (h->p->new_global_model()) flags |= XAYNET_FFI_PARTICIPANT_NEW_GLOBALMODEL;
    return flags;
}

int xaynet_ffi_participant_set_model(XaynetFfiParticipant* h, const void* buffer,
                                     unsigned char data_type, unsigned int len) {
    if (!h || h->consumed || !buffer) return XAYNET_FFI_ERR_NULLPTR;
    size_t n = len;
    switch (data_type) {
        case XAYNET_FFI_DATATYPE_F32:
            h->p->set_model_f32(static_cast<const float*>(buffer), n);
            break;
        case XAYNET_FFI_DATATYPE_F64:
            h->p->set_model_f64(static_cast<const double*>(buffer), n);
            break;
        case XAYNET_FFI_DATATYPE_I32:
            h->p->set_model_i32(static_cast<const int32_t*>(buffer), n);
            break;
        case XAYNET_FFI_DATATYPE_I64:
            h->p->set_model_i64(static_cast<const int64_t*>(buffer), n);
            break;
        default:
            return XAYNET_FFI_ERR_SETMODEL_DATATYPE;
    }
    return XAYNET_FFI_OK;
}

int xaynet_ffi_participant_global_model(XaynetFfiParticipant* h, void* buffer,
                                        unsigned char data_type, unsigned int len) {
    if (!h || h->consumed || !buffer) return XAYNET_FFI_ERR_NULLPTR;
    auto body = h->p->global_model_bincode();
    if (!body) return XAYNET_FFI_GLOBALMODEL_NONE;
    auto m = bincode::decode_option_model(body->data(), body->size());
    if (!m) return XAYNET_FFI_ERR_GLOBALMODEL_IO;
    if (!*m) return XAYNET_FFI_GLOBALMODEL_NONE;
    const auto& model = **m;
    if (model.size() != size_t(len)) return XAYNET_FFI_ERR_GLOBALMODEL_LEN;
    switch (data_type) {
        case XAYNET_FFI_DATATYPE_F32: {
            auto v = mask::model_to_f32(model);
            std::memcpy(buffer, v.data(), v.size() * 4);
            break;
        }
        case XAYNET_FFI_DATATYPE_F64: {
            auto v = mask::model_to_f64(model);
            std::memcpy(buffer, v.data(), v.size() * 8);
            break;
        }
        case XAYNET_FFI_DATATYPE_I32: {
            auto v = mask::model_to_i32(model);
            std::memcpy(buffer, v.data(), v.size() * 4);
            break;
        }
        case XAYNET_FFI_DATATYPE_I64: {
            auto v = mask::model_to_i64(model);
            std::memcpy(buffer, v.data(), v.size() * 8);
            break;
        }
        default:
            return XAYNET_FFI_ERR_GLOBALMODEL_DATATYPE;
    }
    return XAYNET_FFI_OK;
}

int xaynet_ffi_participant_local_model_config(const XaynetFfiParticipant* h, int* data_type_out,
                                              uint64_t* len_out) {
    if (!h || !data_type_out || !len_out) return XAYNET_FFI_ERR_NULLPTR;
    *data_type_out = h->p->model_data_type();
    *len_out = h->p->model_length();
    return XAYNET_FFI_OK;
}

XaynetFfiByteBuffer* xaynet_ffi_participant_save(XaynetFfiParticipant* h) {
    if (!h || h->consumed) return nullptr;
    Bytes native = h->p->save();
    Bytes blob;
    blob.reserve(4 + 32 + 16 + native.size());
    const char magic[4] = {'X', 'A', 'Y', 'P'};
    blob.insert(blob.end(), magic, magic + 4);
    blob.insert(blob.end(), h->sign_seed, h->sign_seed + 32);
    for (uint64_t v : {h->scalar_num, h->scalar_den})
        for (int i = 0; i < 8; ++i) blob.push_back(uint8_t(v >> (8 * i)));
    blob.insert(blob.end(), native.begin(), native.end());
    h->consumed = true;
    auto* b = new XaynetFfiByteBuffer();
    b->len = blob.size();
    b->data = new uint8_t[blob.size()];
    std::memcpy(b->data, blob.data(), blob.size());
    return b;
}

XaynetFfiParticipant* xaynet_ffi_participant_restore(const char* url,
                                                     const XaynetFfiByteBuffer* state) {
    if (!url || !state || !state->data) return nullptr;
    if (state->len >= 4 && std::memcmp(state->data, "XAYP", 4) != 0) {
        // not the native envelope: treat as a reference-format
        // SerializableState bincode blob (xaynet-mobile save() output —
        // its first 4 bytes are a u32 variant index 0..7, never "XAYP")
        std::string host;
        uint16_t port;
        if (!parse_url(url, host, port)) return nullptr;
        auto client = std::make_shared<rest::HttpXaynetClient>(host, port);
        Bytes blob(state->data, state->data + state->len);
        auto p = sdk::Participant::restore_reference(blob, client);
        if (!p) return nullptr;
        auto* h = new XaynetFfiParticipant();
        std::memset(h->sign_seed, 0, 32);  // keys live inside the state
        h->scalar_num = 1;
        h->scalar_den = 1;
        h->p = std::move(p);
        return h;
    }
    if (state->len < 4 + 32 + 16) return nullptr;
    const uint8_t* seed = state->data + 4;
    uint64_t num = 0, den = 0;
    for (int i = 0; i < 8; ++i) num |= uint64_t(state->data[36 + i]) << (8 * i);
    for (int i = 0; i < 8; ++i) den |= uint64_t(state->data[44 + i]) << (8 * i);
    if (den == 0) return nullptr;
    Bytes native(state->data + 52, state->data + state->len);
    // reconstruct via make_participant but with the exact num/den
    std::string host;
    uint16_t port;
    if (!parse_url(url, host, port)) return nullptr;
    sdk::PetSettings st;
    uint8_t pk[32];
    crypto::ed25519_keypair_from_seed(pk, st.sign_sk, seed);
    std::memcpy(st.sign_pk.data(), pk, 32);
    st.scalar = mask::Scalar(num, den);
    auto client = std::make_shared<rest::HttpXaynetClient>(host, port);
    auto p = sdk::Participant::restore(native, client, st);
    if (!p) return nullptr;
    auto* h = new XaynetFfiParticipant();
    h->p = std::move(p);
    std::memcpy(h->sign_seed, seed, 32);
    h->scalar_num = num;
    h->scalar_den = den;
    return h;
}

XaynetFfiByteBuffer* xaynet_ffi_participant_save_reference(XaynetFfiParticipant* h) {
    // xaynet-mobile-compatible checkpoint: bincode SerializableState
    // (participant.rs:236-240); restorable by reference clients
    if (!h || h->consumed) return nullptr;
    Bytes blob = h->p->save_reference();
    h->consumed = true;
    auto* b = new XaynetFfiByteBuffer();
    b->len = blob.size();
    b->data = new uint8_t[blob.size()];
    std::memcpy(b->data, blob.data(), blob.size());
    return b;
}

int xaynet_ffi_byte_buffer_destroy(XaynetFfiByteBuffer* b) {
    if (!b) return XAYNET_FFI_ERR_NULLPTR;
    delete[] b->data;
    delete b;
    return XAYNET_FFI_OK;
}

}  // extern "C"
