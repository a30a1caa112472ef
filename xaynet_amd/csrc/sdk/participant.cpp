#include "participant.h"

#include "../crypto/box.h"
#include "../crypto/curve25519.h"
#include "../crypto/sha2.h"

namespace xaynet::sdk {

using crypto::Sha256;

static bool is_eligible(const uint8_t sig[64], double threshold) {
    if (threshold < 0.0) return false;
    if (threshold > 1.0) return true;
    auto h = Sha256::hash(sig, 64);
    BigUint numer = BigUint::from_bytes_le(h.data(), 32);
    Bytes ff(32, 0xff);
    BigUint denom = BigUint::from_bytes_le(ff.data(), 32);
    Rational lhs(BigInt(numer, false), denom);
    return Rational::cmp(lhs, Rational::from_double(threshold)) <= 0;
}

Participant::Participant(const PetSettings& settings, std::shared_ptr<XaynetClient> client)
    : settings_(settings), client_(std::move(client)) {}

void Participant::tick() {
    made_progress_ = true;
    check_round_freshness();
    if (sending_) {  // finish (or keep retrying) an in-flight send first
        resume_send();
        return;
    }
    switch (phase_) {
        case Phase::NewRound: step_new_round(); break;
        case Phase::Awaiting: made_progress_ = false; break;
        case Phase::Sum: step_sum(); break;
        case Phase::Update: step_update(); break;
        case Phase::Sum2: step_sum2(); break;
    }
}

void Participant::check_round_freshness() {
    auto params = client_->get_round_params();
    if (!params) {
        made_progress_ = false;
        return;
    }
    if (!has_round_ || params->seed != round_.seed) {
        round_ = *params;
        has_round_ = true;
        phase_ = Phase::NewRound;
        task_ = Task::None;
        should_set_model_ = false;
        // a new round implies the previous round published a global model
        // (reference xaynet-mobile participant.rs:282-285)
        new_global_model_ = true;
        local_model_.reset();
        sending_.reset();  // chunks sealed to the old round key are dead
    } else {
        round_ = *params;  // refresh pk etc. (same round)
    }
}

void Participant::step_new_round() {
    if (!has_round_) {
        made_progress_ = false;
        return;
    }
    Bytes seed_sum(round_.seed.begin(), round_.seed.end());
    Bytes seed_update = seed_sum;
    seed_sum.insert(seed_sum.end(), {'s', 'u', 'm'});
    seed_update.insert(seed_update.end(), {'u', 'p', 'd', 'a', 't', 'e'});
    crypto::ed25519_sign(sum_signature_.data(), seed_sum.data(), seed_sum.size(),
                         settings_.sign_sk);
    crypto::ed25519_sign(update_signature_.data(), seed_update.data(), seed_update.size(),
                         settings_.sign_sk);

    if (is_eligible(sum_signature_.data(), round_.sum)) {
        task_ = Task::Sum;
        phase_ = Phase::Sum;
    } else if (is_eligible(update_signature_.data(), round_.update)) {
        task_ = Task::Update;
        should_set_model_ = true;
        phase_ = Phase::Update;
    } else {
        task_ = Task::None;
        phase_ = Phase::Awaiting;
    }
}

std::array<int, 8> Participant::cfg_codes() const {
    const auto& v = round_.mask_config.vect;
    const auto& u = round_.mask_config.unit;
    return {int(v.group), int(v.dtype), int(v.bound), int(v.model),
            int(u.group), int(u.dtype), int(u.bound), int(u.model)};
}

void Participant::begin_send(msg::Tag tag, msg::Payload payload, Phase next_phase) {
    msg::Message m;
    m.participant_pk = settings_.sign_pk;
    m.coordinator_pk = round_.pk;
    m.tag = tag;
    m.payload = std::move(payload);
    PendingSend ps;
    ps.parts = msg::encode_message(m, settings_.sign_sk, settings_.max_message_size,
                                   next_message_id_++);
    ps.next_phase = next_phase;
    sending_ = std::move(ps);
    resume_send();
}

bool Participant::resume_send() {
    if (!sending_) return true;
    while (sending_->next < sending_->parts.size()) {
        const Bytes& part = sending_->parts[sending_->next];
        Bytes sealed = crypto::sealbox_seal(part.data(), part.size(), round_.pk.data());
        if (!client_->send_message(sealed)) {
            // retry THIS chunk next tick (reference sending.rs failed-chunk
            // slot); everything already sent stays sent
            made_progress_ = false;
            return false;
        }
        sending_->next += 1;
    }
    phase_ = sending_->next_phase;
    sending_.reset();
    return true;
}

void Participant::step_sum() {
    crypto::box_keypair(ephm_pk_, ephm_sk_);
    msg::SumPayload p;
    p.sum_signature = sum_signature_;
    std::memcpy(p.ephm_pk.data(), ephm_pk_, 32);
    begin_send(msg::Tag::Sum, p, Phase::Sum2);
}

static mask::MaskObject mask_typed_dispatch(const uint8_t seed[32], const mask::Scalar& sc,
                                            const std::vector<float>& v,
                                            const mask::MaskConfigPair& cfg) {
    return mask::mask_f32(seed, sc, v.data(), v.size(), cfg);
}
static mask::MaskObject mask_typed_dispatch(const uint8_t seed[32], const mask::Scalar& sc,
                                            const std::vector<double>& v,
                                            const mask::MaskConfigPair& cfg) {
    return mask::mask_f64(seed, sc, v.data(), v.size(), cfg);
}
static mask::MaskObject mask_typed_dispatch(const uint8_t seed[32], const mask::Scalar& sc,
                                            const std::vector<int32_t>& v,
                                            const mask::MaskConfigPair& cfg) {
    return mask::mask_i32(seed, sc, v.data(), v.size(), cfg);
}
static mask::MaskObject mask_typed_dispatch(const uint8_t seed[32], const mask::Scalar& sc,
                                            const std::vector<int64_t>& v,
                                            const mask::MaskConfigPair& cfg) {
    return mask::mask_i64(seed, sc, v.data(), v.size(), cfg);
}

void Participant::step_update() {
    if (!local_model_) {
        // waiting for the application to provide the model
        should_set_model_ = true;
        made_progress_ = false;
        return;
    }
    auto sums = client_->get_sums();
    if (!sums || sums->empty()) {
        made_progress_ = false;
        return;
    }
    size_t model_n = std::visit([](const auto& v) { return v.size(); }, *local_model_);
    if (model_n != round_.model_length) {
        // wrong length: nothing sensible to send this round
        made_progress_ = false;
        return;
    }

    // mask the model with a fresh seed: accelerator hook (MI355X K1+K5w)
    // when wired, else the typed CPU fast masker (bit-identical to the
    // rational oracle — see mask_typed in mask/masking.cpp)
    uint8_t seed[32];
    crypto::randombytes(seed, 32);
    std::optional<mask::MaskObject> accel;
    if (mask_hook_) {
        auto cfg = cfg_codes();
        auto wire = std::visit(
            [&](const auto& v) -> std::optional<Bytes> {
                using T = typename std::decay_t<decltype(v)>::value_type;
                int dt = std::is_same_v<T, float>     ? 0
                         : std::is_same_v<T, double>  ? 1
                         : std::is_same_v<T, int32_t> ? 2
                                                      : 3;
                return mask_hook_(seed, dt, v.data(), v.size(), cfg);
            },
            *local_model_);
        if (wire) {
            auto mo = mask::MaskObject::deserialize(wire->data(), wire->size(), nullptr);
            if (mo && mo->vect.count == round_.model_length) accel = std::move(*mo);
        }
    }
    mask::MaskObject masked =
        accel ? std::move(*accel)
              : std::visit(
                    [&](const auto& v) {
                        return mask_typed_dispatch(seed, settings_.scalar, v,
                                                   round_.mask_config);
                    },
                    *local_model_);

    // encrypt the seed to every sum participant's ephemeral pk
    msg::UpdatePayload p;
    p.sum_signature = sum_signature_;
    p.update_signature = update_signature_;
    p.masked = std::move(masked);
    p.local_seed_dict.reserve(sums->size());
    for (const auto& [sum_pk, ephm_pk] : *sums) {
        Bytes enc = crypto::sealbox_seal(seed, 32, ephm_pk.data());
        msg::LocalSeedEntry e;
        e.pk = sum_pk;
        std::memcpy(e.seed.data(), enc.data(), 80);
        p.local_seed_dict.push_back(std::move(e));
    }

    should_set_model_ = false;  // the model is consumed into the message
    begin_send(msg::Tag::Update, std::move(p), Phase::Awaiting);
}

void Participant::step_sum2() {
    auto seeds = client_->get_seeds(settings_.sign_pk);
    if (!seeds) {
        made_progress_ = false;
        return;
    }
    // NOTE: the seed dict keys sum participants by their SIGNING pk; seeds are
    // encrypted to our ephemeral keypair
    std::vector<std::array<uint8_t, 32>> plain_seeds;
    plain_seeds.reserve(seeds->size());
    for (const auto& [update_pk, enc_seed] : *seeds) {
        Bytes seed;
        if (!crypto::sealbox_open(seed, enc_seed.data(), 80, ephm_pk_, ephm_sk_)) continue;
        if (seed.size() != 32) continue;
        std::array<uint8_t, 32> s32;
        std::memcpy(s32.data(), seed.data(), 32);
        plain_seeds.push_back(s32);
    }
    if (plain_seeds.empty()) {
        made_progress_ = false;
        return;
    }
    // derive + modularly aggregate one mask per updater — the hottest
    // client-side loop (reference sum2.rs:170-190). GPU hook: K1+K2.
    std::optional<mask::MaskObject> agg_mask;
    if (sum2_hook_) {
        if (auto wire = sum2_hook_(plain_seeds, round_.model_length, cfg_codes())) {
            auto mo = mask::MaskObject::deserialize(wire->data(), wire->size(), nullptr);
            if (mo && mo->vect.count == round_.model_length) agg_mask = std::move(*mo);
        }
    }
    if (!agg_mask) {
        mask::Aggregation mask_agg(round_.mask_config, round_.model_length);
        for (const auto& s32 : plain_seeds) {
            mask::MaskObject m =
                mask::derive_mask(s32.data(), round_.model_length, round_.mask_config);
            mask_agg.aggregate(m);
        }
        agg_mask = mask_agg.object();
    }

    msg::Sum2Payload p;
    p.sum_signature = sum_signature_;
    p.mask = std::move(*agg_mask);
    begin_send(msg::Tag::Sum2, std::move(p), Phase::Awaiting);
}

void Participant::set_model_f32(const float* w, size_t n) {
    local_model_ = std::vector<float>(w, w + n);
    should_set_model_ = false;
}
void Participant::set_model_f64(const double* w, size_t n) {
    local_model_ = std::vector<double>(w, w + n);
    should_set_model_ = false;
}
void Participant::set_model_i32(const int32_t* w, size_t n) {
    local_model_ = std::vector<int32_t>(w, w + n);
    should_set_model_ = false;
}
void Participant::set_model_i64(const int64_t* w, size_t n) {
    local_model_ = std::vector<int64_t>(w, w + n);
    should_set_model_ = false;
}

std::optional<Bytes> Participant::global_model_bincode() {
    auto m = client_->get_model_bincode();
    // fetch succeeded (even when None): the flag is consumed
    // (reference participant.rs:336-348)
    new_global_model_ = false;
    return m;
}

// save format (framework-native, versioned): v1 | phase | task | has_round |
// round_params bincode | sigs | ephm keys | message id
Bytes Participant::save() const {
    bincode::Writer w;
    w.u8(1);  // version
    w.u8(uint8_t(phase_));
    w.u8(uint8_t(task_));
    w.u8(has_round_ ? 1 : 0);
    Bytes rp = bincode::encode_round_parameters(round_);
    w.u64(rp.size());
    w.raw(rp.data(), rp.size());
    w.raw(sum_signature_.data(), 64);
    w.raw(update_signature_.data(), 64);
    w.raw(ephm_pk_, 32);
    w.raw(ephm_sk_, 32);
    w.u32(next_message_id_);
    return std::move(w.out);
}

std::unique_ptr<Participant> Participant::restore(const Bytes& state,
                                                  std::shared_ptr<XaynetClient> client,
                                                  const PetSettings& settings) {
    bincode::Reader r{state.data(), state.size()};
    if (r.u8() != 1) return nullptr;
    auto p = std::make_unique<Participant>(settings, std::move(client));
    p->phase_ = Phase(r.u8());
    p->task_ = Task(r.u8());
    p->has_round_ = r.u8() != 0;
    uint64_t n = r.u64();
    if (r.fail || r.off + n > state.size()) return nullptr;
    auto rp = bincode::decode_round_parameters(state.data() + r.off, n);
    if (!rp) return nullptr;
    p->round_ = *rp;
    r.off += n;
    r.raw(p->sum_signature_.data(), 64);
    r.raw(p->update_signature_.data(), 64);
    r.raw(p->ephm_pk_, 32);
    r.raw(p->ephm_sk_, 32);
    p->next_message_id_ = uint16_t(r.u32());
    if (r.fail) return nullptr;
    return p;
}

// ------------------------------------------------- reference checkpoint
//
// bincode(SerializableState) — rust/xaynet-sdk/src/state_machine/phase.rs:
//   enum SerializableState { NewRound(State<NewRound>)=0, Awaiting=1, Sum=2,
//     Update=3, Sum2=4, SendingSum=5, SendingUpdate=6, SendingSum2=7 }
//   State<P> = { private: P, shared: SharedState }  (field order as declared)
//   SharedState = { keys: SigningKeyPair{pk 32, sk 64},
//                   scalar: Ratio<BigUint>{numer, denom},
//                   message_size: Option<u64>,
//                   round_params: RoundParameters }
// serde dialect (one place to flip if hardware truth differs — PARITY.md):
// sodiumoxide key newtypes as RAW fixed bytes (the dialect pinned by this
// repo's RoundParameters codec), Signature via its custom &[u8] impl
// (u64 len + 64), Option as u8 0/1, Vec/HashMap with u64 counts, enums as
// u32 variant indices.

static void write_signature(bincode::Writer& w, const uint8_t sig[64]) {
    w.u64(64);
    w.raw(sig, 64);
}
static bool read_signature(bincode::Reader& r, uint8_t sig[64]) {
    if (r.u64() != 64) return false;
    return r.raw(sig, 64);
}

Bytes Participant::save_reference() const {
    bincode::Writer w;
    // variant index
    uint32_t variant;
    switch (phase_) {
        case Phase::NewRound: variant = 0; break;
        case Phase::Awaiting: variant = 1; break;
        case Phase::Sum: variant = 2; break;
        case Phase::Update: variant = 3; break;
        case Phase::Sum2: variant = 4; break;
        default: variant = 0; break;
    }
    w.u32(variant);
    // ---- private (phase) state ----
    switch (phase_) {
        case Phase::NewRound:
        case Phase::Awaiting:
            break;  // unit structs: zero bytes
        case Phase::Sum:
            // Sum { ephm_keys: EncryptKeyPair{pk,sk}, sum_signature }
            w.raw(ephm_pk_, 32);
            w.raw(ephm_sk_, 32);
            write_signature(w, sum_signature_.data());
            break;
        case Phase::Update:
            // Update { sum_signature, update_signature, sum_dict: None,
            //          seed_dict: None, model: None, mask: None } — our
            // update step is atomic, so mid-step options are always None
            write_signature(w, sum_signature_.data());
            write_signature(w, update_signature_.data());
            w.u8(0);
            w.u8(0);
            w.u8(0);
            w.u8(0);
            break;
        case Phase::Sum2:
            // Sum2 { ephm_keys, sum_signature, seed_dict: None,
            //        seeds: None, mask: None }
            w.raw(ephm_pk_, 32);
            w.raw(ephm_sk_, 32);
            write_signature(w, sum_signature_.data());
            w.u8(0);
            w.u8(0);
            w.u8(0);
            break;
    }
    // ---- shared state ----
    w.raw(settings_.sign_pk.data(), 32);
    w.raw(settings_.sign_sk, 64);
    bincode::write_biguint(w, settings_.scalar.numer);
    bincode::write_biguint(w, settings_.scalar.denom);
    // MaxMessageSize(Option<usize>): the reference stores the max MESSAGE
    // size; our setting is the max payload -> add back header + sealbox
    w.u8(1);
    w.u64(uint64_t(settings_.max_message_size) + 136 + 48);
    Bytes rp = bincode::encode_round_parameters(
        has_round_ ? round_ : RoundParameters{});
    w.raw(rp.data(), rp.size());
    return std::move(w.out);
}

std::unique_ptr<Participant> Participant::restore_reference(
    const Bytes& state, std::shared_ptr<XaynetClient> client) {
    bincode::Reader r{state.data(), state.size()};
    uint32_t variant = r.u32();
    if (r.fail || variant > 7) return nullptr;

    Phase phase;
    uint8_t ephm_pk[32] = {}, ephm_sk[32] = {};
    msg::Sig64 sum_sig{}, upd_sig{};

    auto skip_option_bytes = [&]() -> bool {  // Option<Vec<u8>>
        uint8_t tag = r.u8();
        if (tag == 0) return !r.fail;
        uint64_t n = r.u64();
        if (r.fail || !r.need(n)) return false;
        r.off += n;
        return true;
    };
    switch (variant) {
        case 0: phase = Phase::NewRound; break;
        case 1: phase = Phase::Awaiting; break;
        case 2: {
            phase = Phase::Sum;
            if (!r.raw(ephm_pk, 32) || !r.raw(ephm_sk, 32)) return nullptr;
            if (!read_signature(r, sum_sig.data())) return nullptr;
            break;
        }
        case 3: {
            phase = Phase::Update;
            if (!read_signature(r, sum_sig.data())) return nullptr;
            if (!read_signature(r, upd_sig.data())) return nullptr;
            // sum_dict / seed_dict / model / mask mid-step options: tolerate
            // None-only (a reference client parked between micro-steps with
            // Some() state restarts the phase from its fetch step here)
            for (int i = 0; i < 4; ++i) {
                if (r.u8() != 0) return nullptr;
            }
            break;
        }
        case 4: {
            phase = Phase::Sum2;
            if (!r.raw(ephm_pk, 32) || !r.raw(ephm_sk, 32)) return nullptr;
            if (!read_signature(r, sum_sig.data())) return nullptr;
            for (int i = 0; i < 3; ++i) {
                if (r.u8() != 0) return nullptr;
            }
            break;
        }
        case 5: {  // SendingSum { message, failed, next: Sum2 }
            uint32_t enc = r.u32();
            if (enc != 0) return nullptr;  // sum messages are never multipart
            if (!skip_option_bytes()) return nullptr;  // Simple payload
            if (!skip_option_bytes()) return nullptr;  // failed chunk
            phase = Phase::Sum2;
            if (!r.raw(ephm_pk, 32) || !r.raw(ephm_sk, 32)) return nullptr;
            if (!read_signature(r, sum_sig.data())) return nullptr;
            for (int i = 0; i < 3; ++i) {
                if (r.u8() != 0) return nullptr;
            }
            break;
        }
        case 6:    // SendingUpdate { .., next: Awaiting }
        case 7: {  // SendingSum2   { .., next: Awaiting }
            uint32_t enc = r.u32();
            if (enc == 0) {
                if (!skip_option_bytes()) return nullptr;
            } else if (enc == 1) {
                uint8_t k32[32], k64[64];
                if (!r.raw(k32, 32) || !r.raw(k64, 64) || !r.raw(k32, 32)) return nullptr;
                uint64_t n = r.u64();  // data
                if (r.fail || !r.need(n)) return nullptr;
                r.off += n;
                if (!r.need(2)) return nullptr;  // id: u16
                r.off += 2;
                r.u32();                          // tag enum
                r.u64();                          // payload_size
                if (!r.need(2)) return nullptr;   // message_id: u16
                r.off += 2;
            } else {
                return nullptr;
            }
            if (!skip_option_bytes()) return nullptr;  // failed
            phase = Phase::Awaiting;                   // next: Awaiting (unit)
            break;
        }
        default:
            return nullptr;
    }

    // ---- shared ----
    PetSettings st;
    if (!r.raw(st.sign_pk.data(), 32) || !r.raw(st.sign_sk, 64)) return nullptr;
    if (!bincode::read_biguint(r, st.scalar.numer)) return nullptr;
    if (!bincode::read_biguint(r, st.scalar.denom)) return nullptr;
    uint8_t has_size = r.u8();
    uint64_t msg_size = has_size ? r.u64() : 0;
    if (r.fail) return nullptr;
    st.max_message_size = has_size && msg_size > 136 + 48 ? size_t(msg_size - 136 - 48)
                                                          : size_t(1) << 40;
    auto rp = bincode::decode_round_parameters(state.data() + r.off, state.size() - r.off);
    if (!rp) return nullptr;

    auto p = std::make_unique<Participant>(st, std::move(client));
    p->phase_ = phase;
    p->round_ = *rp;
    p->has_round_ = rp->model_length > 0;
    p->sum_signature_ = sum_sig;
    p->update_signature_ = upd_sig;
    std::memcpy(p->ephm_pk_, ephm_pk, 32);
    std::memcpy(p->ephm_sk_, ephm_sk, 32);
    p->task_ = phase == Phase::Sum || phase == Phase::Sum2 ? Task::Sum
               : phase == Phase::Update                    ? Task::Update
                                                           : Task::None;
    return p;
}

}  // namespace xaynet::sdk
