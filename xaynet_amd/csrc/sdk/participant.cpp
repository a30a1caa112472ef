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

bool Participant::send_payload(msg::Tag tag, msg::Payload payload) {
    msg::Message m;
    m.participant_pk = settings_.sign_pk;
    m.coordinator_pk = round_.pk;
    m.tag = tag;
    m.payload = std::move(payload);
    auto parts = msg::encode_message(m, settings_.sign_sk, settings_.max_message_size,
                                     next_message_id_++);
    for (const Bytes& part : parts) {
        Bytes sealed = crypto::sealbox_seal(part.data(), part.size(), round_.pk.data());
        if (!client_->send_message(sealed)) return false;
    }
    return true;
}

void Participant::step_sum() {
    crypto::box_keypair(ephm_pk_, ephm_sk_);
    msg::SumPayload p;
    p.sum_signature = sum_signature_;
    std::memcpy(p.ephm_pk.data(), ephm_pk_, 32);
    if (send_payload(msg::Tag::Sum, p)) {
        phase_ = Phase::Sum2;
    } else {
        made_progress_ = false;
    }
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

    // mask the model with a fresh seed (typed fast masker; bit-identical to
    // the rational oracle — see mask_typed in mask/masking.cpp)
    uint8_t seed[32];
    crypto::randombytes(seed, 32);
    mask::MaskObject masked = std::visit(
        [&](const auto& v) {
            return mask_typed_dispatch(seed, settings_.scalar, v, round_.mask_config);
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

    if (send_payload(msg::Tag::Update, std::move(p))) {
        should_set_model_ = false;
        phase_ = Phase::Awaiting;
    } else {
        made_progress_ = false;
    }
}

void Participant::step_sum2() {
    auto seeds = client_->get_seeds(settings_.sign_pk);
    if (!seeds) {
        made_progress_ = false;
        return;
    }
    // NOTE: the seed dict keys sum participants by their SIGNING pk; seeds are
    // encrypted to our ephemeral keypair
    mask::Aggregation mask_agg(round_.mask_config, round_.model_length);
    size_t decrypted = 0;
    for (const auto& [update_pk, enc_seed] : *seeds) {
        Bytes seed;
        if (!crypto::sealbox_open(seed, enc_seed.data(), 80, ephm_pk_, ephm_sk_)) continue;
        if (seed.size() != 32) continue;
        mask::MaskObject m = mask::derive_mask(seed.data(), round_.model_length,
                                               round_.mask_config);
        mask_agg.aggregate(m);
        decrypted += 1;
    }
    if (decrypted == 0) {
        made_progress_ = false;
        return;
    }

    msg::Sum2Payload p;
    p.sum_signature = sum_signature_;
    p.mask = mask_agg.object();
    if (send_payload(msg::Tag::Sum2, std::move(p))) {
        phase_ = Phase::Awaiting;
    } else {
        made_progress_ = false;
    }
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

}  // namespace xaynet::sdk
