// Participant SDK: PET client state machine + tick-driven wrapper.
//
// Mirrors the reference xaynet-sdk state machine
// (rust/xaynet-sdk/src/state_machine/) and the xaynet-mobile Participant
// surface (tick / save / restore / set_model / task flags,
// rust/xaynet-mobile/src/participant.rs:30-260):
//   NewRound -> {Sum -> SendingSum -> Sum2 -> SendingSum2 | Update ->
//   SendingUpdate} -> Awaiting, with a round-freshness check (GET /params)
//   before every step.
//
// IO is abstracted behind XaynetClient: an in-process implementation talks
// directly to a Coordinator (tests, simulation at scale), and a transport
// bridge lets Python drive it over HTTP (bincode bodies decoded here).
#pragma once

#include <memory>
#include <optional>
#include <variant>

#include "../coordinator/coordinator.h"
#include "../mask/masking.h"
#include "../message/bincode.h"
#include "../message/message.h"

namespace xaynet::sdk {

using bincode::RoundParameters;
using bincode::SumDict;
using bincode::UpdateSeedDict;
using msg::Key32;

class XaynetClient {
  public:
    virtual ~XaynetClient() = default;
    virtual std::optional<RoundParameters> get_round_params() = 0;
    virtual std::optional<SumDict> get_sums() = 0;
    virtual std::optional<UpdateSeedDict> get_seeds(const Key32& pk) = 0;
    virtual std::optional<Bytes> get_model_bincode() = 0;  // Option<Model> body (Some only)
    virtual bool send_message(const Bytes& encrypted) = 0;
};

// in-process client bound to a Coordinator (no network)
class InProcessClient : public XaynetClient {
  public:
    explicit InProcessClient(std::shared_ptr<coord::Coordinator> c) : c_(std::move(c)) {}
    std::optional<RoundParameters> get_round_params() override {
        return c_->round_params_snapshot();
    }
    std::optional<SumDict> get_sums() override {
        auto sd = c_->sum_dict_snapshot();
        if (!sd) return std::nullopt;
        return *sd;
    }
    std::optional<UpdateSeedDict> get_seeds(const Key32& pk) override {
        Bytes b = c_->fetch_seeds(pk);
        auto d = bincode::decode_option_update_seed_dict(b.data(), b.size());
        if (!d || !*d) return std::nullopt;
        return **d;
    }
    std::optional<Bytes> get_model_bincode() override {
        auto m = c_->model_bincode_snapshot();
        if (!m) return std::nullopt;
        return *m;
    }
    bool send_message(const Bytes& encrypted) override {
        c_->handle_encrypted_message(encrypted.data(), encrypted.size());
        return true;  // POST /message always returns 200 (reference rest.rs)
    }

  private:
    std::shared_ptr<coord::Coordinator> c_;
};

enum class Task { None = 0, Sum = 1, Update = 2 };

struct PetSettings {
    uint8_t sign_sk[64];   // Ed25519 (libsodium layout)
    Key32 sign_pk;
    mask::Scalar scalar;
    size_t max_message_size = 4096 - 136 - 48;  // reference default
};

class Participant {
  public:
    Participant(const PetSettings& settings, std::shared_ptr<XaynetClient> client);

    // One state-machine transition (blocking on IO). Returns true if the
    // machine made progress (reference tick + made_progress).
    void tick();

    bool made_progress() const { return made_progress_; }
    bool should_set_model() const { return should_set_model_; }
    bool new_global_model() const { return new_global_model_; }
    Task task() const { return task_; }

    void set_model_f32(const float* w, size_t n);
    void set_model_f64(const double* w, size_t n);
    void set_model_i32(const int32_t* w, size_t n);
    void set_model_i64(const int64_t* w, size_t n);

    std::optional<Bytes> global_model_bincode();  // latest Option<Model> Some-body

    // save/restore (framework-native layout; see participant.cpp)
    Bytes save() const;
    static std::unique_ptr<Participant> restore(const Bytes& state,
                                                std::shared_ptr<XaynetClient> client,
                                                const PetSettings& settings);

    // Reference-compatible checkpoint: the xaynet-sdk `SerializableState`
    // bincode layout (rust/xaynet-sdk/src/state_machine/phase.rs:304-313,
    // xaynet-mobile participant.rs:236-240). Keys/settings/round params are
    // carried IN the state, as the reference does. Our inline-sending design
    // maps the reference's Sending* variants to their `next` state on
    // restore (the round-freshness check re-syncs a stale phase anyway).
    Bytes save_reference() const;
    static std::unique_ptr<Participant> restore_reference(
        const Bytes& state, std::shared_ptr<XaynetClient> client);

    // GPU-offload hooks: when set, the update task's
    // model masking and the sum2 task's mask aggregation run through the
    // accelerator (xaynet_amd.ops on an MI355X) instead of the CPU loops.
    // Each returns the full MaskObject wire bytes, or nullopt to fall back.
    // hooks receive the ROUND's mask config (vect,unit as (group,dtype,
    // bound,model) codes) + model length so one accelerator serves any round
    using MaskModelHook = std::function<std::optional<Bytes>(
        const uint8_t seed[32], int dtype, const void* data, size_t n,
        const std::array<int, 8>& cfg)>;
    using Sum2Hook = std::function<std::optional<Bytes>(
        const std::vector<std::array<uint8_t, 32>>& seeds, size_t length,
        const std::array<int, 8>& cfg)>;
    void set_mask_model_hook(MaskModelHook h) { mask_hook_ = std::move(h); }
    void set_sum2_hook(Sum2Hook h) { sum2_hook_ = std::move(h); }

    // introspection for tests
    int phase_id() const { return int(phase_); }
    const Key32& pk() const { return settings_.sign_pk; }

    // current round's model schema (reference local_model_config): data type
    // per mask::DataType, -1 / 0 before the first round params fetch
    int model_data_type() const { return has_round_ ? int(round_.mask_config.vect.dtype) : -1; }
    uint64_t model_length() const { return has_round_ ? round_.model_length : 0; }

  private:
    enum class Phase : uint8_t {
        NewRound = 0,
        Awaiting,
        Sum,
        Update,
        Sum2,
    };

    std::array<int, 8> cfg_codes() const;
    void check_round_freshness();
    bool resume_send();
    void begin_send(msg::Tag tag, msg::Payload payload, Phase next_phase);
    void step_new_round();
    void step_sum();
    void step_update();
    void step_sum2();

    PetSettings settings_;
    std::shared_ptr<XaynetClient> client_;

    Phase phase_ = Phase::NewRound;
    bool has_round_ = false;
    RoundParameters round_;
    Task task_ = Task::None;
    bool made_progress_ = true;
    bool should_set_model_ = false;
    bool new_global_model_ = false;
    uint64_t sent_model_version_ = 0;

    // round-scoped credentials
    msg::Sig64 sum_signature_{};
    msg::Sig64 update_signature_{};
    uint8_t ephm_pk_[32] = {}, ephm_sk_[32] = {};

    // model provided by the app for the update task — kept TYPED so the
    // update step uses the exact fast masker (mask_f32/...) instead of the
    // rational oracle (~100x slower per weight)
    using TypedModel = std::variant<std::vector<float>, std::vector<double>,
                                    std::vector<int32_t>, std::vector<int64_t>>;
    std::optional<TypedModel> local_model_;
    uint16_t next_message_id_ = 1;

    // resumable sending (reference sending.rs:23-120): a failed POST keeps
    // the remaining chunks and the next tick retries from the FAILED chunk
    // instead of recomposing the whole (possibly multi-hundred-MB) message
    struct PendingSend {
        std::vector<Bytes> parts;  // signed wire messages (unsealed)
        size_t next = 0;
        Phase next_phase = Phase::Awaiting;
    };
    std::optional<PendingSend> sending_;

    MaskModelHook mask_hook_;
    Sum2Hook sum2_hook_;
};

}  // namespace xaynet::sdk
