// The PET coordinator: 7-phase state machine, event bus, request plane and
// message-ingest pipeline.
//
// Architecture mirrors the reference's invariants
// (rust/xaynet-server/src/state_machine/): a single protocol thread owns the
// phase loop and the aggregation state; concurrent ingest threads talk to it
// ONLY through the request queue (mpsc analog) and observe it through the
// event bus (watch analog). Phase gating follows handler.rs: accept up to
// count.max during [0, time.min], then until count.min with
// timeout(time.max - time.min).
//
// The aggregation data plane is pluggable: `AggregationPlane::Cpu` uses the
// exact CPU oracle in-thread; `AggregationPlane::Staged` keeps validated
// masked updates as wire bytes for an external (MI355X GPU) engine that
// aggregates in digit planes — the Python layer drains staged updates and
// supplies the unmasked model at the Unmask phase.
#pragma once

#include <atomic>
#include <condition_variable>
#include <deque>
#include <future>
#include <variant>
#include <functional>
#include <memory>
#include <thread>

#include "../mask/masking.h"
#include "../message/bincode.h"
#include "../message/message.h"
#include "storage.h"

namespace xaynet::coord {

using bincode::RoundParameters;

struct CountParams {
    uint64_t min = 1, max = 100;
};
struct TimeParams {
    double min = 0, max = 3600;  // seconds (fractional allowed for tests)
};
struct PhaseParams {
    CountParams count;
    TimeParams time;
};

struct Settings {
    double sum_prob = 0.5;
    double update_prob = 0.9;
    PhaseParams sum{{1, 100}, {0, 5}};
    PhaseParams update{{3, 10000}, {0, 5}};
    PhaseParams sum2{{1, 100}, {0, 5}};
    mask::MaskConfigPair mask_cfg;
    size_t model_length = 4;
    bool restore = false;
    // multipart reassembly bounds (memory-exhaustion DoS guard): a chunk
    // whose buffering would exceed any of these is rejected, not buffered
    size_t multipart_max_entries = 4096;
    size_t multipart_max_per_pk_bytes = 1ull << 30;   // one in-flight big update
    size_t multipart_max_total_bytes = 4ull << 30;
    // how long the Unmask phase waits for the external (GPU) plane's model
    double unmask_timeout_s = 300.0;
};

enum class PhaseId : uint8_t { Idle = 0, Sum, Update, Sum2, Unmask, Failure, Shutdown };

enum class PipelineError {
    Ok = 0,
    Decrypt,
    Parsing,
    InvalidMessageSignature,
    InvalidCoordinatorPublicKey,
    UnexpectedMessage,
    NotSumEligible,
    NotUpdateEligible,
    MessageRejected,
    MessageDiscarded,
    AggregationFailed,
    Internal,
};

// ------------------------------------------------------------- event bus

struct Events {
    std::mutex mu;
    uint64_t round_id = 0;
    PhaseId phase = PhaseId::Idle;
    RoundParameters params;            // current round parameters
    Bytes params_bincode;              // cached GET /params body
    std::shared_ptr<SumDict> sum_dict;     // nullptr = invalidated
    std::shared_ptr<SeedDict> seed_dict;   // nullptr = invalidated
    std::shared_ptr<Bytes> model_bincode;  // cached Option<Model> body (Some)
    msg::Key32 keys_pk{};              // coordinator encrypt pk
    uint8_t keys_sk[32] = {};          // coordinator encrypt sk (for decryptor)
    uint64_t version = 0;              // bump on any change
};

// ------------------------------------------------------------- requests

struct SumRequest {
    msg::Key32 participant_pk, ephm_pk;
};
struct UpdateRequest {
    msg::Key32 participant_pk;
    std::vector<msg::LocalSeedEntry> local_seed_dict;
    mask::MaskObject masked;
    // the pure per-element checks already ran on the (parallel) ingest
    // thread; the protocol thread only re-checks round-state counts
    bool prevalidated = false;
};
struct Sum2Request {
    msg::Key32 participant_pk;
    mask::MaskObject mask;
};
using StateMachineRequest = std::variant<SumRequest, UpdateRequest, Sum2Request>;

enum class AggregationPlane { Cpu, Staged };

class Coordinator {
  public:
    Coordinator(Settings settings, std::shared_ptr<CoordinatorStorage> store,
                std::shared_ptr<ModelStorage> models,
                AggregationPlane plane = AggregationPlane::Cpu);
    ~Coordinator();

    // run the phase loop on a background thread
    void start();
    void stop();  // request shutdown and join

    // run exactly one phase (test / embedded driving)
    PhaseId run_one_phase();
    PhaseId phase() const { return phase_; }
    uint64_t round_id() const { return round_id_; }

    // ---- ingest (any thread) ----
    PipelineError handle_encrypted_message(const uint8_t* data, size_t len);
    // pre-decrypted/parsed path for in-process participants and tests
    PipelineError handle_message_bytes(const uint8_t* data, size_t len);

    // ---- fetchers (any thread; bincode bodies per the reference REST API) ----
    Bytes fetch_round_params();             // RoundParameters
    Bytes fetch_sum_dict();                 // Option<SumDict>
    Bytes fetch_seeds(const msg::Key32& sum_pk);  // Option<UpdateSeedDict>
    Bytes fetch_model();                    // Option<Model>
    uint64_t events_version();

    // typed accessors (in-process SDK)
    RoundParameters round_params_snapshot();
    std::shared_ptr<SumDict> sum_dict_snapshot();
    std::shared_ptr<SeedDict> seed_dict_snapshot();
    std::shared_ptr<Bytes> model_bincode_snapshot();

    // ---- staged GPU plane ----
    // drain staged masked-update wire bytes (Update phase); empty when none
    std::vector<Bytes> drain_staged_updates();
    // zero-copy variant: pop ONE staged update, writing its vector limbs
    // DIRECTLY into dst (an ingest-ring slot). Returns the limb byte count
    // (0 = none staged / doesn't fit) and fills unit_out with the masked
    // unit limbs. The single memcpy replaces the serialize->pybytes->slice->
    // ring chain (4 copies of a 175 MB update at 25M params).
    size_t pop_staged_vect(uint8_t* dst, size_t cap, Bytes& unit_out);
    size_t staged_count();
    // true when the Unmask phase is waiting for an external unmask result;
    // returns the winning aggregated-mask wire bytes + nb_models
    bool pending_unmask(Bytes& mask_bytes, uint64_t& nb_models);
    // supply the unmasked model (bincode Option<Model> payload = Some body)
    void supply_unmasked_model(const Bytes& model_bincode);

    // checkpoint (bincode CoordinatorState, reference-compatible layout).
    // NOT thread-safe vs a running phase loop: call from the protocol thread
    // (the Idle phase persists automatically) or while the loop is stopped —
    // the reference likewise snapshots only from the Idle phase
    // (phases/idle.rs:143-151).
    Bytes checkpoint_state();
    bool restore_state(const Bytes& state);

    const Settings& settings() const { return settings_; }

  private:
    PhaseId run_idle();
    PhaseId run_sum();
    PhaseId run_update();
    PhaseId run_sum2();
    PhaseId run_unmask();
    PhaseId run_failure();

    // phase gate: pull requests per PhaseParams; calls handler per request
    using Handler = std::function<PipelineError(StateMachineRequest&)>;
    bool process_requests(const PhaseParams& pp, const Handler& h);
    void purge_outdated_requests();

    PipelineError validate_task(const msg::Message& m);
    PipelineError enqueue_and_wait(StateMachineRequest req);

    Settings settings_;
    std::shared_ptr<CoordinatorStorage> store_;
    std::shared_ptr<ModelStorage> models_;
    AggregationPlane plane_;

    Events events_;
    std::atomic<PhaseId> phase_{PhaseId::Idle};
    std::atomic<uint64_t> round_id_{0};

    // round crypto state
    uint8_t encr_pk_[32] = {}, encr_sk_[32] = {};
    msg::Key32 round_seed_{};

    // request queue. span_id: the ingest span that produced this request —
    // the protocol-thread handling is parented to it (the reference threads
    // a tracing::Span through the mpsc tuple, requests.rs:120)
    struct Pending {
        StateMachineRequest req;
        std::shared_ptr<std::promise<PipelineError>> reply;
        uint64_t span_id = 0;
    };
    std::mutex qmu_;
    std::condition_variable qcv_;
    std::deque<Pending> queue_;
    std::atomic<bool> running_{false};
    std::atomic<bool> shutdown_{false};
    std::thread thread_;

    // aggregation state (protocol thread only)
    std::unique_ptr<mask::Aggregation> agg_;
    // staged plane: masked objects MOVED out of the request (no copy on the
    // protocol thread; wire serialization happens in the drain, on the GPU
    // driver's thread). Written by the protocol thread, drained by the
    // external GPU driver thread -> own lock (the request queue lock is NOT
    // held while handlers run).
    std::mutex staged_mu_;
    std::vector<mask::MaskObject> staged_;
    uint64_t staged_nb_models_ = 0;

    // staged unmask handoff
    std::mutex unmask_mu_;
    std::condition_variable unmask_cv_;
    bool unmask_pending_ = false;
    Bytes unmask_mask_bytes_;
    uint64_t unmask_nb_models_ = 0;
    std::optional<Bytes> unmask_result_;

    // multipart reassembly: keyed by (participant_pk, message_id). Chunks
    // may arrive in any order (reference MessageBuilder,
    // services/messages/multipart/service.rs:26-108): the LAST flag records
    // the expected count and the message completes once every id is present.
    struct MultipartEntry {
        std::map<uint16_t, Bytes> chunks;
        int32_t last_id = -1;  // id of the LAST-flagged chunk, -1 = unseen
        size_t bytes = 0;      // Σ chunk payload bytes buffered in this entry
    };
    std::mutex mp_mu_;
    std::map<std::pair<msg::Key32, uint16_t>, MultipartEntry> multipart_;
    size_t multipart_bytes_ = 0;  // global buffered bytes (under mp_mu_)
    // Σ buffered bytes across this pk's in-flight messages (mp_mu_ held)
    size_t multipart_pk_bytes(const msg::Key32& pk) const;
};

}  // namespace xaynet::coord
