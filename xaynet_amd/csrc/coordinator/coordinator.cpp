#include "coordinator.h"

#include <chrono>
#include <future>

#include "../crypto/box.h"
#include "../crypto/curve25519.h"
#include "../crypto/sha2.h"
#include "metrics.h"
#include "trace.h"

namespace xaynet::coord {

using Clock = std::chrono::steady_clock;
using crypto::Sha256;

// Timed condition-variable waits normally use the steady clock
// (pthread_cond_clockwait). This toolchain's libtsan lacks that interceptor
// (verified: no __interceptor_pthread_cond_clockwait), which silently breaks
// every happens-before edge through the mutex and floods TSAN with false
// double-lock/race reports. Under TSAN only, route timed waits through the
// system clock so they hit the intercepted pthread_cond_timedwait.
template <class Pred>
static bool cv_wait_until(std::condition_variable& cv, std::unique_lock<std::mutex>& l,
                          Clock::time_point deadline, Pred pred) {
#if defined(__SANITIZE_THREAD__)
    auto sys = std::chrono::system_clock::now() +
               std::chrono::duration_cast<std::chrono::system_clock::duration>(
                   deadline - Clock::now());
    return cv.wait_until(l, sys, pred);
#else
    return cv.wait_until(l, deadline, pred);
#endif
}

Coordinator::Coordinator(Settings settings, std::shared_ptr<CoordinatorStorage> store,
                         std::shared_ptr<ModelStorage> models, AggregationPlane plane)
    : settings_(std::move(settings)), store_(std::move(store)), models_(std::move(models)),
      plane_(plane) {
    events_.params.sum = settings_.sum_prob;
    events_.params.update = settings_.update_prob;
    events_.params.mask_config = settings_.mask_cfg;
    events_.params.model_length = settings_.model_length;
    if (settings_.restore) {
        if (auto st = store_->coordinator_state()) {
            if (restore_state(*st)) {
                // reload the latest global model and re-broadcast it,
                // validating its length (reference initializer.rs:199-280)
                if (auto id = store_->latest_global_model_id()) {
                    if (auto body = models_->global_model(*id)) {
                        auto m = bincode::decode_option_model(body->data(), body->size());
                        if (m && *m && (*m)->size() == settings_.model_length) {
                            std::lock_guard<std::mutex> l(events_.mu);
                            events_.model_bincode = std::make_shared<Bytes>(std::move(*body));
                            events_.version += 1;
                        }
                    }
                }
            }
        }
    }
}

Coordinator::~Coordinator() { stop(); }

void Coordinator::start() {
    if (running_.exchange(true)) return;
    shutdown_ = false;
    thread_ = std::thread([this] {
        while (!shutdown_) {
            PhaseId next = run_one_phase();
            if (next == PhaseId::Shutdown) break;
        }
        running_ = false;
    });
}

void Coordinator::stop() {
    shutdown_ = true;
    qcv_.notify_all();
    unmask_cv_.notify_all();
    if (thread_.joinable()) thread_.join();
    running_ = false;
    purge_outdated_requests();  // release any ingest workers still waiting
}

static const char* phase_name(PhaseId p) {
    switch (p) {
        case PhaseId::Idle: return "idle";
        case PhaseId::Sum: return "sum";
        case PhaseId::Update: return "update";
        case PhaseId::Sum2: return "sum2";
        case PhaseId::Unmask: return "unmask";
        case PhaseId::Failure: return "failure";
        case PhaseId::Shutdown: return "shutdown";
    }
    return "?";
}

PhaseId Coordinator::run_one_phase() {
    PhaseId cur = phase_.load();
    // reference phase.rs:148: error_span!("run_phase") around each phase
    trace::Span span("run_phase", std::string("phase=") + phase_name(cur) +
                                      " round=" + std::to_string(round_id_.load()));
    PhaseId next;
    switch (cur) {
        case PhaseId::Idle: next = run_idle(); break;
        case PhaseId::Sum: next = run_sum(); break;
        case PhaseId::Update: next = run_update(); break;
        case PhaseId::Sum2: next = run_sum2(); break;
        case PhaseId::Unmask: next = run_unmask(); break;
        case PhaseId::Failure: next = run_failure(); break;
        case PhaseId::Shutdown: return PhaseId::Shutdown;
    }
    purge_outdated_requests();
    phase_ = next;
    {
        std::lock_guard<std::mutex> l(events_.mu);
        events_.phase = next;
        events_.version += 1;
    }
    // reference emit site: state_machine/mod.rs:180-219
    metrics::metric(metrics::Measurement::Phase, double(int(next)), round_id_, int(next));
    return next;
}

// --------------------------------------------------------------- phases

PhaseId Coordinator::run_idle() {
    round_id_ += 1;
    if (!store_->delete_dicts()) return PhaseId::Failure;
    {
        // drop incomplete multipart buffers from the previous round: the
        // fresh coordinator pk makes their remaining chunks unacceptable,
        // so they could never complete (eviction point; the reference's
        // builders die with the service state each round)
        std::lock_guard<std::mutex> l(mp_mu_);
        multipart_.clear();
        multipart_bytes_ = 0;
    }
    {
        // drop updates staged for the GPU by a failed previous round: the GPU
        // driver resets its accumulator on a round-id change, but anything
        // still queued here would otherwise be drained into the fresh round
        std::lock_guard<std::mutex> sl(staged_mu_);
        staged_.clear();
        staged_nb_models_ = 0;
    }

    // fresh round keys
    crypto::box_keypair(encr_pk_, encr_sk_);

    // round seed = sha256(sign(prev_seed || sum_le || update_le)) with an
    // Ed25519 keypair derived from the NEW encrypt secret key
    // (reference phases/idle.rs:84-102)
    uint8_t sign_pk[32], sign_sk[64];
    crypto::ed25519_keypair_from_seed(sign_pk, sign_sk, encr_sk_);
    Bytes msg;
    msg.insert(msg.end(), round_seed_.begin(), round_seed_.end());
    uint8_t fl[8];
    double sum_prob = settings_.sum_prob, upd_prob = settings_.update_prob;
    std::memcpy(fl, &sum_prob, 8);
    msg.insert(msg.end(), fl, fl + 8);
    std::memcpy(fl, &upd_prob, 8);
    msg.insert(msg.end(), fl, fl + 8);
    uint8_t sig[64];
    crypto::ed25519_sign(sig, msg.data(), msg.size(), sign_sk);
    auto h = Sha256::hash(sig, 64);
    std::memcpy(round_seed_.data(), h.data(), 32);

    {
        std::lock_guard<std::mutex> l(events_.mu);
        events_.round_id = round_id_;
        std::memcpy(events_.params.pk.data(), encr_pk_, 32);
        events_.params.seed = round_seed_;
        events_.params.sum = settings_.sum_prob;
        events_.params.update = settings_.update_prob;
        events_.params.mask_config = settings_.mask_cfg;
        events_.params.model_length = settings_.model_length;
        events_.params_bincode = bincode::encode_round_parameters(events_.params);
        std::memcpy(events_.keys_pk.data(), encr_pk_, 32);
        std::memcpy(events_.keys_sk, encr_sk_, 32);
        events_.sum_dict.reset();
        events_.seed_dict.reset();
        events_.version += 1;
    }

    if (!store_->set_coordinator_state(checkpoint_state())) return PhaseId::Failure;
    // reference emit site: phases/idle.rs:158-173
    metrics::metric(metrics::Measurement::RoundTotalNumber, double(round_id_), round_id_,
                    int(PhaseId::Idle));
    metrics::metric(metrics::Measurement::RoundParamSum, settings_.sum_prob, round_id_,
                    int(PhaseId::Idle));
    metrics::metric(metrics::Measurement::RoundParamUpdate, settings_.update_prob, round_id_,
                    int(PhaseId::Idle));
    return PhaseId::Sum;
}

PhaseId Coordinator::run_sum() {
    auto handler = [this](StateMachineRequest& req) -> PipelineError {
        const auto* s = std::get_if<SumRequest>(&req);
        if (!s) return PipelineError::MessageRejected;
        switch (store_->add_sum_participant(s->participant_pk, s->ephm_pk)) {
            case SumPartAddError::Ok: return PipelineError::Ok;
            default: return PipelineError::MessageRejected;
        }
    };
    if (!process_requests(settings_.sum, handler)) return PhaseId::Failure;

    auto sd = store_->sum_dict();
    if (!sd) return PhaseId::Failure;
    {
        std::lock_guard<std::mutex> l(events_.mu);
        events_.sum_dict = std::make_shared<SumDict>(std::move(*sd));
        events_.version += 1;
    }
    return PhaseId::Update;
}

PhaseId Coordinator::run_update() {
    agg_ = std::make_unique<mask::Aggregation>(settings_.mask_cfg, settings_.model_length);
    {
        std::lock_guard<std::mutex> sl(staged_mu_);
        staged_.clear();
        staged_nb_models_ = 0;
    }

    auto handler = [this](StateMachineRequest& req) -> PipelineError {
        auto* u = std::get_if<UpdateRequest>(&req);
        if (!u) return PipelineError::MessageRejected;
        // validate BEFORE the seed dict (reference update.rs:119-140). The
        // elementwise validity pass (175 MB read at 25M params) runs on the
        // ingest threads; only the round-state count caps are checked here
        mask::AggregationError ve = u->prevalidated
                                        ? agg_->validate_counts_only()
                                        : agg_->validate_aggregation(u->masked);
        if (ve != mask::AggregationError::Ok) return PipelineError::AggregationFailed;
        switch (store_->add_local_seed_dict(u->participant_pk, u->local_seed_dict)) {
            case SeedDictAddError::Ok: break;
            default: return PipelineError::MessageRejected;
        }
        if (plane_ == AggregationPlane::Cpu) {
            agg_->aggregate(u->masked);
        } else {
            // move, don't serialize: the request dies after this handler and
            // a 25M-param update is ~175 MB — the wire copy happens in the
            // drain, on the GPU driver's thread, off the serial protocol path
            std::lock_guard<std::mutex> sl(staged_mu_);
            staged_.push_back(std::move(u->masked));
            staged_nb_models_ += 1;
            // keep the CPU aggregation's unit/scalar bookkeeping consistent:
            // staged plane recomputes everything on the GPU, so agg_ only
            // tracks nb_models via set() at unmask time
        }
        return PipelineError::Ok;
    };
    if (!process_requests(settings_.update, handler)) return PhaseId::Failure;

    auto sd = store_->seed_dict();
    if (!sd) return PhaseId::Failure;
    {
        std::lock_guard<std::mutex> l(events_.mu);
        events_.seed_dict = std::make_shared<SeedDict>(std::move(*sd));
        events_.version += 1;
    }
    return PhaseId::Sum2;
}

PhaseId Coordinator::run_sum2() {
    auto handler = [this](StateMachineRequest& req) -> PipelineError {
        const auto* s = std::get_if<Sum2Request>(&req);
        if (!s) return PipelineError::MessageRejected;
        switch (store_->incr_mask_score(s->participant_pk, s->mask.serialize())) {
            case MaskScoreIncrError::Ok: return PipelineError::Ok;
            default: return PipelineError::MessageRejected;
        }
    };
    if (!process_requests(settings_.sum2, handler)) return PhaseId::Failure;
    return PhaseId::Unmask;
}

PhaseId Coordinator::run_unmask() {
    // reference emit site: phases/unmask.rs:137-157
    metrics::metric(metrics::Measurement::MasksTotalNumber,
                    double(store_->number_of_unique_masks()), round_id_, int(PhaseId::Unmask));
    auto best = store_->best_masks(2);
    if (best.empty()) return PhaseId::Failure;

    // unique max vote else AmbiguousMasks (reference unmask.rs:97-115)
    const Bytes* mask_bytes = nullptr;
    uint64_t best_count = 0;
    bool ambiguous = false;
    for (const auto& [mb, count] : best) {
        if (count > best_count) {
            best_count = count;
            mask_bytes = &mb;
            ambiguous = false;
        } else if (count == best_count) {
            ambiguous = true;
        }
    }
    if (!mask_bytes || ambiguous) return PhaseId::Failure;

    Bytes model_bincode;
    if (plane_ == AggregationPlane::Cpu) {
        auto mo = mask::MaskObject::deserialize(mask_bytes->data(), mask_bytes->size(), nullptr);
        if (!mo) return PhaseId::Failure;
        if (agg_->validate_unmasking(*mo) != mask::UnmaskingError::Ok) return PhaseId::Failure;
        auto model = agg_->unmask(*mo);
        model_bincode = bincode::encode_option_model(&model);
    } else {
        // hand off to the external (GPU) plane
        {
            std::lock_guard<std::mutex> l(unmask_mu_);
            unmask_pending_ = true;
            unmask_mask_bytes_ = *mask_bytes;
            unmask_nb_models_ = staged_nb_models_;
            unmask_result_.reset();
        }
        unmask_cv_.notify_all();
        std::unique_lock<std::mutex> l(unmask_mu_);
        cv_wait_until(unmask_cv_, l,
                      Clock::now() + std::chrono::duration_cast<Clock::duration>(
                                         std::chrono::duration<double>(
                                             settings_.unmask_timeout_s)),
                      [this] { return unmask_result_.has_value() || shutdown_.load(); });
        unmask_pending_ = false;
        if (!unmask_result_) return PhaseId::Failure;
        model_bincode = std::move(*unmask_result_);
    }

    auto id = models_->set_global_model(round_id_, round_seed_, model_bincode);
    if (!id) return PhaseId::Failure;
    store_->set_latest_global_model_id(*id);
    {
        std::lock_guard<std::mutex> l(events_.mu);
        events_.model_bincode = std::make_shared<Bytes>(std::move(model_bincode));
        events_.version += 1;
    }
    return PhaseId::Idle;
}

PhaseId Coordinator::run_failure() {
    // wait for storage readiness, then restart the round (reference
    // phases/failure.rs: 5s poll; shortened here, configurable later)
    for (int i = 0; i < 60 && !shutdown_; ++i) {
        if (store_->is_ready() && models_->is_ready()) return PhaseId::Idle;
        std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
    return shutdown_ ? PhaseId::Shutdown : PhaseId::Idle;
}

// --------------------------------------------------------- request plane

bool Coordinator::process_requests(const PhaseParams& pp, const Handler& h) {
    uint64_t accepted = 0;
    auto start = Clock::now();
    auto min_deadline = start + std::chrono::duration_cast<Clock::duration>(
                                    std::chrono::duration<double>(pp.time.min));
    auto max_deadline = start + std::chrono::duration_cast<Clock::duration>(
                                    std::chrono::duration<double>(pp.time.max));

    auto handle_one = [&](Pending& p) {
        // cross-thread edge: parent = the ingest span that enqueued this
        trace::Span span("handle_request",
                         std::string("phase=") + phase_name(phase_.load()) +
                             " round=" + std::to_string(round_id_.load()),
                         p.span_id);
        PipelineError r;
        if (accepted >= pp.count.max) {
            r = PipelineError::MessageDiscarded;
        } else {
            r = h(p.req);
            if (r == PipelineError::Ok) accepted += 1;
        }
        if (trace::enabled()) span.add("result=" + std::to_string(int(r)));
        // reference emit sites: phases/handler.rs Counter
        using metrics::Measurement;
        metrics::metric(r == PipelineError::Ok               ? Measurement::MessageAccepted
                        : r == PipelineError::MessageDiscarded ? Measurement::MessageDiscarded
                                                               : Measurement::MessageRejected,
                        1.0, round_id_, int(phase_.load()));
        if (p.reply) p.reply->set_value(r);
    };

    // phase 1: accept during [0, time.min]
    while (!shutdown_) {
        std::unique_lock<std::mutex> l(qmu_);
        if (!cv_wait_until(qcv_, l, min_deadline, [this] { return !queue_.empty() || shutdown_; }))
            break;  // min time elapsed
        if (shutdown_) return false;
        if (queue_.empty()) break;
        Pending p = std::move(queue_.front());
        queue_.pop_front();
        l.unlock();
        handle_one(p);
        if (Clock::now() >= min_deadline) break;
    }

    // phase 2: until count.min, bounded by time.max
    while (!shutdown_ && accepted < pp.count.min) {
        std::unique_lock<std::mutex> l(qmu_);
        if (!cv_wait_until(qcv_, l, max_deadline, [this] { return !queue_.empty() || shutdown_; }))
            return false;  // timeout without enough messages
        if (shutdown_) return false;
        if (queue_.empty()) {
            if (Clock::now() >= max_deadline) return false;
            continue;
        }
        Pending p = std::move(queue_.front());
        queue_.pop_front();
        l.unlock();
        handle_one(p);
    }
    return !shutdown_;
}

void Coordinator::purge_outdated_requests() {
    std::lock_guard<std::mutex> l(qmu_);
    for (auto& p : queue_) {
        if (p.reply) p.reply->set_value(PipelineError::MessageRejected);
    }
    queue_.clear();
}

PipelineError Coordinator::enqueue_and_wait(StateMachineRequest req) {
    // reject only on explicit shutdown; `running_` may be false when phases
    // are driven manually via run_one_phase (tests, embedded drivers)
    if (shutdown_.load()) return PipelineError::MessageRejected;
    auto prom = std::make_shared<std::promise<PipelineError>>();
    auto fut = prom->get_future();
    {
        std::lock_guard<std::mutex> l(qmu_);
        queue_.push_back(Pending{std::move(req), prom, trace::current_span()});
    }
    qcv_.notify_one();
    // bounded waits with shutdown checks: an ingest worker must never stay
    // parked on a stopped coordinator
    for (int i = 0; i < 3600; ++i) {
        if (fut.wait_for(std::chrono::seconds(1)) == std::future_status::ready)
            return fut.get();
        if (shutdown_.load()) return PipelineError::MessageRejected;
    }
    return PipelineError::Internal;
}

// ------------------------------------------------------------- pipeline

PipelineError Coordinator::handle_encrypted_message(const uint8_t* data, size_t len) {
    trace::Span span("ingest", "bytes=" + std::to_string(len));
    uint8_t pk[32], sk[32];
    {
        std::lock_guard<std::mutex> l(events_.mu);
        std::memcpy(pk, events_.keys_pk.data(), 32);
        std::memcpy(sk, events_.keys_sk, 32);
    }
    Bytes plain;
    if (!crypto::sealbox_open(plain, data, len, pk, sk)) return PipelineError::Decrypt;
    return handle_message_bytes(plain.data(), plain.size());
}

PipelineError Coordinator::validate_task(const msg::Message& m) {
    RoundParameters params = round_params_snapshot();
    Bytes seed_sum(params.seed.begin(), params.seed.end());
    Bytes seed_update = seed_sum;
    seed_sum.insert(seed_sum.end(), {'s', 'u', 'm'});
    seed_update.insert(seed_update.end(), {'u', 'p', 'd', 'a', 't', 'e'});

    // Eligibility (reference sign.rs:186-192: int(sha256(sig))/(2^256-1) <= t)
    // without per-message heap big-int math: precompute, once per threshold,
    // bound = floor(t_num*(2^256-1)/t_den) with t_num/t_den the EXACT binary
    // value of the double threshold — then h <= bound is one 32-byte
    // big-endian-order compare per message. Exact: for integers,
    // h <= floor(A/B) <=> h*B <= A.
    struct Bound {
        double t = -2.0;
        bool always = false, never = false;
        uint8_t le[32];
    };
    static std::mutex bmu;
    static Bound bounds[2];
    auto eligible = [&](const uint8_t sig[64], double threshold, int slot) {
        if (threshold < 0.0) return false;
        if (threshold > 1.0) return true;
        Bound* b;
        {
            std::lock_guard<std::mutex> bl(bmu);
            b = &bounds[slot];
            if (b->t != threshold) {
                Bound nb;
                nb.t = threshold;
                if (threshold <= 0.0) {
                    nb.never = true;
                } else if (threshold >= 1.0) {
                    nb.always = true;
                } else {
                    Rational t = Rational::from_double(threshold);  // exact
                    Bytes ff(32, 0xff);
                    BigUint max256 = BigUint::from_bytes_le(ff.data(), 32);
                    BigUint bound = (t.numer.mag * max256) / t.denom;
                    bound.to_bytes_le_fixed(nb.le, 32);
                }
                *b = nb;
            }
        }
        if (b->never) return false;
        if (b->always) return true;
        auto h = Sha256::hash(sig, 64);
        for (int i = 31; i >= 0; --i) {
            if (h[i] != b->le[i]) return h[i] < b->le[i];
        }
        return true;  // equal
    };

    const uint8_t* sum_sig = nullptr;
    const uint8_t* upd_sig = nullptr;
    if (const auto* s = std::get_if<msg::SumPayload>(&m.payload)) sum_sig = s->sum_signature.data();
    if (const auto* u = std::get_if<msg::UpdatePayload>(&m.payload)) {
        sum_sig = u->sum_signature.data();
        upd_sig = u->update_signature.data();
    }
    if (const auto* s2 = std::get_if<msg::Sum2Payload>(&m.payload))
        sum_sig = s2->sum_signature.data();
    if (!sum_sig) return PipelineError::UnexpectedMessage;

    bool valid_sum = crypto::ed25519_verify(sum_sig, seed_sum.data(), seed_sum.size(),
                                            m.participant_pk.data());
    bool is_summer = valid_sum && eligible(sum_sig, params.sum, 0);

    if (std::holds_alternative<msg::UpdatePayload>(m.payload)) {
        bool valid_upd = upd_sig && crypto::ed25519_verify(upd_sig, seed_update.data(),
                                                           seed_update.size(),
                                                           m.participant_pk.data());
        bool is_updater = !is_summer && valid_upd && eligible(upd_sig, params.update, 1);
        return is_updater ? PipelineError::Ok : PipelineError::NotUpdateEligible;
    }
    return is_summer ? PipelineError::Ok : PipelineError::NotSumEligible;
}

// ------------------------------------------------ streaming multipart parse
//
// Reads a reassembled message's payload directly out of the ordered chunk
// buffers (no concatenation): the reference's bounded-memory
// from_byte_stream (rust/xaynet-core/src/message/traits.rs:27-54,
// services/messages/multipart/buffer.rs:8-60). Validation matches the
// contiguous deserializers byte for byte.
namespace {

struct ChunkCursor {
    const std::map<uint16_t, Bytes>& chunks;
    std::map<uint16_t, Bytes>::const_iterator it;
    size_t off = 0;
    size_t remaining = 0;
    explicit ChunkCursor(const std::map<uint16_t, Bytes>& c) : chunks(c), it(c.begin()) {
        for (const auto& [id, d] : c) remaining += d.size();
    }
    bool read(uint8_t* dst, size_t n) {
        if (n > remaining) return false;
        while (n) {
            while (it != chunks.end() && off == it->second.size()) {
                ++it;
                off = 0;
            }
            if (it == chunks.end()) return false;
            size_t take = std::min(n, it->second.size() - off);
            std::memcpy(dst, it->second.data() + off, take);
            dst += take;
            off += take;
            n -= take;
            remaining -= take;
        }
        return true;
    }
};

std::optional<mask::MaskObject> stream_read_mask_object(ChunkCursor& cur) {
    uint8_t hdr[8];
    if (!cur.read(hdr, 8)) return std::nullopt;
    auto cfg = mask::MaskConfig::from_bytes(hdr);
    if (!cfg) return std::nullopt;
    size_t count = load32_be(hdr + 4);
    size_t bpn = cfg->info().bpn;
    mask::MaskVect v;
    v.cfg = *cfg;
    v.count = count;
    if (count * bpn > cur.remaining) return std::nullopt;
    v.data.resize(count * bpn);
    if (!cur.read(v.data.data(), v.data.size())) return std::nullopt;
    uint8_t uh[4];
    if (!cur.read(uh, 4)) return std::nullopt;
    auto ucfg = mask::MaskConfig::from_bytes(uh);
    if (!ucfg) return std::nullopt;
    mask::MaskUnit u;
    u.cfg = *ucfg;
    u.data.resize(ucfg->info().bpn);
    if (!cur.read(u.data.data(), u.data.size())) return std::nullopt;
    return mask::MaskObject{std::move(v), std::move(u)};
}

std::optional<msg::UpdatePayload> stream_parse_update(ChunkCursor& cur) {
    msg::UpdatePayload u;
    if (!cur.read(u.sum_signature.data(), 64)) return std::nullopt;
    if (!cur.read(u.update_signature.data(), 64)) return std::nullopt;
    auto mo = stream_read_mask_object(cur);
    if (!mo) return std::nullopt;
    u.masked = std::move(*mo);
    uint8_t lenb[4];
    if (!cur.read(lenb, 4)) return std::nullopt;
    uint32_t total = load32_be(lenb);  // INCLUSIVE of the 4 length bytes
    if (total < 4 || (total - 4) % msg::SEED_ENTRY_LEN != 0) return std::nullopt;
    if (total - 4 != cur.remaining) return std::nullopt;  // trailing bytes: error
    size_t n = (total - 4) / msg::SEED_ENTRY_LEN;
    u.local_seed_dict.resize(n);
    for (size_t i = 0; i < n; ++i) {
        if (!cur.read(u.local_seed_dict[i].pk.data(), 32)) return std::nullopt;
        if (!cur.read(u.local_seed_dict[i].seed.data(), 80)) return std::nullopt;
    }
    // duplicate keys are a decode error (reference traits.rs)
    for (size_t i = 0; i < n; ++i)
        for (size_t j = i + 1; j < n; ++j)
            if (u.local_seed_dict[i].pk == u.local_seed_dict[j].pk) return std::nullopt;
    return u;
}

std::optional<msg::Sum2Payload> stream_parse_sum2(ChunkCursor& cur) {
    msg::Sum2Payload s;
    if (!cur.read(s.sum_signature.data(), 64)) return std::nullopt;
    auto mo = stream_read_mask_object(cur);
    if (!mo) return std::nullopt;
    s.mask = std::move(*mo);
    return s;
}

}  // namespace

PipelineError Coordinator::handle_message_bytes(const uint8_t* data, size_t len) {
    // phase filter by tag before the (expensive) signature check
    PhaseId ph = phase_.load();
    if (len < msg::HEADER_LEN) return PipelineError::Parsing;
    uint8_t tag = data[132];
    if (!((tag == 1 && ph == PhaseId::Sum) || (tag == 2 && ph == PhaseId::Update) ||
          (tag == 3 && ph == PhaseId::Sum2))) {
        return PipelineError::UnexpectedMessage;
    }

    auto m = msg::Message::from_bytes(data, len, /*verify=*/true);
    if (!m) return PipelineError::InvalidMessageSignature;

    // coordinator pk must match the current round key
    {
        std::lock_guard<std::mutex> l(events_.mu);
        if (m->coordinator_pk != events_.keys_pk) return PipelineError::InvalidCoordinatorPublicKey;
    }

    if (m->is_multipart) {
        const auto* c = std::get_if<msg::ChunkPayload>(&m->payload);
        if (!c) return PipelineError::Parsing;
        bool complete = false;
        MultipartEntry done;  // moved out of the map under the lock
        {
            std::lock_guard<std::mutex> l(mp_mu_);
            auto key = std::make_pair(m->participant_pk, c->message_id);
            auto it = multipart_.find(key);
            // admission bounds: chunks buffer BEFORE task validation, so a
            // throwaway-keypair flood of never-completing chunk sets would
            // otherwise grow multipart_ unbounded until round end
            size_t add = c->data.size();
            if (it == multipart_.end() && multipart_.size() >= settings_.multipart_max_entries)
                return PipelineError::MessageRejected;
            size_t entry_old = it == multipart_.end() ? 0 : it->second.bytes;
            size_t dup_old = 0;
            if (it != multipart_.end()) {
                auto ch = it->second.chunks.find(c->id);
                if (ch != it->second.chunks.end()) dup_old = ch->second.size();
            }
            if (multipart_bytes_ - dup_old + add > settings_.multipart_max_total_bytes ||
                multipart_pk_bytes(m->participant_pk) - dup_old + add >
                    settings_.multipart_max_per_pk_bytes)
                return PipelineError::MessageRejected;
            auto& entry = multipart_[key];
            entry.chunks[c->id] = c->data;
            entry.bytes = entry_old - dup_old + add;
            multipart_bytes_ = multipart_bytes_ - dup_old + add;
            if (c->last) {
                if (entry.last_id >= 0 && entry.last_id != int32_t(c->id)) {
                    multipart_bytes_ -= entry.bytes;
                    multipart_.erase(key);  // two different LAST ids: corrupt
                    return PipelineError::Parsing;
                }
                entry.last_id = int32_t(c->id);
            }
            // complete once the LAST id is known and ids 0..last are all
            // present (chunks arrive in any order; duplicates overwrite)
            if (entry.last_id >= 0 &&
                entry.chunks.size() == size_t(entry.last_id) + 1) {
                bool ok = true;
                uint16_t expect = 0;
                for (auto& [id, d] : entry.chunks) {
                    if (id != expect++) {
                        ok = false;
                        break;
                    }
                }
                multipart_bytes_ -= entry.bytes;
                done = std::move(entry);
                multipart_.erase(key);
                if (!ok) return PipelineError::Parsing;
                complete = true;
            }
        }
        if (!complete) return PipelineError::Ok;  // buffered; 200 to client
        // STREAMING re-parse (reference from_byte_stream, traits.rs:27-54):
        // the payload is parsed straight out of the chunk buffers — no
        // concatenated copy of a possibly multi-hundred-MB message; the
        // limb bytes move once, into the MaskObject's own storage
        ChunkCursor cur(done.chunks);
        msg::Message inner;
        inner.participant_pk = m->participant_pk;
        inner.coordinator_pk = m->coordinator_pk;
        inner.tag = m->tag;
        switch (m->tag) {
            case msg::Tag::Sum: {
                Bytes small(cur.remaining);
                if (!cur.read(small.data(), small.size())) return PipelineError::Parsing;
                auto p = msg::SumPayload::deserialize(small.data(), small.size());
                if (!p) return PipelineError::Parsing;
                inner.payload = std::move(*p);
                break;
            }
            case msg::Tag::Update: {
                auto p = stream_parse_update(cur);
                if (!p) return PipelineError::Parsing;
                inner.payload = std::move(*p);
                break;
            }
            case msg::Tag::Sum2: {
                auto p = stream_parse_sum2(cur);
                if (!p) return PipelineError::Parsing;
                inner.payload = std::move(*p);
                break;
            }
        }
        m = std::move(inner);
    }

    PipelineError e = validate_task(*m);
    if (e != PipelineError::Ok) return e;

    StateMachineRequest req;
    if (const auto* s = std::get_if<msg::SumPayload>(&m->payload)) {
        req = SumRequest{m->participant_pk, s->ephm_pk};
    } else if (auto* u = std::get_if<msg::UpdatePayload>(&m->payload)) {
        // pure update validation on THIS (concurrent) thread: config and
        // length vs the round settings, element range vs the group order —
        // the serial protocol thread must not stream 175 MB per update
        if (!(u->masked.vect.cfg == settings_.mask_cfg.vect) ||
            !(u->masked.unit.cfg == settings_.mask_cfg.unit) ||
            u->masked.vect.count != settings_.model_length || !u->masked.is_valid())
            return PipelineError::AggregationFailed;
        req = UpdateRequest{m->participant_pk, std::move(u->local_seed_dict),
                            std::move(u->masked), true};
    } else if (auto* s2 = std::get_if<msg::Sum2Payload>(&m->payload)) {
        req = Sum2Request{m->participant_pk, std::move(s2->mask)};
    } else {
        return PipelineError::UnexpectedMessage;
    }
    return enqueue_and_wait(std::move(req));
}

// ------------------------------------------------------------- fetchers

Bytes Coordinator::fetch_round_params() {
    std::lock_guard<std::mutex> l(events_.mu);
    return events_.params_bincode;
}

Bytes Coordinator::fetch_sum_dict() {
    std::shared_ptr<SumDict> sd = sum_dict_snapshot();
    return bincode::encode_option_sum_dict(sd.get());
}

Bytes Coordinator::fetch_seeds(const msg::Key32& sum_pk) {
    std::shared_ptr<SeedDict> sd = seed_dict_snapshot();
    if (!sd) return bincode::encode_option_update_seed_dict(nullptr);
    auto it = sd->find(sum_pk);
    if (it == sd->end()) return bincode::encode_option_update_seed_dict(nullptr);
    return bincode::encode_option_update_seed_dict(&it->second);
}

Bytes Coordinator::fetch_model() {
    std::shared_ptr<Bytes> m = model_bincode_snapshot();
    if (!m) {
        bincode::Writer w;
        w.u8(0);
        return std::move(w.out);
    }
    return *m;
}

uint64_t Coordinator::events_version() {
    std::lock_guard<std::mutex> l(events_.mu);
    return events_.version;
}

RoundParameters Coordinator::round_params_snapshot() {
    std::lock_guard<std::mutex> l(events_.mu);
    return events_.params;
}

std::shared_ptr<SumDict> Coordinator::sum_dict_snapshot() {
    std::lock_guard<std::mutex> l(events_.mu);
    return events_.sum_dict;
}

std::shared_ptr<SeedDict> Coordinator::seed_dict_snapshot() {
    std::lock_guard<std::mutex> l(events_.mu);
    return events_.seed_dict;
}

std::shared_ptr<Bytes> Coordinator::model_bincode_snapshot() {
    std::lock_guard<std::mutex> l(events_.mu);
    return events_.model_bincode;
}

size_t Coordinator::multipart_pk_bytes(const msg::Key32& pk) const {
    // entries are keyed (pk, message_id) in a sorted map: scan the pk's range
    size_t total = 0;
    auto it = multipart_.lower_bound(std::make_pair(pk, uint16_t(0)));
    for (; it != multipart_.end() && it->first.first == pk; ++it) total += it->second.bytes;
    return total;
}

// ------------------------------------------------------------ staged GPU

std::vector<Bytes> Coordinator::drain_staged_updates() {
    // protocol thread appends during Update (under staged_mu_); the GPU
    // driver thread drains concurrently. Serialization (a memcpy of the
    // packed limbs) happens HERE, off the protocol thread.
    std::vector<mask::MaskObject> objs;
    {
        std::lock_guard<std::mutex> l(staged_mu_);
        objs.swap(staged_);
    }
    std::vector<Bytes> out;
    out.reserve(objs.size());
    for (const auto& o : objs) out.push_back(o.serialize());
    return out;
}

size_t Coordinator::pop_staged_vect(uint8_t* dst, size_t cap, Bytes& unit_out) {
    mask::MaskObject obj;
    {
        std::lock_guard<std::mutex> l(staged_mu_);
        if (staged_.empty()) return 0;
        if (staged_.back().vect.data.size() > cap) return 0;
        obj = std::move(staged_.back());
        staged_.pop_back();
    }
    std::memcpy(dst, obj.vect.data.data(), obj.vect.data.size());
    unit_out = obj.unit.data;
    return obj.vect.data.size();
}

size_t Coordinator::staged_count() {
    std::lock_guard<std::mutex> l(staged_mu_);
    return staged_.size();
}

bool Coordinator::pending_unmask(Bytes& mask_bytes, uint64_t& nb_models) {
    std::lock_guard<std::mutex> l(unmask_mu_);
    if (!unmask_pending_) return false;
    mask_bytes = unmask_mask_bytes_;
    nb_models = unmask_nb_models_;
    return true;
}

void Coordinator::supply_unmasked_model(const Bytes& model_bincode) {
    {
        std::lock_guard<std::mutex> l(unmask_mu_);
        unmask_result_ = model_bincode;
    }
    unmask_cv_.notify_all();
}

// ------------------------------------------------------------ checkpoint

Bytes Coordinator::checkpoint_state() {
    // bincode CoordinatorState (reference state_machine/coordinator.rs:94-109):
    // keys{public(32), secret(32)}, round_id u64, round_params,
    // sum/update/sum2 PhaseParameters{count{min,max u64}, time{min,max u64 s}}
    bincode::Writer w;
    w.raw(encr_pk_, 32);
    w.raw(encr_sk_, 32);
    w.u64(round_id_);
    RoundParameters params = round_params_snapshot();
    Bytes pb = bincode::encode_round_parameters(params);
    w.raw(pb.data(), pb.size());
    for (const PhaseParams* pp : {&settings_.sum, &settings_.update, &settings_.sum2}) {
        w.u64(pp->count.min);
        w.u64(pp->count.max);
        w.u64(uint64_t(pp->time.min));
        w.u64(uint64_t(pp->time.max));
    }
    return std::move(w.out);
}

bool Coordinator::restore_state(const Bytes& state) {
    bincode::Reader r{state.data(), state.size()};
    r.raw(encr_pk_, 32);
    r.raw(encr_sk_, 32);
    round_id_ = r.u64();
    size_t params_len = 0;
    auto params = bincode::decode_round_parameters(state.data() + r.off, state.size() - r.off,
                                                   &params_len);
    if (!params || r.fail) return false;
    r.off += params_len;  // exact bytes consumed
    Bytes pb = bincode::encode_round_parameters(*params);
    for (PhaseParams* pp : {&settings_.sum, &settings_.update, &settings_.sum2}) {
        pp->count.min = r.u64();
        pp->count.max = r.u64();
        pp->time.min = double(r.u64());
        pp->time.max = double(r.u64());
    }
    if (r.fail) return false;
    round_seed_ = params->seed;
    settings_.sum_prob = params->sum;
    settings_.update_prob = params->update;
    settings_.mask_cfg = params->mask_config;
    settings_.model_length = params->model_length;
    {
        std::lock_guard<std::mutex> l(events_.mu);
        events_.round_id = round_id_;
        events_.params = *params;
        events_.params_bincode = pb;
        std::memcpy(events_.keys_pk.data(), encr_pk_, 32);
        std::memcpy(events_.keys_sk, encr_sk_, 32);
        events_.version += 1;
    }
    return true;
}

}  // namespace xaynet::coord
