// Coordinator storage interfaces + in-memory backends.
//
// Mirrors the reference Storage trait surface
// (rust/xaynet-server/src/storage/traits.rs:31-311) including the typed
// protocol error codes the Redis Lua scripts return. The in-memory backend is
// the default for the single-node MI355X coordinator (the reference's Redis
// data model maps 1:1 onto these maps; a RESP-backed implementation can slot
// in behind the same interface).
#pragma once

#include <atomic>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <string>
#include <vector>

#include "../common.h"
#include "../message/bincode.h"
#include "../message/message.h"

namespace xaynet::coord {

using msg::Key32;
using msg::EncrSeed80;
using bincode::SumDict;
using bincode::SeedDict;
using bincode::UpdateSeedDict;

enum class SumPartAddError { Ok = 0, AlreadyExists, Storage };

enum class SeedDictAddError {
    Ok = 0,
    LengthMisMatch,
    UnknownSumParticipant,
    UpdatePkAlreadySubmitted,
    UpdatePkAlreadyExistsInUpdateSeedDict,
    Storage,
};

enum class MaskScoreIncrError { Ok = 0, UnknownSumParticipant, MaskAlreadySubmitted, Storage };

class CoordinatorStorage {
  public:
    virtual ~CoordinatorStorage() = default;

    virtual bool set_coordinator_state(const Bytes& state) = 0;
    virtual std::optional<Bytes> coordinator_state() = 0;

    virtual SumPartAddError add_sum_participant(const Key32& pk, const Key32& ephm_pk) = 0;
    virtual std::optional<SumDict> sum_dict() = 0;

    // Validates length vs |sum_dict|, one submission per update pk, known sum
    // pks (reference Redis Lua, coordinator_storage/redis/mod.rs:208-267).
    virtual SeedDictAddError add_local_seed_dict(
        const Key32& update_pk, const std::vector<msg::LocalSeedEntry>& local) = 0;
    virtual std::optional<SeedDict> seed_dict() = 0;

    // Mask-popularity vote (ZINCRBY analog; key = serialized mask object).
    virtual MaskScoreIncrError incr_mask_score(const Key32& sum_pk, const Bytes& mask_bytes) = 0;
    virtual std::vector<std::pair<Bytes, uint64_t>> best_masks(size_t n) = 0;
    virtual uint64_t number_of_unique_masks() = 0;

    virtual bool delete_coordinator_data() = 0;
    virtual bool delete_dicts() = 0;

    virtual bool set_latest_global_model_id(const std::string& id) = 0;
    virtual std::optional<std::string> latest_global_model_id() = 0;

    virtual bool is_ready() = 0;
};

class ModelStorage {
  public:
    virtual ~ModelStorage() = default;
    // id = "<round_id>_<round_seed_hex>" (reference storage/traits.rs:195-198)
    virtual std::optional<std::string> set_global_model(uint64_t round_id,
                                                        const Key32& round_seed,
                                                        const Bytes& model_bincode) = 0;
    virtual std::optional<Bytes> global_model(const std::string& id) = 0;
    virtual bool is_ready() = 0;
};

// ------------------------------------------------------------- in-memory

class InMemoryCoordinatorStorage : public CoordinatorStorage {
  public:
    bool set_coordinator_state(const Bytes& state) override;
    std::optional<Bytes> coordinator_state() override;
    SumPartAddError add_sum_participant(const Key32& pk, const Key32& ephm_pk) override;
    std::optional<SumDict> sum_dict() override;
    SeedDictAddError add_local_seed_dict(const Key32& update_pk,
                                         const std::vector<msg::LocalSeedEntry>& local) override;
    std::optional<SeedDict> seed_dict() override;
    MaskScoreIncrError incr_mask_score(const Key32& sum_pk, const Bytes& mask_bytes) override;
    std::vector<std::pair<Bytes, uint64_t>> best_masks(size_t n) override;
    uint64_t number_of_unique_masks() override;
    bool delete_coordinator_data() override;
    bool delete_dicts() override;
    bool set_latest_global_model_id(const std::string& id) override;
    std::optional<std::string> latest_global_model_id() override;
    bool is_ready() override { return true; }

  private:
    std::mutex mu_;
    std::optional<Bytes> state_;
    SumDict sum_dict_;
    SeedDict seed_dict_;
    std::map<Key32, bool> update_submitted_;
    std::map<Key32, bool> mask_submitted_;
    std::map<Bytes, uint64_t> mask_dict_;
    std::optional<std::string> latest_model_id_;
};

class InMemoryModelStorage : public ModelStorage {
  public:
    std::optional<std::string> set_global_model(uint64_t round_id, const Key32& round_seed,
                                                const Bytes& model_bincode) override;
    std::optional<Bytes> global_model(const std::string& id) override;
    bool is_ready() override { return true; }

  private:
    std::mutex mu_;
    std::map<std::string, Bytes> models_;
};

// ---------------------------------------------------------- fault injection
//
// Test backend (the reference tests phases with mockall storage mocks
// returning Err, state_machine/tests/): fails selected operations a given
// number of times, then behaves like the in-memory store. Drives the
// Failure-phase recovery paths.
class FaultInjectionStorage : public InMemoryCoordinatorStorage {
  public:
    // counters: how many times the next calls of each kind fail
    std::atomic<int> fail_sum_dict{0};
    std::atomic<int> fail_seed_dict{0};
    std::atomic<int> fail_state{0};
    std::atomic<int> fail_best_masks{0};
    std::atomic<int> not_ready{0};

    std::optional<SumDict> sum_dict() override {
        if (take(fail_sum_dict)) return std::nullopt;
        return InMemoryCoordinatorStorage::sum_dict();
    }
    std::optional<SeedDict> seed_dict() override {
        if (take(fail_seed_dict)) return std::nullopt;
        return InMemoryCoordinatorStorage::seed_dict();
    }
    bool set_coordinator_state(const Bytes& state) override {
        if (take(fail_state)) return false;
        return InMemoryCoordinatorStorage::set_coordinator_state(state);
    }
    std::vector<std::pair<Bytes, uint64_t>> best_masks(size_t n) override {
        if (take(fail_best_masks)) return {};
        return InMemoryCoordinatorStorage::best_masks(n);
    }
    bool is_ready() override { return !take(not_ready); }

  private:
    static bool take(std::atomic<int>& c) {
        int v = c.load();
        while (v > 0 && !c.compare_exchange_weak(v, v - 1)) {}
        return v > 0;
    }
};

// ------------------------------------------------------------- file-backed
//
// Durable local equivalents of the reference's network backends (this build
// has no network): coordinator state / latest-model-id land in files under
// `dir` with atomic rename (the Redis persistence analog,
// storage/coordinator_storage/redis/mod.rs); global models are one file per
// `roundid_seedhex` id and an existing id is never overwritten (the S3
// analog, model_storage/s3.rs:190-198). Dict state stays in RAM like the
// reference's Redis working set — restore starts a fresh round and deletes
// dicts anyway (phases/idle.rs).

class FileCoordinatorStorage : public InMemoryCoordinatorStorage {
  public:
    explicit FileCoordinatorStorage(std::string dir);
    bool set_coordinator_state(const Bytes& state) override;
    std::optional<Bytes> coordinator_state() override;
    bool set_latest_global_model_id(const std::string& id) override;
    std::optional<std::string> latest_global_model_id() override;
    bool is_ready() override;

  private:
    std::string dir_;
};

class FileModelStorage : public ModelStorage {
  public:
    explicit FileModelStorage(std::string dir);
    std::optional<std::string> set_global_model(uint64_t round_id, const Key32& round_seed,
                                                const Bytes& model_bincode) override;
    std::optional<Bytes> global_model(const std::string& id) override;
    bool is_ready() override;

  private:
    std::string dir_;
    std::mutex mu_;
};

}  // namespace xaynet::coord
