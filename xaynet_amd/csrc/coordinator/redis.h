// Redis-backed CoordinatorStorage: own RESP2 client over TCP, the reference's
// data model and atomicity (rust/xaynet-server/src/storage/coordinator_storage/
// redis/mod.rs) with the Lua-script invariants expressed as WATCH/MULTI/EXEC
// transactions (same atomicity guarantee: the validating reads are WATCHed, so
// a concurrent mutation aborts the EXEC and the operation retries).
//
// Data model (mirrors redis/mod.rs:1-39):
//   "coordinator_state"        bincode CoordinatorState (raw bytes)
//   "latest_global_model_id"   string id
//   "sum_dict"                 hash: sum_pk(32B)    -> ephm_pk(32B)
//   "seed_dict:<sum_pk_hex>"   hash: update_pk(32B) -> encrypted seed(80B)
//   "update_participants"      set of update_pk that already submitted
//   "update_sum_pks"           set of sum_pk with a seed_dict hash (cleanup)
//   "mask_submitted"           set of sum_pk that already voted
//   "mask_dict"                zset: mask bytes -> score
#pragma once

#include <mutex>
#include <string>
#include <vector>

#include "storage.h"

namespace xaynet::coord {

// Minimal RESP2 value
struct RespValue {
    enum class Kind { Nil, Str, Err, Int, Array } kind = Kind::Nil;
    std::string str;             // Str/Err payload (bulk or simple)
    long long integer = 0;       // Int
    std::vector<RespValue> arr;  // Array
};

class RespClient {
  public:
    RespClient(std::string host, uint16_t port, double timeout_s = 5.0);
    ~RespClient();

    bool connect();
    void close();
    bool connected() const { return fd_ >= 0; }

    // Send one command, read one reply. Auto-reconnects once on I/O failure
    // (reference ConnectionManager behavior, redis/mod.rs:95-101).
    bool command(const std::vector<std::string>& args, RespValue& out);

  private:
    bool send_all(const std::string& buf);
    bool read_value(RespValue& out);
    bool read_line(std::string& line);
    bool fill();

    std::string host_;
    uint16_t port_;
    double timeout_s_;
    int fd_ = -1;
    std::string rbuf_;
    size_t rpos_ = 0;
};

class RedisCoordinatorStorage : public CoordinatorStorage {
  public:
    RedisCoordinatorStorage(std::string host, uint16_t port, double timeout_s = 5.0);

    bool set_coordinator_state(const Bytes& state) override;
    std::optional<Bytes> coordinator_state() override;
    SumPartAddError add_sum_participant(const Key32& pk, const Key32& ephm_pk) override;
    std::optional<SumDict> sum_dict() override;
    SeedDictAddError add_local_seed_dict(
        const Key32& update_pk, const std::vector<msg::LocalSeedEntry>& local) override;
    std::optional<SeedDict> seed_dict() override;
    MaskScoreIncrError incr_mask_score(const Key32& sum_pk, const Bytes& mask_bytes) override;
    std::vector<std::pair<Bytes, uint64_t>> best_masks(size_t n) override;
    uint64_t number_of_unique_masks() override;
    bool delete_coordinator_data() override;
    bool delete_dicts() override;
    bool set_latest_global_model_id(const std::string& id) override;
    std::optional<std::string> latest_global_model_id() override;
    bool is_ready() override;

  private:
    bool cmd(const std::vector<std::string>& args, RespValue& out);

    std::mutex mu_;  // one connection, serialized commands
    RespClient client_;
};

// Redis-backed ModelStorage (global models as raw bincode values under
// "global_model:<id>"; refuse-overwrite via SETNX like the S3 backend).
class RedisModelStorage : public ModelStorage {
  public:
    RedisModelStorage(std::string host, uint16_t port, double timeout_s = 5.0);
    std::optional<std::string> set_global_model(uint64_t round_id, const Key32& round_seed,
                                                const Bytes& model_bincode) override;
    std::optional<Bytes> global_model(const std::string& id) override;
    bool is_ready() override;

  private:
    std::mutex mu_;
    RespClient client_;
};

}  // namespace xaynet::coord
