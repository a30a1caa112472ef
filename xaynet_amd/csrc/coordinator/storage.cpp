#include "storage.h"

#include <sys/stat.h>
#include <sys/types.h>

#include <algorithm>
#include <cctype>
#include <cstdio>

namespace xaynet::coord {

bool InMemoryCoordinatorStorage::set_coordinator_state(const Bytes& state) {
    std::lock_guard<std::mutex> l(mu_);
    state_ = state;
    return true;
}

std::optional<Bytes> InMemoryCoordinatorStorage::coordinator_state() {
    std::lock_guard<std::mutex> l(mu_);
    return state_;
}

SumPartAddError InMemoryCoordinatorStorage::add_sum_participant(const Key32& pk,
                                                                const Key32& ephm_pk) {
    std::lock_guard<std::mutex> l(mu_);
    // HSETNX semantics
    auto [it, inserted] = sum_dict_.emplace(pk, ephm_pk);
    return inserted ? SumPartAddError::Ok : SumPartAddError::AlreadyExists;
}

std::optional<SumDict> InMemoryCoordinatorStorage::sum_dict() {
    std::lock_guard<std::mutex> l(mu_);
    return sum_dict_;
}

SeedDictAddError InMemoryCoordinatorStorage::add_local_seed_dict(
    const Key32& update_pk, const std::vector<msg::LocalSeedEntry>& local) {
    std::lock_guard<std::mutex> l(mu_);
    // reference Lua semantics (redis/mod.rs:208-267): length must equal
    // |sum_dict|; update pk may submit once; every key must be a sum pk
    if (local.size() != sum_dict_.size()) return SeedDictAddError::LengthMisMatch;
    if (update_submitted_.count(update_pk)) return SeedDictAddError::UpdatePkAlreadySubmitted;
    for (const auto& e : local) {
        if (!sum_dict_.count(e.pk)) return SeedDictAddError::UnknownSumParticipant;
        auto it = seed_dict_.find(e.pk);
        if (it != seed_dict_.end() && it->second.count(update_pk))
            return SeedDictAddError::UpdatePkAlreadyExistsInUpdateSeedDict;
    }
    update_submitted_[update_pk] = true;
    for (const auto& e : local) seed_dict_[e.pk][update_pk] = e.seed;
    return SeedDictAddError::Ok;
}

std::optional<SeedDict> InMemoryCoordinatorStorage::seed_dict() {
    std::lock_guard<std::mutex> l(mu_);
    return seed_dict_;
}

MaskScoreIncrError InMemoryCoordinatorStorage::incr_mask_score(const Key32& sum_pk,
                                                               const Bytes& mask_bytes) {
    std::lock_guard<std::mutex> l(mu_);
    if (!sum_dict_.count(sum_pk)) return MaskScoreIncrError::UnknownSumParticipant;
    if (mask_submitted_.count(sum_pk)) return MaskScoreIncrError::MaskAlreadySubmitted;
    mask_submitted_[sum_pk] = true;
    mask_dict_[mask_bytes] += 1;
    return MaskScoreIncrError::Ok;
}

std::vector<std::pair<Bytes, uint64_t>> InMemoryCoordinatorStorage::best_masks(size_t n) {
    std::lock_guard<std::mutex> l(mu_);
    std::vector<std::pair<Bytes, uint64_t>> all(mask_dict_.begin(), mask_dict_.end());
    std::sort(all.begin(), all.end(),
              [](const auto& a, const auto& b) { return a.second > b.second; });
    if (all.size() > n) all.resize(n);
    return all;
}

uint64_t InMemoryCoordinatorStorage::number_of_unique_masks() {
    std::lock_guard<std::mutex> l(mu_);
    return mask_dict_.size();
}

bool InMemoryCoordinatorStorage::delete_coordinator_data() {
    std::lock_guard<std::mutex> l(mu_);
    state_.reset();
    latest_model_id_.reset();
    sum_dict_.clear();
    seed_dict_.clear();
    update_submitted_.clear();
    mask_submitted_.clear();
    mask_dict_.clear();
    return true;
}

bool InMemoryCoordinatorStorage::delete_dicts() {
    std::lock_guard<std::mutex> l(mu_);
    sum_dict_.clear();
    seed_dict_.clear();
    update_submitted_.clear();
    mask_submitted_.clear();
    mask_dict_.clear();
    return true;
}

bool InMemoryCoordinatorStorage::set_latest_global_model_id(const std::string& id) {
    std::lock_guard<std::mutex> l(mu_);
    latest_model_id_ = id;
    return true;
}

std::optional<std::string> InMemoryCoordinatorStorage::latest_global_model_id() {
    std::lock_guard<std::mutex> l(mu_);
    return latest_model_id_;
}

std::optional<std::string> InMemoryModelStorage::set_global_model(uint64_t round_id,
                                                                  const Key32& round_seed,
                                                                  const Bytes& model_bincode) {
    std::lock_guard<std::mutex> l(mu_);
    std::string id = std::to_string(round_id) + "_" + to_hex(round_seed.data(), 32);
    // refuse to overwrite an existing id (reference s3.rs:190-198)
    if (models_.count(id)) return std::nullopt;
    models_[id] = model_bincode;
    return id;
}

std::optional<Bytes> InMemoryModelStorage::global_model(const std::string& id) {
    std::lock_guard<std::mutex> l(mu_);
    auto it = models_.find(id);
    if (it == models_.end()) return std::nullopt;
    return it->second;
}

// ------------------------------------------------------------- file-backed

static bool write_file_atomic(const std::string& path, const Bytes& data) {
    std::string tmp = path + ".tmp";
    FILE* f = fopen(tmp.c_str(), "wb");
    if (!f) return false;
    bool ok = data.empty() || fwrite(data.data(), 1, data.size(), f) == data.size();
    ok = (fflush(f) == 0) && ok;
    fclose(f);
    if (!ok) {
        remove(tmp.c_str());
        return false;
    }
    return rename(tmp.c_str(), path.c_str()) == 0;
}

static std::optional<Bytes> read_file(const std::string& path) {
    FILE* f = fopen(path.c_str(), "rb");
    if (!f) return std::nullopt;
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    Bytes out(n > 0 ? size_t(n) : 0);
    bool ok = out.empty() || fread(out.data(), 1, out.size(), f) == out.size();
    fclose(f);
    if (!ok) return std::nullopt;
    return out;
}

static bool valid_storage_id(const std::string& id) {
    if (id.empty() || id.size() > 128) return false;
    for (char c : id)
        if (!isalnum(uint8_t(c)) && c != '_') return false;
    return true;
}

FileCoordinatorStorage::FileCoordinatorStorage(std::string dir) : dir_(std::move(dir)) {
    mkdir(dir_.c_str(), 0755);  // best-effort; is_ready() reports failures
}

bool FileCoordinatorStorage::set_coordinator_state(const Bytes& state) {
    InMemoryCoordinatorStorage::set_coordinator_state(state);
    return write_file_atomic(dir_ + "/coordinator_state.bin", state);
}

std::optional<Bytes> FileCoordinatorStorage::coordinator_state() {
    if (auto b = read_file(dir_ + "/coordinator_state.bin")) return b;
    return InMemoryCoordinatorStorage::coordinator_state();
}

bool FileCoordinatorStorage::set_latest_global_model_id(const std::string& id) {
    InMemoryCoordinatorStorage::set_latest_global_model_id(id);
    return write_file_atomic(dir_ + "/latest_model_id", Bytes(id.begin(), id.end()));
}

std::optional<std::string> FileCoordinatorStorage::latest_global_model_id() {
    if (auto b = read_file(dir_ + "/latest_model_id"))
        return std::string(b->begin(), b->end());
    return InMemoryCoordinatorStorage::latest_global_model_id();
}

bool FileCoordinatorStorage::is_ready() {
    struct stat st{};
    return stat(dir_.c_str(), &st) == 0 && S_ISDIR(st.st_mode);
}

FileModelStorage::FileModelStorage(std::string dir) : dir_(std::move(dir)) {
    mkdir(dir_.c_str(), 0755);
}

std::optional<std::string> FileModelStorage::set_global_model(uint64_t round_id,
                                                              const Key32& round_seed,
                                                              const Bytes& model_bincode) {
    std::lock_guard<std::mutex> l(mu_);
    std::string id = std::to_string(round_id) + "_" + to_hex(round_seed.data(), 32);
    std::string path = dir_ + "/" + id;
    struct stat st{};
    // refuse to overwrite an existing id (reference s3.rs:190-198)
    if (stat(path.c_str(), &st) == 0) return std::nullopt;
    if (!write_file_atomic(path, model_bincode)) return std::nullopt;
    return id;
}

std::optional<Bytes> FileModelStorage::global_model(const std::string& id) {
    if (!valid_storage_id(id)) return std::nullopt;
    std::lock_guard<std::mutex> l(mu_);
    return read_file(dir_ + "/" + id);
}

bool FileModelStorage::is_ready() {
    struct stat st{};
    return stat(dir_.c_str(), &st) == 0 && S_ISDIR(st.st_mode);
}

}  // namespace xaynet::coord
