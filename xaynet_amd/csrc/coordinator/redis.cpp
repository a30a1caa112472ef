#include "redis.h"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>

namespace xaynet::coord {

// ----------------------------------------------------------- RESP client

RespClient::RespClient(std::string host, uint16_t port, double timeout_s)
    : host_(std::move(host)), port_(port), timeout_s_(timeout_s) {}

RespClient::~RespClient() { close(); }

void RespClient::close() {
    if (fd_ >= 0) {
        ::close(fd_);
        fd_ = -1;
    }
    rbuf_.clear();
    rpos_ = 0;
}

bool RespClient::connect() {
    close();
    addrinfo hints{};
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    addrinfo* res = nullptr;
    char portstr[8];
    snprintf(portstr, sizeof(portstr), "%u", unsigned(port_));
    if (getaddrinfo(host_.c_str(), portstr, &hints, &res) != 0 || !res) return false;
    fd_ = ::socket(res->ai_family, res->ai_socktype, res->ai_protocol);
    if (fd_ < 0) {
        freeaddrinfo(res);
        return false;
    }
    timeval tv;
    tv.tv_sec = long(timeout_s_);
    tv.tv_usec = long((timeout_s_ - double(tv.tv_sec)) * 1e6);
    setsockopt(fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    setsockopt(fd_, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    int one = 1;
    setsockopt(fd_, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    int rc = ::connect(fd_, res->ai_addr, res->ai_addrlen);
    freeaddrinfo(res);
    if (rc != 0) {
        close();
        return false;
    }
    return true;
}

bool RespClient::send_all(const std::string& buf) {
    size_t off = 0;
    while (off < buf.size()) {
        ssize_t w = ::send(fd_, buf.data() + off, buf.size() - off, MSG_NOSIGNAL);
        if (w <= 0) return false;
        off += size_t(w);
    }
    return true;
}

bool RespClient::fill() {
    char tmp[16 * 1024];
    ssize_t r = ::recv(fd_, tmp, sizeof(tmp), 0);
    if (r <= 0) return false;
    rbuf_.append(tmp, size_t(r));
    return true;
}

bool RespClient::read_line(std::string& line) {
    while (true) {
        size_t nl = rbuf_.find("\r\n", rpos_);
        if (nl != std::string::npos) {
            line.assign(rbuf_, rpos_, nl - rpos_);
            rpos_ = nl + 2;
            if (rpos_ > 1 << 20) {  // compact
                rbuf_.erase(0, rpos_);
                rpos_ = 0;
            }
            return true;
        }
        if (!fill()) return false;
    }
}

bool RespClient::read_value(RespValue& out) {
    std::string line;
    if (!read_line(line)) return false;
    if (line.empty()) return false;
    char t = line[0];
    std::string rest = line.substr(1);
    switch (t) {
        case '+':
            out.kind = RespValue::Kind::Str;
            out.str = rest;
            return true;
        case '-':
            out.kind = RespValue::Kind::Err;
            out.str = rest;
            return true;
        case ':':
            out.kind = RespValue::Kind::Int;
            out.integer = strtoll(rest.c_str(), nullptr, 10);
            return true;
        case '$': {
            long long n = strtoll(rest.c_str(), nullptr, 10);
            if (n < 0) {
                out.kind = RespValue::Kind::Nil;
                return true;
            }
            while (rbuf_.size() - rpos_ < size_t(n) + 2) {
                if (!fill()) return false;
            }
            out.kind = RespValue::Kind::Str;
            out.str.assign(rbuf_, rpos_, size_t(n));
            rpos_ += size_t(n) + 2;
            return true;
        }
        case '*': {
            long long n = strtoll(rest.c_str(), nullptr, 10);
            if (n < 0) {
                out.kind = RespValue::Kind::Nil;
                return true;
            }
            out.kind = RespValue::Kind::Array;
            out.arr.resize(size_t(n));
            for (long long i = 0; i < n; ++i) {
                if (!read_value(out.arr[size_t(i)])) return false;
            }
            return true;
        }
        default:
            return false;
    }
}

bool RespClient::command(const std::vector<std::string>& args, RespValue& out) {
    std::string buf = "*" + std::to_string(args.size()) + "\r\n";
    for (const auto& a : args) {
        buf += "$" + std::to_string(a.size()) + "\r\n";
        buf += a;
        buf += "\r\n";
    }
    for (int attempt = 0; attempt < 2; ++attempt) {
        if (fd_ < 0 && !connect()) continue;
        if (send_all(buf) && read_value(out)) return true;
        close();  // retry once on a fresh connection (auto-reconnect)
    }
    return false;
}

// ------------------------------------------------------ coordinator store

static std::string key_bytes(const Key32& k) {
    return std::string(reinterpret_cast<const char*>(k.data()), 32);
}
static std::string hexify(const Key32& k) {
    static const char* d = "0123456789abcdef";
    std::string s;
    s.reserve(64);
    for (uint8_t b : k) {
        s += d[b >> 4];
        s += d[b & 15];
    }
    return s;
}

RedisCoordinatorStorage::RedisCoordinatorStorage(std::string host, uint16_t port,
                                                 double timeout_s)
    : client_(std::move(host), port, timeout_s) {}

bool RedisCoordinatorStorage::cmd(const std::vector<std::string>& args, RespValue& out) {
    return client_.command(args, out);
}

bool RedisCoordinatorStorage::set_coordinator_state(const Bytes& state) {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    return cmd({"SET", "coordinator_state",
                std::string(reinterpret_cast<const char*>(state.data()), state.size())},
               v) &&
           v.kind == RespValue::Kind::Str;
}

std::optional<Bytes> RedisCoordinatorStorage::coordinator_state() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    if (!cmd({"GET", "coordinator_state"}, v) || v.kind != RespValue::Kind::Str) return {};
    return Bytes(v.str.begin(), v.str.end());
}

SumPartAddError RedisCoordinatorStorage::add_sum_participant(const Key32& pk,
                                                             const Key32& ephm_pk) {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    // reference redis/mod.rs: HSETNX sum_dict pk ephm_pk; 0 -> AlreadyExists
    if (!cmd({"HSETNX", "sum_dict", key_bytes(pk), key_bytes(ephm_pk)}, v) ||
        v.kind != RespValue::Kind::Int)
        return SumPartAddError::Storage;
    return v.integer == 1 ? SumPartAddError::Ok : SumPartAddError::AlreadyExists;
}

std::optional<SumDict> RedisCoordinatorStorage::sum_dict() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    if (!cmd({"HGETALL", "sum_dict"}, v) || v.kind != RespValue::Kind::Array) return {};
    SumDict out;
    for (size_t i = 0; i + 1 < v.arr.size(); i += 2) {
        if (v.arr[i].str.size() != 32 || v.arr[i + 1].str.size() != 32) return {};
        Key32 pk, ephm;
        std::memcpy(pk.data(), v.arr[i].str.data(), 32);
        std::memcpy(ephm.data(), v.arr[i + 1].str.data(), 32);
        out.emplace(pk, ephm);
    }
    return out;
}

SeedDictAddError RedisCoordinatorStorage::add_local_seed_dict(
    const Key32& update_pk, const std::vector<msg::LocalSeedEntry>& local) {
    std::lock_guard<std::mutex> l(mu_);
    // The reference enforces these invariants atomically in a Lua script
    // (redis/mod.rs:208-267). Same invariants via WATCH on the validated
    // keys: a concurrent mutation nils the EXEC and we retry.
    for (int attempt = 0; attempt < 8; ++attempt) {
        RespValue v;
        if (!cmd({"WATCH", "sum_dict", "update_participants"}, v)) return SeedDictAddError::Storage;
        // |local| must equal |sum_dict|
        if (!cmd({"HLEN", "sum_dict"}, v) || v.kind != RespValue::Kind::Int)
            return SeedDictAddError::Storage;
        long long hlen = v.integer;
        if ((long long)local.size() != hlen) {
            cmd({"UNWATCH"}, v);
            return SeedDictAddError::LengthMisMatch;
        }
        // every sum_pk must be a sum participant
        for (const auto& e : local) {
            if (!cmd({"HEXISTS", "sum_dict", key_bytes(e.pk)}, v) ||
                v.kind != RespValue::Kind::Int)
                return SeedDictAddError::Storage;
            if (v.integer != 1) {
                cmd({"UNWATCH"}, v);
                return SeedDictAddError::UnknownSumParticipant;
            }
        }
        // one submission per update pk
        if (!cmd({"SISMEMBER", "update_participants", key_bytes(update_pk)}, v) ||
            v.kind != RespValue::Kind::Int)
            return SeedDictAddError::Storage;
        if (v.integer == 1) {
            cmd({"UNWATCH"}, v);
            return SeedDictAddError::UpdatePkAlreadySubmitted;
        }
        if (!cmd({"MULTI"}, v)) return SeedDictAddError::Storage;
        cmd({"SADD", "update_participants", key_bytes(update_pk)}, v);
        for (const auto& e : local) {
            std::string seed(reinterpret_cast<const char*>(e.seed.data()), e.seed.size());
            cmd({"HSETNX", "seed_dict:" + hexify(e.pk), key_bytes(update_pk), seed}, v);
            cmd({"SADD", "update_sum_pks", key_bytes(e.pk)}, v);
        }
        if (!cmd({"EXEC"}, v)) return SeedDictAddError::Storage;
        if (v.kind == RespValue::Kind::Nil) continue;  // raced; retry
        if (v.kind != RespValue::Kind::Array) return SeedDictAddError::Storage;
        // HSETNX results: 0 would mean this update pk already wrote to that
        // sum_pk's hash — unreachable given the update_participants guard,
        // kept as the reference's belt-and-braces error
        for (size_t i = 1; i + 1 < v.arr.size(); i += 2) {
            if (v.arr[i].kind == RespValue::Kind::Int && v.arr[i].integer == 0)
                return SeedDictAddError::UpdatePkAlreadyExistsInUpdateSeedDict;
        }
        return SeedDictAddError::Ok;
    }
    return SeedDictAddError::Storage;
}

std::optional<SeedDict> RedisCoordinatorStorage::seed_dict() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue sd;
    if (!cmd({"HGETALL", "sum_dict"}, sd) || sd.kind != RespValue::Kind::Array) return {};
    SeedDict out;
    for (size_t i = 0; i + 1 < sd.arr.size(); i += 2) {
        Key32 sum_pk;
        if (sd.arr[i].str.size() != 32) return {};
        std::memcpy(sum_pk.data(), sd.arr[i].str.data(), 32);
        RespValue h;
        if (!cmd({"HGETALL", "seed_dict:" + hexify(sum_pk)}, h) ||
            h.kind != RespValue::Kind::Array)
            return {};
        UpdateSeedDict entries;
        for (size_t j = 0; j + 1 < h.arr.size(); j += 2) {
            if (h.arr[j].str.size() != 32 || h.arr[j + 1].str.size() != 80) return {};
            Key32 upk;
            EncrSeed80 seed;
            std::memcpy(upk.data(), h.arr[j].str.data(), 32);
            std::memcpy(seed.data(), h.arr[j + 1].str.data(), 80);
            entries.emplace(upk, seed);
        }
        out.emplace(sum_pk, std::move(entries));
    }
    return out;
}

MaskScoreIncrError RedisCoordinatorStorage::incr_mask_score(const Key32& sum_pk,
                                                            const Bytes& mask_bytes) {
    std::lock_guard<std::mutex> l(mu_);
    // reference Lua (redis/mod.rs:303-339): HEXISTS sum_dict; SADD
    // mask_submitted (dedup); ZINCRBY mask_dict
    for (int attempt = 0; attempt < 8; ++attempt) {
        RespValue v;
        if (!cmd({"WATCH", "mask_submitted"}, v)) return MaskScoreIncrError::Storage;
        if (!cmd({"HEXISTS", "sum_dict", key_bytes(sum_pk)}, v) ||
            v.kind != RespValue::Kind::Int)
            return MaskScoreIncrError::Storage;
        if (v.integer != 1) {
            cmd({"UNWATCH"}, v);
            return MaskScoreIncrError::UnknownSumParticipant;
        }
        if (!cmd({"SISMEMBER", "mask_submitted", key_bytes(sum_pk)}, v) ||
            v.kind != RespValue::Kind::Int)
            return MaskScoreIncrError::Storage;
        if (v.integer == 1) {
            cmd({"UNWATCH"}, v);
            return MaskScoreIncrError::MaskAlreadySubmitted;
        }
        if (!cmd({"MULTI"}, v)) return MaskScoreIncrError::Storage;
        cmd({"SADD", "mask_submitted", key_bytes(sum_pk)}, v);
        cmd({"ZINCRBY", "mask_dict", "1",
             std::string(reinterpret_cast<const char*>(mask_bytes.data()), mask_bytes.size())},
            v);
        if (!cmd({"EXEC"}, v)) return MaskScoreIncrError::Storage;
        if (v.kind == RespValue::Kind::Nil) continue;
        return MaskScoreIncrError::Ok;
    }
    return MaskScoreIncrError::Storage;
}

std::vector<std::pair<Bytes, uint64_t>> RedisCoordinatorStorage::best_masks(size_t n) {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    if (!cmd({"ZREVRANGE", "mask_dict", "0", std::to_string(n ? n - 1 : 0), "WITHSCORES"}, v) ||
        v.kind != RespValue::Kind::Array)
        return {};
    std::vector<std::pair<Bytes, uint64_t>> out;
    for (size_t i = 0; i + 1 < v.arr.size(); i += 2) {
        Bytes mb(v.arr[i].str.begin(), v.arr[i].str.end());
        uint64_t score = uint64_t(strtoull(v.arr[i + 1].str.c_str(), nullptr, 10));
        out.emplace_back(std::move(mb), score);
    }
    return out;
}

uint64_t RedisCoordinatorStorage::number_of_unique_masks() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    if (!cmd({"ZCARD", "mask_dict"}, v) || v.kind != RespValue::Kind::Int) return 0;
    return uint64_t(v.integer);
}

bool RedisCoordinatorStorage::delete_coordinator_data() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    return cmd({"FLUSHDB"}, v) && v.kind != RespValue::Kind::Err;
}

bool RedisCoordinatorStorage::delete_dicts() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    // collect per-sum_pk seed hashes before deleting the tracking set
    std::vector<std::string> del = {"DEL", "sum_dict", "update_participants",
                                    "mask_submitted", "mask_dict", "update_sum_pks"};
    if (cmd({"SMEMBERS", "update_sum_pks"}, v) && v.kind == RespValue::Kind::Array) {
        for (const auto& m : v.arr) {
            if (m.str.size() == 32) {
                Key32 pk;
                std::memcpy(pk.data(), m.str.data(), 32);
                del.push_back("seed_dict:" + hexify(pk));
            }
        }
    }
    return cmd(del, v) && v.kind == RespValue::Kind::Int;
}

bool RedisCoordinatorStorage::set_latest_global_model_id(const std::string& id) {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    return cmd({"SET", "latest_global_model_id", id}, v) && v.kind == RespValue::Kind::Str;
}

std::optional<std::string> RedisCoordinatorStorage::latest_global_model_id() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    if (!cmd({"GET", "latest_global_model_id"}, v) || v.kind != RespValue::Kind::Str) return {};
    return v.str;
}

bool RedisCoordinatorStorage::is_ready() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    return cmd({"PING"}, v) && v.kind == RespValue::Kind::Str;
}

// ----------------------------------------------------------- model store

RedisModelStorage::RedisModelStorage(std::string host, uint16_t port, double timeout_s)
    : client_(std::move(host), port, timeout_s) {}

std::optional<std::string> RedisModelStorage::set_global_model(uint64_t round_id,
                                                               const Key32& round_seed,
                                                               const Bytes& model_bincode) {
    std::lock_guard<std::mutex> l(mu_);
    std::string id = std::to_string(round_id) + "_" + hexify(round_seed);
    RespValue v;
    if (!client_.command({"SETNX", "global_model:" + id,
                          std::string(reinterpret_cast<const char*>(model_bincode.data()),
                                      model_bincode.size())},
                         v) ||
        v.kind != RespValue::Kind::Int)
        return {};
    if (v.integer != 1) return {};  // refuse overwrite (reference s3.rs:190-198)
    return id;
}

std::optional<Bytes> RedisModelStorage::global_model(const std::string& id) {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    if (!client_.command({"GET", "global_model:" + id}, v) || v.kind != RespValue::Kind::Str)
        return {};
    return Bytes(v.str.begin(), v.str.end());
}

bool RedisModelStorage::is_ready() {
    std::lock_guard<std::mutex> l(mu_);
    RespValue v;
    return client_.command({"PING"}, v) && v.kind == RespValue::Kind::Str;
}

}  // namespace xaynet::coord
