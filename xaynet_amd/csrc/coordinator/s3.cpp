#include "s3.h"

namespace xaynet::coord {

static std::string hex32(const Key32& k) {
    static const char* d = "0123456789abcdef";
    std::string s;
    s.reserve(64);
    for (uint8_t b : k) {
        s += d[b >> 4];
        s += d[b & 15];
    }
    return s;
}

S3ModelStorage::S3ModelStorage(std::string host, uint16_t port, std::string bucket,
                               double timeout_s)
    : client_(std::move(host), port, timeout_s), bucket_(std::move(bucket)) {}

bool S3ModelStorage::ensure_bucket() {
    if (bucket_ok_) return true;
    int status = 0;
    Bytes body;
    // PUT /<bucket> creates it; 200 = created, 409 = already owned — both fine
    if (!client_.request("PUT", "/" + bucket_, nullptr, status, body)) return false;
    bucket_ok_ = status == 200 || status == 409;
    return bucket_ok_;
}

std::optional<std::string> S3ModelStorage::set_global_model(uint64_t round_id,
                                                            const Key32& round_seed,
                                                            const Bytes& model_bincode) {
    std::lock_guard<std::mutex> l(mu_);
    if (!ensure_bucket()) return {};
    std::string id = std::to_string(round_id) + "_" + hex32(round_seed);
    std::string path = "/" + bucket_ + "/" + id;
    int status = 0;
    Bytes body;
    // refuse-overwrite (reference s3.rs:190-198): existing object wins.
    // GET (not HEAD): HEAD replies carry Content-Length without a body,
    // which a simple streaming client cannot distinguish generically.
    if (!client_.request("GET", path, nullptr, status, body)) return {};
    if (status == 200) return {};
    if (!client_.request("PUT", path, &model_bincode, status, body)) return {};
    if (status != 200) return {};
    return id;
}

std::optional<Bytes> S3ModelStorage::global_model(const std::string& id) {
    std::lock_guard<std::mutex> l(mu_);
    int status = 0;
    Bytes body;
    if (!client_.request("GET", "/" + bucket_ + "/" + id, nullptr, status, body)) return {};
    if (status != 200) return {};
    return body;
}

bool S3ModelStorage::is_ready() {
    std::lock_guard<std::mutex> l(mu_);
    int status = 0;
    Bytes body;
    if (!client_.request("GET", "/", nullptr, status, body)) return false;
    return status > 0;
}

}  // namespace xaynet::coord
