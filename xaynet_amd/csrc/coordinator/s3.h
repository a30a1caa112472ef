// S3-compatible (minio-style) model storage: path-style HTTP PUT/GET against
// `http://host:port/<bucket>/<id>`, refuse-overwrite, bucket auto-create.
// Mirrors rust/xaynet-server/src/storage/model_storage/s3.rs:68-198 (bucket
// "global-models", id = "<round_id>_<round_seed_hex>", store refuses to
// overwrite an existing id). Unsigned requests (minio dev/anonymous mode);
// credentialed deployments front this with a signing proxy or network policy.
#pragma once

#include <mutex>

#include "../rest/http.h"
#include "storage.h"

namespace xaynet::coord {

class S3ModelStorage : public ModelStorage {
  public:
    S3ModelStorage(std::string host, uint16_t port, std::string bucket = "global-models",
                   double timeout_s = 10.0);

    std::optional<std::string> set_global_model(uint64_t round_id, const Key32& round_seed,
                                                const Bytes& model_bincode) override;
    std::optional<Bytes> global_model(const std::string& id) override;
    bool is_ready() override;

  private:
    bool ensure_bucket();

    std::mutex mu_;
    http::HttpClient client_;
    std::string bucket_;
    bool bucket_ok_ = false;
};

}  // namespace xaynet::coord
