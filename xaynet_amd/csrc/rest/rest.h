// Coordinator REST API + SDK HTTP client — reference parity:
//   POST /message        raw encrypted PET message -> always 200, empty body
//                        (rust/xaynet-server/src/rest.rs:93-101)
//   GET  /params         200 bincode(RoundParameters)      (rest.rs:178-196)
//   GET  /sums           200 bincode(SumDict) | 204        (rest.rs:104-126)
//   GET  /seeds?pk=b64   200 bincode(UpdateSeedDict) | 204 (rest.rs:129-154)
//   GET  /model          200 bincode(Model) | 204          (rest.rs:157-175)
// None is signalled by 204 No Content; bodies are bare bincode values (the
// Option wrapper exists only in-process). pk is standard base64 of the 32-byte
// sum participant signing key, percent-encoded in the query (rest.rs:209-219,
// xaynet-sdk/src/client.rs:159-167).
#pragma once

#include <memory>

#include "../coordinator/coordinator.h"
#include "../sdk/participant.h"
#include "http.h"
#include "tls.h"

namespace xaynet::rest {

// route a parsed request against a coordinator (used by RestServer; exposed
// for tests)
http::Response route(coord::Coordinator& c, const http::Request& req);

class RestServer {
  public:
    RestServer(std::shared_ptr<coord::Coordinator> c, std::string host, uint16_t port,
               int workers = 4);
    // TLS variant (reference rest.rs `tls` feature): server cert/key,
    // optional client-auth trust anchor
    RestServer(std::shared_ptr<coord::Coordinator> c, std::string host, uint16_t port,
               std::string tls_cert, std::string tls_key, std::string tls_client_auth);
    bool start();
    void stop();
    uint16_t port() const;

  private:
    std::shared_ptr<coord::Coordinator> coord_;
    std::unique_ptr<http::HttpServer> server_;
    std::unique_ptr<http::TlsHttpServer> tls_server_;
};

// SDK-side client speaking the REST API (reference xaynet-sdk/src/client.rs)
class HttpXaynetClient : public sdk::XaynetClient {
  public:
    HttpXaynetClient(std::string host, uint16_t port, double timeout_s = 30.0);

    std::optional<bincode::RoundParameters> get_round_params() override;
    std::optional<bincode::SumDict> get_sums() override;
    std::optional<bincode::UpdateSeedDict> get_seeds(const msg::Key32& pk) override;
    std::optional<Bytes> get_model_bincode() override;
    bool send_message(const Bytes& encrypted) override;

  private:
    http::HttpClient client_;
};

// HTTPS variant (reference reqwest TLS options: CA pin, client cert)
class TlsXaynetClient : public sdk::XaynetClient {
  public:
    TlsXaynetClient(std::string host, uint16_t port, std::string ca_file = "",
                    bool insecure = false, std::string cert_file = "",
                    std::string key_file = "", double timeout_s = 30.0);

    std::optional<bincode::RoundParameters> get_round_params() override;
    std::optional<bincode::SumDict> get_sums() override;
    std::optional<bincode::UpdateSeedDict> get_seeds(const msg::Key32& pk) override;
    std::optional<Bytes> get_model_bincode() override;
    bool send_message(const Bytes& encrypted) override;

  private:
    http::TlsHttpClient client_;
};

}  // namespace xaynet::rest
