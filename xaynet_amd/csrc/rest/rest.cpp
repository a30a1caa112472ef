#include "rest.h"

namespace xaynet::rest {

// strip the bincode Option tag from an in-process fetcher body:
// [1, rest...] -> 200 with rest; [0] -> 204
static http::Response option_body(Bytes b) {
    http::Response r;
    if (b.empty() || b[0] == 0) {
        r.status = 204;
        return r;
    }
    b.erase(b.begin());
    r.body = std::move(b);
    return r;
}

http::Response route(coord::Coordinator& c, const http::Request& req) {
    http::Response r;
    if (req.path == "/message") {
        if (req.method != "POST") {
            r.status = 405;
            return r;
        }
        // reference rest.rs:93-101 — errors are logged, 200 regardless
        c.handle_encrypted_message(req.body.data(), req.body.size());
        r.content_type = "text/plain";
        return r;
    }
    if (req.method != "GET") {
        r.status = 405;
        return r;
    }
    if (req.path == "/params") {
        r.body = c.fetch_round_params();
        return r;
    }
    if (req.path == "/sums") return option_body(c.fetch_sum_dict());
    if (req.path == "/model") return option_body(c.fetch_model());
    if (req.path == "/seeds") {
        std::string pk_b64 = http::query_get(req.query, "pk");
        Bytes pk;
        if (!http::base64_decode(pk_b64, pk) || pk.size() != 32) {
            r.status = 400;
            return r;
        }
        msg::Key32 key{};
        std::copy(pk.begin(), pk.end(), key.begin());
        return option_body(c.fetch_seeds(key));
    }
    r.status = 404;
    return r;
}

RestServer::RestServer(std::shared_ptr<coord::Coordinator> c, std::string host, uint16_t port,
                       int workers)
    : coord_(std::move(c)),
      server_(std::make_unique<http::HttpServer>(
          [this](const http::Request& req) { return route(*coord_, req); }, std::move(host), port,
          workers)) {}

RestServer::RestServer(std::shared_ptr<coord::Coordinator> c, std::string host, uint16_t port,
                       std::string tls_cert, std::string tls_key, std::string tls_client_auth)
    : coord_(std::move(c)),
      tls_server_(std::make_unique<http::TlsHttpServer>(
          [this](const http::Request& req) { return route(*coord_, req); }, std::move(host), port,
          std::move(tls_cert), std::move(tls_key), std::move(tls_client_auth))) {}

bool RestServer::start() { return tls_server_ ? tls_server_->start() : server_->start(); }
void RestServer::stop() {
    if (tls_server_) tls_server_->stop();
    if (server_) server_->stop();
}
uint16_t RestServer::port() const { return tls_server_ ? tls_server_->port() : server_->port(); }

// --------------------------------------------------------------- clients
//
// Shared REST-decode logic over any transport with
// `request(method, path, body*, status&, body&)` (plain or TLS).

template <class C>
static std::optional<bincode::RoundParameters> impl_params(C& c) {
    int status;
    Bytes body;
    if (!c.request("GET", "/params", nullptr, status, body) || status != 200)
        return std::nullopt;
    return bincode::decode_round_parameters(body.data(), body.size());
}

template <class C>
static std::optional<bincode::SumDict> impl_sums(C& c) {
    int status;
    Bytes body;
    if (!c.request("GET", "/sums", nullptr, status, body) || status != 200) return std::nullopt;
    body.insert(body.begin(), 1);  // re-wrap as Option Some for the shared decoder
    auto d = bincode::decode_option_sum_dict(body.data(), body.size());
    if (!d || !*d) return std::nullopt;
    return **d;
}

template <class C>
static std::optional<bincode::UpdateSeedDict> impl_seeds(C& c, const msg::Key32& pk) {
    // percent-encode the base64 pk ('+' '/' '=' are reserved in queries)
    std::string b64 = http::base64_encode(pk.data(), pk.size());
    std::string enc;
    for (char ch : b64) {
        if (ch == '+')
            enc += "%2B";
        else if (ch == '/')
            enc += "%2F";
        else if (ch == '=')
            enc += "%3D";
        else
            enc.push_back(ch);
    }
    int status;
    Bytes body;
    if (!c.request("GET", "/seeds?pk=" + enc, nullptr, status, body) || status != 200)
        return std::nullopt;
    body.insert(body.begin(), 1);
    auto d = bincode::decode_option_update_seed_dict(body.data(), body.size());
    if (!d || !*d) return std::nullopt;
    return **d;
}

template <class C>
static std::optional<Bytes> impl_model(C& c) {
    int status;
    Bytes body;
    if (!c.request("GET", "/model", nullptr, status, body) || status != 200) return std::nullopt;
    body.insert(body.begin(), 1);  // Participant expects the Option-tagged Some body
    return body;
}

template <class C>
static bool impl_send(C& c, const Bytes& encrypted) {
    int status;
    Bytes body;
    return c.request("POST", "/message", &encrypted, status, body) && status == 200;
}

HttpXaynetClient::HttpXaynetClient(std::string host, uint16_t port, double timeout_s)
    : client_(std::move(host), port, timeout_s) {}

std::optional<bincode::RoundParameters> HttpXaynetClient::get_round_params() {
    return impl_params(client_);
}
std::optional<bincode::SumDict> HttpXaynetClient::get_sums() { return impl_sums(client_); }
std::optional<bincode::UpdateSeedDict> HttpXaynetClient::get_seeds(const msg::Key32& pk) {
    return impl_seeds(client_, pk);
}
std::optional<Bytes> HttpXaynetClient::get_model_bincode() { return impl_model(client_); }
bool HttpXaynetClient::send_message(const Bytes& encrypted) {
    return impl_send(client_, encrypted);
}

TlsXaynetClient::TlsXaynetClient(std::string host, uint16_t port, std::string ca_file,
                                 bool insecure, std::string cert_file, std::string key_file,
                                 double timeout_s)
    : client_(std::move(host), port, std::move(ca_file), insecure, std::move(cert_file),
              std::move(key_file), timeout_s) {}

std::optional<bincode::RoundParameters> TlsXaynetClient::get_round_params() {
    return impl_params(client_);
}
std::optional<bincode::SumDict> TlsXaynetClient::get_sums() { return impl_sums(client_); }
std::optional<bincode::UpdateSeedDict> TlsXaynetClient::get_seeds(const msg::Key32& pk) {
    return impl_seeds(client_, pk);
}
std::optional<Bytes> TlsXaynetClient::get_model_bincode() { return impl_model(client_); }
bool TlsXaynetClient::send_message(const Bytes& encrypted) {
    return impl_send(client_, encrypted);
}

}  // namespace xaynet::rest
