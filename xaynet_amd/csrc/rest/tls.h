// Optional TLS transport for the REST API (reference rest.rs `tls` feature:
// server authentication via certificate+key, optional client authentication
// via a trust-anchor CA). Blocking accept + one thread per connection over
// OpenSSL; the plaintext epoll server (http.h) stays the default data path.
#pragma once

#include <atomic>
#include <functional>
#include <memory>
#include <string>
#include <thread>
#include <vector>

#include "http.h"

typedef struct ssl_ctx_st SSL_CTX;
typedef struct ssl_st SSL;

namespace xaynet::http {

class TlsHttpServer {
  public:
    // client_ca non-empty => require and verify client certificates
    TlsHttpServer(HttpServer::Handler handler, std::string host, uint16_t port,
                  std::string cert_file, std::string key_file, std::string client_ca = "");
    ~TlsHttpServer();

    bool start();
    void stop();
    uint16_t port() const { return port_; }

  private:
    void accept_loop();
    void serve_conn(int fd);
    void track(int fd, bool add);

    HttpServer::Handler handler_;
    std::string host_;
    uint16_t port_;
    std::string cert_file_, key_file_, client_ca_;

    SSL_CTX* ctx_ = nullptr;
    int listen_fd_ = -1;
    std::atomic<bool> running_{false};
    std::thread accept_thread_;
    std::mutex conn_mu_;
    std::vector<std::thread> conns_;
    std::vector<int> live_fds_;  // shut down on stop() to wake blocked reads
};

class TlsHttpClient {
  public:
    // ca_file empty => trust system roots; insecure => skip verification
    // client cert/key for mutual TLS (reference tls_client_auth)
    TlsHttpClient(std::string host, uint16_t port, std::string ca_file = "",
                  bool insecure = false, std::string cert_file = "", std::string key_file = "",
                  double timeout_s = 30.0);
    ~TlsHttpClient();

    bool request(const std::string& method, const std::string& path_and_query,
                 const Bytes* body, int& status_out, Bytes& body_out);

  private:
    bool connect_();
    void close_();
    bool do_request(const std::string& method, const std::string& pq, const Bytes* body,
                    int& status_out, Bytes& body_out);

    std::string host_;
    uint16_t port_;
    std::string ca_file_, cert_file_, key_file_;
    bool insecure_;
    double timeout_s_;
    SSL_CTX* ctx_ = nullptr;
    SSL* ssl_ = nullptr;
    int fd_ = -1;
};

}  // namespace xaynet::http
