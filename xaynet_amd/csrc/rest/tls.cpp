#include "tls.h"

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <openssl/err.h>
#include <openssl/ssl.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <cstring>

namespace xaynet::http {

static constexpr size_t MAX_TLS_BODY = 256u << 20;   // request bodies
static constexpr size_t MAX_TLS_RESPONSE = 4ull << 30;  // GET /model scale

// --------------------------------------------------------------- server

TlsHttpServer::TlsHttpServer(HttpServer::Handler handler, std::string host, uint16_t port,
                             std::string cert_file, std::string key_file, std::string client_ca)
    : handler_(std::move(handler)), host_(std::move(host)), port_(port),
      cert_file_(std::move(cert_file)), key_file_(std::move(key_file)),
      client_ca_(std::move(client_ca)) {}

TlsHttpServer::~TlsHttpServer() { stop(); }

bool TlsHttpServer::start() {
    ctx_ = SSL_CTX_new(TLS_server_method());
    if (!ctx_) return false;
    SSL_CTX_set_min_proto_version(ctx_, TLS1_2_VERSION);
    if (SSL_CTX_use_certificate_chain_file(ctx_, cert_file_.c_str()) != 1 ||
        SSL_CTX_use_PrivateKey_file(ctx_, key_file_.c_str(), SSL_FILETYPE_PEM) != 1 ||
        SSL_CTX_check_private_key(ctx_) != 1) {
        SSL_CTX_free(ctx_);
        ctx_ = nullptr;
        return false;
    }
    if (!client_ca_.empty()) {
        // mutual TLS (reference tls_client_auth trust anchor)
        if (SSL_CTX_load_verify_locations(ctx_, client_ca_.c_str(), nullptr) != 1) {
            SSL_CTX_free(ctx_);
            ctx_ = nullptr;
            return false;
        }
        SSL_CTX_set_verify(ctx_, SSL_VERIFY_PEER | SSL_VERIFY_FAIL_IF_NO_PEER_CERT, nullptr);
    }

    listen_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (listen_fd_ < 0) return false;
    int one = 1;
    setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons(port_);
    if (host_.empty() || host_ == "0.0.0.0")
        addr.sin_addr.s_addr = INADDR_ANY;
    else if (inet_pton(AF_INET, host_.c_str(), &addr.sin_addr) != 1)
        return false;
    if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0 ||
        listen(listen_fd_, 256) != 0) {
        ::close(listen_fd_);
        listen_fd_ = -1;
        return false;
    }
    if (port_ == 0) {
        socklen_t len = sizeof(addr);
        getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&addr), &len);
        port_ = ntohs(addr.sin_port);
    }
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
    return true;
}

void TlsHttpServer::stop() {
    if (!running_.exchange(false)) return;
    if (listen_fd_ >= 0) {
        ::shutdown(listen_fd_, SHUT_RDWR);
        ::close(listen_fd_);
        listen_fd_ = -1;
    }
    if (accept_thread_.joinable()) accept_thread_.join();
    std::vector<std::thread> conns;
    {
        std::lock_guard<std::mutex> l(conn_mu_);
        conns.swap(conns_);
        for (int fd : live_fds_) ::shutdown(fd, SHUT_RDWR);  // wake blocked SSL_reads
    }
    for (auto& t : conns)
        if (t.joinable()) t.join();
    if (ctx_) {
        SSL_CTX_free(ctx_);
        ctx_ = nullptr;
    }
}

void TlsHttpServer::accept_loop() {
    while (running_) {
        int fd = accept(listen_fd_, nullptr, nullptr);
        if (fd < 0) {
            if (!running_) break;
            continue;
        }
        int one = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
        timeval tv{30, 0};
        setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
        setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
        std::lock_guard<std::mutex> l(conn_mu_);
        if (conns_.size() > 512) {  // reap finished threads
            for (auto& t : conns_)
                if (t.joinable()) t.join();
            conns_.clear();
        }
        conns_.emplace_back([this, fd] { serve_conn(fd); });
    }
}

void TlsHttpServer::track(int fd, bool add) {
    std::lock_guard<std::mutex> l(conn_mu_);
    if (add) {
        live_fds_.push_back(fd);
    } else {
        live_fds_.erase(std::remove(live_fds_.begin(), live_fds_.end(), fd), live_fds_.end());
    }
}

void TlsHttpServer::serve_conn(int fd) {
    track(fd, true);
    SSL* ssl = SSL_new(ctx_);
    SSL_set_fd(ssl, fd);
    if (SSL_accept(ssl) != 1) {
        SSL_free(ssl);
        ::close(fd);
        track(fd, false);
        return;
    }
    Bytes in;
    uint8_t buf[16 * 1024];
    bool keep_alive = true;
    while (running_ && keep_alive) {
        Request req;
        int pr;
        while ((pr = parse_http_request(in, req, keep_alive)) == 0) {
            int r = SSL_read(ssl, buf, sizeof(buf));
            if (r <= 0) goto done;
            in.insert(in.end(), buf, buf + r);
            if (in.size() > MAX_TLS_BODY + 64 * 1024) goto done;
        }
        if (pr < 0) break;
        Response resp;
        try {
            resp = handler_(req);
        } catch (...) {
            resp.status = 500;
        }
        {
            char head[256];
            int hl = snprintf(head, sizeof(head),
                              "HTTP/1.1 %d %s\r\nContent-Type: %s\r\nContent-Length: %zu\r\n"
                              "Connection: %s\r\n\r\n",
                              resp.status, resp.status == 200 ? "OK" : "X",
                              resp.content_type, resp.body.size(),
                              keep_alive ? "keep-alive" : "close");
            if (SSL_write(ssl, head, hl) <= 0) break;
            size_t off = 0;
            while (off < resp.body.size()) {
                int w = SSL_write(ssl, resp.body.data() + off,
                                  int(std::min<size_t>(resp.body.size() - off, 1 << 20)));
                if (w <= 0) goto done;
                off += size_t(w);
            }
        }
    }
done:
    SSL_shutdown(ssl);
    SSL_free(ssl);
    ::close(fd);
    track(fd, false);
}

// --------------------------------------------------------------- client

TlsHttpClient::TlsHttpClient(std::string host, uint16_t port, std::string ca_file, bool insecure,
                             std::string cert_file, std::string key_file, double timeout_s)
    : host_(std::move(host)), port_(port), ca_file_(std::move(ca_file)),
      cert_file_(std::move(cert_file)), key_file_(std::move(key_file)), insecure_(insecure),
      timeout_s_(timeout_s) {}

TlsHttpClient::~TlsHttpClient() {
    close_();
    if (ctx_) SSL_CTX_free(ctx_);
}

void TlsHttpClient::close_() {
    if (ssl_) {
        SSL_shutdown(ssl_);
        SSL_free(ssl_);
        ssl_ = nullptr;
    }
    if (fd_ >= 0) {
        ::close(fd_);
        fd_ = -1;
    }
}

bool TlsHttpClient::connect_() {
    close_();
    if (!ctx_) {
        ctx_ = SSL_CTX_new(TLS_client_method());
        if (!ctx_) return false;
        SSL_CTX_set_min_proto_version(ctx_, TLS1_2_VERSION);
        if (!insecure_) {
            if (!ca_file_.empty()) {
                if (SSL_CTX_load_verify_locations(ctx_, ca_file_.c_str(), nullptr) != 1)
                    return false;
            } else {
                SSL_CTX_set_default_verify_paths(ctx_);
            }
            SSL_CTX_set_verify(ctx_, SSL_VERIFY_PEER, nullptr);
        }
        if (!cert_file_.empty() &&
            (SSL_CTX_use_certificate_chain_file(ctx_, cert_file_.c_str()) != 1 ||
             SSL_CTX_use_PrivateKey_file(ctx_, key_file_.c_str(), SSL_FILETYPE_PEM) != 1))
            return false;
    }
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
    int rc = ::connect(fd_, res->ai_addr, res->ai_addrlen);
    freeaddrinfo(res);
    if (rc != 0) {
        close_();
        return false;
    }
    ssl_ = SSL_new(ctx_);
    SSL_set_fd(ssl_, fd_);
    SSL_set_tlsext_host_name(ssl_, host_.c_str());
    if (!insecure_) {
        // Chain verification alone accepts any cert signed by a trusted CA for ANY
        // domain; pin the expected peer identity so SSL_connect fails on mismatch
        // (reference client is reqwest/rustls which always verifies hostnames).
        X509_VERIFY_PARAM* vp = SSL_get0_param(ssl_);
        in_addr a4{};
        in6_addr a6{};
        bool is_ip = inet_pton(AF_INET, host_.c_str(), &a4) == 1 ||
                     inet_pton(AF_INET6, host_.c_str(), &a6) == 1;
        int ok = is_ip ? X509_VERIFY_PARAM_set1_ip_asc(vp, host_.c_str())
                       : X509_VERIFY_PARAM_set1_host(vp, host_.c_str(), 0);
        if (ok != 1) {
            close_();
            return false;
        }
    }
    if (SSL_connect(ssl_) != 1) {
        close_();
        return false;
    }
    if (!insecure_ && SSL_get_verify_result(ssl_) != X509_V_OK) {
        close_();
        return false;
    }
    return true;
}

bool TlsHttpClient::do_request(const std::string& method, const std::string& pq, const Bytes* body,
                               int& status_out, Bytes& body_out) {
    char head[512];
    int hl = snprintf(head, sizeof(head),
                      "%s %s HTTP/1.1\r\nHost: %s\r\nContent-Length: %zu\r\n"
                      "Connection: keep-alive\r\n\r\n",
                      method.c_str(), pq.c_str(), host_.c_str(), body ? body->size() : 0);
    if (hl <= 0 || SSL_write(ssl_, head, hl) <= 0) return false;
    if (body && !body->empty()) {
        size_t off = 0;
        while (off < body->size()) {
            int w = SSL_write(ssl_, body->data() + off,
                              int(std::min<size_t>(body->size() - off, 1 << 20)));
            if (w <= 0) return false;
            off += size_t(w);
        }
    }
    Bytes buf;
    uint8_t tmp[16 * 1024];
    size_t hdr_end = 0;
    while (true) {
        for (size_t i = hdr_end > 3 ? hdr_end - 3 : 0; i + 3 < buf.size(); ++i) {
            if (buf[i] == '\r' && buf[i + 1] == '\n' && buf[i + 2] == '\r' && buf[i + 3] == '\n') {
                hdr_end = i + 4;
                goto have_headers;
            }
        }
        {
            int r = SSL_read(ssl_, tmp, sizeof(tmp));
            if (r <= 0) return false;
            hdr_end = buf.size();
            buf.insert(buf.end(), tmp, tmp + r);
        }
    }
have_headers: {
    std::string head_s(reinterpret_cast<const char*>(buf.data()), hdr_end);
    size_t sp = head_s.find(' ');
    if (sp == std::string::npos) return false;
    status_out = atoi(head_s.c_str() + sp + 1);
    size_t content_length = 0;
    {
        std::string lower = head_s;
        for (auto& ch : lower) ch = char(tolower(ch));
        size_t cl = lower.find("content-length:");
        if (cl != std::string::npos)
            content_length = strtoull(lower.c_str() + cl + 15, nullptr, 10);
    }
    if (content_length > MAX_TLS_RESPONSE) return false;
    while (buf.size() < hdr_end + content_length) {
        int r = SSL_read(ssl_, tmp, sizeof(tmp));
        if (r <= 0) return false;
        buf.insert(buf.end(), tmp, tmp + r);
    }
    body_out.assign(buf.begin() + hdr_end, buf.begin() + hdr_end + content_length);
    return true;
}
}

bool TlsHttpClient::request(const std::string& method, const std::string& pq, const Bytes* body,
                            int& status_out, Bytes& body_out) {
    if (!ssl_ && !connect_()) return false;
    if (do_request(method, pq, body, status_out, body_out)) return true;
    if (!connect_()) return false;
    return do_request(method, pq, body, status_out, body_out);
}

}  // namespace xaynet::http
