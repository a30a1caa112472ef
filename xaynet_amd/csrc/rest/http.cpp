#include "http.h"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/epoll.h>
#include <sys/eventfd.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>

namespace xaynet::http {

// --------------------------------------------------------------- helpers

static int hexval(char c) {
    if (c >= '0' && c <= '9') return c - '0';
    if (c >= 'a' && c <= 'f') return c - 'a' + 10;
    if (c >= 'A' && c <= 'F') return c - 'A' + 10;
    return -1;
}

std::string percent_decode(const std::string& s) {
    std::string out;
    out.reserve(s.size());
    for (size_t i = 0; i < s.size(); ++i) {
        if (s[i] == '%' && i + 2 < s.size()) {
            int h = hexval(s[i + 1]), l = hexval(s[i + 2]);
            if (h >= 0 && l >= 0) {
                out.push_back(char(h * 16 + l));
                i += 2;
                continue;
            }
        }
        out.push_back(s[i]);
    }
    return out;
}

std::string query_get(const std::string& query, const std::string& key) {
    size_t pos = 0;
    while (pos < query.size()) {
        size_t amp = query.find('&', pos);
        if (amp == std::string::npos) amp = query.size();
        size_t eq = query.find('=', pos);
        if (eq != std::string::npos && eq < amp) {
            if (query.compare(pos, eq - pos, key) == 0)
                return percent_decode(query.substr(eq + 1, amp - eq - 1));
        }
        pos = amp + 1;
    }
    return {};
}

static const char B64[] = "ABCDEFGHIJKLMNOPQRSTUVWXYZabcdefghijklmnopqrstuvwxyz0123456789+/";

std::string base64_encode(const uint8_t* p, size_t n) {
    std::string out;
    out.reserve((n + 2) / 3 * 4);
    for (size_t i = 0; i < n; i += 3) {
        uint32_t v = uint32_t(p[i]) << 16;
        if (i + 1 < n) v |= uint32_t(p[i + 1]) << 8;
        if (i + 2 < n) v |= uint32_t(p[i + 2]);
        out.push_back(B64[(v >> 18) & 63]);
        out.push_back(B64[(v >> 12) & 63]);
        out.push_back(i + 1 < n ? B64[(v >> 6) & 63] : '=');
        out.push_back(i + 2 < n ? B64[v & 63] : '=');
    }
    return out;
}

bool base64_decode(const std::string& s, Bytes& out) {
    static int8_t rev[256];
    static bool init = false;
    if (!init) {
        memset(rev, -1, sizeof(rev));
        for (int i = 0; i < 64; ++i) rev[uint8_t(B64[i])] = int8_t(i);
        init = true;
    }
    out.clear();
    uint32_t buf = 0;
    int bits = 0;
    for (char c : s) {
        if (c == '=' || c == '\n' || c == '\r') continue;
        int8_t v = rev[uint8_t(c)];
        if (v < 0) return false;
        buf = (buf << 6) | uint32_t(v);
        bits += 6;
        if (bits >= 8) {
            bits -= 8;
            out.push_back(uint8_t(buf >> bits));
        }
    }
    return true;
}

static void set_nonblock(int fd) {
    int fl = fcntl(fd, F_GETFL, 0);
    fcntl(fd, F_SETFL, fl | O_NONBLOCK);
}

static const char* status_text(int s) {
    switch (s) {
        case 200: return "OK";
        case 204: return "No Content";
        case 400: return "Bad Request";
        case 404: return "Not Found";
        case 405: return "Method Not Allowed";
        case 411: return "Length Required";
        case 413: return "Payload Too Large";
        default: return "Internal Server Error";
    }
}

static constexpr size_t MAX_BODY = 256u << 20;  // request-body cap, 256 MiB (update messages
                                                // chunk via multipart well below this)
// response cap (client side): GET /model bodies are ~30 B per weight in the
// reference's Vec<Ratio<BigInt>> bincode — a 25M-param model is ~750 MB
static constexpr size_t MAX_RESPONSE = 4ull << 30;

// --------------------------------------------------------------- server

HttpServer::HttpServer(Handler handler, std::string host, uint16_t port, int workers)
    : handler_(std::move(handler)), host_(std::move(host)), port_(port),
      n_workers_(workers < 1 ? 1 : workers) {}

HttpServer::~HttpServer() { stop(); }

bool HttpServer::start() {
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
    if (bind(listen_fd_, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0) {
        ::close(listen_fd_);
        listen_fd_ = -1;
        return false;
    }
    if (port_ == 0) {
        socklen_t len = sizeof(addr);
        getsockname(listen_fd_, reinterpret_cast<sockaddr*>(&addr), &len);
        port_ = ntohs(addr.sin_port);
    }
    if (listen(listen_fd_, 512) != 0) return false;
    set_nonblock(listen_fd_);

    epoll_fd_ = epoll_create1(0);
    wake_fd_ = eventfd(0, EFD_NONBLOCK);
    epoll_event ev{};
    ev.events = EPOLLIN;
    ev.data.u64 = 0;  // listen fd marker
    epoll_ctl(epoll_fd_, EPOLL_CTL_ADD, listen_fd_, &ev);
    epoll_event wev{};
    wev.events = EPOLLIN;
    wev.data.u64 = UINT64_MAX;  // wake fd marker
    epoll_ctl(epoll_fd_, EPOLL_CTL_ADD, wake_fd_, &wev);

    running_ = true;
    io_thread_ = std::thread([this] { io_loop(); });
    for (int i = 0; i < n_workers_; ++i)
        workers_.emplace_back([this] { worker_loop(); });
    return true;
}

void HttpServer::stop() {
    if (!running_.exchange(false)) return;
    // wake everyone
    uint64_t v = 1;
    if (wake_fd_ >= 0) { ssize_t r = write(wake_fd_, &v, 8); (void)r; }
    cv_.notify_all();
    if (io_thread_.joinable()) io_thread_.join();
    for (auto& w : workers_)
        if (w.joinable()) w.join();
    workers_.clear();
    for (auto& [id, c] : conns_)
        if (c.fd >= 0) ::close(c.fd);
    conns_.clear();
    if (listen_fd_ >= 0) ::close(listen_fd_);
    if (epoll_fd_ >= 0) ::close(epoll_fd_);
    if (wake_fd_ >= 0) ::close(wake_fd_);
    listen_fd_ = epoll_fd_ = wake_fd_ = -1;
}

int HttpServer::try_parse(Conn& c, Request& out) {
    return parse_http_request(c.in, out, c.keep_alive);
}

int parse_http_request(Bytes& in, Request& out, bool& keep_alive) {
    // find end of headers
    const char* data = reinterpret_cast<const char*>(in.data());
    size_t n = in.size();
    const char* hdr_end = nullptr;
    for (size_t i = 0; i + 3 < n; ++i) {
        if (data[i] == '\r' && data[i + 1] == '\n' && data[i + 2] == '\r' && data[i + 3] == '\n') {
            hdr_end = data + i;
            break;
        }
    }
    if (!hdr_end) return n > 64 * 1024 ? -1 : 0;  // header cap 64K
    size_t hdr_len = size_t(hdr_end - data);
    std::string head(data, hdr_len);
    size_t le = head.find("\r\n");
    std::string reqline = le == std::string::npos ? head : head.substr(0, le);
    size_t sp1 = reqline.find(' ');
    size_t sp2 = reqline.rfind(' ');
    if (sp1 == std::string::npos || sp2 <= sp1) return -1;
    out.method = reqline.substr(0, sp1);
    std::string target = reqline.substr(sp1 + 1, sp2 - sp1 - 1);
    bool http10 = reqline.compare(sp2 + 1, std::string::npos, "HTTP/1.0") == 0;
    size_t q = target.find('?');
    out.path = q == std::string::npos ? target : target.substr(0, q);
    out.query = q == std::string::npos ? "" : target.substr(q + 1);

    // headers we care about: content-length, connection
    size_t content_length = 0;
    bool have_cl = false;
    keep_alive = !http10;
    size_t pos = le == std::string::npos ? hdr_len : le + 2;
    while (pos < hdr_len) {
        size_t eol = head.find("\r\n", pos);
        if (eol == std::string::npos) eol = hdr_len;
        std::string line = head.substr(pos, eol - pos);
        pos = eol + 2;
        size_t colon = line.find(':');
        if (colon == std::string::npos) continue;
        std::string key = line.substr(0, colon);
        for (auto& ch : key) ch = char(tolower(ch));
        size_t vs = colon + 1;
        while (vs < line.size() && line[vs] == ' ') ++vs;
        std::string val = line.substr(vs);
        if (key == "content-length") {
            have_cl = true;
            content_length = strtoull(val.c_str(), nullptr, 10);
        } else if (key == "connection") {
            for (auto& ch : val) ch = char(tolower(ch));
            if (val == "close") keep_alive = false;
            if (val == "keep-alive") keep_alive = true;
        }
    }
    if (content_length > MAX_BODY) return -1;
    if (out.method == "POST" && !have_cl) return -1;
    size_t total = hdr_len + 4 + content_length;
    if (n < total) return 0;
    out.body.assign(in.begin() + hdr_len + 4, in.begin() + total);
    in.erase(in.begin(), in.begin() + total);
    return 1;
}

void HttpServer::queue_response(uint64_t id, Response r) {
    {
        std::lock_guard<std::mutex> l(mu_);
        done_.emplace_back(id, std::move(r));
    }
    uint64_t v = 1;
    ssize_t rr = write(wake_fd_, &v, 8);
    (void)rr;
}

void HttpServer::worker_loop() {
    while (true) {
        std::pair<uint64_t, Request> job;
        {
            std::unique_lock<std::mutex> l(mu_);
            cv_.wait(l, [this] { return !running_ || !jobs_.empty(); });
            if (!running_ && jobs_.empty()) return;
            job = std::move(jobs_.front());
            jobs_.pop_front();
        }
        Response resp;
        try {
            resp = handler_(job.second);
        } catch (...) {
            resp.status = 500;
            resp.body.clear();
        }
        queue_response(job.first, std::move(resp));
    }
}

void HttpServer::close_conn(uint64_t id) {
    auto it = conns_.find(id);
    if (it == conns_.end()) return;
    epoll_ctl(epoll_fd_, EPOLL_CTL_DEL, it->second.fd, nullptr);
    ::close(it->second.fd);
    conns_.erase(it);
}

void HttpServer::io_loop() {
    std::vector<epoll_event> evs(256);
    while (running_) {
        int n = epoll_wait(epoll_fd_, evs.data(), int(evs.size()), 200);
        if (n < 0) {
            if (errno == EINTR) continue;
            break;
        }
        for (int i = 0; i < n && running_; ++i) {
            uint64_t id = evs[i].data.u64;
            if (id == 0) {
                // accept loop
                while (true) {
                    int fd = accept(listen_fd_, nullptr, nullptr);
                    if (fd < 0) break;
                    set_nonblock(fd);
                    int one = 1;
                    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
                    uint64_t cid = next_id_++;
                    Conn c;
                    c.fd = fd;
                    conns_.emplace(cid, std::move(c));
                    epoll_event ev{};
                    ev.events = EPOLLIN;
                    ev.data.u64 = cid;
                    epoll_ctl(epoll_fd_, EPOLL_CTL_ADD, fd, &ev);
                }
                continue;
            }
            if (id == UINT64_MAX) {
                uint64_t v;
                while (read(wake_fd_, &v, 8) == 8) {}
                // fall through: responses drained below
                continue;
            }
            auto it = conns_.find(id);
            if (it == conns_.end()) continue;
            Conn& c = it->second;
            if (evs[i].events & (EPOLLERR | EPOLLHUP)) {
                close_conn(id);
                continue;
            }
            if (evs[i].events & EPOLLIN) {
                uint8_t buf[64 * 1024];
                bool dead = false;
                while (true) {
                    ssize_t r = read(c.fd, buf, sizeof(buf));
                    if (r > 0) {
                        c.in.insert(c.in.end(), buf, buf + r);
                        if (c.in.size() > MAX_BODY + 64 * 1024) {
                            dead = true;
                            break;
                        }
                    } else if (r == 0) {
                        dead = true;
                        break;
                    } else {
                        if (errno == EAGAIN || errno == EWOULDBLOCK) break;
                        if (errno == EINTR) continue;
                        dead = true;
                        break;
                    }
                }
                if (dead && c.in.empty() && !c.processing) {
                    close_conn(id);
                    continue;
                }
                if (!c.processing) {
                    Request req;
                    int pr = try_parse(c, req);
                    if (pr < 0) {
                        Response bad;
                        bad.status = 400;
                        queue_response(id, std::move(bad));
                        c.processing = true;
                        c.closing = true;
                    } else if (pr == 1) {
                        c.processing = true;
                        {
                            std::lock_guard<std::mutex> l(mu_);
                            jobs_.emplace_back(id, std::move(req));
                        }
                        cv_.notify_one();
                    } else if (dead) {
                        close_conn(id);
                        continue;
                    }
                }
            }
            if (evs[i].events & EPOLLOUT) {
                // flush pending output
                while (c.out_off < c.out.size()) {
                    ssize_t w = write(c.fd, c.out.data() + c.out_off, c.out.size() - c.out_off);
                    if (w > 0)
                        c.out_off += size_t(w);
                    else if (w < 0 && (errno == EAGAIN || errno == EWOULDBLOCK))
                        break;
                    else if (w < 0 && errno == EINTR)
                        continue;
                    else {
                        close_conn(id);
                        goto next_event;
                    }
                }
                if (c.out_off >= c.out.size()) {
                    c.out.clear();
                    c.out_off = 0;
                    if (!c.keep_alive || c.closing) {
                        close_conn(id);
                        continue;
                    }
                    epoll_event ev{};
                    ev.events = EPOLLIN;
                    ev.data.u64 = id;
                    epoll_ctl(epoll_fd_, EPOLL_CTL_MOD, c.fd, &ev);
                }
            }
        next_event:;
        }

        // drain completed responses
        std::deque<std::pair<uint64_t, Response>> done;
        {
            std::lock_guard<std::mutex> l(mu_);
            done.swap(done_);
        }
        for (auto& [id, resp] : done) {
            auto it = conns_.find(id);
            if (it == conns_.end()) continue;
            Conn& c = it->second;
            char head[256];
            int hl = snprintf(head, sizeof(head),
                              "HTTP/1.1 %d %s\r\nContent-Type: %s\r\nContent-Length: %zu\r\n"
                              "Connection: %s\r\n\r\n",
                              resp.status, status_text(resp.status), resp.content_type,
                              resp.body.size(), (c.keep_alive && !c.closing) ? "keep-alive" : "close");
            c.out.insert(c.out.end(), head, head + hl);
            c.out.insert(c.out.end(), resp.body.begin(), resp.body.end());
            c.processing = false;
            // try immediate write; arm EPOLLOUT for the rest
            while (c.out_off < c.out.size()) {
                ssize_t w = write(c.fd, c.out.data() + c.out_off, c.out.size() - c.out_off);
                if (w > 0)
                    c.out_off += size_t(w);
                else
                    break;
            }
            if (c.out_off >= c.out.size()) {
                c.out.clear();
                c.out_off = 0;
                if (!c.keep_alive || c.closing) {
                    close_conn(id);
                    continue;
                }
                // another request may already be buffered (client pipelining not
                // supported, but a keep-alive client may have sent the next one)
                Request req;
                int pr = try_parse(c, req);
                if (pr == 1) {
                    c.processing = true;
                    {
                        std::lock_guard<std::mutex> l(mu_);
                        jobs_.emplace_back(id, std::move(req));
                    }
                    cv_.notify_one();
                } else if (pr < 0) {
                    close_conn(id);
                }
            } else {
                epoll_event ev{};
                ev.events = EPOLLIN | EPOLLOUT;
                ev.data.u64 = id;
                epoll_ctl(epoll_fd_, EPOLL_CTL_MOD, c.fd, &ev);
            }
        }
    }
}

// --------------------------------------------------------------- client

HttpClient::HttpClient(std::string host, uint16_t port, double timeout_s)
    : host_(std::move(host)), port_(port), timeout_s_(timeout_s) {}

HttpClient::~HttpClient() { close_(); }

void HttpClient::close_() {
    if (fd_ >= 0) ::close(fd_);
    fd_ = -1;
}

bool HttpClient::connect_() {
    close_();
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
        close_();
        return false;
    }
    return true;
}

bool HttpClient::send_all(const uint8_t* p, size_t n) {
    size_t off = 0;
    while (off < n) {
        ssize_t w = ::send(fd_, p + off, n - off, MSG_NOSIGNAL);
        if (w > 0)
            off += size_t(w);
        else if (w < 0 && errno == EINTR)
            continue;
        else
            return false;
    }
    return true;
}

bool HttpClient::do_request(const std::string& method, const std::string& pq, const Bytes* body,
                            int& status_out, Bytes& body_out) {
    char head[512];
    int hl = snprintf(head, sizeof(head),
                      "%s %s HTTP/1.1\r\nHost: %s\r\nContent-Length: %zu\r\n"
                      "Connection: keep-alive\r\n\r\n",
                      method.c_str(), pq.c_str(), host_.c_str(), body ? body->size() : 0);
    if (hl <= 0 || !send_all(reinterpret_cast<uint8_t*>(head), size_t(hl))) return false;
    if (body && !body->empty() && !send_all(body->data(), body->size())) return false;

    // read response
    Bytes buf;
    uint8_t tmp[64 * 1024];
    size_t hdr_end = 0;
    while (true) {
        // look for CRLFCRLF
        for (size_t i = hdr_end > 3 ? hdr_end - 3 : 0; i + 3 < buf.size(); ++i) {
            if (buf[i] == '\r' && buf[i + 1] == '\n' && buf[i + 2] == '\r' && buf[i + 3] == '\n') {
                hdr_end = i + 4;
                goto have_headers;
            }
        }
        {
            ssize_t r = recv(fd_, tmp, sizeof(tmp), 0);
            if (r <= 0) return false;
            hdr_end = buf.size();
            buf.insert(buf.end(), tmp, tmp + r);
        }
    }
have_headers: {
    std::string head_s(reinterpret_cast<const char*>(buf.data()), hdr_end);
    // status line: HTTP/1.1 NNN ...
    size_t sp = head_s.find(' ');
    if (sp == std::string::npos) return false;
    status_out = atoi(head_s.c_str() + sp + 1);
    size_t content_length = 0;
    bool keep = true;
    {
        std::string lower = head_s;
        for (auto& ch : lower) ch = char(tolower(ch));
        size_t cl = lower.find("content-length:");
        if (cl != std::string::npos) content_length = strtoull(lower.c_str() + cl + 15, nullptr, 10);
        if (lower.find("connection: close") != std::string::npos) keep = false;
    }
    if (content_length > MAX_RESPONSE) return false;
    while (buf.size() < hdr_end + content_length) {
        ssize_t r = recv(fd_, tmp, sizeof(tmp), 0);
        if (r <= 0) return false;
        buf.insert(buf.end(), tmp, tmp + r);
    }
    body_out.assign(buf.begin() + hdr_end, buf.begin() + hdr_end + content_length);
    if (!keep) close_();
    return true;
}
}

bool HttpClient::request(const std::string& method, const std::string& pq, const Bytes* body,
                         int& status_out, Bytes& body_out) {
    if (fd_ < 0 && !connect_()) return false;
    if (do_request(method, pq, body, status_out, body_out)) return true;
    // stale keep-alive or transient failure: reconnect once
    if (!connect_()) return false;
    return do_request(method, pq, body, status_out, body_out);
}

}  // namespace xaynet::http
