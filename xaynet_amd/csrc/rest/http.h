// Native HTTP/1.1 plumbing for the coordinator REST API and the SDK client.
//
// Server: one epoll IO thread + a bounded worker pool (the concurrency-limit
// analog of the reference's rayon decrypt pool, rest.rs + decryptor.rs).
// Handlers may block (POST /message awaits the protocol thread's oneshot
// reply), so they run on workers, never on the IO thread; each connection
// processes one request at a time (no pipelining), keep-alive supported.
//
// Client: blocking keep-alive connection with one reconnect retry, used by
// the SDK's HttpXaynetClient (reference xaynet-sdk/src/client.rs).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <map>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "../common.h"

namespace xaynet::http {

struct Request {
    std::string method;   // "GET" / "POST"
    std::string path;     // "/seeds"
    std::string query;    // raw query string after '?', percent-encoded
    Bytes body;
};

struct Response {
    int status = 200;
    Bytes body;
    const char* content_type = "application/octet-stream";
};

// percent-decode a URL component (query value); '+' is NOT space-decoded
// (matches the url crate's form used by the reference client).
std::string percent_decode(const std::string& s);
// first value for `key` in a query string, percent-decoded
std::string query_get(const std::string& query, const std::string& key);

std::string base64_encode(const uint8_t* p, size_t n);
bool base64_decode(const std::string& s, Bytes& out);

// parse one complete HTTP/1.1 request from the front of `in`, consuming it;
// returns 1 ok, 0 incomplete, -1 malformed. Shared by the epoll and TLS
// servers.
int parse_http_request(Bytes& in, Request& out, bool& keep_alive);

class HttpServer {
  public:
    using Handler = std::function<Response(const Request&)>;

    HttpServer(Handler handler, std::string host, uint16_t port, int workers = 4);
    ~HttpServer();

    // bind+listen+spawn threads; returns false on bind failure
    bool start();
    void stop();
    uint16_t port() const { return port_; }  // actual port (when constructed with 0)

  private:
    struct Conn {
        int fd = -1;
        Bytes in;
        Bytes out;
        size_t out_off = 0;
        bool processing = false;  // request dispatched, awaiting response
        bool keep_alive = true;
        bool closing = false;
    };

    void io_loop();
    void worker_loop();
    void close_conn(uint64_t id);
    // parse one complete request out of c.in; returns 1 ok, 0 incomplete, -1 bad
    int try_parse(Conn& c, Request& out);
    void queue_response(uint64_t id, Response r);

    Handler handler_;
    std::string host_;
    uint16_t port_;
    int n_workers_;

    int listen_fd_ = -1, epoll_fd_ = -1, wake_fd_ = -1;
    std::atomic<bool> running_{false};
    std::thread io_thread_;
    std::vector<std::thread> workers_;

    std::mutex mu_;
    std::condition_variable cv_;
    std::deque<std::pair<uint64_t, Request>> jobs_;
    std::deque<std::pair<uint64_t, Response>> done_;  // drained by IO thread

    std::map<uint64_t, Conn> conns_;  // IO thread only
    uint64_t next_id_ = 1;
};

class HttpClient {
  public:
    HttpClient(std::string host, uint16_t port, double timeout_s = 30.0);
    ~HttpClient();

    // returns false on transport failure (after one reconnect retry)
    bool request(const std::string& method, const std::string& path_and_query,
                 const Bytes* body, int& status_out, Bytes& body_out);

  private:
    bool connect_();
    bool send_all(const uint8_t* p, size_t n);
    bool do_request(const std::string& method, const std::string& path_and_query,
                    const Bytes* body, int& status_out, Bytes& body_out);
    void close_();

    std::string host_;
    uint16_t port_;
    double timeout_s_;
    int fd_ = -1;
};

}  // namespace xaynet::http
