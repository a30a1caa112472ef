#include "metrics.h"

#include <chrono>
#include <cstdio>

#include "../rest/http.h"

namespace xaynet::metrics {

const char* measurement_name(Measurement m) {
    switch (m) {
        case Measurement::RoundParamSum: return "round_param_sum";
        case Measurement::RoundParamUpdate: return "round_param_update";
        case Measurement::Phase: return "phase";
        case Measurement::MasksTotalNumber: return "masks_total_number";
        case Measurement::RoundTotalNumber: return "round_total_number";
        case Measurement::MessageAccepted: return "message_accepted";
        case Measurement::MessageDiscarded: return "message_discarded";
        case Measurement::MessageRejected: return "message_rejected";
    }
    return "unknown";
}

static std::mutex g_mu;
static std::unique_ptr<Recorder> g_recorder;

void Recorder::install_file(const std::string& path, const std::string&) {
    auto file = std::make_shared<std::string>(path);
    install_sink([file](const std::string& line) {
        FILE* f = fopen(file->c_str(), "ab");
        if (!f) return;
        fwrite(line.data(), 1, line.size(), f);
        fputc('\n', f);
        fclose(f);
    });
}

void Recorder::install_influxdb(const std::string& host, uint16_t port, const std::string& db) {
    // InfluxDB 1.x line-protocol writer: POST /write?db=<db>, one point per
    // request (the reference's per-metric tower dispatch,
    // recorders/influxdb/service.rs:10-16). A slow or down endpoint blocks
    // only the writer thread; the bounded queue then sheds load — the same
    // lossy-under-pressure semantics as the reference's LoadShed/Buffer.
    auto client = std::make_shared<http::HttpClient>(host, port, 5.0);
    std::string path = "/write?db=" + db;
    install_sink([client, path](const std::string& line) {
        Bytes body(line.begin(), line.end());
        int status = 0;
        Bytes resp;
        client->request("POST", path, &body, status, resp);
        // 204 = accepted; failures are dropped silently (lossy by design)
    });
}

void Recorder::install_sink(Sink sink) {
    std::lock_guard<std::mutex> l(g_mu);
    g_recorder = std::make_unique<Recorder>(std::move(sink));
}

void Recorder::uninstall() {
    std::lock_guard<std::mutex> l(g_mu);
    g_recorder.reset();
}

Recorder* Recorder::global() {
    // benign race with install/uninstall at startup/shutdown only
    return g_recorder.get();
}

Recorder::Recorder(Sink sink) : sink_(std::move(sink)) {
    writer_ = std::thread([this] { writer_loop(); });
}

Recorder::~Recorder() {
    running_ = false;
    cv_.notify_all();
    if (writer_.joinable()) writer_.join();
}

void Recorder::record(Measurement m, double value, uint64_t round_id, int phase_id) {
    record_tagged(m, value, round_id, phase_id, "");
}

void Recorder::record_tagged(Measurement m, double value, uint64_t round_id, int phase_id,
                             const std::string& extra_tags) {
    auto ns = std::chrono::duration_cast<std::chrono::nanoseconds>(
                  std::chrono::system_clock::now().time_since_epoch())
                  .count();
    char buf[256];
    int n = snprintf(buf, sizeof(buf), "%s,round_id=%llu,phase=%d%s%s value=%g %lld",
                     measurement_name(m), (unsigned long long)round_id, phase_id,
                     extra_tags.empty() ? "" : ",", extra_tags.c_str(), value, (long long)ns);
    if (n <= 0) return;
    {
        std::lock_guard<std::mutex> l(mu_);
        if (queue_.size() >= MAX_QUEUE) {
            dropped_.fetch_add(1);
            return;  // lossy under pressure, by design
        }
        queue_.emplace_back(buf, size_t(n));
    }
    cv_.notify_one();
}

void Recorder::flush() {
    std::unique_lock<std::mutex> l(mu_);
    cv_.wait_for(l, std::chrono::seconds(5), [this] { return queue_.empty(); });
}

void Recorder::writer_loop() {
    while (true) {
        std::string line;
        {
            std::unique_lock<std::mutex> l(mu_);
            cv_.wait(l, [this] { return !running_ || !queue_.empty(); });
            // shutdown DISCARDS pending points: draining a backlog through a
            // slow sink would block teardown for minutes (metrics are lossy
            // by design; call flush() first when a test needs determinism)
            if (!running_) {
                dropped_.fetch_add(queue_.size());
                queue_.clear();
                return;
            }
            line = std::move(queue_.front());
            queue_.pop_front();
            if (queue_.empty()) cv_.notify_all();  // wake flush()
        }
        try {
            sink_(line);
        } catch (...) {
            dropped_.fetch_add(1);
        }
    }
}

}  // namespace xaynet::metrics
