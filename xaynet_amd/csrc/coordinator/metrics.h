// Metrics: InfluxDB line-protocol recorder behind a lossy bounded queue.
//
// Mirrors the reference's metrics stack (rust/xaynet-server/src/metrics/):
// the 8 Measurement kinds (recorders/influxdb/models.rs:7-31), round/phase
// tags, and the by-design lossy dispatch (metrics are dropped, never block
// the protocol thread, service.rs:10-16). The sink is pluggable: a file
// (append, one line-protocol line per point — the no-network stand-in for
// the InfluxDB HTTP endpoint) or a callback (tests).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <string>
#include <thread>

namespace xaynet::metrics {

enum class Measurement {
    RoundParamSum,
    RoundParamUpdate,
    Phase,
    MasksTotalNumber,
    RoundTotalNumber,
    MessageAccepted,
    MessageDiscarded,
    MessageRejected,
};

const char* measurement_name(Measurement m);

class Recorder {
  public:
    using Sink = std::function<void(const std::string& line)>;

    // global recorder (reference GlobalRecorder OnceCell, metrics/mod.rs:12-80);
    // null until installed — the metric() free function is a no-op then.
    static void install_file(const std::string& path, const std::string& db = "metrics");
    // line-protocol HTTP writer: POST /write?db=<db> to an InfluxDB 1.x
    // endpoint (reference recorders/influxdb/)
    static void install_influxdb(const std::string& host, uint16_t port,
                                 const std::string& db = "metrics");
    static void install_sink(Sink sink);
    static void uninstall();
    static Recorder* global();

    explicit Recorder(Sink sink);
    ~Recorder();

    // non-blocking; drops the point when the queue is full (lossy by design)
    void record(Measurement m, double value, uint64_t round_id, int phase_id);
    void record_tagged(Measurement m, double value, uint64_t round_id, int phase_id,
                       const std::string& extra_tags);

    size_t dropped() const { return dropped_.load(); }
    void flush();

  private:
    void writer_loop();

    Sink sink_;
    std::mutex mu_;
    std::condition_variable cv_;
    std::deque<std::string> queue_;
    std::atomic<size_t> dropped_{0};
    std::atomic<bool> running_{true};
    std::thread writer_;
    static constexpr size_t MAX_QUEUE = 4048;  // reference Buffer<4048>
};

// convenience free function (reference `metric!` macro)
inline void metric(Measurement m, double value, uint64_t round_id, int phase_id) {
    if (Recorder* r = Recorder::global()) r->record(m, value, round_id, phase_id);
}

}  // namespace xaynet::metrics
