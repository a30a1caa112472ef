// Structured tracing spans threading request -> phase.
//
// Mirrors the reference's tracing usage: a span per phase run
// (rust/xaynet-server/src/state_machine/phases/phase.rs:148) and a span
// carried through the request channel so the protocol-thread handling of a
// message is parented to its ingest span
// (rust/xaynet-server/src/state_machine/requests.rs:120).
//
// Design: logfmt lines ("span=<id> parent=<id> name=<..> <fields> dur_us=<n>")
// pushed to a pluggable sink on span END. Disabled by default: a single
// relaxed atomic load gates every call site. Cross-thread parenting is
// explicit (the span id travels in the request struct), same-thread nesting
// via a thread-local current-span id.
#pragma once

#include <atomic>
#include <chrono>
#include <cstdint>
#include <functional>
#include <string>

namespace xaynet::trace {

using Sink = std::function<void(const std::string& line)>;

void install(Sink sink);
void install_file(const std::string& path);
void uninstall();
bool enabled();

// current thread's active span id (0 = none)
uint64_t current_span();

// RAII span: emits one line at destruction. `fields` is pre-formatted
// logfmt (e.g. "phase=Sum round=3"); parent defaults to the thread's
// current span, or pass an explicit id for cross-thread edges.
class Span {
  public:
    Span(const char* name, std::string fields = "", uint64_t explicit_parent = UINT64_MAX);
    ~Span();
    Span(const Span&) = delete;
    Span& operator=(const Span&) = delete;

    uint64_t id() const { return id_; }
    void add(const std::string& more);  // append fields after construction

  private:
    uint64_t id_ = 0;
    uint64_t parent_ = 0;
    uint64_t prev_current_ = 0;
    const char* name_ = nullptr;
    std::string fields_;
    std::chrono::steady_clock::time_point t0_;
    bool live_ = false;
};

}  // namespace xaynet::trace
