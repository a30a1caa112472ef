#include "trace.h"

#include <cstdio>
#include <memory>
#include <mutex>

namespace xaynet::trace {

static std::atomic<bool> g_enabled{false};
static std::mutex g_mu;
static std::shared_ptr<Sink> g_sink;  // shared_ptr: emitters may race uninstall
static std::atomic<uint64_t> g_next_id{1};
static thread_local uint64_t t_current = 0;

void install(Sink sink) {
    std::lock_guard<std::mutex> l(g_mu);
    g_sink = std::make_shared<Sink>(std::move(sink));
    g_enabled.store(true, std::memory_order_release);
}

void install_file(const std::string& path) {
    auto file = std::make_shared<std::string>(path);
    install([file](const std::string& line) {
        FILE* f = fopen(file->c_str(), "ab");
        if (!f) return;
        fwrite(line.data(), 1, line.size(), f);
        fputc('\n', f);
        fclose(f);
    });
}

void uninstall() {
    std::lock_guard<std::mutex> l(g_mu);
    g_enabled.store(false, std::memory_order_release);
    g_sink.reset();
}

bool enabled() { return g_enabled.load(std::memory_order_relaxed); }

uint64_t current_span() { return t_current; }

Span::Span(const char* name, std::string fields, uint64_t explicit_parent) {
    if (!enabled()) return;
    live_ = true;
    name_ = name;
    fields_ = std::move(fields);
    id_ = g_next_id.fetch_add(1, std::memory_order_relaxed);
    parent_ = explicit_parent == UINT64_MAX ? t_current : explicit_parent;
    prev_current_ = t_current;
    t_current = id_;
    t0_ = std::chrono::steady_clock::now();
}

void Span::add(const std::string& more) {
    if (!live_) return;
    if (!fields_.empty()) fields_ += ' ';
    fields_ += more;
}

Span::~Span() {
    if (!live_) return;
    t_current = prev_current_;
    auto us = std::chrono::duration_cast<std::chrono::microseconds>(
                  std::chrono::steady_clock::now() - t0_)
                  .count();
    std::shared_ptr<Sink> sink;
    {
        std::lock_guard<std::mutex> l(g_mu);
        sink = g_sink;
    }
    if (!sink) return;
    char head[96];
    snprintf(head, sizeof(head), "span=%llu parent=%llu name=", (unsigned long long)id_,
             (unsigned long long)parent_);
    std::string line = head;
    line += name_;
    if (!fields_.empty()) {
        line += ' ';
        line += fields_;
    }
    char tail[40];
    snprintf(tail, sizeof(tail), " dur_us=%lld", (long long)us);
    line += tail;
    try {
        (*sink)(line);
    } catch (...) {
    }
}

}  // namespace xaynet::trace
