#include "log.hpp"

#include <execinfo.h>
#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstdlib>
#include <mutex>

namespace mlsl {

static std::atomic<int> g_level{-1};
static std::mutex g_log_mu;

LogLevel GetLogLevel() {
    int lvl = g_level.load(std::memory_order_relaxed);
    if (lvl < 0) {
        lvl = 0;
        if (const char* e = std::getenv("MLSL_LOG_LEVEL")) lvl = std::atoi(e);
        if (lvl < 0) lvl = 0;
        if (lvl > 3) lvl = 3;
        g_level.store(lvl, std::memory_order_relaxed);
    }
    return static_cast<LogLevel>(lvl);
}

void SetLogLevel(LogLevel lvl) {
    g_level.store(static_cast<int>(lvl), std::memory_order_relaxed);
}

static const char* LevelName(LogLevel lvl) {
    switch (lvl) {
        case LogLevel::ERROR: return "ERROR";
        case LogLevel::INFO: return "INFO";
        case LogLevel::DEBUG: return "DEBUG";
        case LogLevel::TRACE: return "TRACE";
    }
    return "?";
}

void LogWrite(LogLevel lvl, const char* func, int line, const char* fmt, ...) {
    using namespace std::chrono;
    auto now = system_clock::now();
    auto us = duration_cast<microseconds>(now.time_since_epoch()).count();
    long tid = static_cast<long>(syscall(SYS_gettid));

    std::lock_guard<std::mutex> lk(g_log_mu);
    std::fprintf(stderr, "[mlsl %s %lld.%06lld tid=%ld %s:%d] ", LevelName(lvl),
                 static_cast<long long>(us / 1000000),
                 static_cast<long long>(us % 1000000), tid, func, line);
    va_list ap;
    va_start(ap, fmt);
    std::vfprintf(stderr, fmt, ap);
    va_end(ap);
    std::fputc('\n', stderr);
    if (lvl == LogLevel::ERROR) PrintBacktrace();
}

void PrintBacktrace() {
    void* frames[32];
    int n = backtrace(frames, 32);
    char** syms = backtrace_symbols(frames, n);
    if (!syms) return;
    std::fprintf(stderr, "[mlsl backtrace, %d frames]\n", n);
    for (int i = 0; i < n; ++i) std::fprintf(stderr, "  #%d %s\n", i, syms[i]);
    std::free(syms);
}

}  // namespace mlsl
