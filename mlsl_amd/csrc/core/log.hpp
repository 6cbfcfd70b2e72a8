// Logging with levels, timestamps, tid and backtrace-on-error.
// Capability parity with reference src/log.{hpp,cpp} (MLSL_LOG levels
// ERROR/INFO/DEBUG/TRACE, env MLSL_LOG_LEVEL, PrintBacktrace).
#pragma once

#include <cstdarg>
#include <cstdio>

namespace mlsl {

enum class LogLevel : int { ERROR = 0, INFO = 1, DEBUG = 2, TRACE = 3 };

LogLevel GetLogLevel();
void SetLogLevel(LogLevel lvl);
void LogWrite(LogLevel lvl, const char* func, int line, const char* fmt, ...)
    __attribute__((format(printf, 4, 5)));
void PrintBacktrace();

#define MLSL_LOG(lvl, ...)                                                     \
    do {                                                                       \
        if (static_cast<int>(::mlsl::LogLevel::lvl) <=                         \
            static_cast<int>(::mlsl::GetLogLevel()))                           \
            ::mlsl::LogWrite(::mlsl::LogLevel::lvl, __func__, __LINE__,        \
                             __VA_ARGS__);                                     \
    } while (0)

}  // namespace mlsl
