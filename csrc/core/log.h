// Minimal leveled logger for infinistore-amd (replaces the reference's spdlog
// sink, cf. /root/reference/src/log.h:11-26 — reimplemented from scratch, no
// external deps; spdlog is not available in this image).
#pragma once

#include <cstdarg>
#include <cstdio>

namespace ifs {

enum class LogLevel : int { kDebug = 0, kInfo = 1, kWarn = 2, kError = 3 };

LogLevel log_level();
void set_log_level(LogLevel lvl);
// Accepts "debug"/"info"/"warning"/"error"; returns false on unknown name.
bool set_log_level(const char* name);

void log_vprintf(LogLevel lvl, const char* file, int line, const char* fmt, va_list ap);
void log_printf(LogLevel lvl, const char* file, int line, const char* fmt, ...)
    __attribute__((format(printf, 4, 5)));

}  // namespace ifs

#define IFS_LOG(lvl, ...)                                               \
    do {                                                                \
        if (static_cast<int>(lvl) >= static_cast<int>(ifs::log_level())) \
            ifs::log_printf(lvl, __FILE__, __LINE__, __VA_ARGS__);      \
    } while (0)

#define DEBUG(...) IFS_LOG(ifs::LogLevel::kDebug, __VA_ARGS__)
#define INFO(...) IFS_LOG(ifs::LogLevel::kInfo, __VA_ARGS__)
#define WARN(...) IFS_LOG(ifs::LogLevel::kWarn, __VA_ARGS__)
#define ERROR(...) IFS_LOG(ifs::LogLevel::kError, __VA_ARGS__)
