#include "log.h"

#include <atomic>
#include <cstring>
#include <ctime>
#include <mutex>

namespace ifs {

static std::atomic<int> g_level{static_cast<int>(LogLevel::kWarn)};
static std::mutex g_mu;

LogLevel log_level() { return static_cast<LogLevel>(g_level.load(std::memory_order_relaxed)); }

void set_log_level(LogLevel lvl) { g_level.store(static_cast<int>(lvl), std::memory_order_relaxed); }

bool set_log_level(const char* name) {
    if (!name) return false;
    if (!strcmp(name, "debug")) set_log_level(LogLevel::kDebug);
    else if (!strcmp(name, "info")) set_log_level(LogLevel::kInfo);
    else if (!strcmp(name, "warning") || !strcmp(name, "warn")) set_log_level(LogLevel::kWarn);
    else if (!strcmp(name, "error")) set_log_level(LogLevel::kError);
    else return false;
    return true;
}

static const char* lvl_name(LogLevel l) {
    switch (l) {
        case LogLevel::kDebug: return "debug";
        case LogLevel::kInfo: return "info";
        case LogLevel::kWarn: return "warn";
        case LogLevel::kError: return "error";
    }
    return "?";
}

void log_vprintf(LogLevel lvl, const char* file, int line, const char* fmt, va_list ap) {
    char msg[2048];
    vsnprintf(msg, sizeof(msg), fmt, ap);
    struct timespec ts;
    clock_gettime(CLOCK_REALTIME, &ts);
    struct tm tmv;
    localtime_r(&ts.tv_sec, &tmv);
    char tbuf[32];
    strftime(tbuf, sizeof(tbuf), "%H:%M:%S", &tmv);
    const char* base = strrchr(file, '/');
    base = base ? base + 1 : file;
    std::lock_guard<std::mutex> lk(g_mu);
    if (lvl >= LogLevel::kWarn)
        fprintf(stderr, "[%s.%03ld][ifs][%s] %s (%s:%d)\n", tbuf, ts.tv_nsec / 1000000,
                lvl_name(lvl), msg, base, line);
    else
        fprintf(stderr, "[%s.%03ld][ifs][%s] %s\n", tbuf, ts.tv_nsec / 1000000, lvl_name(lvl), msg);
    fflush(stderr);
}

void log_printf(LogLevel lvl, const char* file, int line, const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    log_vprintf(lvl, file, line, fmt, ap);
    va_end(ap);
}

}  // namespace ifs
