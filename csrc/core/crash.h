// Crash diagnostics: install SIGSEGV/SIGABRT/SIGBUS/SIGFPE/SIGILL handlers
// that print a backtrace before exiting (role of the reference's
// boost::stacktrace signal handler, utils.cpp:115-122 — implemented with
// glibc execinfo instead; boost is not a dependency here).
#pragma once

#include <csignal>
#include <cstdio>
#include <cstdlib>
#include <execinfo.h>
#include <unistd.h>

namespace ifs {

inline void crash_handler(int sig) {
    void* frames[64];
    int n = backtrace(frames, 64);
    fprintf(stderr, "\n[ifs] fatal signal %d — backtrace (%d frames):\n", sig, n);
    backtrace_symbols_fd(frames, n, STDERR_FILENO);
    signal(sig, SIG_DFL);
    raise(sig);
}

inline void install_crash_handlers() {
    for (int sig : {SIGSEGV, SIGABRT, SIGBUS, SIGFPE, SIGILL}) signal(sig, crash_handler);
}

}  // namespace ifs
