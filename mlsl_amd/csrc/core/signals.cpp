#include "signals.hpp"

#include <csignal>
#include <cstdlib>
#include <cstring>
#include <unistd.h>

#include "log.hpp"

namespace mlsl {

namespace {

const int kSignals[] = {SIGSEGV, SIGBUS, SIGILL, SIGABRT, SIGFPE, SIGTERM};
struct sigaction g_old[sizeof(kSignals) / sizeof(kSignals[0])];
bool g_installed = false;

void Handler(int sig) {
    // async-signal-unsafe calls are acceptable here: the process is dying
    // anyway; the goal is diagnostics + prompt exit so peers unblock.
    std::fprintf(stderr, "[mlsl] fatal signal %d (%s)\n", sig, strsignal(sig));
    PrintBacktrace();
    RestoreSignalHandlers();
    _exit(128 + sig);
}

}  // namespace

void InstallSignalHandlers() {
    if (g_installed) return;
    if (const char* e = std::getenv("MLSL_HANDLE_SIGNALS"))
        if (std::atoi(e) == 0) return;
    struct sigaction sa;
    std::memset(&sa, 0, sizeof(sa));
    sa.sa_handler = Handler;
    sigemptyset(&sa.sa_mask);
    for (size_t i = 0; i < sizeof(kSignals) / sizeof(kSignals[0]); ++i)
        sigaction(kSignals[i], &sa, &g_old[i]);
    g_installed = true;
}

void RestoreSignalHandlers() {
    if (!g_installed) return;
    for (size_t i = 0; i < sizeof(kSignals) / sizeof(kSignals[0]); ++i)
        sigaction(kSignals[i], &g_old[i], nullptr);
    g_installed = false;
}

}  // namespace mlsl
