/* LD_PRELOAD SIGSEGV backtrace helper for GPU-box debugging (no gdb there). */
#define _GNU_SOURCE
#include <execinfo.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <unistd.h>
static void handler(int sig) {
    void* bt[48];
    int n = backtrace(bt, 48);
    dprintf(2, "=== segtrace: signal %d ===\n", sig);
    backtrace_symbols_fd(bt, n, 2);
    _exit(139);
}
__attribute__((constructor)) static void seg_init(void) {
    struct sigaction sa = {0};
    sa.sa_handler = handler;
    sigaction(SIGSEGV, &sa, 0);
    sigaction(SIGBUS, &sa, 0);
}
