/* LD_PRELOAD helper: print a native backtrace on SIGSEGV (debugging aid for
 * the GPU box where gdb is unavailable). Build:
 *   gcc -shared -fPIC -o /tmp/segv.so tools/segv_bt.c
 * Use: LD_PRELOAD=/tmp/segv.so python ... */
#define _GNU_SOURCE
#include <execinfo.h>
#include <signal.h>
#include <stdio.h>
#include <string.h>
#include <unistd.h>

static void handler(int sig) {
    void* frames[64];
    int n = backtrace(frames, 64);
    dprintf(2, "=== SIGSEGV native backtrace (%d frames) ===\n", n);
    backtrace_symbols_fd(frames, n, 2);
    _exit(139);
}

__attribute__((constructor)) static void install(void) {
    struct sigaction sa;
    memset(&sa, 0, sizeof sa);
    sa.sa_handler = handler;
    sa.sa_flags = SA_ONSTACK;
    sigaction(SIGSEGV, &sa, NULL);
}
