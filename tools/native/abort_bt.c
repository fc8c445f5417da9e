/* LD_PRELOAD shim: print a native backtrace when abort() fires (debugging
   a silent SIGABRT on GPU boxes; see docs/notes-round3.md). */
#define _GNU_SOURCE
#include <execinfo.h>
#include <stdio.h>
#include <stdlib.h>
#include <unistd.h>
#include <dlfcn.h>

void abort(void) {
    void *bt[64];
    int n = backtrace(bt, 64);
    fprintf(stderr, "\n=== abort() intercepted; native backtrace (%d) ===\n", n);
    backtrace_symbols_fd(bt, n, 2);
    fflush(NULL);
    void (*real)(void) = dlsym(RTLD_NEXT, "abort");
    real();
    _exit(134);
}
