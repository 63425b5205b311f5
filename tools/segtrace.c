/* Minimal SIGSEGV/SIGABRT native-backtrace preload (the GPU boxes have no
 * gdb and glibc>=2.35 dropped libSegFault). Build:
 *   gcc -shared -fPIC -O1 tools/segtrace.c -o tools/segtrace.so -ldl
 * Use: LD_PRELOAD=$PWD/tools/segtrace.so python ...  (writes to stderr)
 */
#define _GNU_SOURCE
#include <execinfo.h>
#include <signal.h>
#include <stdio.h>
#include <string.h>
#include <unistd.h>

static void handler(int sig, siginfo_t *si, void *ctx) {
  (void)ctx;
  char head[128];
  int n = snprintf(head, sizeof(head),
                   "\n=== segtrace: signal %d at addr %p ===\n", sig,
                   si ? si->si_addr : 0);
  (void)!write(2, head, (size_t)n);
  void *frames[64];
  int depth = backtrace(frames, 64);
  backtrace_symbols_fd(frames, depth, 2);
  (void)!write(2, "=== segtrace end ===\n", 21);
  signal(sig, SIG_DFL);
  raise(sig);
}

__attribute__((constructor)) static void install(void) {
  struct sigaction sa;
  memset(&sa, 0, sizeof(sa));
  sa.sa_sigaction = handler;
  sa.sa_flags = SA_SIGINFO | SA_ONSTACK;
  sigaction(SIGSEGV, &sa, 0);
  sigaction(SIGABRT, &sa, 0);
  sigaction(SIGBUS, &sa, 0);
}
