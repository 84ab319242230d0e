/*
 * binder-amd: static tracepoints.
 *
 * The reference fires DTrace USDT probes op-req-start / op-req-done
 * around every query (lib/server.js:24-29, 472-474, 516-518). On Linux
 * the equivalent is systemtap SDT probes — but <sys/sdt.h> is absent
 * from this build image, so the macros compile away unless BAMD_SDT is
 * defined at build time on a host that has systemtap-sdt-dev. The
 * always-available替代 observability for the same events is the
 * per-query log line with phase timers (server.cpp afterQuery).
 */
#pragma once

#if defined(BAMD_SDT) && __has_include(<sys/sdt.h>)
#include <sys/sdt.h>
#define BAMD_PROBE2(name, a, b) DTRACE_PROBE2(binder, name, a, b)
#else
#define BAMD_PROBE2(name, a, b) \
    do {                        \
    } while (0)
#endif
