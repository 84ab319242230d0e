/*
 * binder-amd: static tracepoints (USDT).
 *
 * The reference fires DTrace USDT probes op-req-start / op-req-done
 * around every query (lib/server.js:24-29, 472-474, 516-518). The
 * Linux equivalent is a systemtap SDT probe: a nop at the probe site
 * plus a .note.stapsdt ELF note describing its address, provider,
 * name, and argument locations. <sys/sdt.h> is absent from this build
 * image, so the note emitter below is written from scratch against
 * the documented stapsdt note layout (type 3, name "stapsdt", desc =
 * {probe addr, link-time base, semaphore addr, provider\0 name\0
 * argfmt\0}); bpftrace/perf/systemtap all consume this format.
 *
 * Zero cost when untraced: one nop on the hot path; no semaphore.
 * Verify with:  readelf -n bin/binderd   (docs/OPERATIONS.md shows the
 * bpftrace usage).
 */
#pragma once

#include <cstdint>

#if defined(__x86_64__) && defined(__ELF__) && !defined(BAMD_NO_SDT)

/* Link-time base symbol: lets consumers compute prelink/PIE offsets.
 * Emitted once per TU (assembler-level .ifndef guard). */
#define BAMD_SDT_BASE                                                  \
    ".ifndef _.stapsdt.base\n"                                         \
    ".pushsection .stapsdt.base,\"aG\",\"progbits\",.stapsdt.base,"    \
    "comdat\n"                                                         \
    ".weak _.stapsdt.base\n"                                           \
    ".hidden _.stapsdt.base\n"                                         \
    "_.stapsdt.base: .space 1\n"                                       \
    ".size _.stapsdt.base, 1\n"                                        \
    ".popsection\n"                                                    \
    ".endif\n"

#define BAMD_PROBE2(name, a, b)                                        \
    do {                                                               \
        __asm__ __volatile__(                                          \
            "990: nop\n"                                               \
            ".pushsection .note.stapsdt,\"\",\"note\"\n"               \
            ".balign 4\n"                                              \
            ".4byte 992f-991f, 994f-993f, 3\n"                         \
            "991: .asciz \"stapsdt\"\n"                                \
            "992: .balign 4\n"                                         \
            "993: .8byte 990b\n"                                       \
            ".8byte _.stapsdt.base\n"                                  \
            ".8byte 0\n" /* no semaphore: always-on nop site */        \
            ".asciz \"binder\"\n"                                      \
            ".asciz \"" name "\"\n"                                    \
            ".asciz \"-8@%0 -8@%1\"\n"                                 \
            "994: .balign 4\n"                                         \
            ".popsection\n" BAMD_SDT_BASE                              \
            :                                                          \
            : "nor"((int64_t)(a)), "nor"((int64_t)(b)));               \
    } while (0)

#else /* non-x86-64/ELF fallback: compiled away */

#define BAMD_PROBE2(name, a, b) \
    do {                        \
    } while (0)

#endif
