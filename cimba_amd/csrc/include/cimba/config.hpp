// cimba_amd — MI355X-native discrete-event simulation engine.
//
// config.hpp: portability layer between host C++ and HIP device code.
//
// Design note (not a port): the reference (ambonvik/cimba) keeps its engine
// in C17 with x86-64 assembly context switches (reference src/cmi_config.h,
// src/port/x86-64/linux/cmi_coroutine_context.asm).  On gfx950 there is no
// stackful coroutine, so the whole engine is written once in C++17 that
// compiles both as host code (CPU executive, unit tests) and as HIP device
// code (one simulation trial per wavefront, engine state in LDS).  The only
// divergence points between host and device are collected here.
#pragma once

#include <stdint.h>
#include <stddef.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#include <hip/hip_runtime.h>
#define CMB_HD __host__ __device__
#define CMB_DEV __device__
#else
#define CMB_HD
#define CMB_DEV
#endif

#define CMB_INLINE CMB_HD inline
#if defined(__HIPCC__)
#define CMB_FORCEINLINE __host__ __device__ __attribute__((always_inline)) inline
#else
#define CMB_FORCEINLINE inline __attribute__((always_inline))
#endif

namespace cmb {

// ---------------------------------------------------------------------------
// Assertions.  The reference has a three-tier assert system
// (reference include/cmb_assert.h:45-84: cmb_assert_debug / _release /
// _always).  We keep the same three tiers.  On the device an assert failure
// records an error code in the engine and aborts the trial (the per-block
// trial-abort flag of SURVEY.md §5.3); on the host it aborts the process.
// ---------------------------------------------------------------------------

#if defined(__HIP_DEVICE_COMPILE__)
CMB_DEV inline void cmb_assert_fail_impl(const char* /*expr*/, const char* /*file*/, int /*line*/) {
    // Device-side hard assert: trap the wavefront.  Engine-level recoverable
    // errors go through Engine::fail() instead; this is for contract bugs.
    __builtin_trap();
}
#else
[[noreturn]] void cmb_assert_fail_impl(const char* expr, const char* file, int line);
#endif

#define cmb_assert_always(expr) \
    do { if (!(expr)) ::cmb::cmb_assert_fail_impl(#expr, __FILE__, __LINE__); } while (0)

#if defined(NASSERT)
#define cmb_assert_release(expr) ((void)0)
#else
#define cmb_assert_release(expr) cmb_assert_always(expr)
#endif

#if defined(NDEBUG)
#define cmb_assert_debug(expr) ((void)0)
#else
#define cmb_assert_debug(expr) cmb_assert_release(expr)
#endif

#define cmb_unused(x) ((void)(x))

// ---------------------------------------------------------------------------
// Small helpers shared by host and device code.
// ---------------------------------------------------------------------------

CMB_FORCEINLINE uint64_t rotl64(uint64_t x, int k) {
    return (x << k) | (x >> (64 - k));
}

CMB_FORCEINLINE uint64_t double_as_u64(double d) {
#if defined(__HIP_DEVICE_COMPILE__)
    return __double_as_longlong(d);
#else
    union { double d; uint64_t u; } c;
    c.d = d;
    return c.u;
#endif
}

CMB_FORCEINLINE double u64_as_double(uint64_t u) {
#if defined(__HIP_DEVICE_COMPILE__)
    return __longlong_as_double(u);
#else
    union { double d; uint64_t u; } c;
    c.u = u;
    return c.d;
#endif
}

}  // namespace cmb
