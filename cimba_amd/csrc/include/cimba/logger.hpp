// Host-side logger + trial-abandon error path — capability parity with
// reference src/cmb_logger.c (32-bit flag-mask logging with 4 reserved
// levels + 28 user bits, cmb_logger.h:54-66; line format
// "[trial] [seed] time func: msg", cmb_logger.c:154-156; mutex-serialized
// output :76-77; cmb_logger_error prints then abandons the trial
// :253-274; cmb_logger_fatal aborts :235-251) and the recovery machinery
// of src/cimba.c:74-76,289-329 (longjmp out of a broken trial).
//
// MI355X redesign: the host executive uses a C++ exception (TrialAbandon)
// instead of longjmp — the engine is POD with no cleanup-order hazards, so
// unwinding is trivially safe; on the DEVICE the equivalent is
// Engine::fail() + the per-trial status word, which the
// executive surfaces as a failed trial exactly like an abandoned host one.
#pragma once

#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <mutex>

namespace cmb {

enum LogFlag : uint32_t {
    LOG_FATAL = 1u << 0,
    LOG_ERROR = 1u << 1,
    LOG_WARNING = 1u << 2,
    LOG_INFO = 1u << 3,
    // bits 4..31 are user flags (reference: 28 user bits)
};

// thrown by logger_error / available to models; caught per-trial by the
// executive, which runs cleanup hooks and continues with the next trial
struct TrialAbandon {
    int32_t code;
};

struct LogCtx {  // per worker thread, set by the executive
    uint32_t trial = 0;
    uint64_t seed = 0;
    double sim_time = 0.0;
    const char* who = "";
};

namespace detail {
inline std::mutex& log_mutex() {
    static std::mutex m;
    return m;
}
inline uint32_t& log_mask() {
    static uint32_t mask = LOG_FATAL | LOG_ERROR | LOG_WARNING | LOG_INFO;
    return mask;
}
inline FILE*& log_stream() {
    static FILE* f = stderr;
    return f;
}
inline LogCtx& log_ctx() {
    static thread_local LogCtx c;
    return c;
}
}  // namespace detail

inline void logger_flags_on(uint32_t bits) { detail::log_mask() |= bits; }
inline void logger_flags_off(uint32_t bits) { detail::log_mask() &= ~bits; }
inline uint32_t logger_flags() { return detail::log_mask(); }
inline void logger_stream_set(FILE* f) { detail::log_stream() = f; }
inline LogCtx& logger_ctx() { return detail::log_ctx(); }

inline void logger_vlog(uint32_t flag, const char* tag, const char* fmt,
                        va_list ap) {
    if (!(detail::log_mask() & flag)) return;
    const LogCtx& c = detail::log_ctx();
    std::lock_guard<std::mutex> lock(detail::log_mutex());
    FILE* f = detail::log_stream();
    fprintf(f, "[%u] [%016llx] %.6f %s %s: ", c.trial,
            (unsigned long long)c.seed, c.sim_time, c.who, tag);
    vfprintf(f, fmt, ap);
    fputc('\n', f);
    fflush(f);
}

inline void logger_info(const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_INFO, "info", fmt, ap);
    va_end(ap);
}
inline void logger_warning(const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_WARNING, "warning", fmt, ap);
    va_end(ap);
}
inline void logger_user(uint32_t flag, const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(flag, "user", fmt, ap);
    va_end(ap);
}
// print then abandon the current trial (reference cmb_logger_error)
[[noreturn]] inline void logger_error(const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_ERROR, "error", fmt, ap);
    va_end(ap);
    throw TrialAbandon{1};
}
// print then abort the program (reference cmb_logger_fatal)
[[noreturn]] inline void logger_fatal(const char* fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_FATAL, "fatal", fmt, ap);
    va_end(ap);
    abort();
}

}  // namespace cmb

// model-visible logging macro: host logs, device no-op (device error path
// is Engine::fail + status word)
#if defined(__HIP_DEVICE_COMPILE__)
#define CMB_LOG_INFO(E, ...) ((void)0)
#define CMB_LOG_WARNING(E, ...) ((void)0)
#else
#define CMB_LOG_INFO(E, ...)                      \
    do {                                          \
        ::cmb::logger_ctx().sim_time = (E).now;   \
        ::cmb::logger_info(__VA_ARGS__);          \
    } while (0)
#define CMB_LOG_WARNING(E, ...)                   \
    do {                                          \
        ::cmb::logger_ctx().sim_time = (E).now;   \
        ::cmb::logger_warning(__VA_ARGS__);       \
    } while (0)
#endif
