/*
 * cimba_amd public C API — the cmb_* surface of the reference
 * (ambonvik/cimba include/cimba.h, cmb_event.h, cmb_process.h,
 * cmb_resource.h, cmb_resourcepool.h, cmb_buffer.h, cmb_objectqueue.h,
 * cmb_priorityqueue.h, cmb_condition.h, cmb_resourceguard.h,
 * cmb_random.h, cmb_datasummary.h, cmb_logger.h) re-exposed over the
 * MI355X-native engine.
 *
 * ONE deliberate difference from the reference (docs/PARITY.md): the
 * reference's processes are stackful coroutines, so blocking calls
 * (cmb_process_hold, cmb_objectqueue_get, ...) suspend mid-function via an
 * assembly context switch.  There is no stack switching on gfx950, so this
 * framework's processes are resumable state machines on BOTH host and
 * device: a process body is written between CMB_PROC_BEGIN/CMB_PROC_END
 * and every blocking call is a CMB_* macro naming a resumption point.
 * Persistent locals live in the user's context struct.  Everything else —
 * the event/clock model, signals, guards with demand predicates, priority
 * ordering, the statistics layer, seeds — follows the reference contract.
 *
 * The simulation context `cmb_sim` is explicit (first argument) instead of
 * thread-local: the same model code can then run as one trial per CPU
 * worker thread or one trial per GPU wavefront, where thread-locals do not
 * exist.
 */
#ifndef CIMBA_AMD_CIMBA_H
#define CIMBA_AMD_CIMBA_H

#include <stddef.h>
#include <stdint.h>
#include <stdio.h>
#include <stdbool.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- signals (reference include/cmb_process.h:60-100) ---- */
#define CMB_PROCESS_SUCCESS INT64_C(0)
#define CMB_PROCESS_PREEMPTED INT64_C(-1)
#define CMB_PROCESS_INTERRUPTED INT64_C(-2)
#define CMB_PROCESS_STOPPED INT64_C(-3)
#define CMB_PROCESS_CANCELLED INT64_C(-4)
#define CMB_PROCESS_TIMEOUT INT64_C(-5)

#define CMB_UNLIMITED INT32_C(0x7FFFFFFF)

typedef struct cmb_sim cmb_sim;           /* per-trial engine (opaque) */
typedef struct cmb_process cmb_process;   /* process handle (opaque) */
typedef struct cmb_objectqueue cmb_objectqueue;
typedef struct cmb_priorityqueue cmb_priorityqueue;
typedef struct cmb_resource cmb_resource;
typedef struct cmb_resourcepool cmb_resourcepool;
typedef struct cmb_buffer cmb_buffer;
typedef struct cmb_condition cmb_condition;

typedef void (cmb_process_func)(cmb_sim* sim, cmb_process* me, void* ctx);
typedef void (cmb_event_func)(cmb_sim* sim, void* subject, void* object);
typedef bool (cmb_demand_func)(cmb_sim* sim, void* ctx);

/* ---- experiment executive (reference src/cimba.c) ---- */
typedef void (cimba_trial_func)(cmb_sim* sim, void* trial);

/* Run `n` trials of `size` bytes each from `experiment`; per-trial seeds
 * derive from master_seed via fmix64.  Returns the failed-trial count. */
uint64_t cimba_run(void* experiment, uint64_t n, size_t size,
                   cimba_trial_func* trial_fn, uint64_t master_seed,
                   int nthreads);
void cimba_threads_use(int nthreads);       /* default worker count */
uint64_t cimba_trials_remaining(void);
void cimba_trial_abandon(cmb_sim* sim);     /* abandon current trial */
void cimba_thread_hooks_set(void (*init_fn)(int), void (*exit_fn)(int));
void cimba_trial_cleanup_set(void (*cleanup_fn)(uint64_t));
uint32_t cmb_sim_trial_index(const cmb_sim* sim);
uint64_t cmb_sim_trial_seed(const cmb_sim* sim);
uint64_t cmb_sim_events_dispatched(const cmb_sim* sim);

/* ---- clock & events (reference include/cmb_event.h) ---- */
double cmb_time(const cmb_sim* sim);
uint64_t cmb_event_schedule(cmb_sim* sim, cmb_event_func* action,
                            void* subject, void* object, double time,
                            int priority);
bool cmb_event_cancel(cmb_sim* sim, uint64_t handle);
bool cmb_event_reschedule(cmb_sim* sim, uint64_t handle, double time,
                          int priority);
/* wildcard pattern ops; pass NULL action/subject/object as ANY */
uint64_t cmb_event_pattern_count(cmb_sim* sim, cmb_event_func* action,
                                 void* subject, void* object);
uint64_t cmb_event_pattern_cancel(cmb_sim* sim, cmb_event_func* action,
                                  void* subject, void* object);
void cmb_event_queue_execute(cmb_sim* sim);  /* run until empty */
void cmb_event_queue_execute_until(cmb_sim* sim, double until);

/* ---- processes (reference include/cmb_process.h) ---- */
cmb_process* cmb_process_spawn(cmb_sim* sim, const char* name,
                               cmb_process_func* fn, void* ctx,
                               int priority);
void cmb_process_start(cmb_sim* sim, cmb_process* p);
void cmb_process_start_at(cmb_sim* sim, cmb_process* p, double delay);
void cmb_process_interrupt(cmb_sim* sim, cmb_process* p, int64_t sig);
void cmb_process_stop(cmb_sim* sim, cmb_process* p);
void cmb_process_resume(cmb_sim* sim, cmb_process* p);
void cmb_process_priority_set(cmb_sim* sim, cmb_process* p, int priority);
int64_t cmb_process_priority(const cmb_sim* sim, const cmb_process* p);
const char* cmb_process_name(const cmb_sim* sim, const cmb_process* p);
int cmb_process_state(const cmb_sim* sim, const cmb_process* p);
int64_t cmb_process_signal(const cmb_sim* sim, const cmb_process* p);
void* cmb_process_context(const cmb_sim* sim, const cmb_process* p);

/* protothread plumbing used by the CMB_* macros below */
int cmb_proc_pc_(const cmb_sim* sim, const cmb_process* p);
void cmb_proc_set_pc_(cmb_sim* sim, cmb_process* p, int pc);
void cmb_proc_resumed_(cmb_sim* sim, cmb_process* p);
void cmb_hold_setup_(cmb_sim* sim, cmb_process* p, double duration);
int cmb_wait_process_setup_(cmb_sim* sim, cmb_process* p, cmb_process* tgt);
void cmb_wait_event_setup_(cmb_sim* sim, cmb_process* p, uint64_t handle);
void cmb_proc_finish_(cmb_sim* sim, cmb_process* p);
void cmb_timer_arm_(cmb_sim* sim, cmb_process* p, double delay, int64_t sig);
void cmb_timer_disarm_(cmb_sim* sim, cmb_process* p);

/* ---- multiple concurrent timers per process (reference
 * cmb_process_timer_add/set/cancel/clear, cmb_process.c:514-580).
 * Slot 0 is reserved for the blocking-call timeout (cmb_timer_arm_);
 * user slots are 1..CMB_PROCESS_TIMERS-1.  A firing timer wakes the
 * process's current blocking call with `sig`. ---- */
#define CMB_PROCESS_TIMERS 4
bool cmb_process_timer_add(cmb_sim* sim, cmb_process* p, int slot,
                           double delay, int64_t sig);
void cmb_process_timer_cancel(cmb_sim* sim, cmb_process* p, int slot);
void cmb_process_timer_clear(cmb_sim* sim, cmb_process* p);
bool cmb_process_timer_pending(const cmb_sim* sim, const cmb_process* p,
                               int slot);
int cmb_sim_ok_(const cmb_sim* sim);

/* queue/resource try+wait plumbing */
bool cmb_queue_try_put_(cmb_sim* sim, cmb_objectqueue* q, cmb_process* p,
                        void* object);
bool cmb_queue_try_get_(cmb_sim* sim, cmb_objectqueue* q, cmb_process* p,
                        void** object);
void cmb_queue_wait_space_(cmb_sim* sim, cmb_objectqueue* q, cmb_process* p);
void cmb_queue_wait_object_(cmb_sim* sim, cmb_objectqueue* q, cmb_process* p);
bool cmb_pqueue_try_put_(cmb_sim* sim, cmb_priorityqueue* q, cmb_process* p,
                         void* object, int priority);
bool cmb_pqueue_try_get_(cmb_sim* sim, cmb_priorityqueue* q, cmb_process* p,
                         void** object);
void cmb_pqueue_wait_space_(cmb_sim* sim, cmb_priorityqueue* q,
                            cmb_process* p);
void cmb_pqueue_wait_object_(cmb_sim* sim, cmb_priorityqueue* q,
                             cmb_process* p);
bool cmb_resource_try_acquire_(cmb_sim* sim, cmb_resource* r, cmb_process* p);
bool cmb_resource_try_preempt_(cmb_sim* sim, cmb_resource* r, cmb_process* p);
void cmb_resource_wait_(cmb_sim* sim, cmb_resource* r, cmb_process* p);
int32_t cmb_pool_try_take_(cmb_sim* sim, cmb_resourcepool* r, cmb_process* p,
                           int32_t want);
bool cmb_pool_try_take_all_(cmb_sim* sim, cmb_resourcepool* r, cmb_process* p,
                            int32_t want);
void cmb_pool_wait_(cmb_sim* sim, cmb_resourcepool* r, cmb_process* p);
void cmb_pool_wait_ge_(cmb_sim* sim, cmb_resourcepool* r, cmb_process* p,
                       int32_t amount);
bool cmb_buffer_try_get_(cmb_sim* sim, cmb_buffer* b, cmb_process* p,
                         int64_t amount);
bool cmb_buffer_try_put_(cmb_sim* sim, cmb_buffer* b, cmb_process* p,
                         int64_t amount);
void cmb_buffer_wait_level_(cmb_sim* sim, cmb_buffer* b, cmb_process* p,
                            int64_t amount);
void cmb_buffer_wait_space_(cmb_sim* sim, cmb_buffer* b, cmb_process* p,
                            int64_t amount);
void cmb_condition_wait_setup_(cmb_sim* sim, cmb_condition* c, cmb_process* p,
                               cmb_demand_func* demand, void* ctx);

/* ---- the protothread macro surface ----
 * Process bodies:
 *   void body(cmb_sim* sim, cmb_process* me, void* vctx) {
 *       struct my_ctx* ctx = vctx;              // persistent locals
 *       CMB_PROC_BEGIN(me);
 *       CMB_HOLD(sim, me, 1.5);
 *       ...
 *       CMB_PROC_END(sim, me);
 *   }
 * One CMB_* blocking macro per source line.
 */
#define CMB_PROC_BEGIN(sim, me) switch (cmb_proc_pc_((sim), (me))) { case 0:
#define CMB_PROC_END(sim, me) } cmb_proc_finish_((sim), (me)); return;

#define CMB_SIGNAL(sim, me) cmb_process_signal((sim), (me))

#define CMB_YIELD_(sim, me)                      \
    cmb_proc_set_pc_((sim), (me), __LINE__);     \
    return;                                      \
    case __LINE__:                               \
        cmb_proc_resumed_((sim), (me));

#define CMB_HOLD(sim, me, dur)                 \
    do {                                       \
        cmb_hold_setup_((sim), (me), (dur));   \
        CMB_YIELD_(sim, me);                   \
    } while (0)

#define CMB_WAIT_PROCESS(sim, me, tgt)                         \
    do {                                                       \
        if (cmb_wait_process_setup_((sim), (me), (tgt))) {     \
            CMB_YIELD_(sim, me);                               \
        }                                                      \
    } while (0)

#define CMB_WAIT_EVENT(sim, me, handle)                    \
    do {                                                   \
        cmb_wait_event_setup_((sim), (me), (handle));      \
        CMB_YIELD_(sim, me);                               \
    } while (0)

#define CMB_BLOCKING_LOOP_(sim, me, try_expr, wait_stmt)                  \
    for (;;) {                                                            \
        if (try_expr) break;                                              \
        if (!cmb_sim_ok_(sim)) break;                                     \
        wait_stmt;                                                        \
        CMB_YIELD_(sim, me);                                              \
        if (cmb_process_signal((sim), (me)) != CMB_PROCESS_SUCCESS) break; \
    }

#define CMB_OBJECTQUEUE_PUT(sim, me, q, obj)                              \
    CMB_BLOCKING_LOOP_(sim, me, cmb_queue_try_put_((sim), (q), (me), (obj)), \
                       cmb_queue_wait_space_((sim), (q), (me)))

#define CMB_OBJECTQUEUE_GET(sim, me, q, objp)                             \
    CMB_BLOCKING_LOOP_(sim, me, cmb_queue_try_get_((sim), (q), (me), (objp)), \
                       cmb_queue_wait_object_((sim), (q), (me)))

#define CMB_PRIORITYQUEUE_PUT(sim, me, q, obj, pri)                        \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_pqueue_try_put_((sim), (q), (me), (obj), (pri)), \
                       cmb_pqueue_wait_space_((sim), (q), (me)))

#define CMB_PRIORITYQUEUE_GET(sim, me, q, objp)                            \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_pqueue_try_get_((sim), (q), (me), (objp)),      \
                       cmb_pqueue_wait_object_((sim), (q), (me)))

#define CMB_RESOURCE_ACQUIRE(sim, me, r)                                   \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_resource_try_acquire_((sim), (r), (me)),        \
                       cmb_resource_wait_((sim), (r), (me)))

#define CMB_RESOURCE_PREEMPT(sim, me, r)                                   \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_resource_try_preempt_((sim), (r), (me)),        \
                       cmb_resource_wait_((sim), (r), (me)))

#define CMB_RESOURCEPOOL_ACQUIRE_ALL(sim, me, r, amount)                   \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_pool_try_take_all_((sim), (r), (me), (amount)), \
                       cmb_pool_wait_ge_((sim), (r), (me), (amount)))

#define CMB_RESOURCEPOOL_PREEMPT(sim, me, r, amount)                       \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_pool_try_preempt_((sim), (r), (me), (amount)),  \
                       cmb_pool_wait_ge_((sim), (r), (me), (amount)))

#define CMB_BUFFER_GET(sim, me, b, amount)                                 \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_buffer_try_get_((sim), (b), (me), (amount)),    \
                       cmb_buffer_wait_level_((sim), (b), (me), (amount)))

#define CMB_BUFFER_PUT(sim, me, b, amount)                                 \
    CMB_BLOCKING_LOOP_(sim, me,                                            \
                       cmb_buffer_try_put_((sim), (b), (me), (amount)),    \
                       cmb_buffer_wait_space_((sim), (b), (me), (amount)))

#define CMB_CONDITION_WAIT(sim, me, c, demand, ctx)                        \
    do {                                                                   \
        cmb_condition_wait_setup_((sim), (c), (me), (demand), (ctx));      \
        CMB_YIELD_(sim, me);                                               \
    } while (0)

/* ---- toolkit create/initialize/query (non-blocking side) ---- */
cmb_objectqueue* cmb_objectqueue_create(cmb_sim* sim);
void cmb_objectqueue_initialize(cmb_sim* sim, cmb_objectqueue* q,
                                const char* name, int32_t capacity);
uint64_t cmb_objectqueue_length(const cmb_sim* sim, const cmb_objectqueue* q);
/* 1-based position of `object` in the queue; 0 = not present (reference
 * cmb_objectqueue_position) */
uint64_t cmb_objectqueue_position(const cmb_sim* sim,
                                  const cmb_objectqueue* q,
                                  const void* object);
void cmb_objectqueue_recording_start(cmb_sim* sim, cmb_objectqueue* q);
void cmb_objectqueue_recording_stop(cmb_sim* sim, cmb_objectqueue* q);
/* time-weighted length stats while recording: mean/stddev/min/max */
void cmb_objectqueue_stats(cmb_sim* sim, const cmb_objectqueue* q,
                           double out4[4]);

cmb_priorityqueue* cmb_priorityqueue_create(cmb_sim* sim);
void cmb_priorityqueue_initialize(cmb_sim* sim, cmb_priorityqueue* q,
                                  const char* name, int32_t capacity);
uint64_t cmb_priorityqueue_length(const cmb_sim* sim,
                                  const cmb_priorityqueue* q);

cmb_resource* cmb_resource_create(cmb_sim* sim);
void cmb_resource_initialize(cmb_sim* sim, cmb_resource* r, const char* name);
void cmb_resource_release(cmb_sim* sim, cmb_resource* r, cmb_process* p);
bool cmb_resource_in_use(const cmb_sim* sim, const cmb_resource* r);
cmb_process* cmb_resource_holder(const cmb_sim* sim, const cmb_resource* r);
void cmb_resource_recording_start(cmb_sim* sim, cmb_resource* r);
void cmb_resource_stats(cmb_sim* sim, const cmb_resource* r, double out4[4]);

cmb_resourcepool* cmb_resourcepool_create(cmb_sim* sim);
void cmb_resourcepool_initialize(cmb_sim* sim, cmb_resourcepool* r,
                                 const char* name, int32_t capacity);
void cmb_resourcepool_release(cmb_sim* sim, cmb_resourcepool* r,
                              cmb_process* holder, int32_t amount);
int32_t cmb_resourcepool_holding(const cmb_sim* sim,
                                 const cmb_resourcepool* r,
                                 const cmb_process* p);
bool cmb_pool_try_preempt_(cmb_sim* sim, cmb_resourcepool* r, cmb_process* p,
                           int32_t want);
int32_t cmb_resourcepool_capacity(const cmb_sim* sim,
                                  const cmb_resourcepool* r);
int32_t cmb_resourcepool_in_use(const cmb_sim* sim, const cmb_resourcepool* r);
int32_t cmb_resourcepool_available(const cmb_sim* sim,
                                   const cmb_resourcepool* r);

cmb_buffer* cmb_buffer_create(cmb_sim* sim);
void cmb_buffer_initialize(cmb_sim* sim, cmb_buffer* b, const char* name,
                           int64_t capacity, int64_t initial_level);
int64_t cmb_buffer_level(const cmb_sim* sim, const cmb_buffer* b);
int64_t cmb_buffer_capacity(const cmb_sim* sim, const cmb_buffer* b);

cmb_condition* cmb_condition_create(cmb_sim* sim);
void cmb_condition_initialize(cmb_sim* sim, cmb_condition* c,
                              const char* name);
uint64_t cmb_condition_signal(cmb_sim* sim, cmb_condition* c);

/* ---- debug dumps & reports (reference cmb_event_queue_print,
 * cmb_resource_print_report et al.) ---- */
void cmb_event_queue_print(cmb_sim* sim, FILE* out);
void cmb_resource_print_report(cmb_sim* sim, const cmb_resource* r,
                               FILE* out);
void cmb_resourcepool_print_report(cmb_sim* sim, const cmb_resourcepool* r,
                                   FILE* out);
void cmb_objectqueue_report_print(cmb_sim* sim, const cmb_objectqueue* q,
                                  FILE* out);

/* ---- RNG (reference include/cmb_random.h; per-trial stream) ---- */
uint64_t cmb_random_sfc64(cmb_sim* sim);
uint64_t cmb_random_fmix64(uint64_t x);
uint64_t cmb_random_curseed(const cmb_sim* sim);
double cmb_random_uniform(cmb_sim* sim, double lo, double hi);
bool cmb_random_flip(cmb_sim* sim, double p);
int64_t cmb_random_bernoulli(cmb_sim* sim, double p);
double cmb_random_std_normal(cmb_sim* sim);
double cmb_random_normal(cmb_sim* sim, double mu, double sigma);
double cmb_random_std_exponential(cmb_sim* sim);
double cmb_random_exponential(cmb_sim* sim, double mean);
double cmb_random_lognormal(cmb_sim* sim, double mu, double sigma);
double cmb_random_logistic(cmb_sim* sim, double loc, double scale);
double cmb_random_cauchy(cmb_sim* sim, double loc, double scale);
double cmb_random_rayleigh(cmb_sim* sim, double sigma);
double cmb_random_weibull(cmb_sim* sim, double shape, double scale);
double cmb_random_pareto(cmb_sim* sim, double shape, double scale);
double cmb_random_triangular(cmb_sim* sim, double lo, double mode, double hi);
double cmb_random_pert(cmb_sim* sim, double lo, double mode, double hi);
double cmb_random_std_gamma(cmb_sim* sim, double alpha);
double cmb_random_gamma(cmb_sim* sim, double shape, double scale);
double cmb_random_erlang(cmb_sim* sim, int64_t k, double mean);
double cmb_random_hypoexponential(cmb_sim* sim, double m1, double m2);
double cmb_random_hyperexponential(cmb_sim* sim, double p, double m1,
                                   double m2);
double cmb_random_std_beta(cmb_sim* sim, double a, double b);
double cmb_random_beta(cmb_sim* sim, double a, double b, double lo, double hi);
double cmb_random_chisquared(cmb_sim* sim, double k);
double cmb_random_std_t_dist(cmb_sim* sim, double df);
double cmb_random_t_dist(cmb_sim* sim, double df, double loc, double scale);
double cmb_random_f_dist(cmb_sim* sim, double d1, double d2);
int64_t cmb_random_geometric(cmb_sim* sim, double p);
int64_t cmb_random_poisson(cmb_sim* sim, double mean);
int64_t cmb_random_binomial(cmb_sim* sim, int64_t n, double p);
int64_t cmb_random_negative_binomial(cmb_sim* sim, double r, double p);
int64_t cmb_random_pascal(cmb_sim* sim, int64_t r, double p);
int64_t cmb_random_discrete_uniform(cmb_sim* sim, int64_t lo, int64_t hi);
int64_t cmb_random_dice(cmb_sim* sim, int64_t sides);
int64_t cmb_random_discrete_nonuniform(cmb_sim* sim, const double* weights,
                                       int64_t n);
int64_t cmb_random_loaded_dice(cmb_sim* sim, const double* weights,
                               int64_t sides);
uint64_t cmb_random_hwseed(void);

/* ---- running summaries (reference cmb_datasummary / cmb_wtdsummary) ---- */
typedef struct cmb_datasummary {
    double n, mean, m2, m3, m4, mn, mx;
} cmb_datasummary;
void cmb_datasummary_initialize(cmb_datasummary* s);
void cmb_datasummary_add(cmb_datasummary* s, double x);
void cmb_datasummary_merge(cmb_datasummary* s, const cmb_datasummary* o);
double cmb_datasummary_count(const cmb_datasummary* s);
double cmb_datasummary_mean(const cmb_datasummary* s);
double cmb_datasummary_variance(const cmb_datasummary* s);
double cmb_datasummary_stddev(const cmb_datasummary* s);
double cmb_datasummary_skewness(const cmb_datasummary* s);
double cmb_datasummary_kurtosis(const cmb_datasummary* s);
double cmb_datasummary_minimum(const cmb_datasummary* s);
double cmb_datasummary_maximum(const cmb_datasummary* s);

typedef struct cmb_wtdsummary {
    double n, sumw, mean, m2, mn, mx;
} cmb_wtdsummary;
void cmb_wtdsummary_initialize(cmb_wtdsummary* s);
void cmb_wtdsummary_add(cmb_wtdsummary* s, double x, double w);
void cmb_wtdsummary_merge(cmb_wtdsummary* s, const cmb_wtdsummary* o);
double cmb_wtdsummary_mean(const cmb_wtdsummary* s);
double cmb_wtdsummary_variance(const cmb_wtdsummary* s);

/* ---- logger (reference include/cmb_logger.h) ---- */
#define CMB_LOGGER_FATAL (1u << 0)
#define CMB_LOGGER_ERROR (1u << 1)
#define CMB_LOGGER_WARNING (1u << 2)
#define CMB_LOGGER_INFO (1u << 3)
void cmb_logger_flags_on(uint32_t flags);
void cmb_logger_flags_off(uint32_t flags);
/* pluggable sim-time formatter (reference cmb_logger_timeformatter_set);
 * fmt writes into buf (>= 32 bytes); NULL restores the default */
typedef void (cmb_timeformatter_func)(double t, char* buf, size_t bufsz);
void cmb_logger_timeformatter_set(cmb_timeformatter_func* fmt);
void cmb_logger_info(cmb_sim* sim, const char* fmt, ...);
void cmb_logger_warning(cmb_sim* sim, const char* fmt, ...);
void cmb_logger_error(cmb_sim* sim, const char* fmt, ...); /* abandons trial */

#ifdef __cplusplus
}
#endif

#endif /* CIMBA_AMD_CIMBA_H */
