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
#include <stdarg.h>
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

/* ---- engine envelope (defaults; rebuild libcimba with
 * -DCIMBA_C_MAX_PROC=N etc. to change — cimba_amd/csrc/host/capi.cpp).
 * Within a trial: up to 16384 live processes, 64 each of object queues /
 * resources / pools / buffers / conditions, 32 priority queues, 65536
 * pending user events.  The event heap (16K fast + 48K growth tier),
 * object queues (8K + 24K) and priority queues (4K + 12K) grow into a
 * spill tier automatically; a trial aborts with a clean status only on
 * true exhaustion. ---- */

typedef struct cmb_sim cmb_sim;           /* per-trial engine (opaque) */
typedef struct cmb_process cmb_process;   /* process handle (opaque) */
typedef struct cmb_objectqueue cmb_objectqueue;
typedef struct cmb_priorityqueue cmb_priorityqueue;
typedef struct cmb_resource cmb_resource;
typedef struct cmb_resourcepool cmb_resourcepool;
typedef struct cmb_buffer cmb_buffer;
typedef struct cmb_condition cmb_condition;
typedef struct cmb_timeseries cmb_timeseries;  /* full API further below */

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
/* reference-compat executive aliases/extras */
typedef void (cimba_thread_init_func)(int worker);
typedef void (cimba_thread_exit_func)(int worker);
typedef void (cimba_trial_cleanup_func)(uint64_t trial);
uint64_t cimba_run_experiment(void* experiment, uint64_t n, size_t size,
                              cimba_trial_func* trial_fn,
                              uint64_t master_seed, int nthreads);
uint32_t cimba_trial_index(const cmb_sim* sim);   /* = cmb_sim_trial_index */
uint64_t cimba_trials_total(void);
int cimba_threads_num(void);                       /* configured workers */
/* per-worker user context (reference cimba_thread_context: e.g. streams) */
void cimba_thread_context_set(void* ctx);
void* cimba_thread_context(void);
int cimba_thread_id(void);                         /* current worker index */
const char* cimba_version(void);

/* ---- clock & events (reference include/cmb_event.h) ---- */
double cmb_time(const cmb_sim* sim);
uint64_t cmb_event_schedule(cmb_sim* sim, cmb_event_func* action,
                            void* subject, void* object, double time,
                            int priority);
bool cmb_event_cancel(cmb_sim* sim, uint64_t handle);
bool cmb_event_reschedule(cmb_sim* sim, uint64_t handle, double time,
                          int priority);
/* wildcard pattern ops.  Pass CMB_ANY_ACTION / CMB_ANY_SUBJECT /
 * CMB_ANY_OBJECT to match any value in that position (reference
 * include/cmb_event.h wildcard contract); NULL is a CONCRETE value and
 * matches only events scheduled with a NULL in that position. */
#define CMB_ANY_ACTION ((cmb_event_func*)~(uintptr_t)0)
#define CMB_ANY_SUBJECT ((void*)~(uintptr_t)0)
#define CMB_ANY_OBJECT ((void*)~(uintptr_t)0)
uint64_t cmb_event_pattern_count(cmb_sim* sim, cmb_event_func* action,
                                 void* subject, void* object);
uint64_t cmb_event_pattern_cancel(cmb_sim* sim, cmb_event_func* action,
                                  void* subject, void* object);
void cmb_event_queue_execute(cmb_sim* sim);  /* run until empty */
void cmb_event_queue_execute_until(cmb_sim* sim, double until);
/* single-step + queue introspection (reference cmb_event_execute_next,
 * cmb_event_queue_count/is_empty/clear, cmb_event_is_scheduled,
 * cmb_event_time/priority, cmb_event_reprioritize) */
bool cmb_event_execute_next(cmb_sim* sim);
uint64_t cmb_event_queue_count(const cmb_sim* sim);
bool cmb_event_queue_is_empty(const cmb_sim* sim);
void cmb_event_queue_clear(cmb_sim* sim);
bool cmb_event_is_scheduled(const cmb_sim* sim, uint64_t handle);
double cmb_event_time(const cmb_sim* sim, uint64_t handle);
int cmb_event_priority(const cmb_sim* sim, uint64_t handle);
bool cmb_event_reprioritize(cmb_sim* sim, uint64_t handle, int priority);
uint64_t cmb_event_pattern_find(cmb_sim* sim, cmb_event_func* action,
                                void* subject, void* object);
/* handle of the currently / most recently executed event, 0 if none
 * (reference cmb_event_current; takes the sim context instead of TLS) */
uint64_t cmb_event_current(const cmb_sim* sim);

/* ---- processes (reference include/cmb_process.h) ---- */
/* currently executing process, NULL from the dispatcher / trial function
 * (reference cmb_process_current; sim context instead of coroutine TLS) */
cmb_process* cmb_process_current(const cmb_sim* sim);
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
/* reference-compat process aliases */
void cmb_process_kill(cmb_sim* sim, cmb_process* p);     /* = stop */
int cmb_process_status(const cmb_sim* sim, const cmb_process* p);
void cmb_process_name_set(cmb_sim* sim, cmb_process* p, const char* name);
/* two-phase lifecycle (reference cmb_process_create + _initialize);
 * spawn() remains the one-call form.  terminate/destroy are no-ops: the
 * engine owns all process storage (POD, recycled by proc slots). */
cmb_process* cmb_process_create(cmb_sim* sim);
void cmb_process_initialize(cmb_sim* sim, cmb_process* p, const char* name,
                            cmb_process_func* fn, void* ctx, int priority);
void cmb_process_terminate(cmb_sim* sim, cmb_process* p);
void cmb_process_destroy(cmb_sim* sim, cmb_process* p);
void cmb_process_exit_value_set_(cmb_sim* sim, cmb_process* p, void* value);
void* cmb_process_exit_value(const cmb_sim* sim, const cmb_process* p);
/* reset a pending timer's delay (reference cmb_process_timer_set) */
bool cmb_process_timer_set(cmb_sim* sim, cmb_process* p, int slot,
                           double delay, int64_t sig);
#define cmb_process_timers_clear cmb_process_timer_clear

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
 * user slots are 1..CMB_PROCESS_TIMERS-1 and the public wrappers REJECT
 * slot 0 (timer_add/set return false; cancel/pending treat it as absent;
 * clear skips it), so a user timer can never hijack the next CMB_HOLD.
 * A firing timer wakes the process's current blocking call with
 * `sig`. ---- */
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

/* greedy partial pool acquisition (reference cmb_resourcepool_acquire,
 * include/cmb_resourcepool.h:15-19); `remvar` = persistent int32 lvalue */
#define CMB_RESOURCEPOOL_ACQUIRE(sim, me, r, amount, remvar)               \
    do {                                                                   \
        (remvar) = (amount);                                               \
        for (;;) {                                                         \
            (remvar) -= cmb_pool_try_take_((sim), (r), (me), (remvar));    \
            if ((remvar) <= 0) break;                                      \
            if (!cmb_sim_ok_(sim)) break;                                  \
            cmb_pool_wait_((sim), (r), (me));                              \
            CMB_YIELD_(sim, me);                                           \
            if (cmb_process_signal((sim), (me)) != CMB_PROCESS_SUCCESS)    \
                break;                                                     \
        }                                                                  \
    } while (0)

/* ---- reference-named blocking-call aliases (lower-case macro forms of
 * the CMB_* macros above, so model code reads like the reference's; the
 * extra (sim, me) arguments are the explicit-context difference) ---- */
#define cmb_process_hold(sim, me, dur) CMB_HOLD(sim, me, dur)
#define cmb_process_wait_process(sim, me, tgt) CMB_WAIT_PROCESS(sim, me, tgt)
#define cmb_process_wait_event(sim, me, h) CMB_WAIT_EVENT(sim, me, h)
#define cmb_objectqueue_put(sim, me, q, obj) \
    CMB_OBJECTQUEUE_PUT(sim, me, q, obj)
#define cmb_objectqueue_get(sim, me, q, objp) \
    CMB_OBJECTQUEUE_GET(sim, me, q, objp)
#define cmb_priorityqueue_put(sim, me, q, obj, pri) \
    CMB_PRIORITYQUEUE_PUT(sim, me, q, obj, pri)
#define cmb_priorityqueue_get(sim, me, q, objp) \
    CMB_PRIORITYQUEUE_GET(sim, me, q, objp)
#define cmb_resource_acquire(sim, me, r) CMB_RESOURCE_ACQUIRE(sim, me, r)
#define cmb_resource_preempt(sim, me, r) CMB_RESOURCE_PREEMPT(sim, me, r)
#define cmb_resourcepool_acquire(sim, me, r, amount, remvar) \
    CMB_RESOURCEPOOL_ACQUIRE(sim, me, r, amount, remvar)
#define cmb_resourcepool_acquire_all(sim, me, r, amount) \
    CMB_RESOURCEPOOL_ACQUIRE_ALL(sim, me, r, amount)
#define cmb_resourcepool_preempt(sim, me, r, amount) \
    CMB_RESOURCEPOOL_PREEMPT(sim, me, r, amount)
#define cmb_buffer_get(sim, me, b, amount) CMB_BUFFER_GET(sim, me, b, amount)
#define cmb_buffer_put(sim, me, b, amount) CMB_BUFFER_PUT(sim, me, b, amount)
#define cmb_condition_wait(sim, me, c, demand, ctx) \
    CMB_CONDITION_WAIT(sim, me, c, demand, ctx)
#define cmb_resourceguard_wait(sim, me, g, demand, ctx) \
    CMB_RESOURCEGUARD_WAIT(sim, me, g, demand, ctx)
/* finish the process, recording an exit value readable by waiters
 * (reference cmb_process_exit / coroutine return value) */
#define cmb_process_exit(sim, me, value)                  \
    do {                                                  \
        cmb_process_exit_value_set_((sim), (me), (value)); \
        cmb_proc_finish_((sim), (me));                    \
        return;                                           \
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
/* recorded queue-length history while recording is on (reference
 * cmb_objectqueue_history); NULL before the first recording_start */
cmb_timeseries* cmb_objectqueue_history(const cmb_sim* sim,
                                        const cmb_objectqueue* q);
/* time-weighted length stats while recording: mean/stddev/min/max */
void cmb_objectqueue_stats(cmb_sim* sim, const cmb_objectqueue* q,
                           double out4[4]);

cmb_priorityqueue* cmb_priorityqueue_create(cmb_sim* sim);
void cmb_priorityqueue_initialize(cmb_sim* sim, cmb_priorityqueue* q,
                                  const char* name, int32_t capacity);
uint64_t cmb_priorityqueue_length(const cmb_sim* sim,
                                  const cmb_priorityqueue* q);
uint64_t cmb_priorityqueue_space(const cmb_sim* sim,
                                 const cmb_priorityqueue* q);
/* 1-based retrieval position of `object`; 0 = absent */
uint64_t cmb_priorityqueue_position(const cmb_sim* sim,
                                    const cmb_priorityqueue* q,
                                    const void* object);
bool cmb_priorityqueue_cancel(cmb_sim* sim, cmb_priorityqueue* q,
                              const void* object);
bool cmb_priorityqueue_reprioritize(cmb_sim* sim, cmb_priorityqueue* q,
                                    const void* object, int priority);
void cmb_priorityqueue_recording_start(cmb_sim* sim, cmb_priorityqueue* q);
void cmb_priorityqueue_recording_stop(cmb_sim* sim, cmb_priorityqueue* q);
cmb_timeseries* cmb_priorityqueue_history(const cmb_sim* sim,
                                          const cmb_priorityqueue* q);
void cmb_priorityqueue_report_print(cmb_sim* sim, const cmb_priorityqueue* q,
                                    FILE* out);
uint64_t cmb_objectqueue_space(const cmb_sim* sim, const cmb_objectqueue* q);
const char* cmb_objectqueue_name(const cmb_sim* sim,
                                 const cmb_objectqueue* q);
const char* cmb_resource_name(const cmb_sim* sim, const cmb_resource* r);
const char* cmb_resourcepool_get_name(const cmb_sim* sim,
                                      const cmb_resourcepool* r);
const char* cmb_buffer_name(const cmb_sim* sim, const cmb_buffer* b);
#define cmb_priorityqueue_name(sim, q) "priorityqueue"
#define cmb_resource_held(sim, r) cmb_resource_in_use((sim), (r))

cmb_resource* cmb_resource_create(cmb_sim* sim);
void cmb_resource_initialize(cmb_sim* sim, cmb_resource* r, const char* name);
void cmb_resource_release(cmb_sim* sim, cmb_resource* r, cmb_process* p);
bool cmb_resource_in_use(const cmb_sim* sim, const cmb_resource* r);
cmb_process* cmb_resource_holder(const cmb_sim* sim, const cmb_resource* r);
void cmb_resource_recording_start(cmb_sim* sim, cmb_resource* r);
void cmb_resource_recording_stop(cmb_sim* sim, cmb_resource* r);
/* reference aliases */
void cmb_resource_start_recording(cmb_sim* sim, cmb_resource* r);
void cmb_resource_stop_recording(cmb_sim* sim, cmb_resource* r);
/* busy (0/1) history (reference cmb_resource_history) */
cmb_timeseries* cmb_resource_history(const cmb_sim* sim,
                                     const cmb_resource* r);
bool cmb_resource_available(const cmb_sim* sim, const cmb_resource* r);
bool cmb_resource_held_by_process(const cmb_sim* sim, const cmb_resource* r,
                                  const cmb_process* p);
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
int32_t cmb_resourcepool_held(const cmb_sim* sim, const cmb_resourcepool* r);
int32_t cmb_resourcepool_held_by_process(const cmb_sim* sim,
                                         const cmb_resourcepool* r,
                                         const cmb_process* p);
void cmb_resourcepool_start_recording(cmb_sim* sim, cmb_resourcepool* r);
void cmb_resourcepool_stop_recording(cmb_sim* sim, cmb_resourcepool* r);
/* units-in-use history (reference cmb_resourcepool_get_history) */
cmb_timeseries* cmb_resourcepool_get_history(const cmb_sim* sim,
                                             const cmb_resourcepool* r);
#define cmb_resourcepool_history cmb_resourcepool_get_history
void cmb_resourcepool_stats(cmb_sim* sim, const cmb_resourcepool* r,
                            double out4[4]);

cmb_buffer* cmb_buffer_create(cmb_sim* sim);
void cmb_buffer_initialize(cmb_sim* sim, cmb_buffer* b, const char* name,
                           int64_t capacity, int64_t initial_level);
int64_t cmb_buffer_level(const cmb_sim* sim, const cmb_buffer* b);
int64_t cmb_buffer_capacity(const cmb_sim* sim, const cmb_buffer* b);
int64_t cmb_buffer_space(const cmb_sim* sim, const cmb_buffer* b);
void cmb_buffer_recording_start(cmb_sim* sim, cmb_buffer* b);
void cmb_buffer_recording_stop(cmb_sim* sim, cmb_buffer* b);
/* level history (reference cmb_buffer_history) */
cmb_timeseries* cmb_buffer_history(const cmb_sim* sim, const cmb_buffer* b);
void cmb_buffer_stats(cmb_sim* sim, const cmb_buffer* b, double out4[4]);
void cmb_buffer_print_report(cmb_sim* sim, const cmb_buffer* b, FILE* out);

cmb_condition* cmb_condition_create(cmb_sim* sim);
void cmb_condition_initialize(cmb_sim* sim, cmb_condition* c,
                              const char* name);
uint64_t cmb_condition_signal(cmb_sim* sim, cmb_condition* c);
/* wake a specific waiter with SIG_CANCELLED (reference
 * cmb_condition_cancel/remove) */
bool cmb_condition_cancel(cmb_sim* sim, cmb_condition* c, cmb_process* p);
/* observe another object's guard: any signal there re-evaluates this
 * condition's waiters (reference cmb_condition_subscribe/unsubscribe and
 * cmb_resourceguard_register/unregister) */
void cmb_condition_subscribe_resource(cmb_sim* sim, cmb_condition* c,
                                      cmb_resource* r);
void cmb_condition_subscribe_queue(cmb_sim* sim, cmb_condition* c,
                                   cmb_objectqueue* q);
void cmb_condition_subscribe_pool(cmb_sim* sim, cmb_condition* c,
                                  cmb_resourcepool* r);
/* drop every observer registration pointing at this condition */
void cmb_condition_unsubscribe_all(cmb_sim* sim, cmb_condition* c);
#define cmb_condition_unsubscribe cmb_condition_unsubscribe_all

/* ---- resource guard as a public class (reference cmb_resourceguard):
 * the generic demand-predicate wait queue.  Backed by the same guard
 * machinery as conditions; signal() evaluates the FRONT waiter only
 * (priority desc, entry asc, FIFO), matching the reference contract. ---- */
typedef struct cmb_resourceguard cmb_resourceguard;
typedef bool (cmb_resourceguard_demand_func)(cmb_sim* sim, void* ctx);
typedef cmb_resourceguard_demand_func cmb_condition_demand_func;
#define cmb_condition_remove cmb_condition_cancel
#define cmb_resourceguard_remove cmb_resourceguard_cancel
#define cmb_resourceguard_register cmb_condition_subscribe_resource
#define cmb_resourceguard_unregister cmb_condition_unsubscribe_all
cmb_resourceguard* cmb_resourceguard_create(cmb_sim* sim);
void cmb_resourceguard_initialize(cmb_sim* sim, cmb_resourceguard* g,
                                  const char* name);
/* the guard EMBEDDED in a built-in object (reference cmb_resource_guard /
 * cmb_resourcepool_guard return &obj->guard): signal it after changing
 * external state a waiter's demand depends on, wait on it with
 * CMB_RESOURCEGUARD_WAIT, cancel waiters with cmb_resourceguard_cancel */
cmb_resourceguard* cmb_resource_guard(cmb_sim* sim, cmb_resource* r);
cmb_resourceguard* cmb_resourcepool_guard(cmb_sim* sim, cmb_resourcepool* p);
bool cmb_resourceguard_signal(cmb_sim* sim, cmb_resourceguard* g);
bool cmb_resourceguard_cancel(cmb_sim* sim, cmb_resourceguard* g,
                              cmb_process* p);
void cmb_guard_wait_setup_(cmb_sim* sim, cmb_resourceguard* g,
                           cmb_process* p, cmb_resourceguard_demand_func* fn,
                           void* ctx);
#define CMB_RESOURCEGUARD_WAIT(sim, me, g, demand, ctx)                    \
    do {                                                                   \
        cmb_guard_wait_setup_((sim), (g), (me), (demand), (ctx));          \
        CMB_YIELD_(sim, me);                                               \
    } while (0)

/* ---- debug dumps & reports (reference cmb_event_queue_print,
 * cmb_resource_print_report et al.) ---- */
void cmb_event_queue_print(cmb_sim* sim, FILE* out);
/* user-event labeler for queue dumps (reference cmb_event_print_formatter):
 * return a static/thread-local string naming the event, or NULL to fall
 * back to the default kind/handle line */
typedef const char* (cmb_event_print_formatter)(cmb_event_func* action,
                                                const void* subject,
                                                const void* object);
void cmb_event_queue_print_formatted(cmb_sim* sim, FILE* out,
                                     cmb_event_print_formatter* epf);
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
uint64_t cmb_random_splitmix64(uint64_t* state);
/* reseed this trial's stream (reference cmb_random_initialize) */
void cmb_random_initialize(cmb_sim* sim, uint64_t seed);
/* Vose alias tables (reference cmb_random_alias_*) */
typedef struct cmb_alias cmb_alias;
cmb_alias* cmb_random_alias_create(const double* weights, int64_t n);
void cmb_random_alias_destroy(cmb_alias* a);
int64_t cmb_random_alias_sample(cmb_sim* sim, const cmb_alias* a);
int64_t cmb_random_alias_draw(cmb_sim* sim, const cmb_alias* a);

/* ---- sample container & time series (reference cmb_dataset /
 * cmb_timeseries; host-side) ---- */
typedef struct cmb_dataset cmb_dataset;
cmb_dataset* cmb_dataset_create(void);
void cmb_dataset_destroy(cmb_dataset* d);
void cmb_dataset_reset(cmb_dataset* d);
void cmb_dataset_add(cmb_dataset* d, double x);
uint64_t cmb_dataset_count(const cmb_dataset* d);
double cmb_dataset_min(cmb_dataset* d);
double cmb_dataset_max(cmb_dataset* d);
double cmb_dataset_median(cmb_dataset* d);
double cmb_dataset_quantile(cmb_dataset* d, double q);
void cmb_dataset_sort(cmb_dataset* d);
void cmb_dataset_merge(cmb_dataset* d, const cmb_dataset* o);
void cmb_dataset_copy(cmb_dataset* dst, const cmb_dataset* src);
struct cmb_datasummary cmb_dataset_summarize(const cmb_dataset* d);
void cmb_dataset_acf(const cmb_dataset* d, double* out, int maxlag);
void cmb_dataset_pacf(const cmb_dataset* d, double* out, int maxlag);
void cmb_dataset_fivenum_print(cmb_dataset* d, FILE* out);
void cmb_dataset_histogram(cmb_dataset* d, int nbins, int64_t* out_counts);
void cmb_dataset_histogram_print(cmb_dataset* d, int nbins, FILE* out);
#define cmb_dataset_print_histogram cmb_dataset_histogram_print
void cmb_dataset_correlogram_print(const cmb_dataset* d, int maxlag,
                                   FILE* out);
void cmb_dataset_print(const cmb_dataset* d, FILE* out);

typedef struct cmb_timeseries cmb_timeseries;
cmb_timeseries* cmb_timeseries_create(void);
void cmb_timeseries_destroy(cmb_timeseries* t);
void cmb_timeseries_reset(cmb_timeseries* t);
void cmb_timeseries_add(cmb_timeseries* t, double x, double time);
uint64_t cmb_timeseries_count(const cmb_timeseries* t);
struct cmb_wtdsummary cmb_timeseries_summarize(const cmb_timeseries* t,
                                               double end_time);
double cmb_timeseries_median(const cmb_timeseries* t, double end_time);
void cmb_timeseries_print(const cmb_timeseries* t, FILE* out);
void cmb_timeseries_copy(cmb_timeseries* dst, const cmb_timeseries* src);
/* record the series' end time (reference cmb_timeseries_finalize); the
 * summarize/median wrappers then use it when end_time < 0 */
void cmb_timeseries_finalize(cmb_timeseries* t, double end_time);
double cmb_timeseries_min(const cmb_timeseries* t);
double cmb_timeseries_max(const cmb_timeseries* t);
void cmb_timeseries_fivenum_print(const cmb_timeseries* t, FILE* out);
void cmb_timeseries_histogram_print(const cmb_timeseries* t, int nbins,
                                    FILE* out);
void cmb_timeseries_correlogram_print(const cmb_timeseries* t, int maxlag,
                                      FILE* out);
/* x-sorted / t-sorted copies (reference sort_x/sort_t) */
void cmb_timeseries_sort_x(cmb_timeseries* t);
void cmb_timeseries_sort_t(cmb_timeseries* t);

/* ---- running summaries (reference cmb_datasummary / cmb_wtdsummary) ---- */
typedef struct cmb_datasummary {
    double n, mean, m2, m3, m4, mn, mx;
} cmb_datasummary;
cmb_datasummary* cmb_datasummary_create(void);
void cmb_datasummary_destroy(cmb_datasummary* s);
#define cmb_datasummary_terminate(s) ((void)0)
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
#define cmb_datasummary_min cmb_datasummary_minimum
#define cmb_datasummary_max cmb_datasummary_maximum
#define cmb_datasummary_reset cmb_datasummary_initialize
#define cmb_wtdsummary_reset cmb_wtdsummary_initialize

typedef struct cmb_wtdsummary {
    double n, sumw, mean, m2, m3, m4, mn, mx;
} cmb_wtdsummary;
cmb_wtdsummary* cmb_wtdsummary_create(void);
void cmb_wtdsummary_destroy(cmb_wtdsummary* s);
#define cmb_wtdsummary_terminate(s) ((void)0)
void cmb_wtdsummary_initialize(cmb_wtdsummary* s);
void cmb_wtdsummary_add(cmb_wtdsummary* s, double x, double w);
void cmb_wtdsummary_merge(cmb_wtdsummary* s, const cmb_wtdsummary* o);
double cmb_wtdsummary_mean(const cmb_wtdsummary* s);
double cmb_wtdsummary_variance(const cmb_wtdsummary* s);
double cmb_wtdsummary_stddev(const cmb_wtdsummary* s);
double cmb_wtdsummary_skewness(const cmb_wtdsummary* s);
double cmb_wtdsummary_kurtosis(const cmb_wtdsummary* s);
double cmb_wtdsummary_count(const cmb_wtdsummary* s);
double cmb_wtdsummary_min(const cmb_wtdsummary* s);
double cmb_wtdsummary_max(const cmb_wtdsummary* s);
void cmb_wtdsummary_print(const cmb_wtdsummary* s, FILE* out);
void cmb_datasummary_print(const struct cmb_datasummary* s, FILE* out);

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
void cmb_logger_vfprintf(cmb_sim* sim, uint32_t flag, const char* fmt,
                         va_list ap);
void cmb_logger_info(cmb_sim* sim, const char* fmt, ...);
void cmb_logger_warning(cmb_sim* sim, const char* fmt, ...);
void cmb_logger_error(cmb_sim* sim, const char* fmt, ...); /* abandons trial */
void cmb_logger_fatal(cmb_sim* sim, const char* fmt, ...); /* aborts program */
void cmb_logger_user(cmb_sim* sim, uint32_t flag, const char* fmt, ...);

/* ---- asserts (reference include/cmb_assert.h three tiers) ---- */
void cmb_assert_failed_(const char* expr, const char* file, int line);
#define cmb_assert_always(expr) \
    ((expr) ? (void)0 : cmb_assert_failed_(#expr, __FILE__, __LINE__))
#ifdef NASSERT
#define cmb_assert_release(expr) ((void)0)
#else
#define cmb_assert_release(expr) cmb_assert_always(expr)
#endif
#ifdef NDEBUG
#define cmb_assert_debug(expr) ((void)0)
#else
#define cmb_assert_debug(expr) cmb_assert_release(expr)
#endif
#define cmb_unused(x) ((void)(x))

/* ---- no-op lifecycle verbs for engine-owned objects (reference
 * create/initialize pairs with explicit terminate/destroy; here per-trial
 * storage is engine-owned POD, reset wholesale at trial start) ---- */
#define cmb_objectqueue_terminate(sim, q) ((void)0)
#define cmb_objectqueue_destroy(sim, q) ((void)0)
#define cmb_priorityqueue_terminate(sim, q) ((void)0)
#define cmb_priorityqueue_destroy(sim, q) ((void)0)
#define cmb_resource_terminate(sim, r) ((void)0)
#define cmb_resource_destroy(sim, r) ((void)0)
#define cmb_resourcepool_terminate(sim, r) ((void)0)
#define cmb_resourcepool_destroy(sim, r) ((void)0)
#define cmb_buffer_terminate(sim, b) ((void)0)
#define cmb_buffer_destroy(sim, b) ((void)0)
#define cmb_condition_terminate(sim, c) ((void)0)
#define cmb_condition_destroy(sim, c) ((void)0)
#define cmb_resourceguard_terminate(sim, g) ((void)0)
#define cmb_resourceguard_destroy(sim, g) ((void)0)
#define cmb_event_queue_create(sim) ((void)0)
#define cmb_event_queue_initialize(sim, t0) ((void)0) /* engine-init'd */
#define cmb_event_queue_terminate(sim) ((void)0)
#define cmb_random_terminate(sim) ((void)0)
#define cmb_dataset_initialize(d, name) ((void)0)
#define cmb_dataset_terminate(d) ((void)0)
#define cmb_timeseries_initialize(t, name) ((void)0)
#define cmb_timeseries_terminate(t) ((void)0)

#ifdef __cplusplus
}
#endif

#endif /* CIMBA_AMD_CIMBA_H */
