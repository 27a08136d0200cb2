// Implementation of the public cmb_* C API (include/cimba.h) over the
// MI355X-native engine: a runtime-configured model ("CModel") whose
// process bodies, event actions and demand predicates are C function
// pointers.  Host-side counterpart of the reference's whole public
// surface (reference src/cimba.c, cmb_event.c, cmb_process.c,
// cmb_resource*.c, cmb_buffer.c, cmb_objectqueue.c, cmb_priorityqueue.c,
// cmb_condition.c, cmb_random.c, cmb_datasummary.c, cmb_logger.c).
// engine headers first: cimba.h defines CMB_* macros (incl. CMB_UNLIMITED)
// that would otherwise collide with the engine's identifiers
#include "cimba/dataset.hpp"
#include "cimba/engine.hpp"
#include "cimba/logger.hpp"
#include "cimba/runner.hpp"

#include "../../../include/cimba.h"

#include <algorithm>
#include <atomic>
#include <cstring>
#include <ctime>
#include <thread>
#include <vector>

using namespace cmb;

namespace {

// ---- host C API envelope -------------------------
// Build-time configurable via -DCIMBA_C_MAX_PROC=... etc.; the defaults
// are sized for host memory (each worker thread owns one engine of this
// shape, ~tens of MB) and far beyond the reference's typical scale.  The
// engine arrays are still bounded — the reference grows without bound —
// but the event heap and object queues spill to a growth tier (SPILL_*
// below), and true exhaustion aborts the trial cleanly.
#ifndef CIMBA_C_MAX_PROC
#define CIMBA_C_MAX_PROC 16384
#endif
#ifndef CIMBA_C_UEV
#define CIMBA_C_UEV 65536
#endif
#ifndef CIMBA_C_NUM_QUEUES
#define CIMBA_C_NUM_QUEUES 64
#endif
#ifndef CIMBA_C_NUM_RES
#define CIMBA_C_NUM_RES 64
#endif
#ifndef CIMBA_C_NUM_POOLS
#define CIMBA_C_NUM_POOLS 64
#endif
#ifndef CIMBA_C_NUM_BUFS
#define CIMBA_C_NUM_BUFS 64
#endif
#ifndef CIMBA_C_NUM_PQ
#define CIMBA_C_NUM_PQ 32
#endif
#ifndef CIMBA_C_NUM_COND
#define CIMBA_C_NUM_COND 64
#endif

constexpr int C_MAX_PROC = CIMBA_C_MAX_PROC;
constexpr int C_UEV = CIMBA_C_UEV;  // pending user events
constexpr int C_NAME = 32;  // reference CMB_PROCESS_NAMEBUF_SZ

struct CModel : ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = C_MAX_PROC;
        static constexpr int MAX_EV = 16384;
        static constexpr int SPILL_EV = 49152;   // events grow to 64K total
        static constexpr bool EV_MAP = true;     // O(1) cancel/reschedule
        static constexpr int TIMERS = 4;
        static constexpr int NUM_QUEUES = CIMBA_C_NUM_QUEUES;
        static constexpr int QCAP = 8192;
        static constexpr int SPILL_Q = 24576;    // queues grow to 32K total
        static constexpr int NUM_RES = CIMBA_C_NUM_RES;
        static constexpr int NUM_POOLS = CIMBA_C_NUM_POOLS;
        static constexpr int NUM_BUFS = CIMBA_C_NUM_BUFS;
        static constexpr int NUM_PQ = CIMBA_C_NUM_PQ;
        static constexpr int PQCAP = 4096;
        static constexpr int SPILL_PQ = 12288;   // PQs grow to 16K total
        static constexpr int NUM_COND = CIMBA_C_NUM_COND;
    };
    struct Params {
        cmb_sim* sim;
    };
    struct Frame {};
    struct Result {};
    struct UEv {
        cmb_event_func* fn;
        void* subj;
        void* obj;
        int32_t next_free;  // -1 = in use
    };
    struct Globals {
        cmb_process_func* fn[C_MAX_PROC];
        void* ctx[C_MAX_PROC];
        char name[C_MAX_PROC][C_NAME];
        void* exit_value[C_MAX_PROC];
        cmb_demand_func* dem_fn[C_MAX_PROC];
        void* dem_ctx[C_MAX_PROC];
        UEv uev[C_UEV];
        int32_t uev_free;  // freelist head
        uint32_t cur_event;  // most recently dispatched user event handle
        int16_t cur_proc;    // currently executing process idx, -1 = none
        int32_t nproc, nq, npq, nres, npool, nbuf, ncond;
        char qname[Cfg::NUM_QUEUES][C_NAME];
        char rname[Cfg::NUM_RES][C_NAME];
        char plname[Cfg::NUM_POOLS][C_NAME];
        char bname[Cfg::NUM_BUFS][C_NAME];
    };

    template <class E_>
    static void step(E_& E, int pidx);
    template <class E_>
    static bool demand(E_& E, int pidx, uint8_t kind, uint32_t ctx);
    template <class E_>
    static void on_event(E_& E, const EvEntry& ev);
    template <class E_>
    static void setup(E_&) {}
    template <class E_>
    static void finish(E_&, Result&) {}
};

using CEngine = Engine<CModel>;

}  // namespace

/* host timeseries wrapper (also the cmb_timeseries representation) */
struct CTimeseries {
    Timeseries ts;
    double end = -1.0;
};

/* host-side recorded histories (reference cmb_*_history: a timeseries of
 * the object's level, appended on every successful state change while
 * recording is on; the C API is host-only, so these live beside the sim
 * instead of inside the device-shaped engine Storage) */
struct SimHist {
    CTimeseries* q[CModel::Cfg::NUM_QUEUES] = {};
    CTimeseries* pq[CModel::Cfg::NUM_PQ] = {};
    CTimeseries* res[CModel::Cfg::NUM_RES] = {};
    CTimeseries* pool[CModel::Cfg::NUM_POOLS] = {};
    CTimeseries* buf[CModel::Cfg::NUM_BUFS] = {};
    void reset() {
        auto z = [](CTimeseries** a, int n) {
            for (int i = 0; i < n; ++i) {
                delete a[i];
                a[i] = nullptr;
            }
        };
        z(q, CModel::Cfg::NUM_QUEUES);
        z(pq, CModel::Cfg::NUM_PQ);
        z(res, CModel::Cfg::NUM_RES);
        z(pool, CModel::Cfg::NUM_POOLS);
        z(buf, CModel::Cfg::NUM_BUFS);
    }
    ~SimHist() { reset(); }
};

struct cmb_sim {
    CEngine* E;
    CModel::Params params;
    uint64_t seed;
    SimHist* hist = nullptr;
};
namespace {
using CStorage = CEngine::Storage;

// append a level sample at the current sim time, if a history is live
void hist_add(const cmb_sim* s, CTimeseries* h, double v) {
    if (h) h->ts.add(v, s->E->now);
}
}

namespace {

// ---- handle encoding: index+1 as pointer ----
template <class T>
T* enc(int idx) {
    return (T*)(uintptr_t)(idx + 1);
}
template <class T>
int dec(const T* h) {
    return (int)((uintptr_t)h - 1);
}

void init_globals(CEngine& E) {
    CModel::Globals& g = E.globals;
    std::memset(&g, 0, sizeof(g));
    for (int i = 0; i < C_UEV; ++i) g.uev[i].next_free = i + 1;
    g.uev[C_UEV - 1].next_free = -1;
    g.uev_free = 0;
    g.cur_proc = -1;
}

int uev_alloc(CEngine& E, cmb_event_func* fn, void* subj, void* obj) {
    CModel::Globals& g = E.globals;
    const int s = g.uev_free;
    if (s < 0) {
        E.fail(ST_HEAP_FULL);
        return -1;
    }
    g.uev_free = g.uev[s].next_free;
    g.uev[s] = {fn, subj, obj, -1};
    return s;
}

void uev_free_slot(CEngine& E, int s) {
    CModel::Globals& g = E.globals;
    g.uev[s].next_free = g.uev_free;
    g.uev_free = s;
}

template <class E_>
void CModel::step(E_& E, int pidx) {
    cmb_sim* sim = E.params->sim;
    Globals& g = E.globals;
    const int16_t prev = g.cur_proc;
    g.cur_proc = (int16_t)pidx;
    if (g.fn[pidx]) g.fn[pidx](sim, enc<cmb_process>(pidx), g.ctx[pidx]);
    g.cur_proc = prev;
}

template <class E_>
bool CModel::demand(E_& E, int pidx, uint8_t kind, uint32_t /*ctx*/) {
    if (kind != DEM_USER) return false;
    Globals& g = E.globals;
    if (!g.dem_fn[pidx]) return false;
    return g.dem_fn[pidx](E.params->sim, g.dem_ctx[pidx]);
}

template <class E_>
void CModel::on_event(E_& E, const EvEntry& ev) {
    if (ev.kind != EV_USER) return;
    const int s = (int)ev.b;
    UEv u = E.globals.uev[s];
    uev_free_slot(E, s);
    E.globals.cur_event = ev.handle;
    if (u.fn) u.fn(E.params->sim, u.subj, u.obj);
}

// ---- executive globals (reference cimba.c statics) ----
std::atomic<uint64_t> g_next{0};
std::atomic<uint64_t> g_total{0};
int g_default_threads = 0;
thread_local void* g_thread_ctx = nullptr;
thread_local int g_thread_id = 0;
void (*g_thread_init)(int) = nullptr;
void (*g_thread_exit)(int) = nullptr;
void (*g_trial_cleanup)(uint64_t) = nullptr;

}  // namespace

extern "C" {

/* ---- executive ---- */

uint64_t cimba_run(void* experiment, uint64_t n, size_t size,
                   cimba_trial_func* trial_fn, uint64_t master_seed,
                   int nthreads) {
    if (nthreads <= 0) nthreads = g_default_threads;
    if (nthreads <= 0) nthreads = (int)std::thread::hardware_concurrency();
    if (nthreads <= 0) nthreads = 1;
    if ((uint64_t)nthreads > n) nthreads = (int)(n ? n : 1);

    g_next.store(0);
    g_total.store(n);
    std::atomic<uint64_t> failed{0};

    auto worker = [&](int widx) {
        g_thread_id = widx;
        if (g_thread_init) g_thread_init(widx);
        auto store = std::make_unique<CStorage>();
        auto eng = std::make_unique<CEngine>(*store);
        auto slab = std::make_unique<CEngine::Spill>();  // HBM-analog tier
        eng->set_spill(slab.get());
        cmb_sim sim;
        sim.E = eng.get();
        sim.params.sim = &sim;
        SimHist hist;
        sim.hist = &hist;
        for (;;) {
            const uint64_t t = g_next.fetch_add(1);
            if (t >= n) break;
            const uint64_t seed = trial_seed(master_seed, t);
            sim.seed = seed;
            logger_ctx().trial = (uint32_t)t;
            logger_ctx().seed = seed;
            logger_ctx().sim_time = 0.0;
            eng->init(&sim.params, seed, (uint32_t)t);
            init_globals(*eng);
            hist.reset();
            try {
                trial_fn(&sim, (char*)experiment + t * size);
            } catch (const TrialAbandon&) {
                eng->fail(ST_USER_ABORT);
                if (g_trial_cleanup) g_trial_cleanup(t);
            }
            if (eng->status != ST_OK) failed.fetch_add(1);
        }
        if (g_thread_exit) g_thread_exit(widx);
    };

    if (nthreads == 1) {
        worker(0);
    } else {
        std::vector<std::thread> th;
        for (int i = 0; i < nthreads; ++i) th.emplace_back(worker, i);
        for (auto& t : th) t.join();
    }
    return failed.load();
}

void cimba_threads_use(int nthreads) { g_default_threads = nthreads; }
uint64_t cimba_run_experiment(void* experiment, uint64_t n, size_t size,
                              cimba_trial_func* trial_fn,
                              uint64_t master_seed, int nthreads) {
    return cimba_run(experiment, n, size, trial_fn, master_seed, nthreads);
}
uint32_t cimba_trial_index(const cmb_sim* s) { return s->E->trial_index; }
uint64_t cimba_trials_total(void) { return g_total.load(); }
int cimba_threads_num(void) { return g_default_threads; }
void cimba_thread_context_set(void* ctx) { g_thread_ctx = ctx; }
void* cimba_thread_context(void) { return g_thread_ctx; }
int cimba_thread_id(void) { return g_thread_id; }
const char* cimba_version(void) { return "cimba-mi355x 0.1.0"; }
uint64_t cimba_trials_remaining(void) {
    const uint64_t next = g_next.load(), total = g_total.load();
    return next >= total ? 0 : total - next;
}
void cimba_trial_abandon(cmb_sim*) { throw TrialAbandon{1}; }
void cimba_thread_hooks_set(void (*init_fn)(int), void (*exit_fn)(int)) {
    g_thread_init = init_fn;
    g_thread_exit = exit_fn;
}
void cimba_trial_cleanup_set(void (*cleanup_fn)(uint64_t)) {
    g_trial_cleanup = cleanup_fn;
}
uint32_t cmb_sim_trial_index(const cmb_sim* s) { return s->E->trial_index; }
uint64_t cmb_sim_trial_seed(const cmb_sim* s) { return s->seed; }
uint64_t cmb_sim_events_dispatched(const cmb_sim* s) {
    return s->E->ev_dispatched;
}

/* ---- clock & events ---- */

double cmb_time(const cmb_sim* s) { return s->E->now; }

uint64_t cmb_event_schedule(cmb_sim* s, cmb_event_func* action, void* subject,
                            void* object, double time, int priority) {
    const int slot = uev_alloc(*s->E, action, subject, object);
    if (slot < 0) return 0;
    return s->E->schedule(EV_USER, 0, 0, (uint64_t)slot, time, priority);
}

bool cmb_event_cancel(cmb_sim* s, uint64_t handle) {
    EvEntry out;
    if (!s->E->evq.cancel((uint32_t)handle, &out)) return false;
    if (out.kind == EV_USER) uev_free_slot(*s->E, (int)out.b);
    if (s->E->n_event_waiters)
        s->E->wake_event_waiters((uint32_t)handle, SIG_CANCELLED);
    return true;
}

bool cmb_event_reschedule(cmb_sim* s, uint64_t handle, double time,
                          int priority) {
    return s->E->event_reschedule((uint32_t)handle, time, priority);
}

uint64_t cmb_event_pattern_count(cmb_sim* s, cmb_event_func* action,
                                 void* subject, void* object) {
    uint64_t cnt = 0;
    auto& q = s->E->evq;
    for (int32_t i = 0; i < q.n; ++i) {
        if (q.at(i).kind != EV_USER) continue;
        const auto& u = s->E->globals.uev[(int)q.at(i).b];
        if ((action == CMB_ANY_ACTION || u.fn == action) &&
            (subject == CMB_ANY_SUBJECT || u.subj == subject) &&
            (object == CMB_ANY_OBJECT || u.obj == object))
            ++cnt;
    }
    return cnt;
}

uint64_t cmb_event_pattern_cancel(cmb_sim* s, cmb_event_func* action,
                                  void* subject, void* object) {
    // restart the scan after every removal: remove_at(i) refills slot i
    // with the last heap entry and sift_up can carry that not-yet-examined
    // entry ABOVE i, where a single ascending scan would miss it (same
    // fix as HashHeap::pattern_cancel; caught by tests/test_hashheap.py)
    uint64_t cnt = 0;
    auto& q = s->E->evq;
    for (;;) {
        int32_t hit = -1;
        for (int32_t i = 0; i < q.n; ++i) {
            if (q.at(i).kind != EV_USER) continue;
            const auto& u = s->E->globals.uev[(int)q.at(i).b];
            if ((action == CMB_ANY_ACTION || u.fn == action) &&
                (subject == CMB_ANY_SUBJECT || u.subj == subject) &&
                (object == CMB_ANY_OBJECT || u.obj == object)) {
                hit = i;
                break;
            }
        }
        if (hit < 0) return cnt;
        const uint32_t h = q.at(hit).handle;
        uev_free_slot(*s->E, (int)q.at(hit).b);
        q.remove_at(hit);
        if (s->E->n_event_waiters)
            s->E->wake_event_waiters(h, SIG_CANCELLED);
        ++cnt;
    }
}

void cmb_event_queue_execute(cmb_sim* s) {
    s->E->run(1.0e308, UINT64_C(0xFFFFFFFFFFFFFFFF));
}
bool cmb_event_execute_next(cmb_sim* s) { return s->E->dispatch_one(); }
uint64_t cmb_event_queue_count(const cmb_sim* s) {
    return (uint64_t)s->E->evq.n;
}
bool cmb_event_queue_is_empty(const cmb_sim* s) { return s->E->evq.empty(); }
void cmb_event_queue_clear(cmb_sim* s) {
    auto& q = s->E->evq;
    for (int32_t i = 0; i < q.n; ++i)
        if (q.at(i).kind == EV_USER) uev_free_slot(*s->E, (int)q.at(i).b);
    q.reset();
}
static int32_t find_slot_(const cmb_sim* s, uint64_t handle) {
    return s->E->evq.find_index((uint32_t)handle);
}
bool cmb_event_is_scheduled(const cmb_sim* s, uint64_t handle) {
    return find_slot_(s, handle) >= 0;
}
double cmb_event_time(const cmb_sim* s, uint64_t handle) {
    const int32_t i = find_slot_(s, handle);
    return i >= 0 ? s->E->evq.at(i).t : -1.0;
}
int cmb_event_priority(const cmb_sim* s, uint64_t handle) {
    const int32_t i = find_slot_(s, handle);
    if (i < 0) return 0;
    return 32767 - (int)(s->E->evq.at(i).pseq >> 48);
}
bool cmb_event_reprioritize(cmb_sim* s, uint64_t handle, int priority) {
    const int32_t i = find_slot_(s, handle);
    if (i < 0) return false;
    const double t = s->E->evq.at(i).t;
    return s->E->event_reschedule((uint32_t)handle, t, priority);
}
uint64_t cmb_event_pattern_find(cmb_sim* s, cmb_event_func* action,
                                void* subject, void* object) {
    const auto& q = s->E->evq;
    for (int32_t i = 0; i < q.n; ++i) {
        if (q.at(i).kind != EV_USER) continue;
        const auto& u = s->E->globals.uev[(int)q.at(i).b];
        if ((action == CMB_ANY_ACTION || u.fn == action) &&
            (subject == CMB_ANY_SUBJECT || u.subj == subject) &&
            (object == CMB_ANY_OBJECT || u.obj == object))
            return q.at(i).handle;
    }
    return 0;
}
void cmb_event_queue_execute_until(cmb_sim* s, double until) {
    s->E->run(until, UINT64_C(0xFFFFFFFFFFFFFFFF));
}
uint64_t cmb_event_current(const cmb_sim* s) {
    // reference cmb_event.h:189 — handle of the currently / most
    // recently executed user event, 0 if none yet
    return (uint64_t)s->E->globals.cur_event;
}

/* ---- processes ---- */

cmb_process* cmb_process_current(const cmb_sim* s) {
    // reference cmb_process.h:256 — the process whose body is executing,
    // NULL from the dispatcher / trial function (no coroutines here: the
    // trampoline in CModel::step tracks the running protothread instead)
    const int16_t p = s->E->globals.cur_proc;
    return p < 0 ? nullptr : enc<cmb_process>((int)p);
}

cmb_process* cmb_process_spawn(cmb_sim* s, const char* name,
                               cmb_process_func* fn, void* ctx,
                               int priority) {
    CEngine& E = *s->E;
    const int idx = E.proc_alloc();
    if (idx < 0) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    E.proc_init(idx, 0, priority);
    E.globals.fn[idx] = fn;
    E.globals.ctx[idx] = ctx;
    std::strncpy(E.globals.name[idx], name ? name : "", C_NAME - 1);
    E.globals.name[idx][C_NAME - 1] = 0;
    return enc<cmb_process>(idx);
}

void cmb_process_start(cmb_sim* s, cmb_process* p) {
    s->E->proc_start(dec(p));
}
void cmb_process_start_at(cmb_sim* s, cmb_process* p, double delay) {
    s->E->proc_start(dec(p), delay);
}
void cmb_process_interrupt(cmb_sim* s, cmb_process* p, int64_t sig) {
    s->E->proc_interrupt(dec(p), sig);
}
void cmb_process_stop(cmb_sim* s, cmb_process* p) { s->E->proc_stop(dec(p)); }
void cmb_process_resume(cmb_sim* s, cmb_process* p) {
    s->E->proc_interrupt(dec(p), SIG_SUCCESS);
}
void cmb_process_priority_set(cmb_sim* s, cmb_process* p, int priority) {
    s->E->proc_priority_set(dec(p), priority);
}
int64_t cmb_process_priority(const cmb_sim* s, const cmb_process* p) {
    return s->E->procs[dec(p)].priority;
}
const char* cmb_process_name(const cmb_sim* s, const cmb_process* p) {
    return s->E->globals.name[dec(p)];
}
int cmb_process_state(const cmb_sim* s, const cmb_process* p) {
    return (int)s->E->procs[dec(p)].state;
}
int64_t cmb_process_signal(const cmb_sim* s, const cmb_process* p) {
    return s->E->procs[dec(p)].sig;
}
void* cmb_process_context(const cmb_sim* s, const cmb_process* p) {
    return s->E->globals.ctx[dec(p)];
}
void cmb_process_kill(cmb_sim* s, cmb_process* p) { s->E->proc_stop(dec(p)); }
int cmb_process_status(const cmb_sim* s, const cmb_process* p) {
    return (int)s->E->procs[dec(p)].state;
}
void cmb_process_name_set(cmb_sim* s, cmb_process* p, const char* name) {
    std::strncpy(s->E->globals.name[dec(p)], name ? name : "", C_NAME - 1);
    s->E->globals.name[dec(p)][C_NAME - 1] = 0;
}
cmb_process* cmb_process_create(cmb_sim* s) {
    CEngine& E = *s->E;
    const int idx = E.proc_alloc();
    if (idx < 0) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    E.globals.fn[idx] = nullptr;
    E.globals.ctx[idx] = nullptr;
    E.globals.exit_value[idx] = nullptr;
    return enc<cmb_process>(idx);
}
void cmb_process_initialize(cmb_sim* s, cmb_process* p, const char* name,
                            cmb_process_func* fn, void* ctx, int priority) {
    CEngine& E = *s->E;
    const int idx = dec(p);
    E.proc_init(idx, 0, priority);
    E.globals.fn[idx] = fn;
    E.globals.ctx[idx] = ctx;
    cmb_process_name_set(s, p, name);
}
void cmb_process_terminate(cmb_sim*, cmb_process*) {}
void cmb_process_destroy(cmb_sim*, cmb_process*) {}
void cmb_process_exit_value_set_(cmb_sim* s, cmb_process* p, void* value) {
    s->E->globals.exit_value[dec(p)] = value;
}
void* cmb_process_exit_value(const cmb_sim* s, const cmb_process* p) {
    return s->E->globals.exit_value[dec(p)];
}
bool cmb_process_timer_set(cmb_sim* s, cmb_process* p, int slot,
                           double delay, int64_t sig) {
    if (slot < 1 || slot >= CModel::Cfg::TIMERS) return false;  // 0 reserved
    s->E->timer_cancel(s->E->procs[dec(p)], slot);
    return s->E->timer_add(s->E->procs[dec(p)], slot, delay, sig);
}

int cmb_proc_pc_(const cmb_sim* s, const cmb_process* p) {
    return s->E->procs[dec(p)].pc;
}
void cmb_proc_set_pc_(cmb_sim* s, cmb_process* p, int pc) {
    s->E->procs[dec(p)].pc = (int16_t)pc;
}
void cmb_proc_resumed_(cmb_sim* s, cmb_process* p) {
    s->E->await_cleanup(s->E->procs[dec(p)]);
}
void cmb_hold_setup_(cmb_sim* s, cmb_process* p, double duration) {
    s->E->hold_setup(s->E->procs[dec(p)], duration);
}
int cmb_wait_process_setup_(cmb_sim* s, cmb_process* p, cmb_process* tgt) {
    return s->E->wait_proc_setup(s->E->procs[dec(p)], dec(tgt)) ? 1 : 0;
}
void cmb_wait_event_setup_(cmb_sim* s, cmb_process* p, uint64_t handle) {
    s->E->wait_event_setup(s->E->procs[dec(p)], (uint32_t)handle);
}
void cmb_proc_finish_(cmb_sim* s, cmb_process* p) {
    s->E->proc_finish(s->E->procs[dec(p)]);
}
void cmb_timer_arm_(cmb_sim* s, cmb_process* p, double delay, int64_t sig) {
    s->E->timer_add(s->E->procs[dec(p)], 0, delay, sig);
}
void cmb_timer_disarm_(cmb_sim* s, cmb_process* p) {
    s->E->timer_cancel(s->E->procs[dec(p)], 0);
}
bool cmb_process_timer_add(cmb_sim* s, cmb_process* p, int slot,
                           double delay, int64_t sig) {
    if (slot < 1 || slot >= CModel::Cfg::TIMERS) return false;  // 0 reserved
    return s->E->timer_add(s->E->procs[dec(p)], slot, delay, sig);
}
void cmb_process_timer_cancel(cmb_sim* s, cmb_process* p, int slot) {
    if (slot < 1 || slot >= CModel::Cfg::TIMERS) return;  // 0 reserved
    s->E->timer_cancel(s->E->procs[dec(p)], slot);
}
void cmb_process_timer_clear(cmb_sim* s, cmb_process* p) {
    for (int t = 1; t < CModel::Cfg::TIMERS; ++t)  // slot 0 reserved
        s->E->timer_cancel(s->E->procs[dec(p)], t);
}
bool cmb_process_timer_pending(const cmb_sim* s, const cmb_process* p,
                               int slot) {
    if (slot < 1 || slot >= CModel::Cfg::TIMERS) return false;  // 0 reserved
    return s->E->procs[dec(p)].timers[slot] != 0;
}
int cmb_sim_ok_(const cmb_sim* s) { return s->E->status == ST_OK; }

/* ---- toolkit ---- */

cmb_objectqueue* cmb_objectqueue_create(cmb_sim* s) {
    CEngine& E = *s->E;
    if (E.globals.nq >= CModel::Cfg::NUM_QUEUES) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    return enc<cmb_objectqueue>(E.globals.nq++);
}
void cmb_objectqueue_initialize(cmb_sim* s, cmb_objectqueue* q,
                                const char* name, int32_t capacity) {
    s->E->queues[dec(q)].limit =
        capacity == CMB_UNLIMITED ? CMB_UNLIMITED : capacity;
    std::strncpy(s->E->globals.qname[dec(q)], name ? name : "", C_NAME - 1);
    s->E->globals.qname[dec(q)][C_NAME - 1] = 0;
}
const char* cmb_objectqueue_name(const cmb_sim* s, const cmb_objectqueue* q) {
    return s->E->globals.qname[dec(q)];
}
uint64_t cmb_objectqueue_space(const cmb_sim* s, const cmb_objectqueue* q) {
    const auto& Q = s->E->queues[dec(q)];
    const int32_t lim =
        Q.limit == CMB_UNLIMITED ? CModel::Cfg::QCAP : Q.limit;
    return (uint64_t)(lim - Q.len);
}
uint64_t cmb_objectqueue_length(const cmb_sim* s, const cmb_objectqueue* q) {
    return (uint64_t)s->E->queues[dec(q)].len;
}
uint64_t cmb_objectqueue_position(const cmb_sim* s, const cmb_objectqueue* q,
                                  const void* object) {
    const auto& Q = s->E->queues[dec(q)];
    for (int32_t i = 0; i < Q.len; ++i) {
        const int32_t idx = (Q.head + i) % CModel::Cfg::QCAP;
        if (Q.ring[idx] == (uint64_t)object) return (uint64_t)i + 1;
    }
    return 0;
}
void cmb_objectqueue_recording_start(cmb_sim* s, cmb_objectqueue* q) {
    auto& Q = s->E->queues[dec(q)];
    Q.recording = 1;
    Q.len_stats.reset();
    Q.t_last = s->E->now;
    if (s->hist && !s->hist->q[dec(q)])
        s->hist->q[dec(q)] = new CTimeseries;
    if (s->hist) hist_add(s, s->hist->q[dec(q)], (double)Q.len);
}
cmb_timeseries* cmb_objectqueue_history(const cmb_sim* s,
                                        const cmb_objectqueue* q) {
    // reference cmb_objectqueue.h:226 — queue-length history (recorded
    // while recording is on; NULL before the first recording_start)
    return s->hist ? (cmb_timeseries*)s->hist->q[dec(q)] : NULL;
}
void cmb_objectqueue_recording_stop(cmb_sim* s, cmb_objectqueue* q) {
    s->E->queues[dec(q)].recording = 0;
}
void cmb_objectqueue_stats(cmb_sim* s, const cmb_objectqueue* q,
                           double out4[4]) {
    auto Q = s->E->queues[dec(q)];  // copy
    Q.len_stats.add((double)Q.len, s->E->now - Q.t_last);
    out4[0] = Q.len_stats.mean;
    out4[1] = Q.len_stats.stddev();
    out4[2] = Q.len_stats.mn;
    out4[3] = Q.len_stats.mx;
}

bool cmb_queue_try_put_(cmb_sim* s, cmb_objectqueue* q, cmb_process* p,
                        void* object) {
    if (!s->E->q_try_put(dec(q), s->E->procs[dec(p)], (uint64_t)object))
        return false;
    const auto& Q = s->E->queues[dec(q)];
    if (Q.recording && s->hist)
        hist_add(s, s->hist->q[dec(q)], (double)Q.len);
    return true;
}
bool cmb_queue_try_get_(cmb_sim* s, cmb_objectqueue* q, cmb_process* p,
                        void** object) {
    uint64_t v = 0;
    if (!s->E->q_try_get(dec(q), s->E->procs[dec(p)], &v)) return false;
    *object = (void*)v;
    const auto& Q = s->E->queues[dec(q)];
    if (Q.recording && s->hist)
        hist_add(s, s->hist->q[dec(q)], (double)Q.len);
    return true;
}
void cmb_queue_wait_space_(cmb_sim* s, cmb_objectqueue* q, cmb_process* p) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.queues[dec(q)].g_rear, DEM_QSPACE,
                 (uint32_t)dec(q));
}
void cmb_queue_wait_object_(cmb_sim* s, cmb_objectqueue* q, cmb_process* p) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.queues[dec(q)].g_front, DEM_QOBJ,
                 (uint32_t)dec(q));
}

cmb_priorityqueue* cmb_priorityqueue_create(cmb_sim* s) {
    CEngine& E = *s->E;
    if (E.globals.npq >= CModel::Cfg::NUM_PQ) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    return enc<cmb_priorityqueue>(E.globals.npq++);
}
void cmb_priorityqueue_initialize(cmb_sim* s, cmb_priorityqueue* q,
                                  const char* /*name*/, int32_t capacity) {
    s->E->pqueues[dec(q)].limit =
        capacity == CMB_UNLIMITED ? CModel::Cfg::PQCAP : capacity;
}
uint64_t cmb_priorityqueue_length(const cmb_sim* s,
                                  const cmb_priorityqueue* q) {
    return (uint64_t)s->E->pqueues[dec(q)].len;
}
uint64_t cmb_priorityqueue_space(const cmb_sim* s,
                                 const cmb_priorityqueue* q) {
    const auto& Q = s->E->pqueues[dec(q)];
    return (uint64_t)(Q.limit - Q.len);
}
uint64_t cmb_priorityqueue_position(const cmb_sim* s,
                                    const cmb_priorityqueue* q,
                                    const void* object) {
    const auto& Q = s->E->pqueues[dec(q)];
    for (int32_t i = 0; i < Q.len; ++i) {
        if (Q.val[i] == (uint64_t)object) {
            uint64_t pos = 1;
            for (int32_t j = 0; j < Q.len; ++j)
                if (Q.key[j] < Q.key[i]) ++pos;
            return pos;
        }
    }
    return 0;
}
static void pq_remove_at_(cmb_sim* s, int qi, int32_t i) {
    auto& Q = s->E->pqueues[qi];
    --Q.len;
    Q.key[i] = Q.key[Q.len];
    Q.val[i] = Q.val[Q.len];
    for (int32_t r = Q.len / 2 - 1; r >= 0; --r) {  // full re-heapify
        int32_t n0 = r;
        for (;;) {
            int32_t cle = 2 * n0 + 1;
            if (cle >= Q.len) break;
            if (cle + 1 < Q.len && Q.key[cle + 1] < Q.key[cle]) ++cle;
            if (Q.key[n0] <= Q.key[cle]) break;
            std::swap(Q.key[n0], Q.key[cle]);
            std::swap(Q.val[n0], Q.val[cle]);
            n0 = cle;
        }
    }
}
bool cmb_priorityqueue_cancel(cmb_sim* s, cmb_priorityqueue* q,
                              const void* object) {
    auto& Q = s->E->pqueues[dec(q)];
    for (int32_t i = 0; i < Q.len; ++i) {
        if (Q.val[i] == (uint64_t)object) {
            pq_remove_at_(s, dec(q), i);
            s->E->guard_signal(Q.g_rear);  // space freed
            return true;
        }
    }
    return false;
}
bool cmb_priorityqueue_reprioritize(cmb_sim* s, cmb_priorityqueue* q,
                                    const void* object, int priority) {
    auto& Q = s->E->pqueues[dec(q)];
    for (int32_t i = 0; i < Q.len; ++i) {
        if (Q.val[i] == (uint64_t)object) {
            pq_remove_at_(s, dec(q), i);
            const uint64_t key =
                ((uint64_t)(uint16_t)(32767 - priority) << 32) | Q.seq++;
            int32_t j = Q.len++;
            while (j > 0) {
                const int32_t par = (j - 1) / 2;
                if (Q.key[par] <= key) break;
                Q.key[j] = Q.key[par];
                Q.val[j] = Q.val[par];
                j = par;
            }
            Q.key[j] = key;
            Q.val[j] = (uint64_t)object;
            return true;
        }
    }
    return false;
}
void cmb_priorityqueue_recording_start(cmb_sim* s, cmb_priorityqueue* q) {
    auto& Q = s->E->pqueues[dec(q)];
    Q.recording = 1;
    Q.len_stats.reset();
    Q.t_last = s->E->now;
    if (s->hist && !s->hist->pq[dec(q)])
        s->hist->pq[dec(q)] = new CTimeseries;
    if (s->hist) hist_add(s, s->hist->pq[dec(q)], (double)Q.len);
}
cmb_timeseries* cmb_priorityqueue_history(const cmb_sim* s,
                                          const cmb_priorityqueue* q) {
    // reference cmb_priorityqueue.h:258
    return s->hist ? (cmb_timeseries*)s->hist->pq[dec(q)] : NULL;
}
void cmb_priorityqueue_recording_stop(cmb_sim* s, cmb_priorityqueue* q) {
    s->E->pqueues[dec(q)].recording = 0;
}
void cmb_priorityqueue_report_print(cmb_sim* s, const cmb_priorityqueue* q,
                                    FILE* out) {
    if (!out) out = stderr;
    auto Q = s->E->pqueues[dec(q)];  // copy
    Q.len_stats.add((double)Q.len, s->E->now - Q.t_last);
    fprintf(out, "pqueue report @ t=%.6f: length=%d mean=%.4f sd=%.4f\n",
            s->E->now, Q.len, Q.len_stats.mean, Q.len_stats.stddev());
}
bool cmb_pqueue_try_put_(cmb_sim* s, cmb_priorityqueue* q, cmb_process* p,
                         void* object, int priority) {
    if (!s->E->pq_try_put(dec(q), s->E->procs[dec(p)], (uint64_t)object,
                          priority))
        return false;
    const auto& Q = s->E->pqueues[dec(q)];
    if (Q.recording && s->hist)
        hist_add(s, s->hist->pq[dec(q)], (double)Q.len);
    return true;
}
bool cmb_pqueue_try_get_(cmb_sim* s, cmb_priorityqueue* q, cmb_process* p,
                         void** object) {
    uint64_t v = 0;
    if (!s->E->pq_try_get(dec(q), s->E->procs[dec(p)], &v)) return false;
    *object = (void*)v;
    const auto& Q = s->E->pqueues[dec(q)];
    if (Q.recording && s->hist)
        hist_add(s, s->hist->pq[dec(q)], (double)Q.len);
    return true;
}
void cmb_pqueue_wait_space_(cmb_sim* s, cmb_priorityqueue* q,
                            cmb_process* p) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.pqueues[dec(q)].g_rear, DEM_PQSP,
                 (uint32_t)dec(q));
}
void cmb_pqueue_wait_object_(cmb_sim* s, cmb_priorityqueue* q,
                             cmb_process* p) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.pqueues[dec(q)].g_front, DEM_PQOBJ,
                 (uint32_t)dec(q));
}

cmb_resource* cmb_resource_create(cmb_sim* s) {
    CEngine& E = *s->E;
    if (E.globals.nres >= CModel::Cfg::NUM_RES) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    return enc<cmb_resource>(E.globals.nres++);
}
void cmb_resource_initialize(cmb_sim* s, cmb_resource* r, const char* name) {
    std::strncpy(s->E->globals.rname[dec(r)], name ? name : "", C_NAME - 1);
    s->E->globals.rname[dec(r)][C_NAME - 1] = 0;
}
const char* cmb_resource_name(const cmb_sim* s, const cmb_resource* r) {
    return s->E->globals.rname[dec(r)];
}
void cmb_resource_release(cmb_sim* s, cmb_resource* r, cmb_process*) {
    s->E->resource_release(dec(r));
    const auto& R = s->E->resources[dec(r)];
    if (R.recording && s->hist) hist_add(s, s->hist->res[dec(r)], 0.0);
}
bool cmb_resource_in_use(const cmb_sim* s, const cmb_resource* r) {
    return s->E->resources[dec(r)].holder >= 0;
}
cmb_process* cmb_resource_holder(const cmb_sim* s, const cmb_resource* r) {
    const int h = s->E->resources[dec(r)].holder;
    return h < 0 ? nullptr : enc<cmb_process>(h);
}
void cmb_resource_recording_stop(cmb_sim* s, cmb_resource* r) {
    s->E->resources[dec(r)].recording = 0;
}
void cmb_resource_start_recording(cmb_sim* s, cmb_resource* r) {
    cmb_resource_recording_start(s, r);
}
void cmb_resource_stop_recording(cmb_sim* s, cmb_resource* r) {
    cmb_resource_recording_stop(s, r);
}
bool cmb_resource_available(const cmb_sim* s, const cmb_resource* r) {
    return s->E->resources[dec(r)].holder < 0;
}
bool cmb_resource_held_by_process(const cmb_sim* s, const cmb_resource* r,
                                  const cmb_process* p) {
    return s->E->resources[dec(r)].holder == (int16_t)dec(p);
}
void cmb_resource_recording_start(cmb_sim* s, cmb_resource* r) {
    auto& R = s->E->resources[dec(r)];
    R.recording = 1;
    R.busy.reset();
    R.t_last = s->E->now;
    if (s->hist && !s->hist->res[dec(r)])
        s->hist->res[dec(r)] = new CTimeseries;
    if (s->hist)
        hist_add(s, s->hist->res[dec(r)], R.holder >= 0 ? 1.0 : 0.0);
}
cmb_timeseries* cmb_resource_history(const cmb_sim* s,
                                     const cmb_resource* r) {
    // reference cmb_resource.h:236 — busy (0/1) history
    return s->hist ? (cmb_timeseries*)s->hist->res[dec(r)] : NULL;
}
void cmb_resource_stats(cmb_sim* s, const cmb_resource* r, double out4[4]) {
    auto R = s->E->resources[dec(r)];  // copy
    R.busy.add(R.holder >= 0 ? 1.0 : 0.0, s->E->now - R.t_last);
    out4[0] = R.busy.mean;
    out4[1] = R.busy.stddev();
    out4[2] = R.busy.mn;
    out4[3] = R.busy.mx;
}
bool cmb_resource_try_acquire_(cmb_sim* s, cmb_resource* r, cmb_process* p) {
    if (!s->E->res_try_acquire(dec(r), s->E->procs[dec(p)])) return false;
    const auto& R = s->E->resources[dec(r)];
    if (R.recording && s->hist) hist_add(s, s->hist->res[dec(r)], 1.0);
    return true;
}
bool cmb_resource_try_preempt_(cmb_sim* s, cmb_resource* r, cmb_process* p) {
    return s->E->res_try_preempt(dec(r), s->E->procs[dec(p)]);
}
void cmb_resource_wait_(cmb_sim* s, cmb_resource* r, cmb_process* p) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.resources[dec(r)].gid, DEM_RES,
                 (uint32_t)dec(r));
}

cmb_resourcepool* cmb_resourcepool_create(cmb_sim* s) {
    CEngine& E = *s->E;
    if (E.globals.npool >= CModel::Cfg::NUM_POOLS) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    return enc<cmb_resourcepool>(E.globals.npool++);
}
void cmb_resourcepool_initialize(cmb_sim* s, cmb_resourcepool* r,
                                 const char* name, int32_t capacity) {
    s->E->pools[dec(r)].capacity = capacity;
    std::strncpy(s->E->globals.plname[dec(r)], name ? name : "", C_NAME - 1);
    s->E->globals.plname[dec(r)][C_NAME - 1] = 0;
}
const char* cmb_resourcepool_get_name(const cmb_sim* s,
                                      const cmb_resourcepool* r) {
    return s->E->globals.plname[dec(r)];
}
void cmb_resourcepool_release(cmb_sim* s, cmb_resourcepool* r,
                              cmb_process* holder, int32_t amount) {
    s->E->pool_release_for(dec(r), dec(holder), amount);
    const auto& P = s->E->pools[dec(r)];
    if (P.recording && s->hist)
        hist_add(s, s->hist->pool[dec(r)], (double)P.in_use);
}
int32_t cmb_resourcepool_holding(const cmb_sim* s, const cmb_resourcepool* r,
                                 const cmb_process* p) {
    return s->E->pool_holding(dec(r), dec(p));
}
bool cmb_pool_try_preempt_(cmb_sim* s, cmb_resourcepool* r, cmb_process* p,
                           int32_t want) {
    return s->E->pool_try_preempt(dec(r), s->E->procs[dec(p)], want);
}
int32_t cmb_resourcepool_capacity(const cmb_sim* s,
                                  const cmb_resourcepool* r) {
    return s->E->pools[dec(r)].capacity;
}
int32_t cmb_resourcepool_in_use(const cmb_sim* s, const cmb_resourcepool* r) {
    return s->E->pools[dec(r)].in_use;
}
int32_t cmb_resourcepool_available(const cmb_sim* s,
                                   const cmb_resourcepool* r) {
    return s->E->pools[dec(r)].capacity - s->E->pools[dec(r)].in_use;
}
int32_t cmb_resourcepool_held(const cmb_sim* s, const cmb_resourcepool* r) {
    return s->E->pools[dec(r)].in_use;
}
int32_t cmb_resourcepool_held_by_process(const cmb_sim* s,
                                         const cmb_resourcepool* r,
                                         const cmb_process* p) {
    return s->E->pool_holding(dec(r), dec(p));
}
void cmb_resourcepool_start_recording(cmb_sim* s, cmb_resourcepool* r) {
    auto& P = s->E->pools[dec(r)];
    P.recording = 1;
    P.use_stats.reset();
    P.t_last = s->E->now;
    if (s->hist && !s->hist->pool[dec(r)])
        s->hist->pool[dec(r)] = new CTimeseries;
    if (s->hist) hist_add(s, s->hist->pool[dec(r)], (double)P.in_use);
}
cmb_timeseries* cmb_resourcepool_get_history(const cmb_sim* s,
                                             const cmb_resourcepool* r) {
    // reference cmb_resourcepool.h:293 — units-in-use history
    return s->hist ? (cmb_timeseries*)s->hist->pool[dec(r)] : NULL;
}
void cmb_resourcepool_stop_recording(cmb_sim* s, cmb_resourcepool* r) {
    s->E->pools[dec(r)].recording = 0;
}
void cmb_resourcepool_stats(cmb_sim* s, const cmb_resourcepool* r,
                            double out4[4]) {
    auto P = s->E->pools[dec(r)];  // copy
    P.use_stats.add((double)P.in_use, s->E->now - P.t_last);
    out4[0] = P.use_stats.mean;
    out4[1] = P.use_stats.stddev();
    out4[2] = P.use_stats.mn;
    out4[3] = P.use_stats.mx;
}
int32_t cmb_pool_try_take_(cmb_sim* s, cmb_resourcepool* r, cmb_process* p,
                           int32_t want) {
    const int32_t got = s->E->pool_try_take(dec(r), s->E->procs[dec(p)], want);
    const auto& P = s->E->pools[dec(r)];
    if (got > 0 && P.recording && s->hist)
        hist_add(s, s->hist->pool[dec(r)], (double)P.in_use);
    return got;
}
bool cmb_pool_try_take_all_(cmb_sim* s, cmb_resourcepool* r, cmb_process* p,
                            int32_t want) {
    if (!s->E->pool_try_take_all(dec(r), s->E->procs[dec(p)], want))
        return false;
    const auto& P = s->E->pools[dec(r)];
    if (P.recording && s->hist)
        hist_add(s, s->hist->pool[dec(r)], (double)P.in_use);
    return true;
}
void cmb_pool_wait_(cmb_sim* s, cmb_resourcepool* r, cmb_process* p) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.pools[dec(r)].gid, DEM_POOL,
                 (uint32_t)dec(r));
}
void cmb_pool_wait_ge_(cmb_sim* s, cmb_resourcepool* r, cmb_process* p,
                       int32_t amount) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.pools[dec(r)].gid, DEM_POOL_GE,
                 (uint32_t)dec(r) | ((uint32_t)amount << 8));
}

cmb_buffer* cmb_buffer_create(cmb_sim* s) {
    CEngine& E = *s->E;
    if (E.globals.nbuf >= CModel::Cfg::NUM_BUFS) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    return enc<cmb_buffer>(E.globals.nbuf++);
}
void cmb_buffer_initialize(cmb_sim* s, cmb_buffer* b, const char* name,
                           int64_t capacity, int64_t initial_level) {
    std::strncpy(s->E->globals.bname[dec(b)], name ? name : "", C_NAME - 1);
    s->E->globals.bname[dec(b)][C_NAME - 1] = 0;
    auto& B = s->E->buffers[dec(b)];
    B.capacity = capacity == CMB_UNLIMITED ? INT64_MAX / 2 : capacity;
    B.level = initial_level;
}
int64_t cmb_buffer_level(const cmb_sim* s, const cmb_buffer* b) {
    return s->E->buffers[dec(b)].level;
}
int64_t cmb_buffer_capacity(const cmb_sim* s, const cmb_buffer* b) {
    return s->E->buffers[dec(b)].capacity;
}
const char* cmb_buffer_name(const cmb_sim* s, const cmb_buffer* b) {
    return s->E->globals.bname[dec(b)];
}
int64_t cmb_buffer_space(const cmb_sim* s, const cmb_buffer* b) {
    const auto& B = s->E->buffers[dec(b)];
    return B.capacity - B.level;
}
void cmb_buffer_recording_start(cmb_sim* s, cmb_buffer* b) {
    auto& B = s->E->buffers[dec(b)];
    B.recording = 1;
    B.level_stats.reset();
    B.t_last = s->E->now;
    if (s->hist && !s->hist->buf[dec(b)])
        s->hist->buf[dec(b)] = new CTimeseries;
    if (s->hist) hist_add(s, s->hist->buf[dec(b)], (double)B.level);
}
cmb_timeseries* cmb_buffer_history(const cmb_sim* s, const cmb_buffer* b) {
    // reference cmb_buffer.h:230 — level history
    return s->hist ? (cmb_timeseries*)s->hist->buf[dec(b)] : NULL;
}
void cmb_buffer_recording_stop(cmb_sim* s, cmb_buffer* b) {
    s->E->buffers[dec(b)].recording = 0;
}
void cmb_buffer_stats(cmb_sim* s, const cmb_buffer* b, double out4[4]) {
    auto B = s->E->buffers[dec(b)];  // copy
    B.level_stats.add((double)B.level, s->E->now - B.t_last);
    out4[0] = B.level_stats.mean;
    out4[1] = B.level_stats.stddev();
    out4[2] = B.level_stats.mn;
    out4[3] = B.level_stats.mx;
}
void cmb_buffer_print_report(cmb_sim* s, const cmb_buffer* b, FILE* out) {
    if (!out) out = stderr;
    double st[4];
    cmb_buffer_stats(s, b, st);
    fprintf(out,
            "buffer report @ t=%.6f: level=%lld mean=%.4f sd=%.4f\n",
            s->E->now, (long long)s->E->buffers[dec(b)].level, st[0], st[1]);
}
bool cmb_buffer_try_get_(cmb_sim* s, cmb_buffer* b, cmb_process* p,
                         int64_t amount) {
    if (!s->E->buf_try_get(dec(b), s->E->procs[dec(p)], amount))
        return false;
    const auto& B = s->E->buffers[dec(b)];
    if (B.recording && s->hist)
        hist_add(s, s->hist->buf[dec(b)], (double)B.level);
    return true;
}
bool cmb_buffer_try_put_(cmb_sim* s, cmb_buffer* b, cmb_process* p,
                         int64_t amount) {
    if (!s->E->buf_try_put(dec(b), s->E->procs[dec(p)], amount))
        return false;
    const auto& B = s->E->buffers[dec(b)];
    if (B.recording && s->hist)
        hist_add(s, s->hist->buf[dec(b)], (double)B.level);
    return true;
}
void cmb_buffer_wait_level_(cmb_sim* s, cmb_buffer* b, cmb_process* p,
                            int64_t amount) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.buffers[dec(b)].g_get, DEM_BUF_GE,
                 (uint32_t)dec(b) | ((uint32_t)amount << 8));
}
void cmb_buffer_wait_space_(cmb_sim* s, cmb_buffer* b, cmb_process* p,
                            int64_t amount) {
    CEngine& E = *s->E;
    E.guard_wait(E.procs[dec(p)], E.buffers[dec(b)].g_put, DEM_BUF_SP,
                 (uint32_t)dec(b) | ((uint32_t)amount << 8));
}

cmb_condition* cmb_condition_create(cmb_sim* s) {
    CEngine& E = *s->E;
    if (E.globals.ncond >= CModel::Cfg::NUM_COND) {
        E.fail(ST_BAD_STATE);
        return nullptr;
    }
    return enc<cmb_condition>(E.globals.ncond++);
}
void cmb_condition_initialize(cmb_sim*, cmb_condition*, const char*) {}
uint64_t cmb_condition_signal(cmb_sim* s, cmb_condition* c) {
    return s->E->condition_signal(dec(c));
}
void cmb_condition_wait_setup_(cmb_sim* s, cmb_condition* c, cmb_process* p,
                               cmb_demand_func* demand, void* ctx) {
    CEngine& E = *s->E;
    const int pidx = dec(p);
    E.globals.dem_fn[pidx] = demand;
    E.globals.dem_ctx[pidx] = ctx;
    E.guard_wait(E.procs[pidx], E.conds[dec(c)].gid, DEM_USER, 0);
}
bool cmb_condition_cancel(cmb_sim* s, cmb_condition* c, cmb_process* p) {
    CEngine& E = *s->E;
    auto& pr = E.procs[dec(p)];
    if (pr.await_kind != AW_GUARD || pr.gid != E.conds[dec(c)].gid)
        return false;
    E.proc_interrupt(dec(p), SIG_CANCELLED);
    return true;
}
void cmb_condition_subscribe_resource(cmb_sim* s, cmb_condition* c,
                                      cmb_resource* r) {
    s->E->condition_observe(dec(c), s->E->resources[dec(r)].gid);
}
void cmb_condition_subscribe_queue(cmb_sim* s, cmb_condition* c,
                                   cmb_objectqueue* q) {
    s->E->condition_observe(dec(c), s->E->queues[dec(q)].g_front);
}
void cmb_condition_subscribe_pool(cmb_sim* s, cmb_condition* c,
                                  cmb_resourcepool* r) {
    s->E->condition_observe(dec(c), s->E->pools[dec(r)].gid);
}
void cmb_condition_unsubscribe_all(cmb_sim* s, cmb_condition* c) {
    CEngine& E = *s->E;
    for (int g = 0; g < CEngine::NGUARD; ++g)
        if (E.guards[g].observer == (int16_t)dec(c)) E.guards[g].observer = -1;
}

/* resource guard = a condition whose signal() evaluates the front waiter
 * only (the reference's guard contract).  A guard handle is either a
 * condition (standalone guard) or a TAGGED engine guard id: the guards
 * embedded in built-in objects (reference cmb_resource.h:205 /
 * cmb_resourcepool.h:263 return `&rp->guard`) are exposed the same way
 * here, as handles that signal / wait on / cancel against the object's
 * own guard queue. */
static const uintptr_t RG_TAG = (uintptr_t)1 << 30;
static int rg_gid(const cmb_sim* s, const cmb_resourceguard* g) {
    const uintptr_t v = (uintptr_t)g;
    if (v & RG_TAG) return (int)((v & ~RG_TAG) - 1);
    return s->E->conds[dec((const cmb_condition*)g)].gid;
}
cmb_resourceguard* cmb_resourceguard_create(cmb_sim* s) {
    return (cmb_resourceguard*)cmb_condition_create(s);
}
void cmb_resourceguard_initialize(cmb_sim*, cmb_resourceguard*,
                                  const char*) {}
cmb_resourceguard* cmb_resource_guard(cmb_sim* s, cmb_resource* r) {
    return (cmb_resourceguard*)(
        ((uintptr_t)(s->E->resources[dec(r)].gid + 1)) | RG_TAG);
}
cmb_resourceguard* cmb_resourcepool_guard(cmb_sim* s, cmb_resourcepool* p) {
    return (cmb_resourceguard*)(
        ((uintptr_t)(s->E->pools[dec(p)].gid + 1)) | RG_TAG);
}
bool cmb_resourceguard_signal(cmb_sim* s, cmb_resourceguard* g) {
    return s->E->guard_signal(rg_gid(s, g));
}
bool cmb_resourceguard_cancel(cmb_sim* s, cmb_resourceguard* g,
                              cmb_process* p) {
    CEngine& E = *s->E;
    auto& pr = E.procs[dec(p)];
    if (pr.await_kind != AW_GUARD || pr.gid != rg_gid(s, g)) return false;
    E.proc_interrupt(dec(p), SIG_CANCELLED);
    return true;
}
void cmb_guard_wait_setup_(cmb_sim* s, cmb_resourceguard* g, cmb_process* p,
                           cmb_resourceguard_demand_func* fn, void* ctx) {
    CEngine& E = *s->E;
    const int pidx = dec(p);
    E.globals.dem_fn[pidx] = (cmb_demand_func*)fn;
    E.globals.dem_ctx[pidx] = ctx;
    E.guard_wait(E.procs[pidx], rg_gid(s, g), DEM_USER, 0);
}

/* ---- debug dumps & reports ---- */

void cmb_event_queue_print_formatted(cmb_sim* s, FILE* out,
                                     cmb_event_print_formatter* epf) {
    // reference cmb_event_queue_print(FILE*, cmb_event_print_formatter*):
    // the formatter labels user events from (action, subject, object)
    if (!out) out = stderr;
    auto& q = s->E->evq;
    fprintf(out, "event queue @ t=%.6f: %d pending\n", s->E->now, q.n);
    for (int32_t i = 0; i < q.n; ++i) {
        const auto& e = q.at(i);
        const char* label = NULL;
        if (epf && e.kind == EV_USER) {
            const auto& u = s->E->globals.uev[(int)e.b];
            label = epf(u.fn, u.subj, u.obj);
        }
        if (label)
            fprintf(out, "  [%2d] t=%.6f handle=%u %s\n", i, e.t, e.handle,
                    label);
        else
            fprintf(out, "  [%2d] t=%.6f kind=%u a=%u handle=%u\n", i, e.t,
                    (unsigned)e.kind, (unsigned)e.a, e.handle);
    }
}
void cmb_event_queue_print(cmb_sim* s, FILE* out) {
    cmb_event_queue_print_formatted(s, out, NULL);
}

void cmb_resource_print_report(cmb_sim* s, const cmb_resource* r, FILE* out) {
    if (!out) out = stderr;
    double st[4];
    cmb_resource_stats(s, r, st);
    fprintf(out,
            "resource report @ t=%.6f: holder=%d utilization mean=%.4f "
            "sd=%.4f\n",
            s->E->now, (int)s->E->resources[dec(r)].holder, st[0], st[1]);
}

void cmb_resourcepool_print_report(cmb_sim* s, const cmb_resourcepool* r,
                                   FILE* out) {
    if (!out) out = stderr;
    const auto& pl = s->E->pools[dec(r)];
    auto copy = pl;
    copy.use_stats.add((double)copy.in_use, s->E->now - copy.t_last);
    fprintf(out,
            "pool report @ t=%.6f: capacity=%d in_use=%d mean_busy=%.4f\n",
            s->E->now, pl.capacity, pl.in_use, copy.use_stats.mean);
}

void cmb_objectqueue_report_print(cmb_sim* s, const cmb_objectqueue* q,
                                  FILE* out) {
    if (!out) out = stderr;
    double st[4];
    cmb_objectqueue_stats(s, q, st);
    fprintf(out,
            "queue report @ t=%.6f: length=%d mean=%.4f sd=%.4f max=%.0f\n",
            s->E->now, (int)s->E->queues[dec(q)].len, st[0], st[1], st[3]);
}

/* ---- RNG ---- */

uint64_t cmb_random_sfc64(cmb_sim* s) { return s->E->rng.next(); }
uint64_t cmb_random_fmix64(uint64_t x) { return fmix64(x); }
uint64_t cmb_random_curseed(const cmb_sim* s) { return s->seed; }
double cmb_random_uniform(cmb_sim* s, double lo, double hi) {
    return s->E->rng.uniform(lo, hi);
}
bool cmb_random_flip(cmb_sim* s, double p) { return s->E->rng.flip(p); }
int64_t cmb_random_bernoulli(cmb_sim* s, double p) {
    return s->E->rng.bernoulli(p);
}
double cmb_random_std_normal(cmb_sim* s) { return s->E->rng.std_normal(); }
double cmb_random_normal(cmb_sim* s, double mu, double sg) {
    return s->E->rng.normal(mu, sg);
}
double cmb_random_std_exponential(cmb_sim* s) {
    return s->E->rng.std_exponential();
}
double cmb_random_exponential(cmb_sim* s, double mean) {
    return s->E->rng.exponential(mean);
}
double cmb_random_lognormal(cmb_sim* s, double mu, double sg) {
    return s->E->rng.lognormal(mu, sg);
}
double cmb_random_logistic(cmb_sim* s, double a, double b) {
    return s->E->rng.logistic(a, b);
}
double cmb_random_cauchy(cmb_sim* s, double a, double b) {
    return s->E->rng.cauchy(a, b);
}
double cmb_random_rayleigh(cmb_sim* s, double sg) {
    return s->E->rng.rayleigh(sg);
}
double cmb_random_weibull(cmb_sim* s, double k, double l) {
    return s->E->rng.weibull(k, l);
}
double cmb_random_pareto(cmb_sim* s, double a, double b) {
    return s->E->rng.pareto(a, b);
}
double cmb_random_triangular(cmb_sim* s, double lo, double mo, double hi) {
    return s->E->rng.triangular(lo, mo, hi);
}
double cmb_random_pert(cmb_sim* s, double lo, double mo, double hi) {
    return s->E->rng.pert(lo, mo, hi);
}
double cmb_random_std_gamma(cmb_sim* s, double a) {
    return s->E->rng.std_gamma(a);
}
double cmb_random_gamma(cmb_sim* s, double k, double t) {
    return s->E->rng.gamma(k, t);
}
double cmb_random_erlang(cmb_sim* s, int64_t k, double m) {
    return s->E->rng.erlang(k, m);
}
double cmb_random_hypoexponential(cmb_sim* s, double a, double b) {
    return s->E->rng.hypoexponential(a, b);
}
double cmb_random_hyperexponential(cmb_sim* s, double p, double a, double b) {
    return s->E->rng.hyperexponential(p, a, b);
}
double cmb_random_std_beta(cmb_sim* s, double a, double b) {
    return s->E->rng.std_beta(a, b);
}
double cmb_random_beta(cmb_sim* s, double a, double b, double lo, double hi) {
    return s->E->rng.beta(a, b, lo, hi);
}
double cmb_random_chisquared(cmb_sim* s, double k) {
    return s->E->rng.chisquared(k);
}
double cmb_random_std_t_dist(cmb_sim* s, double df) {
    return s->E->rng.std_t_dist(df);
}
double cmb_random_t_dist(cmb_sim* s, double df, double a, double b) {
    return s->E->rng.t_dist(df, a, b);
}
double cmb_random_f_dist(cmb_sim* s, double a, double b) {
    return s->E->rng.f_dist(a, b);
}
int64_t cmb_random_geometric(cmb_sim* s, double p) {
    return s->E->rng.geometric(p);
}
int64_t cmb_random_poisson(cmb_sim* s, double m) { return s->E->rng.poisson(m); }
int64_t cmb_random_binomial(cmb_sim* s, int64_t n, double p) {
    return s->E->rng.binomial(n, p);
}
int64_t cmb_random_negative_binomial(cmb_sim* s, double r, double p) {
    return s->E->rng.negative_binomial(r, p);
}
int64_t cmb_random_pascal(cmb_sim* s, int64_t r, double p) {
    return s->E->rng.pascal(r, p);
}
int64_t cmb_random_discrete_uniform(cmb_sim* s, int64_t lo, int64_t hi) {
    return s->E->rng.discrete_uniform(lo, hi);
}
int64_t cmb_random_dice(cmb_sim* s, int64_t n) { return s->E->rng.dice(n); }
int64_t cmb_random_discrete_nonuniform(cmb_sim* s, const double* w,
                                       int64_t n) {
    return s->E->rng.discrete_nonuniform(w, n);
}
int64_t cmb_random_loaded_dice(cmb_sim* s, const double* w, int64_t n) {
    return s->E->rng.loaded_dice(w, n);
}
uint64_t cmb_random_splitmix64(uint64_t* state) { return splitmix64(*state); }
void cmb_random_initialize(cmb_sim* s, uint64_t seed) {
    s->E->rng.seed(seed);
    s->seed = seed;
}
struct cmb_alias_impl {
    std::vector<double> prob;
    std::vector<int32_t> alias;
    int64_t n;
};
cmb_alias* cmb_random_alias_create(const double* weights, int64_t n) {
    auto* a = new cmb_alias_impl;
    a->n = n;
    a->prob.resize((size_t)n);
    a->alias.resize((size_t)n);
    std::vector<int32_t> scratch((size_t)(2 * n));
    alias_build(weights, n, a->prob.data(), a->alias.data(), scratch.data());
    return (cmb_alias*)a;
}
void cmb_random_alias_destroy(cmb_alias* a) { delete (cmb_alias_impl*)a; }
int64_t cmb_random_alias_sample(cmb_sim* s, const cmb_alias* va) {
    const auto* a = (const cmb_alias_impl*)va;
    AliasTable t{a->prob.data(), a->alias.data(), a->n};
    return t.sample(s->E->rng);
}
int64_t cmb_random_alias_draw(cmb_sim* s, const cmb_alias* a) {
    return cmb_random_alias_sample(s, a);
}
uint64_t cmb_random_hwseed(void) {
    // host entropy (reference RDSEED asm; /dev/urandom is the portable
    // equivalent — see docs/PARITY.md)
    uint64_t x = 0;
    FILE* f = fopen("/dev/urandom", "rb");
    if (f) {
        if (fread(&x, sizeof(x), 1, f) != 1) x = 0;
        fclose(f);
    }
    if (!x) x = (uint64_t)time(nullptr) ^ 0x9E3779B97F4A7C15ULL;
    return x;
}

/* ---- dataset & timeseries (host) ---- */

cmb_dataset* cmb_dataset_create(void) { return (cmb_dataset*)new Dataset; }
void cmb_dataset_destroy(cmb_dataset* d) { delete (Dataset*)d; }
void cmb_dataset_reset(cmb_dataset* d) { ((Dataset*)d)->clear(); }
void cmb_dataset_add(cmb_dataset* d, double x) { ((Dataset*)d)->add(x); }
uint64_t cmb_dataset_count(const cmb_dataset* d) {
    return (uint64_t)((const Dataset*)d)->size();
}
double cmb_dataset_min(cmb_dataset* d) { return ((Dataset*)d)->quantile(0.0); }
double cmb_dataset_max(cmb_dataset* d) { return ((Dataset*)d)->quantile(1.0); }
double cmb_dataset_median(cmb_dataset* d) { return ((Dataset*)d)->median(); }
double cmb_dataset_quantile(cmb_dataset* d, double q) {
    return ((Dataset*)d)->quantile(q);
}
void cmb_dataset_sort(cmb_dataset* d) { ((Dataset*)d)->sort(); }
void cmb_dataset_merge(cmb_dataset* d, const cmb_dataset* o) {
    ((Dataset*)d)->merge(*(const Dataset*)o);
}
void cmb_dataset_copy(cmb_dataset* dst, const cmb_dataset* src) {
    *(Dataset*)dst = *(const Dataset*)src;
}
struct cmb_datasummary cmb_dataset_summarize(const cmb_dataset* d) {
    const DataSummary s = ((const Dataset*)d)->summarize();
    cmb_datasummary out;
    std::memcpy(&out, &s, sizeof(out));
    return out;
}
void cmb_dataset_acf(const cmb_dataset* d, double* out, int maxlag) {
    auto v = ((const Dataset*)d)->acf(maxlag);
    for (int i = 0; i < maxlag; ++i) out[i] = v[(size_t)i];
}
void cmb_dataset_pacf(const cmb_dataset* d, double* out, int maxlag) {
    auto v = ((const Dataset*)d)->pacf(maxlag);
    for (int i = 0; i < maxlag; ++i) out[i] = v[(size_t)i];
}
void cmb_dataset_histogram(cmb_dataset* d, int nbins, int64_t* out_counts) {
    auto hcounts = ((Dataset*)d)->histogram(nbins);
    for (int i = 0; i < nbins; ++i) out_counts[i] = hcounts[(size_t)i];
}
void cmb_dataset_fivenum_print(cmb_dataset* d, FILE* out) {
    if (!out) out = stdout;
    double f[5];
    ((Dataset*)d)->fivenum(f);
    fprintf(out, "min %.6g  Q1 %.6g  median %.6g  Q3 %.6g  max %.6g\n",
            f[0], f[1], f[2], f[3], f[4]);
}
void cmb_dataset_histogram_print(cmb_dataset* d, int nbins, FILE* out) {
    if (!out) out = stdout;
    auto h = ((Dataset*)d)->histogram(nbins);
    int64_t peak = 1;
    for (auto c : h) peak = c > peak ? c : peak;
    for (int i = 0; i < nbins; ++i) {
        const int bars = (int)((h[(size_t)i] * 50) / peak);
        fprintf(out, "%3d | %-50.*s %lld\n", i, bars,
                "##################################################",
                (long long)h[(size_t)i]);
    }
}
void cmb_dataset_correlogram_print(const cmb_dataset* d, int maxlag,
                                   FILE* out) {
    if (!out) out = stdout;
    auto a = ((const Dataset*)d)->acf(maxlag);
    auto p = ((const Dataset*)d)->pacf(maxlag);
    const double ci = 1.96 / sqrt((double)((const Dataset*)d)->size());
    fprintf(out, "lag      acf     pacf   (95%% CI +-%.4f)\n", ci);
    for (int k = 0; k < maxlag; ++k)
        fprintf(out, "%3d  %7.4f  %7.4f\n", k + 1, a[(size_t)k],
                p[(size_t)k]);
}
void cmb_dataset_print(const cmb_dataset* d, FILE* out) {
    if (!out) out = stdout;
    const auto& v = ((const Dataset*)d)->values();
    for (double x : v) fprintf(out, "%.9g\n", x);
}

static double ts_end_(const CTimeseries* t, double end_time) {
    if (end_time >= 0.0) return end_time;
    if (t->end >= 0.0) return t->end;
    return t->ts.times().empty() ? 0.0 : t->ts.times().back();
}
cmb_timeseries* cmb_timeseries_create(void) {
    return (cmb_timeseries*)new CTimeseries;
}
void cmb_timeseries_destroy(cmb_timeseries* t) { delete (CTimeseries*)t; }
void cmb_timeseries_reset(cmb_timeseries* t) {
    *(CTimeseries*)t = CTimeseries();
}
void cmb_timeseries_add(cmb_timeseries* t, double x, double time) {
    ((CTimeseries*)t)->ts.add(x, time);
}
uint64_t cmb_timeseries_count(const cmb_timeseries* t) {
    return (uint64_t)((const CTimeseries*)t)->ts.size();
}
void cmb_timeseries_finalize(cmb_timeseries* t, double end_time) {
    ((CTimeseries*)t)->end = end_time;
}
void cmb_timeseries_copy(cmb_timeseries* dst, const cmb_timeseries* src) {
    *(CTimeseries*)dst = *(const CTimeseries*)src;
}
struct cmb_wtdsummary cmb_timeseries_summarize(const cmb_timeseries* t,
                                               double end_time) {
    const auto* ct = (const CTimeseries*)t;
    const WtdSummary s = ct->ts.summarize(ts_end_(ct, end_time));
    cmb_wtdsummary out;
    std::memcpy(&out, &s, sizeof(out));
    return out;
}
double cmb_timeseries_median(const cmb_timeseries* t, double end_time) {
    const auto* ct = (const CTimeseries*)t;
    return ct->ts.median(ts_end_(ct, end_time));
}
static Dataset ts_values_(const CTimeseries* t) {
    Dataset d;
    for (double x : t->ts.values()) d.add(x);
    return d;
}
double cmb_timeseries_min(const cmb_timeseries* t) {
    Dataset d = ts_values_((const CTimeseries*)t);
    return d.quantile(0.0);
}
double cmb_timeseries_max(const cmb_timeseries* t) {
    Dataset d = ts_values_((const CTimeseries*)t);
    return d.quantile(1.0);
}
void cmb_timeseries_fivenum_print(const cmb_timeseries* t, FILE* out) {
    Dataset d = ts_values_((const CTimeseries*)t);
    cmb_dataset_fivenum_print((cmb_dataset*)&d, out);
}
void cmb_timeseries_histogram_print(const cmb_timeseries* t, int nbins,
                                    FILE* out) {
    Dataset d = ts_values_((const CTimeseries*)t);
    cmb_dataset_histogram_print((cmb_dataset*)&d, nbins, out);
}
void cmb_timeseries_correlogram_print(const cmb_timeseries* t, int maxlag,
                                      FILE* out) {
    // reference cmb_timeseries.h:342 (delegates to the dataset version)
    Dataset d = ts_values_((const CTimeseries*)t);
    cmb_dataset_correlogram_print((cmb_dataset*)&d, maxlag, out);
}
void cmb_timeseries_sort_x(cmb_timeseries* t) {
    auto* ct = (CTimeseries*)t;
    std::vector<std::pair<double, double>> v;
    for (size_t i = 0; i < ct->ts.size(); ++i)
        v.emplace_back(ct->ts.values()[i], ct->ts.times()[i]);
    std::sort(v.begin(), v.end());
    Timeseries fresh;
    for (auto& pr : v) fresh.add(pr.first, pr.second);
    ct->ts = fresh;
}
void cmb_timeseries_sort_t(cmb_timeseries* t) {
    auto* ct = (CTimeseries*)t;
    std::vector<std::pair<double, double>> v;
    for (size_t i = 0; i < ct->ts.size(); ++i)
        v.emplace_back(ct->ts.times()[i], ct->ts.values()[i]);
    std::sort(v.begin(), v.end());
    Timeseries fresh;
    for (auto& pr : v) fresh.add(pr.second, pr.first);
    ct->ts = fresh;
}
void cmb_timeseries_print(const cmb_timeseries* t, FILE* out) {
    if (!out) out = stdout;
    const auto* ts = &((const CTimeseries*)t)->ts;
    for (size_t i = 0; i < ts->size(); ++i)
        fprintf(out, "%.9g %.9g\n", ts->times()[i], ts->values()[i]);
}

/* ---- summaries ---- */

cmb_datasummary* cmb_datasummary_create(void) {
    auto* s = new cmb_datasummary;
    reinterpret_cast<DataSummary*>(s)->reset();
    return s;
}
void cmb_datasummary_destroy(cmb_datasummary* s) { delete s; }
cmb_wtdsummary* cmb_wtdsummary_create(void) {
    auto* s = new cmb_wtdsummary;
    reinterpret_cast<WtdSummary*>(s)->reset();
    return s;
}
void cmb_wtdsummary_destroy(cmb_wtdsummary* s) { delete s; }
void cmb_datasummary_initialize(cmb_datasummary* s) {
    reinterpret_cast<DataSummary*>(s)->reset();
}
void cmb_datasummary_add(cmb_datasummary* s, double x) {
    reinterpret_cast<DataSummary*>(s)->add(x);
}
void cmb_datasummary_merge(cmb_datasummary* s, const cmb_datasummary* o) {
    reinterpret_cast<DataSummary*>(s)->merge(
        *reinterpret_cast<const DataSummary*>(o));
}
double cmb_datasummary_count(const cmb_datasummary* s) { return s->n; }
double cmb_datasummary_mean(const cmb_datasummary* s) { return s->mean; }
double cmb_datasummary_variance(const cmb_datasummary* s) {
    return reinterpret_cast<const DataSummary*>(s)->variance();
}
double cmb_datasummary_stddev(const cmb_datasummary* s) {
    return reinterpret_cast<const DataSummary*>(s)->stddev();
}
double cmb_datasummary_skewness(const cmb_datasummary* s) {
    return reinterpret_cast<const DataSummary*>(s)->skewness();
}
double cmb_datasummary_kurtosis(const cmb_datasummary* s) {
    return reinterpret_cast<const DataSummary*>(s)->kurtosis();
}
double cmb_datasummary_minimum(const cmb_datasummary* s) { return s->mn; }
double cmb_datasummary_maximum(const cmb_datasummary* s) { return s->mx; }

void cmb_wtdsummary_initialize(cmb_wtdsummary* s) {
    reinterpret_cast<WtdSummary*>(s)->reset();
}
void cmb_wtdsummary_add(cmb_wtdsummary* s, double x, double w) {
    reinterpret_cast<WtdSummary*>(s)->add(x, w);
}
void cmb_wtdsummary_merge(cmb_wtdsummary* s, const cmb_wtdsummary* o) {
    reinterpret_cast<WtdSummary*>(s)->merge(
        *reinterpret_cast<const WtdSummary*>(o));
}
double cmb_wtdsummary_mean(const cmb_wtdsummary* s) { return s->mean; }
double cmb_wtdsummary_variance(const cmb_wtdsummary* s) {
    return reinterpret_cast<const WtdSummary*>(s)->variance();
}
double cmb_wtdsummary_stddev(const cmb_wtdsummary* s) {
    return reinterpret_cast<const WtdSummary*>(s)->stddev();
}
double cmb_wtdsummary_skewness(const cmb_wtdsummary* s) {
    return reinterpret_cast<const WtdSummary*>(s)->skewness();
}
double cmb_wtdsummary_kurtosis(const cmb_wtdsummary* s) {
    return reinterpret_cast<const WtdSummary*>(s)->kurtosis();
}
double cmb_wtdsummary_count(const cmb_wtdsummary* s) { return s->n; }
double cmb_wtdsummary_min(const cmb_wtdsummary* s) { return s->mn; }
double cmb_wtdsummary_max(const cmb_wtdsummary* s) { return s->mx; }
void cmb_wtdsummary_print(const cmb_wtdsummary* s, FILE* out) {
    if (!out) out = stdout;
    fprintf(out,
            "wtd summary: n=%.0f w=%.6g mean=%.6g sd=%.6g min=%.6g "
            "max=%.6g\n",
            s->n, s->sumw, s->mean, cmb_wtdsummary_stddev(s), s->mn, s->mx);
}
void cmb_datasummary_print(const struct cmb_datasummary* s, FILE* out) {
    if (!out) out = stdout;
    fprintf(out,
            "summary: n=%.0f mean=%.6g sd=%.6g skew=%.4f kurt=%.4f "
            "min=%.6g max=%.6g\n",
            s->n, s->mean, cmb_datasummary_stddev(s),
            cmb_datasummary_skewness(s), cmb_datasummary_kurtosis(s), s->mn,
            s->mx);
}

/* ---- logger ---- */

void cmb_logger_flags_on(uint32_t flags) { logger_flags_on(flags); }
void cmb_logger_flags_off(uint32_t flags) { logger_flags_off(flags); }
static cmb_timeformatter_func* g_timefmt = nullptr;
void cmb_logger_timeformatter_set(cmb_timeformatter_func* fmt) {
    g_timefmt = fmt;
}
static void apply_timefmt(cmb_sim* s) {
    logger_ctx().sim_time = s->E->now;
    if (g_timefmt) {
        static thread_local char buf[48];
        g_timefmt(s->E->now, buf, sizeof(buf));
        logger_ctx().who = buf;
    }
}
void cmb_logger_info(cmb_sim* s, const char* fmt, ...) {
    apply_timefmt(s);
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_INFO, "info", fmt, ap);
    va_end(ap);
}
void cmb_logger_warning(cmb_sim* s, const char* fmt, ...) {
    apply_timefmt(s);
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_WARNING, "warning", fmt, ap);
    va_end(ap);
}
void cmb_logger_error(cmb_sim* s, const char* fmt, ...) {
    apply_timefmt(s);
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_ERROR, "error", fmt, ap);
    va_end(ap);
    throw TrialAbandon{1};
}
void cmb_logger_vfprintf(cmb_sim* s, uint32_t flag, const char* fmt,
                         va_list ap) {
    apply_timefmt(s);
    logger_vlog(flag, "user", fmt, ap);
}
void cmb_logger_fatal(cmb_sim* s, const char* fmt, ...) {
    apply_timefmt(s);
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(LOG_FATAL, "fatal", fmt, ap);
    va_end(ap);
    abort();
}
void cmb_logger_user(cmb_sim* s, uint32_t flag, const char* fmt, ...) {
    apply_timefmt(s);
    va_list ap;
    va_start(ap, fmt);
    logger_vlog(flag, "user", fmt, ap);
    va_end(ap);
}
void cmb_assert_failed_(const char* expr, const char* file, int line) {
    fprintf(stderr, "cimba assertion failed: %s (%s:%d)\n", expr, file, line);
    fflush(stderr);
    abort();
}

}  // extern "C"
