// Scenario model: trace-based semantic tests of the process/guard/toolkit
// layer — the counterpart of reference test/test_process.c,
// test_resource.c, test_resourceguard.c, test_resourcepool.c,
// test_buffer.c, test_condition.c, test_priorityqueue.c, test_event.c
// (SURVEY.md §4.1).  Each scenario spawns a few processes that exercise one
// primitive and append (time, code) records to a trace; the Python tests
// assert the exact sequences.  The same model compiles for host and gfx950
// so GPU runs must produce identical traces (device-semantics parity).
#pragma once

#include "../include/cimba/engine.hpp"
#include "../include/cimba/stats.hpp"
#if !defined(__HIP_DEVICE_COMPILE__)
#include "../include/cimba/logger.hpp"
#endif

namespace cmb_models {

struct Scenario : cmb::ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = 6;
        static constexpr int MAX_EV = 32;
        static constexpr int TIMERS = 2;
        static constexpr int NUM_QUEUES = 1;
        static constexpr int QCAP = 16;
        static constexpr int NUM_RES = 1;
        static constexpr int NUM_POOLS = 1;
        static constexpr int NUM_BUFS = 1;
        static constexpr int NUM_PQ = 1;
        static constexpr int PQCAP = 16;
        static constexpr int NUM_COND = 1;
    };

    // scenario ids
    enum Which : int32_t {
        W_HOLD_ORDER = 1,
        W_INTERRUPT = 2,
        W_RES_PRIORITY = 3,
        W_RES_TIMEOUT = 4,
        W_PREEMPT = 5,
        W_POOL_PARTIAL = 6,
        W_BUFFER = 7,
        W_CONDITION = 8,
        W_STOP_WAIT = 9,
        W_PQUEUE = 10,
        W_WAIT_EVENT = 11,
        W_WAIT_EVENT_CANCEL = 12,
        W_STALE_GRANT = 13,
        W_ABANDON = 14,
        W_POOL_PREEMPT = 15,
        W_HEAP_OVERFLOW = 16,   // schedule past MAX_EV -> ST_HEAP_FULL
        W_QUEUE_OVERFLOW = 17,  // put past physical QCAP -> ST_QUEUE_FULL
        W_COND_OBSERVER = 18,   // condition observes the resource guard
        W_TIMESERIES = 19,      // device-side TimeseriesRec recording
    };

    struct Params {
        int32_t which;
    };

    struct TraceEv {
        double t;
        int32_t code;
        int32_t pad_;
    };
    struct Globals {
        TraceEv ev[96];
        int32_t n;
        int32_t aux;       // condition state variable
        uint32_t uev;      // user event handle for wait_event scenarios
        cmb::TimeseriesRec<16> qlen;  // scenario 19: device-side series
    };
    struct Frame {
        int64_t a, b;
        int32_t rem;
        double d;
    };
    struct Result {
        TraceEv ev[96];
        int32_t n;
        int32_t status;
        uint64_t events;
    };

    template <class E_>
    CMB_FORCEINLINE static void trace(E_& E, int pidx, int tag) {
        Globals& g = E.globals;
        if (g.n < 96) {
            g.ev[g.n].t = E.now;
            g.ev[g.n].code = pidx * 1000 + tag;
            ++g.n;
        }
    }

    // trace tags
    enum Tag : int32_t {
        T_START = 1,
        T_WAKE = 2,
        T_DONE = 3,
        T_ACQ = 10,
        T_REL = 11,
        T_GOT = 20,   // +value for queue/pq payloads
        T_SIG = 100,  // +(-sig): 100=SUCCESS, 101=PREEMPTED, 102=INTERRUPTED,
                      // 103=STOPPED, 104=CANCELLED, 105=TIMEOUT
        T_USER_EV = 900,
    };

    template <class E_>
    CMB_FORCEINLINE static int sigtag(E_& E, typename E_::ProcT* self) {
        const long long s = (long long)self->sig;
        return T_SIG + (s <= 0 && s >= -5 ? (int)(-s) : 50);
    }

    // ---- process bodies ---------------------------------------------------

    enum Func : uint8_t {
        F_HOLDER = 0,       // hold a.d twice, tracing
        F_INTERRUPTER,      // hold a, interrupt proc b with sig 42
        F_SLEEPER,          // hold 10, trace resulting signal
        F_RES_USER,         // acquire res0 (after initial hold d), hold a, release
        F_RES_TIMEOUT,      // timeout-armed acquire
        F_PREEMPTOR,        // hold d, preempt res0, hold a, release
        F_POOL_USER,        // acquire a units (after hold d), hold b, release
        F_BUF_PRODUCER,     // 3x { hold 1; put 3 }
        F_BUF_CONSUMER,     // get a units, trace
        F_COND_WAITER,      // wait until aux >= a
        F_COND_SETTER,      // hold 1, aux=a, signal condition
        F_PROC_WAITER,      // wait_process(b), trace signal
        F_STOPPER,          // hold d, stop proc b
        F_LONG_RES_HOLDER,  // acquire res0, hold 100
        F_PQ_PRODUCER,      // put (val 1,pri 0) (val 2,pri 5) (val 3,pri 5)
        F_PQ_CONSUMER,      // hold 1, get 3 items, trace values
        F_EVENT_WAITER,     // wait_event(globals.uev), trace signal
        F_EVENT_CANCELLER,  // hold d, cancel globals.uev
        F_Q_GETTER_TMO,     // timeout-armed queue get (stale-grant scenario)
        F_Q_GETTER,         // plain queue get, trace value
        F_Q_PUTTER,         // hold d, put value a
        F_ABANDONER,        // hold 1, then abandon the trial (host logger
                            // error path; device: Engine::fail)
        F_POOL_PREEMPTOR,   // hold d, preempt a units of pool 0, hold b,
                            // release
        F_TS_PUTTER,        // put at t=1,2,3 recording queue length
        F_TS_GETTER,        // get at t=2.5, 4 recording queue length
        F_COND_RES_FREE_WAITER,  // condition-wait until resource 0 free
        F_OVERFLOWER,       // schedule a user events (heap abort path)
        F_Q_FLOODER,        // put a objects without a consumer
    };

    template <class E_>
    CMB_FORCEINLINE static void body(E_& E, typename E_::ProcT* self) {
        const int me = E.pidx_of(self);
        Frame& f = E.frames[me];
        Globals& g = E.globals;
        switch (self->func) {
        case F_HOLDER: {
            CMB_BEGIN();
            trace(E, me, T_START);
            CMB_HOLD(f.d);
            trace(E, me, T_WAKE);
            CMB_HOLD(f.d);
            trace(E, me, T_DONE);
            CMB_END();
        }
        case F_INTERRUPTER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            E.proc_interrupt((int)f.b, 42);
            trace(E, me, T_DONE);
            CMB_END();
        }
        case F_SLEEPER: {
            CMB_BEGIN();
            trace(E, me, T_START);
            CMB_HOLD(10.0);
            trace(E, me, (int)(400 + self->sig));  // 442 when interrupted w/ 42
            CMB_END();
        }
        case F_RES_USER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            CMB_RES_ACQUIRE(0);
            trace(E, me, T_ACQ);
            CMB_HOLD((double)f.a);
            CMB_RES_RELEASE(0);
            trace(E, me, T_REL);
            CMB_END();
        }
        case F_RES_TIMEOUT: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            E.timeout_arm(*self, (double)f.a);
            CMB_RES_ACQUIRE(0);
            trace(E, me, sigtag(E, self));
            if (CMB_SIG() == cmb::SIG_SUCCESS) {
                E.timeout_disarm(*self);
                CMB_HOLD(1.0);
                CMB_RES_RELEASE(0);
            }
            CMB_END();
        }
        case F_PREEMPTOR: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            CMB_RES_PREEMPT(0);
            trace(E, me, T_ACQ);
            CMB_HOLD((double)f.a);
            CMB_RES_RELEASE(0);
            trace(E, me, T_REL);
            CMB_END();
        }
        case F_POOL_USER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            CMB_POOL_ACQUIRE(0, (int32_t)f.a, f.rem);
            trace(E, me, T_ACQ);
            CMB_HOLD((double)f.b);
            // a preemptor may have taken units mid-hold: release what we
            // still hold (reference preemption contract)
            if (CMB_SIG() != cmb::SIG_SUCCESS) trace(E, me, sigtag(E, self));
            CMB_POOL_RELEASE(0, E.pool_holding(0, me));
            trace(E, me, T_REL);
            CMB_END();
        }
        case F_TS_PUTTER: {
            CMB_BEGIN();
            for (f.b = 0; f.b < 3; ++f.b) {
                CMB_HOLD(1.0);
                CMB_QPUT(0, (uint64_t)f.b);
                g.qlen.add((double)E.queues[0].len, E.now);
            }
            CMB_END();
        }
        case F_TS_GETTER: {
            CMB_BEGIN();
            CMB_HOLD(2.5);
            CMB_QGET(0, (uint64_t*)&f.a);
            g.qlen.add((double)E.queues[0].len, E.now);
            CMB_HOLD(1.5);
            CMB_QGET(0, (uint64_t*)&f.a);
            g.qlen.add((double)E.queues[0].len, E.now);
            CMB_END();
        }
        case F_COND_RES_FREE_WAITER: {
            CMB_BEGIN();
            CMB_COND_WAIT(0, cmb::DEM_USER + 1, 0);
            trace(E, me, T_WAKE);
            CMB_END();
        }
        case F_OVERFLOWER: {
            CMB_BEGIN();
            for (f.b = 0; f.b < f.a && E.status == cmb::ST_OK; ++f.b)
                E.schedule(cmb::EV_USER, 0, 0, 0, 1e9 + (double)f.b, 0);
            CMB_END();
        }
        case F_Q_FLOODER: {
            CMB_BEGIN();
            for (f.b = 0; f.b < f.a; ++f.b) {
                CMB_QPUT(0, (uint64_t)f.b);
                if (CMB_SIG() != cmb::SIG_SUCCESS) break;
            }
            CMB_END();
        }
        case F_POOL_PREEMPTOR: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            CMB_POOL_PREEMPT(0, (int32_t)f.a);
            trace(E, me, T_ACQ);
            CMB_HOLD((double)f.b);
            CMB_POOL_RELEASE(0, (int32_t)f.a);
            trace(E, me, T_REL);
            CMB_END();
        }
        case F_BUF_PRODUCER: {
            CMB_BEGIN();
            for (f.a = 0; f.a < 3; ++f.a) {
                CMB_HOLD(1.0);
                CMB_BUF_PUT(0, 3);
                trace(E, me, T_REL);
            }
            CMB_END();
        }
        case F_BUF_CONSUMER: {
            CMB_BEGIN();
            CMB_BUF_GET(0, f.a);
            trace(E, me, T_GOT);
            CMB_END();
        }
        case F_COND_WAITER: {
            CMB_BEGIN();
            CMB_COND_WAIT(0, cmb::DEM_USER, (uint32_t)f.a);
            trace(E, me, T_WAKE);
            CMB_END();
        }
        case F_COND_SETTER: {
            CMB_BEGIN();
            CMB_HOLD(1.0);
            g.aux = (int32_t)f.a;
            E.condition_signal(0);
            CMB_HOLD(1.0);
            g.aux = (int32_t)f.b;
            E.condition_signal(0);
            CMB_END();
        }
        case F_PROC_WAITER: {
            CMB_BEGIN();
            CMB_WAIT_PROCESS((int)f.b);
            trace(E, me, sigtag(E, self));
            CMB_END();
        }
        case F_STOPPER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            E.proc_stop((int)f.b);
            trace(E, me, T_DONE);
            CMB_END();
        }
        case F_LONG_RES_HOLDER: {
            CMB_BEGIN();
            CMB_RES_ACQUIRE(0);
            trace(E, me, T_ACQ);
            CMB_HOLD(100.0);
            if (E.resources[0].holder == me) {
                CMB_RES_RELEASE(0);
                trace(E, me, T_REL);
            } else {
                // preempted mid-hold: the resource is no longer ours
                trace(E, me, sigtag(E, self));
            }
            CMB_END();
        }
        case F_PQ_PRODUCER: {
            CMB_BEGIN();
            CMB_PQPUT(0, 1, 0);
            CMB_PQPUT(0, 2, 5);
            CMB_PQPUT(0, 3, 5);
            CMB_END();
        }
        case F_PQ_CONSUMER: {
            CMB_BEGIN();
            CMB_HOLD(1.0);
            for (f.b = 0; f.b < 3; ++f.b) {
                CMB_PQGET(0, (uint64_t*)&f.a);
                trace(E, me, (int)(T_GOT + f.a));
            }
            CMB_END();
        }
        case F_EVENT_WAITER: {
            CMB_BEGIN();
            CMB_WAIT_EVENT(g.uev);
            trace(E, me, sigtag(E, self));
            CMB_END();
        }
        case F_EVENT_CANCELLER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            E.event_cancel(g.uev);
            CMB_END();
        }
        case F_Q_GETTER_TMO: {
            CMB_BEGIN();
            E.timeout_arm(*self, (double)f.a);
            CMB_QGET(0, (uint64_t*)&f.b);
            trace(E, me, sigtag(E, self));
            if (CMB_SIG() == cmb::SIG_SUCCESS) {
                E.timeout_disarm(*self);
                trace(E, me, (int)(T_GOT + f.b));
            }
            CMB_END();
        }
        case F_Q_GETTER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            CMB_QGET(0, (uint64_t*)&f.b);
            trace(E, me, (int)(T_GOT + f.b));
            CMB_END();
        }
        case F_Q_PUTTER: {
            CMB_BEGIN();
            CMB_HOLD(f.d);
            CMB_QPUT(0, (uint64_t)f.a);
            trace(E, me, T_REL);
            CMB_END();
        }
        case F_ABANDONER: {
            CMB_BEGIN();
            CMB_HOLD(1.0);
#if defined(__HIP_DEVICE_COMPILE__)
            E.fail(cmb::ST_USER_ABORT);  // device trial-abort flag
#else
            cmb::logger_error("scenario abandon at t=%g", E.now);
#endif
            CMB_END();
        }
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        body(E, &E.procs[pidx]);
    }

    template <class E_>
    CMB_FORCEINLINE static bool demand(E_& E, int /*pidx*/, uint8_t kind,
                                       uint32_t ctx) {
        if (kind == cmb::DEM_USER) return E.globals.aux >= (int32_t)ctx;
        if (kind == cmb::DEM_USER + 1) return E.resources[0].holder < 0;
        return false;
    }

    template <class E_>
    CMB_FORCEINLINE static void on_event(E_& E, const cmb::EvEntry& ev) {
        if (ev.kind == cmb::EV_USER) trace(E, (int)ev.a, T_USER_EV);
    }

    // spawn helper
    template <class E_>
    CMB_FORCEINLINE static void sp(E_& E, int pidx, uint8_t func, int pri,
                                   int64_t a, int64_t b, double d) {
        E.proc_init(pidx, func, pri);
        E.frames[pidx].a = a;
        E.frames[pidx].b = b;
        E.frames[pidx].d = d;
        E.frames[pidx].rem = 0;
        E.proc_start(pidx);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        Globals& g = E.globals;
        g.n = 0;
        g.aux = 0;
        g.uev = 0;
        const int32_t which = E.params->which;
        switch (which) {
        case W_HOLD_ORDER:
            // two holders, same times, different priorities: at each time
            // step the higher-priority proc's events run first
            sp(E, 0, F_HOLDER, 0, 0, 0, 1.5);
            sp(E, 1, F_HOLDER, 5, 0, 0, 1.5);
            break;
        case W_INTERRUPT:
            sp(E, 0, F_SLEEPER, 0, 0, 0, 0.0);
            sp(E, 1, F_INTERRUPTER, 0, 0, /*target*/ 0, 1.0);
            break;
        case W_RES_PRIORITY:
            // p0 takes the resource at t=0 for 5; p1 (pri 0) queues at t=1,
            // p2 (pri 9) queues at t=2: release at t=5 grants p2 first
            sp(E, 0, F_RES_USER, 0, /*hold*/ 5, 0, 0.0);
            sp(E, 1, F_RES_USER, 0, 1, 0, 1.0);
            sp(E, 2, F_RES_USER, 9, 1, 0, 2.0);
            break;
        case W_RES_TIMEOUT:
            sp(E, 0, F_RES_USER, 0, /*hold*/ 5, 0, 0.0);
            sp(E, 1, F_RES_TIMEOUT, 0, /*tmo*/ 2, 0, 0.5);
            break;
        case W_PREEMPT:
            sp(E, 0, F_LONG_RES_HOLDER, 0, 0, 0, 0.0);
            sp(E, 1, F_PREEMPTOR, 5, /*hold*/ 2, 0, 1.0);
            break;
        case W_POOL_PARTIAL:
            // capacity 10; p0 takes 7 at t=0 for 3; p1 wants 6 at t=1:
            // takes 3 immediately, waits, gets remaining 3 at t=3
            E.pools[0].capacity = 10;
            sp(E, 0, F_POOL_USER, 0, 7, /*hold*/ 3, 0.0);
            sp(E, 1, F_POOL_USER, 0, 6, 1, 1.0);
            break;
        case W_BUFFER:
            E.buffers[0].capacity = 100;
            sp(E, 0, F_BUF_PRODUCER, 0, 0, 0, 0.0);
            sp(E, 1, F_BUF_CONSUMER, 0, /*amount*/ 5, 0, 0.0);
            break;
        case W_CONDITION:
            sp(E, 0, F_COND_WAITER, 0, /*threshold*/ 3, 0, 0.0);
            sp(E, 1, F_COND_WAITER, 0, /*threshold*/ 7, 0, 0.0);
            sp(E, 2, F_COND_SETTER, 0, /*first*/ 5, /*second*/ 8, 0.0);
            break;
        case W_STOP_WAIT:
            sp(E, 0, F_LONG_RES_HOLDER, 0, 0, 0, 0.0);
            sp(E, 1, F_PROC_WAITER, 0, 0, /*target*/ 0, 0.0);
            sp(E, 2, F_STOPPER, 0, 0, /*target*/ 0, 1.0);
            sp(E, 3, F_RES_USER, 0, /*hold*/ 1, 0, 0.5);  // gets res after stop
            break;
        case W_PQUEUE:
            sp(E, 0, F_PQ_PRODUCER, 0, 0, 0, 0.0);
            sp(E, 1, F_PQ_CONSUMER, 0, 0, 0, 0.0);
            break;
        case W_WAIT_EVENT:
        case W_WAIT_EVENT_CANCEL:
            g.uev = E.schedule(cmb::EV_USER, /*a*/ 7, 0, 0, 5.0, 0);
            sp(E, 0, F_EVENT_WAITER, 0, 0, 0, 0.0);
            if (which == W_WAIT_EVENT_CANCEL)
                sp(E, 1, F_EVENT_CANCELLER, 0, 0, 0, 2.0);
            break;
        case W_STALE_GRANT:
            // putter arms its hold FIRST (earlier FIFO seq at t=1), then the
            // timeout getter arms its timer: at t=1 the put fires first and
            // schedules a grant for p1, then p1's timeout fires (earlier seq
            // than the grant) and removes p1 — the grant must pass on to p2
            sp(E, 0, F_Q_PUTTER, 0, /*value*/ 77, 0, 1.0);
            sp(E, 1, F_Q_GETTER_TMO, 0, /*tmo*/ 1.0, 0, 0.0);
            sp(E, 2, F_Q_GETTER, 0, 0, 0, 0.5);
            break;
        case W_ABANDON:
            sp(E, 0, F_HOLDER, 0, 0, 0, 1.5);   // a bystander process
            sp(E, 1, F_ABANDONER, 0, 0, 0, 0.0);
            break;
        case W_HEAP_OVERFLOW:
            sp(E, 0, F_OVERFLOWER, 0, /*count*/ 100, 0, 0.0);
            break;
        case W_QUEUE_OVERFLOW:
            // queue limit CMB_UNLIMITED but physical QCAP=16: the 17th
            // put must abort the trial, not deadlock
            E.queues[0].limit = cmb::CMB_UNLIMITED;
            sp(E, 0, F_Q_FLOODER, 0, /*count*/ 64, 0, 0.0);
            break;
        case W_TIMESERIES:
            E.globals.qlen.reset();
            sp(E, 0, F_TS_PUTTER, 0, 0, 0, 0.0);
            sp(E, 1, F_TS_GETTER, 0, 0, 0, 0.0);
            break;
        case W_COND_OBSERVER:
            // p1 waits on a condition whose predicate is "resource 0 is
            // free"; the condition OBSERVES the resource's guard
            // (reference observer registration,
            // include/cmb_resourceguard.h:48-52): the release signal is
            // forwarded and wakes p1 without anyone signaling the
            // condition directly.  demand id DEM_USER+1 -> aux unused.
            E.condition_observe(0, E.resources[0].gid);
            sp(E, 0, F_RES_USER, 0, /*hold*/ 2, 0, 0.0);
            sp(E, 1, F_COND_RES_FREE_WAITER, 0, 0, 0, 0.0);
            break;
        case W_POOL_PREEMPT:
            // cap 4: p0 (pri 0) holds 3 for 10; p1 (pri 5) preempts 2 at
            // t=1 (1 free + 1 reclaimed); p0's hold returns SIG_PREEMPTED
            // and it releases its remaining 2 units
            E.pools[0].capacity = 4;
            sp(E, 0, F_POOL_USER, 0, 3, /*hold*/ 10, 0.0);
            sp(E, 1, F_POOL_PREEMPTOR, 5, 2, /*hold*/ 2, 1.0);
            break;
        default:
            E.fail(cmb::ST_USER_ABORT);
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.n = E.globals.n;
        for (int i = 0; i < r.n; ++i) r.ev[i] = E.globals.ev[i];
        r.status = E.status;
        r.events = E.ev_dispatched;
        // scenario 19: summarize the device-recorded series into the
        // trace as (weighted-mean x 1000, count) pseudo-entries
        if (E.params->which == W_TIMESERIES && r.n + 1 < 96) {
            cmb::WtdSummary ws;
            ws.reset();
            E.globals.qlen.summarize(ws, 5.0);
            r.ev[r.n].t = ws.mean;
            r.ev[r.n].code = (int32_t)E.globals.qlen.len;
            ++r.n;
        }
    }
};

}  // namespace cmb_models
