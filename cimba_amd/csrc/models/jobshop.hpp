// Closed job-shop / resource-pool network — BASELINE.json config 4
// ("Job-shop / resource-pool network (cmb_resourcepool + cmb_objectqueue),
// 10k entities per replication").
//
// NJOBS permanent job processes cycle through 3 machine stations
// (resource pools with capacities 3/2/4); each job class (pidx % 3) visits
// the stations in a different order with class-specific service means.
// A replication completes when TOTAL_ENTITIES jobs have finished their
// route.  Exercises pool contention, partial acquisition and the guard
// priority ordering under load; throughput/utilization sanity is checked
// host-side (tests/test_mg1_jobshop.py).
#pragma once

#include "../include/cimba/engine.hpp"

namespace cmb_models {

struct JobShop : cmb::ModelBase {
    static constexpr int NST = 3;  // stations

    struct Cfg {
        static constexpr int MAX_PROC = 24;  // permanent jobs
        static constexpr int MAX_EV = 64;
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 1;   // finished-entity log queue
        static constexpr int QCAP = 32;
        static constexpr int NUM_RES = 0;
        static constexpr int NUM_POOLS = NST;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };

    struct Params {
        uint64_t total_entities;  // completed routes per replication
        double think_mean;        // inter-cycle think time
        double srv_mean[NST];     // per-station mean service times
        int32_t njobs;            // active job processes (<= MAX_PROC)
        int32_t pad_;
    };

    struct Result {
        uint64_t completed;
        double makespan;          // sim time at completion
        double busy_time[NST];    // total unit-busy time per station
        uint64_t events;
        int32_t status;
        int32_t pad_;
    };

    struct Frame {
        int32_t leg;     // current leg of the route (0..NST-1)
        int32_t rem;     // pool partial-acquire scratch
        int32_t units;   // units requested at current station
    };

    struct Globals {
        uint64_t completed;
    };

    static constexpr int CAPACITY[NST] = {3, 2, 4};
    // route[class][leg] = station
    CMB_FORCEINLINE static int route(int cls, int leg) {
        // class 0: 0-1-2, class 1: 1-2-0, class 2: 2-0-1
        return (cls + leg) % NST;
    }

    template <class E_>
    CMB_FORCEINLINE static void body(E_& E, typename E_::ProcT* self) {
        const Params& P = *E.params;
        const int me = E.pidx_of(self);
        Frame& f = E.frames[me];
        CMB_BEGIN();
        while (E.globals.completed < P.total_entities) {
            CMB_HOLD(E.rng.exponential(P.think_mean));
            for (f.leg = 0; f.leg < NST; ++f.leg) {
                f.units = 1 + (int32_t)(me % 2);  // some jobs need 2 units
                // station expression is re-evaluated inside the macro's
                // retry loop, so no local may be declared across the yield.
                // all-or-nothing acquire: greedy partial acquisition would
                // deadlock here (partial holders can exhaust a station)
                CMB_POOL_ACQUIRE_ALL(route(me % NST, f.leg), f.units);
                if (CMB_SIG() != cmb::SIG_SUCCESS) break;
                CMB_HOLD(E.rng.exponential(P.srv_mean[route(me % NST, f.leg)]));
                CMB_POOL_RELEASE(route(me % NST, f.leg), f.units);
            }
            E.globals.completed += 1u;
        }
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        body(E, &E.procs[pidx]);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        const Params& P = *E.params;
        E.globals.completed = 0;
        for (int s = 0; s < NST; ++s) {
            E.pools[s].capacity = CAPACITY[s];
            E.pools[s].recording = 1;  // time-weighted units-in-use stats
        }
        const int nj = P.njobs < Cfg::MAX_PROC ? P.njobs : Cfg::MAX_PROC;
        for (int j = 0; j < nj; ++j) {
            E.proc_init(j, 0, /*priority*/ j % 3);
            E.frames[j].leg = 0;
            E.frames[j].rem = 0;
            E.proc_start(j);
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.completed = E.globals.completed;
        r.makespan = E.now;
        for (int s = 0; s < NST; ++s) {
            // close the recording interval and export mean units-in-use x T
            E.pools[s].use_stats.add((double)E.pools[s].in_use,
                                     E.now - E.pools[s].t_last);
            r.busy_time[s] = E.pools[s].use_stats.mean *
                             E.pools[s].use_stats.sumw;
        }
        r.events = E.ev_dispatched;
        r.status = E.status;
    }
};

}  // namespace cmb_models
