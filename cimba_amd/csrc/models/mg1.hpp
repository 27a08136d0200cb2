// M/G/1 queue with a cmb-resource server and selectable service-time
// distribution — BASELINE.json config 3 ("M/G/1 with cmb_resource +
// ziggurat-normal service times") and the counterpart of the reference's
// integration experiment (test/test_cimba.c: M/G/1, 4 variability levels x
// 5 utilizations x 10 replications).
//
// Validation: Pollaczek-Khinchine — E[Wq] = lambda*E[S^2] / (2(1-rho)),
// E[T] = E[Wq] + E[S]; tests/test_mg1.py checks the simulated mean system
// time against it for each service distribution.
#pragma once

#include "../include/cimba/engine.hpp"

namespace cmb_models {

struct MG1 : cmb::ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = 2;
        static constexpr int MAX_EV = 16;
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 1;
        static constexpr int QCAP = 1024;   // heavier tails than M/M/1
        static constexpr int SPILL_Q = 31744;  // HBM spill: 32K total, so
        // no heavy-tail workload aborts (r01 envelope: ~1 abort / 5e5
        // trials at lognormal SCV=4, rho=0.8 — now served by the slab)
        static constexpr int NUM_RES = 1;
        static constexpr int NUM_POOLS = 0;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };

    enum Dist : int32_t {
        D_EXPONENTIAL = 0,  // M/M/1 sanity point
        D_GAMMA = 1,        // SCV via shape = 1/scv
        D_LOGNORMAL = 2,    // SCV via sigma^2 = ln(1+scv)
        D_NORMAL = 3,       // ziggurat normal, cv = sqrt(scv), clipped at 0
    };

    struct Params {
        double arr_mean;   // mean interarrival time (1/lambda)
        double srv_mean;   // mean service time
        double srv_scv;    // squared coefficient of variation of service
        uint64_t num_objects;
        int32_t dist;
        int32_t pad_;
    };

    struct Result {
        uint64_t obj_cnt;
        double sum_system;  // total time in system
        double sum_queue;   // total waiting-for-server time
        uint64_t events;
        int32_t status;
        int32_t pad_;
    };

    union Frame {
        struct {
            uint64_t i;
        } arr;
        struct {
            uint64_t obj;
            double t_start_srv;
        } srv;
    };

    struct Globals {
        uint64_t cnt;
        double sum_system;
        double sum_queue;
    };

    enum Func : uint8_t { F_ARRIVAL = 0, F_SERVER = 1 };

    template <class E_>
    CMB_FORCEINLINE static double service_time(E_& E) {
        const Params& P = *E.params;
        switch (P.dist) {
            case D_GAMMA:
                return E.rng.gamma(1.0 / P.srv_scv, P.srv_mean * P.srv_scv);
            case D_LOGNORMAL: {
                const double s2 = log(1.0 + P.srv_scv);
                return E.rng.lognormal(log(P.srv_mean) - 0.5 * s2, sqrt(s2));
            }
            case D_NORMAL: {
                const double sd = P.srv_mean * sqrt(P.srv_scv);
                const double x = E.rng.normal(P.srv_mean, sd);
                return x > 0.0 ? x : 0.0;
            }
            default:
                return E.rng.exponential(P.srv_mean);
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void body(E_& E, typename E_::ProcT* self) {
        const Params& P = *E.params;
        const int me = E.pidx_of(self);
        if (self->func == F_ARRIVAL) {
            auto& f = E.frames[me].arr;
            CMB_BEGIN();
            for (f.i = 0; f.i < P.num_objects; ++f.i) {
                CMB_HOLD(E.rng.exponential(P.arr_mean));
                CMB_QPUT(0, cmb::double_as_u64(E.now));
                if (CMB_SIG() != cmb::SIG_SUCCESS) break;
            }
            CMB_END();
        } else {
            auto& f = E.frames[me].srv;
            CMB_BEGIN();
            for (;;) {
                CMB_QGET(0, &f.obj);
                if (CMB_SIG() != cmb::SIG_SUCCESS) break;
                CMB_RES_ACQUIRE(0);
                if (CMB_SIG() != cmb::SIG_SUCCESS) break;
                E.globals.sum_queue += E.now - cmb::u64_as_double(f.obj);
                CMB_HOLD(service_time(E));
                CMB_RES_RELEASE(0);
                E.globals.sum_system += E.now - cmb::u64_as_double(f.obj);
                E.globals.cnt += 1u;
            }
            CMB_END();
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        body(E, &E.procs[pidx]);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        E.queues[0].limit = cmb::CMB_UNLIMITED;
        E.globals.cnt = 0;
        E.globals.sum_system = 0.0;
        E.globals.sum_queue = 0.0;
        E.proc_init(0, F_ARRIVAL, 0);
        E.proc_init(1, F_SERVER, 0);
        E.proc_start(0);
        E.proc_start(1);
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.obj_cnt = E.globals.cnt;
        r.sum_system = E.globals.sum_system;
        r.sum_queue = E.globals.sum_queue;
        r.events = E.ev_dispatched;
        r.status = E.status;
    }
};

}  // namespace cmb_models
