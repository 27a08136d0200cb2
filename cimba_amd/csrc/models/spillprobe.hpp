// Spill-probe model: a deliberately tiny fast tier (8-entry heap, 8-slot
// queue ring) over a large spill declaration, so EVERY trial crosses the
// LDS/scratch -> HBM slab boundary many times.  Exists to exercise the
// device spill machinery (lazy slab claim atomics, two-tier heap/queue
// traversal, pool-exhaustion abort) deterministically in CI — the
// production models (MG1's 32K queue) overflow their fast tier only once
// per ~5e5 trials, which proves the no-abort property but would let a
// broken slab path hide.  Host run of the same model is the bitwise
// reference (tests/test_gpu.py::test_device_spill_probe).
#pragma once

#include "../include/cimba/engine.hpp"

namespace cmb_models {

struct SpillProbe : cmb::ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = 2;
        static constexpr int MAX_EV = 8;
        static constexpr int SPILL_EV = 56;
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 1;
        static constexpr int QCAP = 8;
        static constexpr int SPILL_Q = 504;  // bursts reach ~150 deep
        static constexpr int NUM_RES = 0;
        static constexpr int NUM_POOLS = 0;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };

    struct Params {
        uint64_t num_objects;
    };

    struct Result {
        uint64_t obj_cnt;
        double sum_wait;
        uint64_t events;
        int32_t status;
        int32_t pad_;
    };

    struct Frame {
        uint64_t u;
    };
    struct Globals {
        uint64_t cnt;
        double sum;
    };

    // overloaded producer (rho >> 1) so the queue builds deep into the
    // spill segment, then drains back through the ring repeatedly
    template <class E_>
    CMB_FORCEINLINE static void arrival(E_& E, typename E_::ProcT* self) {
        auto& f = E.frames[0];
        CMB_BEGIN();
        for (f.u = 0; f.u < E.params->num_objects; ++f.u) {
            CMB_HOLD(E.rng.exponential(0.05));
            CMB_QPUT(0, cmb::double_as_u64(E.now));
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;
        }
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void service(E_& E, typename E_::ProcT* self) {
        auto& f = E.frames[1];
        CMB_BEGIN();
        for (;;) {
            CMB_QGET(0, &f.u);
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;
            CMB_HOLD(E.rng.exponential(1.0));
            E.globals.sum += E.now - cmb::u64_as_double(f.u);
            E.globals.cnt += 1u;
        }
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        if (pidx == 0)
            arrival(E, &E.procs[pidx]);
        else
            service(E, &E.procs[pidx]);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        E.globals.cnt = 0;
        E.globals.sum = 0.0;
        E.proc_init(0, 0, 0);
        E.proc_init(1, 1, 0);
        E.proc_start(0);
        E.proc_start(1);
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.obj_cnt = E.globals.cnt;
        r.sum_wait = E.globals.sum;
        r.events = E.ev_dispatched;
        r.status = E.status;
    }
};

}  // namespace cmb_models
