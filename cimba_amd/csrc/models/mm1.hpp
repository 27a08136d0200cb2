// M/M/1 queue benchmark model — the reference's headline benchmark
// (reference benchmark/MM1_single.c / MM1_multi.c: arrival + service
// processes, one unlimited object queue, exponential interarrival/service,
// stop after num_objects objects; avg system time expected 1/(mu-lambda)).
//
// Written in the cimba_amd protothread style: each process body is a
// resumable state machine; the object is the arrival timestamp (a double in
// the queue's 64-bit payload), replacing the reference's mempool-allocated
// void* object (MM1_multi.c:56-63).
#pragma once

#include "../include/cimba/engine.hpp"

namespace cmb_models {

struct MM1 : cmb::ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = 2;
        static constexpr int MAX_EV = 16;
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 1;
        static constexpr int QCAP = 512;  // P(Q > 512) at rho=0.9 ~ 4e-24
        static constexpr int NUM_RES = 0;
        static constexpr int NUM_POOLS = 0;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };

    struct Params {
        double arr_mean;       // 1 / arrival rate
        double srv_mean;       // 1 / service rate
        uint64_t num_objects;  // objects generated per trial
    };

    struct Result {
        uint64_t obj_cnt;
        double sum_wait;
        uint64_t events;
        int32_t status;
        int32_t pad_;
    };

    struct ArrFrame {
        uint64_t i;
    };
    struct SrvFrame {
        uint64_t obj;  // bit-cast arrival time
    };
    // per-process persistent locals; frames[0] = arrival, frames[1] = service
    union Frame {
        ArrFrame arr;
        SrvFrame srv;
    };
    // model-wide per-trial accumulators
    struct Globals {
        uint64_t cnt;
        double sum;
    };

    enum Func : uint8_t { F_ARRIVAL = 0, F_SERVICE = 1 };

    // reference MM1_multi.c:56-69 arrivalfunc
    template <class E_>
    CMB_FORCEINLINE static void arrival(E_& E, typename E_::ProcT* self) {
        const Params& P = *E.params;
        ArrFrame& f = E.frames[0].arr;
        CMB_BEGIN();
        for (f.i = 0; f.i < P.num_objects; ++f.i) {
            CMB_HOLD(E.rng.exponential(P.arr_mean));
            CMB_QPUT(0, cmb::double_as_u64(E.now));
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;
        }
        CMB_END();
    }

    // reference MM1_multi.c:71-88 servicefunc; accumulators are in acc_of(E)
    template <class E_>
    CMB_FORCEINLINE static void service(E_& E, typename E_::ProcT* self) {
        const Params& P = *E.params;
        SrvFrame& f = E.frames[1].srv;
        CMB_BEGIN();
        for (;;) {
            CMB_QGET(0, &f.obj);
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;
            CMB_HOLD(E.rng.exponential(P.srv_mean));
            acc_of(E).sum += E.now - cmb::u64_as_double(f.obj);
            acc_of(E).cnt += 1u;
        }
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static Globals& acc_of(E_& E) {
        return E.globals;
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        auto* self = &E.procs[pidx];
        if (self->func == F_ARRIVAL)
            arrival(E, self);
        else
            service(E, self);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        E.queues[0].limit = cmb::CMB_UNLIMITED;
        acc_of(E).cnt = 0;
        acc_of(E).sum = 0.0;
        E.proc_init(0, F_ARRIVAL, 0);
        E.proc_init(1, F_SERVICE, 0);
        E.proc_start(0);
        E.proc_start(1);
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.obj_cnt = acc_of(E).cnt;
        r.sum_wait = acc_of(E).sum;
        r.events = E.ev_dispatched;
        r.status = E.status;
    }
};

}  // namespace cmb_models
