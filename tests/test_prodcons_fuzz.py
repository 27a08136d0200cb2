"""Randomized blocking-path stress — producers/consumers with random
holds, random queue choice, priorities, and timeout-armed gets, run for
hundreds of independently-seeded trials.  This exercises the guard
grant / stale-grant / revalidate protocol (reference
cmb_resourceguard.c:163-228 semantics) under schedules the 19 hand
scenarios cannot enumerate, with conservation + per-producer FIFO as
the oracle: every token put is consumed exactly once, and tokens from
one producer come out of each queue in increasing sequence order."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HARNESS = r"""
#include "cimba/engine.hpp"
#include "cimba/runner.hpp"

#include <cstdio>
#include <cstdlib>

// 3 producers scatter K tokens each over 2 bounded queues (QCAP 8 ->
// constant two-sided contention); 2 consumers per queue drain them with
// timeout-armed gets.  The trial ends when the event queue empties.
struct ProdCons : cmb::ModelBase {
    static constexpr int NPROD = 3, NCONS = 4, K = 200;
    struct Cfg {
        static constexpr int MAX_PROC = 8;
        static constexpr int MAX_EV = 64;
        static constexpr int TIMERS = 2;
        static constexpr int NUM_QUEUES = 2;
#ifdef CMB_FUZZ_SPILL
        // spill variant: a 4-slot ring under the same logical limit (8),
        // so every full-queue episode crosses the slab boundary while
        // the two-sided contention the fuzz targets is unchanged —
        // results must be BITWISE identical to the plain build
        static constexpr int QCAP = 4;
        static constexpr int SPILL_Q = 16;
#else
        static constexpr int QCAP = 8;
#endif
        static constexpr int NUM_RES = 0;
        static constexpr int NUM_POOLS = 0;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };
    struct Params {};
    struct Result {
        uint32_t consumed;
        uint32_t timeouts;
        uint8_t order_ok, drained, status_ok, pad_;
    };
    struct PFrame {
        uint32_t i;
        uint32_t q;
    };
    struct CFrame {
        uint64_t obj;
        uint32_t q;
    };
    union Frame {
        PFrame p;
        CFrame c;
    };
    struct Globals {
        uint32_t consumed;
        uint32_t timeouts;
        uint32_t producers_done;
        uint32_t last_seq[NPROD][2];  // 1 + last consumed seq, per queue
        uint8_t order_ok;
    };

    template <class E_>
    CMB_FORCEINLINE static void producer(E_& E, typename E_::ProcT* self,
                                         int me) {
        PFrame& f = E.frames[me].p;
        CMB_BEGIN();
        for (f.i = 0; f.i < K; ++f.i) {
            CMB_HOLD(E.rng.exponential(1.0));
            f.q = (uint32_t)E.rng.below(2);
            CMB_QPUT((int)f.q, ((uint64_t)me << 32) | f.i);
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;  // not expected
        }
        E.globals.producers_done++;
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void consumer(E_& E, typename E_::ProcT* self,
                                         int me) {
        CFrame& f = E.frames[me].c;
        f.q = (uint32_t)(me - NPROD) & 1u;  // two consumers per queue
        CMB_BEGIN();
        for (;;) {
            E.timeout_arm(*self, 3.0 + E.rng.u01());
            CMB_QGET((int)E.frames[me].c.q, &E.frames[me].c.obj);
            if (CMB_SIG() == cmb::SIG_TIMEOUT) {
                E.globals.timeouts++;
                if (E.globals.producers_done == NPROD &&
                    E.q_length((int)E.frames[me].c.q) == 0)
                    break;     // drained: exit instead of re-arming forever
                continue;      // spurious timeout under contention: retry
            }
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;
            E.timeout_disarm(*self);
            {
                const uint32_t prod = (uint32_t)(E.frames[me].c.obj >> 32);
                const uint32_t seq = (uint32_t)E.frames[me].c.obj;
                uint32_t& last = E.globals.last_seq[prod][E.frames[me].c.q];
                if (seq + 1 <= last) E.globals.order_ok = 0;  // FIFO broken
                last = seq + 1;
                E.globals.consumed++;
            }
            CMB_HOLD(E.rng.exponential(0.5));
        }
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        auto* self = &E.procs[pidx];
        if (pidx < NPROD)
            producer(E, self, pidx);
        else
            consumer(E, self, pidx);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        E.globals = Globals{};
        E.globals.order_ok = 1;
        E.queues[0].limit = 8;  // logical capacity (ring + spill tiers)
        E.queues[1].limit = 8;
        for (int i = 0; i < NPROD + NCONS; ++i) {
            const int pidx = E.proc_alloc();
            // random priorities shuffle the guard grant order per trial
            E.proc_init(pidx, 0, (int)E.rng.below(5) - 2);
            E.proc_start(pidx);
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.consumed = E.globals.consumed;
        r.timeouts = E.globals.timeouts;
        r.order_ok = E.globals.order_ok;
        r.drained = (E.q_length(0) == 0 && E.q_length(1) == 0) ? 1 : 0;
        r.status_ok = (E.status == cmb::ST_OK) ? 1 : 0;
    }
};

int main(int argc, char** argv) {
    const uint64_t seed = argc > 1 ? strtoull(argv[1], nullptr, 0) : 1;
    const uint64_t ntrials = argc > 2 ? strtoull(argv[2], nullptr, 0) : 400;
    std::vector<ProdCons::Result> out(ntrials);
    ProdCons::Params P;
    const auto rep = cmb::run_host<ProdCons>(P, seed, ntrials, 0, out.data());
    if (rep.failed != 0) {
        std::fprintf(stderr, "FAIL: %llu trials failed\n",
                     (unsigned long long)rep.failed);
        return 1;
    }
    uint64_t timeouts = 0;
    for (uint64_t t = 0; t < ntrials; ++t) {
        const auto& r = out[t];
        if (r.consumed != ProdCons::NPROD * ProdCons::K || !r.order_ok ||
            !r.drained || !r.status_ok) {
            std::fprintf(stderr,
                         "FAIL trial %llu: consumed=%u order=%u drained=%u "
                         "status=%u\n",
                         (unsigned long long)t, r.consumed, r.order_ok,
                         r.drained, r.status_ok);
            return 1;
        }
        timeouts += r.timeouts;
    }
    std::printf("prodcons fuzz OK: %llu trials, %llu timeouts exercised\n",
                (unsigned long long)ntrials, (unsigned long long)timeouts);
    return 0;
}
"""


def _compile(d, extra):
    src = d / f"pcfuzz{len(extra)}.cpp"
    src.write_text(HARNESS)
    exe = str(d / f"pcfuzz{len(extra)}")
    r = subprocess.run(
        ["g++", "-std=c++17", "-O2", "-g", *extra,
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe, "-lpthread"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    return exe


@pytest.fixture(scope="module")
def harness(tmp_path_factory):
    return _compile(tmp_path_factory.mktemp("pcfuzz"), [])


@pytest.fixture(scope="module")
def harness_spill(tmp_path_factory):
    return _compile(tmp_path_factory.mktemp("pcfuzzsp"),
                    ["-DCMB_FUZZ_SPILL"])


@pytest.mark.parametrize("seed", [1, 99, 0xFEED])
def test_prodcons_fuzz(harness, seed):
    r = subprocess.run([harness, str(seed), "400"], capture_output=True,
                       text=True, timeout=600)
    assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
    assert "prodcons fuzz OK" in r.stdout
    # the timeout path must actually fire across the batch
    timeouts = int(r.stdout.split("trials,")[1].split()[0])
    assert timeouts > 0


@pytest.mark.parametrize("seed", [1, 99, 0xFEED])
def test_prodcons_fuzz_spill_bitwise(harness, harness_spill, seed):
    """The spill build (4-slot ring + slab under the same logical limit)
    must reproduce the plain build's randomized schedules BITWISE:
    physical placement of queue entries may never change semantics."""
    a = subprocess.run([harness, str(seed), "400"], capture_output=True,
                       text=True, timeout=600)
    b = subprocess.run([harness_spill, str(seed), "400"],
                       capture_output=True, text=True, timeout=600)
    assert a.returncode == 0 and b.returncode == 0, (a.stderr, b.stderr)
    assert a.stdout == b.stdout
