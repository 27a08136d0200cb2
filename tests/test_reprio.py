"""Live reprioritization of a waiting process (reference
cmb_process_priority_set while queued on a resourceguard): the guard must
grant in the NEW priority order.  Runs the same model on BOTH guard
implementations — the <=64-process bitmask path (packed grank word,
refreshed by proc_priority_set) and the >64-process intrusive-list path
(live comparator) — and requires the identical grant order."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HARNESS = r"""
#include "cimba/engine.hpp"
#include "cimba/runner.hpp"

#include <cstdio>
#include <memory>

#define CHECK(x)                                                       \
    do {                                                               \
        if (!(x)) {                                                    \
            std::fprintf(stderr, "CHECK failed %s:%d: %s\n", __FILE__, \
                         __LINE__, #x);                                \
            return 1;                                                  \
        }                                                              \
    } while (0)

// MAXP selects the guard implementation: <=64 -> waiter bitmask + packed
// rank word; >64 -> intrusive list + live field comparator
template <int MAXP>
struct Reprio : cmb::ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = MAXP;
        static constexpr int MAX_EV = 64;
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 0;
        static constexpr int QCAP = 1;
        static constexpr int NUM_RES = 1;
        static constexpr int NUM_POOLS = 0;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };
    struct Params {};
    struct Result {
        int order[4];
        int n;
        int32_t status;
    };
    struct Frame {};
    struct Globals {
        int order[4];
        int n;
    };

    enum Func : uint8_t { F_HOLDER = 0, F_WAITER = 1, F_BOSS = 2 };

    // holds the resource for 10 time units, then releases
    template <class E_>
    CMB_FORCEINLINE static void holder(E_& E, typename E_::ProcT* self) {
        CMB_BEGIN();
        CMB_RES_ACQUIRE(0);
        CMB_HOLD(10.0);
        CMB_RES_RELEASE(0);
        CMB_END();
    }

    // waiters 1..3 queue up at t=1 with priorities 1, 2, 3; record the
    // order in which they win the resource
    template <class E_>
    CMB_FORCEINLINE static void waiter(E_& E, typename E_::ProcT* self) {
        CMB_BEGIN();
        CMB_HOLD(1.0);
        CMB_RES_ACQUIRE(0);
        E.globals.order[E.globals.n++] = E.pidx_of(self);
        CMB_HOLD(1.0);
        CMB_RES_RELEASE(0);
        CMB_END();
    }

    // at t=5 (while all three wait), the boss inverts the priorities:
    // waiter 1 (pri 1) -> 9, waiter 3 (pri 3) -> 0
    template <class E_>
    CMB_FORCEINLINE static void boss(E_& E, typename E_::ProcT* self) {
        CMB_BEGIN();
        CMB_HOLD(5.0);
        E.proc_priority_set(1, 9);
        E.proc_priority_set(3, 0);
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        auto* self = &E.procs[pidx];
        if (self->func == F_HOLDER)
            holder(E, self);
        else if (self->func == F_WAITER)
            waiter(E, self);
        else
            boss(E, self);
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        E.globals.n = 0;
        E.proc_init(0, F_HOLDER, 0);
        for (int w = 1; w <= 3; ++w) E.proc_init(w, F_WAITER, w);
        E.proc_init(4, F_BOSS, 0);
        for (int i = 0; i <= 4; ++i) E.proc_start(i);
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.n = E.globals.n;
        for (int i = 0; i < 4; ++i)
            r.order[i] = i < r.n ? E.globals.order[i] : -1;
        r.status = E.status;
    }
};

template <int MAXP>
static int run_one(const char* tag) {
    using M = Reprio<MAXP>;
    auto store = std::make_unique<typename cmb::Engine<M>::Storage>();
    cmb::Engine<M> E(*store);
    typename M::Params P{};
    E.init(&P, 1, 0);
    M::setup(E);
    E.run(1.0e308, 100000);
    typename M::Result r;
    M::finish(E, r);
    CHECK(r.status == cmb::ST_OK);
    CHECK(r.n == 3);
    // after the t=5 inversion: waiter 1 has pri 9 (highest), waiter 2
    // keeps 2, waiter 3 drops to 0 -> grant order 1, 2, 3
    CHECK(r.order[0] == 1);
    CHECK(r.order[1] == 2);
    CHECK(r.order[2] == 3);
    std::printf("reprio %s OK (order %d %d %d)\n", tag, r.order[0],
                r.order[1], r.order[2]);
    return 0;
}

int main() {
    if (run_one<8>("mask/grank")) return 1;     // bitmask + packed rank
    if (run_one<80>("list/fields")) return 1;   // intrusive list path
    std::puts("reprio suite OK");
    return 0;
}
"""


@pytest.fixture(scope="module")
def harness(tmp_path_factory):
    d = tmp_path_factory.mktemp("reprio")
    src = d / "reprio.cpp"
    src.write_text(HARNESS)
    exe = str(d / "reprio")
    r = subprocess.run(
        ["g++", "-std=c++17", "-O2", "-g",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe, "-lpthread"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    return exe


def test_reprio_while_waiting(harness):
    r = subprocess.run([harness], capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
    assert "reprio suite OK" in r.stdout
