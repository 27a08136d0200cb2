"""HBM spill tier: bounded fast tier + overflow slab.

The reference grows its hash-heap and queues by unbounded doubling
(/root/reference/src/cmi_hashheap.c:380-432); the MI355X engine keeps a
bounded fast tier (LDS/scratch) and claims a spill slab (HBM on device,
heap on host) on first overflow.  These tests pin:
  - FIFO order across queue spill/unspill cycles
  - heap dispatch order and cancel-by-handle across the tier boundary
  - bit-exact equivalence of a spilling config vs a large-capacity one
  - the clean abort path when no slab is available (true exhaustion)
"""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HARNESS = r"""
#include "cimba/engine.hpp"
#include "cimba/runner.hpp"

#include <cstdio>
#include <cstdlib>
#include <memory>
#include <vector>

#define CHECK(x)                                                       \
    do {                                                               \
        if (!(x)) {                                                    \
            std::fprintf(stderr, "CHECK failed %s:%d: %s\n", __FILE__, \
                         __LINE__, #x);                                \
            return 1;                                                  \
        }                                                              \
    } while (0)

// burst M/M/1-ish model, templated on Cfg so the spilling (QCAP 8 +
// SPILL_Q 120) and non-spilling (QCAP 128) configurations share the
// exact same body -> results must be bitwise equal
template <class CfgT>
struct Burst : cmb::ModelBase {
    using Cfg = CfgT;
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
    template <class E_>
    static void arrival(E_& E, typename E_::ProcT* self) {
        auto& f = E.frames[0];
        CMB_BEGIN();
        for (f.u = 0; f.u < E.params->num_objects; ++f.u) {
            CMB_HOLD(E.rng.exponential(0.05));  // rho >> 1: queue builds
            CMB_QPUT(0, cmb::double_as_u64(E.now));
            if (CMB_SIG() != cmb::SIG_SUCCESS) break;
        }
        CMB_END();
    }
    template <class E_>
    static void service(E_& E, typename E_::ProcT* self) {
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
    static void step(E_& E, int pidx) {
        if (pidx == 0)
            arrival(E, &E.procs[pidx]);
        else
            service(E, &E.procs[pidx]);
    }
    template <class E_>
    static void setup(E_& E) {
        E.globals.cnt = 0;
        E.globals.sum = 0.0;
        E.proc_init(0, 0, 0);
        E.proc_init(1, 1, 0);
        E.proc_start(0);
        E.proc_start(1);
    }
    template <class E_>
    static void finish(E_& E, Result& r) {
        r.obj_cnt = E.globals.cnt;
        r.sum_wait = E.globals.sum;
        r.events = E.ev_dispatched;
        r.status = E.status;
    }
};

struct CfgSpill {
    static constexpr int MAX_PROC = 2;
    static constexpr int MAX_EV = 8;
    static constexpr int SPILL_EV = 64;
    static constexpr int TIMERS = 1;
    static constexpr int NUM_QUEUES = 1;
    static constexpr int QCAP = 8;
    static constexpr int SPILL_Q = 120;
    static constexpr int NUM_RES = 0;
    static constexpr int NUM_POOLS = 0;
    static constexpr int NUM_BUFS = 0;
    static constexpr int NUM_PQ = 1;
    static constexpr int PQCAP = 8;
    static constexpr int SPILL_PQ = 120;
    static constexpr int NUM_COND = 0;
};
// map-enabled variant: the handle->index back-map plus the spill tier
// together (the host C API shape), for the open-addressing/backward-
// shift-deletion storm below
struct CfgMap {
    static constexpr int MAX_PROC = 2;
    static constexpr int MAX_EV = 32;
    static constexpr int SPILL_EV = 480;
    static constexpr bool EV_MAP = true;
    static constexpr int TIMERS = 1;
    static constexpr int NUM_QUEUES = 1;
    static constexpr int QCAP = 8;
    static constexpr int SPILL_Q = 120;
    static constexpr int NUM_RES = 0;
    static constexpr int NUM_POOLS = 0;
    static constexpr int NUM_BUFS = 0;
    static constexpr int NUM_PQ = 0;
    static constexpr int PQCAP = 1;
    static constexpr int NUM_COND = 0;
};

struct CfgBig {
    static constexpr int MAX_PROC = 2;
    static constexpr int MAX_EV = 72;
    static constexpr int TIMERS = 1;
    static constexpr int NUM_QUEUES = 1;
    static constexpr int QCAP = 128;
    static constexpr int NUM_RES = 0;
    static constexpr int NUM_POOLS = 0;
    static constexpr int NUM_BUFS = 0;
    static constexpr int NUM_PQ = 0;
    static constexpr int PQCAP = 1;
    static constexpr int NUM_COND = 0;
};

using MSpill = Burst<CfgSpill>;
using MBig = Burst<CfgBig>;
using ESpill = cmb::Engine<MSpill>;

// 1) FIFO across repeated spill/unspill cycles, driven directly
static int test_queue_fifo() {
    auto store = std::make_unique<ESpill::Storage>();
    auto slab = std::make_unique<ESpill::Spill>();
    ESpill E(*store);
    E.set_spill(slab.get());
    MSpill::Params P{0};
    E.init(&P, 42, 0);
    MSpill::setup(E);
    auto& p = E.procs[0];
    uint64_t next_put = 0, next_get = 0;
    // phase bursts crossing the ring boundary in both directions
    const int plan[][2] = {{60, 0}, {0, 20}, {30, 0}, {0, 70},
                           {100, 0}, {0, 100}};
    for (auto& ph : plan) {
        for (int i = 0; i < ph[0]; ++i)
            CHECK(E.q_try_put(0, p, next_put++));
        for (int i = 0; i < ph[1]; ++i) {
            uint64_t v = ~0ull;
            CHECK(E.q_try_get(0, p, &v));
            CHECK(v == next_get++);
        }
        CHECK(E.q_length(0) == (int64_t)(next_put - next_get));
        CHECK(E.status == cmb::ST_OK);
    }
    CHECK(E.q_length(0) == 0);
    // capacity: QCAP + SPILL_Q total; one more must abort
    for (int i = 0; i < 128; ++i) CHECK(E.q_try_put(0, p, i));
    CHECK(!E.q_try_put(0, p, 999));  // limit reached (no abort: limit)
    CHECK(E.status == cmb::ST_OK);
    std::puts("queue fifo across spill OK");
    return 0;
}

// 2) heap order + cancel across the tier boundary
static int test_heap_spill() {
    auto store = std::make_unique<ESpill::Storage>();
    auto slab = std::make_unique<ESpill::Spill>();
    ESpill E(*store);
    E.set_spill(slab.get());
    MSpill::Params P{0};
    E.init(&P, 7, 0);
    cmb::Rng r;
    r.seed(123);
    uint32_t handles[64];
    double times[64];
    for (int i = 0; i < 64; ++i) {
        times[i] = r.uniform(0.0, 100.0);
        handles[i] = E.schedule(cmb::EV_USER, 0, 0, (uint64_t)i, times[i], 0);
        CHECK(handles[i] != 0);
    }
    CHECK(E.status == cmb::ST_OK);
    CHECK(E.evq.n == 64);  // 8 fast + 56 spilled
    // cancel five entries that sit in the spill region (indices >= 8)
    int cancelled = 0;
    for (int i = 60; i < 64 && cancelled < 5; ++i) {
        CHECK(E.event_cancel(handles[i]));
        ++cancelled;
    }
    double last = -1.0;
    int popped = 0;
    while (!E.evq.empty()) {
        cmb::EvEntry ev = E.evq.pop();
        CHECK(ev.t >= last);
        last = ev.t;
        ++popped;
    }
    CHECK(popped == 64 - cancelled);
    std::puts("heap spill order+cancel OK");
    return 0;
}

// 3) spilling config == big config, bitwise, across seeded trials
static int test_equivalence() {
    const uint64_t ntrials = 64, objs = 100;
    std::vector<MSpill::Result> a(ntrials);
    std::vector<MBig::Result> b(ntrials);
    MSpill::Params Pa{objs};
    MBig::Params Pb{objs};
    auto ra = cmb::run_host<MSpill>(Pa, 2024, ntrials, 0, a.data());
    auto rb = cmb::run_host<MBig>(Pb, 2024, ntrials, 0, b.data());
    CHECK(ra.failed == 0);
    CHECK(rb.failed == 0);
    for (uint64_t i = 0; i < ntrials; ++i) {
        CHECK(a[i].obj_cnt == b[i].obj_cnt);
        CHECK(a[i].sum_wait == b[i].sum_wait);  // bitwise
        CHECK(a[i].events == b[i].events);
        CHECK(a[i].status == 0 && b[i].status == 0);
    }
    std::puts("spill == big-capacity bitwise OK");
    return 0;
}

// 4) true exhaustion still aborts cleanly: no slab, ring full
static int test_exhaustion_abort() {
    auto store = std::make_unique<ESpill::Storage>();
    ESpill E(*store);  // NO spill slab, NO pool
    MSpill::Params P{0};
    E.init(&P, 1, 0);
    MSpill::setup(E);
    auto& p = E.procs[0];
    for (int i = 0; i < CfgSpill::QCAP; ++i) CHECK(E.q_try_put(0, p, i));
    CHECK(!E.q_try_put(0, p, 999));
    CHECK(E.status == cmb::ST_QUEUE_FULL);  // clean abort, no crash
    std::puts("exhaustion abort OK");
    return 0;
}

// 4b) priority-queue spill: puts cross the fast/slab boundary; gets come
// back in exact (priority desc, FIFO) order
static int test_pq_spill() {
    auto store = std::make_unique<ESpill::Storage>();
    auto slab = std::make_unique<ESpill::Spill>();
    ESpill E(*store);
    E.set_spill(slab.get());
    MSpill::Params P{0};
    E.init(&P, 5, 0);
    MSpill::setup(E);
    auto& p = E.procs[0];
    cmb::Rng r;
    r.seed(77);
    // 100 entries (8 fast + 92 spilled), random priorities
    int pri[100];
    for (int i = 0; i < 100; ++i) {
        pri[i] = (int)(r.next() % 7);
        CHECK(E.pq_try_put(0, p, ((uint64_t)pri[i] << 32) | (uint64_t)i,
                           pri[i]));
    }
    CHECK(E.status == cmb::ST_OK);
    int last_pri = 1 << 30;
    uint64_t last_seq_at_pri = 0;
    for (int i = 0; i < 100; ++i) {
        uint64_t v = 0;
        CHECK(E.pq_try_get(0, p, &v));
        const int vp = (int)(v >> 32);
        const uint64_t vi = v & 0xFFFFFFFFull;
        CHECK(vp <= last_pri);
        if (vp == last_pri) CHECK(vi > last_seq_at_pri);  // FIFO in class
        last_pri = vp;
        last_seq_at_pri = vi;
    }
    CHECK(E.pqueues[0].len == 0);  // fully drained
    std::puts("pq spill order OK");
    return 0;
}

// 5) back-map storm: schedule / random-cancel / reschedule across the
// tier boundary with the map ON; a scan oracle checks every lookup
static int test_backmap_storm() {
    using MMap = Burst<CfgMap>;
    using EMap = cmb::Engine<MMap>;
    auto store = std::make_unique<EMap::Storage>();
    auto slab = std::make_unique<EMap::Spill>();
    EMap E(*store);
    E.set_spill(slab.get());
    MMap::Params P{0};
    E.init(&P, 99, 0);
    cmb::Rng r;
    r.seed(4242);
    uint32_t h[512] = {0};
    int live = 0;
    for (int round = 0; round < 4000; ++round) {
        const uint64_t op = r.next() % 100;
        if (op < 55 && live < 512) {  // schedule
            for (int i = 0; i < 512; ++i)
                if (!h[i]) {
                    h[i] = E.schedule(cmb::EV_USER, 0, 0, (uint64_t)i,
                                      r.uniform(0.0, 1e6), 0);
                    CHECK(h[i] != 0);
                    ++live;
                    break;
                }
        } else if (op < 80 && live > 0) {  // cancel a random live handle
            int k = (int)(r.next() % 512);
            while (!h[k]) k = (k + 1) % 512;
            CHECK(E.event_cancel(h[k]));
            CHECK(!E.event_cancel(h[k]));  // double-cancel must miss
            h[k] = 0;
            --live;
        } else if (live > 0) {  // reschedule
            int k = (int)(r.next() % 512);
            while (!h[k]) k = (k + 1) % 512;
            CHECK(E.event_reschedule(h[k], r.uniform(0.0, 1e6), 0));
        }
        CHECK(E.evq.n == live);
        // oracle: the map agrees with a full scan for one random handle
        if (live > 0) {
            int k = (int)(r.next() % 512);
            while (!h[k]) k = (k + 1) % 512;
            const int32_t idx = E.evq.find_index(h[k]);
            CHECK(idx >= 0 && E.evq.at(idx).handle == h[k]);
        }
    }
    // drain in nondecreasing time order
    double last = -1.0;
    while (!E.evq.empty()) {
        cmb::EvEntry ev = E.evq.pop();
        CHECK(ev.t >= last);
        last = ev.t;
    }
    std::puts("backmap storm OK");
    return 0;
}

int main() {
    if (test_queue_fifo()) return 1;
    if (test_heap_spill()) return 1;
    if (test_equivalence()) return 1;
    if (test_exhaustion_abort()) return 1;
    if (test_pq_spill()) return 1;
    if (test_backmap_storm()) return 1;
    std::puts("spill suite OK");
    return 0;
}
"""


@pytest.fixture(scope="module")
def harness(tmp_path_factory):
    d = tmp_path_factory.mktemp("spill")
    src = d / "spill.cpp"
    src.write_text(HARNESS)
    exe = str(d / "spill")
    r = subprocess.run(
        ["g++", "-std=c++17", "-O2", "-g",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe, "-lpthread"], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    return exe


def test_spill_suite(harness):
    r = subprocess.run([harness], capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
    assert "spill suite OK" in r.stdout


def test_mg1_spill_config_available():
    """MG1 now declares a spill tier (SPILL_Q) — heavy-tail service at
    SCV=4 must run without aborts on the host path."""
    import cimba_amd as ca

    r = ca.mg1_host(ntrials=32, num_objects=4000, arr_rate=0.8,
                    srv_mean=1.0, srv_scv=4.0, dist=2, seed=11, threads=0)
    assert r["trials_ok"] == 32, r


def test_spillprobe_host_path():
    """The spill-probe model on the host engine: every trial crosses the
    8-entry fast tier into the slab hundreds of times."""
    import cimba_amd as ca

    r = ca._C.spillprobe_run(ntrials=32, num_objects=150, seed=11, gpu=False)
    assert r["trials_ok"] == 32, r
    assert r["total_events"] > 32 * 250
