"""Randomized model-check of the interaction-toolkit containers —
counterpart of the reference's test_objectqueue.c / test_priorityqueue.c /
test_buffer.c / test_resourcepool.c unit suites, generalized over random
op sequences.  A C++ harness drives the engine's non-blocking try-ops
directly (the blocking macros are these try-ops plus guard waits, so the
state transitions exercised here ARE the ones the blocking paths commit)
and cross-checks every transition against naive independent models:
queue vs deque, priority queue vs stable sort, buffer vs level counter,
pool vs per-proc holdings table."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HARNESS = r"""
#include "cimba/engine.hpp"

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <deque>
#include <vector>

struct FuzzModel : cmb::ModelBase {
    struct Cfg {
        static constexpr int MAX_PROC = 4;
        static constexpr int MAX_EV = 16;
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 1;
        static constexpr int QCAP = 32;
        static constexpr int NUM_RES = 1;
        static constexpr int NUM_POOLS = 1;
        static constexpr int NUM_BUFS = 1;
        static constexpr int NUM_PQ = 1;
        static constexpr int PQCAP = 32;
        static constexpr int NUM_COND = 1;
    };
    struct Params {};
    struct Result {};
    struct Frame {};
    template <class E_>
    static void step(E_&, int) {}
    template <class E_>
    static void setup(E_&) {}
    template <class E_>
    static void finish(E_&, Result&) {}
};

using E = cmb::Engine<FuzzModel>;

#define CHECK(c)                                                  \
    do {                                                          \
        if (!(c)) {                                               \
            std::fprintf(stderr, "FAIL %s:%d op=%lu\n", __FILE__, \
                         __LINE__, (unsigned long)op);            \
            return 1;                                             \
        }                                                         \
    } while (0)

int main(int argc, char** argv) {
    const uint64_t seed = argc > 1 ? strtoull(argv[1], nullptr, 0) : 1;
    const uint64_t nops = argc > 2 ? strtoull(argv[2], nullptr, 0) : 30000;
    static E::Storage st;
    E eng(st);
    FuzzModel::Params P;
    eng.init(&P, seed, 0);
    // three procs (distinct pool holders); none ever blocks, so every
    // guard stays empty and the no-queue-jump gate is always open
    for (int i = 0; i < 3; ++i) {
        const int pidx = eng.proc_alloc();
        eng.proc_init(pidx, 0, i - 1);  // priorities -1/0/1: preempt order
    }
    eng.queues[0].limit = 24;  // below QCAP: exercises the limit path
    eng.pqueues[0].limit = 24;
    eng.pools[0].capacity = 10;
    eng.buffers[0].capacity = 50;

    // independent models
    std::deque<uint64_t> mq;
    struct PqEnt { int pri; uint64_t seq, val; };
    std::vector<PqEnt> mpq;
    uint64_t mseq = 0;
    int64_t mlevel = 0;
    int32_t mheld[3] = {0, 0, 0};

    cmb::Rng r;
    r.seed(seed ^ 0xABCD);
    uint64_t next_val = 1;
    for (uint64_t op = 0; op < nops; ++op) {
        const int who = (int)r.below(3);
        auto& p = eng.procs[who];
        switch (r.below(11)) {
        case 0: {  // queue put
            const uint64_t v = next_val++;
            const bool ok = eng.q_try_put(0, p, v);
            CHECK(ok == (mq.size() < 24));
            if (ok) mq.push_back(v);
            break;
        }
        case 1: {  // queue get: strict FIFO
            uint64_t v = 0;
            const bool ok = eng.q_try_get(0, p, &v);
            CHECK(ok == !mq.empty());
            if (ok) {
                CHECK(v == mq.front());
                mq.pop_front();
            }
            break;
        }
        case 2: {  // pq put
            const uint64_t v = next_val++;
            const int pri = (int)r.below(4) - 1;
            const bool ok = eng.pq_try_put(0, p, v, pri);
            CHECK(ok == (mpq.size() < 24));
            if (ok) mpq.push_back({pri, mseq++, v});
            break;
        }
        case 3: {  // pq get: highest priority, FIFO among equals
            uint64_t v = 0;
            const bool ok = eng.pq_try_get(0, p, &v);
            CHECK(ok == !mpq.empty());
            if (ok) {
                size_t best = 0;
                for (size_t i = 1; i < mpq.size(); ++i)
                    if (mpq[i].pri > mpq[best].pri ||
                        (mpq[i].pri == mpq[best].pri &&
                         mpq[i].seq < mpq[best].seq))
                        best = i;
                CHECK(v == mpq[best].val);
                mpq.erase(mpq.begin() + (long)best);
            }
            break;
        }
        case 4: {  // buffer put
            const int64_t amt = 1 + (int64_t)r.below(8);
            const bool ok = eng.buf_try_put(0, p, amt);
            CHECK(ok == (50 - mlevel >= amt));
            if (ok) mlevel += amt;
            break;
        }
        case 5: {  // buffer get
            const int64_t amt = 1 + (int64_t)r.below(8);
            const bool ok = eng.buf_try_get(0, p, amt);
            CHECK(ok == (mlevel >= amt));
            if (ok) mlevel -= amt;
            break;
        }
        case 6: {  // pool greedy take
            const int32_t want = 1 + (int32_t)r.below(5);
            const int32_t avail = 10 - (mheld[0] + mheld[1] + mheld[2]);
            const int32_t got = eng.pool_try_take(0, p, want);
            CHECK(got == (avail < want ? avail : want));
            mheld[who] += got;
            break;
        }
        case 7: {  // pool all-or-nothing take
            const int32_t want = 1 + (int32_t)r.below(5);
            const int32_t avail = 10 - (mheld[0] + mheld[1] + mheld[2]);
            const bool ok = eng.pool_try_take_all(0, p, want);
            CHECK(ok == (avail >= want));
            if (ok) mheld[who] += want;
            break;
        }
        case 8: {  // pool partial release
            if (mheld[who] == 0) break;
            const int32_t amt = 1 + (int32_t)r.below((uint32_t)mheld[who]);
            eng.pool_release(0, p, amt);
            mheld[who] -= amt;
            break;
        }
        case 9: {  // pool preempt: all-or-nothing from lower-pri holders
            const int32_t want = 1 + (int32_t)r.below(5);
            const int32_t in_use = mheld[0] + mheld[1] + mheld[2];
            const int32_t free_units = 10 - in_use;
            int32_t reclaimable = 0;
            for (int i = 0; i < 3; ++i)
                if (i != who && (i - 1) < (who - 1))  // strictly lower pri
                    reclaimable += mheld[i];
            const bool expect = free_units + reclaimable >= want;
            const bool ok = eng.pool_try_preempt(0, p, want);
            CHECK(ok == expect);
            if (ok) {
                int32_t need = want - free_units;
                while (need > 0) {  // lowest priority first, then index
                    int victim = -1;
                    for (int i = 0; i < 3; ++i) {
                        if (i == who || mheld[i] <= 0) continue;
                        if ((i - 1) >= (who - 1)) continue;
                        if (victim < 0 || (i - 1) < (victim - 1)) victim = i;
                    }
                    const int32_t take =
                        mheld[victim] < need ? mheld[victim] : need;
                    mheld[victim] -= take;
                    need -= take;
                }
                mheld[who] += want;
            }
            break;
        }
        default: {  // invariants snapshot
            CHECK(eng.q_length(0) == (int64_t)mq.size());
            CHECK(eng.pqueues[0].len == (int32_t)mpq.size());
            CHECK(eng.buffers[0].level == mlevel);
            CHECK(eng.pools[0].in_use == mheld[0] + mheld[1] + mheld[2]);
            for (int i = 0; i < 3; ++i)
                CHECK(eng.pool_holding(0, i) == mheld[i]);
            break;
        }
        }
        CHECK(eng.status == cmb::ST_OK);
    }
    // drain the queue and pq fully: total order check
    uint64_t op = nops;
    {
        auto& p = eng.procs[0];
        uint64_t v;
        while (!mq.empty()) {
            CHECK(eng.q_try_get(0, p, &v) && v == mq.front());
            mq.pop_front();
        }
        CHECK(!eng.q_try_get(0, p, &v));
        std::stable_sort(mpq.begin(), mpq.end(),
                         [](const PqEnt& a, const PqEnt& b) {
                             return a.pri != b.pri ? a.pri > b.pri
                                                   : a.seq < b.seq;
                         });
        for (const auto& ent : mpq)
            CHECK(eng.pq_try_get(0, p, &v) && v == ent.val);
        CHECK(!eng.pq_try_get(0, p, &v));
    }
    std::puts("toolkit fuzz OK");
    return 0;
}
"""


@pytest.fixture(scope="module")
def harness(tmp_path_factory):
    d = tmp_path_factory.mktemp("tkfuzz")
    src = d / "tkfuzz.cpp"
    src.write_text(HARNESS)
    exe = str(d / "tkfuzz")
    r = subprocess.run(
        ["g++", "-std=c++17", "-O2", "-g",
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src),
         os.path.join(ROOT, "cimba_amd", "csrc", "host", "support.cpp"),
         "-o", exe], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    return exe


@pytest.mark.parametrize("seed", [1, 42, 0xC0FFEE, 2026])
def test_toolkit_fuzz(harness, seed):
    r = subprocess.run([harness, str(seed), "30000"], capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
    assert "toolkit fuzz OK" in r.stdout
