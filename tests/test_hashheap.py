"""Randomized model-check of the future-event list — counterpart of the
reference's test/test_hashheap.c stress test.  Compiles a C++ harness
that drives HashHeap with random push/pop/cancel/reschedule/pattern ops
and cross-checks every result against a naive O(n^2) linear-scan model
(an independent algorithm, so agreement pins both ordering and the
remove_at/sift repair paths)."""
import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

HARNESS = r"""
#include "cimba/hashheap.hpp"
#include "cimba/rng.hpp"

#include <cstdio>
#include <cstdlib>
#include <vector>

using cmb::EvEntry;
using cmb::HashHeap;
using cmb::ev_pseq;

// independent reference: unsorted vector + linear scan for the minimum
struct Naive {
    std::vector<EvEntry> v;
    int min_idx() const {
        int m = 0;
        for (size_t i = 1; i < v.size(); ++i)
            if (cmb::ev_less(v[i], v[m])) m = (int)i;
        return m;
    }
};

#define CHECK(c)                                                     \
    do {                                                             \
        if (!(c)) {                                                  \
            std::fprintf(stderr, "FAIL %s:%d op=%lu\n", __FILE__,    \
                         __LINE__, (unsigned long)op);               \
            return 1;                                                \
        }                                                            \
    } while (0)

int main(int argc, char** argv) {
    const uint64_t seed = argc > 1 ? strtoull(argv[1], nullptr, 0) : 1;
    const uint64_t nops = argc > 2 ? strtoull(argv[2], nullptr, 0) : 20000;
#ifdef CMB_HEAP_TIERED
    // two-tier + back-map instantiation (the host C API shape, scaled
    // down): an 8-entry fast tier under a 504-entry spill slab, so the
    // fuzz crosses the boundary constantly, with the open-addressing
    // handle map live on every op
    constexpr int CAP = 8;
    constexpr int SCAP = 504;
    static EvEntry buf[CAP];
    static EvEntry slab[SCAP];
    static cmb::EvMapSlot mapbuf[HashHeap<CAP, SCAP, true>::MSIZE];
    HashHeap<CAP, SCAP, true> h(buf);
    h.map = mapbuf;
    h.map_clear();
    h.attach_spill(slab, SCAP);
#else
    constexpr int CAP = 512;
    static EvEntry buf[CAP];
    HashHeap<CAP> h(buf);
#endif
    Naive m;
    cmb::Rng r;
    r.seed(seed);
    uint32_t next_handle = 1;
    uint64_t seq = 0;
    for (uint64_t op = 0; op < nops; ++op) {
        const uint32_t k = (uint32_t)r.below(100);
        if (k < 45) {  // push
            if (h.full()) continue;
            EvEntry ev{};
            ev.t = (double)r.below(64);  // small domain -> many ties
            const int pri = (int)r.below(5) - 2;
            ev.pseq = ev_pseq(pri, seq++);
            ev.handle = next_handle++;
            ev.kind = (uint16_t)(16 + r.below(3));
            ev.a = (uint16_t)r.below(4);
            ev.b = r.below(3);
            CHECK(h.push(ev));
            m.v.push_back(ev);
        } else if (k < 80) {  // pop
            CHECK(h.empty() == m.v.empty());
            if (h.empty()) continue;
            const EvEntry got = h.pop();
            const int mi = m.min_idx();
            // the comparator is a total order over (t, pseq) and every
            // pseq is unique, so min is unique -> exact handle match
            CHECK(got.handle == m.v[(size_t)mi].handle);
            CHECK(got.t == m.v[(size_t)mi].t);
            m.v.erase(m.v.begin() + mi);
        } else if (k < 90) {  // cancel a (maybe-live) handle
            const uint32_t victim = 1 + (uint32_t)r.below(next_handle);
            bool live = false;
            for (size_t i = 0; i < m.v.size(); ++i)
                if (m.v[i].handle == victim) {
                    live = true;
                    m.v.erase(m.v.begin() + i);
                    break;
                }
            CHECK(h.cancel(victim) == live);
        } else if (k < 96) {  // reschedule a (maybe-live) handle
            const uint32_t victim = 1 + (uint32_t)r.below(next_handle);
            const double nt = (double)r.below(64);
            const uint64_t np = ev_pseq((int)r.below(5) - 2, seq++);
            bool live = false;
            for (auto& ev : m.v)
                if (ev.handle == victim) {
                    live = true;
                    ev.t = nt;
                    ev.pseq = np;
                    break;
                }
            CHECK(h.reschedule(victim, nt, np) == live);
        } else {  // pattern count / cancel over (kind, a, b) wildcards
            const uint16_t pk = r.below(2) ? (uint16_t)(16 + r.below(3))
                                           : (uint16_t)0xFFFF;
            const uint16_t pa = r.below(2) ? (uint16_t)r.below(4)
                                           : (uint16_t)0xFFFF;
            const bool mb = r.below(2) != 0;
            const uint64_t pb = r.below(3);
            auto match = [&](const EvEntry& ev) {
                return (pk == 0xFFFF || ev.kind == pk) &&
                       (pa == 0xFFFF || ev.a == pa) && (!mb || ev.b == pb);
            };
            int32_t want = 0;
            for (const auto& ev : m.v) want += match(ev) ? 1 : 0;
            CHECK(h.pattern_count(pk, pa, mb, pb) == want);
            if (r.below(2)) {
                CHECK(h.pattern_cancel(pk, pa, mb, pb) == want);
                for (size_t i = 0; i < m.v.size();)
                    if (match(m.v[i]))
                        m.v.erase(m.v.begin() + i);
                    else
                        ++i;
            }
        }
        CHECK((size_t)h.n == m.v.size());
    }
    // drain: full ordering check
    uint64_t op = nops;
    while (!h.empty()) {
        const EvEntry got = h.pop();
        const int mi = m.min_idx();
        CHECK(got.handle == m.v[(size_t)mi].handle);
        m.v.erase(m.v.begin() + mi);
    }
    CHECK(m.v.empty());
    std::puts("heap fuzz OK");
    return 0;
}
"""


def _compile(d, extra):
    src = d / f"heapfuzz{len(extra)}.cpp"
    src.write_text(HARNESS)
    exe = str(d / f"heapfuzz{len(extra)}")
    r = subprocess.run(
        ["g++", "-std=c++17", "-O2", "-g", *extra,
         "-I", os.path.join(ROOT, "cimba_amd", "csrc", "include"),
         str(src), "-o", exe], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    return exe


@pytest.fixture(scope="module")
def harness(tmp_path_factory):
    return _compile(tmp_path_factory.mktemp("heapfuzz"), [])


@pytest.fixture(scope="module")
def harness_tiered(tmp_path_factory):
    return _compile(tmp_path_factory.mktemp("heapfuzzt"),
                    ["-DCMB_HEAP_TIERED"])


@pytest.mark.parametrize("seed", [1, 2, 0xDEADBEEF, 12345, 777])
def test_heap_fuzz(harness, seed):
    r = subprocess.run([harness, str(seed), "20000"], capture_output=True,
                       text=True, timeout=300)
    assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
    assert "heap fuzz OK" in r.stdout


@pytest.mark.parametrize("seed", [1, 2, 0xDEADBEEF, 12345, 777])
def test_heap_fuzz_tiered_with_backmap(harness_tiered, seed):
    """Same model-check over the two-tier + handle-map instantiation:
    every push/pop/cancel/reschedule/pattern op crosses or probes the
    fast/slab boundary and the open-addressing map simultaneously."""
    r = subprocess.run([harness_tiered, str(seed), "20000"],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (r.stdout, r.stderr[-2000:])
    assert "heap fuzz OK" in r.stdout
