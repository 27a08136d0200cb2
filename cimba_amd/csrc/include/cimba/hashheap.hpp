// cimba_amd future-event list: fixed-capacity binary min-heap keyed by
// (time asc, priority desc, FIFO seq asc), with O(1)-amortized cancel by
// handle through a slot back-map.
//
// Counterpart of reference src/cmi_hashheap.c (binary min-heap + open
// addressing hash map in one allocation, pattern find/count/cancel,
// cmi_hashheap.h:335-357).  Redesigned for gfx950: the reference needs a
// hash map because its 64-bit handles are opaque; here the heap lives in
// LDS with a bounded capacity, so a handle is (slot-generation, pool index)
// and cancellation resolves by a direct scan over a small heap — no hash
// map, no tombstones, no growth path in the hot loop.  Host-side large
// models use the same structure with a bigger CAP (heap-allocated engine).
#pragma once

#include "config.hpp"

namespace cmb {

// Event payload carried inline in the heap (reference keeps 4 pointers;
// we keep {kind, a, c, b} — kind = action id, a = subject (proc index or
// similar), c = auxiliary u32 (guard id / timer slot), b = 64-bit payload
// (signal value, object word)).
struct EvEntry {
    // NOTE: a "peek-hot head first" reorder ({t, kind, a, handle} in the
    // leading 16 B) was measured 2-4% SLOWER across all three lane
    // models: with the heap-top register cache a peek costs no memory at
    // all, and splitting the (t, pseq) comparator pair across chunks
    // taxes every sift compare instead.  Comparator-pair-first stays.
    double t;        // activation time
    uint64_t pseq;   // (INT16_MAX - priority) << 48 | seq  → min == next
    uint64_t b;      // payload
    uint32_t handle; // unique id for cancel/reschedule
    uint16_t kind;
    uint16_t a;
    uint32_t c;
    uint32_t pad_;
};

CMB_FORCEINLINE uint64_t ev_pseq(int priority, uint64_t seq) {
    const uint16_t k = (uint16_t)(32767 - priority);
    return ((uint64_t)k << 48) | (seq & UINT64_C(0xFFFFFFFFFFFF));
}

// lexicographic (t, pseq): earlier time first; at equal time higher
// priority first, then FIFO (same comparator contract as reference
// cmb_event.c:78-103)
CMB_FORCEINLINE bool ev_less(const EvEntry& x, const EvEntry& y) {
    if (x.t != y.t) return x.t < y.t;
    return x.pseq < y.pseq;
}

// The heap is a VIEW: `e` references a caller-owned entry array (LDS or
// HBM) while the size `n` lives in the view itself — on the device the
// view is part of the register-resident Engine context, so the hot-loop
// size checks and the count never round-trip through LDS.
//
// SCAP > 0 enables a SECOND TIER: entries [CAP, CAP+cap2) live in a
// spill buffer (HBM slab on the device, heap memory on the host) that
// the engine attaches on first overflow — the MI355X counterpart of the
// reference's unbounded doubling growth (cmi_hashheap.c:380-432).  The
// heap is index-contiguous across the tiers, so deep (cold) entries
// land in the slow tier naturally while the hot top stays fast.  With
// SCAP == 0 every access compiles to the plain fast-tier load — zero
// cost for models that never spill.
// MAP enables the handle->index BACK-MAP (reference cmi_hashheap.c
// open-addressing contract, :89-120): cancel/reschedule/find become an
// O(1) hash lookup + O(log n) sift instead of an O(n) scan.  Meant for
// the host C API configuration (MAX_EV 16384+); device models keep
// MAP=false and the O(small-n) scan, which wins at LDS heap sizes.
// Open addressing, linear probing, backward-shift deletion, fixed <=50%
// load (map is 2x total heap capacity, power of two).
struct EvMapSlot {
    uint32_t key;  // event handle; 0 = empty (handles start at 1)
    int32_t idx;   // heap index
};

constexpr int32_t cmb_pow2_atleast(int32_t v) {
    int32_t p = 2;
    while (p < v) p <<= 1;
    return p;
}

template <int CAP, int SCAP = 0, bool MAP = false>
struct HashHeap {
    static constexpr int32_t MSIZE =
        MAP ? cmb_pow2_atleast(2 * (CAP + SCAP)) : 2;

    EvEntry (&e)[CAP];
    EvEntry* e2;     // spill tier base (null until attached)
    EvMapSlot* map;  // back-map slots (storage-owned when MAP)
    int32_t cap2;    // usable spill entries (0 until attached)
    int32_t n;
    // register-cached copy of e[0] (valid while n > 0): the heap top is
    // read by every peek and every pop — the hottest dependent load in
    // the dispatch chain — and every mutation flows through place(), so
    // the cache is maintained by construction
    EvEntry top_c;

    CMB_FORCEINLINE explicit HashHeap(EvEntry (&buf)[CAP])
        : e(buf), e2(nullptr), map(nullptr), cap2(0), n(0) {}

    // ---- back-map primitives (compiled out when !MAP) ----
    static CMB_FORCEINLINE uint32_t mhash(uint32_t h) {
        if constexpr (!MAP) return 0;
        return (h * 2654435769u) >> (32 - __builtin_ctz((uint32_t)MSIZE));
    }
    CMB_FORCEINLINE void map_clear() {
        if constexpr (MAP) {
            for (int32_t i = 0; i < MSIZE; ++i) map[i].key = 0;
        }
    }
    CMB_FORCEINLINE void map_set(uint32_t key, int32_t idx) {
        if constexpr (MAP) {
            uint32_t i = mhash(key);
            for (;;) {
                if (map[i].key == key || map[i].key == 0) {
                    map[i].key = key;
                    map[i].idx = idx;
                    return;
                }
                i = (i + 1) & (MSIZE - 1);
            }
        }
    }
    CMB_FORCEINLINE int32_t map_get(uint32_t key) const {
        if constexpr (!MAP) return -1;
        uint32_t i = mhash(key);
        for (;;) {
            if (map[i].key == key) return map[i].idx;
            if (map[i].key == 0) return -1;
            i = (i + 1) & (MSIZE - 1);
        }
    }
    CMB_FORCEINLINE void map_erase(uint32_t key) {
        if constexpr (MAP) {
            uint32_t i = mhash(key);
            for (;;) {
                if (map[i].key == key) break;
                if (map[i].key == 0) return;
                i = (i + 1) & (MSIZE - 1);
            }
            // backward-shift deletion keeps probe chains intact with no
            // tombstones, so no compaction pass is ever needed
            uint32_t j = i;
            for (;;) {
                map[i].key = 0;
                for (;;) {
                    j = (j + 1) & (MSIZE - 1);
                    if (map[j].key == 0) return;
                    const uint32_t h = mhash(map[j].key);
                    if (i <= j ? (h <= i || h > j) : (h <= i && h > j))
                        break;
                }
                map[i] = map[j];
                i = j;
            }
        }
    }

    // placement: every heap move goes through here so the map AND the
    // top cache track it
    CMB_FORCEINLINE void place(int32_t i, const EvEntry& ev) {
        at(i) = ev;
        if (i == 0) top_c = ev;
        map_set(ev.handle, i);
    }

    CMB_FORCEINLINE EvEntry& at(int32_t i) {
        if constexpr (SCAP > 0) {
            return i < CAP ? e[i] : e2[i - CAP];
        } else {
            return e[i];
        }
    }
    CMB_FORCEINLINE const EvEntry& at(int32_t i) const {
        if constexpr (SCAP > 0) {
            return i < CAP ? e[i] : e2[i - CAP];
        } else {
            return e[i];
        }
    }

    CMB_FORCEINLINE void attach_spill(EvEntry* buf, int32_t cap) {
        e2 = buf;
        cap2 = cap < SCAP ? cap : SCAP;
    }

    CMB_FORCEINLINE int32_t capacity() const {
        if constexpr (SCAP > 0) return CAP + cap2;
        else return CAP;
    }

    CMB_FORCEINLINE void reset() {
        n = 0;
        if (map) map_clear();
    }
    CMB_FORCEINLINE bool empty() const { return n == 0; }
    CMB_FORCEINLINE bool full() const { return n == capacity(); }
    CMB_FORCEINLINE const EvEntry& top() const { return top_c; }

    CMB_FORCEINLINE void sift_up(int32_t i) {
        EvEntry tmp = at(i);
        while (i > 0) {
            const int32_t p = (i - 1) >> 1;
            if (!ev_less(tmp, at(p))) break;
            place(i, at(p));
            i = p;
        }
        place(i, tmp);
    }

    CMB_FORCEINLINE void sift_down(int32_t i) {
        EvEntry tmp = at(i);
        for (;;) {
            int32_t c = 2 * i + 1;
            if (c >= n) break;
            if (c + 1 < n && ev_less(at(c + 1), at(c))) ++c;
            if (!ev_less(at(c), tmp)) break;
            place(i, at(c));
            i = c;
        }
        place(i, tmp);
    }

    // returns false when full (caller attaches spill or aborts the trial)
    CMB_FORCEINLINE bool push(const EvEntry& ev) {
        if (n == capacity()) return false;
        at(n) = ev;
        sift_up(n);
        ++n;
        return true;
    }

    CMB_FORCEINLINE EvEntry pop() {
        EvEntry out = top_c;
        map_erase(out.handle);
        --n;
        if (n > 0) {
            place(0, at(n));
            sift_down(0);
        }
        return out;
    }

    // find heap index by handle: O(1) with the back-map, O(n) scan else
    CMB_FORCEINLINE int32_t find_index(uint32_t handle) const {
        if constexpr (MAP) {
            const int32_t i = map_get(handle);
            return (i >= 0 && i < n) ? i : -1;
        }
        for (int32_t i = 0; i < n; ++i)
            if (at(i).handle == handle) return i;
        return -1;
    }

    CMB_FORCEINLINE bool cancel(uint32_t handle, EvEntry* out = nullptr) {
        const int32_t i = find_index(handle);
        if (i < 0) return false;
        if (out) *out = at(i);
        remove_at(i);
        return true;
    }

    CMB_FORCEINLINE void remove_at(int32_t i) {
        map_erase(at(i).handle);
        --n;
        if (i == n) return;
        place(i, at(n));
        sift_down(i);
        sift_up(i);
    }

    // reschedule (reference cmb_event_reschedule): new time, keep payload
    CMB_FORCEINLINE bool reschedule(uint32_t handle, double t, uint64_t pseq) {
        const int32_t i = find_index(handle);
        if (i < 0) return false;
        at(i).t = t;
        at(i).pseq = pseq;
        sift_down(i);
        sift_up(i);
        return true;
    }

    // wildcard pattern ops over (kind, a, b) — reference
    // cmb_event_pattern_find/count/cancel (cmb_event.c:537-580) with
    // CMB_ANY_* wildcards.  kind == 0xFFFF / a == 0xFFFF / b-match disabled
    // via match_b=false act as wildcards.
    CMB_FORCEINLINE int32_t pattern_count(uint16_t kind, uint16_t a, bool match_b,
                                          uint64_t b) const {
        int32_t cnt = 0;
        for (int32_t i = 0; i < n; ++i) {
            if ((kind == 0xFFFF || at(i).kind == kind) &&
                (a == 0xFFFF || at(i).a == a) && (!match_b || at(i).b == b))
                ++cnt;
        }
        return cnt;
    }

    CMB_FORCEINLINE int32_t pattern_cancel(uint16_t kind, uint16_t a, bool match_b,
                                           uint64_t b) {
        // restart the scan after every removal: remove_at(i) refills slot i
        // with the last entry, and sift_up can carry that NOT-yet-examined
        // entry above i where a single ascending scan would never revisit
        // it (caught by the tests/test_hashheap.py fuzzer)
        int32_t cnt = 0;
        for (;;) {
            int32_t hit = -1;
            for (int32_t i = 0; i < n; ++i) {
                if ((kind == 0xFFFF || at(i).kind == kind) &&
                    (a == 0xFFFF || at(i).a == a) &&
                    (!match_b || at(i).b == b)) {
                    hit = i;
                    break;
                }
            }
            if (hit < 0) return cnt;
            remove_at(hit);
            ++cnt;
        }
    }

    CMB_FORCEINLINE uint32_t pattern_find(uint16_t kind, uint16_t a, bool match_b,
                                          uint64_t b) const {
        for (int32_t i = 0; i < n; ++i) {
            if ((kind == 0xFFFF || at(i).kind == kind) &&
                (a == 0xFFFF || at(i).a == a) && (!match_b || at(i).b == b))
                return at(i).handle;
        }
        return 0;  // 0 = no handle
    }
};

}  // namespace cmb
