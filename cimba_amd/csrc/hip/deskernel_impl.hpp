// Shared template implementation for the per-model DES kernel TUs
// (deskernel.hip = M/M/1 + device utils, deskernel_mg1.hip,
// deskernel_jobshop.hip, deskernel_scenario.hip).  Split so hipcc
// compiles the models in PARALLEL: each instantiation inlines the whole
// engine at -O3, and one combined TU took ~8 min.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

#include "../include/cimba/runner.hpp"

namespace cmb_dk {

using cmb::Engine;

#define HIP_TRY(x)                                    \
    do {                                              \
        hipError_t err_ = (x);                        \
        if (err_ != hipSuccess) return (int)err_;     \
    } while (0)



// generic trial-per-wavefront kernel; WPB = waves (= trials) per workgroup;
// MINW = requested waves/SIMD (caps the register allocator; occupancy knob)
template <class Model, int WPB, int MINW = 1>
__global__ __launch_bounds__(WPB * 64, MINW) __attribute__((flatten)) void trial_kernel(
    typename Model::Params P, uint64_t master_seed, uint64_t trial_base,
    uint32_t ntrials, double until, uint64_t max_events,
    typename Model::Result* __restrict__ out,
    typename Engine<Model>::Spill* __restrict__ sp_arena,
    int32_t* __restrict__ sp_cur, int32_t sp_cap) {
    // arrays in LDS; the Engine context (clock, seq, handles, status, RNG
    // state, heap size) is a per-lane local -> register-resident
    __shared__ typename Engine<Model>::Storage st[WPB];
    const int w = (int)(threadIdx.x >> 6);
    if ((threadIdx.x & 63) != 0) return;  // lane 0 of each wave drives
    Engine<Model> E(st[w]);
    E.set_spill_pool(sp_arena, sp_cur, sp_cap);
    const uint32_t stride = gridDim.x * WPB;
    for (uint32_t trial = blockIdx.x * WPB + (uint32_t)w; trial < ntrials;
         trial += stride) {
        E.init(&P, cmb::trial_seed(master_seed, trial_base + trial),
               (uint32_t)(trial_base + trial));
        Model::setup(E);
        E.run(until, max_events);
        Model::finish(E, out[trial]);
    }
}




// Device spill pool: slabs claimed lazily by trials that overflow their
// fast tier (Engine::claim_spill).  Sized far above the measured overflow
// rate (~1 trial per 5e5 at MG1 lognormal SCV=4, rho=0.8) and slabs stay
// claimed by a lane across its later trials, so 4096 covers any batch.
template <class Model>
struct DevSpillPool {
    using Spill = typename cmb::Engine<Model>::Spill;
    Spill* arena = nullptr;
    int32_t* cursor = nullptr;
    int32_t cap = 0;
    int alloc(uint64_t ntrials) {
        if constexpr (!cmb::Engine<Model>::NEEDS_SPILL) {
            (void)ntrials;
            return 0;
        } else {
            const char* e = getenv("CIMBA_SPILL_SLABS");
            uint64_t want = e ? (uint64_t)atoll(e) : 4096;
            if (want > ntrials) want = ntrials;
            if (want == 0) want = 1;
            cap = (int32_t)want;
            hipError_t err = hipMalloc(&arena, sizeof(Spill) * want);
            if (err != hipSuccess) return (int)err;
            err = hipMalloc(&cursor, sizeof(int32_t));
            if (err != hipSuccess) return (int)err;
            return (int)hipMemset(cursor, 0, sizeof(int32_t));
        }
    }
    void release() {
        if (arena) hipFree(arena);
        if (cursor) hipFree(cursor);
        arena = nullptr;
        cursor = nullptr;
    }
};

// host-side launcher: upload params, launch, copy per-trial results back
template <class Model, int WPB, int MINW = 1>
int run_trials_gpu(const typename Model::Params& P, uint64_t ntrials,
                   uint64_t seed, uint64_t trial_base, double until,
                   uint64_t max_events, double* elapsed_ms,
                   typename Model::Result* host_out) {
    using Result = typename Model::Result;
    Result* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_out, sizeof(Result) * ntrials));
    const uint32_t want_blocks = (uint32_t)((ntrials + WPB - 1) / WPB);
    const uint32_t grid = want_blocks < 16384u ? want_blocks : 16384u;
    DevSpillPool<Model> pool;
    HIP_TRY((hipError_t)pool.alloc(ntrials));

    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL((trial_kernel<Model, WPB, MINW>), dim3(grid),
                       dim3(WPB * 64), 0, 0, P, seed, trial_base,
                       (uint32_t)ntrials, until, max_events, d_out,
                       pool.arena, pool.cursor, pool.cap);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(host_out, d_out, sizeof(Result) * ntrials,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
    pool.release();
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

// lane-parallel variant: one trial per LANE (64 per wave), per-lane
// Storage in HBM (L2/L3-cached — an MM1 store is ~5 KB, so a full
// launch's working set sits in the 256 MiB L3).  The engine context is
// per-lane registers either way; the SIMT cost is divergence across the
// (event-kind, resume-pc) dispatch, the win is 64 trials per instruction
// stream.  Measured A/B against the wave-per-trial kernel in profiles/.
template <class Model, int MINW = 1>
__global__ __launch_bounds__(256, MINW) __attribute__((flatten)) void lane_trial_kernel(
    typename Model::Params P, uint64_t master_seed, uint64_t trial_base,
    uint32_t ntrials, double until, uint64_t max_events,
    typename Model::Result* __restrict__ out,
    typename Engine<Model>::Storage* __restrict__ stores,
    typename Engine<Model>::Spill* __restrict__ sp_arena,
    int32_t* __restrict__ sp_cur, int32_t sp_cap) {
    const uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
    const uint32_t stride = gridDim.x * blockDim.x;
    Engine<Model> E(stores[gid]);
    E.set_spill_pool(sp_arena, sp_cur, sp_cap);
    for (uint32_t trial = gid; trial < ntrials; trial += stride) {
        E.init(&P, cmb::trial_seed(master_seed, trial_base + trial),
               (uint32_t)(trial_base + trial));
        Model::setup(E);
        E.run(until, max_events);
        Model::finish(E, out[trial]);
    }
}

// scratch variant: the per-trial Storage is a kernel LOCAL, so it lives
// in private (scratch) memory — which the hardware interleaves per lane.
// Same-field accesses across the 64 lanes of a wave therefore COALESCE
// into wide transactions: the AoS->SoA transposition for free.
template <class Model, int MINW = 1>
__global__ __launch_bounds__(256, MINW) void lane_scratch_kernel(
    typename Model::Params P, uint64_t master_seed, uint64_t trial_base,
    uint32_t ntrials, double until, uint64_t max_events,
    typename Model::Result* __restrict__ out,
    typename Engine<Model>::Spill* __restrict__ sp_arena,
    int32_t* __restrict__ sp_cur, int32_t sp_cap) {
    const uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
    const uint32_t stride = gridDim.x * blockDim.x;
    typename Engine<Model>::Storage st;  // per-lane scratch (HW-swizzled)
    Engine<Model> E(st);
    E.set_spill_pool(sp_arena, sp_cur, sp_cap);
    for (uint32_t trial = gid; trial < ntrials; trial += stride) {
        E.init(&P, cmb::trial_seed(master_seed, trial_base + trial),
               (uint32_t)(trial_base + trial));
        Model::setup(E);
        E.run(until, max_events);
        Model::finish(E, out[trial]);
    }
}

// ---------------------------------------------------------------------------
// Path-converged lane kernel: the divergence fix for the lane regime.
//
// The plain lane kernels run 64 independent trials per wave; at each
// dispatch step the 64 lanes sit at random (event-kind, process-func,
// resume-pc) points, so the wave serializes over every distinct path
// present and VALUUtilization measured 20-28%
// (profiles/r01_lane_divergence.md).  Here each lane still owns ONE trial
// in scratch, but dispatch is gated by a wave vote: every iteration the
// lanes peek their next dispatch path (Engine::peek_path), poll up to 4
// distinct candidates (__shfl from the first lane of each unseen group),
// and ONLY the lanes on the most-popular path dispatch.  Idle minority
// lanes cost less than serializing all paths.  Measurement history and
// the shipped-default matrix: profiles/r02_conv_divergence.md (first
// A/B: 3.93 vs 3.33 G ev/s, profiles/logs/r2_ab1.log; final with the
// heap-top cache: 4.69 G and conv wins for every lane model).
//
// Trial refill is wave-synchronous via the stride loop — all lanes of a
// wave start their next trial together, so the low-utilization tail is
// one trial-length VARIANCE, not a full trial.  (The earlier design —
// K parked trials per lane with per-lane atomic refill — measured 21%
// WORSE at any batch that needs refill: staggered per-lane trial ages
// leave each wave grinding out stragglers at ~50% liveness after the
// pool empties.  profiles/logs (r2 sweep data); kept here as the measured
// justification for the no-refill shape.)
// ---------------------------------------------------------------------------
template <class Model, int MINW = 1>
__global__ __launch_bounds__(256, MINW) void conv_lane_kernel(
    typename Model::Params P, uint64_t master_seed, uint64_t trial_base,
    uint32_t ntrials, double until, uint64_t max_events,
    typename Model::Result* __restrict__ out,
    typename Engine<Model>::Spill* __restrict__ sp_arena,
    int32_t* __restrict__ sp_cur, int32_t sp_cap) {
    using Eng = Engine<Model>;
    constexpr uint32_t DONE = Eng::PATH_DONE;
    const uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
    const uint32_t stride = gridDim.x * blockDim.x;
    typename Eng::Storage st;  // per-lane scratch (HW-swizzled)
    Eng E(st);
    E.set_spill_pool(sp_arena, sp_cur, sp_cap);
    for (uint32_t trial = gid;; trial += stride) {
        const bool have = trial < ntrials;
        if (__ballot(have) == 0) break;
        if (have) {
            E.init(&P, cmb::trial_seed(master_seed, trial_base + trial),
                   (uint32_t)(trial_base + trial));
            Model::setup(E);
        }
        uint32_t mypath = have ? E.peek_path(until, max_events) : DONE;
        for (;;) {
            const uint64_t live = __ballot(mypath != DONE);
            if (live == 0) break;
            // poll up to 4 distinct paths; serve the most popular
            uint64_t rem = live;
            uint32_t best = 0;
            int bestv = -1;
            for (int it = 0; it < 4 && rem; ++it) {
                const int src = __ffsll((unsigned long long)rem) - 1;
                const uint32_t cand = __shfl(mypath, src);
                const uint64_t m = __ballot(mypath == cand);
                const int v = __popcll(m);
                if (v > bestv) {
                    bestv = v;
                    best = cand;
                }
                rem &= ~m;
            }
            if (mypath == best) {
                E.dispatch_one();
                if (E.ev_dispatched >= max_events)
                    E.fail(cmb::ST_EVENT_LIMIT);
                mypath = E.peek_path(until, max_events);
            }
        }
        if (have) {
            if (!E.evq.empty() && E.evq.top().t > until) E.now = until;
            Model::finish(E, out[trial]);
        }
    }
}

template <class Model, int MINW>
int run_trials_gpu_conv(const typename Model::Params& P, uint64_t ntrials,
                        uint64_t seed, uint64_t trial_base, double until,
                        uint64_t max_events, double* elapsed_ms,
                        typename Model::Result* host_out, uint32_t blocks) {
    using Result = typename Model::Result;
    if (blocks == 0) blocks = 16384u;
    const uint32_t want = (uint32_t)((ntrials + 255) / 256);
    const uint32_t grid = want < blocks ? want : blocks;
    if (getenv("CIMBA_CONV_DEBUG"))
        fprintf(stderr, "[conv] grid=%u lanes=%u ntrials=%llu\n", grid,
                grid * 256u, (unsigned long long)ntrials);
    Result* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_out, sizeof(Result) * ntrials));
    DevSpillPool<Model> pool;
    HIP_TRY((hipError_t)pool.alloc(ntrials));
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL((conv_lane_kernel<Model, MINW>), dim3(grid),
                       dim3(256), 0, 0, P, seed, trial_base,
                       (uint32_t)ntrials, until, max_events, d_out,
                       pool.arena, pool.cursor, pool.cap);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(host_out, d_out, sizeof(Result) * ntrials,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
    pool.release();
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

template <class Model, int MINW>
int run_trials_gpu_lane_scratch(const typename Model::Params& P,
                                uint64_t ntrials, uint64_t seed,
                                uint64_t trial_base, double until,
                                uint64_t max_events, double* elapsed_ms,
                                typename Model::Result* host_out,
                                uint32_t blocks) {
    using Result = typename Model::Result;
    const uint32_t want = (uint32_t)((ntrials + 255) / 256);
    const uint32_t grid = want < blocks ? want : blocks;
    Result* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_out, sizeof(Result) * ntrials));
    DevSpillPool<Model> pool;
    HIP_TRY((hipError_t)pool.alloc(ntrials));
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL((lane_scratch_kernel<Model, MINW>), dim3(grid),
                       dim3(256), 0, 0, P, seed, trial_base,
                       (uint32_t)ntrials, until, max_events, d_out,
                       pool.arena, pool.cursor, pool.cap);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(host_out, d_out, sizeof(Result) * ntrials,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
    pool.release();
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

template <class Model, int MINW>
int run_trials_gpu_lane(const typename Model::Params& P, uint64_t ntrials,
                        uint64_t seed, uint64_t trial_base, double until,
                        uint64_t max_events, double* elapsed_ms,
                        typename Model::Result* host_out, uint32_t blocks) {
    using Result = typename Model::Result;
    using St = typename Engine<Model>::Storage;
    const uint32_t want = (uint32_t)((ntrials + 255) / 256);
    const uint32_t grid = want < blocks ? want : blocks;
    Result* d_out = nullptr;
    St* d_st = nullptr;
    HIP_TRY(hipMalloc(&d_out, sizeof(Result) * ntrials));
    HIP_TRY(hipMalloc(&d_st, sizeof(St) * (size_t)grid * 256));
    DevSpillPool<Model> pool;
    HIP_TRY((hipError_t)pool.alloc(ntrials));
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL((lane_trial_kernel<Model, MINW>), dim3(grid),
                       dim3(256), 0, 0, P, seed, trial_base,
                       (uint32_t)ntrials, until, max_events, d_out, d_st,
                       pool.arena, pool.cursor, pool.cap);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(host_out, d_out, sizeof(Result) * ntrials,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
    HIP_TRY(hipFree(d_st));
    pool.release();
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

// scratch-kernel front end: MINW occupancy knob (CIMBA_SCRATCH_MINW)
template <class Model>
int run_scratch_auto(const typename Model::Params& P, uint64_t ntrials,
                     uint64_t seed, uint64_t trial_base, double until,
                     uint64_t max_events, double* elapsed_ms,
                     typename Model::Result* host_out, uint32_t blocks) {
    const char* me = getenv("CIMBA_SCRATCH_MINW");
    const int minw = me ? atoi(me) : 1;
    if (minw >= 4)
        return run_trials_gpu_lane_scratch<Model, 4>(
            P, ntrials, seed, trial_base, until, max_events, elapsed_ms,
            host_out, blocks);
    return run_trials_gpu_lane_scratch<Model, 1>(
        P, ntrials, seed, trial_base, until, max_events, elapsed_ms,
        host_out, blocks);
}

// converged-kernel front end: grid (CIMBA_CONV_BLOCKS, 0 = one lane per
// trial capped at 16384 blocks) and register budget (CIMBA_CONV_MINW)
template <class Model>
int run_conv_auto(const typename Model::Params& P, uint64_t ntrials,
                  uint64_t seed, uint64_t trial_base, double until,
                  uint64_t max_events, double* elapsed_ms,
                  typename Model::Result* host_out) {
    const char* be = getenv("CIMBA_CONV_BLOCKS");
    const uint32_t blocks = be ? (uint32_t)atoi(be) : 0u;
    // MINW (waves/SIMD floor) trades registers for occupancy: the engine
    // context spills into hardware-swizzled scratch (L1-friendly) and the
    // extra resident waves hide the dispatch chain's memory latency — the
    // dominant term at 131 VGPRs / 3 waves (r2 sweeps).
    // measured (profiles/r02_conv_divergence.md): MINW=4 is the sweet
    // spot, 6/8 overspill everywhere — only 1 and 4 stay instantiated
    const char* me = getenv("CIMBA_CONV_MINW");
    const int minw = me ? atoi(me) : 4;
    if (minw >= 4)
        return run_trials_gpu_conv<Model, 4>(P, ntrials, seed, trial_base,
                                             until, max_events, elapsed_ms,
                                             host_out, blocks);
    return run_trials_gpu_conv<Model, 1>(P, ntrials, seed, trial_base,
                                         until, max_events, elapsed_ms,
                                         host_out, blocks);
}



}  // namespace cmb_dk
