// Host experiment executive: the CPU-plumbing counterpart of reference
// src/cimba.c cimba_run: worker threads claim trials from
// a shared atomic index, run the model trial function, write results back
// into the experiment array.  Per-thread hooks (reference
// cimba_thread_hooks_set, cimba.c:150-157: e.g. per-thread GPU streams),
// trial-abandon recovery (reference __builtin_setjmp/longjmp,
// cimba.c:74-76,289-329 — here a C++ exception over a POD engine) and the
// failed-trial count (cimba.c:399) are all supported.  The GPU path
// (hip/deskernel.hip) replaces worker threads with wavefronts; this host
// path exists for BASELINE.json config 1 (CPU plumbing) and for CPU-only
// unit testing of the exact same engine code.
#pragma once

#include "engine.hpp"
#include "logger.hpp"

#include <atomic>
#include <functional>
#include <memory>
#include <thread>
#include <vector>

namespace cmb {

// per-trial seed derivation from a master seed (reference seed discipline:
// master seed -> fmix64 per-trial seeds, include/cimba.h:126-148)
CMB_FORCEINLINE uint64_t trial_seed(uint64_t master, uint64_t idx) {
    return fmix64(master ^ (UINT64_C(0x9E3779B97F4A7C15) * (idx + 1)));
}

struct RunLimits {
    double until = 1.0e308;
    uint64_t max_events = UINT64_C(0xFFFFFFFFFFFFFFFF);
};

// reference cimba_thread_hooks_set / cimba_trial_cleanup_set
struct RunHooks {
    std::function<void(int)> thread_init;          // worker index
    std::function<void(int)> thread_exit;
    std::function<void(uint64_t)> trial_cleanup;   // abandoned trial index
};

struct RunReport {
    uint64_t trials = 0;
    uint64_t failed = 0;     // aborted (status != 0) or abandoned trials
    uint64_t abandoned = 0;  // subset of failed: TrialAbandon thrown
};

// `trial_base` offsets the GLOBAL trial index: seeds derive from
// trial_base + t, so a run sharded across ranks/devices simulates the
// identical trial set as a single-device run of the same total — any
// world size reproduces bit-identical per-trial streams.
template <class Model>
RunReport run_host(const typename Model::Params& params, uint64_t master_seed,
                   uint64_t ntrials, int nthreads,
                   typename Model::Result* out, RunLimits limits = {},
                   const RunHooks* hooks = nullptr, uint64_t trial_base = 0) {
    if (nthreads <= 0) {
        nthreads = (int)std::thread::hardware_concurrency();
        if (nthreads <= 0) nthreads = 1;
    }
    if ((uint64_t)nthreads > ntrials) nthreads = (int)(ntrials ? ntrials : 1);

    std::atomic<uint64_t> next{0};  // reference work-claim: cimba.c:280
    std::atomic<uint64_t> failed{0};
    std::atomic<uint64_t> abandoned{0};

    auto worker = [&](int widx) {
        if (hooks && hooks->thread_init) hooks->thread_init(widx);
        // one engine per worker, reused across trials (reference: per-thread
        // event queue reset between trials, cimba.c:332)
        auto store = std::make_unique<typename Engine<Model>::Storage>();
        auto eng = std::make_unique<Engine<Model>>(*store);
        std::unique_ptr<typename Engine<Model>::Spill> slab;
        if constexpr (Engine<Model>::NEEDS_SPILL) {
            slab = std::make_unique<typename Engine<Model>::Spill>();
            eng->set_spill(slab.get());
        }
        for (;;) {
            const uint64_t t = next.fetch_add(1, std::memory_order_relaxed);
            if (t >= ntrials) break;
            const uint64_t gt = trial_base + t;  // global trial index
            const uint64_t seed = trial_seed(master_seed, gt);
            logger_ctx().trial = (uint32_t)gt;
            logger_ctx().seed = seed;
            logger_ctx().sim_time = 0.0;
            eng->init(&params, seed, (uint32_t)gt);
            try {
                Model::setup(*eng);
                eng->run(limits.until, limits.max_events);
            } catch (const TrialAbandon& ab) {
                // recovery: the engine is POD — re-init on the next trial is
                // the whole cleanup (reference memregistry LIFO teardown)
                eng->fail(ST_USER_ABORT);
                abandoned.fetch_add(1, std::memory_order_relaxed);
                if (hooks && hooks->trial_cleanup) hooks->trial_cleanup(t);
            }
            Model::finish(*eng, out[t]);
            if (eng->status != ST_OK)
                failed.fetch_add(1, std::memory_order_relaxed);
        }
        if (hooks && hooks->thread_exit) hooks->thread_exit(widx);
    };

    if (nthreads == 1) {
        worker(0);
    } else {
        std::vector<std::thread> threads;
        threads.reserve(nthreads);
        for (int i = 0; i < nthreads; ++i) threads.emplace_back(worker, i);
        for (auto& th : threads) th.join();
    }
    return RunReport{ntrials, failed.load(), abandoned.load()};
}

}  // namespace cmb
