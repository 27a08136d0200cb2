// Host experiment executive: the CPU-plumbing counterpart of reference
// src/cimba.c cimba_run: worker threads claim trials from
// a shared atomic index, run the model trial function, write results back
// into the experiment array.  The GPU path (hip/deskernel.hip) replaces
// worker threads with wavefronts; this host path exists for BASELINE.json
// config 1 (CPU plumbing) and for CPU-only unit testing of the exact same
// engine code.
#pragma once

#include "engine.hpp"

#include <atomic>
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
    double until;
    uint64_t max_events;
};

template <class Model>
void run_host(const typename Model::Params& params, uint64_t master_seed,
              uint64_t ntrials, int nthreads, typename Model::Result* out,
              RunLimits limits = {1.0e308, UINT64_C(0xFFFFFFFFFFFFFFFF)}) {
    if (nthreads <= 0) {
        nthreads = (int)std::thread::hardware_concurrency();
        if (nthreads <= 0) nthreads = 1;
    }
    if ((uint64_t)nthreads > ntrials) nthreads = (int)(ntrials ? ntrials : 1);

    std::atomic<uint64_t> next{0};  // reference work-claim: cimba.c:280
    auto worker = [&]() {
        // one engine per worker, reused across trials (reference: per-thread
        // event queue reset between trials, cimba.c:332)
        auto eng = std::make_unique<Engine<Model>>();
        for (;;) {
            const uint64_t t = next.fetch_add(1, std::memory_order_relaxed);
            if (t >= ntrials) break;
            eng->init(&params, trial_seed(master_seed, t), (uint32_t)t);
            Model::setup(*eng);
            eng->run(limits.until, limits.max_events);
            Model::finish(*eng, out[t]);
        }
    };

    if (nthreads == 1) {
        worker();
        return;
    }
    std::vector<std::thread> threads;
    threads.reserve(nthreads);
    for (int i = 0; i < nthreads; ++i) threads.emplace_back(worker);
    for (auto& th : threads) th.join();
}

}  // namespace cmb
