// M/G/1 DES kernel entry point (templates in deskernel_impl.hpp).
#include "deskernel_impl.hpp"

#include "../models/mg1.hpp"

using cmb::Engine;
using cmb_models::MG1;
using namespace cmb_dk;

static_assert(sizeof(Engine<MG1>::Storage) * 4 < 64 * 1024, "MG1 LDS plan");

extern "C" {

// MG1: results array provided by caller (per-trial)
int cimba_mg1_gpu_run(uint64_t ntrials, const void* params, uint64_t seed,
                      uint64_t trial_base, int device, double* elapsed_ms,
                      void* results_out) {
    HIP_TRY(hipSetDevice(device));
    const char* lane = getenv("CIMBA_MG1_LANE");
    const uint64_t nt_ = ntrials;
    if (lane ? atoi(lane) != 0 : nt_ >= 32768) {
        // measured: vote-gated conv at MINW=4 = 4.19 G ev/s vs 3.69 G
        // scratch — MG1 has a wider path mix (service-distribution
        // branches), so path convergence pays where it did not for M/M/1
        const int lane_mode = lane ? atoi(lane) : 3;
        if (lane_mode == 3)
            return run_conv_auto<MG1>(*(const MG1::Params*)params, ntrials,
                                      seed, trial_base, 1.0e308,
                                      UINT64_C(0xFFFFFFFFFFFFFFFF),
                                      elapsed_ms, (MG1::Result*)results_out);
        return run_scratch_auto<MG1>(
            *(const MG1::Params*)params, ntrials, seed, trial_base, 1.0e308,
            UINT64_C(0xFFFFFFFFFFFFFFFF), elapsed_ms,
            (MG1::Result*)results_out, 2048u);
    }
    const char* mw = getenv("CIMBA_MG1_MINW");
    const int minw = mw ? atoi(mw) : 4;
    if (minw >= 4)
        return run_trials_gpu<MG1, 4, 4>(*(const MG1::Params*)params, ntrials,
                                         seed, trial_base, 1.0e308,
                                         UINT64_C(0xFFFFFFFFFFFFFFFF),
                                         elapsed_ms,
                                         (MG1::Result*)results_out);
    if (minw == 3)
        return run_trials_gpu<MG1, 4, 3>(*(const MG1::Params*)params, ntrials,
                                         seed, trial_base, 1.0e308,
                                         UINT64_C(0xFFFFFFFFFFFFFFFF),
                                         elapsed_ms,
                                         (MG1::Result*)results_out);
    return run_trials_gpu<MG1, 4>(*(const MG1::Params*)params, ntrials, seed, trial_base,
                                  1.0e308, UINT64_C(0xFFFFFFFFFFFFFFFF),
                                  elapsed_ms, (MG1::Result*)results_out);
}

}  // extern "C"
