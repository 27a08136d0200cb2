// gfx950 DES kernel entry points for the M/M/1 flagship model, plus the
// device utility calls.  The kernel/launcher templates live in
// deskernel_impl.hpp; MG1 / JobShop / Scenario entries compile in their
// own TUs (deskernel_mg1/jobshop/scenario.hip) so hipcc builds the
// models in parallel (the combined TU took ~8 min at -O3).
#include "deskernel_impl.hpp"

#include "../models/mm1.hpp"

using cmb::Engine;
using cmb_models::MM1;
using namespace cmb_dk;

// LDS budget check: 4 engine stores per workgroup, several workgroups/CU
static_assert(sizeof(Engine<MM1>::Storage) * 4 < 60 * 1024, "MM1 LDS plan");

extern "C" {

struct Mm1GpuOut {
    double elapsed_ms;
    uint64_t total_events;
    uint64_t total_objs;
    double total_wait;
    uint64_t trials_ok;
    int32_t first_bad_status;
    int32_t pad_;
};

// per_trial_avg_out (nullable): per-trial average system times, for the
// host-vs-device divergence-count test (tests/test_gpu.py)
int cimba_mm1_gpu_run_pt(uint64_t ntrials, double arr_mean, double srv_mean,
                         uint64_t num_objects, uint64_t seed,
                         uint64_t trial_base, int device, double until,
                         uint64_t max_events, Mm1GpuOut* out,
                         double* per_trial_avg_out);

int cimba_mm1_gpu_run(uint64_t ntrials, double arr_mean, double srv_mean,
                      uint64_t num_objects, uint64_t seed,
                      uint64_t trial_base, int device, double until,
                      uint64_t max_events, Mm1GpuOut* out) {
    return cimba_mm1_gpu_run_pt(ntrials, arr_mean, srv_mean, num_objects,
                                seed, trial_base, device, until, max_events,
                                out, nullptr);
}

int cimba_mm1_gpu_run_pt(uint64_t ntrials, double arr_mean, double srv_mean,
                         uint64_t num_objects, uint64_t seed,
                         uint64_t trial_base, int device, double until,
                         uint64_t max_events, Mm1GpuOut* out,
                         double* per_trial_avg_out) {
    HIP_TRY(hipSetDevice(device));
    MM1::Params P{arr_mean, srv_mean, num_objects};
    std::vector<MM1::Result> res(ntrials);
    const char* mw = getenv("CIMBA_MM1_MINW");
    const int minw = mw ? atoi(mw) : 5;  // measured best (profiles/)
    // lane-parallel auto policy: 64 trials/wave needs a large batch to
    // fill the chip; below the threshold the wave-per-trial kernel wins
    const char* lane = getenv("CIMBA_MM1_LANE");
    const bool use_lane = lane ? atoi(lane) != 0 : ntrials >= 32768;
    int rc;
    if (use_lane) {
        const char* lb = getenv("CIMBA_MM1_LANE_BLOCKS");
        const uint32_t blocks = lb ? (uint32_t)atoi(lb) : 2048u;
        // measured (profiles/r02_conv_divergence.md): after the heap-top
        // register cache the vote-gated conv kernel wins for M/M/1 too
        // (4.69 vs 4.55 G ev/s at the bench batch)
        const int lane_mode = lane ? atoi(lane) : 3;
        if (lane_mode == 1)  // explicit HBM-lane variant
            rc = run_trials_gpu_lane<MM1, 1>(P, ntrials, seed, trial_base, until,
                                             max_events, &out->elapsed_ms,
                                             res.data(), blocks);
        else if (lane_mode == 2)  // scratch (HW lane-interleaved)
            rc = run_scratch_auto<MM1>(P, ntrials, seed, trial_base, until,
                                       max_events, &out->elapsed_ms,
                                       res.data(), blocks);
        else  // default: path-converged K-trials-per-lane (divergence fix)
            rc = run_conv_auto<MM1>(P, ntrials, seed, trial_base, until,
                                    max_events, &out->elapsed_ms, res.data());
        goto aggregate;
    }
    switch (minw) {
        case 5:
            rc = run_trials_gpu<MM1, 4, 5>(P, ntrials, seed, trial_base, until,
                                           max_events, &out->elapsed_ms,
                                           res.data());
            break;
        case 6:
            rc = run_trials_gpu<MM1, 4, 6>(P, ntrials, seed, trial_base, until,
                                           max_events, &out->elapsed_ms,
                                           res.data());
            break;
        default:
            rc = run_trials_gpu<MM1, 4, 4>(P, ntrials, seed, trial_base, until,
                                           max_events, &out->elapsed_ms,
                                           res.data());
    }
aggregate:
    if (rc) return rc;
    out->total_events = 0;
    out->total_objs = 0;
    out->total_wait = 0.0;
    out->trials_ok = 0;
    out->first_bad_status = 0;
    for (uint64_t i = 0; i < ntrials; ++i) {
        out->total_events += res[i].events;
        out->total_objs += res[i].obj_cnt;
        out->total_wait += res[i].sum_wait;
        if (per_trial_avg_out)
            per_trial_avg_out[i] =
                res[i].obj_cnt ? res[i].sum_wait / (double)res[i].obj_cnt
                               : 0.0;
        if (res[i].status == 0)
            ++out->trials_ok;
        else if (out->first_bad_status == 0)
            out->first_bad_status = res[i].status;
    }
    return 0;
}

int cimba_gpu_device_count(int* n) {
    hipError_t err = hipGetDeviceCount(n);
    if (err != hipSuccess) {
        *n = 0;
        return (int)err;
    }
    return 0;
}

int cimba_gpu_sync(void) { return (int)hipDeviceSynchronize(); }

}  // extern "C"
