// gfx950 DES kernels: one simulation trial per wavefront, engine state in
// LDS — the device realization of BASELINE.json's north star ("each trial
// becomes a HIP thread-block ... future-event list and small-object
// allocator in LDS ... thousands of replications advance per launch").
//
// Mapping: a workgroup of WPB*64 threads carries WPB independent trials,
// one per wavefront, each with a private Engine<Model> in LDS.  Lane 0 of
// each wave drives its trial's event loop (the trial-internal scheduler is
// sequential by construction — SURVEY.md §7 hard part (b): throughput comes
// from thousands of concurrent trials, exactly as the reference relies on
// cores).  WPB > 1 raises CU residency past the workgroups-per-CU limit a
// 64-thread block would hit; WPB=1 gives the literal block-per-trial
// mapping.  No __syncthreads anywhere: waves never share state.
#include <hip/hip_runtime.h>

#include "../models/mm1.hpp"
#include "../models/scenarios.hpp"
#include "../include/cimba/runner.hpp"

namespace {

using cmb::Engine;
using cmb_models::MM1;
using cmb_models::Scenario;

constexpr int WPB = 4;  // waves (= trials) per workgroup

using EngMM1 = Engine<MM1>;

// LDS budget check: WPB engines must fit the 160 KiB LDS of a gfx950 CU
// with room for multiple resident workgroups.
static_assert(sizeof(EngMM1) * WPB < 60 * 1024, "MM1 engine too big for LDS plan");

__global__ __launch_bounds__(WPB * 64) __attribute__((flatten)) void mm1_kernel(
    MM1::Params P, uint64_t master_seed, uint32_t ntrials, double until,
    uint64_t max_events, MM1::Result* __restrict__ out) {
    __shared__ EngMM1 eng[WPB];
    const int w = (int)(threadIdx.x >> 6);
    if ((threadIdx.x & 63) != 0) return;  // lane 0 of each wave drives
    EngMM1& E = eng[w];
    const uint32_t stride = gridDim.x * WPB;
    for (uint32_t trial = blockIdx.x * WPB + (uint32_t)w; trial < ntrials;
         trial += stride) {
        E.init(&P, cmb::trial_seed(master_seed, trial), trial);
        MM1::setup(E);
        E.run(until, max_events);
        MM1::finish(E, out[trial]);
    }
}

// single-trial semantic-parity kernel: runs one Scenario trial on-device;
// the trace must match the host engine exactly (tests/test_gpu.py)
__global__ __launch_bounds__(64) __attribute__((flatten)) void scenario_kernel(
    Scenario::Params P, Scenario::Result* __restrict__ out) {
    __shared__ Engine<Scenario> eng;
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    eng.init(&P, 123, 0);
    Scenario::setup(eng);
    eng.run(1.0e308, 100000);
    Scenario::finish(eng, *out);
}

#define HIP_TRY(x)                                    \
    do {                                              \
        hipError_t err_ = (x);                        \
        if (err_ != hipSuccess) return (int)err_;     \
    } while (0)

}  // namespace

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

// Launch ntrials M/M/1 replications on `device`; aggregates results on the
// host.  Returns 0 or a hipError_t.
int cimba_mm1_gpu_run(uint64_t ntrials, double arr_mean, double srv_mean,
                      uint64_t num_objects, uint64_t seed, int device,
                      double until, uint64_t max_events, Mm1GpuOut* out) {
    HIP_TRY(hipSetDevice(device));
    MM1::Params P{arr_mean, srv_mean, num_objects};
    MM1::Result* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_out, sizeof(MM1::Result) * ntrials));

    const uint32_t want_blocks = (uint32_t)((ntrials + WPB - 1) / WPB);
    const uint32_t grid = want_blocks < 16384u ? want_blocks : 16384u;

    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL(mm1_kernel, dim3(grid), dim3(WPB * 64), 0, 0, P, seed,
                       (uint32_t)ntrials, until, max_events, d_out);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));

    std::vector<MM1::Result> res(ntrials);
    HIP_TRY(hipMemcpy(res.data(), d_out, sizeof(MM1::Result) * ntrials,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));

    out->elapsed_ms = (double)ms;
    out->total_events = 0;
    out->total_objs = 0;
    out->total_wait = 0.0;
    out->trials_ok = 0;
    out->first_bad_status = 0;
    for (uint64_t i = 0; i < ntrials; ++i) {
        out->total_events += res[i].events;
        out->total_objs += res[i].obj_cnt;
        out->total_wait += res[i].sum_wait;
        if (res[i].status == 0)
            ++out->trials_ok;
        else if (out->first_bad_status == 0)
            out->first_bad_status = res[i].status;
    }
    return 0;
}

int cimba_scenario_gpu_run(int which, void* result_out) {
    Scenario::Params P{which};
    Scenario::Result* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_out, sizeof(Scenario::Result)));
    hipLaunchKernelGGL(scenario_kernel, dim3(1), dim3(64), 0, 0, P, d_out);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipMemcpy(result_out, d_out, sizeof(Scenario::Result),
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
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
