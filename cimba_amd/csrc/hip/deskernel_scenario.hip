// Scenario semantic-parity kernel (templates in deskernel_impl.hpp):
// runs one hand-scenario trial on-device; the trace must match the host
// engine exactly (tests/test_scenarios.py).
#include "deskernel_impl.hpp"

#include "../models/scenarios.hpp"
#include "../models/spillprobe.hpp"

using cmb::Engine;
using cmb_models::Scenario;
using cmb_models::SpillProbe;
using namespace cmb_dk;

namespace {

// single-trial semantic-parity kernel: runs one Scenario trial on-device;
// the trace must match the host engine exactly (tests/test_gpu.py)
__global__ __launch_bounds__(64) __attribute__((flatten)) void scenario_kernel(
    Scenario::Params P, Scenario::Result* __restrict__ out) {
    __shared__ Engine<Scenario>::Storage st;
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    Engine<Scenario> eng(st);
    eng.init(&P, 123, 0);
    Scenario::setup(eng);
    eng.run(1.0e308, 100000);
    Scenario::finish(eng, *out);
}

}  // namespace

extern "C" {

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

// spill-probe entry (models/spillprobe.hpp): every trial crosses the
// fast->slab boundary, exercising lazy claim atomics + the two-tier
// heap/queue paths on the device; runs through the default conv kernel
int cimba_spillprobe_gpu_run(uint64_t ntrials, uint64_t num_objects,
                             uint64_t seed, uint64_t trial_base, int device,
                             double* elapsed_ms, void* results_out) {
    HIP_TRY(hipSetDevice(device));
    SpillProbe::Params P{num_objects};
    return run_trials_gpu_conv<SpillProbe, 4>(
        P, ntrials, seed, trial_base, 1.0e308,
        UINT64_C(0xFFFFFFFFFFFFFFFF), elapsed_ms,
        (SpillProbe::Result*)results_out, 0u);
}

}  // extern "C"
