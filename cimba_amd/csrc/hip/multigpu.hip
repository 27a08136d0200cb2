// Native multi-GPU experiment fan-out over RCCL/xGMI — the BASELINE.json
// north-star collective pattern without any Python in the loop:
// ncclBroadcast of the model parameters from GPU 0, per-device
// trial-shard launches (the same lane/wave DES kernels), then
// ncclAllReduce of the per-device DataSummary partials (raw-moment form:
// sums reduce with ncclSum; min/max with ncclMin/ncclMax).  Payloads are
// KB-scale, so this is latency-bound — exactly SURVEY.md §5.8's analysis:
// the hot trial loops stay share-nothing on each GPU and only statistics
// cross xGMI.
//
// This is the single-process form (one host thread per device with
// ncclGroupStart/End).  The per-rank multi-process form used by bench.py
// (torch.distributed backend "nccl" = this same RCCL) measures scaling;
// this runner exists for C/C++ deployments without Python and is covered
// by a GPU test at ndev=1 (the collectives run with nranks=1).
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include "../models/mm1.hpp"
#include "../include/cimba/runner.hpp"
#include "../include/cimba/stats.hpp"

#include <algorithm>
#include <chrono>
#include <memory>
#include <thread>
#include <vector>

namespace {

using cmb::DataSummary;
using cmb::Engine;
using cmb_models::MM1;

#define HIP_TRY(x)                                    \
    do {                                              \
        hipError_t err_ = (x);                        \
        if (err_ != hipSuccess) return (int)err_;     \
    } while (0)
#define NCCL_TRY(x)                                          \
    do {                                                     \
        ncclResult_t nr_ = (x);                              \
        if (nr_ != ncclSuccess) return 10000 + (int)nr_;     \
    } while (0)

// same lane-parallel scratch kernel shape as deskernel.hip's default
__global__ __launch_bounds__(256) void mm1_lane_kernel_mg(
    MM1::Params P, uint64_t master_seed, uint32_t trial_lo, uint32_t trial_hi,
    double until, uint64_t max_events, MM1::Result* __restrict__ out) {
    const uint32_t gid = blockIdx.x * blockDim.x + threadIdx.x;
    const uint32_t stride = gridDim.x * blockDim.x;
    typename Engine<MM1>::Storage st;
    Engine<MM1> E(st);
    for (uint32_t trial = trial_lo + gid; trial < trial_hi; trial += stride) {
        E.init(&P, cmb::trial_seed(master_seed, trial), trial);
        MM1::setup(E);
        E.run(until, max_events);
        MM1::finish(E, out[trial - trial_lo]);
    }
}

// per-device reduction kernel: per-trial avg system times -> raw moment
// sums (n, Sx, Sx2, Sx3, Sx4, events, min, max), ALL on-device so only
// 8 doubles cross xGMI per GPU (the event total included — no per-trial
// Result round-trip to the host)
__global__ __launch_bounds__(256) void mm1_summarize_kernel(
    const MM1::Result* __restrict__ res, uint32_t n,
    double* __restrict__ out8) {
    __shared__ double acc[6][256];
    __shared__ double amin[256], amax[256];
    const uint32_t tid = threadIdx.x;
    double s[6] = {0, 0, 0, 0, 0, 0};
    double mn = 1e308, mx = -1e308;
    for (uint32_t i = blockIdx.x * blockDim.x + tid; i < n;
         i += gridDim.x * blockDim.x) {
        const double x =
            res[i].obj_cnt ? res[i].sum_wait / (double)res[i].obj_cnt : 0.0;
        const double x2 = x * x;
        s[0] += 1.0;
        s[1] += x;
        s[2] += x2;
        s[3] += x2 * x;
        s[4] += x2 * x2;
        s[5] += (double)res[i].events;
        mn = x < mn ? x : mn;
        mx = x > mx ? x : mx;
    }
    for (int k = 0; k < 6; ++k) acc[k][tid] = s[k];
    amin[tid] = mn;
    amax[tid] = mx;
    __syncthreads();
    for (uint32_t w = 128; w > 0; w >>= 1) {
        if (tid < w) {
            for (int k = 0; k < 6; ++k) acc[k][tid] += acc[k][tid + w];
            amin[tid] = amin[tid + w] < amin[tid] ? amin[tid + w] : amin[tid];
            amax[tid] = amax[tid + w] > amax[tid] ? amax[tid + w] : amax[tid];
        }
        __syncthreads();
    }
    if (tid == 0) {
        for (int k = 0; k < 6; ++k) atomicAdd(&out8[k], acc[k][0]);
        // min/max via CAS loops (few blocks)
        unsigned long long* pmn = (unsigned long long*)&out8[6];
        unsigned long long old = *pmn, assumed;
        do {
            assumed = old;
            if (__longlong_as_double(assumed) <= amin[0]) break;
            old = atomicCAS(pmn, assumed, __double_as_longlong(amin[0]));
        } while (old != assumed);
        unsigned long long* pmx = (unsigned long long*)&out8[7];
        old = *pmx;
        do {
            assumed = old;
            if (__longlong_as_double(assumed) >= amax[0]) break;
            old = atomicCAS(pmx, assumed, __double_as_longlong(amax[0]));
        } while (old != assumed);
    }
}

struct DevCtx {
    int dev;
    ncclComm_t comm;
    hipStream_t stream;
    MM1::Params* d_params;
    MM1::Result* d_res;
    double* d_sums;   // [8]: n, Sx, Sx2, Sx3, Sx4, min, max, events
    double* d_red;    // reduced [8]
    uint32_t lo, hi;
    int rc;
};

}  // namespace

extern "C" {

// Run `ntrials` M/M/1 replications sharded across `ndev` GPUs (-1 = all
// visible) in ONE process; RCCL broadcasts the parameters and reduces the
// statistics.  out10: [n, mean, var, min, max, events, ndev, ms, 0, 0].
int cimba_mm1_multigpu_rccl(uint64_t ntrials, double arr_mean,
                            double srv_mean, uint64_t num_objects,
                            uint64_t seed, int ndev, double* out10) {
    int avail = 0;
    HIP_TRY(hipGetDeviceCount(&avail));
    if (ndev <= 0 || ndev > avail) ndev = avail;
    if (ndev == 0) return -1;

    std::vector<DevCtx> ctx((size_t)ndev);
    std::vector<ncclComm_t> comms((size_t)ndev);
    std::vector<int> devs((size_t)ndev);
    for (int i = 0; i < ndev; ++i) devs[(size_t)i] = i;
    NCCL_TRY(ncclCommInitAll(comms.data(), ndev, devs.data()));

    MM1::Params host_params{arr_mean, srv_mean, num_objects};

    const auto t0 = std::chrono::steady_clock::now();
    auto worker = [&](int i) {
        DevCtx& c = ctx[(size_t)i];
        c.dev = i;
        c.comm = comms[(size_t)i];
        c.rc = 0;
        auto TRY = [&](hipError_t e) {
            if (e != hipSuccess && c.rc == 0) c.rc = (int)e;
            return e == hipSuccess;
        };
        auto NTRY = [&](ncclResult_t e) {
            if (e != ncclSuccess && c.rc == 0) c.rc = 10000 + (int)e;
            return e == ncclSuccess;
        };
        if (!TRY(hipSetDevice(i))) return;
        if (!TRY(hipStreamCreate(&c.stream))) return;
        c.lo = (uint32_t)(ntrials * (uint64_t)i / (uint64_t)ndev);
        c.hi = (uint32_t)(ntrials * (uint64_t)(i + 1) / (uint64_t)ndev);
        const uint32_t mine = c.hi - c.lo;
        if (!TRY(hipMalloc(&c.d_params, sizeof(MM1::Params)))) return;
        if (i == 0 &&
            !TRY(hipMemcpyAsync(c.d_params, &host_params,
                                sizeof(MM1::Params), hipMemcpyHostToDevice,
                                c.stream)))
            return;
        // the north-star broadcast: parameters from rank 0 over xGMI
        if (!NTRY(ncclBroadcast(c.d_params, c.d_params,
                                sizeof(MM1::Params), ncclChar, 0, c.comm,
                                c.stream)))
            return;
        MM1::Params p_local;
        if (!TRY(hipMemcpyAsync(&p_local, c.d_params, sizeof(MM1::Params),
                                hipMemcpyDeviceToHost, c.stream)))
            return;
        if (!TRY(hipStreamSynchronize(c.stream))) return;

        if (!TRY(hipMalloc(&c.d_res, sizeof(MM1::Result) * mine))) return;
        if (!TRY(hipMalloc(&c.d_sums, sizeof(double) * 8))) return;
        if (!TRY(hipMalloc(&c.d_red, sizeof(double) * 8))) return;
        const double init[8] = {0, 0, 0, 0, 0, 0, 1e308, -1e308};
        if (!TRY(hipMemcpyAsync(c.d_sums, init, sizeof(init),
                                hipMemcpyHostToDevice, c.stream)))
            return;
        const uint32_t grid =
            std::min((mine + 255u) / 256u, 2048u);
        hipLaunchKernelGGL(mm1_lane_kernel_mg, dim3(grid), dim3(256), 0,
                           c.stream, p_local, seed, c.lo, c.hi, 1.0e308,
                           UINT64_C(0xFFFFFFFFFFFFFFFF), c.d_res);
        if (!TRY(hipGetLastError())) return;
        hipLaunchKernelGGL(mm1_summarize_kernel, dim3(64), dim3(256), 0,
                           c.stream, c.d_res, mine, c.d_sums);
        if (!TRY(hipGetLastError())) return;
        // the north-star reduce: statistics over xGMI
        // (sums: moments + event total, all summed on-device; min / max)
        if (!NTRY(ncclAllReduce(c.d_sums, c.d_red, 6, ncclDouble, ncclSum,
                                c.comm, c.stream)))
            return;
        if (!NTRY(ncclAllReduce(c.d_sums + 6, c.d_red + 6, 1, ncclDouble,
                                ncclMin, c.comm, c.stream)))
            return;
        if (!NTRY(ncclAllReduce(c.d_sums + 7, c.d_red + 7, 1, ncclDouble,
                                ncclMax, c.comm, c.stream)))
            return;
        if (!TRY(hipStreamSynchronize(c.stream))) return;
    };

    {
        std::vector<std::thread> th;
        for (int i = 0; i < ndev; ++i) th.emplace_back(worker, i);
        for (auto& t : th) t.join();
    }
    const double ms = std::chrono::duration<double, std::milli>(
                          std::chrono::steady_clock::now() - t0)
                          .count();

    int rc = 0;
    for (auto& c : ctx)
        if (c.rc && !rc) rc = c.rc;

    double red[8] = {0};
    if (!rc) {
        HIP_TRY(hipSetDevice(0));
        HIP_TRY(hipMemcpy(red, ctx[0].d_red, sizeof(red),
                          hipMemcpyDeviceToHost));
    }
    for (auto& c : ctx) {
        hipSetDevice(c.dev);
        if (c.d_params) hipFree(c.d_params);
        if (c.d_res) hipFree(c.d_res);
        if (c.d_sums) hipFree(c.d_sums);
        if (c.d_red) hipFree(c.d_red);
        if (c.stream) hipStreamDestroy(c.stream);
        ncclCommDestroy(c.comm);
    }
    if (rc) return rc;

    const double n = red[0];
    const double mean = n > 0 ? red[1] / n : 0.0;
    const double var = n > 1 ? (red[2] - n * mean * mean) / (n - 1.0) : 0.0;
    out10[0] = n;
    out10[1] = mean;
    out10[2] = var;
    out10[3] = red[6];  // min
    out10[4] = red[7];  // max
    out10[5] = red[5];  // total events (sum-reduced)
    out10[6] = (double)ndev;
    out10[7] = ms;
    out10[8] = 0.0;
    out10[9] = 0.0;
    return 0;
}

}  // extern "C"
