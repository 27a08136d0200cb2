// gfx950 bulk-sampling kernels: the device-wide batched counterpart of the
// reference's ziggurat samplers (reference src/cmb_random.c:148-175 +
// codegen tables; BASELINE.json north star: "ziggurat normal/exponential
// sampling as a hand-written CDNA4 kernel ... shown with rocprof
// counters").
//
// Unlike the in-trial RNG (one sequential stream per trial wavefront),
// these kernels run all 64 lanes of every wave: lane-parallel independent
// sfc64 streams (fmix64-derived per-lane seeds), grid-stride over the
// output, coalesced 8B stores.  The ziggurat common path is ~15 ALU ops +
// one 8B constant-cache table read per sample; rejection handling is
// per-lane (divergence cost ~2-3% at the published acceptance rates).
// A fused moment-reduction variant accumulates raw sums (n, Sx, Sx2, Sx3,
// Sx4, min, max) through wave shuffles + one atomic per wave, so 1e9+
// sample statistical tests never materialize the samples.
#include <hip/hip_runtime.h>

#include "../include/cimba/rng.hpp"

namespace {

using cmb::Rng;

enum DistId : int {
    DIST_U01 = 0,
    DIST_STD_NORMAL = 1,
    DIST_STD_EXPONENTIAL = 2,
    DIST_GAMMA = 3,    // p0 = shape
    DIST_POISSON = 4,  // p0 = mean
};

__device__ __forceinline__ double sample_one(Rng& r, int dist, double p0) {
    switch (dist) {
        case DIST_STD_NORMAL: return r.std_normal();
        case DIST_STD_EXPONENTIAL: return r.std_exponential();
        case DIST_GAMMA: return r.std_gamma(p0);
        case DIST_POISSON: return (double)r.poisson(p0);
        default: return r.u01();
    }
}

__global__ __launch_bounds__(256) void sample_kernel(
    int dist, double p0, uint64_t n, uint64_t seed, double* __restrict__ out) {
    const uint64_t gid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    Rng r;
    r.seed(cmb::fmix64(seed ^ (gid * UINT64_C(0x9E3779B97F4A7C15) + 1)));
    for (uint64_t i = gid; i < n; i += stride) {
        out[i] = sample_one(r, dist, p0);
    }
}

struct MomentAcc {
    double n, s1, s2, s3, s4, mn, mx;
};

__global__ __launch_bounds__(256) void sample_moments_kernel(
    int dist, double p0, uint64_t n, uint64_t seed,
    MomentAcc* __restrict__ out) {
    const uint64_t gid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    Rng r;
    r.seed(cmb::fmix64(seed ^ (gid * UINT64_C(0x9E3779B97F4A7C15) + 1)));
    double s1 = 0, s2 = 0, s3 = 0, s4 = 0;
    double mn = 1e308, mx = -1e308;
    uint64_t cnt = 0;
    for (uint64_t i = gid; i < n; i += stride) {
        const double x = sample_one(r, dist, p0);
        const double x2 = x * x;
        s1 += x;
        s2 += x2;
        s3 += x2 * x;
        s4 += x2 * x2;
        mn = x < mn ? x : mn;
        mx = x > mx ? x : mx;
        ++cnt;
    }
    // wave-level butterfly reduction over the 64 lanes
    for (int off = 32; off > 0; off >>= 1) {
        s1 += __shfl_down(s1, off);
        s2 += __shfl_down(s2, off);
        s3 += __shfl_down(s3, off);
        s4 += __shfl_down(s4, off);
        cnt += (uint64_t)__shfl_down((unsigned long long)cnt, off);
        const double omn = __shfl_down(mn, off);
        const double omx = __shfl_down(mx, off);
        mn = omn < mn ? omn : mn;
        mx = omx > mx ? omx : mx;
    }
    if ((threadIdx.x & 63) == 0) {
        atomicAdd(&out->n, (double)cnt);
        atomicAdd(&out->s1, s1);
        atomicAdd(&out->s2, s2);
        atomicAdd(&out->s3, s3);
        atomicAdd(&out->s4, s4);
        // f64 atomic min/max via CAS loop (rare after wave reduce)
        unsigned long long* pmn = (unsigned long long*)&out->mn;
        unsigned long long old = *pmn, assumed;
        do {
            assumed = old;
            if (__longlong_as_double(assumed) <= mn) break;
            old = atomicCAS(pmn, assumed, __double_as_longlong(mn));
        } while (old != assumed);
        unsigned long long* pmx = (unsigned long long*)&out->mx;
        old = *pmx;
        do {
            assumed = old;
            if (__longlong_as_double(assumed) >= mx) break;
            old = atomicCAS(pmx, assumed, __double_as_longlong(mx));
        } while (old != assumed);
    }
}


// MFMA-assisted variant (the north star asked for "MFMA for the batched
// table lookups/rejection math"; the A/B and the analysis live in
// profiles/r02_mfma_ziggurat.md).  The rejection/table math itself has
// no GEMM shape — per-sample work is independent elementwise int64/f64
// ops — so the only productive MFMA mapping is the MOMENT ACCUMULATION:
// v_mfma_f64_16x16x4_f64 with B = ones folds each lane's power value
// into the accumulator tile, offloading the 4 f64 adds per sample from
// the VALU onto the (otherwise idle) matrix pipe.
typedef __attribute__((ext_vector_type(4))) double f64x4;

__global__ __launch_bounds__(256) void sample_moments_mfma_kernel(
    int dist, double p0, uint64_t n, uint64_t seed,
    MomentAcc* __restrict__ out) {
    const uint64_t gid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
    Rng r;
    r.seed(cmb::fmix64(seed ^ (gid * UINT64_C(0x9E3779B97F4A7C15) + 1)));
    // D[i][j] += sum_k A[i][k] * B[k][j]; with B = ones every lane's A
    // value lands in its row's running sums — 4 power accumulators
    f64x4 d1 = {0, 0, 0, 0}, d2 = {0, 0, 0, 0};
    f64x4 d3 = {0, 0, 0, 0}, d4 = {0, 0, 0, 0};
    double mn = 1e308, mx = -1e308;
    uint64_t cnt = 0;
    for (uint64_t i = gid; i < n; i += stride) {
        const double x = sample_one(r, dist, p0);
        const double x2 = x * x;
        d1 = __builtin_amdgcn_mfma_f64_16x16x4f64(x, 1.0, d1, 0, 0, 0);
        d2 = __builtin_amdgcn_mfma_f64_16x16x4f64(x2, 1.0, d2, 0, 0, 0);
        d3 = __builtin_amdgcn_mfma_f64_16x16x4f64(x2 * x, 1.0, d3, 0, 0, 0);
        d4 = __builtin_amdgcn_mfma_f64_16x16x4f64(x2 * x2, 1.0, d4, 0, 0, 0);
        mn = x < mn ? x : mn;
        mx = x > mx ? x : mx;
        ++cnt;
    }
    // every D column holds the same row sums (B = ones): col 0 lanes own
    // rows (lane>>4)*4+r; fold rows via the same wave butterfly
    double s1 = d1[0] + d1[1] + d1[2] + d1[3];
    double s2 = d2[0] + d2[1] + d2[2] + d2[3];
    double s3 = d3[0] + d3[1] + d3[2] + d3[3];
    double s4 = d4[0] + d4[1] + d4[2] + d4[3];
    if ((threadIdx.x & 15) != 0) {  // only col-0 lanes carry real sums
        s1 = s2 = s3 = s4 = 0.0;
    }
    for (int off = 32; off > 0; off >>= 1) {
        s1 += __shfl_down(s1, off);
        s2 += __shfl_down(s2, off);
        s3 += __shfl_down(s3, off);
        s4 += __shfl_down(s4, off);
        cnt += (uint64_t)__shfl_down((unsigned long long)cnt, off);
        const double omn = __shfl_down(mn, off);
        const double omx = __shfl_down(mx, off);
        mn = omn < mn ? omn : mn;
        mx = omx > mx ? omx : mx;
    }
    if ((threadIdx.x & 63) == 0) {
        atomicAdd(&out->n, (double)cnt);
        atomicAdd(&out->s1, s1);
        atomicAdd(&out->s2, s2);
        atomicAdd(&out->s3, s3);
        atomicAdd(&out->s4, s4);
        unsigned long long* pmn = (unsigned long long*)&out->mn;
        unsigned long long old = *pmn, assumed;
        do {
            assumed = old;
            if (__longlong_as_double(assumed) <= mn) break;
            old = atomicCAS(pmn, assumed, __double_as_longlong(mn));
        } while (old != assumed);
        unsigned long long* pmx = (unsigned long long*)&out->mx;
        old = *pmx;
        do {
            assumed = old;
            if (__longlong_as_double(assumed) >= mx) break;
            old = atomicCAS(pmx, assumed, __double_as_longlong(mx));
        } while (old != assumed);
    }
}

#define HIP_TRY(x)                                    \
    do {                                              \
        hipError_t err_ = (x);                        \
        if (err_ != hipSuccess) return (int)err_;     \
    } while (0)

}  // namespace

extern "C" {

int cimba_sample_gpu(int dist, double p0, uint64_t n, uint64_t seed,
                     int device, double* host_out, double* elapsed_ms) {
    HIP_TRY(hipSetDevice(device));
    double* d_out = nullptr;
    HIP_TRY(hipMalloc(&d_out, n * sizeof(double)));
    const uint32_t grid = 2048;
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL(sample_kernel, dim3(grid), dim3(256), 0, 0, dist, p0,
                       n, seed, d_out);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(host_out, d_out, n * sizeof(double),
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_out));
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

int cimba_sample_moments_gpu2(int dist, double p0, uint64_t n,
                              uint64_t seed, int device, int use_mfma,
                              double* out7, double* elapsed_ms);

int cimba_sample_moments_gpu(int dist, double p0, uint64_t n, uint64_t seed,
                             int device, double* out7, double* elapsed_ms) {
    return cimba_sample_moments_gpu2(dist, p0, n, seed, device, 0, out7,
                                     elapsed_ms);
}

int cimba_sample_moments_gpu2(int dist, double p0, uint64_t n,
                              uint64_t seed, int device, int use_mfma,
                              double* out7, double* elapsed_ms) {
    HIP_TRY(hipSetDevice(device));
    MomentAcc h{0, 0, 0, 0, 0, 1e308, -1e308};
    MomentAcc* d = nullptr;
    HIP_TRY(hipMalloc(&d, sizeof(MomentAcc)));
    HIP_TRY(hipMemcpy(d, &h, sizeof(MomentAcc), hipMemcpyHostToDevice));
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    if (use_mfma)
        hipLaunchKernelGGL(sample_moments_mfma_kernel, dim3(2048), dim3(256),
                           0, 0, dist, p0, n, seed, d);
    else
        hipLaunchKernelGGL(sample_moments_kernel, dim3(2048), dim3(256), 0,
                           0, dist, p0, n, seed, d);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(&h, d, sizeof(MomentAcc), hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d));
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    out7[0] = h.n;
    out7[1] = h.s1;
    out7[2] = h.s2;
    out7[3] = h.s3;
    out7[4] = h.s4;
    out7[5] = h.mn;
    out7[6] = h.mx;
    return 0;
}

}  // extern "C"
