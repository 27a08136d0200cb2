// Terrain kernels for gfx950 — the MI355X counterpart of the reference's
// AWACS terrain stack (reference tutorial/tut_5_2.cu terrain_generate_kernel
// :107-118, terrain_stats_kernel :171-178, prime_altitudes_kernel :557-580,
// raymarch_kernel LOS masking :1304; see also tut_5_3.cu multi-GPU variants).
//
// Design differences from the reference (cimba/terrain.hpp header comment):
// fmix64 lattice hash instead of a __constant__ permutation table, plain
// HBM float buffer + explicit bilinear instead of texture objects, and the
// LOS ray-march is WAVE-cooperative: the reference assigns one 32-thread
// warp per queued target; here one 64-lane wave handles one query with the
// march samples strided across lanes and a ballot-style any() reduce — a
// single round of __any keeps the march O(nsteps/64) per query.
//
// The heightmap stays resident on the GPU across calls (the reference keeps
// its ~14 GB terrain on-device, tut_5_2.c:12); handles are opaque pointers.
#include "cimba/terrain.hpp"

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define HIP_TRY(x)                                                  \
    do {                                                            \
        hipError_t err_ = (x);                                      \
        if (err_ != hipSuccess) {                                   \
            std::fprintf(stderr, "HIP error %s at %s:%d\n",         \
                         hipGetErrorString(err_), __FILE__, __LINE__); \
            return -1;                                              \
        }                                                           \
    } while (0)

namespace {

using cmb::TerrainDesc;

__global__ __launch_bounds__(256) void terrain_generate_kernel(
    float* __restrict__ h, TerrainDesc T) {
    const size_t n = (size_t)T.cols * T.rows;
    const size_t stride = (size_t)gridDim.x * blockDim.x;
    for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        h[i] = cmb::th_texel_height(T, (int32_t)(i % T.cols),
                                    (int32_t)(i / T.cols));
}

struct TerrainStats {
    double s1, s2, mn, mx;
};

__global__ __launch_bounds__(256) void terrain_stats_kernel(
    const float* __restrict__ h, size_t n, TerrainStats* __restrict__ out) {
    const size_t stride = (size_t)gridDim.x * blockDim.x;
    double s1 = 0, s2 = 0, mn = 1e308, mx = -1e308;
    for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        const double x = (double)h[i];
        s1 += x;
        s2 += x * x;
        mn = x < mn ? x : mn;
        mx = x > mx ? x : mx;
    }
    for (int off = 32; off > 0; off >>= 1) {
        s1 += __shfl_down(s1, off);
        s2 += __shfl_down(s2, off);
        const double omn = __shfl_down(mn, off);
        const double omx = __shfl_down(mx, off);
        mn = omn < mn ? omn : mn;
        mx = omx > mx ? omx : mx;
    }
    if ((threadIdx.x & 63) == 0) {
        atomicAdd(&out->s1, s1);
        atomicAdd(&out->s2, s2);
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

// one thread per query point: bilinear height sample (the reference's
// prime_altitudes_kernel shape)
__global__ __launch_bounds__(256) void terrain_sample_kernel(
    const float* __restrict__ h, TerrainDesc T, const float* __restrict__ xs,
    const float* __restrict__ ys, float* __restrict__ out, uint32_t n) {
    const uint32_t stride = gridDim.x * blockDim.x;
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride)
        out[i] = cmb::th_sample(h, T, xs[i], ys[i]);
}

// one WAVE per LOS query: lanes take march samples k, k+64, k+128, ...
// and a single __any() resolves the query (wave64 replacement for the
// reference's warp-per-target raymarch loop)
__global__ __launch_bounds__(256) void terrain_los_kernel(
    const float* __restrict__ h, TerrainDesc T,
    const float* __restrict__ q,  // 6 floats per query: x0 y0 z0 x1 y1 z1
    uint8_t* __restrict__ vis, uint32_t nq, int nsteps) {
    const uint32_t lane = threadIdx.x & 63u;
    const uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    const uint32_t nwaves = (gridDim.x * blockDim.x) >> 6;
    for (uint32_t i = wave; i < nq; i += nwaves) {
        const float x0 = q[i * 6 + 0], y0 = q[i * 6 + 1], z0 = q[i * 6 + 2];
        const float x1 = q[i * 6 + 3], y1 = q[i * 6 + 4], z1 = q[i * 6 + 5];
        bool blocked = false;
        for (int k = (int)lane; k < nsteps && !blocked; k += 64)
            blocked = cmb::th_los_blocked_at(h, T, x0, y0, z0, x1, y1, z1,
                                             nsteps, k);
        const bool any_blocked = __any(blocked);
        if (lane == 0) vis[i] = any_blocked ? 0 : 1;
    }
}

uint32_t grid_for(size_t n) {
    size_t want = (n + 255) / 256;
    // >= 8 workgroups per XCD even for small inputs; cap well above 256 CUs
    if (want < 64) want = 64;
    if (want > 16384) want = 16384;
    return (uint32_t)want;
}

}  // namespace

extern "C" {

// builds the heightmap on `device`, returns an opaque handle (device ptr)
int cimba_terrain_gpu_build(int cols, int rows, double base, double amp,
                            int octaves, uint64_t seed, int device,
                            void** handle_out) {
    HIP_TRY(hipSetDevice(device));
    const size_t n = (size_t)cols * rows;
    float* d_h = nullptr;
    HIP_TRY(hipMalloc(&d_h, n * sizeof(float)));
    TerrainDesc T{cols, rows, 0.0f, 0.0f, 1.0f, 1.0f,
                  (float)base, (float)amp, octaves, seed};
    hipLaunchKernelGGL(terrain_generate_kernel, dim3(grid_for(n)), dim3(256),
                       0, 0, d_h, T);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipDeviceSynchronize());
    *handle_out = (void*)d_h;
    return 0;
}

int cimba_terrain_gpu_stats(void* handle, int cols, int rows,
                            double out4[4]) {
    const size_t n = (size_t)cols * rows;
    TerrainStats* d_s = nullptr;
    HIP_TRY(hipMalloc(&d_s, sizeof(TerrainStats)));
    const TerrainStats init{0.0, 0.0, 1e308, -1e308};
    HIP_TRY(hipMemcpy(d_s, &init, sizeof(init), hipMemcpyHostToDevice));
    hipLaunchKernelGGL(terrain_stats_kernel, dim3(grid_for(n)), dim3(256), 0,
                       0, (const float*)handle, n, d_s);
    HIP_TRY(hipGetLastError());
    TerrainStats s;
    HIP_TRY(hipMemcpy(&s, d_s, sizeof(s), hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_s));
    const double mean = s.s1 / (double)n;
    out4[0] = mean;
    out4[1] = s.s2 / (double)n - mean * mean;  // population variance
    out4[2] = s.mn;
    out4[3] = s.mx;
    return 0;
}

int cimba_terrain_gpu_sample(void* handle, int cols, int rows, double base,
                             double amp, int octaves, uint64_t seed,
                             const float* xs, const float* ys, float* out,
                             uint32_t n) {
    TerrainDesc T{cols, rows, 0.0f, 0.0f, 1.0f, 1.0f,
                  (float)base, (float)amp, octaves, seed};
    float *d_x = nullptr, *d_y = nullptr, *d_o = nullptr;
    HIP_TRY(hipMalloc(&d_x, n * sizeof(float)));
    HIP_TRY(hipMalloc(&d_y, n * sizeof(float)));
    HIP_TRY(hipMalloc(&d_o, n * sizeof(float)));
    HIP_TRY(hipMemcpy(d_x, xs, n * sizeof(float), hipMemcpyHostToDevice));
    HIP_TRY(hipMemcpy(d_y, ys, n * sizeof(float), hipMemcpyHostToDevice));
    hipLaunchKernelGGL(terrain_sample_kernel, dim3(grid_for(n)), dim3(256),
                       0, 0, (const float*)handle, T, d_x, d_y, d_o, n);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipMemcpy(out, d_o, n * sizeof(float), hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_x));
    HIP_TRY(hipFree(d_y));
    HIP_TRY(hipFree(d_o));
    return 0;
}

int cimba_terrain_gpu_los(void* handle, int cols, int rows, double base,
                          double amp, int octaves, uint64_t seed,
                          const float* queries, uint8_t* vis, uint32_t nq,
                          int nsteps, double* elapsed_ms) {
    TerrainDesc T{cols, rows, 0.0f, 0.0f, 1.0f, 1.0f,
                  (float)base, (float)amp, octaves, seed};
    float* d_q = nullptr;
    uint8_t* d_v = nullptr;
    HIP_TRY(hipMalloc(&d_q, (size_t)nq * 6 * sizeof(float)));
    HIP_TRY(hipMalloc(&d_v, nq));
    HIP_TRY(hipMemcpy(d_q, queries, (size_t)nq * 6 * sizeof(float),
                      hipMemcpyHostToDevice));
    hipEvent_t t0, t1;
    HIP_TRY(hipEventCreate(&t0));
    HIP_TRY(hipEventCreate(&t1));
    HIP_TRY(hipEventRecord(t0));
    hipLaunchKernelGGL(terrain_los_kernel, dim3(grid_for((size_t)nq * 64)),
                       dim3(256), 0, 0, (const float*)handle, T, d_q, d_v,
                       nq, nsteps);
    HIP_TRY(hipGetLastError());
    HIP_TRY(hipEventRecord(t1));
    HIP_TRY(hipEventSynchronize(t1));
    float ms = 0.f;
    HIP_TRY(hipEventElapsedTime(&ms, t0, t1));
    if (elapsed_ms) *elapsed_ms = (double)ms;
    HIP_TRY(hipMemcpy(vis, d_v, nq, hipMemcpyDeviceToHost));
    HIP_TRY(hipFree(d_q));
    HIP_TRY(hipFree(d_v));
    HIP_TRY(hipEventDestroy(t0));
    HIP_TRY(hipEventDestroy(t1));
    return 0;
}

int cimba_terrain_gpu_free(void* handle) {
    HIP_TRY(hipFree(handle));
    return 0;
}

}  // extern "C"
