// Synthetic terrain + line-of-sight masking — the MI355X equivalent of the
// reference's AWACS terrain stack (reference tutorial/tut_5_2.cu:107-118
// terrain_generate_kernel, :171-178 terrain_stats_kernel, :557-580
// prime_altitudes_kernel, :1304 raymarch_kernel LOS masking).
//
// Own design, not a port: the reference builds ridged Perlin noise with a
// __constant__ permutation table into a CUDA texture object and ray-marches
// with per-warp cooperation.  Here the lattice hash is fmix64 (no tables to
// stage), the heightmap is a plain float buffer with an explicit bilinear
// sampler (gfx950 has no benefit from the texture path for this access
// pattern), and the LOS march is wave64-cooperative (terrain_kernel.hip).
// Every function below is CMB_FORCEINLINE host+device: the host build IS
// the numerics reference for the GPU kernels (tests/test_terrain.py).
#pragma once

#include "config.hpp"
#include "rng.hpp"  // fmix64

#include <cstdint>

namespace cmb {

struct TerrainDesc {
    int32_t cols, rows;
    float x0, y0;     // world coords of texel (0,0)
    float dx, dy;     // world units per texel
    float base, amp;  // height = base + amp * ridged_fbm
    int32_t octaves;
    uint64_t seed;
};

// lattice hash -> [-1, 1)
CMB_FORCEINLINE float th_lattice(int32_t ix, int32_t iy, uint64_t seed) {
    const uint64_t h = fmix64(seed ^ (uint64_t)(uint32_t)ix ^
                              ((uint64_t)(uint32_t)iy << 32));
    return (float)((int64_t)(h >> 11) - (int64_t)(1ull << 52)) *
           (1.0f / (float)(1ull << 52));
}

CMB_FORCEINLINE float th_fade(float t) {
    // quintic smoothstep: C2-continuous across lattice cells
    return t * t * t * (t * (t * 6.0f - 15.0f) + 10.0f);
}

// value noise at lattice scale 1
CMB_FORCEINLINE float th_vnoise(float x, float y, uint64_t seed) {
    const float fx = floorf(x), fy = floorf(y);
    const int32_t ix = (int32_t)fx, iy = (int32_t)fy;
    const float tx = th_fade(x - fx), ty = th_fade(y - fy);
    const float v00 = th_lattice(ix, iy, seed);
    const float v10 = th_lattice(ix + 1, iy, seed);
    const float v01 = th_lattice(ix, iy + 1, seed);
    const float v11 = th_lattice(ix + 1, iy + 1, seed);
    const float a = v00 + (v10 - v00) * tx;
    const float b = v01 + (v11 - v01) * tx;
    return a + (b - a) * ty;
}

// ridged fractional Brownian motion in [0, ~2]; x,y in texel units
CMB_FORCEINLINE float th_ridged_fbm(float x, float y, int octaves,
                                    uint64_t seed) {
    float sum = 0.0f, amp = 1.0f, norm = 0.0f;
    float fx = x * (1.0f / 64.0f), fy = y * (1.0f / 64.0f);
    for (int o = 0; o < octaves; ++o) {
        const float n = th_vnoise(fx, fy, seed + (uint64_t)o * 0x9E3779B9u);
        sum += amp * (1.0f - fabsf(n));  // ridge: fold around zero
        norm += amp;
        amp *= 0.5f;
        fx *= 2.013f;  // slightly irrational lacunarity breaks lattice echo
        fy *= 2.013f;
    }
    return sum / norm;
}

CMB_FORCEINLINE float th_texel_height(const TerrainDesc& T, int32_t c,
                                      int32_t r) {
    return T.base + T.amp * th_ridged_fbm((float)c, (float)r, T.octaves,
                                          T.seed);
}

// bilinear sample of a built heightmap at world (x, y), edge-clamped
CMB_FORCEINLINE float th_sample(const float* __restrict__ h,
                                const TerrainDesc& T, float x, float y) {
    float cx = (x - T.x0) / T.dx, cy = (y - T.y0) / T.dy;
    cx = cx < 0.0f ? 0.0f : cx;
    cy = cy < 0.0f ? 0.0f : cy;
    const float mx = (float)(T.cols - 1), my = (float)(T.rows - 1);
    cx = cx > mx ? mx : cx;
    cy = cy > my ? my : cy;
    const float fx = floorf(cx), fy = floorf(cy);
    int32_t c0 = (int32_t)fx, r0 = (int32_t)fy;
    const int32_t c1 = c0 + 1 < T.cols ? c0 + 1 : c0;
    const int32_t r1 = r0 + 1 < T.rows ? r0 + 1 : r0;
    const float tx = cx - fx, ty = cy - fy;
    const float v00 = h[(size_t)r0 * T.cols + c0];
    const float v10 = h[(size_t)r0 * T.cols + c1];
    const float v01 = h[(size_t)r1 * T.cols + c0];
    const float v11 = h[(size_t)r1 * T.cols + c1];
    const float a = v00 + (v10 - v00) * tx;
    const float b = v01 + (v11 - v01) * tx;
    return a + (b - a) * ty;
}

// Is sample index k of nsteps along the ray (terrain above the ray at the
// interior sample point)?  Exposed per-sample so the device LOS kernel can
// stride samples across the 64 lanes of a wave; the host loops k serially.
CMB_FORCEINLINE bool th_los_blocked_at(const float* __restrict__ h,
                                       const TerrainDesc& T, float x0,
                                       float y0, float z0, float x1, float y1,
                                       float z1, int nsteps, int k) {
    const float t = (float)(k + 1) / (float)(nsteps + 1);  // interior only
    const float x = x0 + (x1 - x0) * t;
    const float y = y0 + (y1 - y0) * t;
    const float z = z0 + (z1 - z0) * t;
    return th_sample(h, T, x, y) > z;
}

CMB_FORCEINLINE bool th_los_clear(const float* __restrict__ h,
                                  const TerrainDesc& T, float x0, float y0,
                                  float z0, float x1, float y1, float z1,
                                  int nsteps) {
    for (int k = 0; k < nsteps; ++k)
        if (th_los_blocked_at(h, T, x0, y0, z0, x1, y1, z1, nsteps, k))
            return false;
    return true;
}

}  // namespace cmb
