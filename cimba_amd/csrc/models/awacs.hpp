// AWACS radar surveillance model — BASELINE.json config 5 ("AWACS
// 1000-target radar model (reference tutorial/tut_5_1.c family), per-target
// RCS/beamforming compute as MFMA batched GEMM").
//
// Reference architecture: trials on CPU threads call
// CUDA kernels per dwell through pinned-memory round trips
// (tut_5_3.c:859-887, tut_5_3.cu:1473 sensor_gpu_step).  MI355X-native
// inversion: the trial ALREADY runs on the GPU, so dwell physics is a
// wave-parallel phase of the same kernel — no host round trip, no streams:
// lane 0 drives the event loop; at each dwell the radar process yields a
// physics request and all 64 lanes compute the dwell (kinematics + phased-
// array beamforming) before the event loop resumes.  Targets are SoA data
// in the trial's globals (the reference's 1000 target coroutines become
// maneuver EVENTS + per-dwell batch physics — the idiomatic mapping when
// 1000 x 72 B process records will not fit LDS).
//
// Physics: ELEM-element uniform linear array, BEAMS fixed beams.  Per
// dwell, per target: steering response a_e = exp(i*pi*e*sin(az)); received
// power at beam b: |sum_e a_e * conj(w_eb)|^2; detection draw vs a
// range^-4 SNR model.  On gfx950 the [64 targets x ELEM] x [ELEM x BEAMS]
// complex products run on MFMA (mfma_f32_16x16x4f32, exact f32 — see
// hip/awacs_kernel.hip); the host path computes the identical f32 math
// scalar, which is the numerics reference for the device test.
#pragma once

#include "../include/cimba/engine.hpp"

#include <math.h>

// Dwell macro: on the host, run the physics inline (scalar); on the
// device, set the physics request and yield to the kernel's wave-parallel
// physics phase (awacs_kernel.hip resumes the process afterwards).
#if defined(__HIP_DEVICE_COMPILE__)
#define CMB_AWACS_DWELL()                 \
    do {                                  \
        E.globals.phys_request = 1;       \
        self->pc = (int16_t)__LINE__;     \
        return;                           \
        case __LINE__:;                   \
    } while (0)
#else
#define CMB_AWACS_DWELL() cmb_models::AWACS::physics_all(E)
#endif

namespace cmb_models {

struct AWACS : cmb::ModelBase {
    static constexpr int MAX_T = 1024;   // target capacity
    static constexpr int ELEM = 16;      // array elements
    static constexpr int BEAMS = 16;     // fixed beams

    struct Cfg {
        static constexpr int MAX_PROC = 2;   // radar + spare
        static constexpr int MAX_EV = 1536;  // ~1 maneuver event per target
        static constexpr int TIMERS = 1;
        static constexpr int NUM_QUEUES = 0;
        static constexpr int QCAP = 1;
        static constexpr int NUM_RES = 0;
        static constexpr int NUM_POOLS = 0;
        static constexpr int NUM_BUFS = 0;
        static constexpr int NUM_PQ = 0;
        static constexpr int PQCAP = 1;
        static constexpr int NUM_COND = 0;
    };

    struct Params {
        double duration;       // sim seconds
        double dwell;          // dwell period (reference: 0.04 s)
        double maneuver_mean;  // mean time between target maneuvers
        int32_t ntargets;
        int32_t pad_;
        double area;           // half-width of the surveillance box [m]
        double speed;          // target speed scale [m/s]
        double snr_ref;        // SNR at reference range
    };

    struct Result {
        uint64_t detections;
        uint64_t dwells;
        uint64_t maneuvers;
        uint64_t events;
        double sum_power;  // accumulated best-beam power (diagnostic)
        int32_t status;
        int32_t pad_;
    };

    struct Frame {
        uint64_t n;
    };

    enum : uint16_t { EV_MANEUVER = cmb::EV_USER + 1 };

    struct Globals {
        // target SoA (f32 for the MFMA path)
        float x[MAX_T], y[MAX_T], vx[MAX_T], vy[MAX_T], rcs[MAX_T];
        uint32_t det_cnt[MAX_T];
        // beam steering weights w[e][b], complex
        float wr[ELEM][BEAMS], wi[ELEM][BEAMS];
        int32_t nt;
        int32_t phys_request;  // device: radar yielded for a physics phase
        uint64_t detections;
        uint64_t dwells;
        uint64_t maneuvers;
        double sum_power;
        double last_t;  // time of previous dwell (kinematics dt)
    };

    enum Func : uint8_t { F_RADAR = 0 };

    // counter-based parallel-safe uniform for per-target detection draws
    // (identical on host and device; the trial's sfc64 stream stays with
    // the lane-0 event loop)
    CMB_FORCEINLINE static double draw_u01(uint64_t seed, uint64_t dwell,
                                           uint32_t t) {
        const uint64_t h =
            cmb::fmix64(seed ^ (dwell * UINT64_C(0x9E3779B97F4A7C15)) ^
                        ((uint64_t)t << 40));
        return (double)(h >> 11) * 0x1.0p-53;
    }

    // scalar single-target dwell physics (host path AND the numerics
    // reference for the device MFMA path) — f32 math to match MFMA
    CMB_FORCEINLINE static float target_power(const Globals& g, int t) {
        const float r2 = g.x[t] * g.x[t] + g.y[t] * g.y[t] + 1.0f;
        const float az = atan2f(g.y[t], g.x[t]);
        const float s = sinf(az);
        float ar[ELEM], ai[ELEM];
        for (int e = 0; e < ELEM; ++e) {
            const float ph = 3.14159265358979f * (float)e * s;
            ar[e] = cosf(ph);
            ai[e] = sinf(ph);
        }
        float best = 0.0f;
        for (int b = 0; b < BEAMS; ++b) {
            float re = 0.0f, im = 0.0f;
            for (int e = 0; e < ELEM; ++e) {
                // a * conj(w) summed over elements
                re += ar[e] * g.wr[e][b] + ai[e] * g.wi[e][b];
                im += ai[e] * g.wr[e][b] - ar[e] * g.wi[e][b];
            }
            const float p = re * re + im * im;
            best = p > best ? p : best;
        }
        // normalized beamforming gain x RCS / r^4 path loss
        return best * g.rcs[t] / (r2 * r2);
    }

    template <class E_>
    CMB_FORCEINLINE static void advance_target(E_& E, int t, float dt) {
        Globals& g = E.globals;
        g.x[t] += g.vx[t] * dt;
        g.y[t] += g.vy[t] * dt;
        // toroidal wrap inside the surveillance box (reference kinematics
        // wrap, tut_5_2.cu:612-670)
        const float a = (float)E.params->area;
        if (g.x[t] > a) g.x[t] -= 2.0f * a;
        if (g.x[t] < -a) g.x[t] += 2.0f * a;
        if (g.y[t] > a) g.y[t] -= 2.0f * a;
        if (g.y[t] < -a) g.y[t] += 2.0f * a;
    }

    // one target's detection draw for this dwell (shared by the scalar
    // host path and the wave-parallel device kernel; counter-based so any
    // lane can evaluate any target race-free and host==device)
    CMB_FORCEINLINE static bool detect_draw(uint32_t trial, uint64_t dwell,
                                            uint32_t t, float power,
                                            double snr_ref) {
        const double snr = (double)power * snr_ref;
        const double pd = snr / (1.0 + snr);  // soft detection curve
        return draw_u01(trial, dwell, t) < pd;
    }

    // host-path dwell: scalar over all targets (device: awacs_kernel.hip
    // runs the same math wave-parallel with MFMA beamforming)
    template <class E_>
    CMB_FORCEINLINE static void physics_all(E_& E) {
        Globals& g = E.globals;
        const float dt = (float)(E.now - g.last_t);
        g.last_t = E.now;
        for (int t = 0; t < g.nt; ++t) {
            advance_target(E, t, dt);
            const float p = target_power(g, t);
            if (detect_draw(E.trial_index, g.dwells, (uint32_t)t, p,
                            E.params->snr_ref)) {
                g.det_cnt[t] += 1u;
                g.detections += 1u;
            }
            g.sum_power += (double)p;
        }
        g.dwells += 1u;
    }

    template <class E_>
    CMB_FORCEINLINE static void body(E_& E, typename E_::ProcT* self) {
        const Params& P = *E.params;
        Frame& f = E.frames[0];
        CMB_BEGIN();
        for (f.n = 0; E.now < P.duration; ++f.n) {
            CMB_AWACS_DWELL();  // physics phase (see macro below)
            CMB_HOLD(P.dwell);
        }
        CMB_END();
    }

    template <class E_>
    CMB_FORCEINLINE static void step(E_& E, int pidx) {
        body(E, &E.procs[pidx]);
    }

    template <class E_>
    CMB_FORCEINLINE static void on_event(E_& E, const cmb::EvEntry& ev) {
        if (ev.kind != EV_MANEUVER) return;
        Globals& g = E.globals;
        const int t = (int)ev.a;
        // re-randomize heading (the reference's per-target maneuver logic,
        // driven here by DES events instead of 1000 coroutines)
        const double ang = E.rng.uniform(0.0, 2.0 * M_PI);
        const float sp = (float)(E.params->speed * (0.5 + E.rng.u01()));
        g.vx[t] = sp * (float)cos(ang);
        g.vy[t] = sp * (float)sin(ang);
        g.maneuvers += 1u;
        if (E.now < E.params->duration) {
            E.schedule(EV_MANEUVER, (uint16_t)t, 0, 0,
                       E.now + E.rng.exponential(E.params->maneuver_mean), 0);
        }
    }

    template <class E_>
    CMB_FORCEINLINE static void setup(E_& E) {
        const Params& P = *E.params;
        Globals& g = E.globals;
        g.nt = P.ntargets < MAX_T ? P.ntargets : MAX_T;
        g.phys_request = 0;
        g.detections = 0;
        g.dwells = 0;
        g.maneuvers = 0;
        g.sum_power = 0.0;
        g.last_t = 0.0;
        // beams uniformly over sin-space; w[e][b] = exp(i*pi*e*sin_b)
        for (int b = 0; b < BEAMS; ++b) {
            const float sb = -0.9375f + 0.125f * (float)b;  // 16 beams
            for (int e = 0; e < ELEM; ++e) {
                const float ph = 3.14159265358979f * (float)e * sb;
                g.wr[e][b] = cosf(ph);
                g.wi[e][b] = sinf(ph);
            }
        }
        for (int t = 0; t < g.nt; ++t) {
            g.x[t] = (float)E.rng.uniform(-P.area, P.area);
            g.y[t] = (float)E.rng.uniform(-P.area, P.area);
            const double ang = E.rng.uniform(0.0, 2.0 * M_PI);
            const float sp = (float)(P.speed * (0.5 + E.rng.u01()));
            g.vx[t] = sp * (float)cos(ang);
            g.vy[t] = sp * (float)sin(ang);
            g.rcs[t] = (float)E.rng.exponential(1.0);  // Swerling-1 RCS
            g.det_cnt[t] = 0;
            E.schedule(EV_MANEUVER, (uint16_t)t, 0, 0,
                       E.rng.exponential(P.maneuver_mean), 0);
        }
        E.proc_init(0, F_RADAR, 10);
        E.proc_start(0);
    }

    template <class E_>
    CMB_FORCEINLINE static void finish(E_& E, Result& r) {
        r.detections = E.globals.detections;
        r.dwells = E.globals.dwells;
        r.maneuvers = E.globals.maneuvers;
        r.events = E.ev_dispatched;
        r.sum_power = E.globals.sum_power;
        r.status = E.status;
    }
};

}  // namespace cmb_models
