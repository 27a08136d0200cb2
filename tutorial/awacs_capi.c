/*
 * CPU AWACS radar surveillance on the cmb_* C API — the capability
 * counterpart of the reference's tutorial/tut_5_1.c (CPU-only AWACS with
 * one coroutine per target), written as protothread process bodies with
 * the explicit sim context (porting notes in docs/PARITY.md).
 *
 * One PROCESS PER TARGET (the reference's structure — unlike the
 * GPU-resident model in cimba_amd/csrc/models/awacs.hpp, which turns
 * maneuvers into events): each target holds exponential maneuver times
 * and re-randomizes its heading; a radar process wakes every dwell,
 * advances the beam, and runs detection draws on the targets inside the
 * beam; a watch process CONDITION-waits until the radar has confirmed
 * `TRACK_GOAL` distinct targets, then records the time-to-goal.  With
 * 1000 target processes this exercises the host engine's process table,
 * timer discipline and condition machinery at the reference tutorial's
 * scale.
 *
 * Build:
 *   gcc -std=c11 -Iinclude tutorial/awacs_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -lm -o awacs_capi
 */
#include <cimba.h>

#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#ifndef M_PI
#define M_PI 3.14159265358979323846
#endif

#define N_TARGETS 1000u
#define N_TRIALS 8u
#define TRACK_GOAL 200u

static const double duration_s = 600.0;
static const double dwell_s = 0.04;
static const double maneuver_mean_s = 5.0;
static const double area_m = 50000.0;
static const double speed_m_s = 250.0;
static const double rot_rate = 0.6283185307179586; /* 6 RPM */
static const double beamwidth = 0.0262;
static const double snr_ref = 2.0e15;

struct trial {
    uint64_t seed;
    uint64_t detections;
    uint64_t dwells;
    uint64_t tracked;     /* distinct targets ever detected */
    double t_goal;        /* sim time when TRACK_GOAL reached, -1 if never */
};

struct target {
    double x, y, vx, vy, rcs;
    uint8_t tracked;
};

struct world {
    cmb_sim* sim;
    struct target tgt[N_TARGETS];
    cmb_condition* tracks_cond;
    uint64_t detections;
    uint64_t dwells;
    uint64_t tracked;
    double t_goal;
};

static _Thread_local struct world g_world;

static void target_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct target* t = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        cmb_process_hold(sim, me,
                         cmb_random_exponential(sim, maneuver_mean_s));
        if (cmb_time(sim) >= duration_s) break;
        {
            const double ang = cmb_random_uniform(sim, 0.0, 2.0 * M_PI);
            const double sp =
                speed_m_s * (0.5 + cmb_random_uniform(sim, 0.0, 1.0));
            t->vx = sp * cos(ang);
            t->vy = sp * sin(ang);
        }
    }
    CMB_PROC_END(sim, me);
}

static void radar_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct world* w = vctx;
    CMB_PROC_BEGIN(sim, me);
    while (cmb_time(sim) < duration_s) {
        cmb_process_hold(sim, me, dwell_s);
        {
            const double bdir = fmod(rot_rate * cmb_time(sim), 2.0 * M_PI);
            const double gate = 0.5 * (beamwidth + rot_rate * dwell_s);
            for (unsigned i = 0; i < N_TARGETS; ++i) {
                struct target* t = &w->tgt[i];
                /* piecewise-linear kinematics, advanced at observation */
                t->x += t->vx * dwell_s;
                t->y += t->vy * dwell_s;
                if (t->x > area_m) t->x -= 2.0 * area_m;
                if (t->x < -area_m) t->x += 2.0 * area_m;
                if (t->y > area_m) t->y -= 2.0 * area_m;
                if (t->y < -area_m) t->y += 2.0 * area_m;
                double d = atan2(t->y, t->x) - bdir;
                while (d > M_PI) d -= 2.0 * M_PI;
                while (d < -M_PI) d += 2.0 * M_PI;
                if (fabs(d) > gate) continue;
                {
                    const double r2 = t->x * t->x + t->y * t->y + 1.0;
                    const double snr = snr_ref * t->rcs / (r2 * r2);
                    const double pd = snr / (1.0 + snr);
                    if (cmb_random_uniform(sim, 0.0, 1.0) < pd) {
                        w->detections += 1u;
                        if (!t->tracked) {
                            t->tracked = 1u;
                            w->tracked += 1u;
                            /* a new track may satisfy the watcher */
                            cmb_condition_signal(sim, w->tracks_cond);
                        }
                    }
                }
            }
            w->dwells += 1u;
        }
    }
    CMB_PROC_END(sim, me);
}

static bool goal_reached(cmb_sim* sim, void* vctx) {
    (void)sim;
    return ((struct world*)vctx)->tracked >= TRACK_GOAL;
}

static void watch_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct world* w = vctx;
    CMB_PROC_BEGIN(sim, me);
    while (!goal_reached(sim, w)) {
        cmb_condition_wait(sim, me, w->tracks_cond, goal_reached, w);
        if (CMB_SIGNAL(sim, me) != CMB_PROCESS_SUCCESS) {
            cmb_proc_finish_(sim, me);
            return;
        }
    }
    w->t_goal = cmb_time(sim);
    CMB_PROC_END(sim, me);
}

static void trial_fn(cmb_sim* sim, void* exp_slot) {
    struct trial* trl = exp_slot;
    struct world* w = &g_world;
    memset(w, 0, sizeof *w);
    w->sim = sim;
    w->t_goal = -1.0;
    w->tracks_cond = cmb_condition_create(sim);
    cmb_condition_initialize(sim, w->tracks_cond, "tracks");

    for (unsigned i = 0; i < N_TARGETS; ++i) {
        struct target* t = &w->tgt[i];
        t->x = cmb_random_uniform(sim, -area_m, area_m);
        t->y = cmb_random_uniform(sim, -area_m, area_m);
        const double ang = cmb_random_uniform(sim, 0.0, 2.0 * M_PI);
        const double sp =
            speed_m_s * (0.5 + cmb_random_uniform(sim, 0.0, 1.0));
        t->vx = sp * cos(ang);
        t->vy = sp * sin(ang);
        t->rcs = cmb_random_exponential(sim, 1.0);
        char name[32];
        snprintf(name, sizeof name, "tgt-%u", i);
        cmb_process* p = cmb_process_spawn(sim, name, target_proc, t, 0);
        cmb_process_start(sim, p);
    }
    cmb_process* radar = cmb_process_spawn(sim, "radar", radar_proc, w, 10);
    cmb_process* watch = cmb_process_spawn(sim, "watch", watch_proc, w, 5);
    cmb_process_start(sim, radar);
    cmb_process_start(sim, watch);

    cmb_event_queue_execute(sim);

    trl->detections = w->detections;
    trl->dwells = w->dwells;
    trl->tracked = w->tracked;
    trl->t_goal = w->t_goal;
}

int main(int argc, char** argv) {
    const uint64_t master = argc > 1 ? strtoull(argv[1], NULL, 0)
                                     : 0x34f05c64d7ad598full;
    struct trial exp[N_TRIALS];
    memset(exp, 0, sizeof exp);
    const uint64_t failed =
        cimba_run(exp, N_TRIALS, sizeof(struct trial), trial_fn, master, 0);
    double tsum = 0.0;
    uint64_t det = 0;
    unsigned goal_ok = 0;
    for (unsigned i = 0; i < N_TRIALS; ++i) {
        det += exp[i].detections;
        if (exp[i].t_goal >= 0.0) {
            tsum += exp[i].t_goal;
            ++goal_ok;
        }
        printf("trial %u: %llu detections, %llu tracks, goal at %.1f s\n", i,
               (unsigned long long)exp[i].detections,
               (unsigned long long)exp[i].tracked, exp[i].t_goal);
    }
    printf("%u trials (%llu failed): %llu detections total; "
           "%u reached the %u-track goal, mean t=%.1f s\n",
           N_TRIALS, (unsigned long long)failed, (unsigned long long)det,
           goal_ok, TRACK_GOAL, goal_ok ? tsum / goal_ok : -1.0);
    return failed != 0;
}
