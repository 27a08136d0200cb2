/*
 * Harbor simulation with abandoned trials — the capability counterpart of
 * the reference's tutorial/tut_4_3.c on the cimba-mi355x C API (explicit
 * sim context + protothread process bodies; porting notes in
 * docs/PARITY.md).
 *
 * The model: ships (SMALL/LARGE) arrive at a tidal harbor.  A ship may
 * dock only when the harbormaster CONDITION holds: enough water under its
 * keel (tide + dredged depth), a berth of its size class free, and enough
 * tugboats idle.  Docking and undocking need a multi-unit acquire from the
 * shared tug POOL plus a radio slot on the comms RESOURCE; unloading holds
 * a PERT-distributed time at the berth.  A weather process drives the wind
 * (slows tug operations); a tide process drives the water level and
 * signals the harbormaster on every rise.  With small probability a ship
 * hits a simulated data error and ABANDONS THE TRIAL through
 * cmb_logger_error — exercising the executive's longjmp-equivalent
 * recovery, the user trial-cleanup hook, and the failed-trial count that
 * cimba_run returns (reference cimba.c:289-329 semantics).
 *
 * Experiment: 3 arrival-rate scenarios x N_REPS replications, run
 * multithreaded by cimba_run; per-trial outputs are the mean time in
 * harbor, ships served, and the tug/berth utilization reports.
 *
 * Build:
 *   gcc -std=c11 -Iinclude tutorial/harbor_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -lm -o harbor_capi
 */
#include <cimba.h>

#define _USE_MATH_DEFINES
#include <math.h>
#ifndef M_PI
#define M_PI 3.14159265358979323846
#endif
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#define USERFLAG_SHIP 0x00000001u

#define N_SCENARIOS 3u
#define N_REPS 10u
#define MAX_SHIPS_ACTIVE 256u

static const double arrival_rate[N_SCENARIOS] = {0.5, 0.55, 0.625};
static const double percent_large = 0.25;
static const double base_depth = 15.5;   /* dredged depth at zero tide */
static const double tide_amp = 1.2;      /* tidal amplitude */
static const double tide_period = 12.4;  /* hours */
static const unsigned num_tugs = 10u;
static const unsigned num_berths[2] = {6u, 3u}; /* SMALL, LARGE */
static const double unloading_avg[2] = {8.0, 12.0};
static const double draft[2] = {14.2, 16.1};
static const unsigned tugs_needed[2] = {1u, 3u};
static const double abandon_prob = 2e-4;
static const double duration_h = 24.0 * 30;

enum ship_size { SMALL = 0, LARGE = 1 };

struct trial {
    double lambda;     /* in: arrivals per hour */
    double avg_in_harbor;
    double tug_util;
    uint64_t served;
    uint64_t arrived;
};

struct ship;

/* per-trial world, allocated on the trial stack */
struct world {
    cmb_sim* sim;
    struct trial* trl;
    cmb_resourcepool* tugs;
    cmb_resourcepool* berths[2];
    cmb_resource* comms;
    cmb_condition* harbormaster;
    double tide;        /* current tide height above datum */
    double wind;        /* current wind speed */
    double sum_in_harbor;
    uint64_t served;
    uint64_t arrived;
    struct ship* freelist;
};

/* a ship's persistent protothread locals */
struct ship {
    struct world* w;
    cmb_process* proc;
    struct ship* next;  /* freelist link */
    int size;
    double t_arr;
    int32_t rem;        /* greedy pool-acquire scratch */
};

/* per-worker-thread slab: trials run concurrently across workers */
static _Thread_local struct ship ship_slab[MAX_SHIPS_ACTIVE];

/* dock-ready predicate for the harbormaster condition: evaluated by the
 * engine on every signal (reference cmb_condition demand contract) */
static bool is_ready_to_dock(cmb_sim* sim, void* vctx) {
    struct ship* sh = vctx;
    struct world* w = sh->w;
    (void)sim;
    if (base_depth + w->tide < draft[sh->size] + 0.3) return false;
    if (cmb_resourcepool_available(w->sim, w->berths[sh->size]) < 1) return false;
    if (cmb_resourcepool_available(w->sim, w->tugs) <
        (int32_t)tugs_needed[sh->size])
        return false;
    return true;
}

static void ship_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct ship* sh = vctx;
    struct world* w = sh->w;
    CMB_PROC_BEGIN(sim, me);

    cmb_logger_user(sim, USERFLAG_SHIP, "%s arrives",
                    cmb_process_name(sim, me));
    sh->t_arr = cmb_time(sim);

    /* loop to absorb spurious wakeups: another ship can grab the tide
     * window first (reference tut_4_3 pattern) */
    while (!is_ready_to_dock(sim, sh)) {
        cmb_condition_wait(sim, me, w->harbormaster, is_ready_to_dock, sh);
        if (CMB_SIGNAL(sim, me) != CMB_PROCESS_SUCCESS) {
            cmb_proc_finish_(sim, me);  /* early exit (stopped trial) */
            return;
        }
    }

    cmb_resourcepool_acquire_all(sim, me, w->berths[sh->size], 1);
    cmb_resourcepool_acquire(sim, me, w->tugs, (int32_t)tugs_needed[sh->size],
                             sh->rem);

    /* radio clearance, then docking (wind slows the tugs) */
    cmb_resource_acquire(sim, me, w->comms);
    cmb_process_hold(sim, me, cmb_random_gamma(sim, 5.0, 0.01));
    cmb_resource_release(sim, w->comms, me);
    cmb_process_hold(sim, me, cmb_random_pert(sim, 0.4, 0.5, 0.8) *
                                  (1.0 + 0.05 * w->wind));

    /* simulated data error: abandon the whole trial (recovery path) */
    if (cmb_random_bernoulli(sim, abandon_prob))
        cmb_logger_error(sim, "ship hit a data error, abandoning trial");

    /* docked: dismiss the tugs, unload */
    cmb_resourcepool_release(sim, w->tugs, me, (int32_t)tugs_needed[sh->size]);
    cmb_condition_signal(sim, w->harbormaster);
    cmb_process_hold(sim, me,
                     cmb_random_pert(sim, 0.75 * unloading_avg[sh->size],
                                     unloading_avg[sh->size],
                                     2.0 * unloading_avg[sh->size]));

    /* leave: tugs again, radio, undock, free everything */
    cmb_resourcepool_acquire(sim, me, w->tugs, (int32_t)tugs_needed[sh->size],
                             sh->rem);
    cmb_resource_acquire(sim, me, w->comms);
    cmb_process_hold(sim, me, cmb_random_gamma(sim, 5.0, 0.01));
    cmb_resource_release(sim, w->comms, me);
    cmb_process_hold(sim, me, cmb_random_pert(sim, 0.4, 0.5, 0.8) *
                                  (1.0 + 0.05 * w->wind));
    cmb_resourcepool_release(sim, w->berths[sh->size], me, 1);
    cmb_resourcepool_release(sim, w->tugs, me, (int32_t)tugs_needed[sh->size]);
    cmb_condition_signal(sim, w->harbormaster);

    cmb_logger_user(sim, USERFLAG_SHIP, "%s departs",
                    cmb_process_name(sim, me));
    w->sum_in_harbor += cmb_time(sim) - sh->t_arr;
    w->served += 1u;
    sh->next = w->freelist;  /* recycle the slab slot */
    w->freelist = sh;
    CMB_PROC_END(sim, me);
}

static void arrivals_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct world* w = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        cmb_process_hold(sim, me,
                         cmb_random_exponential(sim, 1.0 / w->trl->lambda));
        if (cmb_time(sim) >= duration_h) break;
        if (w->freelist == NULL) {
            cmb_logger_warning(sim, "ship slab exhausted, arrival dropped");
            continue;
        }
        {
            struct ship* sh = w->freelist;
            w->freelist = sh->next;
            sh->w = w;
            sh->size =
                cmb_random_bernoulli(sim, percent_large) ? LARGE : SMALL;
            sh->rem = 0;
            char name[32];
            snprintf(name, sizeof name, "ship-%llu",
                     (unsigned long long)w->arrived);
            sh->proc = cmb_process_spawn(sim, name, ship_proc, sh, 0);
            cmb_process_start(sim, sh->proc);
            w->arrived += 1u;
        }
    }
    CMB_PROC_END(sim, me);
}

static void tide_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct world* w = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        w->tide = tide_amp * sin(2.0 * M_PI * cmb_time(sim) / tide_period);
        /* rising water can unblock waiting ships */
        cmb_condition_signal(sim, w->harbormaster);
        if (cmb_time(sim) >= duration_h) break;
        cmb_process_hold(sim, me, 0.25);
    }
    CMB_PROC_END(sim, me);
}

static void weather_proc(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct world* w = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        /* mean-reverting wind walk, clamped at calm */
        w->wind += 0.3 * (5.0 - w->wind) +
                   cmb_random_normal(sim, 0.0, 1.0);
        if (w->wind < 0.0) w->wind = 0.0;
        if (cmb_time(sim) >= duration_h) break;
        cmb_process_hold(sim, me, 1.0);
    }
    CMB_PROC_END(sim, me);
}

static int g_cleanups = 0; /* abandoned-trial cleanup hook count */

static void trial_cleanup(uint64_t trial_idx) {
    (void)trial_idx;
    __atomic_fetch_add(&g_cleanups, 1, __ATOMIC_RELAXED);
}

static void trial_fn(cmb_sim* sim, void* exp_slot) {
    struct trial* trl = exp_slot;
    struct world w;
    memset(&w, 0, sizeof w);
    w.sim = sim;
    w.trl = trl;
    w.freelist = NULL;
    for (unsigned i = 0; i < MAX_SHIPS_ACTIVE; ++i) {
        ship_slab[i].next = w.freelist; /* per-trial recycle list */
        w.freelist = &ship_slab[i];
    }

    w.tugs = cmb_resourcepool_create(sim);
    cmb_resourcepool_initialize(sim, w.tugs, "tugs", (int32_t)num_tugs);
    cmb_resourcepool_start_recording(sim, w.tugs);
    for (int s = 0; s < 2; ++s) {
        w.berths[s] = cmb_resourcepool_create(sim);
        cmb_resourcepool_initialize(sim, w.berths[s],
                                    s == SMALL ? "berths-S" : "berths-L",
                                    (int32_t)num_berths[s]);
    }
    w.comms = cmb_resource_create(sim);
    cmb_resource_initialize(sim, w.comms, "comms");
    w.harbormaster = cmb_condition_create(sim);
    cmb_condition_initialize(sim, w.harbormaster, "harbormaster");

    cmb_process* arr = cmb_process_spawn(sim, "arrivals", arrivals_proc, &w, 0);
    cmb_process* tide = cmb_process_spawn(sim, "tide", tide_proc, &w, 5);
    cmb_process* wx = cmb_process_spawn(sim, "weather", weather_proc, &w, 5);
    cmb_process_start(sim, arr);
    cmb_process_start(sim, tide);
    cmb_process_start(sim, wx);

    cmb_event_queue_execute(sim);

    trl->served = w.served;
    trl->arrived = w.arrived;
    trl->avg_in_harbor = w.served ? w.sum_in_harbor / (double)w.served : 0.0;
    {
        double st[4]; /* time-weighted units-in-use: mean, var, min, max */
        cmb_resourcepool_stats(sim, w.tugs, st);
        trl->tug_util = st[0] / (double)num_tugs;
    }
}

int main(int argc, char** argv) {
    uint64_t master = argc > 1 ? strtoull(argv[1], NULL, 0)
                               : 0x34f05c64d7ad598full;
    struct trial exp[N_SCENARIOS * N_REPS];
    memset(exp, 0, sizeof exp);
    for (unsigned s = 0; s < N_SCENARIOS; ++s)
        for (unsigned r = 0; r < N_REPS; ++r)
            exp[s * N_REPS + r].lambda = arrival_rate[s];

    cimba_trial_cleanup_set(trial_cleanup);
    const uint64_t failed = cimba_run(exp, N_SCENARIOS * N_REPS,
                                      sizeof(struct trial), trial_fn,
                                      master, 0);

    for (unsigned s = 0; s < N_SCENARIOS; ++s) {
        double avg = 0.0, util = 0.0;
        uint64_t served = 0;
        unsigned ok = 0;
        for (unsigned r = 0; r < N_REPS; ++r) {
            const struct trial* t = &exp[s * N_REPS + r];
            if (t->served == 0) continue; /* abandoned trial */
            avg += t->avg_in_harbor;
            util += t->tug_util;
            served += t->served;
            ++ok;
        }
        printf("lambda=%.3f: %u/%u trials ok, %llu ships, "
               "avg time in harbor %.2f h, tug utilization %.1f%%\n",
               arrival_rate[s], ok, N_REPS, (unsigned long long)served,
               ok ? avg / ok : 0.0, ok ? 100.0 * util / ok : 0.0);
    }
    printf("abandoned trials: %llu (cleanup hook ran %d times)\n",
           (unsigned long long)failed, g_cleanups);
    if ((int)failed != g_cleanups) {
        fprintf(stderr, "FAIL: cleanup hook count != failed count\n");
        return 1;
    }
    return 0;
}
