/*
 * M/M/1 benchmark written against the public cmb_* C API (include/cimba.h)
 * — the C-API twin of cimba_amd/csrc/models/mm1.hpp and the counterpart of
 * the reference's benchmark/MM1_multi.c (arrival + service processes, one
 * unlimited object queue, avg system time expected 1/(mu - lambda) = 10).
 *
 * Shows the state-machine process style that replaces the reference's
 * stackful coroutines (see include/cimba.h header comment): persistent
 * locals live in the context struct, blocking calls are CMB_* macros.
 *
 * Build (after `python cimba_amd/_build.py`):
 *   gcc -std=c11 -Iinclude tutorial/mm1_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -o mm1_capi
 */
#include <cimba.h>

#include <math.h>
#include <stdio.h>
#include <stdlib.h>

#define NUM_OBJECTS 100000u
#define ARRIVAL_RATE 0.9
#define SERVICE_RATE 1.0
#define NUM_TRIALS 20

struct trial {
    double avg_wait;
    uint64_t obj_cnt;
    uint64_t events;
};

struct mm1_ctx {
    cmb_objectqueue* queue;
    struct trial* trl;
    /* persistent process locals (must survive blocking macros) */
    uint64_t i;         /* arrival loop counter */
    void* put_obj;      /* arrival: object being enqueued */
    void* obj;          /* service: fetched object */
    double sum_wait;
    uint64_t cnt;
};

union timebox {
    double d;
    void* p;
};

static void arrival_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct mm1_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (ctx->i = 0; ctx->i < NUM_OBJECTS; ctx->i++) {
        CMB_HOLD(sim, me, cmb_random_exponential(sim, 1.0 / ARRIVAL_RATE));
        /* the object is its arrival time, boxed in the pointer payload;
         * stored in ctx so it survives a blocking PUT */
        {
            union timebox box;
            box.d = cmb_time(sim);
            ctx->put_obj = box.p;
        }
        CMB_OBJECTQUEUE_PUT(sim, me, ctx->queue, ctx->put_obj);
    }
    CMB_PROC_END(sim, me);
}

static void service_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct mm1_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        CMB_OBJECTQUEUE_GET(sim, me, ctx->queue, &ctx->obj);
        if (CMB_SIGNAL(sim, me) != CMB_PROCESS_SUCCESS) break;
        CMB_HOLD(sim, me, cmb_random_exponential(sim, 1.0 / SERVICE_RATE));
        {
            union timebox box;
            box.p = ctx->obj;
            ctx->sum_wait += cmb_time(sim) - box.d;
            ctx->cnt += 1u;
        }
    }
    CMB_PROC_END(sim, me);
}

static void run_trial(cmb_sim* sim, void* vtrl) {
    struct trial* trl = vtrl;
    struct mm1_ctx ctx = {0};
    ctx.trl = trl;

    ctx.queue = cmb_objectqueue_create(sim);
    cmb_objectqueue_initialize(sim, ctx.queue, "Queue", CMB_UNLIMITED);

    cmb_process* arrival =
        cmb_process_spawn(sim, "Arrival", arrival_body, &ctx, 0);
    cmb_process* service =
        cmb_process_spawn(sim, "Service", service_body, &ctx, 0);
    cmb_process_start(sim, arrival);
    cmb_process_start(sim, service);

    cmb_event_queue_execute(sim);

    trl->obj_cnt = ctx.cnt;
    trl->avg_wait = ctx.cnt ? ctx.sum_wait / (double)ctx.cnt : 0.0;
    trl->events = cmb_sim_events_dispatched(sim);
}

int main(void) {
    struct trial experiment[NUM_TRIALS] = {0};

    uint64_t failed = cimba_run(experiment, NUM_TRIALS, sizeof(*experiment),
                                run_trial, 0x34f05c64d7ad598fULL, 0);

    cmb_datasummary summary;
    cmb_datasummary_initialize(&summary);
    uint64_t events = 0;
    for (unsigned i = 0; i < NUM_TRIALS; i++) {
        cmb_datasummary_add(&summary, experiment[i].avg_wait);
        events += experiment[i].events;
    }
    const double mean = cmb_datasummary_mean(&summary);
    const double sdev = cmb_datasummary_stddev(&summary);
    printf("Average system time %f +- %f (n %u, expected %f), "
           "events %llu, failed %llu\n",
           mean, sdev, (unsigned)cmb_datasummary_count(&summary),
           1.0 / (SERVICE_RATE - ARRIVAL_RATE), (unsigned long long)events,
           (unsigned long long)failed);
    /* exit nonzero if statistically broken (tests parse this) */
    if (failed != 0 || mean < 7.0 || mean > 13.0) return 1;
    return 0;
}
