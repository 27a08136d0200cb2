/*
 * M/G/1 queue with a resource server, wait timeouts, a condition variable
 * and report printing — a wider tour of the cmb_* C API than
 * tutorial/mm1_capi.c (counterpart of the reference's tutorial
 * progression and of test/test_cimba.c's M/G/1 experiment).
 *
 * Model: arrival process enqueues jobs (timestamps); server process takes
 * each job, ACQUIRES the server resource, holds a gamma-distributed
 * service time, releases.  A monitor process waits on a condition
 * variable until `served >= threshold`, then records the time.  Arrivals
 * use a timeout-armed put so a full queue drops jobs instead of blocking
 * forever (drop counting).
 *
 * Build:
 *   gcc -std=c11 -Iinclude tutorial/mg1_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -lm -o mg1_capi
 */
#include <cimba.h>

#include <math.h>
#include <stdio.h>
#include <stdlib.h>

#define NUM_JOBS 20000u
#define ARRIVAL_RATE 0.8
#define SERVICE_MEAN 1.0
#define SERVICE_SCV 0.25
#define NUM_TRIALS 8
#define MONITOR_THRESHOLD 10000u

struct trial {
    double avg_system;
    double t_threshold;  /* sim time when MONITOR_THRESHOLD jobs done */
    uint64_t served;
    uint64_t dropped;
};

struct mg1_ctx {
    cmb_objectqueue* queue;
    cmb_resource* server;
    cmb_condition* done_cond;
    struct trial* trl;
    /* persistent locals */
    uint64_t i;
    void* put_obj;
    void* obj;
    double sum_system;
    uint64_t served;
    uint64_t dropped;
    double t_threshold;
};

union timebox {
    double d;
    void* p;
};

static double service_time(cmb_sim* sim) {
    /* gamma with mean SERVICE_MEAN, SCV SERVICE_SCV */
    return cmb_random_gamma(sim, 1.0 / SERVICE_SCV,
                            SERVICE_MEAN * SERVICE_SCV);
}

static void arrival_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct mg1_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (ctx->i = 0; ctx->i < NUM_JOBS; ctx->i++) {
        CMB_HOLD(sim, me, cmb_random_exponential(sim, 1.0 / ARRIVAL_RATE));
        {
            union timebox box;
            box.d = cmb_time(sim);
            ctx->put_obj = box.p;
        }
        /* timeout-armed put: drop the job if no space within 5 time units */
        cmb_timer_arm_(sim, me, 5.0, CMB_PROCESS_TIMEOUT);
        CMB_OBJECTQUEUE_PUT(sim, me, ctx->queue, ctx->put_obj);
        if (CMB_SIGNAL(sim, me) == CMB_PROCESS_TIMEOUT) {
            ctx->dropped++;
        } else {
            cmb_timer_disarm_(sim, me);
        }
    }
    CMB_PROC_END(sim, me);
}

static void server_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct mg1_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        CMB_OBJECTQUEUE_GET(sim, me, ctx->queue, &ctx->obj);
        if (CMB_SIGNAL(sim, me) != CMB_PROCESS_SUCCESS) break;
        CMB_RESOURCE_ACQUIRE(sim, me, ctx->server);
        CMB_HOLD(sim, me, service_time(sim));
        cmb_resource_release(sim, ctx->server, me);
        {
            union timebox box;
            box.p = ctx->obj;
            ctx->sum_system += cmb_time(sim) - box.d;
        }
        ctx->served++;
        if (ctx->served == MONITOR_THRESHOLD)
            cmb_condition_signal(sim, ctx->done_cond);
    }
    CMB_PROC_END(sim, me);
}

static bool threshold_reached(cmb_sim* sim, void* vctx) {
    (void)sim;
    const struct mg1_ctx* ctx = vctx;
    return ctx->served >= MONITOR_THRESHOLD;
}

static void monitor_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct mg1_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    CMB_CONDITION_WAIT(sim, me, ctx->done_cond, threshold_reached, ctx);
    ctx->t_threshold = cmb_time(sim);
    CMB_PROC_END(sim, me);
}

static void run_trial(cmb_sim* sim, void* vtrl) {
    struct trial* trl = vtrl;
    struct mg1_ctx ctx = {0};
    ctx.trl = trl;

    ctx.queue = cmb_objectqueue_create(sim);
    cmb_objectqueue_initialize(sim, ctx.queue, "Jobs", 64);
    cmb_objectqueue_recording_start(sim, ctx.queue);
    ctx.server = cmb_resource_create(sim);
    cmb_resource_initialize(sim, ctx.server, "Server");
    cmb_resource_recording_start(sim, ctx.server);
    ctx.done_cond = cmb_condition_create(sim);
    cmb_condition_initialize(sim, ctx.done_cond, "Done");

    cmb_process* a = cmb_process_spawn(sim, "Arrival", arrival_body, &ctx, 0);
    cmb_process* s = cmb_process_spawn(sim, "Server", server_body, &ctx, 0);
    cmb_process* m = cmb_process_spawn(sim, "Monitor", monitor_body, &ctx, 1);
    cmb_process_start(sim, a);
    cmb_process_start(sim, s);
    cmb_process_start(sim, m);

    cmb_event_queue_execute(sim);

    if (cmb_sim_trial_index(sim) == 0) {
        cmb_objectqueue_report_print(sim, ctx.queue, stdout);
        cmb_resource_print_report(sim, ctx.server, stdout);
    }

    trl->served = ctx.served;
    trl->dropped = ctx.dropped;
    trl->avg_system = ctx.served ? ctx.sum_system / (double)ctx.served : 0.0;
    trl->t_threshold = ctx.t_threshold;
}

int main(void) {
    struct trial experiment[NUM_TRIALS] = {{0}};
    uint64_t failed = cimba_run(experiment, NUM_TRIALS, sizeof(*experiment),
                                run_trial, 0x1234abcdULL, 0);

    cmb_datasummary sys;
    cmb_datasummary_initialize(&sys);
    uint64_t served = 0, dropped = 0;
    for (unsigned i = 0; i < NUM_TRIALS; i++) {
        cmb_datasummary_add(&sys, experiment[i].avg_system);
        served += experiment[i].served;
        dropped += experiment[i].dropped;
    }
    /* Pollaczek-Khinchine expectation for rho=0.8, SCV=0.25 */
    const double lam = ARRIVAL_RATE, m = SERVICE_MEAN;
    const double es2 = SERVICE_SCV * m * m + m * m;
    const double expect = lam * es2 / (2.0 * (1.0 - lam * m)) + m;
    printf("M/G/1 avg system time %f +- %f (PK theory %f), served %llu, "
           "dropped %llu, failed %llu\n",
           cmb_datasummary_mean(&sys), cmb_datasummary_stddev(&sys), expect,
           (unsigned long long)served, (unsigned long long)dropped,
           (unsigned long long)failed);
    const double mean = cmb_datasummary_mean(&sys);
    if (failed != 0 || fabs(mean - expect) / expect > 0.15) return 1;
    return 0;
}
