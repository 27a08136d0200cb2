/*
 * API tour: exercises the C surface that the model tutorials don't —
 * dataset/timeseries statistics, event introspection (is_scheduled /
 * time / priority / reschedule / reprioritize / pattern ops), alias
 * tables, queue position, multi-slot timers, summaries with weighted
 * moments.  Every call is assertion-checked; exit 0 means the whole
 * surface behaves.
 *
 * Build:
 *   gcc -std=c11 -Iinclude tutorial/api_tour_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -lm -o api_tour
 */
#include <cimba.h>

#include <assert.h>
#include <math.h>
#include <stdio.h>
#include <stdlib.h>

static int g_fired = 0;

static void noop_event(cmb_sim* sim, void* subject, void* object) {
    (void)sim;
    (void)object;
    g_fired += (int)(intptr_t)subject;
}

struct tour_ctx {
    cmb_objectqueue* q;
    uint64_t ev_a, ev_b, ev_c;
    void* got;
};

static void tour_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct tour_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);

    assert(cmb_process_current(sim) == me);

    /* ---- event introspection ---- */
    ctx->ev_a = cmb_event_schedule(sim, noop_event, (void*)1, NULL, 5.0, 0);
    ctx->ev_b = cmb_event_schedule(sim, noop_event, (void*)2, NULL, 7.0, 3);
    ctx->ev_c = cmb_event_schedule(sim, noop_event, (void*)4, NULL, 9.0, 0);
    assert(cmb_event_queue_count(sim) >= 3);
    assert(cmb_event_is_scheduled(sim, ctx->ev_a));
    assert(fabs(cmb_event_time(sim, ctx->ev_a) - 5.0) < 1e-12);
    assert(cmb_event_priority(sim, ctx->ev_b) == 3);
    assert(cmb_event_reschedule(sim, ctx->ev_a, 6.0, 1));
    assert(fabs(cmb_event_time(sim, ctx->ev_a) - 6.0) < 1e-12);
    assert(cmb_event_reprioritize(sim, ctx->ev_b, -2));
    assert(cmb_event_priority(sim, ctx->ev_b) == -2);
    assert(cmb_event_pattern_count(sim, noop_event, CMB_ANY_SUBJECT,
                                   CMB_ANY_OBJECT) == 3);
    assert(cmb_event_pattern_find(sim, CMB_ANY_ACTION, (void*)4,
                                  CMB_ANY_OBJECT) == ctx->ev_c);
    assert(cmb_event_pattern_cancel(sim, CMB_ANY_ACTION, (void*)4,
                                    CMB_ANY_OBJECT) == 1);
    assert(!cmb_event_is_scheduled(sim, ctx->ev_c));

    /* ---- multi-slot timers: slot 1 fires first, slot 2 canceled ---- */
    cmb_process_timer_add(sim, me, 1, 1.0, 111);
    cmb_process_timer_add(sim, me, 2, 2.0, 222);
    assert(cmb_process_timer_pending(sim, me, 1));
    cmb_process_timer_cancel(sim, me, 2);
    assert(!cmb_process_timer_pending(sim, me, 2));
    CMB_HOLD(sim, me, 100.0);
    assert(CMB_SIGNAL(sim, me) == 111);  /* interrupted by slot-1 timer */
    assert(fabs(cmb_time(sim) - 1.0) < 1e-12);

    /* ---- queue position ---- */
    CMB_OBJECTQUEUE_PUT(sim, me, ctx->q, (void*)10);
    CMB_OBJECTQUEUE_PUT(sim, me, ctx->q, (void*)20);
    CMB_OBJECTQUEUE_PUT(sim, me, ctx->q, (void*)30);
    assert(cmb_objectqueue_position(sim, ctx->q, (void*)10) == 1);
    assert(cmb_objectqueue_position(sim, ctx->q, (void*)30) == 3);
    assert(cmb_objectqueue_position(sim, ctx->q, (void*)99) == 0);
    CMB_OBJECTQUEUE_GET(sim, me, ctx->q, &ctx->got);
    assert(ctx->got == (void*)10);

    /* let the two remaining noop events run */
    CMB_HOLD(sim, me, 50.0);
    CMB_PROC_END(sim, me);
}

static const char* tour_fmt(cmb_event_func* a, const void* s,
                            const void* o) {
    (void)a;
    (void)o;
    return s == (const void*)7 ? "noop(7)" : NULL;
}

/* ---- embedded resource guard: wait on cmb_resource_guard() with a
 * custom demand, woken by the release-driven guard signal ---- */
struct rg_ctx {
    cmb_resource* res;
    double t0;
    int got;
};

static bool res_free_demand(cmb_sim* sim, void* vctx) {
    const struct rg_ctx* c = vctx;
    return cmb_resource_available(sim, c->res);
}

static void rg_holder(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct rg_ctx* c = vctx;
    CMB_PROC_BEGIN(sim, me);
    CMB_RESOURCE_ACQUIRE(sim, me, c->res);
    CMB_HOLD(sim, me, 5.0);
    cmb_resource_release(sim, c->res, me);
    CMB_PROC_END(sim, me);
}

static void rg_waiter(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct rg_ctx* c = vctx;
    CMB_PROC_BEGIN(sim, me);
    CMB_HOLD(sim, me, 1.0); /* let the holder grab the resource first */
    while (!cmb_resource_available(sim, c->res)) {
        CMB_RESOURCEGUARD_WAIT(sim, me, cmb_resource_guard(sim, c->res),
                               res_free_demand, c);
    }
    c->got = 1;
    assert(fabs(cmb_time(sim) - c->t0 - 5.0) < 1e-12);
    CMB_PROC_END(sim, me);
}

static void run_trial(cmb_sim* sim, void* vtrl) {
    (void)vtrl;
    static struct tour_ctx ctx;
    assert(cmb_process_current(sim) == NULL); /* dispatcher context */
    ctx.q = cmb_objectqueue_create(sim);
    cmb_objectqueue_initialize(sim, ctx.q, "TourQ", 16);
    assert(cmb_objectqueue_space(sim, ctx.q) == 16);
    cmb_process* p = cmb_process_spawn(sim, "Tour", tour_body, &ctx, 0);
    cmb_process_start(sim, p);
    cmb_event_queue_execute(sim);
    assert(g_fired == 1 + 2);  /* events a+b fired; c canceled */
    assert(cmb_event_current(sim) == ctx.ev_b); /* b (t=7) ran last */

    /* ---- built-in object guards ---- */
    static struct rg_ctx rc;
    rc.res = cmb_resource_create(sim);
    cmb_resource_initialize(sim, rc.res, "RGRes");
    rc.t0 = cmb_time(sim);
    rc.got = 0;
    cmb_resource_recording_start(sim, rc.res);
    cmb_process* ph = cmb_process_spawn(sim, "Holder", rg_holder, &rc, 0);
    cmb_process* pw = cmb_process_spawn(sim, "Waiter", rg_waiter, &rc, 0);
    cmb_process_start(sim, ph);
    cmb_process_start(sim, pw);
    cmb_event_queue_execute(sim);
    assert(rc.got == 1);
    /* recorded busy history: 0 at start, 1 at acquire, 0 at release */
    cmb_timeseries* rh = cmb_resource_history(sim, rc.res);
    assert(rh != NULL);
    assert(cmb_timeseries_count(rh) == 3);
    assert(cmb_timeseries_min(rh) == 0.0 && cmb_timeseries_max(rh) == 1.0);
    /* formatted event-queue dump */
    uint64_t ev = cmb_event_schedule(sim, noop_event, (void*)7, NULL, 1e9, 0);
    FILE* nul2 = fopen("/dev/null", "w");
    cmb_event_queue_print_formatted(sim, nul2, tour_fmt);
    fclose(nul2);
    assert(cmb_event_cancel(sim, ev));

    /* pool guard handle + signal with no waiters = no grant */
    cmb_resourcepool* pl = cmb_resourcepool_create(sim);
    cmb_resourcepool_initialize(sim, pl, "RGPool", 4);
    assert(cmb_resourcepool_guard(sim, pl) != NULL);
    assert(!cmb_resourceguard_signal(sim, cmb_resourcepool_guard(sim, pl)));

    /* ---- per-trial RNG helpers ---- */
    double w[4] = {1, 2, 3, 4};
    cmb_alias* al = cmb_random_alias_create(w, 4);
    long counts[4] = {0, 0, 0, 0};
    for (int i = 0; i < 40000; i++)
        counts[cmb_random_alias_sample(sim, al)]++;
    cmb_random_alias_destroy(al);
    for (int k = 0; k < 4; k++) {
        const double frac = counts[k] / 40000.0;
        assert(fabs(frac - (k + 1) / 10.0) < 0.02);
    }
}

int main(void) {
    char dummy = 0;
    uint64_t failed = cimba_run(&dummy, 1, 1, run_trial, 42, 1);
    assert(failed == 0);

    /* ---- dataset / timeseries / summaries (host-side) ---- */
    cmb_dataset* d = cmb_dataset_create();
    for (int i = 0; i < 101; i++) cmb_dataset_add(d, (double)i);
    assert(cmb_dataset_count(d) == 101);
    assert(fabs(cmb_dataset_median(d) - 50.0) < 1e-12);
    assert(cmb_dataset_min(d) == 0.0 && cmb_dataset_max(d) == 100.0);
    struct cmb_datasummary ds = cmb_dataset_summarize(d);
    assert(fabs(cmb_datasummary_mean(&ds) - 50.0) < 1e-12);
    int64_t hist[10];
    cmb_dataset_histogram(d, 10, hist);
    int64_t tot = 0;
    for (int i = 0; i < 10; i++) tot += hist[i];
    assert(tot == 101);
    double acf[3];
    cmb_dataset_acf(d, acf, 3);
    assert(acf[0] > 0.9); /* a ramp is highly autocorrelated */
    cmb_dataset_destroy(d);

    cmb_timeseries* ts = cmb_timeseries_create();
    cmb_timeseries_add(ts, 1.0, 0.0);
    cmb_timeseries_add(ts, 3.0, 2.0);
    cmb_timeseries_finalize(ts, 4.0);
    struct cmb_wtdsummary ws = cmb_timeseries_summarize(ts, -1.0);
    assert(fabs(cmb_wtdsummary_mean(&ws) - 2.0) < 1e-12); /* (1*2+3*2)/4 */
    assert(cmb_timeseries_min(ts) == 1.0 && cmb_timeseries_max(ts) == 3.0);
    cmb_timeseries_destroy(ts);

    /* timeseries correlogram (delegates to the dataset version) */
    cmb_timeseries* t2 = cmb_timeseries_create();
    for (int i = 0; i < 64; i++)
        cmb_timeseries_add(t2, (double)(i % 5), (double)i);
    FILE* nul = fopen("/dev/null", "w");
    cmb_timeseries_correlogram_print(t2, 5, nul);
    fclose(nul);
    cmb_timeseries_destroy(t2);

    /* weighted skewness sanity: symmetric data -> ~0 */
    cmb_wtdsummary* w2 = cmb_wtdsummary_create();
    cmb_wtdsummary_add(w2, -1.0, 2.0);
    cmb_wtdsummary_add(w2, 0.0, 3.0);
    cmb_wtdsummary_add(w2, 1.0, 2.0);
    assert(fabs(cmb_wtdsummary_skewness(w2)) < 1e-12);
    cmb_wtdsummary_destroy(w2);

    uint64_t st = 7;
    assert(cmb_random_splitmix64(&st) != cmb_random_splitmix64(&st));

    printf("api tour OK\n");
    return 0;
}
