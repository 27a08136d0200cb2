/*
 * 1000 concurrent producer processes + one consumer — the process-count
 * scale point of the reference's AWACS tutorial ("1000 target
 * coroutines", README.md:321-329), on the state-machine process model.
 * Each producer holds an exponential delay, then puts a token; a single
 * consumer drains the queue.  Also exercises multi-slot timers: every
 * producer keeps a user timer (slot 1) ticking as a heartbeat that
 * interrupts its hold with a user signal.
 *
 * Build:
 *   gcc -std=c11 -Iinclude tutorial/manyprocs_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -lm -o manyprocs_capi
 */
#include <cimba.h>

#include <stdio.h>
#include <stdlib.h>

#define NPROD 1000
#define TOKENS_PER_PROD 50
#define SIG_HEARTBEAT 42

struct shared {
    cmb_objectqueue* queue;
    uint64_t consumed;
    uint64_t heartbeats;
};

struct prod_ctx {
    struct shared* sh;
    uint32_t i;
};

struct cons_ctx {
    struct shared* sh;
    void* obj;
};

static void producer_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct prod_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (ctx->i = 0; ctx->i < TOKENS_PER_PROD; ctx->i++) {
        /* heartbeat timer on user slot 1: fires mid-hold now and then */
        if (!cmb_process_timer_pending(sim, me, 1))
            cmb_process_timer_add(sim, me, 1, 7.5, SIG_HEARTBEAT);
        CMB_HOLD(sim, me, cmb_random_exponential(sim, 1.0));
        if (CMB_SIGNAL(sim, me) == SIG_HEARTBEAT) {
            ctx->sh->heartbeats++;
            continue; /* interrupted hold: token not produced this round */
        }
        CMB_OBJECTQUEUE_PUT(sim, me, ctx->sh->queue, (void*)(uintptr_t)(ctx->i + 1));
    }
    cmb_process_timer_clear(sim, me);
    CMB_PROC_END(sim, me);
}

static void consumer_body(cmb_sim* sim, cmb_process* me, void* vctx) {
    struct cons_ctx* ctx = vctx;
    CMB_PROC_BEGIN(sim, me);
    for (;;) {
        CMB_OBJECTQUEUE_GET(sim, me, ctx->sh->queue, &ctx->obj);
        if (CMB_SIGNAL(sim, me) != CMB_PROCESS_SUCCESS) break;
        ctx->sh->consumed++;
    }
    CMB_PROC_END(sim, me);
}

struct trial {
    uint64_t consumed;
    uint64_t heartbeats;
    uint64_t events;
};

static struct prod_ctx g_prod[NPROD]; /* per-trial reset inside run_trial */

static void run_trial(cmb_sim* sim, void* vtrl) {
    struct trial* trl = vtrl;
    static struct shared sh; /* single-threaded per trial via cimba_run(1) */
    sh.consumed = 0;
    sh.heartbeats = 0;
    sh.queue = cmb_objectqueue_create(sim);
    cmb_objectqueue_initialize(sim, sh.queue, "Tokens", CMB_UNLIMITED);

    static struct cons_ctx cc;
    cc.sh = &sh;
    cc.obj = NULL;
    cmb_process* cons =
        cmb_process_spawn(sim, "Consumer", consumer_body, &cc, 1);
    cmb_process_start(sim, cons);

    for (int i = 0; i < NPROD; i++) {
        g_prod[i].sh = &sh;
        g_prod[i].i = 0;
        char name[32];
        snprintf(name, sizeof(name), "Prod%d", i);
        cmb_process* p =
            cmb_process_spawn(sim, name, producer_body, &g_prod[i], 0);
        cmb_process_start(sim, p);
    }

    cmb_event_queue_execute(sim);

    trl->consumed = sh.consumed;
    trl->heartbeats = sh.heartbeats;
    trl->events = cmb_sim_events_dispatched(sim);
}

int main(void) {
    struct trial trl = {0};
    uint64_t failed = cimba_run(&trl, 1, sizeof(trl), run_trial,
                                0xfeedface, 1);
    printf("1000-process run: consumed %llu, heartbeats %llu, "
           "events %llu, failed %llu\n",
           (unsigned long long)trl.consumed,
           (unsigned long long)trl.heartbeats,
           (unsigned long long)trl.events, (unsigned long long)failed);
    /* every token produced must be consumed; heartbeats replace tokens */
    if (failed != 0) return 1;
    if (trl.consumed + trl.heartbeats != (uint64_t)NPROD * TOKENS_PER_PROD)
        return 1;
    if (trl.heartbeats == 0) return 1; /* timers must actually fire */
    return 0;
}
