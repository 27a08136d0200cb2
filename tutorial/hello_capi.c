/*
 * The smallest possible cimba-mi355x program — the "hello, simulated
 * world" entry of the graded tutorial series (reference tutorial/hello.c
 * counterpart): one process, three holds, the simulation clock.
 *
 * Build:
 *   gcc -std=c11 -Iinclude tutorial/hello_capi.c -Lcimba_amd -lcimba \
 *       -Wl,-rpath,$PWD/cimba_amd -lm -o hello
 */
#include <cimba.h>

#include <stdio.h>

static void hello_proc(cmb_sim* sim, cmb_process* me, void* ctx) {
    (void)ctx;
    CMB_PROC_BEGIN(sim, me);
    printf("t=%5.1f  hello from %s\n", cmb_time(sim),
           cmb_process_name(sim, me));
    cmb_process_hold(sim, me, 1.5);
    printf("t=%5.1f  still here\n", cmb_time(sim));
    cmb_process_hold(sim, me, cmb_random_exponential(sim, 2.0));
    printf("t=%5.1f  one random hold later\n", cmb_time(sim));
    CMB_PROC_END(sim, me);
}

static void trial(cmb_sim* sim, void* exp) {
    (void)exp;
    cmb_process* p = cmb_process_spawn(sim, "hello", hello_proc, NULL, 0);
    cmb_process_start(sim, p);
    cmb_event_queue_execute(sim);
    printf("simulation ended at t=%.3f after %llu events\n", cmb_time(sim),
           (unsigned long long)cmb_sim_events_dispatched(sim));
}

int main(void) {
    char exp[1];
    const uint64_t failed = cimba_run(exp, 1, 1, trial, 42, 1);
    return failed != 0;
}
