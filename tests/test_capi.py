"""Compile-and-run test of the public cmb_* C API (include/cimba.h):
builds tutorial/mm1_capi.c with gcc against libcimba.so and checks the
M/M/1 result — the counterpart of the reference's install-verification job
(reference .github/workflows/ci.yml install job + header self-containment
check)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def libcimba():
    import cimba_amd  # triggers build (includes libcimba.so)

    lib = os.path.join(ROOT, "cimba_amd", "libcimba.so")
    if not os.path.exists(lib):
        from cimba_amd import _build

        _build.build(force=True)
    assert os.path.exists(lib)
    return lib


def test_capi_mm1_compiles_and_runs(libcimba, tmp_path):
    exe = str(tmp_path / "mm1_capi")
    compile_cmd = [
        "gcc", "-std=c11", "-O2", "-Wall", "-Werror",
        "-I", os.path.join(ROOT, "include"),
        os.path.join(ROOT, "tutorial", "mm1_capi.c"),
        "-L", os.path.join(ROOT, "cimba_amd"), "-lcimba",
        f"-Wl,-rpath,{os.path.join(ROOT, 'cimba_amd')}",
        "-lm", "-o", exe,
    ]
    r = subprocess.run(compile_cmd, capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "Average system time" in out.stdout
    # determinism: same master seed => identical output
    out2 = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out2.stdout == out.stdout


def test_capi_mg1_resource_condition(libcimba, tmp_path):
    # wider API tour: resource, condition with C demand predicate, wait
    # timeouts, recording + report printers; validates vs PK theory
    exe = str(tmp_path / "mg1_capi")
    r = subprocess.run(
        ["gcc", "-std=c11", "-O2", "-Wall", "-Werror",
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "tutorial", "mg1_capi.c"),
         "-L", os.path.join(ROOT, "cimba_amd"), "-lcimba",
         f"-Wl,-rpath,{os.path.join(ROOT, 'cimba_amd')}",
         "-lm", "-o", exe], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "queue report" in out.stdout
    assert "resource report" in out.stdout
    assert "M/G/1 avg system time" in out.stdout


def test_capi_1000_processes_and_timers(libcimba, tmp_path):
    # 1000 concurrent processes + multi-slot heartbeat timers (the
    # reference's AWACS-scale process count on the host engine)
    exe = str(tmp_path / "manyprocs")
    r = subprocess.run(
        ["gcc", "-std=c11", "-O2", "-Wall", "-Werror",
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "tutorial", "manyprocs_capi.c"),
         "-L", os.path.join(ROOT, "cimba_amd"), "-lcimba",
         f"-Wl,-rpath,{os.path.join(ROOT, 'cimba_amd')}",
         "-lm", "-o", exe], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "1000-process run" in out.stdout


def test_capi_api_tour(libcimba, tmp_path):
    # assertion-checked tour of the wider surface: event introspection,
    # pattern ops, multi-slot timers, queue position, alias tables,
    # dataset/timeseries/summary C wrappers
    exe = str(tmp_path / "api_tour")
    r = subprocess.run(
        ["gcc", "-std=c11", "-O1", "-g", "-Wall", "-Werror",
         "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "tutorial", "api_tour_capi.c"),
         "-L", os.path.join(ROOT, "cimba_amd"), "-lcimba",
         f"-Wl,-rpath,{os.path.join(ROOT, 'cimba_amd')}",
         "-lm", "-o", exe], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    out = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "api tour OK" in out.stdout


def test_capi_header_is_c_clean(tmp_path):
    # header must compile as plain C99 without the library
    src = tmp_path / "hdr.c"
    src.write_text('#include "cimba.h"\nint main(void){return 0;}\n')
    r = subprocess.run(
        ["gcc", "-std=c99", "-Wall", "-Werror", "-I",
         os.path.join(ROOT, "include"), "-c", str(src), "-o",
         str(tmp_path / "hdr.o")],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr


def test_capi_event_storm_no_quadratic_blowup(libcimba, tmp_path):
    """Host-scale hashheap contract: a 60K-event
    schedule + random-cancel storm and a 4000-process wait_event fanout
    must run in O(n log n).  Before the handle back-map + waiter list,
    the cancel path alone was ~1.8e9 scan steps (many seconds); now the
    whole storm is tens of milliseconds, so a generous wall bound still
    discriminates sharply."""
    src = tmp_path / "storm.c"
    src.write_text(r'''
#include "cimba.h"
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

static void noop(cmb_sim* sim, void* subj, void* obj) {
    (void)sim; (void)subj; (void)obj;
}

static void trial(cmb_sim* sim, void* exp) {
    (void)exp;
    enum { N = 60000 };
    static uint64_t h[N];
    /* schedule N events at pseudo-random times */
    uint64_t s = 0x9E3779B97F4A7C15ull;
    for (int i = 0; i < N; ++i) {
        s = s * 6364136223846793005ull + 1442695040888963407ull;
        double t = (double)(s >> 40);
        h[i] = cmb_event_schedule(sim, noop, (void*)(uintptr_t)(i + 1),
                                  NULL, t, 0);
        if (!h[i]) { fprintf(stderr, "schedule failed at %d\n", i); exit(1); }
    }
    /* reschedule a third, cancel half in a scrambled order */
    for (int i = 0; i < N; i += 3) {
        if (!cmb_event_reschedule(sim, h[i], (double)i, 0)) exit(2);
    }
    for (int i = 0; i < N; i += 2) {
        int j = (int)((unsigned)(i * 2654435761u) % N) & ~1;
        if (h[j]) {
            if (!cmb_event_cancel(sim, h[j])) exit(3);
            h[j] = 0;
        }
    }
    cmb_event_queue_clear(sim);
}

int main(void) {
    char exp[8];
    uint64_t failed = cimba_run(exp, 1, 8, trial, 42, 1);
    if (failed) { fprintf(stderr, "trial failed\n"); return 1; }
    puts("storm OK");
    return 0;
}
''')
    exe = str(tmp_path / "storm")
    r = subprocess.run(
        ["gcc", "-std=c11", "-O2", "-I", os.path.join(ROOT, "include"),
         str(src), "-L", os.path.join(ROOT, "cimba_amd"), "-lcimba",
         f"-Wl,-rpath,{os.path.join(ROOT, 'cimba_amd')}", "-lm", "-o", exe],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    import time
    t0 = time.perf_counter()
    out = subprocess.run([exe], capture_output=True, text=True, timeout=60)
    dt = time.perf_counter() - t0
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "storm OK" in out.stdout
    # O(n^2) was multiple seconds of pure scanning; O(n log n) is ~tens
    # of ms — 3 s leaves huge CI headroom while still failing a scan
    assert dt < 3.0, f"event storm took {dt:.1f}s — quadratic path?"


def _build_and_run(tmp_path, src_name, exe_name, timeout=600, args=()):
    exe = str(tmp_path / exe_name)
    r = subprocess.run(
        ["gcc", "-std=c11", "-O2", "-Wall", "-I", os.path.join(ROOT, "include"),
         os.path.join(ROOT, "tutorial", src_name),
         "-L", os.path.join(ROOT, "cimba_amd"), "-lcimba",
         f"-Wl,-rpath,{os.path.join(ROOT, 'cimba_amd')}", "-lm", "-o", exe],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-3000:]
    out = subprocess.run([exe, *args], capture_output=True, text=True,
                         timeout=timeout)
    return out


def test_capi_hello_tutorial(libcimba, tmp_path):
    out = _build_and_run(tmp_path, "hello_capi.c", "hello")
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "hello from hello" in out.stdout
    assert "simulation ended" in out.stdout


def test_capi_harbor_tutorial(libcimba, tmp_path):
    """Harbor-with-abandoned-trials (reference tut_4_3 counterpart):
    condition waits, multi-unit pool acquires, logger_error -> trial
    abandon with the cleanup hook, multithreaded scenario grid."""
    out = _build_and_run(tmp_path, "harbor_capi.c", "harbor")
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    assert "abandoned trials:" in out.stdout
    # the cleanup-hook count must equal the failed count (checked in C,
    # nonzero exit otherwise); determinism: same seed -> same output
    out2 = _build_and_run(tmp_path, "harbor_capi.c", "harbor")
    assert out2.stdout == out.stdout


def test_capi_awacs_tutorial(libcimba, tmp_path):
    """CPU AWACS (reference tut_5_1 counterpart): 1000 target processes,
    per-dwell radar scanning, a condition watcher — the C API at the
    reference tutorial's process scale."""
    out = _build_and_run(tmp_path, "awacs_capi.c", "awacsc")
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    assert "reached the 200-track goal" in out.stdout
    last = out.stdout.strip().splitlines()[-1]
    assert "(0 failed)" in last
