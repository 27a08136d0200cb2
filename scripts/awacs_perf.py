"""AWACS pipeline GPU probe: correctness scale + the published-point beat
(reference: 300 trials x 6 h sim, 1000 targets, 0.04 s dwells = 78 s wall
on a 32-core 3970X + 2x RTX 3090, README.md:321-329)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca

# short warmup + scaling probe
for dur in (60.0, 600.0):
    t0 = time.perf_counter()
    r = ca._C.awacs_gpu(ntrials=300, duration=dur, ntargets=1000, seed=5, device=0)
    dt = time.perf_counter() - t0
    print(f"dur={dur:6.0f}s 300 trials: wall={dt:7.2f}s kernel={r['elapsed_ms']/1e3:7.2f}s "
          f"det={r['total_detections']} illum={r['total_illuminated']} shld={r['total_shielded']} ok={r['trials_ok']}", flush=True)
# full published point: 300 trials x 6 h
t0 = time.perf_counter()
r = ca._C.awacs_gpu(ntrials=300, duration=21600.0, ntargets=1000, seed=5, device=0)
dt = time.perf_counter() - t0
print(f"PUBLISHED POINT 300x6h: wall={dt:.1f}s kernel={r['elapsed_ms']/1e3:.1f}s "
      f"(reference 78 s) det={r['total_detections']} shld={r['total_shielded']} ok={r['trials_ok']}", flush=True)
