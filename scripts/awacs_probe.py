import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca
def run(tag, probe):
    os.environ["CIMBA_AWACS_PROBE"] = str(probe)
    try:
        t0 = time.perf_counter()
        r = ca._C.awacs_gpu(ntrials=300, duration=240.0, ntargets=1000, seed=5, device=0)
        print(f"{tag:26s}: {time.perf_counter()-t0:6.2f}s det={r['total_detections']}", flush=True)
    finally:
        del os.environ["CIMBA_AWACS_PROBE"]
run("warmup (full)", 0)
run("full pipeline", 0)
run("skip survivor loop", 1)
run("skip clutter/CFAR", 2)
run("skip LOS", 4)
run("skip MFMA beamforming", 8)
run("skip surv+MFMA", 9)
