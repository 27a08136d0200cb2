import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca

SEED = 0x34F05C64D7AD598F

def mm1(tag, n, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mm1_gpu(ntrials=n, num_objects=10000, seed=SEED, device=0)
        print(f"mm1 {tag:28s} N={n}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s ok={r['trials_ok']==n}", flush=True)
    finally:
        for k in env: del os.environ[k]

mm1("warmup", 262144, {})
for n in (262144, 524288, 1048576):
    mm1("conv K=1", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_DEBUG": "1"})
for n in (524288, 1048576):
    mm1("conv K=1 B=2048", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_BLOCKS": "2048"})
    mm1("conv K=1 B=4096", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_BLOCKS": "4096"})
    mm1("conv K=2", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_K": "2", "CIMBA_CONV_DEBUG": "1"})
    mm1("conv K=2 B=2048", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_K": "2", "CIMBA_CONV_BLOCKS": "2048"})
    mm1("scratch", n, {"CIMBA_MM1_LANE": "2"})
    mm1("scratch B=4096", n, {"CIMBA_MM1_LANE": "2", "CIMBA_MM1_LANE_BLOCKS": "4096"})
