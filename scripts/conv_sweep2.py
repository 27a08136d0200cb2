import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca

N, OBJ, SEED = 262144, 10000, 0x34F05C64D7AD598F

def mm1(tag, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mm1_gpu(ntrials=N, num_objects=OBJ, seed=SEED, device=0)
        print(f"mm1 {tag}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s ok={r['trials_ok']}", flush=True)
    finally:
        for k in env: del os.environ[k]

def mg1(tag, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mg1_gpu(ntrials=N, num_objects=OBJ, arr_rate=0.8, srv_mean=1.0,
                       srv_scv=0.25, dist=3, seed=SEED, device=0)
        print(f"mg1 {tag}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]

mm1("warmup", {})
mm1("conv K=1 MINW=1", {"CIMBA_MM1_LANE": "3"})
mm1("conv K=1 MINW=4", {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_MINW": "4"})
mm1("conv K=1 B=2048", {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_BLOCKS": "2048"})
mm1("conv K=1 again ", {"CIMBA_MM1_LANE": "3"})
mg1("scratch", {"CIMBA_MG1_LANE": "2"})
mg1("conv K=1", {"CIMBA_MG1_LANE": "3"})
mg1("conv K=1 MINW=4", {"CIMBA_MG1_LANE": "3", "CIMBA_CONV_MINW": "4"})
