"""Vote-gated conv kernel vs plain lane kernels across batch sizes."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca

SEED = 0x34F05C64D7AD598F

def mm1(tag, n, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mm1_gpu(ntrials=n, num_objects=10000, seed=SEED, device=0)
        print(f"mm1 {tag:24s} N={n}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s ok={r['trials_ok']==n}", flush=True)
    finally:
        for k in env: del os.environ[k]

def mg1(tag, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mg1_gpu(ntrials=262144, num_objects=10000, arr_rate=0.8, srv_mean=1.0,
                       srv_scv=0.25, dist=3, seed=SEED, device=0)
        print(f"mg1 {tag}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]

def js(tag, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.jobshop_gpu(ntrials=65536, entities=10000, njobs=24, seed=SEED, device=0)
        print(f"js  {tag}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]

mm1("warmup", 262144, {})
for n in (262144, 524288, 1048576):
    mm1("conv M8", n, {"CIMBA_MM1_LANE": "3"})
    mm1("conv M6", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_MINW": "6"})
    mm1("conv M4", n, {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_MINW": "4"})
    mm1("scratch", n, {"CIMBA_MM1_LANE": "2"})
    mm1("scratch M4", n, {"CIMBA_MM1_LANE": "2", "CIMBA_SCRATCH_MINW": "4"})
    mm1("scratch M6", n, {"CIMBA_MM1_LANE": "2", "CIMBA_SCRATCH_MINW": "6"})
mg1("scratch ", {"CIMBA_MG1_LANE": "2"})
mg1("scr M4  ", {"CIMBA_MG1_LANE": "2", "CIMBA_SCRATCH_MINW": "4"})
mg1("conv    ", {"CIMBA_MG1_LANE": "3"})
mg1("conv M6 ", {"CIMBA_MG1_LANE": "3", "CIMBA_CONV_MINW": "6"})
mg1("conv M4 ", {"CIMBA_MG1_LANE": "3", "CIMBA_CONV_MINW": "4"})
js("hbm ", {"CIMBA_JS_LANE": "1"})
js("conv", {"CIMBA_JS_LANE": "3"})
js("conv M6", {"CIMBA_JS_LANE": "3", "CIMBA_CONV_MINW": "6"})
