import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca
SEED = 0x34F05C64D7AD598F
def js(tag, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.jobshop_gpu(ntrials=131072, entities=10000, njobs=24, seed=SEED, device=0)
        print(f"js  {tag:16s}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s ok={r['trials_ok']==131072}", flush=True)
    finally:
        for k in env: del os.environ[k]
def mm1(tag, n, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mm1_gpu(ntrials=n, num_objects=10000, seed=SEED, device=0)
        print(f"mm1 {tag:16s} N={n}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]
def mg1(tag, env):
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mg1_gpu(ntrials=524288, num_objects=10000, arr_rate=0.8, srv_mean=1.0,
                       srv_scv=0.25, dist=3, seed=SEED, device=0)
        print(f"mg1 {tag:16s}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]
js("warmup", {})
js("hbm", {"CIMBA_JS_LANE": "1"})
js("hbm M3", {"CIMBA_JS_LANE": "1", "CIMBA_JS_LANE_MINW": "3"})
js("scratch", {"CIMBA_JS_LANE": "2"})
js("conv M4", {"CIMBA_JS_LANE": "3"})
js("conv M6", {"CIMBA_JS_LANE": "3", "CIMBA_CONV_MINW": "6"})
mm1("scratch(default)", 524288, {})
mg1("conv(default)", {})
