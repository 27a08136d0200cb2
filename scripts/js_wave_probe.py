import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca
SEED = 0x34F05C64D7AD598F
for tag, env in [("hbm-lane(default)", {}), ("wave M4", {"CIMBA_JS_LANE": "0"}),
                 ("wave M5", {"CIMBA_JS_LANE": "0", "CIMBA_JS_MINW": "5"})]:
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.jobshop_gpu(ntrials=131072, entities=10000, njobs=24, seed=SEED, device=0)
        print(f"js {tag:18s}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s ok={r['trials_ok']==131072}", flush=True)
    finally:
        for k in env: del os.environ[k]
