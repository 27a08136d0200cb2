import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import os, sys, time
import cimba_amd as ca

N, OBJ, SEED = 262144, 10000, 0x34F05C64D7AD598F
def run(tag, env):
    for k, v in env.items():
        os.environ[k] = v
    try:
        r = ca.mm1_gpu(ntrials=N, num_objects=OBJ, seed=SEED, device=0)
        ev_s = r["total_events"] / (r["elapsed_ms"] * 1e-3)
        print(f"{tag}: {ev_s/1e9:.3f} G ev/s  elapsed={r['elapsed_ms']:.0f} ms trials_ok={r['trials_ok']}", flush=True)
    finally:
        for k in env:
            del os.environ[k]
    return ev_s

run("warmup(scratch)", {"CIMBA_MM1_LANE": "2"})
run("scratch  ", {"CIMBA_MM1_LANE": "2"})
run("hbm-lane ", {"CIMBA_MM1_LANE": "1"})
for k in ("1", "2", "4", "8"):
    run(f"conv K={k}", {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_K": k, "CIMBA_CONV_DEBUG": "1"})
for k in ("2", "4"):
    for b in ("256", "512", "1024"):
        run(f"conv K={k} B={b}", {"CIMBA_MM1_LANE": "3", "CIMBA_CONV_K": k, "CIMBA_CONV_BLOCKS": b})
# MG1 + JobShop conv A/B
for tag, env in [("mg1 scratch", {"CIMBA_MG1_LANE": "2"}), ("mg1 conv4", {"CIMBA_MG1_LANE": "3"})]:
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.mg1_gpu(ntrials=N, num_objects=OBJ, arr_rate=0.8, srv_mean=1.0, srv_scv=0.25, dist=3, seed=SEED, device=0)
        print(f"{tag}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]
for tag, env in [("js hbm", {"CIMBA_JS_LANE": "1"}), ("js conv4", {"CIMBA_JS_LANE": "3"})]:
    for k, v in env.items(): os.environ[k] = v
    try:
        r = ca.jobshop_gpu(ntrials=65536, entities=10000, njobs=24, seed=SEED, device=0)
        print(f"{tag}: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
    finally:
        for k in env: del os.environ[k]
