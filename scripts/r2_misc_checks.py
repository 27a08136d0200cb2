import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import cimba_amd as ca

os.environ["CIMBA_JS_LANE"] = "3"; os.environ["CIMBA_CONV_MINW"] = "4"
r = ca.jobshop_gpu(ntrials=131072, entities=10000, njobs=24, seed=5, device=0)
print(f"js conv M4: {r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
del os.environ["CIMBA_JS_LANE"]; del os.environ["CIMBA_CONV_MINW"]

# VERDICT item 4 done-criterion: MG1 lognormal SCV=4 rho=0.8, ~1e9 objects, zero aborts
r = ca.mg1_gpu(ntrials=131072, num_objects=8000, arr_rate=0.8, srv_mean=1.0,
               srv_scv=4.0, dist=2, seed=5, device=0)
print(f"mg1 lognormal SCV=4 1.05e9 objects: trials_ok={r['trials_ok']}/131072 "
      f"first_bad={r['first_bad_status']} rate={r['total_events']/(r['elapsed_ms']*1e-3)/1e9:.3f} G ev/s", flush=True)
