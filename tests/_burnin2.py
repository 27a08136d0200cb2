import time
import cimba_amd as ca
t0 = time.time()
rounds = 0
while time.time() - t0 < 1500:  # 25 min
    i = rounds + 777
    r = ca.mm1_gpu(ntrials=262144, num_objects=10000, seed=i, device=0)
    assert r["trials_ok"] == 262144 and 9.0 < r["avg_system_time"] < 11.0, r
    g = ca.mg1_gpu(ntrials=262144, num_objects=5000, srv_scv=2.0, dist=1, seed=i * 3, device=0)
    assert g["trials_ok"] == 262144, g
    j = ca.jobshop_gpu(ntrials=131072, entities=500, njobs=24, seed=i * 7, device=0)
    assert j["trials_ok"] == 131072, j
    a = ca._C.awacs_gpu(ntrials=2048, duration=15.0, ntargets=1000, seed=i * 11, device=0)
    assert a["trials_ok"] == 2048, a
    rr = ca._C.mm1_multigpu_rccl(ntrials=8192, num_objects=5000, seed=i * 13)
    assert int(rr["n"]) == 8192, rr
    rounds += 1
print("burn-in2 complete: %d rounds in %.0f s, all OK" % (rounds, time.time() - t0))
