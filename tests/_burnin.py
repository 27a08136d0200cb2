import time
import cimba_amd as ca

t0 = time.time()
rounds = 0
while time.time() - t0 < 900:  # 15 min mixed burn-in
    i = rounds
    r = ca.mm1_gpu(ntrials=262144, num_objects=10000, seed=1000 + i, device=0)
    assert r["trials_ok"] == 262144, r
    g = ca.mg1_gpu(ntrials=262144, num_objects=5000, seed=2000 + i, device=0)
    assert g["trials_ok"] == 262144, g
    j = ca.jobshop_gpu(ntrials=131072, entities=500, njobs=24, seed=3000 + i, device=0)
    assert j["trials_ok"] == 131072, j
    a = ca._C.awacs_gpu(ntrials=1024, duration=20.0, ntargets=1000, seed=4000 + i, device=0)
    assert a["trials_ok"] == 1024, a
    for w in list(range(1, 14)) + [15, 16, 17]:
        s = ca._C.scenario_gpu(w)
        assert (s["status"] == 0) == (w < 16), (w, s["status"])
    m = ca._C.rng_moments_gpu("std_normal", 0.0, 1 << 27, 5000 + i, 0)
    assert abs(m["mean"]) < 1e-3
    rounds += 1
print("burn-in complete: %d mixed rounds in %.0f s, all OK" % (rounds, time.time() - t0))
