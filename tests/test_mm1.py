"""Host-engine M/M/1 integration tests — the counterpart of the reference's
benchmark/MM1_multi.c correctness expectations (avg system time
1/(mu-lambda)) plus determinism-by-seed (reference seed-replay discipline,
SURVEY.md §5.4)."""
import cimba_amd as ca


def test_mm1_mean_system_time():
    r = ca.mm1_host(ntrials=8, num_objects=50_000, seed=99, threads=4)
    assert r["trials_ok"] == 8
    assert r["total_objects"] == 8 * 50_000
    # rho=0.9: E[T] = 1/(mu-lambda) = 10; 400k objects -> tight-ish
    assert 8.5 < r["avg_system_time"] < 11.5


def test_mm1_deterministic_by_seed():
    a = ca.mm1_host(ntrials=4, num_objects=5_000, seed=7, threads=1)
    b = ca.mm1_host(ntrials=4, num_objects=5_000, seed=7, threads=4)
    # per-trial seeding makes results independent of thread count/schedule
    assert a["total_wait"] == b["total_wait"]
    assert a["total_events"] == b["total_events"]
    assert a["per_trial_avg"] == b["per_trial_avg"]
    c = ca.mm1_host(ntrials=4, num_objects=5_000, seed=8, threads=1)
    assert c["total_wait"] != a["total_wait"]


def test_mm1_event_rate_sane():
    # ~2.1 events per object (hold wake + queue grant + service hold)
    r = ca.mm1_host(ntrials=2, num_objects=20_000, seed=5, threads=2)
    per_obj = r["total_events"] / r["total_objects"]
    assert 1.8 < per_obj < 3.5


def test_mm1_zero_trial_edge():
    r = ca.mm1_host(ntrials=1, num_objects=1, seed=1, threads=1)
    assert r["trials_ok"] == 1
    assert r["total_objects"] == 1


def test_seed_replay_single_trial():
    """Seed-replay (reference checkpoint substitute): any
    trial of an experiment reruns identically from (master_seed, index)."""
    full = ca.mm1_host(ntrials=8, num_objects=3000, seed=999, threads=4)
    # replay trial 5 alone: a 1-trial experiment whose master seed is
    # chosen so fmix-derived trial-0 seed equals the original trial-5 seed
    # -> use the internal trial_seed to verify determinism directly
    s5 = ca.trial_seed(999, 5)
    s5b = ca.trial_seed(999, 5)
    assert s5 == s5b != ca.trial_seed(999, 6)
    again = ca.mm1_host(ntrials=8, num_objects=3000, seed=999, threads=1)
    assert again["per_trial_avg"][5] == full["per_trial_avg"][5]


def test_models_registry():
    from cimba_amd.models import MODELS, run

    assert MODELS == ("awacs", "jobshop", "mg1", "mm1")
    r = run("mm1", backend="cpu", ntrials=2, num_objects=2000, seed=4,
            threads=1)
    assert r["trials_ok"] == 2
