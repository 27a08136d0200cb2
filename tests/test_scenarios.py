"""Golden semantic traces for the process/guard/toolkit layer.

Each expectation below is derived BY HAND from the engine contract
(reference semantics): event ordering is (time asc,
priority desc, FIFO), guards grant front-waiter-only with stale-grant
pass-on, blocking calls return signals per include/cmb_process.h:60-100.
Codes: pidx*1000 + tag (tag: 1 start, 2 wake, 3 done, 10 acquire,
11 release, 20+v got-value, 100+|sig| signal, 900 user event).
"""
import pytest

import cimba_amd as ca

EXPECTED = {
    # hold ordering: higher event priority first at equal times
    1: [(0.0, 1001), (0.0, 1), (1.5, 1002), (1.5, 2), (3.0, 1003), (3.0, 3)],
    # interrupt delivers user signal 42 into a hold
    2: [(0.0, 1), (1.0, 1003), (1.0, 442)],
    # resource grants by (priority desc, FIFO): p2 (pri 9) beats p1
    3: [(0.0, 10), (5.0, 11), (5.0, 2010), (6.0, 2011), (6.0, 1010), (7.0, 1011)],
    # resource wait timeout -> SIG_TIMEOUT (105), guard entry removed
    4: [(0.0, 10), (2.5, 1105), (5.0, 11)],
    # preempt: taker acquires, holder's hold returns SIG_PREEMPTED (101)
    5: [(0.0, 10), (1.0, 1010), (1.0, 101), (3.0, 1011)],
    # pool greedy partial acquisition: 3 now + 3 when released
    6: [(0.0, 10), (3.0, 11), (3.0, 1010), (4.0, 1011)],
    # buffer: consumer unblocks when level first reaches 5 (t=2)
    7: [(1.0, 11), (2.0, 11), (2.0, 1020), (3.0, 11)],
    # condition: each signal wakes exactly the satisfied waiters
    8: [(1.0, 2), (2.0, 1002)],
    # stop: killed holder's resource passes on; waiter gets SIG_STOPPED (103)
    9: [(0.0, 10), (1.0, 2003), (1.0, 3010), (1.0, 1103), (2.0, 3011)],
    # priority queue: (pri desc, FIFO) -> values 2, 3, then 1
    10: [(1.0, 1022), (1.0, 1023), (1.0, 1021)],
    # wait_event: waiter woken when the event executes
    11: [(5.0, 7900), (5.0, 100)],
    # event cancel wakes waiters with SIG_CANCELLED (104)
    12: [(2.0, 104)],
    # stale grant passes on: granted waiter times out at the same
    # timestamp, the grant must not be lost -> next waiter gets the object
    13: [(1.0, 11), (1.0, 1105), (1.0, 2097)],
    # pool preemption: higher-priority taker reclaims units from a
    # lower-priority holder; victim's hold returns SIG_PREEMPTED (101)
    # and it releases only its remaining holding
    15: [(0.0, 10), (1.0, 1010), (1.0, 101), (1.0, 11), (3.0, 1011)],
    # condition observing a resource guard: the release signal at t=2 is
    # forwarded to the condition and wakes the predicate waiter
    18: [(0.0, 10), (2.0, 11), (2.0, 1002)],
}


@pytest.mark.parametrize("which", sorted(EXPECTED))
def test_scenario_host(which):
    r = ca._C.scenario_host(which)
    assert r["status"] == 0
    assert r["trace"] == EXPECTED[which], (
        f"scenario {which}: {r['trace']} != {EXPECTED[which]}")


@pytest.mark.gpu
@pytest.mark.parametrize("which", sorted(EXPECTED))
def test_scenario_gpu_matches_host(which):
    g = ca._C.scenario_gpu(which)
    assert g["status"] == 0
    assert g["trace"] == EXPECTED[which], (
        f"GPU scenario {which}: {g['trace']} != {EXPECTED[which]}")
    h = ca._C.scenario_host(which)
    assert g["events"] == h["events"]


@pytest.mark.parametrize("which,status", [(16, 1), (17, 2)])
def test_failure_paths_abort_cleanly(which, status):
    """Capacity overflow = trial abort with a status code (SURVEY.md §5.3
    per-block trial-abort flag), never a hang or corruption."""
    r = ca._C.scenario_host(which)
    assert r["status"] == status  # ST_HEAP_FULL=1 / ST_QUEUE_FULL=2


@pytest.mark.gpu
@pytest.mark.parametrize("which,status", [(16, 1), (17, 2)])
def test_failure_paths_abort_cleanly_gpu(which, status):
    r = ca._C.scenario_gpu(which)
    assert r["status"] == status


def _expected_ts():
    # puts at 1,2,3 (len 1,2,3 after; getter takes at 2.5 -> len 1, at 4 ->
    # len 1... trace: put@1 len1, put@2 len2, get@2.5 len1, put@3 len2,
    # get@4 len1; weighted over [1,5): 1*1+2*0.5+1*0.5+2*1+1*1 / 4
    wmean = (1*1.0 + 2*0.5 + 1*0.5 + 2*1.0 + 1*1.0) / 4.0
    return wmean, 5


def test_scenario_timeseries_device_recorder_host():
    r = ca._C.scenario_host(19)
    assert r["status"] == 0
    wmean, n = _expected_ts()
    t, code = r["trace"][-1]
    assert code == n
    assert abs(t - wmean) < 1e-12


@pytest.mark.gpu
def test_scenario_timeseries_device_recorder_gpu():
    r = ca._C.scenario_gpu(19)
    assert r["status"] == 0
    wmean, n = _expected_ts()
    t, code = r["trace"][-1]
    assert code == n
    assert abs(t - wmean) < 1e-12
