"""Property-based invariant tests for the CPU oracle env (hypothesis).

The differential HIP/CPU parity suite (test_hip_env_parity.py) proves
GPU == CPU on sampled action streams; these tests probe the CPU oracle
itself across randomized specs, traces, and masked-action streams —
invariants must hold at EVERY step, on every generated configuration,
including the Gandiva and elastic variants.
"""
import numpy as np
from hypothesis import given, settings, strategies as st

from rlgpuschedule_amd.env.cpu_ref import ClusterEnvRef
from rlgpuschedule_amd.env.spec import EnvSpec
from rlgpuschedule_amd.traces.synthetic import philly_like


def _mk_env(nodes, gpn, gandiva, elastic, jobs, seed, obs_version=1,
            resize=False, K=16, P=256, S=8):
    spec = EnvSpec(num_nodes=nodes, gpus_per_node=gpn, horizon=80_000,
                   gandiva=gandiva, elastic=elastic, obs_version=obs_version,
                   elastic_resize=resize and elastic, queue_visible=K,
                   queue_cap=P, max_span=S)
    tr = philly_like(jobs, seed=seed, max_job_gpus=spec.max_job_gpus,
                     mean_interarrival_s=40.0, median_duration_s=300.0,
                     elastic=elastic)
    return spec, ClusterEnvRef(spec, tr)


@settings(max_examples=60, deadline=None)
@given(nodes=st.integers(1, 6), gpn=st.sampled_from([2, 4, 8]),
       gandiva=st.booleans(), elastic=st.booleans(),
       jobs=st.integers(4, 48), seed=st.integers(0, 10_000),
       aseed=st.integers(0, 10_000), obs_version=st.sampled_from([1, 2]),
       resize=st.booleans(), K=st.sampled_from([2, 8, 16, 32]),
       P=st.sampled_from([8, 64, 256]), S=st.sampled_from([1, 2, 8]))
def test_invariants_hold_under_random_masked_actions(
        nodes, gpn, gandiva, elastic, jobs, seed, aseed, obs_version, resize,
        K, P, S):
    if K > P:
        K = P
    spec, env = _mk_env(nodes, gpn, gandiva, elastic, jobs, seed,
                        obs_version, resize, K=K, P=P, S=S)
    env.reset()
    rng = np.random.default_rng(aseed)
    for t in range(200):
        mask = env.action_mask()
        valid = np.flatnonzero(mask)
        assert valid.size >= 1          # noop is always legal
        a = int(rng.choice(valid))
        obs, rew, done, info = env.step(a)
        env.validate_invariants()
        assert obs.shape == (spec.obs_dim,)
        assert np.isfinite(obs).all()
        assert np.isfinite(rew)
        if done:
            break


@settings(max_examples=15, deadline=None)
@given(nodes=st.integers(1, 4), gpn=st.sampled_from([4, 8]),
       jobs=st.integers(4, 32), seed=st.integers(0, 10_000))
def test_deterministic_replay_same_fingerprint(nodes, gpn, jobs, seed):
    spec, env1 = _mk_env(nodes, gpn, False, False, jobs, seed)
    _, env2 = _mk_env(nodes, gpn, False, False, jobs, seed)
    env1.reset()
    env2.reset()
    rng1 = np.random.default_rng(77)
    rng2 = np.random.default_rng(77)
    for t in range(64):
        a1 = int(rng1.choice(np.flatnonzero(env1.action_mask())))
        a2 = int(rng2.choice(np.flatnonzero(env2.action_mask())))
        assert a1 == a2
        _, r1, d1, _ = env1.step(a1)
        _, r2, d2, _ = env2.step(a2)
        assert r1 == r2 and d1 == d2
        if d1:
            break
    f1, f2 = env1.state_fingerprint(), env2.state_fingerprint()
    assert f1.keys() == f2.keys()
    for k in f1:
        assert np.array_equal(np.asarray(f1[k]), np.asarray(f2[k])), k


@settings(max_examples=15, deadline=None)
@given(jobs=st.integers(8, 48), seed=st.integers(0, 10_000))
def test_all_jobs_eventually_complete_under_greedy_schedule(jobs, seed):
    """Schedule-whenever-possible must drain the whole trace before the
    horizon cap (work conservation: nothing is lost or stuck)."""
    spec, env = _mk_env(2, 8, False, False, jobs, seed)
    env.reset()
    done = False
    for t in range(5000):
        mask = env.action_mask()
        sched = np.flatnonzero(mask[: spec.queue_visible])
        a = int(sched[0]) if sched.size else spec.noop_action
        _, _, done, info = env.step(a)
        if done:
            break
    assert done
    assert env.completed == jobs


@settings(max_examples=20, deadline=None)
@given(n=st.integers(4, 64), seed=st.integers(0, 100_000),
       maxg=st.sampled_from([8, 16, 64]), elastic=st.booleans(),
       fail=st.sampled_from([0.0, 0.1]))
def test_trace_generator_wellformed(n, seed, maxg, elastic, fail):
    """philly_like invariants: sorted arrivals, positive work, demands in
    [1, max_job_gpus], 1 <= min_gpus <= gpus (== gpus when not elastic)."""
    tr = philly_like(n, seed=seed, max_job_gpus=maxg, elastic=elastic,
                     fail_frac=fail)
    assert len(tr) == n
    arr = np.asarray(tr.arrival_us)
    assert (np.diff(arr) >= 0).all()
    assert (np.asarray(tr.work_us) >= 1).all()
    g = np.asarray(tr.gpus)
    mg = np.asarray(tr.min_gpus)
    assert ((g >= 1) & (g <= maxg)).all()
    assert ((mg >= 1) & (mg <= g)).all()
    if not elastic:
        assert (mg == g).all()
    # determinism
    tr2 = philly_like(n, seed=seed, max_job_gpus=maxg, elastic=elastic,
                      fail_frac=fail)
    assert np.array_equal(arr, np.asarray(tr2.arrival_us))
    assert np.array_equal(g, np.asarray(tr2.gpus))
