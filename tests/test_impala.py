"""V-trace + IMPALA tests."""
import dataclasses

import numpy as np
import pytest
import torch

from rlgpuschedule_amd.ops.gae_ref import gae_advantages, vtrace_reference


def test_vtrace_onpolicy_reduces_to_gae():
    torch.manual_seed(0)
    T, E = 24, 7
    r = torch.randn(T, E)
    v = torch.randn(T, E)
    d = (torch.rand(T, E) < 0.1).to(torch.uint8)
    lv = torch.randn(E)
    lp = torch.randn(T, E)
    gamma, lam = 0.99, 0.95
    vs, pg = vtrace_reference(r, v, d, lv, lp, lp, gamma, 1.0, 1.0, lam)
    adv, ret = gae_advantages(r, v, d, lv, gamma, lam)
    torch.testing.assert_close(vs - v, adv, atol=1e-5, rtol=1e-5)


def test_vtrace_truncation_bounds():
    # very off-policy: rho should truncate at rho_bar
    T, E = 8, 4
    r = torch.ones(T, E)
    v = torch.zeros(T, E)
    d = torch.zeros(T, E, dtype=torch.uint8)
    lv = torch.zeros(E)
    lp_cur = torch.full((T, E), 5.0)
    lp_beh = torch.zeros(T, E)   # ratio = e^5, truncated to 1.0
    vs1, pg1 = vtrace_reference(r, v, d, lv, lp_cur, lp_beh, 0.9, 1.0, 1.0, 1.0)
    vs2, pg2 = vtrace_reference(r, v, d, lv, lp_beh, lp_beh, 0.9, 1.0, 1.0, 1.0)
    torch.testing.assert_close(vs1, vs2)  # truncation makes them identical


def test_impala_cpu_trains():
    from rlgpuschedule_amd.trainer import CONFIGS, train
    cfg = dataclasses.replace(
        CONFIGS["c5_impala"], num_envs=2, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=16)
    out = train(cfg, iterations=3, log_fn=None)
    assert np.isfinite(out["pi_loss"])


@pytest.mark.gpu
def test_vtrace_kernel_matches_reference():
    from rlgpuschedule_amd.ops.nn import vtrace_scan
    torch.manual_seed(3)
    T, E = 80, 700
    r = torch.randn(T, E, device="cuda")
    v = torch.randn(T, E, device="cuda")
    d = (torch.rand(T, E, device="cuda") < 0.05).to(torch.uint8)
    lv = torch.randn(E, device="cuda")
    lpc = torch.randn(T, E, device="cuda") * 0.3
    lpb = torch.randn(T, E, device="cuda") * 0.3
    vs, pg = vtrace_scan(r, v, d, lv, lpc, lpb, 0.99, 1.0, 1.0, 0.95)
    vs_r, pg_r = vtrace_reference(r, v, d, lv, lpc, lpb, 0.99, 1.0, 1.0, 0.95)
    torch.testing.assert_close(vs, vs_r, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(pg, pg_r, atol=1e-5, rtol=1e-5)


@pytest.mark.gpu
def test_impala_gpu_async_pipeline():
    from rlgpuschedule_amd.trainer import CONFIGS, train
    cfg = dataclasses.replace(
        CONFIGS["c5_impala"], num_envs=256, jobs_per_env=32, rollout_T=16)
    out = train(cfg, iterations=4, log_fn=None)
    assert np.isfinite(out["pi_loss"])
    assert "episode_stats" in out


@pytest.mark.gpu
def test_hip_vtrace_learner_matches_reference():
    import copy
    from helpers import make_buf
    from rlgpuschedule_amd.agents.impala import ImpalaConfig, VtraceLearner
    from rlgpuschedule_amd.agents.impala_hip import HipVtraceLearner
    from rlgpuschedule_amd.env.spec import cluster64
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(0)
    spec = cluster64()
    T, E = 16, 64
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = ImpalaConfig(norm_reward=False)
    ref = VtraceLearner(pol_a, cfg)
    hip = HipVtraceLearner(pol_b, cfg, use_graph=False)
    buf = make_buf(T, E, spec, "cuda", seed=2)
    s_ref = ref.update(buf)
    s_hip = hip.update(buf)
    for k in ("pi_loss", "v_loss", "entropy"):
        assert abs(s_ref[k] - s_hip[k]) < 5e-3, (k, s_ref[k], s_hip[k])
    # note: ref uses torch Adam + clip; hip uses the fused Adam — same math
    for (na, pa), (_, pb) in zip(pol_a.named_parameters(),
                                 pol_b.named_parameters()):
        torch.testing.assert_close(pa, pb, atol=3e-4, rtol=2e-3), na


@pytest.mark.gpu
def test_impala_gpu_hip_learner_pipeline():
    import dataclasses
    from rlgpuschedule_amd.trainer import CONFIGS, train
    cfg = dataclasses.replace(
        CONFIGS["c5_impala"], num_envs=512, jobs_per_env=32, rollout_T=16)
    out = train(cfg, iterations=6, log_fn=None)
    assert np.isfinite(out["pi_loss"])


def test_vtrace_correction_engages_under_staleness():
    """with behavior_sync=4 the rollout is genuinely
    off-policy (stored behavior logp != current policy logp) and the
    V-trace arm's update differs from the plain-GAE (use_vtrace=False)
    arm; with behavior_sync=1 (on-policy, lam=1) the two coincide."""
    import copy
    import dataclasses

    import torch

    from rlgpuschedule_amd.agents.impala import ImpalaConfig, VtraceLearner
    from rlgpuschedule_amd.agents.rollout import Collector
    from rlgpuschedule_amd.env.spec import cluster16
    from rlgpuschedule_amd.models.policy import evaluate_actions
    from rlgpuschedule_amd.trainer import CONFIGS, make_env, make_policy

    cfg = dataclasses.replace(
        CONFIGS["c5_impala"], spec=cluster16(), num_envs=8, jobs_per_env=16,
        rollout_T=16, use_hip_env=False, device="cpu", hidden=32, seed=7)
    dev = torch.device("cpu")

    def run(sync, vtrace, iters=8):
        torch.manual_seed(123)
        icfg = ImpalaConfig(behavior_sync=sync, use_vtrace=vtrace)
        env = make_env(cfg, dev, 0)
        pol = make_policy(cfg, dev)
        lr = VtraceLearner(pol, icfg)
        behavior = copy.deepcopy(pol)
        col = Collector(env, behavior, cfg.rollout_T, dev, seed=1)
        stale = 0.0
        for it in range(iters):
            if it % icfg.behavior_sync == 0:
                behavior.load_state_dict(pol.state_dict())
            buf = col.collect()
            with torch.no_grad():
                T, E = buf.reward.shape
                logits, _, _ = pol(buf.obs.reshape(T * E, -1),
                                   buf.mask.reshape(T * E, -1), None)
                lp_cur, _ = evaluate_actions(logits,
                                             buf.actions.reshape(T * E))
                stale = max(stale, float(
                    (lp_cur - buf.logp.reshape(T * E)).abs().max()))
            lr.update(buf)
        psum = float(sum(p.detach().sum() for p in pol.parameters()))
        return stale, psum

    # staleness is real at sync=4
    stale4_vt, p4_vt = run(4, True)
    assert stale4_vt > 1e-4, "behavior_sync=4 must produce off-policy data"
    # and the correction changes the learned parameters
    _, p4_gae = run(4, False)
    assert p4_vt != p4_gae
    # on-policy (sync=1, lam=1): V-trace reduces to GAE -> identical runs
    _, p1_vt = run(1, True)
    _, p1_gae = run(1, False)
    assert abs(p1_vt - p1_gae) < 1e-5
