"""Gradient checks: hand-written training kernels vs torch autograd."""
import copy
import ctypes

import numpy as np
import pytest
import torch

from rlgpuschedule_amd.env.spec import cluster64


def test_train_so_builds_and_exports():
    from rlgpuschedule_amd.ops.build import build_lib
    so = build_lib("librlsched_train.so")
    lib = ctypes.CDLL(str(so))
    for fn in ("rlsched_fwd_train", "rlsched_ppo_head", "rlsched_bwd_dx",
               "rlsched_bwd_dw", "rlsched_colsum"):
        assert getattr(lib, fn) is not None


def _setup(E=1000, seed=0):
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(seed)
    spec = cluster64()
    D, A, H = spec.obs_dim, spec.num_actions, 256
    policy = MLPPolicy(D, A, H).cuda()
    obs = torch.randn(E, D, device="cuda")
    mask = (torch.rand(E, A, device="cuda") < 0.6).to(torch.uint8)
    mask[:, spec.noop_action] = 1
    rng = torch.Generator(device="cuda").manual_seed(seed + 1)
    # actions valid under mask
    probs = mask.float() + 1e-9
    actions = torch.multinomial(probs, 1, generator=rng).squeeze(1)
    old_logp = -torch.rand(E, device="cuda") * 2
    old_v = torch.randn(E, device="cuda")
    adv = torch.randn(E, device="cuda")
    ret = torch.randn(E, device="cuda")
    return spec, policy, obs, mask, actions, old_logp, old_v, adv, ret


@pytest.mark.gpu
def test_fwd_train_matches_torch():
    from rlgpuschedule_amd.models.hip_train import HEADC, TrainMirror
    from rlgpuschedule_amd.ops.train import fwd_train
    spec, policy, obs, mask, *_ = _setup()
    E, D, H = obs.shape[0], obs.shape[1], 256
    mi = TrainMirror(policy)
    h1 = torch.empty(E, H, device="cuda")
    h2 = torch.empty(E, H, device="cuda")
    head = torch.empty(E, HEADC, device="cuda")
    fwd_train(obs, mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth, mi.bh, h1, h2, head)
    with torch.no_grad():
        t1 = torch.tanh(policy.trunk[0](obs))
        t2 = torch.tanh(policy.trunk[1](t1))
        logits = policy.pi(t2)
        value = policy.v(t2).squeeze(-1)
    torch.testing.assert_close(h1, t1, atol=2e-5, rtol=1e-4)
    torch.testing.assert_close(h2, t2, atol=2e-5, rtol=1e-4)
    torch.testing.assert_close(head[:, : spec.num_actions], logits,
                               atol=2e-4, rtol=1e-4)
    torch.testing.assert_close(head[:, spec.num_actions], value,
                               atol=2e-4, rtol=1e-4)


def _torch_ppo_loss(policy, obs, mask, actions, old_logp, old_v, adv, ret,
                    clip=0.2, vf=0.5, ent_c=0.01):
    from rlgpuschedule_amd.models.policy import evaluate_actions
    logits, value, _ = policy(obs, mask, None)
    logp, entropy = evaluate_actions(logits, actions)
    ratio = (logp - old_logp).exp()
    s1 = ratio * adv
    s2 = ratio.clamp(1 - clip, 1 + clip) * adv
    pi_loss = -torch.min(s1, s2).mean()
    v_clip = old_v + (value - old_v).clamp(-clip, clip)
    v_loss = 0.5 * torch.max((value - ret) ** 2, (v_clip - ret) ** 2).mean()
    return pi_loss + vf * v_loss - ent_c * entropy.mean()


@pytest.mark.gpu
def test_full_backward_matches_autograd():
    from rlgpuschedule_amd.models.hip_train import HEADC, TrainMirror
    from rlgpuschedule_amd.ops.train import (bwd_dw, bwd_dx, colsum,
                                             fwd_train, ppo_head)
    spec, policy, obs, mask, actions, old_logp, old_v, adv, ret = _setup(
        E=1024)
    E, D, H, A = obs.shape[0], obs.shape[1], 256, spec.num_actions
    # torch reference grads
    loss = _torch_ppo_loss(policy, obs, mask, actions, old_logp, old_v,
                           adv, ret)
    loss.backward()
    ref = {n: p.grad.clone() for n, p in policy.named_parameters()}

    mi = TrainMirror(policy)
    z = lambda *s: torch.zeros(*s, device="cuda")
    h1, h2 = z(E, H), z(E, H)
    head, dhead = z(E, HEADC), z(E, HEADC)
    dh2, dh1 = z(E, H), z(E, H)
    stats = z(4)
    fwd_train(obs, mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth, mi.bh, h1, h2, head)
    ppo_head(head, mask, actions, old_logp, old_v, adv, ret,
             0.2, 0.5, 0.01, dhead, stats)
    bwd_dx(dhead, mi.whT, h2, dh2)
    bwd_dx(dh2, mi.w2T, h1, dh1)
    dW1, dW2, dWh = z(D, H), z(H, H), z(H, HEADC)
    db1, db2, dbh = z(H), z(H), z(HEADC)
    bwd_dw(h2, dhead, dWh)
    bwd_dw(h1, dh2, dW2)
    bwd_dw(obs, dh1, dW1)
    colsum(dhead, dbh)
    colsum(dh2, db2)
    colsum(dh1, db1)

    tol = dict(atol=5e-5, rtol=5e-3)
    torch.testing.assert_close(dWh[:, :A].t(), ref["pi.weight"], **tol)
    torch.testing.assert_close(dbh[:A], ref["pi.bias"], **tol)
    torch.testing.assert_close(dWh[:, A], ref["v.weight"][0], **tol)
    torch.testing.assert_close(dbh[A:A + 1], ref["v.bias"], **tol)
    torch.testing.assert_close(dW2.t(), ref["trunk.1.weight"], **tol)
    torch.testing.assert_close(db2, ref["trunk.1.bias"], **tol)
    torch.testing.assert_close(dW1.t(), ref["trunk.0.weight"], **tol)
    torch.testing.assert_close(db1, ref["trunk.0.bias"], **tol)
    # loss value parity (stats[0] + vf*stats[1] - ent*stats[2], /E)
    ref_loss = float(loss.detach())
    hip_loss = float((stats[0] + 0.5 * stats[1] - 0.01 * stats[2]) / E)
    assert abs(ref_loss - hip_loss) < 5e-3


@pytest.mark.gpu
def test_hip_ppo_learner_matches_reference():
    from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    from helpers import make_buf as _make_buf
    torch.manual_seed(0)
    spec = cluster64()
    T, E = 16, 64
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False)
    ref = PPOLearner(pol_a, cfg)
    hip = HipPPOLearner(pol_b, cfg, use_graph=False)
    buf = _make_buf(T, E, spec, "cuda")
    s_ref = ref.update(buf)
    s_hip = hip.update(buf)
    for k in ("pi_loss", "v_loss", "entropy", "clipfrac"):
        assert abs(s_ref[k] - s_hip[k]) < 5e-3, (k, s_ref[k], s_hip[k])
    for (na, pa), (nb, pb) in zip(pol_a.named_parameters(),
                                  pol_b.named_parameters()):
        torch.testing.assert_close(pa, pb, atol=2e-4, rtol=1e-3)


@pytest.mark.gpu
def test_hip_ppo_learner_graph_trains():
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    from helpers import make_buf as _make_buf
    torch.manual_seed(2)
    spec = cluster64()
    pol = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    hip = HipPPOLearner(pol, PPOConfig(epochs=2, minibatches=4),
                        use_graph=True)
    buf = _make_buf(32, 128, spec, "cuda", seed=5)
    for _ in range(3):
        stats = hip.update(buf)
    for k, v in stats.items():
        assert np.isfinite(v), (k, v)
    for p in pol.parameters():
        assert torch.isfinite(p).all()


@pytest.mark.gpu
def test_hip_adam_matches_torch_adam():
    """Fused clip+Adam vs torch clip_grad_norm_ + Adam over 3 steps."""
    from rlgpuschedule_amd.ops.train import HipAdam
    torch.manual_seed(11)
    D, H, A, HEADC = 84, 256, 17, 32
    # params in torch layout
    w1 = torch.randn(H, D, device="cuda")
    b1 = torch.randn(H, device="cuda")
    piw = torch.randn(A, H, device="cuda")
    vw = torch.randn(1, H, device="cuda")
    ref = {k: v.clone() for k, v in
           dict(w1=w1, b1=b1, piw=piw, vw=vw).items()}
    n = D * H + H + H * HEADC
    gflat = torch.zeros(n, device="cuda")
    dW1 = gflat[: D * H].view(D, H)
    db1 = gflat[D * H: D * H + H]
    dWh = gflat[D * H + H:].view(H, HEADC)
    segs = [
        dict(goff=0, R=D, C=H, c0=0, c1=H, param=w1, transposed=True),
        dict(goff=D * H, R=1, C=H, c0=0, c1=H, param=b1, transposed=False),
        dict(goff=D * H + H, R=H, C=HEADC, c0=0, c1=A, param=piw,
             transposed=True),
        dict(goff=D * H + H, R=H, C=HEADC, c0=A, c1=A + 1, param=vw,
             transposed=True),
    ]
    opt = HipAdam(gflat, segs, lr=1e-2, eps=1e-5, max_norm=0.5)
    refp = [torch.nn.Parameter(ref["w1"]), torch.nn.Parameter(ref["b1"]),
            torch.nn.Parameter(ref["piw"]), torch.nn.Parameter(ref["vw"])]
    topt = torch.optim.Adam(refp, lr=1e-2, eps=1e-5)
    g = torch.Generator(device="cuda").manual_seed(3)
    for step in range(3):
        dW1.copy_(torch.randn(D, H, device="cuda", generator=g))
        db1.copy_(torch.randn(H, device="cuda", generator=g))
        dWh.zero_()
        dWh[:, :A].copy_(torch.randn(H, A, device="cuda", generator=g))
        dWh[:, A].copy_(torch.randn(H, device="cuda", generator=g))
        # torch side: same grads in param layout
        refp[0].grad = dW1.t().contiguous()
        refp[1].grad = db1.clone()
        refp[2].grad = dWh[:, :A].t().contiguous()
        refp[3].grad = dWh[:, A:A + 1].t().contiguous()
        opt.step()
        torch.nn.utils.clip_grad_norm_(refp, 0.5)
        topt.step()
    torch.testing.assert_close(w1, refp[0].detach(), atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(b1, refp[1].detach(), atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(piw, refp[2].detach(), atol=1e-6, rtol=1e-5)
    torch.testing.assert_close(vw, refp[3].detach(), atol=1e-6, rtol=1e-5)


@pytest.mark.gpu
def test_bf16_update_path_close_to_fp32():
    """Mixed-precision update: gradient DIRECTION must track the fp32 path
    (cosine > 0.995 per weight tensor after one full-batch update), and a
    few updates stay finite."""
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    from helpers import make_buf
    torch.manual_seed(5)
    spec = cluster64()
    T, E = 16, 128
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    pol_b = copy.deepcopy(pol_a)
    p0 = {n: p.detach().clone() for n, p in pol_a.named_parameters()}
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False)
    l32 = HipPPOLearner(pol_a, cfg, use_graph=False, dtype="fp32")
    l16 = HipPPOLearner(pol_b, cfg, use_graph=False, dtype="bf16")
    buf = make_buf(T, E, spec, "cuda", seed=8)
    l32.update(buf)
    l16.update(buf)
    for (n, pa), (_, pb) in zip(pol_a.named_parameters(),
                                pol_b.named_parameters()):
        da = (pa - p0[n]).flatten()
        db = (pb - p0[n]).flatten()
        cos = torch.nn.functional.cosine_similarity(da, db, dim=0).item()
        assert cos > 0.995, (n, cos)
    # multi-update stability (graph mode)
    pol_c = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    lg = HipPPOLearner(pol_c, PPOConfig(epochs=1, minibatches=2),
                       use_graph=True, dtype="bf16")
    for _ in range(3):
        stats = lg.update(buf)
    for k, v in stats.items():
        assert np.isfinite(v), (k, v)


@pytest.mark.gpu
def test_hip_ppo_kl_penalty_matches_reference():
    """kl_coef>0 (k3 KL-to-behavior) parity vs torch autograd reference."""
    from helpers import make_buf as _make_buf
    from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(11)
    spec = cluster64()
    T, E = 16, 64
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False, kl_coef=0.3)
    ref = PPOLearner(pol_a, cfg)
    hip = HipPPOLearner(pol_b, cfg, use_graph=False)
    buf = _make_buf(T, E, spec, "cuda", seed=8)
    s_ref = ref.update(buf)
    s_hip = hip.update(buf)
    # pi_loss stat includes the kl penalty on both sides
    for k in ("pi_loss", "v_loss", "entropy", "clipfrac"):
        assert abs(s_ref[k] - s_hip[k]) < 5e-3, (k, s_ref[k], s_hip[k])
    for (na, pa), (nb, pb) in zip(pol_a.named_parameters(),
                                  pol_b.named_parameters()):
        torch.testing.assert_close(pa, pb, atol=2e-4, rtol=1e-3)


@pytest.mark.gpu
def test_hip_ppo_entropy_floor_matches_reference():
    """ent_floor/ent_floor_coef (per-row hinge entropy bonus) parity vs
    torch autograd reference. Floor is set mid-range so the gate is
    exercised in BOTH states across rows (row entropies vary with the
    action-mask population)."""
    from helpers import make_buf as _make_buf
    from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(13)
    spec = cluster64()
    T, E = 16, 64
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False,
                    ent_floor=2.0, ent_floor_coef=0.05)
    ref = PPOLearner(pol_a, cfg)
    hip = HipPPOLearner(pol_b, cfg, use_graph=False)
    buf = _make_buf(T, E, spec, "cuda", seed=21)
    s_ref = ref.update(buf)
    s_hip = hip.update(buf)
    for k in ("pi_loss", "v_loss", "entropy", "clipfrac"):
        assert abs(s_ref[k] - s_hip[k]) < 5e-3, (k, s_ref[k], s_hip[k])
    for (na, pa), (nb, pb) in zip(pol_a.named_parameters(),
                                  pol_b.named_parameters()):
        torch.testing.assert_close(pa, pb, atol=3e-4, rtol=1e-3)


@pytest.mark.gpu
def test_hip_a2c_learner_matches_reference():
    """HipA2CLearner (fused head + RMSprop-mode step) vs torch A2CLearner."""
    import copy

    from helpers import make_buf as _make_buf
    from rlgpuschedule_amd.agents.a2c_hip import HipA2CLearner
    from rlgpuschedule_amd.agents.ppo import A2CConfig, A2CLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(5)
    spec = cluster64()
    T, E = 16, 64
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = A2CConfig(norm_reward=False)
    ref = A2CLearner(pol_a, cfg)
    hip = HipA2CLearner(pol_b, cfg, use_graph=False)
    buf = _make_buf(T, E, spec, "cuda", seed=14)
    s_ref = ref.update(buf)
    s_hip = hip.update(buf)
    for k in ("pi_loss", "v_loss", "entropy"):
        assert abs(s_ref[k] - s_hip[k]) < 5e-3, (k, s_ref[k], s_hip[k])
    for (na, pa), (nb, pb) in zip(pol_a.named_parameters(),
                                  pol_b.named_parameters()):
        torch.testing.assert_close(pa, pb, atol=3e-4, rtol=2e-3)


@pytest.mark.gpu
def test_hip_a2c_graph_trains_c4():
    """config c4 (gandiva+elastic+resize) end-to-end on the HIP A2C path."""
    import dataclasses

    import numpy as np

    from rlgpuschedule_amd.trainer import CONFIGS, train
    cfg = dataclasses.replace(CONFIGS["c4_gandiva_a2c"], num_envs=256,
                              jobs_per_env=32)
    out = train(cfg, iterations=5, log_fn=None)
    assert np.isfinite(out["pi_loss"])
    from rlgpuschedule_amd.agents.a2c_hip import HipA2CLearner  # noqa: F401


@pytest.mark.gpu
def test_capture_failure_falls_back_eager(monkeypatch):
    """inject a hipGraph-capture failure and
    assert the learner warns, runs eager, and still trains (the
    agree_all rank-consistency path is the single-rank degenerate case
    here; the 2-rank gloo version is tests/test_dist.py)."""
    import warnings

    from helpers import make_buf as _make_buf
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(3)
    spec = cluster64()
    pol = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    lr = HipPPOLearner(pol, PPOConfig(epochs=1, minibatches=1),
                       use_graph=True)
    monkeypatch.setattr(
        lr, "_capture",
        lambda buf: (_ for _ in ()).throw(RuntimeError("injected")))
    buf = _make_buf(8, 64, spec, "cuda", seed=2)
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        stats = lr.update(buf)
    assert any("capture" in str(x.message).lower() for x in w)
    assert lr.use_graph is False and lr._graph is None
    assert all(abs(v) < 1e6 for v in stats.values())
    lr.update(buf)   # eager steady state keeps working
