"""Fused GRU actor kernel vs torch GRUPolicy reference."""
import numpy as np
import pytest
import torch

from rlgpuschedule_amd.env.spec import cluster256


@pytest.mark.gpu
def test_gru_actor_matches_torch():
    from rlgpuschedule_amd.models.hip_actor import HipActorGRU
    from rlgpuschedule_amd.models.policy import GRUPolicy, sample_categorical
    torch.manual_seed(0)
    spec = cluster256()
    D, A, H = spec.obs_dim, spec.num_actions, 256
    E = 1000
    policy = GRUPolicy(D, A, H).cuda()
    # non-trivial cell state
    for p in policy.cell.parameters():
        with torch.no_grad():
            p.mul_(2.0)
    obs = torch.randn(E, D, device="cuda")
    mask = (torch.rand(E, A, device="cuda") < 0.5).to(torch.uint8)
    mask[:, spec.noop_action] = 1
    u = torch.rand(E, device="cuda")
    hx = torch.randn(E, H, device="cuda") * 0.5
    done_prev = (torch.rand(E, device="cuda") < 0.3).to(torch.uint8)

    actor = HipActorGRU(policy)
    actions = torch.empty(E, dtype=torch.int64, device="cuda")
    logp = torch.empty(E, device="cuda")
    value = torch.empty(E, device="cuda")
    hx_out = torch.empty(E, H, device="cuda")
    actor.act(obs, mask, u, done_prev, hx, hx_out, actions, logp, value)

    with torch.no_grad():
        hx_ref = hx * (1.0 - done_prev.float()).unsqueeze(-1)
        logits, v_ref, hx_new = policy(obs, mask, hx_ref)
        a_ref, lp_ref, _ = sample_categorical(logits, u)

    torch.testing.assert_close(hx_out, hx_new, atol=5e-5, rtol=1e-4)
    torch.testing.assert_close(value, v_ref, atol=3e-4, rtol=1e-3)
    agree = actions == a_ref
    assert agree.float().mean().item() > 0.995
    torch.testing.assert_close(logp[agree], lp_ref[agree], atol=3e-4,
                               rtol=1e-3)


@pytest.mark.gpu
def test_gru_rollout_engine_and_recurrent_ppo():
    from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
    from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
    from rlgpuschedule_amd.env.hip_env import HipVectorEnv
    from rlgpuschedule_amd.models.policy import GRUPolicy
    from rlgpuschedule_amd.env.spec import cluster64
    spec = cluster64(horizon=2_000)
    env = HipVectorEnv(spec, 128, trace_kind="philly", jobs_per_env=16,
                       seed=3)
    policy = GRUPolicy(spec.obs_dim, spec.num_actions, 64).cuda()
    eng = GpuRolloutEngine(env, policy, T=16, seed=0, use_graph=True)
    learner = PPOLearner(policy, PPOConfig(epochs=1, minibatches=2))
    for _ in range(3):
        buf = eng.collect()
        stats = learner.update(buf)
    assert np.isfinite(stats["pi_loss"])
    # action legality + hidden-state reset plumbing
    for t in range(16):
        chosen = buf.mask[t].gather(1, buf.actions[t].unsqueeze(1))
        assert (chosen == 1).all()
    assert torch.isfinite(buf.h0).all()


@pytest.mark.gpu
def test_gru_bf16_actor_close_to_torch():
    from rlgpuschedule_amd.models.hip_actor import HipActorGRUBf16
    from rlgpuschedule_amd.models.policy import GRUPolicy, sample_categorical
    import torch
    torch.manual_seed(7)
    spec = cluster256()
    D, A, H = spec.obs_dim, spec.num_actions, 256
    E = 1000
    policy = GRUPolicy(D, A, H).cuda()
    obs = torch.randn(E, D, device="cuda")
    mask = (torch.rand(E, A, device="cuda") < 0.5).to(torch.uint8)
    mask[:, spec.noop_action] = 1
    u = torch.rand(E, device="cuda")
    hx = torch.randn(E, H, device="cuda") * 0.5
    done_prev = (torch.rand(E, device="cuda") < 0.3).to(torch.uint8)
    actor = HipActorGRUBf16(policy)
    actions = torch.empty(E, dtype=torch.int64, device="cuda")
    logp = torch.empty(E, device="cuda")
    value = torch.empty(E, device="cuda")
    hx_out = torch.empty(E, H, device="cuda")
    actor.act(obs, mask, u, done_prev, hx, hx_out, actions, logp, value)
    with torch.no_grad():
        hx_ref = hx * (1.0 - done_prev.float()).unsqueeze(-1)
        logits, v_ref, hx_new = policy(obs, mask, hx_ref)
        a_ref, lp_ref, _ = sample_categorical(logits, u)
    torch.testing.assert_close(hx_out, hx_new, atol=5e-2, rtol=5e-2)
    torch.testing.assert_close(value, v_ref, atol=8e-2, rtol=8e-2)
    assert (actions == a_ref).float().mean().item() > 0.95
    chosen = mask.gather(1, actions.unsqueeze(1))
    assert (chosen == 1).all()


@pytest.mark.gpu
def test_gru_sharded_engine():
    import torch
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_gru_hip import HipGRUPPOLearner
    from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
    from rlgpuschedule_amd.env.hip_env import HipVectorEnv
    from rlgpuschedule_amd.env.spec import cluster64
    from rlgpuschedule_amd.models.policy import GRUPolicy
    spec = cluster64(horizon=50_000)
    envs = [HipVectorEnv(spec, 128, trace_kind="philly", jobs_per_env=24,
                         seed=21 + s) for s in range(2)]
    policy = GRUPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    eng = GpuRolloutEngine(envs, policy, T=16, seed=1, use_graph=True,
                           actor_dtype="bf16")
    learner = HipGRUPPOLearner(policy, PPOConfig(epochs=1, minibatches=2),
                               use_graph=False)
    for _ in range(3):
        buf = eng.collect()
        stats = learner.update(buf)
    assert np.isfinite(stats["pi_loss"])
    for t in range(16):
        chosen = buf.mask[t].gather(1, buf.actions[t].unsqueeze(1))
        assert (chosen == 1).all()
