"""CPU tests for the training stack: GAE, PPO/A2C, rollout, checkpoints."""
import numpy as np
import pytest
import torch

from rlgpuschedule_amd.agents.ppo import (A2CConfig, A2CLearner, PPOConfig,
                                          PPOLearner)
from rlgpuschedule_amd.agents.rollout import Collector, TorchEnvAdapter
from rlgpuschedule_amd.ckpt.io import load_policy, save_policy
from rlgpuschedule_amd.env.gym_api import CpuVectorEnv
from rlgpuschedule_amd.env.spec import cluster16
from rlgpuschedule_amd.models.policy import (GRUPolicy, MLPPolicy,
                                             sample_categorical)
from rlgpuschedule_amd.ops.gae_ref import gae_advantages


def test_gae_matches_hand_rollout():
    # 3-step, 1-env hand computation
    r = torch.tensor([[1.0], [1.0], [1.0]])
    v = torch.tensor([[0.5], [0.4], [0.3]])
    d = torch.tensor([[0], [0], [0]], dtype=torch.uint8)
    last_v = torch.tensor([0.2])
    g, l = 0.9, 0.8
    adv, ret = gae_advantages(r, v, d, last_v, g, l)
    d2 = 1 + g * 0.2 - 0.3
    d1 = 1 + g * 0.3 - 0.4
    d0 = 1 + g * 0.4 - 0.5
    a2 = d2
    a1 = d1 + g * l * a2
    a0 = d0 + g * l * a1
    np.testing.assert_allclose(adv.squeeze().numpy(), [a0, a1, a2], rtol=1e-6)
    np.testing.assert_allclose(ret.squeeze().numpy(),
                               (adv + v).squeeze().numpy(), rtol=1e-6)


def test_gae_resets_at_done():
    r = torch.zeros(3, 1)
    v = torch.zeros(3, 1)
    d = torch.tensor([[0], [1], [0]], dtype=torch.uint8)
    last_v = torch.tensor([100.0])
    adv, _ = gae_advantages(r, v, d, last_v, 0.99, 0.95)
    assert adv[1, 0].item() == 0.0  # bootstrap cut at done
    assert adv[2, 0].item() != 0.0  # last step bootstraps last_v


def test_sample_categorical_inverse_cdf():
    logits = torch.log(torch.tensor([[0.1, 0.2, 0.7]]))
    a, lp, ent = sample_categorical(logits, torch.tensor([0.05]))
    assert a.item() == 0
    a, _, _ = sample_categorical(logits, torch.tensor([0.25]))
    assert a.item() == 1
    a, _, _ = sample_categorical(logits, torch.tensor([0.95]))
    assert a.item() == 2
    assert abs(lp.item() - np.log(0.1)) < 1e-5


def _tiny_env(num_envs=4, **kw):
    spec = cluster16(horizon=10_000)
    venv = CpuVectorEnv(spec, num_envs, trace_kind="philly", jobs_per_env=16,
                        seed=3, mean_interarrival_s=30.0,
                        median_duration_s=60.0, **kw)
    return spec, TorchEnvAdapter(venv, "cpu")


@pytest.mark.parametrize("policy_cls", [MLPPolicy, GRUPolicy])
def test_ppo_runs_and_losses_finite(policy_cls):
    spec, env = _tiny_env()
    pol = policy_cls(spec.obs_dim, spec.num_actions, 32)
    learner = PPOLearner(pol, PPOConfig(epochs=2, minibatches=2))
    col = Collector(env, pol, T=16, device="cpu", seed=0)
    for _ in range(2):
        buf = col.collect()
        stats = learner.update(buf)
    for k, val in stats.items():
        assert np.isfinite(val), (k, val)


def test_a2c_runs():
    spec, env = _tiny_env()
    pol = MLPPolicy(spec.obs_dim, spec.num_actions, 32)
    learner = A2CLearner(pol, A2CConfig())
    col = Collector(env, pol, T=16, device="cpu", seed=0)
    stats = learner.update(col.collect())
    assert np.isfinite(stats["pi_loss"])


def test_rollout_buffer_consistency():
    spec, env = _tiny_env()
    pol = MLPPolicy(spec.obs_dim, spec.num_actions, 32)
    col = Collector(env, pol, T=8, device="cpu", seed=0)
    buf = col.collect()
    assert buf.obs.shape == (8, 4, spec.obs_dim)
    # logp corresponds to stored action under stored obs/mask
    logits, v, _ = pol(buf.obs[0], buf.mask[0], None)
    from rlgpuschedule_amd.models.policy import evaluate_actions
    lp, _ = evaluate_actions(logits, buf.actions[0])
    np.testing.assert_allclose(lp.detach().numpy(),
                               buf.logp[0].detach().numpy(), atol=1e-5)
    np.testing.assert_allclose(v.detach().numpy(),
                               buf.value[0].detach().numpy(), atol=1e-5)
    # masked action legality: every stored action was valid under its mask
    for t in range(8):
        chosen = buf.mask[t].gather(1, buf.actions[t].unsqueeze(1))
        assert (chosen == 1).all()


def test_checkpoint_roundtrip_pt_and_safetensors(tmp_path):
    spec, _ = _tiny_env()
    pol = MLPPolicy(spec.obs_dim, spec.num_actions, 32)
    pol2 = MLPPolicy(spec.obs_dim, spec.num_actions, 32)
    for suffix in (".pt", ".safetensors"):
        p = tmp_path / f"ck{suffix}"
        save_policy(p, pol, spec=spec, meta={"iter": 7})
        with torch.no_grad():
            for q in pol2.parameters():
                q.add_(1.0)
        meta = load_policy(p, pol2)
        for a, b in zip(pol.parameters(), pol2.parameters()):
            assert torch.equal(a, b)
        if suffix == ".pt":
            assert meta["meta"]["iter"] == 7
            assert meta["env_spec"]["num_nodes"] == spec.num_nodes


def test_trainer_config1_random_cpu():
    from rlgpuschedule_amd.trainer import CONFIGS, train
    import dataclasses
    cfg = dataclasses.replace(CONFIGS["c1_cpu_random"], num_envs=2,
                              jobs_per_env=16, rollout_T=8)
    out = train(cfg, iterations=2, log_fn=None)
    assert "episode_stats" in out


def test_trainer_config2_smoke_cpu():
    from rlgpuschedule_amd.trainer import CONFIGS, train
    import dataclasses
    cfg = dataclasses.replace(
        CONFIGS["c2_ppo_mlp"], num_envs=2, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=32)
    out = train(cfg, iterations=2, log_fn=None)
    assert np.isfinite(out["pi_loss"])


def test_multi_seed_selection_cpu():
    """train_multi_seed: selection snapshots, disjoint held-out eval, and
    winner chosen by selection-pool JCT (never the held-out number)."""
    import dataclasses

    from rlgpuschedule_amd.agents.selection import train_multi_seed
    from rlgpuschedule_amd.trainer import CONFIGS
    cfg = dataclasses.replace(
        CONFIGS["c2_ppo_mlp"], num_envs=4, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=16)
    cfg = dataclasses.replace(
        cfg, ppo=dataclasses.replace(cfg.ppo, epochs=1, minibatches=1,
                                     kl_coef=0.2))
    res = train_multi_seed(cfg, [1, 2], iterations=4, select_every=2,
                           select_kw=dict(num_envs=2, min_episodes=2,
                                          max_steps=3000),
                           eval_kw=dict(num_envs=2, min_episodes=2,
                                        max_steps=3000))
    assert len(res["runs"]) == 2
    for r in res["runs"]:
        assert r["selected_at"] > 0
        assert r["heldout_jct_us"] > 0
    assert res["best_seed"] in (1, 2)
    win = min(res["runs"], key=lambda r: r["selection_jct_us"])
    assert res["best_seed"] == win["seed"]
    assert res["best_selection_jct_us"] == win["selection_jct_us"]
    # best_state loads into a fresh policy
    import torch

    from rlgpuschedule_amd.trainer import make_policy
    pol = make_policy(cfg, torch.device("cpu"))
    pol.load_state_dict(res["best_state"])


def test_cli_train_sweep_cpu(tmp_path):
    from rlgpuschedule_amd.cli import main
    rc = main(["train-sweep", "--config", "c2_ppo_mlp", "--seeds", "1",
               "--iterations", "2", "--envs", "2", "--jobs-per-env", "8",
               "--select-every", "1",
               "--save-best", str(tmp_path / "best.pt")])
    assert rc == 0


def test_multi_seed_selection_disabled_is_json_safe():
    """select_every=0: no snapshots, result must serialize as strict JSON
    (no Infinity) and carry the final policies' held-out evals."""
    import dataclasses
    import json

    from rlgpuschedule_amd.agents.selection import train_multi_seed
    from rlgpuschedule_amd.trainer import CONFIGS
    cfg = dataclasses.replace(
        CONFIGS["c2_ppo_mlp"], num_envs=2, jobs_per_env=8, rollout_T=8,
        use_hip_env=False, device="cpu", hidden=16)
    cfg = dataclasses.replace(
        cfg, ppo=dataclasses.replace(cfg.ppo, epochs=1, minibatches=1))
    res = train_multi_seed(cfg, [1], iterations=2, select_every=0,
                           eval_kw=dict(num_envs=2, min_episodes=2,
                                        max_steps=3000))
    assert "best_seed" not in res
    json.dumps(res, allow_nan=False)   # raises on inf/nan
