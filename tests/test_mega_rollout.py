"""Fused whole-rollout megakernel (csrc/rollout_mega.hip): must be
BIT-IDENTICAL to the two-kernel rollout path given the same policy
weights, env seed and uniforms — same fp32 MFMA actor math, same env
transition code (shared env_device.h)."""
import ctypes

import pytest
import torch

from rlgpuschedule_amd.env.spec import cluster16, cluster64


def test_mega_so_builds_and_exports():
    from rlgpuschedule_amd.ops.build import build_lib
    so = build_lib("librlsched_mega.so")
    lib = ctypes.CDLL(str(so))
    assert lib.rlsched_mega_rollout is not None
    assert lib.rlsched_source_hash is not None


def _mk(spec, E, T, seed, mode, tkw):
    from rlgpuschedule_amd.agents.rollout_gpu import GpuRolloutEngine
    from rlgpuschedule_amd.env.hip_env import HipVectorEnv
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(900 + seed)
    env = HipVectorEnv(spec, E, trace_kind="philly", jobs_per_env=32,
                       seed=seed, **tkw)
    policy = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    eng = GpuRolloutEngine(env, policy, T, seed=77, use_graph=False,
                           rollout=mode)
    return env, policy, eng


@pytest.mark.gpu
@pytest.mark.parametrize("spec_kw,tkw", [
    (dict(), dict()),
    (dict(gandiva=True, elastic=True, elastic_resize=True, obs_version=2),
     dict(elastic=True, fail_frac=0.1)),
])
def test_mega_rollout_bit_parity(spec_kw, tkw):
    spec = cluster64(horizon=50_000, **spec_kw)
    E, T = 128, 16
    env_a, pol_a, eng_a = _mk(spec, E, T, seed=5, mode="twokernel", tkw=tkw)
    env_b, pol_b, eng_b = _mk(spec, E, T, seed=5, mode="mega", tkw=tkw)
    pol_b.load_state_dict(pol_a.state_dict())
    eng_b.actors[0].refresh()
    for it in range(3):
        a = eng_a.collect()
        b = eng_b.collect()
        torch.cuda.synchronize()
        for name in ("obs", "mask", "actions", "logp", "value", "reward",
                     "done", "last_value"):
            ta, tb = getattr(a, name), getattr(b, name)
            assert torch.equal(ta, tb), \
                f"iter {it}: {name} diverges ({(ta != tb).sum()} cells)"
        assert torch.equal(eng_a.obs_last, eng_b.obs_last), f"iter {it}"
        assert torch.equal(eng_a.mask_last, eng_b.mask_last)
    sa, sb = env_a.episode_stats(), env_b.episode_stats()
    assert sa == sb


@pytest.mark.gpu
def test_mega_rollout_trains():
    """HipPPOLearner on mega-collected rollouts: 5 iterations, finite."""
    import numpy as np

    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_hip import HipPPOLearner
    spec = cluster64(horizon=50_000)
    env, pol, eng = _mk(spec, 256, 16, seed=9, mode="mega", tkw={})
    learner = HipPPOLearner(pol, PPOConfig(epochs=1, minibatches=2),
                            use_graph=False)
    for _ in range(5):
        stats = learner.update(eng.collect())
        eng.actors[0].refresh()
    torch.cuda.synchronize()
    for k, v in stats.items():
        assert np.isfinite(v), (k, v)
