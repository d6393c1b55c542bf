"""FusedPPOLearner (HIP loss kernel + graph) vs reference PPOLearner."""
import copy

import numpy as np
import pytest
import torch

from rlgpuschedule_amd.env.spec import cluster64


from helpers import make_buf as _make_buf


@pytest.mark.gpu
def test_fused_ppo_matches_reference_fullbatch():
    from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
    from rlgpuschedule_amd.agents.ppo_gpu import FusedPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(0)
    spec = cluster64()
    T, E = 16, 64
    pol_a = MLPPolicy(spec.obs_dim, spec.num_actions, 128).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False)
    ref = PPOLearner(pol_a, cfg)
    fused = FusedPPOLearner(pol_b, cfg, use_graph=False)
    buf = _make_buf(T, E, spec, "cuda")
    s_ref = ref.update(buf)
    s_fused = fused.update(buf)
    for k in ("pi_loss", "v_loss", "entropy", "clipfrac"):
        assert abs(s_ref[k] - s_fused[k]) < 5e-3, (k, s_ref[k], s_fused[k])
    for pa, pb in zip(pol_a.parameters(), pol_b.parameters()):
        torch.testing.assert_close(pa, pb, atol=1e-4, rtol=1e-3)


@pytest.mark.gpu
def test_fused_ppo_graph_mode_trains():
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_gpu import FusedPPOLearner
    from rlgpuschedule_amd.models.policy import MLPPolicy
    torch.manual_seed(1)
    spec = cluster64()
    T, E = 32, 128
    pol = MLPPolicy(spec.obs_dim, spec.num_actions, 256).cuda()
    fused = FusedPPOLearner(pol, PPOConfig(epochs=2, minibatches=4),
                            use_graph=True)
    p0 = [p.detach().clone() for p in pol.parameters()]
    buf = _make_buf(T, E, spec, "cuda", seed=3)
    for i in range(3):
        stats = fused.update(buf)
        for k, v in stats.items():
            assert np.isfinite(v), (k, v)
    changed = any(not torch.equal(a, b)
                  for a, b in zip(p0, pol.parameters()))
    assert changed
    for p in pol.parameters():
        assert torch.isfinite(p).all()
