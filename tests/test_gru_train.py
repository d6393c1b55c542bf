"""Gradient checks for the GRU BPTT kernels vs torch autograd."""
import copy

import numpy as np
import pytest
import torch

from rlgpuschedule_amd.env.spec import cluster64


def _mk(T, Emb, D, H, A, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    obs = torch.randn(T, Emb, D, generator=g).cuda()
    done = (torch.rand(T, Emb, generator=g) < 0.08).to(torch.uint8).cuda()
    h0 = (torch.randn(Emb, H, generator=g) * 0.5).cuda()
    return obs, done, h0


@pytest.mark.gpu
def test_gru_seq_fwd_matches_torch_replay():
    from rlgpuschedule_amd.models.gru_train_mirror import GruTrainMirror
    from rlgpuschedule_amd.models.policy import GRUPolicy
    from rlgpuschedule_amd.ops.gru import gru_seq_fwd
    torch.manual_seed(1)
    T, Emb, H = 12, 96, 256
    spec = cluster64()
    D, A = spec.obs_dim, spec.num_actions
    pol = GRUPolicy(D, A, H).cuda()
    obs, done, h0 = _mk(T, Emb, D, H, A)
    mi = GruTrainMirror(pol)
    B = T * Emb
    z = lambda: torch.zeros(B, H, device="cuda")
    xs, rs, zs, ns, ghns, hprevs, hs = [z() for _ in range(7)]
    gru_seq_fwd(obs, done, h0, mi.wtin, mi.bin, mi.wih, mi.bih, mi.whh,
                mi.bhh, xs, rs, zs, ns, ghns, hprevs, hs)
    # torch replay
    with torch.no_grad():
        hx = h0.clone()
        for t in range(T):
            x = torch.tanh(pol.inp(obs[t]))
            hx_new = pol.cell(x, hx)
            torch.testing.assert_close(
                xs.view(T, Emb, H)[t], x, atol=3e-5, rtol=1e-4)
            torch.testing.assert_close(
                hs.view(T, Emb, H)[t], hx_new, atol=5e-5, rtol=1e-4)
            torch.testing.assert_close(
                hprevs.view(T, Emb, H)[t], hx, atol=5e-5, rtol=1e-4)
            hx = hx_new * (1.0 - done[t].float()).unsqueeze(-1)


@pytest.mark.gpu
def test_hip_gru_learner_matches_reference():
    from helpers import make_buf
    from rlgpuschedule_amd.agents.ppo import PPOConfig, PPOLearner
    from rlgpuschedule_amd.agents.ppo_gru_hip import HipGRUPPOLearner
    from rlgpuschedule_amd.models.policy import GRUPolicy
    torch.manual_seed(0)
    spec = cluster64()
    T, E, H = 12, 64, 256
    pol_a = GRUPolicy(spec.obs_dim, spec.num_actions, H).cuda()
    pol_b = copy.deepcopy(pol_a)
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False)
    ref = PPOLearner(pol_a, cfg)
    hip = HipGRUPPOLearner(pol_b, cfg, use_graph=False)
    buf = make_buf(T, E, spec, "cuda", seed=4)
    # recurrent buffers need h0 and realistic done flags
    buf.h0 = torch.randn(E, H, device="cuda") * 0.3
    s_ref = ref.update(buf)
    s_hip = hip.update(buf)
    for k in ("pi_loss", "v_loss", "entropy", "clipfrac"):
        assert abs(s_ref[k] - s_hip[k]) < 5e-3, (k, s_ref[k], s_hip[k])
    for (na, pa), (nb, pb) in zip(pol_a.named_parameters(),
                                  pol_b.named_parameters()):
        torch.testing.assert_close(pa, pb, atol=3e-4, rtol=2e-3), na


@pytest.mark.gpu
def test_hip_gru_learner_graph_trains():
    from helpers import make_buf
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_gru_hip import HipGRUPPOLearner
    from rlgpuschedule_amd.models.policy import GRUPolicy
    torch.manual_seed(3)
    spec = cluster64()
    T, E, H = 16, 128, 256
    pol = GRUPolicy(spec.obs_dim, spec.num_actions, H).cuda()
    hip = HipGRUPPOLearner(pol, PPOConfig(epochs=2, minibatches=2),
                           use_graph=True)
    buf = make_buf(T, E, spec, "cuda", seed=9)
    buf.h0 = torch.randn(E, H, device="cuda") * 0.3
    for _ in range(3):
        stats = hip.update(buf)
    for k, v in stats.items():
        assert np.isfinite(v), (k, v)
    for p in pol.parameters():
        assert torch.isfinite(p).all()


@pytest.mark.gpu
def test_gru_bf16_update_close_to_fp32():
    from helpers import make_buf
    from rlgpuschedule_amd.agents.ppo import PPOConfig
    from rlgpuschedule_amd.agents.ppo_gru_hip import HipGRUPPOLearner
    from rlgpuschedule_amd.models.policy import GRUPolicy
    torch.manual_seed(6)
    spec = cluster64()
    T, E, H = 16, 128, 256
    pol_a = GRUPolicy(spec.obs_dim, spec.num_actions, H).cuda()
    pol_b = copy.deepcopy(pol_a)
    p0 = {n: p.detach().clone() for n, p in pol_a.named_parameters()}
    cfg = PPOConfig(epochs=1, minibatches=1, norm_reward=False)
    l32 = HipGRUPPOLearner(pol_a, cfg, use_graph=False, dtype="fp32")
    l16 = HipGRUPPOLearner(pol_b, cfg, use_graph=False, dtype="bf16")
    buf = make_buf(T, E, spec, "cuda", seed=13)
    buf.h0 = torch.randn(E, H, device="cuda") * 0.3
    l32.update(buf)
    l16.update(buf)
    for (n, pa), (_, pb) in zip(pol_a.named_parameters(),
                                pol_b.named_parameters()):
        da = (pa - p0[n]).flatten()
        db = (pb - p0[n]).flatten()
        cos = torch.nn.functional.cosine_similarity(da, db, dim=0).item()
        assert cos > 0.98, (n, cos)
    # graph-mode stability
    pol_c = GRUPolicy(spec.obs_dim, spec.num_actions, H).cuda()
    lg = HipGRUPPOLearner(pol_c, PPOConfig(epochs=1, minibatches=2),
                          use_graph=True, dtype="bf16")
    for _ in range(3):
        stats = lg.update(buf)
    for k, v in stats.items():
        assert np.isfinite(v), (k, v)
