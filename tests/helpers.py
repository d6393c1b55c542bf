"""Shared test fixtures/helpers."""
import torch


def make_buf(T, E, spec, device, seed=0):
    from rlgpuschedule_amd.agents.rollout import RolloutBuffer
    g = torch.Generator(device="cpu").manual_seed(seed)
    buf = RolloutBuffer(T, E, spec.obs_dim, spec.num_actions, device)
    buf.obs.copy_(torch.randn(T, E, spec.obs_dim, generator=g))
    mask = (torch.rand(T, E, spec.num_actions, generator=g) < 0.5)
    mask[..., spec.noop_action] = True
    buf.mask.copy_(mask.to(torch.uint8))
    a = torch.full((T, E), spec.noop_action, dtype=torch.int64)
    buf.actions.copy_(a)
    buf.logp.copy_(torch.randn(T, E, generator=g) * 0.1 - 2.0)
    buf.value.copy_(torch.randn(T, E, generator=g))
    buf.reward.copy_(-torch.rand(T, E, generator=g))
    buf.done.copy_((torch.rand(T, E, generator=g) < 0.02).to(torch.uint8))
    buf.last_value.copy_(torch.randn(E, generator=g))
    return buf
