"""GAE(λ) reference implementation (torch, works on CPU and GPU).

The MI355X fast path is the HIP segmented reverse scan in
csrc/nn_kernels.hip (ops/nn.py: gae_scan); numerics tests compare it
against this fp32 reference.
"""
from __future__ import annotations

import torch


@torch.no_grad()
def gae_advantages(reward: torch.Tensor, value: torch.Tensor,
                   done: torch.Tensor, last_value: torch.Tensor,
                   gamma: float, lam: float):
    """reward/value/done: [T, E]; last_value: [E]. done[t]=1 means the
    transition at t ended an episode (obs[t+1] is a fresh reset).
    Returns (adv [T,E], returns [T,E])."""
    T, E = reward.shape
    adv = torch.zeros_like(reward)
    lastgaelam = torch.zeros(E, device=reward.device, dtype=reward.dtype)
    nd = 1.0 - done.to(reward.dtype)
    for t in range(T - 1, -1, -1):
        next_v = value[t + 1] if t < T - 1 else last_value
        delta = reward[t] + gamma * next_v * nd[t] - value[t]
        lastgaelam = delta + gamma * lam * nd[t] * lastgaelam
        adv[t] = lastgaelam
    return adv, adv + value
