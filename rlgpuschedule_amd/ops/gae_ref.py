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


@torch.no_grad()
def vtrace_reference(reward, value, done, last_value, lp_cur, lp_beh,
                     gamma, rho_bar=1.0, c_bar=1.0, lam=1.0):
    """Torch fp32 reference of the HIP vtrace_kernel (same recurrence)."""
    T, E = reward.shape
    vs = torch.zeros_like(reward)
    pg_adv = torch.zeros_like(reward)
    acc = torch.zeros(E, dtype=reward.dtype, device=reward.device)
    next_v = last_value.clone()
    vs_next = last_value.clone()
    nd = 1.0 - done.to(reward.dtype)
    ratio = (lp_cur - lp_beh).exp()
    rho = ratio.clamp(max=rho_bar)
    c = lam * ratio.clamp(max=c_bar)
    for t in range(T - 1, -1, -1):
        delta = rho[t] * (reward[t] + gamma * next_v * nd[t] - value[t])
        acc = delta + gamma * c[t] * nd[t] * acc
        vs[t] = value[t] + acc
        pg_adv[t] = rho[t] * (reward[t] + gamma * vs_next * nd[t] - value[t])
        vs_next = vs[t]
        next_v = value[t]
    return vs, pg_adv
