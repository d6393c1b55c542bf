"""Running reward normalization (CleanRL-style NormalizeReward).

Rewards are divided by the running std of DISCOUNTED returns so value
targets stay O(1) — without this, cluster-scheduling returns reach 1e4
and the clipped value loss (clip 0.2) can never track them. The scale is
a device tensor updated with an EMA of the batch variance; all ops are
torch tensor ops (hipGraph-capturable). Episode stats in the env are
computed from RAW rewards and are unaffected.
"""
from __future__ import annotations

import torch


class RewardNormalizer:
    def __init__(self, gamma: float, device, ema: float = 0.99,
                 eps: float = 1e-8):
        self.gamma = gamma
        self.ema = ema
        self.eps = eps
        self.var = torch.ones((), device=device)
        self.initialized = torch.zeros((), device=device)

    @torch.no_grad()
    def __call__(self, reward: torch.Tensor, done: torch.Tensor) -> None:
        """Scale reward [T,E] in place (done [T,E] cuts the return)."""
        if reward.is_cuda:
            from ..ops.nn import gae_scan
            zeros = torch.zeros_like(reward[0])
            ret, _ = gae_scan(reward, torch.zeros_like(reward),
                              done, zeros, self.gamma, 1.0)
        else:
            from ..ops.gae_ref import gae_advantages
            ret, _ = gae_advantages(reward, torch.zeros_like(reward), done,
                                    torch.zeros_like(reward[0]),
                                    self.gamma, 1.0)
        bvar = ret.pow(2).mean()  # zero-mean assumption (returns <= 0 here
        # have large |mean|; second moment is the right scale regardless)
        # first batch: adopt bvar, then EMA
        new = torch.where(self.initialized > 0,
                          self.ema * self.var + (1 - self.ema) * bvar, bvar)
        self.var.copy_(new)
        self.initialized.fill_(1.0)
        reward.mul_(1.0 / (self.var.sqrt() + self.eps))
