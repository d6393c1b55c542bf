"""HipActorMLP — device-side mirror of MLPPolicy for the fused actor kernel.

Keeps transposed fp32 weight copies ([in,out] layout for coalesced
B-fragment reads) in sync with the torch parameters; refresh() is a few
device copies (capturable, run once per PPO iteration after the update).
"""
from __future__ import annotations

import torch

from ..ops import nn as hipnn

HEADC = 32


class HipActorMLP:
    def __init__(self, policy):
        assert not policy.is_recurrent and len(policy.trunk) == 2, \
            "fused actor kernel supports the 2-layer MLP flagship"
        self.policy = policy
        D, H, A = policy.obs_dim, policy.hidden, policy.num_actions
        assert D % 4 == 0 and H % 64 == 0 and A + 1 <= HEADC
        dev = policy.pi.weight.device
        self.D, self.H, self.A = D, H, A
        self.wt1 = torch.empty(D, H, device=dev)
        self.b1 = torch.empty(H, device=dev)
        self.wt2 = torch.empty(H, H, device=dev)
        self.b2 = torch.empty(H, device=dev)
        self.wth = torch.zeros(H, HEADC, device=dev)
        self.bh = torch.zeros(HEADC, device=dev)
        self.refresh()

    @torch.no_grad()
    def refresh(self):
        p = self.policy
        self.wt1.copy_(p.trunk[0].weight.t())
        self.b1.copy_(p.trunk[0].bias)
        self.wt2.copy_(p.trunk[1].weight.t())
        self.b2.copy_(p.trunk[1].bias)
        self.wth[:, : self.A].copy_(p.pi.weight.t())
        self.wth[:, self.A].copy_(p.v.weight[0])
        self.bh[: self.A].copy_(p.pi.bias)
        self.bh[self.A] = p.v.bias[0]

    def act(self, obs, mask, u, actions, logp, value):
        """Sample actions in-place into the given out tensors (one launch)."""
        hipnn.actor_mlp(obs, mask, u, self.wt1, self.b1, self.wt2, self.b2,
                        self.wth, self.bh, actions, logp, value)
        return actions, logp, value

    def values(self, obs, value_out, mask):
        """Bootstrap-value pass: value_only kernel writes ONLY value, so the
        action/logp/u slots are dead pointers (value_out reused). `mask` is
        only used for its shape-A metadata in the binding."""
        hipnn.actor_mlp(obs, mask, value_out, self.wt1, self.b1, self.wt2,
                        self.b2, self.wth, self.bh, value_out, value_out,
                        value_out, value_only=True)
        return value_out
