"""Policy/value networks: MLP and GRU heads (torch reference path).

These torch modules define the parameter layout and the numerics reference.
The MI355X fast path (hand-written MFMA GEMM forward/backward + fused
sampling, rlgpuschedule_amd/csrc/nn_kernels.hip) consumes the SAME
parameter tensors, so checkpoints are interchangeable between paths.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

NEG_INF = -1e9


def masked_logits(logits: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
    return logits + (mask.to(logits.dtype) - 1.0) * (-NEG_INF)


def _ortho_init(layer: nn.Linear, gain: float) -> nn.Linear:
    nn.init.orthogonal_(layer.weight, gain)
    nn.init.zeros_(layer.bias)
    return layer


class MLPPolicy(nn.Module):
    """obs -> tanh MLP trunk -> (action logits, value)."""

    is_recurrent = False

    def __init__(self, obs_dim: int, num_actions: int, hidden: int = 256,
                 layers: int = 2):
        super().__init__()
        self.obs_dim, self.num_actions, self.hidden = obs_dim, num_actions, hidden
        dims = [obs_dim] + [hidden] * layers
        self.trunk = nn.ModuleList(
            _ortho_init(nn.Linear(dims[i], dims[i + 1]), math.sqrt(2))
            for i in range(layers))
        self.pi = _ortho_init(nn.Linear(hidden, num_actions), 0.01)
        self.v = _ortho_init(nn.Linear(hidden, 1), 1.0)

    def forward(self, obs: torch.Tensor, mask: torch.Tensor | None = None,
                hx: torch.Tensor | None = None):
        x = obs
        for lin in self.trunk:
            x = torch.tanh(lin(x))
        logits = self.pi(x)
        if mask is not None:
            logits = masked_logits(logits, mask)
        return logits, self.v(x).squeeze(-1), None

    def initial_state(self, batch: int, device) -> None:
        return None


class GRUPolicy(nn.Module):
    """obs -> linear+tanh -> GRU cell -> (logits, value). Recurrent over the
    partially-observed queue state (BASELINE.json config 3)."""

    is_recurrent = True

    def __init__(self, obs_dim: int, num_actions: int, hidden: int = 256):
        super().__init__()
        self.obs_dim, self.num_actions, self.hidden = obs_dim, num_actions, hidden
        self.inp = _ortho_init(nn.Linear(obs_dim, hidden), math.sqrt(2))
        self.cell = nn.GRUCell(hidden, hidden)
        self.pi = _ortho_init(nn.Linear(hidden, num_actions), 0.01)
        self.v = _ortho_init(nn.Linear(hidden, 1), 1.0)

    def forward(self, obs: torch.Tensor, mask: torch.Tensor | None = None,
                hx: torch.Tensor | None = None):
        x = torch.tanh(self.inp(obs))
        hx = self.cell(x, hx)
        logits = self.pi(hx)
        if mask is not None:
            logits = masked_logits(logits, mask)
        return logits, self.v(hx).squeeze(-1), hx

    def initial_state(self, batch: int, device) -> torch.Tensor:
        return torch.zeros(batch, self.hidden, device=device)


def sample_categorical(logits: torch.Tensor, u: torch.Tensor | None = None):
    """Sample actions + logprob + entropy from (masked) logits.

    With `u` (uniform [0,1) noise [B]) uses the inverse-CDF on device —
    deterministic given u, mirrored by the HIP fused actor kernel via the
    Gumbel trick equivalence is NOT used; keep CDF order (ascending index).
    """
    logp = F.log_softmax(logits, dim=-1)
    p = logp.exp()
    if u is None:
        actions = torch.multinomial(p, 1).squeeze(-1)
    else:
        cdf = p.cumsum(-1)
        actions = (cdf < u.unsqueeze(-1)).sum(-1).clamp(max=logits.shape[-1] - 1)
    lp = logp.gather(-1, actions.unsqueeze(-1)).squeeze(-1)
    entropy = -(p * logp).sum(-1)
    return actions, lp, entropy


def evaluate_actions(logits: torch.Tensor, actions: torch.Tensor):
    logp = F.log_softmax(logits, dim=-1)
    lp = logp.gather(-1, actions.unsqueeze(-1)).squeeze(-1)
    entropy = -(logp.exp() * logp).sum(-1)
    return lp, entropy
