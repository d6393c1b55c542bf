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
        assert H % 64 == 0 and A + 1 <= HEADC
        dev = policy.pi.weight.device
        self.D, self.H, self.A = D, H, A
        self.Dp = (D + 15) & ~15
        # packed layouts (see csrc/nn_kernels.hip kperm note):
        # wt[K,H] -> view(K/16, 4(s), 4(kk), H).permute(0,2,3,1) contiguous
        self.wt1 = torch.zeros(self.Dp // 16, 4, H, 4, device=dev)
        self.b1 = torch.empty(H, device=dev)
        self.wt2 = torch.empty(H // 16, 4, H, 4, device=dev)
        self.b2 = torch.empty(H, device=dev)
        self.wth = torch.zeros(H // 16, 4, HEADC, 4, device=dev)
        self.bh = torch.zeros(HEADC, device=dev)
        self.refresh()

    @staticmethod
    def _pack(wt: torch.Tensor) -> torch.Tensor:
        """[K,H] -> [K/16, 4(kk), H, 4(s)] packed for f32x4 B-fragment reads."""
        K, H = wt.shape
        return wt.view(K // 16, 4, 4, H).permute(0, 2, 3, 1)

    @torch.no_grad()
    def refresh(self):
        p = self.policy
        D, H, A = self.D, self.H, self.A
        if not hasattr(self, "_wt1_full"):
            self._wt1_full = self.wt1.new_zeros(self.Dp, H)
            self._wth_full = self.wth.new_zeros(H, HEADC)
        wt1_full = self._wt1_full
        wt1_full[:D].copy_(p.trunk[0].weight.t())
        self.wt1.copy_(self._pack(wt1_full))
        self.b1.copy_(p.trunk[0].bias)
        self.wt2.copy_(self._pack(p.trunk[1].weight.t()))
        self.b2.copy_(p.trunk[1].bias)
        wth_full = self._wth_full
        wth_full[:, :A].copy_(p.pi.weight.t())
        wth_full[:, A].copy_(p.v.weight[0])
        self.wth.copy_(self._pack(wth_full))
        self.bh[:A].copy_(p.pi.bias)
        self.bh[A] = p.v.bias[0]

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


class HipActorGRU:
    """Device-side mirror of GRUPolicy for the fused GRU actor kernel.

    Packs: input layer [Dp,H], GRU gate weights weight_ih/_hh transposed to
    [H, 3H] (torch GRUCell gate order r,z,n), combined head [H,32]."""

    def __init__(self, policy):
        assert policy.is_recurrent
        D, H, A = policy.obs_dim, policy.hidden, policy.num_actions
        assert H % 64 == 0 and A + 1 <= HEADC
        dev = policy.pi.weight.device
        self.policy = policy
        self.D, self.H, self.A = D, H, A
        self.Dp = (D + 15) & ~15
        self.wtin = torch.zeros(self.Dp // 16, 4, H, 4, device=dev)
        self.bin = torch.empty(H, device=dev)
        self.wih = torch.empty(H // 16, 4, 3 * H, 4, device=dev)
        self.bih = torch.empty(3 * H, device=dev)
        self.whh = torch.empty(H // 16, 4, 3 * H, 4, device=dev)
        self.bhh = torch.empty(3 * H, device=dev)
        self.wth = torch.zeros(H // 16, 4, HEADC, 4, device=dev)
        self.bh = torch.zeros(HEADC, device=dev)
        self.refresh()

    _pack = staticmethod(HipActorMLP._pack)

    @torch.no_grad()
    def refresh(self):
        p = self.policy
        D, H, A = self.D, self.H, self.A
        if not hasattr(self, "_win_full"):
            self._win_full = self.wtin.new_zeros(self.Dp, H)
            self._wth_full = self.wth.new_zeros(H, HEADC)
        self._win_full[:D].copy_(p.inp.weight.t())
        self.wtin.copy_(self._pack(self._win_full))
        self.bin.copy_(p.inp.bias)
        self.wih.copy_(self._pack(p.cell.weight_ih.t().contiguous()))
        self.bih.copy_(p.cell.bias_ih)
        self.whh.copy_(self._pack(p.cell.weight_hh.t().contiguous()))
        self.bhh.copy_(p.cell.bias_hh)
        self._wth_full[:, :A].copy_(p.pi.weight.t())
        self._wth_full[:, A].copy_(p.v.weight[0])
        self.wth.copy_(self._pack(self._wth_full))
        self.bh[:A].copy_(p.pi.bias)
        self.bh[A] = p.v.bias[0]

    def act(self, obs, mask, u, done_prev, hx, hx_out, actions, logp, value):
        from ..ops.nn import gru_actor
        gru_actor(obs, mask, u, done_prev, hx, hx_out, self.wtin, self.bin,
                  self.wih, self.bih, self.whh, self.bhh, self.wth, self.bh,
                  actions, logp, value)

    def values(self, obs, mask, done_prev, hx, hx_scratch, value_out):
        from ..ops.nn import gru_actor
        gru_actor(obs, mask, value_out, done_prev, hx, hx_scratch, self.wtin,
                  self.bin, self.wih, self.bih, self.whh, self.bhh, self.wth,
                  self.bh, value_out, value_out, value_out, value_only=True)


class HipActorMLPBf16:
    """bf16-MFMA variant of HipActorMLP (~13x the f32 MFMA rate; the actor
    is MFMA-issue-bound). Weights bf16, accumulation/head/sampling f32.
    Packing: wt[K,N].view(K/32, 4(kk), 8(j), N).permute(0,1,3,2) — the
    contiguous per-lane k-map shared by A and B fragments."""

    def __init__(self, policy):
        assert not policy.is_recurrent and len(policy.trunk) == 2
        self.policy = policy
        D, H, A = policy.obs_dim, policy.hidden, policy.num_actions
        assert H % 64 == 0 and A + 1 <= HEADC
        dev = policy.pi.weight.device
        self.D, self.H, self.A = D, H, A
        self.Dp = (D + 31) & ~31
        self.wt1 = torch.zeros(self.Dp // 32, 4, H, 8, device=dev,
                               dtype=torch.bfloat16)
        self.b1 = torch.empty(H, device=dev)
        self.wt2 = torch.empty(H // 32, 4, H, 8, device=dev,
                               dtype=torch.bfloat16)
        self.b2 = torch.empty(H, device=dev)
        self.wth = torch.zeros(H // 32, 4, HEADC, 8, device=dev,
                               dtype=torch.bfloat16)
        self.bh = torch.zeros(HEADC, device=dev)
        self.refresh()

    @staticmethod
    def _pack8(wt: torch.Tensor) -> torch.Tensor:
        K, N = wt.shape
        return wt.view(K // 32, 4, 8, N).permute(0, 1, 3, 2)

    @torch.no_grad()
    def refresh(self):
        p = self.policy
        D, H, A = self.D, self.H, self.A
        if not hasattr(self, "_wt1_full"):
            self._wt1_full = self.b1.new_zeros(self.Dp, H)
            self._wth_full = self.b1.new_zeros(H, HEADC)
        self._wt1_full[:D].copy_(p.trunk[0].weight.t())
        self.wt1.copy_(self._pack8(self._wt1_full))
        self.b1.copy_(p.trunk[0].bias)
        self.wt2.copy_(self._pack8(p.trunk[1].weight.t()))
        self.b2.copy_(p.trunk[1].bias)
        self._wth_full[:, :A].copy_(p.pi.weight.t())
        self._wth_full[:, A].copy_(p.v.weight[0])
        self.wth.copy_(self._pack8(self._wth_full))
        self.bh[:A].copy_(p.pi.bias)
        self.bh[A] = p.v.bias[0]

    def act(self, obs, mask, u, actions, logp, value):
        from ..ops.nn import actor_bf16
        actor_bf16(obs, mask, u, self.wt1, self.b1, self.wt2, self.b2,
                   self.wth, self.bh, actions, logp, value)
        return actions, logp, value

    def values(self, obs, value_out, mask):
        from ..ops.nn import actor_bf16
        actor_bf16(obs, mask, value_out, self.wt1, self.b1, self.wt2,
                   self.b2, self.wth, self.bh, value_out, value_out,
                   value_out, value_only=True)
        return value_out


class HipActorGRUBf16:
    """bf16-MFMA variant of HipActorGRU (f32 accumulate/gates/sampling)."""

    def __init__(self, policy):
        assert policy.is_recurrent
        D, H, A = policy.obs_dim, policy.hidden, policy.num_actions
        assert H % 64 == 0 and A + 1 <= HEADC
        dev = policy.pi.weight.device
        self.policy = policy
        self.D, self.H, self.A = D, H, A
        self.Dp = (D + 31) & ~31
        bf = torch.bfloat16
        self.wtin = torch.zeros(self.Dp // 32, 4, H, 8, device=dev, dtype=bf)
        self.bin = torch.empty(H, device=dev)
        self.wih = torch.empty(H // 32, 4, 3 * H, 8, device=dev, dtype=bf)
        self.bih = torch.empty(3 * H, device=dev)
        self.whh = torch.empty(H // 32, 4, 3 * H, 8, device=dev, dtype=bf)
        self.bhh = torch.empty(3 * H, device=dev)
        self.wth = torch.zeros(H // 32, 4, HEADC, 8, device=dev, dtype=bf)
        self.bh = torch.zeros(HEADC, device=dev)
        self.refresh()

    _pack8 = staticmethod(HipActorMLPBf16._pack8)

    @torch.no_grad()
    def refresh(self):
        p = self.policy
        D, H, A = self.D, self.H, self.A
        if not hasattr(self, "_win_full"):
            self._win_full = self.bin.new_zeros(self.Dp, H)
            self._wth_full = self.bin.new_zeros(H, HEADC)
        self._win_full[:D].copy_(p.inp.weight.t())
        self.wtin.copy_(self._pack8(self._win_full))
        self.bin.copy_(p.inp.bias)
        self.wih.copy_(self._pack8(p.cell.weight_ih.t().contiguous()))
        self.bih.copy_(p.cell.bias_ih)
        self.whh.copy_(self._pack8(p.cell.weight_hh.t().contiguous()))
        self.bhh.copy_(p.cell.bias_hh)
        self._wth_full[:, :A].copy_(p.pi.weight.t())
        self._wth_full[:, A].copy_(p.v.weight[0])
        self.wth.copy_(self._pack8(self._wth_full))
        self.bh[:A].copy_(p.pi.bias)
        self.bh[A] = p.v.bias[0]

    def act(self, obs, mask, u, done_prev, hx, hx_out, actions, logp, value):
        from ..ops.nn import gru_bf16
        gru_bf16(obs, mask, u, done_prev, hx, hx_out, self.wtin, self.bin,
                 self.wih, self.bih, self.whh, self.bhh, self.wth, self.bh,
                 actions, logp, value)

    def values(self, obs, mask, done_prev, hx, hx_scratch, value_out):
        from ..ops.nn import gru_bf16
        gru_bf16(obs, mask, value_out, done_prev, hx, hx_scratch, self.wtin,
                 self.bin, self.wih, self.bih, self.whh, self.bhh, self.wth,
                 self.bh, value_out, value_out, value_out, value_only=True)
