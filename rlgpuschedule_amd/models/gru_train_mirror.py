"""GruTrainMirror — packed weights for the GRU BPTT kernels.

Extends the GRU actor mirror with the backward-direction packs:
  dx   = dgi @ weight_ih          -> wih_bwd = pack(weight_ih [3H,H])
  dh'  = dgh @ weight_hh          -> whh_bwd = pack(weight_hh [3H,H])
  dh_ext = dhead @ Wh^T           -> whT     = pack(wth_full.t() [32,H])
refresh() after every optimizer step.
"""
from __future__ import annotations

import torch

from .hip_actor import HEADC, HipActorGRU


class GruTrainMirror(HipActorGRU):
    def __init__(self, policy):
        H = policy.hidden
        dev = policy.pi.weight.device
        self.wih_bwd = torch.empty(3 * H // 16, 4, H, 4, device=dev)
        self.whh_bwd = torch.empty(3 * H // 16, 4, H, 4, device=dev)
        self.whT = torch.empty(HEADC // 16, 4, H, 4, device=dev)
        super().__init__(policy)

    @torch.no_grad()
    def refresh(self):
        super().refresh()
        p = self.policy
        self.wih_bwd.copy_(self._pack(p.cell.weight_ih.contiguous()))
        self.whh_bwd.copy_(self._pack(p.cell.weight_hh.contiguous()))
        if not hasattr(self, "_whT_full"):
            self._whT_full = self.whT.new_zeros(HEADC, self.H)
        self._whT_full[: self.A].copy_(p.pi.weight)
        self._whT_full[self.A].copy_(p.v.weight[0])
        self.whT.copy_(self._pack(self._whT_full))


class GruTrainMirrorBf16:
    """bf16 packed8 mirrors for the mixed-precision GRU BPTT path."""

    def __init__(self, policy):
        from .hip_actor import HipActorGRUBf16
        self._fwd = HipActorGRUBf16(policy)
        H = policy.hidden
        dev = policy.pi.weight.device
        bf = torch.bfloat16
        self.wih_bwd = torch.empty(3 * H // 32, 4, H, 8, device=dev, dtype=bf)
        self.whh_bwd = torch.empty(3 * H // 32, 4, H, 8, device=dev, dtype=bf)
        self.whT = torch.empty(32 // 32, 4, H, 8, device=dev, dtype=bf)
        self._whT_full = torch.zeros(HEADC, H, device=dev)
        self.policy = policy
        self.refresh()

    def __getattr__(self, name):
        if name in ("wtin", "bin", "wih", "bih", "whh", "bhh", "wth", "bh"):
            return getattr(self._fwd, name)
        raise AttributeError(name)

    @torch.no_grad()
    def refresh(self):
        self._fwd.refresh()
        p = self.policy
        self.wih_bwd.copy_(self._fwd._pack8(p.cell.weight_ih.contiguous()))
        self.whh_bwd.copy_(self._fwd._pack8(p.cell.weight_hh.contiguous()))
        self._whT_full[: p.num_actions].copy_(p.pi.weight)
        self._whT_full[p.num_actions].copy_(p.v.weight[0])
        self.whT.copy_(self._fwd._pack8(self._whT_full))
