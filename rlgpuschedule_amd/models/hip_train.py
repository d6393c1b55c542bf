"""TrainMirror — packed weight mirrors for the hand-written update path.

Extends the actor mirror (forward weights) with the packed TRANSPOSED
weights the backward dX GEMMs need:
  dh2 = dhead @ Wh^T  -> whT packed, K=HEADC
  dh1 = dh2 @ W2^T    -> w2T packed, K=H  (W2^T = trunk[1].weight as-is,
                          since the forward weight is weight.t())
refresh() must run after EVERY optimizer step (torch copies, capturable).
"""
from __future__ import annotations

import torch

from .hip_actor import HEADC, HipActorMLP


class TrainMirror(HipActorMLP):
    def __init__(self, policy):
        H = policy.hidden
        dev = policy.pi.weight.device
        self.w2T = torch.empty(H // 16, 4, H, 4, device=dev)
        self.whT = torch.empty(HEADC // 16, 4, H, 4, device=dev)
        super().__init__(policy)

    @torch.no_grad()
    def refresh(self):
        super().refresh()
        p = self.policy
        self.w2T.copy_(self._pack(p.trunk[1].weight))
        # Wh^T [HEADC, H]: rows [0,A) = pi.weight, row A = v.weight
        if not hasattr(self, "_whT_full"):
            self._whT_full = self.whT.new_zeros(HEADC, self.H)
        self._whT_full[: self.A].copy_(p.pi.weight)
        self._whT_full[self.A].copy_(p.v.weight[0])
        self.whT.copy_(self._pack(self._whT_full))


class TrainMirrorBf16:
    """bf16 packed8 mirrors for the mixed-precision update path: forward
    weights (wt1/wt2/wth) + transposed backward weights (w2T/whT)."""

    def __init__(self, policy):
        from .hip_actor import HipActorMLPBf16
        self._fwd = HipActorMLPBf16(policy)   # wt1/b1/wt2/b2/wth/bh packed8
        H, A = policy.hidden, policy.num_actions
        dev = policy.pi.weight.device
        bf = torch.bfloat16
        self.w2T = torch.empty(H // 32, 4, H, 8, device=dev, dtype=bf)
        self.whT = torch.empty(HEADC // 32, 4, H, 8, device=dev, dtype=bf)
        self._whT_full = torch.zeros(HEADC, H, device=dev)
        self.policy = policy
        self.refresh()

    def __getattr__(self, name):
        if name in ("wt1", "b1", "wt2", "b2", "wth", "bh"):
            return getattr(self._fwd, name)
        raise AttributeError(name)

    @torch.no_grad()
    def refresh(self):
        self._fwd.refresh()
        p = self.policy
        self.w2T.copy_(self._fwd._pack8(p.trunk[1].weight))
        self._whT_full[: p.num_actions].copy_(p.pi.weight)
        self._whT_full[p.num_actions].copy_(p.v.weight[0])
        self.whT.copy_(self._fwd._pack8(self._whT_full))
