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
