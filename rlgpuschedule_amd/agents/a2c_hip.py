"""HipA2CLearner — the A2C update entirely in hand-written HIP.

A2C's loss is IMPALA's head objective with GAE targets instead of the
V-trace scan: L = -adv*logp + 0.5*vf*(v - ret)^2 - ent*H, one gradient
step per rollout. So this learner reuses the whole IMPALA/PPO machinery:

  gae_scan           HIP segmented reverse scan -> adv, ret (lam=1 by
                     default: plain discounted returns, per A2CConfig)
  fwd_train[_bf16]   MFMA forward, activations saved
  vtrace_head pass2  fused loss forward+backward -> dhead (pg_adv=adv,
                     vs=ret; pass 1 is unnecessary — no IS correction)
  bwd_dx/bwd_dw/colsum + fused clip step in RMSprop mode (HipAdam
  algo="rmsprop" matches agents/ppo.py::A2CLearner's torch.optim.RMSprop)

Parity: tests/test_train_kernels.py::test_hip_a2c_learner_matches_reference.
"""
from __future__ import annotations

import torch

from ..models.hip_train import HEADC, TrainMirror, TrainMirrorBf16
from ..ops.nn import gae_scan
from ..ops.train import (HipAdam, bwd_dw, bwd_dw_bf16, bwd_dx, bwd_dx_bf16,
                         colsum, fwd_train, fwd_train_bf16, vtrace_head)
from .ppo import A2CConfig
from .rollout import RolloutBuffer


class HipA2CLearner:
    def __init__(self, policy, cfg: A2CConfig, grad_sync=None,
                 use_graph: bool = True, dtype: str = "fp32"):
        assert not policy.is_recurrent
        assert dtype in ("fp32", "bf16")
        self.policy = policy
        self.cfg = cfg
        self.dtype = dtype
        self.grad_sync = grad_sync
        self.use_graph = use_graph
        self.mirror = (TrainMirrorBf16(policy) if dtype == "bf16"
                       else TrainMirror(policy))
        self._graphs = {}
        self._s = None
        self.opt = None

    def _alloc(self, buf: RolloutBuffer):
        T, E = buf.reward.shape
        B = T * E
        D = buf.obs.shape[-1]
        A = buf.mask.shape[-1]
        H = self.policy.hidden
        dev = buf.reward.device
        self._B, self._A, self._D, self._H = B, A, D, H
        adt = torch.bfloat16 if self.dtype == "bf16" else torch.float32
        z = lambda *s: torch.zeros(*s, device=dev)
        self._s = s = {
            "h1": torch.zeros(B, H, device=dev, dtype=adt),
            "h2": torch.zeros(B, H, device=dev, dtype=adt),
            "head": z(B, HEADC), "dhead": z(B, HEADC),
            "dh2": z(B, H), "dh1": z(B, H),
            "adv": z(T, E), "ret": z(T, E),
            "lp_scratch": z(B), "v_scratch": z(B),
            "stats": z(4),
        }
        gflat = torch.zeros(D * H + H * H + H * HEADC + 2 * H + HEADC,
                            device=dev)
        off = 0
        offs = {}
        for name, shape in (("dW1", (D, H)), ("dW2", (H, H)),
                            ("dWh", (H, HEADC)), ("db1", (H,)),
                            ("db2", (H,)), ("dbh", (HEADC,))):
            offs[name] = off
            n = 1
            for sd in shape:
                n *= sd
            s[name] = gflat[off:off + n].view(*shape)
            off += n
        s["gflat"] = gflat
        pol = self.policy
        from ..parallel.dist import allreduce_flat
        W = lambda name, R, C, c0, c1, p: dict(
            goff=offs[name], R=R, C=C, c0=c0, c1=c1, param=p, transposed=True)
        Bs = lambda name, C, c0, c1, p: dict(
            goff=offs[name], R=1, C=C, c0=c0, c1=c1, param=p, transposed=False)
        segs = [
            W("dW1", D, H, 0, H, pol.trunk[0].weight),
            W("dW2", H, H, 0, H, pol.trunk[1].weight),
            W("dWh", H, HEADC, 0, A, pol.pi.weight),
            W("dWh", H, HEADC, A, A + 1, pol.v.weight),
            Bs("db1", H, 0, H, pol.trunk[0].bias),
            Bs("db2", H, 0, H, pol.trunk[1].bias),
            Bs("dbh", HEADC, 0, A, pol.pi.bias),
            Bs("dbh", HEADC, A, A + 1, pol.v.bias),
        ]
        self.opt = HipAdam(gflat, segs, lr=self.cfg.lr, eps=1e-5,
                           betas=(0.0, 0.99),   # beta2 = RMSprop alpha
                           max_norm=self.cfg.max_grad_norm, algo="rmsprop",
                           grad_reduce=allreduce_flat
                           if self.grad_sync is not None else None)
        from ..ckpt.io import resolve_pending_opt_state
        resolve_pending_opt_state(self)

    def load_optim_state(self, sd):
        """Stash checkpointed optimizer state; applied when _alloc creates
        the real optimizer (HipAdam is lazy — see ckpt.io)."""
        self._pending_opt_state = sd

    def _update_body(self, buf: RolloutBuffer):
        cfg = self.cfg
        s = self._s
        mi = self.mirror
        if cfg.norm_reward:
            if not hasattr(self, "_rnorm"):
                from .rewardnorm import RewardNormalizer
                self._rnorm = RewardNormalizer(cfg.gamma, buf.reward.device)
            self._rnorm(buf.reward, buf.done)
        B = self._B
        obs = buf.obs.reshape(B, -1)
        mask = buf.mask.reshape(B, -1)
        acts = buf.actions.reshape(B)
        gae_scan(buf.reward, buf.value, buf.done, buf.last_value,
                 cfg.gamma, cfg.lam, adv=s["adv"], ret=s["ret"])
        mi.refresh()
        if self.dtype == "bf16":
            fwd_train_bf16(obs, mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth, mi.bh,
                           s["h1"], s["h2"], s["head"])
        else:
            fwd_train(obs, mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth, mi.bh,
                      s["h1"], s["h2"], s["head"])
        s["stats"].zero_()
        vtrace_head(s["head"], mask, acts, s["adv"].view(B),
                    s["ret"].view(B), s["lp_scratch"], s["v_scratch"],
                    s["dhead"], s["stats"], cfg.vf_coef, cfg.ent_coef,
                    logp_only=False)
        if self.dtype == "bf16":
            bwd_dx_bf16(s["dhead"], mi.whT, s["h2"], s["dh2"])
            bwd_dx_bf16(s["dh2"], mi.w2T, s["h1"], s["dh1"])
            s["gflat"].zero_()
            bwd_dw_bf16(s["h2"], s["dhead"], s["dWh"])
            bwd_dw_bf16(s["h1"], s["dh2"], s["dW2"])
            bwd_dw_bf16(obs, s["dh1"], s["dW1"])
        else:
            bwd_dx(s["dhead"], mi.whT, s["h2"], s["dh2"])
            bwd_dx(s["dh2"], mi.w2T, s["h1"], s["dh1"])
            s["gflat"].zero_()
            bwd_dw(s["h2"], s["dhead"], s["dWh"])
            bwd_dw(s["h1"], s["dh2"], s["dW2"])
            bwd_dw(obs, s["dh1"], s["dW1"])
        colsum(s["dhead"], s["dbh"])
        colsum(s["dh2"], s["db2"])
        colsum(s["dh1"], s["db1"])
        self.opt.step()
        mi.refresh()

    def update(self, buf: RolloutBuffer) -> dict:
        if self._s is None:
            self._alloc(buf)
        key = id(buf)
        if self.use_graph and key not in self._graphs:
            try:
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    self._update_body(buf)
                torch.cuda.current_stream().wait_stream(side)
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._update_body(buf)
                self._graphs[key] = g
            except Exception as e:
                import warnings
                warnings.warn(f"hipGraph capture of the A2C update failed "
                              f"({e!r}); falling back to eager execution")
                self.use_graph = False
            from ..parallel.dist import agree_all
            if not agree_all(self.use_graph):   # rank-consistent fallback
                self.use_graph = False
        if self.use_graph:
            self._graphs[key].replay()
        else:
            self._update_body(buf)
        st = self._s["stats"].cpu()
        n = self._B
        return {
            "pi_loss": st[0].item() / n,
            "v_loss": st[1].item() / n,
            "entropy": st[2].item() / n,
            "mean_reward": buf.reward.mean().item(),
        }
