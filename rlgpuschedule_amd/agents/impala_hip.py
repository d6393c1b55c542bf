"""HipVtraceLearner — the IMPALA update entirely in hand-written HIP.

One gradient step per rollout (no minibatching, per IMPALA):
  fwd_train[_bf16]   MFMA forward over all T*E rows (activations saved)
  vtrace_head pass1  logp + value under CURRENT params (for the scan)
  vtrace_scan        truncated-IS targets vs / pg_adv (HIP reverse scan)
  vtrace_head pass2  fused loss forward+backward -> dhead
  bwd_dx/bwd_dw/colsum + fused clip+Adam (same machinery as the PPO path)
Bootstrap value = buf.last_value (behavior-policy value at collection),
matching agents/impala.py::VtraceLearner. Parity test:
tests/test_impala.py::test_hip_vtrace_learner_matches_reference.
"""
from __future__ import annotations

import torch

from ..models.hip_train import HEADC, TrainMirror, TrainMirrorBf16
from ..ops.nn import vtrace_scan
from ..ops.train import (HipAdam, bwd_dw, bwd_dw_bf16, bwd_dx, bwd_dx_bf16,
                         colsum, fwd_train, fwd_train_bf16, vtrace_head)
from .impala import ImpalaConfig
from .rollout import RolloutBuffer


class HipVtraceLearner:
    def __init__(self, policy, cfg: ImpalaConfig, grad_sync=None,
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
        self._graphs = {}   # per-rollout-buffer graphs (IMPALA double-buffers)
        self._s = None
        self.opt = None   # HipAdam, created in _alloc

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
            "lp_cur": z(T, E), "value": z(T, E),
            "vs": z(T, E), "pg_adv": z(T, E),
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
                           max_norm=self.cfg.max_grad_norm,
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
        T, E = buf.reward.shape
        B = self._B
        A, H = self._A, self._H
        obs = buf.obs.reshape(B, -1)
        mask = buf.mask.reshape(B, -1)
        acts = buf.actions.reshape(B)
        mi.refresh()
        if self.dtype == "bf16":
            fwd_train_bf16(obs, mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth, mi.bh,
                           s["h1"], s["h2"], s["head"])
        else:
            fwd_train(obs, mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth, mi.bh,
                      s["h1"], s["h2"], s["head"])
        lp_flat = s["lp_cur"].view(B)
        v_flat = s["value"].view(B)
        vtrace_head(s["head"], mask, acts, v_flat, v_flat, lp_flat, v_flat,
                    s["dhead"], s["stats"], cfg.vf_coef, cfg.ent_coef,
                    logp_only=True)
        vtrace_scan(buf.reward, s["value"], buf.done, buf.last_value,
                    s["lp_cur"], buf.logp, cfg.gamma, cfg.rho_bar,
                    cfg.c_bar, cfg.lam, vs=s["vs"], pg_adv=s["pg_adv"])
        s["stats"].zero_()
        vtrace_head(s["head"], mask, acts, s["pg_adv"].view(B),
                    s["vs"].view(B), lp_flat, v_flat, s["dhead"], s["stats"],
                    cfg.vf_coef, cfg.ent_coef, logp_only=False)
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
                warnings.warn(f"hipGraph capture of the IMPALA update failed "
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
