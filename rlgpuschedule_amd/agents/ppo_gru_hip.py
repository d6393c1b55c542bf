"""HipGRUPPOLearner — recurrent PPO update entirely in hand-written HIP.

Replaces the torch GRUCell BPTT replay (hundreds of launches, ~1.2 s/iter
at config-3 scale) with:
  gru_seq_fwd   ONE launch: whole T-step sequence forward (blocks own
                their 16 envs end-to-end; activations saved for BPTT)
  head_fwd      batched logits/value head over T*Emb rows
  ppo_head      fused masked loss forward+backward -> dhead
  bwd_dx        dh_ext = dhead @ Wh^T (plain GEMM)
  gru_seq_bwd   ONE launch: reverse-time sweep (gate grads + carry GEMM)
  bwd_dx        dx = (dgi @ Wih) * tanh'(x)
  bwd_dw x4 / colsum x4   weight/bias grads contracted over T*Emb
then grad assembly, global-norm clip and capturable Adam (torch), all
inside one hipGraph per iteration. Minibatches are over ENVS (sequences
stay contiguous), matching agents/ppo.py::_update_recurrent.

Gradient checks: tests/test_gru_train.py vs torch autograd.
"""
from __future__ import annotations

import torch

from ..models.gru_train_mirror import (HEADC, GruTrainMirror,
                                       GruTrainMirrorBf16)
from ..ops.gru import (gru_seq_fwd, gru_seq_bwd, gru_seq_fwd_bf16,
                       gru_seq_bwd_bf16, head_fwd, head_fwd_bf16)
from ..ops.nn import gae_scan
from ..ops.train import (bwd_dw, bwd_dw_bf16, bwd_dx, bwd_dx_bf16, colsum,
                         ppo_head)
from .ppo import PPOConfig
from .rollout import RolloutBuffer


class HipGRUPPOLearner:
    def __init__(self, policy, cfg: PPOConfig, grad_sync=None,
                 use_graph: bool = True, use_hip_adam: bool = True,
                 dtype: str = "fp32"):
        assert policy.is_recurrent
        assert dtype in ("fp32", "bf16")
        self.policy = policy
        self.cfg = cfg
        self.dtype = dtype
        self.grad_sync = grad_sync
        self.use_hip_adam = use_hip_adam
        self.hip_opt = None
        self.use_graph = use_graph
        self.opt = torch.optim.Adam(policy.parameters(), lr=cfg.lr, eps=1e-5,
                                    capturable=self.use_graph)
        self.mirror = (GruTrainMirrorBf16(policy) if dtype == "bf16"
                       else GruTrainMirror(policy))
        self.gen = torch.Generator(device="cuda")
        self.gen.manual_seed(24680)
        self._graph = None
        self._s = None
        for p in policy.parameters():
            if p.grad is None:
                p.grad = torch.zeros_like(p)

    def _alloc(self, buf: RolloutBuffer):
        cfg = self.cfg
        T, E = buf.reward.shape
        D = buf.obs.shape[-1]
        A = buf.mask.shape[-1]
        H = self.policy.hidden
        Emb = max(E // cfg.minibatches, 16)
        B = T * Emb
        dev = buf.reward.device
        self._T, self._E, self._Emb, self._B = T, E, Emb, B
        self._A, self._D, self._H = A, D, H
        z = lambda *s: torch.zeros(*s, device=dev)
        self._s = {
            "adv": z(T, E), "ret": z(T, E),
            "perm": torch.zeros(cfg.epochs, E, dtype=torch.int64, device=dev),
            "stats": z(4),
            "obs": z(T, Emb, D),
            "done": torch.zeros(T, Emb, dtype=torch.uint8, device=dev),
            "mask": torch.zeros(T, Emb, A, dtype=torch.uint8, device=dev),
            "acts": torch.zeros(T, Emb, dtype=torch.int64, device=dev),
            "olp": z(T, Emb), "ov": z(T, Emb),
            "advm": z(T, Emb), "retm": z(T, Emb),
            "h0": z(Emb, H),
            **{k: torch.zeros(B, H, device=dev,
                              dtype=(torch.bfloat16 if self.dtype == "bf16"
                                     else torch.float32))
               for k in ("xs", "rs", "zs", "ns", "ghns", "hprevs", "hs")},
            "head": z(B, HEADC), "dhead": z(B, HEADC),
            "dh_ext": z(T, Emb, H),
            "dgi": z(B, 3 * H), "dgh": z(B, 3 * H),
            "dxpre": z(B, H),
        }
        gflat = torch.zeros(D * H + H + 2 * (H * 3 * H + 3 * H)
                            + H * HEADC + HEADC, device=dev)
        off = 0

        def carve(*shape):
            nonlocal off
            n = 1
            for sd in shape:
                n *= sd
            t = gflat[off:off + n].view(*shape)
            off += n
            return t
        self._s["gflat"] = gflat
        offs = {}
        for name, shape in (("dWin", (D, H)), ("dbin", (H,)),
                            ("dWih", (H, 3 * H)), ("dbih", (3 * H,)),
                            ("dWhh", (H, 3 * H)), ("dbhh", (3 * H,)),
                            ("dWh", (H, HEADC)), ("dbh", (HEADC,))):
            offs[name] = off
            self._s[name] = carve(*shape)
        if self.use_hip_adam:
            from ..ops.train import HipAdam
            from ..parallel.dist import allreduce_flat
            pol = self.policy
            W = lambda name, R, C, c0, c1, p: dict(
                goff=offs[name], R=R, C=C, c0=c0, c1=c1, param=p,
                transposed=True)
            Bs = lambda name, C, c0, c1, p: dict(
                goff=offs[name], R=1, C=C, c0=c0, c1=c1, param=p,
                transposed=False)
            segs = [
                W("dWin", D, H, 0, H, pol.inp.weight),
                W("dWih", H, 3 * H, 0, 3 * H, pol.cell.weight_ih),
                W("dWhh", H, 3 * H, 0, 3 * H, pol.cell.weight_hh),
                W("dWh", H, HEADC, 0, A, pol.pi.weight),
                W("dWh", H, HEADC, A, A + 1, pol.v.weight),
                Bs("dbin", H, 0, H, pol.inp.bias),
                Bs("dbih", 3 * H, 0, 3 * H, pol.cell.bias_ih),
                Bs("dbhh", 3 * H, 0, 3 * H, pol.cell.bias_hh),
                Bs("dbh", HEADC, 0, A, pol.pi.bias),
                Bs("dbh", HEADC, A, A + 1, pol.v.bias),
            ]
            self.hip_opt = HipAdam(
                gflat, segs, lr=cfg.lr, eps=1e-5,
                max_norm=cfg.max_grad_norm,
                grad_reduce=allreduce_flat if self.grad_sync is not None
                else None)
            self.opt = self.hip_opt
        from ..ckpt.io import resolve_pending_opt_state
        resolve_pending_opt_state(self)

    def load_optim_state(self, sd):
        """Stash checkpointed optimizer state; applied when _alloc creates
        the real optimizer (HipAdam is lazy — see ckpt.io)."""
        self._pending_opt_state = sd

    def _refill_perm(self):
        for ep in range(self.cfg.epochs):
            torch.randperm(self._E, generator=self.gen, device="cuda",
                           out=self._s["perm"][ep])

    def _update_body(self, buf: RolloutBuffer):
        cfg = self.cfg
        s = self._s
        mi = self.mirror
        pol = self.policy
        if cfg.norm_reward:
            if not hasattr(self, "_rnorm"):
                from .rewardnorm import RewardNormalizer
                self._rnorm = RewardNormalizer(cfg.gamma, buf.reward.device)
            self._rnorm(buf.reward, buf.done)
        gae_scan(buf.reward, buf.value, buf.done, buf.last_value,
                 cfg.gamma, cfg.lam, adv=s["adv"], ret=s["ret"])
        adv = s["adv"]
        if cfg.norm_adv:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        s["stats"].zero_()
        mi.refresh()
        T, Emb, B = self._T, self._Emb, self._B
        A, H = self._A, self._H
        for ep in range(cfg.epochs):
            for i in range(cfg.minibatches):
                ix = s["perm"][ep, i * Emb:(i + 1) * Emb]
                torch.index_select(buf.obs, 1, ix, out=s["obs"])
                torch.index_select(buf.done, 1, ix, out=s["done"])
                torch.index_select(buf.mask, 1, ix, out=s["mask"])
                torch.index_select(buf.actions, 1, ix, out=s["acts"])
                torch.index_select(buf.logp, 1, ix, out=s["olp"])
                torch.index_select(buf.value, 1, ix, out=s["ov"])
                torch.index_select(adv, 1, ix, out=s["advm"])
                torch.index_select(s["ret"], 1, ix, out=s["retm"])
                torch.index_select(buf.h0, 0, ix, out=s["h0"])
                if self.dtype == "bf16":
                    gru_seq_fwd_bf16(s["obs"], s["done"], s["h0"], mi.wtin,
                                     mi.bin, mi.wih, mi.bih, mi.whh, mi.bhh,
                                     s["xs"], s["rs"], s["zs"], s["ns"],
                                     s["ghns"], s["hprevs"], s["hs"])
                    head_fwd_bf16(s["hs"], mi.wth, mi.bh, s["head"])
                else:
                    gru_seq_fwd(s["obs"], s["done"], s["h0"], mi.wtin,
                                mi.bin, mi.wih, mi.bih, mi.whh, mi.bhh,
                                s["xs"], s["rs"], s["zs"], s["ns"],
                                s["ghns"], s["hprevs"], s["hs"])
                    head_fwd(s["hs"], mi.wth, mi.bh, s["head"])
                ppo_head(s["head"], s["mask"].view(B, A),
                         s["acts"].view(B), s["olp"].view(B),
                         s["ov"].view(B), s["advm"].reshape(B),
                         s["retm"].view(B), cfg.clip, cfg.vf_coef,
                         cfg.ent_coef, s["dhead"], s["stats"],
                         kl_coef=getattr(cfg, "kl_coef", 0.0),
                         ent_floor=getattr(cfg, "ent_floor", 0.0),
                         ent_floor_coef=getattr(cfg, "ent_floor_coef", 0.0))
                if self.dtype == "bf16":
                    bwd_dx_bf16(s["dhead"], mi.whT, None,
                                s["dh_ext"].view(B, H))
                    gru_seq_bwd_bf16(s["dh_ext"], s["done"], s["rs"],
                                     s["zs"], s["ns"], s["ghns"],
                                     s["hprevs"], mi.whh_bwd, s["dgi"],
                                     s["dgh"])
                    bwd_dx_bf16(s["dgi"], mi.wih_bwd, s["xs"], s["dxpre"])
                    s["gflat"].zero_()
                    bwd_dw_bf16(s["hs"], s["dhead"], s["dWh"])
                    colsum(s["dhead"], s["dbh"])
                    bwd_dw_bf16(s["xs"], s["dgi"], s["dWih"])
                    colsum(s["dgi"], s["dbih"])
                    bwd_dw_bf16(s["hprevs"], s["dgh"], s["dWhh"])
                    colsum(s["dgh"], s["dbhh"])
                    bwd_dw_bf16(s["obs"].view(B, self._D), s["dxpre"],
                                s["dWin"])
                    colsum(s["dxpre"], s["dbin"])
                else:
                    bwd_dx(s["dhead"], mi.whT, None, s["dh_ext"].view(B, H))
                    gru_seq_bwd(s["dh_ext"], s["done"], s["rs"], s["zs"],
                                s["ns"], s["ghns"], s["hprevs"], mi.whh_bwd,
                                s["dgi"], s["dgh"])
                    bwd_dx(s["dgi"], mi.wih_bwd, s["xs"], s["dxpre"])
                    s["gflat"].zero_()
                    bwd_dw(s["hs"], s["dhead"], s["dWh"])
                    colsum(s["dhead"], s["dbh"])
                    bwd_dw(s["xs"], s["dgi"], s["dWih"])
                    colsum(s["dgi"], s["dbih"])
                    bwd_dw(s["hprevs"], s["dgh"], s["dWhh"])
                    colsum(s["dgh"], s["dbhh"])
                    bwd_dw(s["obs"].view(B, self._D), s["dxpre"], s["dWin"])
                    colsum(s["dxpre"], s["dbin"])
                if self.hip_opt is not None:
                    self.hip_opt.step()   # fused clip+Adam on gflat
                else:
                    pol.inp.weight.grad.copy_(s["dWin"].t())
                    pol.inp.bias.grad.copy_(s["dbin"])
                    pol.cell.weight_ih.grad.copy_(s["dWih"].t())
                    pol.cell.bias_ih.grad.copy_(s["dbih"])
                    pol.cell.weight_hh.grad.copy_(s["dWhh"].t())
                    pol.cell.bias_hh.grad.copy_(s["dbhh"])
                    pol.pi.weight.grad.copy_(s["dWh"][:, :A].t())
                    pol.pi.bias.grad.copy_(s["dbh"][:A])
                    pol.v.weight.grad.copy_(s["dWh"][:, A:A + 1].t())
                    pol.v.bias.grad.copy_(s["dbh"][A:A + 1])
                    if self.grad_sync is not None:
                        self.grad_sync(self.policy)
                    torch.nn.utils.clip_grad_norm_(pol.parameters(),
                                                   cfg.max_grad_norm)
                    self.opt.step()
                mi.refresh()

    def _capture(self, buf: RolloutBuffer):
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            self._update_body(buf)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._update_body(buf)
        self._graph = g

    def update(self, buf: RolloutBuffer) -> dict:
        if self._s is None:
            self._alloc(buf)
        self._refill_perm()
        if self.use_graph and self._graph is None:
            try:
                self._capture(buf)
            except Exception as e:
                import warnings
                warnings.warn(f"hipGraph capture of the GRU update failed "
                              f"({e!r}); falling back to eager execution")
                self.use_graph = False
                self._graph = None
            from ..parallel.dist import agree_all
            if not agree_all(self.use_graph):   # rank-consistent fallback
                self.use_graph = False
                self._graph = None
        if self.use_graph:
            self._graph.replay()
        else:
            self._update_body(buf)
        st = self._s["stats"].cpu()
        n = self.cfg.epochs * self.cfg.minibatches * self._B
        return {
            "pi_loss": st[0].item() / n,
            "v_loss": st[1].item() / n,
            "entropy": st[2].item() / n,
            "clipfrac": st[3].item() / n,
            "mean_reward": buf.reward.mean().item(),
        }
