"""HipPPOLearner — the PPO update with EVERY compute stage hand-written HIP.

Per minibatch (8 kernel launches + a few small torch copies, all inside one
captured hipGraph per iteration):
  fwd_train      MFMA forward saving h1,h2,head
  ppo_head       masked softmax + ratio-clip + value-clip + entropy, fused
                 forward AND backward (dhead)
  bwd_dx x2      dh2, dh1 via MFMA GEMMs with fused tanh'
  bwd_dw x3      dWh, dW2, dW1 (LDS-tiled split-batch MFMA + atomics)
  colsum x3      bias grads
then grad assembly into param.grad (transposed torch copies), global-norm
clip and Adam (torch, capturable). GAE is the HIP scan.

Numerics: tests/test_train_kernels.py (gradient check vs torch autograd)
and tests/test_ppo_gpu.py (parameter-level parity with the reference
learner).
"""
from __future__ import annotations

import torch

from ..models.hip_train import HEADC, TrainMirror, TrainMirrorBf16
from ..ops.nn import gae_scan
from ..ops.train import (bwd_dw, bwd_dw_bf16, bwd_dx, bwd_dx_bf16, colsum,
                         fwd_train, fwd_train_bf16, gather_mb, ppo_head)
from .ppo import PPOConfig
from .rollout import RolloutBuffer


class HipPPOLearner:
    def __init__(self, policy, cfg: PPOConfig, grad_sync=None,
                 use_graph: bool = True, use_hip_adam: bool = True,
                 dtype: str = "fp32"):
        """dtype="bf16": mixed-precision update (bf16 MFMA GEMMs +
        bf16-saved activations, f32 loss/grads/Adam). "fp32" matches the
        torch reference to gradient-check tolerance."""
        assert not policy.is_recurrent
        assert dtype in ("fp32", "bf16")
        self.policy = policy
        self.cfg = cfg
        self.dtype = dtype
        self.grad_sync = grad_sync
        self.use_hip_adam = use_hip_adam
        self.hip_opt = None
        # graph capture is attempted even with DP grad sync (RCCL ops are
        # capturable on recent ROCm); on capture failure we fall back to
        # eager execution (update() try/except)
        self.use_graph = use_graph
        self.opt = torch.optim.Adam(policy.parameters(), lr=cfg.lr, eps=1e-5,
                                    capturable=self.use_graph)
        self.mirror = (TrainMirrorBf16(policy) if dtype == "bf16"
                       else TrainMirror(policy))
        self.gen = torch.Generator(device="cuda")
        self.gen.manual_seed(7654321)
        self._graph = None
        self._bufs = None
        for p in policy.parameters():
            if p.grad is None:
                p.grad = torch.zeros_like(p)

    def _alloc(self, buf: RolloutBuffer):
        cfg = self.cfg
        T, E = buf.reward.shape
        B = T * E
        dev = buf.reward.device
        A = buf.mask.shape[-1]
        D = buf.obs.shape[-1]
        H = self.policy.hidden
        mb = B // cfg.minibatches
        self._B, self._mb, self._A, self._D, self._H = B, mb, A, D, H
        z = lambda *shape: torch.zeros(*shape, device=dev)
        self._s = {
            "adv": z(T, E), "ret": z(T, E),
            "perm": torch.zeros(cfg.epochs, B, dtype=torch.int64, device=dev),
            "stats": z(4),
            "o": z(mb, D), "m": torch.zeros(mb, A, dtype=torch.uint8, device=dev),
            "a": torch.zeros(mb, dtype=torch.int64, device=dev),
            "olp": z(mb), "ov": z(mb), "advm": z(mb), "retm": z(mb),
            "h1": torch.zeros(mb, H, device=dev,
                              dtype=(torch.bfloat16 if self.dtype == "bf16"
                                     else torch.float32)),
            "h2": torch.zeros(mb, H, device=dev,
                              dtype=(torch.bfloat16 if self.dtype == "bf16"
                                     else torch.float32)),
            "head": z(mb, HEADC),
            "dhead": z(mb, HEADC), "dh2": z(mb, H), "dh1": z(mb, H),
        }
        # grad buffers as views of ONE flat tensor -> a single zero_() per
        # minibatch instead of six
        gflat = torch.zeros(D * H + H * H + H * HEADC + 2 * H + HEADC,
                            device=dev)
        off = 0
        def carve(*shape):
            nonlocal off
            n = 1
            for sdim in shape:
                n *= sdim
            t = gflat[off:off + n].view(*shape)
            off += n
            return t
        self._s["gflat"] = gflat
        offs = {}
        for name, shape in (("dW1", (D, H)), ("dW2", (H, H)),
                            ("dWh", (H, HEADC)), ("db1", (H,)),
                            ("db2", (H,)), ("dbh", (HEADC,))):
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
                W("dW1", D, H, 0, H, pol.trunk[0].weight),
                W("dW2", H, H, 0, H, pol.trunk[1].weight),
                W("dWh", H, HEADC, 0, A, pol.pi.weight),
                W("dWh", H, HEADC, A, A + 1, pol.v.weight),
                Bs("db1", H, 0, H, pol.trunk[0].bias),
                Bs("db2", H, 0, H, pol.trunk[1].bias),
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
            torch.randperm(self._B, generator=self.gen, device="cuda",
                           out=self._s["perm"][ep])

    def _update_body(self, buf: RolloutBuffer):
        cfg = self.cfg
        if cfg.norm_reward:
            if not hasattr(self, "_rnorm"):
                from .rewardnorm import RewardNormalizer
                self._rnorm = RewardNormalizer(cfg.gamma, buf.reward.device)
            self._rnorm(buf.reward, buf.done)
        s = self._s
        pol = self.policy
        mi = self.mirror
        gae_scan(buf.reward, buf.value, buf.done, buf.last_value,
                 cfg.gamma, cfg.lam, adv=s["adv"], ret=s["ret"])
        T, E = buf.reward.shape
        B = T * E
        obs = buf.obs.reshape(B, -1)
        mask = buf.mask.reshape(B, -1)
        acts = buf.actions.reshape(B)
        old_logp = buf.logp.reshape(B)
        old_v = buf.value.reshape(B)
        adv = s["adv"].reshape(B)
        ret = s["ret"].reshape(B)
        if cfg.norm_adv:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        s["stats"].zero_()
        mi.refresh()
        A = self._A
        for ep in range(cfg.epochs):
            for i in range(cfg.minibatches):
                ix = s["perm"][ep, i * self._mb:(i + 1) * self._mb]
                gather_mb(ix, obs, mask, acts, old_logp, old_v,
                          adv.contiguous(), ret, s["o"], s["m"], s["a"],
                          s["olp"], s["ov"], s["advm"], s["retm"])
                if self.dtype == "bf16":
                    fwd_train_bf16(s["o"], mi.wt1, mi.b1, mi.wt2, mi.b2,
                                   mi.wth, mi.bh, s["h1"], s["h2"], s["head"])
                else:
                    fwd_train(s["o"], mi.wt1, mi.b1, mi.wt2, mi.b2, mi.wth,
                              mi.bh, s["h1"], s["h2"], s["head"])
                ppo_head(s["head"], s["m"], s["a"], s["olp"], s["ov"],
                         s["advm"], s["retm"], cfg.clip, cfg.vf_coef,
                         cfg.ent_coef, s["dhead"], s["stats"],
                         kl_coef=getattr(cfg, "kl_coef", 0.0),
                         ent_floor=getattr(cfg, "ent_floor", 0.0),
                         ent_floor_coef=getattr(cfg, "ent_floor_coef", 0.0))
                if self.dtype == "bf16":
                    bwd_dx_bf16(s["dhead"], mi.whT, s["h2"], s["dh2"])
                    bwd_dx_bf16(s["dh2"], mi.w2T, s["h1"], s["dh1"])
                    s["gflat"].zero_()
                    bwd_dw_bf16(s["h2"], s["dhead"], s["dWh"])
                    bwd_dw_bf16(s["h1"], s["dh2"], s["dW2"])
                    bwd_dw_bf16(s["o"], s["dh1"], s["dW1"])
                else:
                    bwd_dx(s["dhead"], mi.whT, s["h2"], s["dh2"])
                    bwd_dx(s["dh2"], mi.w2T, s["h1"], s["dh1"])
                    s["gflat"].zero_()
                    bwd_dw(s["h2"], s["dhead"], s["dWh"])
                    bwd_dw(s["h1"], s["dh2"], s["dW2"])
                    bwd_dw(s["o"], s["dh1"], s["dW1"])
                colsum(s["dhead"], s["dbh"])
                colsum(s["dh2"], s["db2"])
                colsum(s["dh1"], s["db1"])
                if self.hip_opt is not None:
                    self.hip_opt.step()   # fused clip+Adam on gflat
                else:
                    # torch fallback: assemble into param grads
                    pol.trunk[0].weight.grad.copy_(s["dW1"].t())
                    pol.trunk[0].bias.grad.copy_(s["db1"])
                    pol.trunk[1].weight.grad.copy_(s["dW2"].t())
                    pol.trunk[1].bias.grad.copy_(s["db2"])
                    pol.pi.weight.grad.copy_(s["dWh"][:, :A].t())
                    pol.pi.bias.grad.copy_(s["dbh"][:A])
                    pol.v.weight.grad.copy_(s["dWh"][:, A:A + 1].t())
                    pol.v.bias.grad.copy_(s["dbh"][A:A + 1])
                    if self.grad_sync is not None:
                        self.grad_sync(self.policy)
                    torch.nn.utils.clip_grad_norm_(pol.parameters(),
                                                   cfg.max_grad_norm)
                    self.opt.step()
                mi.refresh()   # params changed; next minibatch needs them

    def _capture(self, buf: RolloutBuffer):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._update_body(buf)   # warmup (autograd/Adam/comm state)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._update_body(buf)
        self._graph = g

    def update(self, buf: RolloutBuffer) -> dict:
        if self._bufs is None:
            self._alloc(buf)
            self._bufs = True
        self._refill_perm()
        if self.use_graph and self._graph is None:
            try:
                self._capture(buf)
            except Exception as e:
                import warnings
                warnings.warn(f"hipGraph capture of the PPO update failed "
                              f"({e!r}); falling back to eager execution")
                self.use_graph = False
                self._graph = None
            # rank-consistent decision: if any DP rank failed to capture,
            # all ranks run eager so collective sequences stay aligned
            from ..parallel.dist import agree_all
            if not agree_all(self.use_graph):
                self.use_graph = False
                self._graph = None
        if self.use_graph:
            self._graph.replay()
        else:
            self._update_body(buf)
        st = self._s["stats"].cpu()
        n = self.cfg.epochs * self.cfg.minibatches * self._mb
        return {
            "pi_loss": st[0].item() / n,
            "v_loss": st[1].item() / n,
            "entropy": st[2].item() / n,
            "clipfrac": st[3].item() / n,
            "mean_reward": buf.reward.mean().item(),
        }
