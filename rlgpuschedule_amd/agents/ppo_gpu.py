"""FusedPPOLearner — GPU fast path for the PPO update.

The loss head (masked softmax, ratio-clip objective, clipped value loss,
entropy, and ALL their gradients) is ONE HIP kernel (rlsched_ppo_loss),
exposed through a custom autograd Function; trunk GEMMs run through torch
(hipBLASLt MFMA) until the hand-written training GEMMs land. The entire
update (GAE scan + epochs x minibatches of forward/backward/Adam) is
captured in a hipGraph and replayed each iteration; minibatch permutations
are refilled into a fixed index buffer outside the graph.

Numerics: tests/test_ppo_gpu.py checks parameter-level agreement with the
reference PPOLearner (agents/ppo.py).
"""
from __future__ import annotations

import torch

from ..ops.nn import gae_scan, ppo_loss_fused
from .ppo import PPOConfig
from .rollout import RolloutBuffer


class _PpoHeadLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, value, actions, old_logp, old_value, adv, ret,
                clip, vf_coef, ent_coef, stats, dlogits, dvalue):
        ppo_loss_fused(logits, value, actions, old_logp, old_value, adv, ret,
                       clip, vf_coef, ent_coef, dlogits, dvalue, stats)
        ctx.save_for_backward(dlogits, dvalue)
        return logits.new_zeros(())

    @staticmethod
    def backward(ctx, g):
        dlogits, dvalue = ctx.saved_tensors
        return (dlogits * g, dvalue * g) + (None,) * 11


class FusedPPOLearner:
    def __init__(self, policy, cfg: PPOConfig, grad_sync=None,
                 use_graph: bool = True):
        assert not policy.is_recurrent, "fused path is the MLP flagship"
        self.policy = policy
        self.cfg = cfg
        self.grad_sync = grad_sync
        # graph capture is attempted even with DP grad sync (RCCL ops are
        # capturable on recent ROCm); on capture failure we fall back to
        # eager execution (update() try/except)
        self.use_graph = use_graph
        # capturable Adam so the optimizer step can live inside the graph
        self.opt = torch.optim.Adam(policy.parameters(), lr=cfg.lr, eps=1e-5,
                                    capturable=self.use_graph)
        self.gen = torch.Generator(device="cuda")
        self.gen.manual_seed(1234567)
        self._graph = None
        self._bufs = None
        self._nup = cfg.epochs * cfg.minibatches

    def _alloc(self, buf: RolloutBuffer):
        cfg = self.cfg
        T, E = buf.reward.shape
        B = T * E
        dev = buf.reward.device
        A = buf.mask.shape[-1]
        mb = B // cfg.minibatches
        self._B, self._mb = B, mb
        bs = {
            "adv": torch.zeros(T, E, device=dev),
            "ret": torch.zeros(T, E, device=dev),
            "perm": torch.zeros(cfg.epochs, B, dtype=torch.int64, device=dev),
            "stats": torch.zeros(4, device=dev),
            "dlogits": torch.zeros(mb, A, device=dev),
            "dvalue": torch.zeros(mb, device=dev),
        }
        self._bufs = bs

    def _refill_perm(self):
        for ep in range(self.cfg.epochs):
            torch.randperm(self._B, generator=self.gen, device="cuda",
                           out=self._bufs["perm"][ep])

    def _update_body(self, buf: RolloutBuffer):
        cfg = self.cfg
        if cfg.norm_reward:
            if not hasattr(self, "_rnorm"):
                from .rewardnorm import RewardNormalizer
                self._rnorm = RewardNormalizer(cfg.gamma, buf.reward.device)
            self._rnorm(buf.reward, buf.done)
        bs = self._bufs
        gae_scan(buf.reward, buf.value, buf.done, buf.last_value,
                 cfg.gamma, cfg.lam, adv=bs["adv"], ret=bs["ret"])
        T, E = buf.reward.shape
        B = T * E
        obs = buf.obs.reshape(B, -1)
        mask = buf.mask.reshape(B, -1)
        acts = buf.actions.reshape(B)
        old_logp = buf.logp.reshape(B)
        old_v = buf.value.reshape(B)
        adv = bs["adv"].reshape(B)
        ret = bs["ret"].reshape(B)
        if cfg.norm_adv:
            adv = (adv - adv.mean()) / (adv.std() + 1e-8)
        bs["stats"].zero_()
        for ep in range(cfg.epochs):
            for i in range(cfg.minibatches):
                ix = bs["perm"][ep, i * self._mb:(i + 1) * self._mb]
                logits, value, _ = self.policy(obs[ix], mask[ix], None)
                loss = _PpoHeadLoss.apply(
                    logits, value, acts[ix], old_logp[ix], old_v[ix],
                    adv[ix], ret[ix], cfg.clip, cfg.vf_coef, cfg.ent_coef,
                    bs["stats"], bs["dlogits"], bs["dvalue"])
                self.opt.zero_grad(set_to_none=False)
                loss.backward()
                if self.grad_sync is not None:
                    self.grad_sync(self.policy)
                torch.nn.utils.clip_grad_norm_(self.policy.parameters(),
                                               cfg.max_grad_norm)
                self.opt.step()

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
        if self.use_graph:
            self._graph.replay()
        else:
            self._update_body(buf)
        st = self._bufs["stats"].cpu()
        n = self._nup * self._mb
        return {
            "pi_loss": st[0].item() / n,
            "v_loss": st[1].item() / n,  # kernel already applies the 0.5
            "entropy": st[2].item() / n,
            "clipfrac": st[3].item() / n,
            "mean_reward": buf.reward.mean().item(),
        }
