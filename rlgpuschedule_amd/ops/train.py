"""ctypes bindings for the hand-written training kernels
(librlsched_train.so): activation-saving MLP forward, masked PPO loss head
(fwd+bwd), backward dX / dW MFMA GEMMs, bias column sums."""
from __future__ import annotations

import ctypes

import numpy as np
import torch

from .build import build_lib, verify_loaded

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        so = build_lib("librlsched_train.so")
        lib = ctypes.CDLL(str(so))
        verify_loaded(lib, "librlsched_train.so")
        for fn, na in (("rlsched_fwd_train", 3), ("rlsched_ppo_head", 4),
                       ("rlsched_bwd_dx", 3), ("rlsched_bwd_dw", 3),
                       ("rlsched_colsum", 3)):
            f = getattr(lib, fn)
            f.restype = ctypes.c_int
            f.argtypes = [ctypes.c_void_p] * na
        _lib = lib
    return _lib


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


def _ptrs(*tensors: torch.Tensor) -> np.ndarray:
    return np.array([t.data_ptr() for t in tensors], dtype=np.uint64)


def _call(fn_name, ip, bufs, stream, fp=None):
    ipa = np.asarray(ip, np.int64)
    args = [ipa.ctypes.data_as(ctypes.c_void_p)]
    if fp is not None:
        fpa = np.asarray(fp, np.float64)
        args.append(fpa.ctypes.data_as(ctypes.c_void_p))
    args += [bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(stream)]
    rc = getattr(load(), fn_name)(*args)
    if rc != 0:
        raise RuntimeError(f"{fn_name} failed: {rc}")


def fwd_train(obs, wt1, b1, wt2, b2, wth, bh, h1, h2, head):
    """Forward pass saving activations: h1,h2 [B,H], head [B,32]."""
    B, D = obs.shape
    H = wt2.shape[2]
    _call("rlsched_fwd_train", [B, D, H],
          _ptrs(obs, wt1, b1, wt2, b2, wth, bh, h1, h2, head), _stream(obs))


def ppo_head(head, mask, actions, old_logp, old_value, adv, ret,
             clip, vf_coef, ent_coef, dhead, stats, kl_coef=0.0,
             ent_floor=0.0, ent_floor_coef=0.0):
    """Masked PPO loss on head rows; writes dhead + accumulates stats[4].
    kl_coef>0 adds the k3 KL-to-behavior penalty (folded into pi stat).
    ent_floor/ent_floor_coef: rows with entropy < ent_floor get an extra
    entropy bonus (per-row hinge against entropy collapse)."""
    B, A = mask.shape
    _call("rlsched_ppo_head", [B, A],
          _ptrs(head, mask, actions, old_logp, old_value, adv, ret,
                dhead, stats),
          _stream(head), fp=[clip, vf_coef, ent_coef, 1.0 / B, kl_coef,
                             ent_floor, ent_floor_coef])


def bwd_dx(dY, wtp, h, dX):
    """dX = (dY @ W^T) [* (1 - h^2) if h is not None].
    dY [B,K], wtp packed [K,N], dX [B,N]."""
    B, K = dY.shape
    N = dX.shape[1]
    hp = h.data_ptr() if h is not None else 0
    ipa = np.asarray([B, K, N], np.int64)
    bufs = np.array([dY.data_ptr(), wtp.data_ptr(), hp, dX.data_ptr()],
                    dtype=np.uint64)
    rc = load().rlsched_bwd_dx(
        ipa.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_void_p(_stream(dY)))
    if rc != 0:
        raise RuntimeError(f"rlsched_bwd_dx failed: {rc}")


def bwd_dw(X, dY, dW, bc: int = 1024):
    """dW[M,N] = X^T @ dY (accumulated; caller zeroes dW first)."""
    B, M = X.shape
    N = dY.shape[1]
    _call("rlsched_bwd_dw", [B, M, N, bc], _ptrs(X, dY, dW), _stream(X))


def colsum(dY, db):
    """db += column sums of dY (zero db first)."""
    B, N = dY.shape
    _call("rlsched_colsum", [B, N], _ptrs(dY, db), _stream(dY))


def gather_mb(ix, obs, mask, acts, olp, ov, adv, ret,
              o_obs, o_mask, o_acts, o_olp, o_ov, o_adv, o_ret):
    """Gather a PPO minibatch (7 arrays) in ONE kernel launch."""
    lib = load()
    if not hasattr(lib, "_gather_ready"):
        lib.rlsched_gather_mb.restype = ctypes.c_int
        lib.rlsched_gather_mb.argtypes = [ctypes.c_void_p] * 3
        lib._gather_ready = True
    MB = ix.shape[0]
    D = obs.shape[1]
    A = mask.shape[1]
    _call("rlsched_gather_mb", [MB, D, A],
          _ptrs(ix, obs, mask, acts, olp, ov, adv, ret,
                o_obs, o_mask, o_acts, o_olp, o_ov, o_adv, o_ret),
          _stream(obs))


def _load_adam():
    lib = load()
    if not hasattr(lib, "_adam_ready"):
        lib.rlsched_adam_bump.restype = ctypes.c_int
        lib.rlsched_adam_bump.argtypes = [ctypes.c_void_p] * 4
        lib.rlsched_adam_step.restype = ctypes.c_int
        lib.rlsched_adam_step.argtypes = [ctypes.c_void_p] * 4
        lib._adam_ready = True
    return lib


class HipAdam:
    """Fused clip+Adam over the flat grad buffer (2 kernel launches).

    segments: list of dicts with keys
      goff (flat offset), R, C (full grad row stride), c0, c1 (column
      slice), param (torch tensor, f32), transposed (grad [in,out] ->
      param [out,in]; biases/slices use the compact layout).
    Duck-types torch optimizers for checkpointing (state_dict).
    """

    def __init__(self, gflat, segments, lr=3e-4, betas=(0.9, 0.999),
                 eps=1e-5, max_norm=0.5, grad_reduce=None, algo="adam"):
        assert algo in ("adam", "rmsprop")  # rmsprop: betas[1] is alpha
        self.algo = algo
        self.gflat = gflat
        self.lr, self.betas, self.eps = lr, betas, eps
        self.max_norm = max_norm
        self.grad_reduce = grad_reduce  # DP hook: all-reduce gflat
        dev = gflat.device
        total = sum(s["R"] * (s["c1"] - s["c0"]) for s in segments)
        self.m = torch.zeros(total, device=dev)
        self.v = torch.zeros(total, device=dev)
        self.t = torch.zeros(1, device=dev)
        self.bc = torch.ones(2, device=dev)
        self.coef = torch.ones(1, device=dev)
        ip = [len(segments)]
        moff = 0
        for s in segments:
            ip += [s["goff"], moff, s["R"], s["C"], s["c0"], s["c1"]]
            moff += s["R"] * (s["c1"] - s["c0"])
        ip += [1 if s["transposed"] else 0 for s in segments]
        ip += [1 if algo == "rmsprop" else 0]
        self._ip = np.asarray(ip, np.int64)
        self._bump_ip = np.zeros(1, np.int64)
        self._bump_fp = np.asarray(betas, np.float64)
        self._fp = np.asarray([lr, betas[0], betas[1], eps], np.float64)
        self._bufs = np.array(
            [gflat.data_ptr(), self.m.data_ptr(), self.v.data_ptr(),
             self.coef.data_ptr(), self.bc.data_ptr()]
            + [s["param"].data_ptr() for s in segments], dtype=np.uint64)
        self._bump_bufs = np.array([self.t.data_ptr(), self.bc.data_ptr()],
                                   dtype=np.uint64)

    def step(self):
        if self.grad_reduce is not None:
            self.grad_reduce(self.gflat)
        norm = torch.linalg.vector_norm(self.gflat).reshape(1)
        torch.clamp(self.max_norm / (norm + 1e-6), max=1.0, out=self.coef)
        lib = _load_adam()
        st = _stream(self.gflat)
        rc = lib.rlsched_adam_bump(
            self._bump_ip.ctypes.data_as(ctypes.c_void_p),
            self._bump_fp.ctypes.data_as(ctypes.c_void_p),
            self._bump_bufs.ctypes.data_as(ctypes.c_void_p),
            ctypes.c_void_p(st))
        if rc == 0:
            rc = lib.rlsched_adam_step(
                self._ip.ctypes.data_as(ctypes.c_void_p),
                self._fp.ctypes.data_as(ctypes.c_void_p),
                self._bufs.ctypes.data_as(ctypes.c_void_p),
                ctypes.c_void_p(st))
        if rc != 0:
            raise RuntimeError(f"hip adam failed: {rc}")

    # torch-optimizer duck typing for ckpt/io.py
    def state_dict(self):
        return {"hip_adam": True, "m": self.m, "v": self.v, "t": self.t,
                "bc": self.bc}

    def load_state_dict(self, sd):
        if not (isinstance(sd, dict) and sd.get("hip_adam")):
            raise ValueError(
                "HipAdam.load_state_dict got non-HipAdam optimizer state "
                "(e.g. torch Adam); refusing to restore silently — use "
                "ckpt.io.apply_optim_state for cross-path dispatch")
        self.m.copy_(sd["m"].to(self.m.device))
        self.v.copy_(sd["v"].to(self.v.device))
        self.t.copy_(sd["t"].to(self.t.device))
        if "bc" in sd:
            self.bc.copy_(sd["bc"].to(self.bc.device))
        else:
            # legacy checkpoint without the bias-correction denominators
            # (bc = 1 - beta^t, see adam_bump_kernel): recompute from t
            # so bias correction stays right after resume
            t = float(sd["t"].reshape(-1)[0].item())
            self.bc.copy_(torch.tensor(
                [1.0 - self.betas[0] ** t, 1.0 - self.betas[1] ** t],
                device=self.bc.device))


def _load_bf16():
    lib = load()
    if not hasattr(lib, "_tbf16_ready"):
        for fn in ("rlsched_fwd_train_bf16", "rlsched_bwd_dx_bf16",
                   "rlsched_bwd_dw_bf16"):
            f = getattr(lib, fn)
            f.restype = ctypes.c_int
            f.argtypes = [ctypes.c_void_p] * 3
        lib._tbf16_ready = True
    return lib


def fwd_train_bf16(obs, wt1, b1, wt2, b2, wth, bh, h1, h2, head):
    """bf16-MFMA training forward; h1/h2 saved bf16, head f32."""
    _load_bf16()
    B, D = obs.shape
    H = h1.shape[1]
    _call("rlsched_fwd_train_bf16", [B, D, H],
          _ptrs(obs, wt1, b1, wt2, b2, wth, bh, h1, h2, head), _stream(obs))


def bwd_dx_bf16(dY, wtp, h, dX):
    """dX = (dY @ W^T) [* (1-h^2)], bf16 MFMA. dY/dX f32, h bf16 or None."""
    _load_bf16()
    B, K = dY.shape
    N = dX.shape[1]
    hp = h.data_ptr() if h is not None else 0
    ipa = np.asarray([B, K, N], np.int64)
    bufs = np.array([dY.data_ptr(), wtp.data_ptr(), hp, dX.data_ptr()],
                    dtype=np.uint64)
    rc = load().rlsched_bwd_dx_bf16(
        ipa.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_void_p(_stream(dY)))
    if rc != 0:
        raise RuntimeError(f"rlsched_bwd_dx_bf16 failed: {rc}")


def bwd_dw_bf16(X, dY, dW, bc: int = 1024):
    """dW[M,N] = X^T @ dY, bf16 MFMA (X f32 or bf16; dY f32; dW f32,
    caller zeroes)."""
    _load_bf16()
    B, M = X.shape
    N = dY.shape[1]
    _call("rlsched_bwd_dw_bf16",
          [B, M, N, bc, 1 if X.dtype == torch.bfloat16 else 0],
          _ptrs(X, dY, dW), _stream(dY))


def vtrace_head(head, mask, actions, pg_adv, vs, lp_cur, value, dhead,
                stats, vf_coef, ent_coef, logp_only=False):
    """IMPALA loss head. Pass 1 (logp_only): fills lp_cur + value for the
    V-trace scan. Pass 2: loss sums into stats[0..2] + dhead gradient."""
    lib = load()
    if not hasattr(lib, "_vth_ready"):
        lib.rlsched_vtrace_head.restype = ctypes.c_int
        lib.rlsched_vtrace_head.argtypes = [ctypes.c_void_p] * 4
        lib._vth_ready = True
    B, A = mask.shape
    _call("rlsched_vtrace_head", [B, A, 1 if logp_only else 0],
          _ptrs(head, mask, actions, pg_adv, vs, lp_cur, value, dhead, stats),
          _stream(head), fp=[vf_coef, ent_coef, 1.0 / B])
