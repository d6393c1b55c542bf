"""ctypes bindings for the hand-written training kernels
(librlsched_train.so): activation-saving MLP forward, masked PPO loss head
(fwd+bwd), backward dX / dW MFMA GEMMs, bias column sums."""
from __future__ import annotations

import ctypes

import numpy as np
import torch

from .build import build_lib

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        so = build_lib("librlsched_train.so")
        lib = ctypes.CDLL(str(so))
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
             clip, vf_coef, ent_coef, dhead, stats):
    """Masked PPO loss on head rows; writes dhead + accumulates stats[4]."""
    B, A = mask.shape
    _call("rlsched_ppo_head", [B, A],
          _ptrs(head, mask, actions, old_logp, old_value, adv, ret,
                dhead, stats),
          _stream(head), fp=[clip, vf_coef, ent_coef, 1.0 / B])


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
