"""ctypes bindings for the HIP NN kernels (librlsched_nn.so): GAE scan and
the fused MLP actor. Launches are hipGraph-capturable."""
from __future__ import annotations

import ctypes

import numpy as np
import torch

from .build import build_lib, verify_loaded

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        so = build_lib("librlsched_nn.so")
        lib = ctypes.CDLL(str(so))
        verify_loaded(lib, "librlsched_nn.so")
        lib.rlsched_gae.restype = ctypes.c_int
        lib.rlsched_gae.argtypes = [ctypes.c_void_p] * 4
        lib.rlsched_actor_mlp.restype = ctypes.c_int
        lib.rlsched_actor_mlp.argtypes = [ctypes.c_void_p] * 3
        _lib = lib
    return _lib


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


def _ptrs(*tensors: torch.Tensor) -> np.ndarray:
    return np.array([t.data_ptr() for t in tensors], dtype=np.uint64)


def gae_scan(reward: torch.Tensor, value: torch.Tensor, done: torch.Tensor,
             last_value: torch.Tensor, gamma: float, lam: float,
             adv: torch.Tensor | None = None, ret: torch.Tensor | None = None):
    """HIP GAE(λ) segmented reverse scan. reward/value [T,E] f32,
    done [T,E] u8, last_value [E]. Returns (adv, ret)."""
    T, E = reward.shape
    assert done.dtype == torch.uint8
    adv = torch.empty_like(reward) if adv is None else adv
    ret = torch.empty_like(reward) if ret is None else ret
    ip = np.array([T, E], np.int64)
    fp = np.array([gamma, lam], np.float64)
    bufs = _ptrs(reward, value, done, last_value, adv, ret)
    rc = load().rlsched_gae(
        ip.ctypes.data_as(ctypes.c_void_p), fp.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(reward)))
    if rc != 0:
        raise RuntimeError(f"rlsched_gae failed: {rc}")
    return adv, ret


def actor_mlp(obs, mask, u, wt1, b1, wt2, b2, wth, bh, actions, logp, value,
              value_only: bool = False):
    """Fused MLP policy forward + masked inverse-CDF sampling (one launch)."""
    E, D = obs.shape
    H = wt2.shape[2]  # wt2 packed [H/16, 4, H, 4]
    A = mask.shape[1]
    ip = np.array([E, D, H, A, 1 if value_only else 0], np.int64)
    bufs = _ptrs(obs, mask, u, wt1, b1, wt2, b2, wth, bh, actions, logp, value)
    rc = load().rlsched_actor_mlp(
        ip.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(obs)))
    if rc != 0:
        raise RuntimeError(f"rlsched_actor_mlp failed: {rc} "
                           f"(check D%4==0, H%64==0, A<=31)")
    return actions, logp, value


def _load_ppo():
    lib = load()
    if not hasattr(lib, "_ppo_ready"):
        lib.rlsched_ppo_loss.restype = ctypes.c_int
        lib.rlsched_ppo_loss.argtypes = [ctypes.c_void_p] * 4
        lib._ppo_ready = True
    return lib


def ppo_loss_fused(logits, value, actions, old_logp, old_value, adv, ret,
                   clip, vf_coef, ent_coef, dlogits, dvalue, stats):
    """Fused PPO objective forward+backward. `stats` [4] f32 is ACCUMULATED
    (zero it first): [pi_loss_sum, v_loss_sum, entropy_sum, clipfrac_sum].
    dlogits/dvalue get the loss gradients (mean-scaled)."""
    B, A = logits.shape
    ip = np.array([B, A], np.int64)
    fp = np.array([clip, vf_coef, ent_coef, 1.0 / B], np.float64)
    bufs = _ptrs(logits, value, actions, old_logp, old_value, adv, ret,
                 dlogits, dvalue, stats)
    rc = _load_ppo().rlsched_ppo_loss(
        ip.ctypes.data_as(ctypes.c_void_p), fp.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(logits)))
    if rc != 0:
        raise RuntimeError(f"rlsched_ppo_loss failed: {rc}")


def _load_vtrace():
    lib = load()
    if not hasattr(lib, "_vtrace_ready"):
        lib.rlsched_vtrace.restype = ctypes.c_int
        lib.rlsched_vtrace.argtypes = [ctypes.c_void_p] * 4
        lib._vtrace_ready = True
    return lib


def vtrace_scan(reward, value, done, last_value, lp_cur, lp_beh,
                gamma, rho_bar=1.0, c_bar=1.0, lam=1.0,
                vs=None, pg_adv=None):
    """HIP V-trace reverse scan. All [T,E] f32 (done u8). Returns (vs, pg_adv)."""
    T, E = reward.shape
    vs = torch.empty_like(reward) if vs is None else vs
    pg_adv = torch.empty_like(reward) if pg_adv is None else pg_adv
    ip = np.array([T, E], np.int64)
    fp = np.array([gamma, rho_bar, c_bar, lam], np.float64)
    bufs = _ptrs(reward, value, done, last_value, lp_cur, lp_beh, vs, pg_adv)
    rc = _load_vtrace().rlsched_vtrace(
        ip.ctypes.data_as(ctypes.c_void_p), fp.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(reward)))
    if rc != 0:
        raise RuntimeError(f"rlsched_vtrace failed: {rc}")
    return vs, pg_adv


def _load_gru():
    lib = load()
    if not hasattr(lib, "_gru_ready"):
        lib.rlsched_gru_actor.restype = ctypes.c_int
        lib.rlsched_gru_actor.argtypes = [ctypes.c_void_p] * 3
        lib._gru_ready = True
    return lib


def gru_actor(obs, mask, u, done_prev, hx, hx_out, wtin, bin_, wih, bih,
              whh, bhh, wth, bh, actions, logp, value,
              value_only: bool = False):
    """Fused GRU policy step + masked sampling (one launch).
    done_prev: uint8 [E] tensor, or None to skip the hidden-state reset."""
    E, D = obs.shape
    H = hx.shape[1]
    A = mask.shape[1]
    ip = np.array([E, D, H, A, 1 if value_only else 0], np.int64)
    dp = done_prev.data_ptr() if done_prev is not None else 0
    bufs = np.array([obs.data_ptr(), mask.data_ptr(), u.data_ptr(), dp,
                     hx.data_ptr(), hx_out.data_ptr(), wtin.data_ptr(),
                     bin_.data_ptr(), wih.data_ptr(), bih.data_ptr(),
                     whh.data_ptr(), bhh.data_ptr(), wth.data_ptr(),
                     bh.data_ptr(), actions.data_ptr(), logp.data_ptr(),
                     value.data_ptr()], dtype=np.uint64)
    rc = _load_gru().rlsched_gru_actor(
        ip.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(obs)))
    if rc != 0:
        raise RuntimeError(f"rlsched_gru_actor failed: {rc}")


def _load_bf16():
    lib = load()
    if not hasattr(lib, "_bf16_ready"):
        lib.rlsched_actor_bf16.restype = ctypes.c_int
        lib.rlsched_actor_bf16.argtypes = [ctypes.c_void_p] * 3
        lib._bf16_ready = True
    return lib


def actor_bf16(obs, mask, u, wt1, b1, wt2, b2, wth, bh, actions, logp, value,
               value_only: bool = False):
    """bf16-MFMA fused MLP actor (f32 accumulate/head/sampling)."""
    E, D = obs.shape
    H = wt2.shape[2]  # packed [H/32, 4, H, 8]
    A = mask.shape[1]
    ip = np.array([E, D, H, A, 1 if value_only else 0], np.int64)
    bufs = _ptrs(obs, mask, u, wt1, b1, wt2, b2, wth, bh, actions, logp, value)
    rc = _load_bf16().rlsched_actor_bf16(
        ip.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(obs)))
    if rc != 0:
        raise RuntimeError(f"rlsched_actor_bf16 failed: {rc}")


def gru_bf16(obs, mask, u, done_prev, hx, hx_out, wtin, bin_, wih, bih,
             whh, bhh, wth, bh, actions, logp, value,
             value_only: bool = False):
    """bf16-MFMA fused GRU actor step."""
    lib = load()
    if not hasattr(lib, "_grub_ready"):
        lib.rlsched_gru_bf16.restype = ctypes.c_int
        lib.rlsched_gru_bf16.argtypes = [ctypes.c_void_p] * 3
        lib._grub_ready = True
    E, D = obs.shape
    H = hx.shape[1]
    A = mask.shape[1]
    ip = np.array([E, D, H, A, 1 if value_only else 0], np.int64)
    dp = done_prev.data_ptr() if done_prev is not None else 0
    bufs = np.array([obs.data_ptr(), mask.data_ptr(), u.data_ptr(), dp,
                     hx.data_ptr(), hx_out.data_ptr(), wtin.data_ptr(),
                     bin_.data_ptr(), wih.data_ptr(), bih.data_ptr(),
                     whh.data_ptr(), bhh.data_ptr(), wth.data_ptr(),
                     bh.data_ptr(), actions.data_ptr(), logp.data_ptr(),
                     value.data_ptr()], dtype=np.uint64)
    rc = lib.rlsched_gru_bf16(
        ip.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(_stream(obs)))
    if rc != 0:
        raise RuntimeError(f"rlsched_gru_bf16 failed: {rc}")
