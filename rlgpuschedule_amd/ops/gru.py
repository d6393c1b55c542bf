"""ctypes bindings for the GRU BPTT kernels (librlsched_gru.so)."""
from __future__ import annotations

import ctypes

import numpy as np
import torch

from .build import build_lib, verify_loaded

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        so = build_lib("librlsched_gru.so")
        lib = ctypes.CDLL(str(so))
        verify_loaded(lib, "librlsched_gru.so")
        for fn in ("rlsched_gru_seq_fwd", "rlsched_gru_seq_bwd",
                   "rlsched_head_fwd"):
            f = getattr(lib, fn)
            f.restype = ctypes.c_int
            f.argtypes = [ctypes.c_void_p] * 3
        _lib = lib
    return _lib


def _stream(t: torch.Tensor) -> int:
    return torch.cuda.current_stream(t.device).cuda_stream


def _call(fn, ip, tensors, stream):
    ipa = np.asarray(ip, np.int64)
    bufs = np.array([t.data_ptr() for t in tensors], dtype=np.uint64)
    rc = getattr(load(), fn)(
        ipa.ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p), ctypes.c_void_p(stream))
    if rc != 0:
        raise RuntimeError(f"{fn} failed: {rc}")


def gru_seq_fwd(obs, done, h0, wtin, bin_, wih, bih, whh, bhh,
                xs, rs, zs, ns, ghns, hprevs, hs):
    """Whole-sequence GRU forward (ONE launch). obs [T,Emb,D], done [T,Emb]
    u8, h0 [Emb,H]; saves all per-step activations for BPTT."""
    T, Emb, D = obs.shape
    H = h0.shape[1]
    _call("rlsched_gru_seq_fwd", [T, Emb, D, H],
          (obs, done, h0, wtin, bin_, wih, bih, whh, bhh,
           xs, rs, zs, ns, ghns, hprevs, hs), _stream(obs))


def gru_seq_bwd(dh_ext, done, rs, zs, ns, ghns, hprevs, whh_bwd,
                dgi_all, dgh_all):
    """Whole-sequence GRU backward (ONE launch): reverse time sweep with
    the carry GEMM in-kernel; emits gate pre-activation grads."""
    T, Emb, H = dh_ext.shape
    _call("rlsched_gru_seq_bwd", [T, Emb, H],
          (dh_ext, done, rs, zs, ns, ghns, hprevs, whh_bwd,
           dgi_all, dgh_all), _stream(dh_ext))


def head_fwd(h, wth, bh, head):
    """head[B,32] = h[B,H] @ wth + bh."""
    B, H = h.shape
    _call("rlsched_head_fwd", [B, H], (h, wth, bh, head), _stream(h))


def _load_bf16():
    lib = load()
    if not hasattr(lib, "_gbf16_ready"):
        for fn in ("rlsched_gru_seq_fwd_bf16", "rlsched_gru_seq_bwd_bf16",
                   "rlsched_head_fwd_bf16"):
            f = getattr(lib, fn)
            f.restype = ctypes.c_int
            f.argtypes = [ctypes.c_void_p] * 3
        lib._gbf16_ready = True
    return lib


def gru_seq_fwd_bf16(obs, done, h0, wtin, bin_, wih, bih, whh, bhh,
                     xs, rs, zs, ns, ghns, hprevs, hs):
    _load_bf16()
    T, Emb, D = obs.shape
    H = h0.shape[1]
    _call("rlsched_gru_seq_fwd_bf16", [T, Emb, D, H],
          (obs, done, h0, wtin, bin_, wih, bih, whh, bhh,
           xs, rs, zs, ns, ghns, hprevs, hs), _stream(obs))


def gru_seq_bwd_bf16(dh_ext, done, rs, zs, ns, ghns, hprevs, whh_bwd,
                     dgi_all, dgh_all):
    _load_bf16()
    T, Emb, H = dh_ext.shape
    _call("rlsched_gru_seq_bwd_bf16", [T, Emb, H],
          (dh_ext, done, rs, zs, ns, ghns, hprevs, whh_bwd,
           dgi_all, dgh_all), _stream(dh_ext))


def head_fwd_bf16(h, wth, bh, head):
    _load_bf16()
    B, H = h.shape
    _call("rlsched_head_fwd_bf16", [B, H], (h, wth, bh, head), _stream(head))
