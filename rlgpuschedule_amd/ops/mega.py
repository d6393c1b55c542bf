"""ctypes binding for the fused whole-rollout megakernel
(librlsched_mega.so). One launch = T steps of (actor forward + sampling +
env transition) + the GAE bootstrap value — see csrc/rollout_mega.hip.

Buffer ABI: the env's 34 pointers (envstep.BUF_ORDER) followed by
  wt1,b1,wt2,b2,wth,bh, u_all,
  obs,mask,actions,logp,value,reward,done  (rollout [T,E,...]),
  obs_last, mask_last, last_value.
"""
from __future__ import annotations

import ctypes

import numpy as np

from .build import build_lib, verify_loaded
from .envstep import BUF_ORDER

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        so = build_lib("librlsched_mega.so")
        lib = ctypes.CDLL(str(so))
        verify_loaded(lib, "librlsched_mega.so")
        lib.rlsched_mega_rollout.restype = ctypes.c_int
        lib.rlsched_mega_rollout.argtypes = [ctypes.c_void_p] * 3
        _lib = lib
    return _lib


def mega_rollout(env, actor, buf, u_all, obs_last, mask_last, stream: int):
    """Run the whole T-step rollout in one kernel launch.

    env: HipVectorEnv (single shard). actor: HipActorMLP (fp32 packed
    weights). buf: RolloutBuffer with obs[0]/mask[0] pre-filled.
    """
    T, E = buf.reward.shape
    params = np.concatenate([
        env.binding.params,
        np.asarray([T, actor.H], np.int64),
    ])
    ptrs = [env.binding._ptrs[k] for k in BUF_ORDER]
    ptrs += [actor.wt1.data_ptr(), actor.b1.data_ptr(),
             actor.wt2.data_ptr(), actor.b2.data_ptr(),
             actor.wth.data_ptr(), actor.bh.data_ptr(),
             u_all.data_ptr(),
             buf.obs.data_ptr(), buf.mask.data_ptr(),
             buf.actions.data_ptr(), buf.logp.data_ptr(),
             buf.value.data_ptr(),
             buf.reward.data_ptr(), buf.done.data_ptr(),
             obs_last.data_ptr(), mask_last.data_ptr(),
             buf.last_value.data_ptr()]
    bufs = np.asarray(ptrs, dtype=np.uint64)
    rc = load().rlsched_mega_rollout(
        np.ascontiguousarray(params).ctypes.data_as(ctypes.c_void_p),
        bufs.ctypes.data_as(ctypes.c_void_p),
        ctypes.c_void_p(stream))
    if rc != 0:
        raise RuntimeError(f"rlsched_mega_rollout failed: {rc}")
