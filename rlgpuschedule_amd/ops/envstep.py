"""ctypes binding for the batched HIP env kernels (librlsched_env.so).

The buffer-pointer order here is the ABI contract with csrc/env_common.h
(RlsBufs field order). Launches are enqueued on the provided HIP stream and
are hipGraph-capturable (no allocation/sync inside).
"""
from __future__ import annotations

import ctypes

import numpy as np

from .build import build_lib, verify_loaded

BUF_ORDER = [
    "trace_arr", "trace_gpus", "trace_work", "trace_ming",
    "job_lo", "job_hi", "base",
    "clock", "cursor", "node_free", "pend_job", "pend_override", "pend_count",
    "run_valid", "run_job", "run_gpus", "run_finish", "run_start", "run_alloc",
    "run_count", "decisions", "completed", "jct_sum", "reward_sum",
    "actions", "obs", "rewards", "done", "mask",
    "ep_count", "ep_return", "ep_jct_sum", "ep_completed", "ep_decisions",
]
NUM_BUFS = len(BUF_ORDER)
assert NUM_BUFS == 34

_lib = None


def load() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        so = build_lib("librlsched_env.so")
        lib = ctypes.CDLL(str(so))
        verify_loaded(lib, "librlsched_env.so")
        for fn in ("rlsched_env_reset", "rlsched_env_step"):
            f = getattr(lib, fn)
            f.restype = ctypes.c_int
            f.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p]
        lib.rlsched_device_sync.restype = ctypes.c_int
        lib.rlsched_device_sync.argtypes = []
        _lib = lib
    return _lib


class EnvStepBinding:
    def __init__(self, params: np.ndarray, buf_ptrs: dict[str, int]):
        assert params.dtype == np.int64 and len(params) == 17
        self.lib = load()
        self.params = np.ascontiguousarray(params)
        self._ptrs = dict(buf_ptrs)
        self._rebuild()

    def _rebuild(self):
        missing = [k for k in BUF_ORDER if k not in self._ptrs]
        assert not missing, f"unbound buffers: {missing}"
        self.bufs = np.array([self._ptrs[k] for k in BUF_ORDER], dtype=np.uint64)

    def set_ptr(self, name: str, ptr: int):
        self._ptrs[name] = ptr
        self._rebuild()

    def _call(self, fn, stream: int, overrides: dict[str, int] | None):
        bufs = self.bufs
        if overrides:
            ptrs = dict(self._ptrs)
            ptrs.update(overrides)
            bufs = np.array([ptrs[k] for k in BUF_ORDER], dtype=np.uint64)
        rc = fn(
            self.params.ctypes.data_as(ctypes.c_void_p),
            bufs.ctypes.data_as(ctypes.c_void_p),
            ctypes.c_void_p(stream),
        )
        if rc != 0:
            raise RuntimeError(f"HIP env kernel launch failed: hipError {rc}")

    def reset(self, stream: int, **overrides):
        self._call(self.lib.rlsched_env_reset, stream, overrides or None)

    def step(self, stream: int, **overrides):
        self._call(self.lib.rlsched_env_step, stream, overrides or None)
