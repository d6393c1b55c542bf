"""rlgpuschedule_amd — MI355X-native RL framework for GPU-cluster scheduling.

Built from scratch (not a port) to the capability spec of
matthewygf/RLGPUSchedule: a batched on-device discrete-event GPU-cluster
simulator (HIP/CDNA4 step kernel), MLP/GRU policy+value heads, GAE, PPO/A2C
and IMPALA-style async actor/learner, with data parallelism over RCCL/xGMI.

Public API (lazy imports; heavy deps load on first use):
    EnvSpec, cluster16/64/256/1024, ClusterSchedEnv, CpuVectorEnv,
    HipVectorEnv, MLPPolicy, GRUPolicy, TrainConfig, CONFIGS, train,
    JobTrace, philly_like, poisson
"""
__version__ = "0.1.0"

_EXPORTS = {
    "EnvSpec": "rlgpuschedule_amd.env.spec",
    "cluster16": "rlgpuschedule_amd.env.spec",
    "cluster64": "rlgpuschedule_amd.env.spec",
    "cluster256": "rlgpuschedule_amd.env.spec",
    "cluster1024": "rlgpuschedule_amd.env.spec",
    "ClusterSchedEnv": "rlgpuschedule_amd.env.gym_api",
    "CpuVectorEnv": "rlgpuschedule_amd.env.gym_api",
    "HipVectorEnv": "rlgpuschedule_amd.env.hip_env",
    "MLPPolicy": "rlgpuschedule_amd.models.policy",
    "GRUPolicy": "rlgpuschedule_amd.models.policy",
    "TrainConfig": "rlgpuschedule_amd.trainer",
    "CONFIGS": "rlgpuschedule_amd.trainer",
    "train": "rlgpuschedule_amd.trainer",
    "JobTrace": "rlgpuschedule_amd.traces.synthetic",
    "philly_like": "rlgpuschedule_amd.traces.synthetic",
    "poisson": "rlgpuschedule_amd.traces.synthetic",
}


def __getattr__(name):
    if name in _EXPORTS:
        import importlib
        mod = importlib.import_module(_EXPORTS[name])
        return getattr(mod, name)
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")


def __dir__():
    return sorted(list(globals()) + list(_EXPORTS))
