"""rlgpuschedule_amd — MI355X-native RL framework for GPU-cluster scheduling.

Built from scratch (not a port) to the capability spec of
matthewygf/RLGPUSchedule: a batched on-device discrete-event GPU-cluster
simulator (HIP/CDNA4 step kernel), MLP/GRU policy+value heads, GAE, PPO/A2C
and IMPALA-style async actor/learner, with data parallelism over RCCL/xGMI.
"""
__version__ = "0.1.0"
