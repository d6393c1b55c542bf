"""Static environment configuration.

Normative semantics live in docs/ENV_SPEC.md. The reference repo
(matthewygf/RLGPUSchedule) was mounted empty this round, so
the spec is our own design built to BASELINE.json's capability list.
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class EnvSpec:
    num_nodes: int
    gpus_per_node: int = 8
    queue_visible: int = 16          # K: schedulable slots
    queue_cap: int = 256             # P: pending FIFO capacity
    run_cap: int = 0                 # R: 0 -> min(C, 512)
    max_span: int = 8                # S: max nodes one job may span
    idle_tick_us: int = 1_000_000
    time_scale_us: int = 10_000_000
    horizon: int = 50_000
    gandiva: bool = False            # adds MIGRATE/TIMESLICE actions (v2)
    elastic: bool = False            # elastic multi-GPU jobs (v3)
    obs_version: int = 1             # 2 adds 12 global features (v4)
    elastic_resize: bool = False     # SHRINK/EXPAND running jobs (v5)

    def __post_init__(self):
        if self.run_cap == 0:
            object.__setattr__(self, "run_cap", min(self.total_gpus, 512))
        assert self.num_nodes >= 1 and self.gpus_per_node >= 1
        assert self.queue_visible <= self.queue_cap
        assert not self.elastic_resize or self.elastic, \
            "elastic_resize requires elastic jobs (min_gpus < gpus)" 

    @property
    def total_gpus(self) -> int:
        return self.num_nodes * self.gpus_per_node

    @property
    def num_actions(self) -> int:
        return (self.queue_visible + 1 + (2 if self.gandiva else 0)
                + (2 if self.elastic_resize else 0))

    @property
    def noop_action(self) -> int:
        return self.queue_visible

    @property
    def migrate_action(self) -> int:
        assert self.gandiva
        return self.queue_visible + 1

    @property
    def timeslice_action(self) -> int:
        assert self.gandiva
        return self.queue_visible + 2

    @property
    def shrink_action(self) -> int:
        assert self.elastic_resize
        return self.queue_visible + 1 + (2 if self.gandiva else 0)

    @property
    def expand_action(self) -> int:
        return self.shrink_action + 1

    # obs v2 extras: 8-bucket node free-fraction histogram, pending
    # overflow beyond the visible window, max single-node free fraction,
    # trace-consumption progress, episode progress (docs/ENV_SPEC.md v4)
    OBS_V2_EXTRA = 12

    @property
    def obs_dim(self) -> int:
        return (4 + 5 * self.queue_visible
                + (self.OBS_V2_EXTRA if self.obs_version >= 2 else 0))

    @property
    def max_job_gpus(self) -> int:
        return min(self.total_gpus, self.max_span * self.gpus_per_node)


# The five BASELINE.json configs reference these cluster shapes.
def cluster16(**kw) -> EnvSpec:
    return EnvSpec(num_nodes=4, gpus_per_node=4, **kw)


def cluster64(**kw) -> EnvSpec:
    return EnvSpec(num_nodes=8, gpus_per_node=8, **kw)


def cluster256(**kw) -> EnvSpec:
    return EnvSpec(num_nodes=32, gpus_per_node=8, **kw)


def cluster1024(**kw) -> EnvSpec:
    return EnvSpec(num_nodes=128, gpus_per_node=8, **kw)
