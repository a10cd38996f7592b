"""Round-robin multi-task sampler over the 12 training datasets.

The 12-in-1 recipe (paper via README.md:6 of the reference) cycles tasks so
every optimizer step trains one task's batch. Determinism requirement: all
DP ranks must draw the SAME task each step (the gradient layout must agree
for the bucketed all-reduce — parallel/ddp.py), while each rank draws a
DIFFERENT data shard of that task. The sampler state (step counter + per-task
epoch counters) is checkpointable (SURVEY.md §5 checkpoint obligation:
model+optimizer+task-sampler state).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

from ..tasks import TRAINING_DATASETS


@dataclass
class SamplerState:
    step: int = 0
    per_task_steps: Dict[str, int] = field(default_factory=dict)


class RoundRobinTaskSampler:
    def __init__(
        self,
        datasets: Sequence[str] = TRAINING_DATASETS,
        rank: int = 0,
        world_size: int = 1,
        weights: Optional[Dict[str, int]] = None,
    ):
        """weights: optional integer repeats per dataset within one cycle
        (the 12-in-1 recipe oversamples small datasets by epoch-stretching;
        integer repeats per round-robin cycle approximate that)."""
        self.rank = rank
        self.world_size = world_size
        self.cycle: List[str] = []
        for d in datasets:
            self.cycle.extend([d] * (weights or {}).get(d, 1))
        self.state = SamplerState(per_task_steps={d: 0 for d in datasets})

    def next_task(self) -> str:
        task = self.cycle[self.state.step % len(self.cycle)]
        self.state.step += 1
        self.state.per_task_steps[task] = self.state.per_task_steps.get(task, 0) + 1
        return task

    def shard_seed(self, task: str) -> int:
        """Per-(task, step, rank) seed so ranks draw disjoint synthetic
        shards deterministically. Uses a stable hash (builtin hash() is
        per-process salted and would desynchronize ranks)."""
        import zlib

        base = zlib.crc32(
            f"{task}:{self.state.per_task_steps.get(task, 0)}".encode()
        ) & 0x3FFFFFFF
        return base * self.world_size + self.rank

    # -- checkpoint --------------------------------------------------------
    def state_dict(self) -> dict:
        return {
            "step": self.state.step,
            "per_task_steps": dict(self.state.per_task_steps),
        }

    def load_state_dict(self, sd: dict) -> None:
        self.state.step = int(sd["step"])
        self.state.per_task_steps = dict(sd["per_task_steps"])
