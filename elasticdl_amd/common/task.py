"""Task/shard data model shared by master, workers and tests.

Mirrors the reference protocol types (elasticai_api/proto/elasticai_api.proto:9-105):
TaskType enum, Shard{name, start, end, indices}, Task{task_id, shard,
model_version, type, extended_config}.
"""

from dataclasses import dataclass, field
from typing import Dict, List, Optional


class TaskType:
    TRAINING = "training"
    EVALUATION = "evaluation"
    PREDICTION = "prediction"
    WAIT = "wait"
    TRAIN_END_CALLBACK = "train_end_callback"
    NONE = "none"  # empty task: no work left, worker should exit


@dataclass
class Shard:
    name: str
    start: int
    end: int
    # optional explicit record indices (record-level shuffle support,
    # reference task_manager.py:319-321)
    indices: Optional[List[int]] = None

    @property
    def size(self) -> int:
        return self.end - self.start


@dataclass
class Task:
    task_id: int
    shard: Optional[Shard]
    type: str
    model_version: int = -1
    extended_config: Dict = field(default_factory=dict)

    def to_wire(self) -> Dict:
        d = {
            "task_id": self.task_id,
            "type": self.type,
            "model_version": self.model_version,
            "extended_config": self.extended_config,
        }
        if self.shard is not None:
            d["shard"] = {
                "name": self.shard.name,
                "start": self.shard.start,
                "end": self.shard.end,
                "indices": self.shard.indices,
            }
        return d

    @staticmethod
    def from_wire(d: Dict) -> "Task":
        shard = None
        if "shard" in d and d["shard"] is not None:
            s = d["shard"]
            shard = Shard(s["name"], s["start"], s["end"], s.get("indices"))
        return Task(
            task_id=d["task_id"],
            shard=shard,
            type=d["type"],
            model_version=d.get("model_version", -1),
            extended_config=d.get("extended_config", {}),
        )


def is_empty_task(task: Task) -> bool:
    return task.type == TaskType.NONE
