"""Pending-task accounting when batches straddle shard boundaries
(SURVEY hard-part: data_shard_service.py:111-148 semantics)."""

from elasticdl_amd.common.task import Shard, Task, TaskType
from elasticdl_amd.worker.data_shard_service import DataShardService


class FakeMC:
    def __init__(self, tasks):
        self.tasks = list(tasks)
        self.reported = []

    def get_task(self):
        if self.tasks:
            return self.tasks.pop(0)
        return Task(0, None, TaskType.NONE)

    def report_task_result(self, task_id, err_message=""):
        self.reported.append((task_id, err_message))


def make_tasks(sizes):
    return [
        Task(i + 1, Shard("f", 0, s), TaskType.TRAINING)
        for i, s in enumerate(sizes)
    ]


def test_batch_straddles_two_shards():
    mc = FakeMC(make_tasks([32, 32]))
    svc = DataShardService(mc, batch_size=50)
    svc.fetch_task()
    svc.fetch_task()
    # one 50-record batch: finishes task 1 (32) and consumes 18 of task 2
    assert svc.report_batch_done(50) is True
    assert mc.reported == [(1, "")]
    assert svc.pending_count == 1
    # next 14 records: task 2 complete
    assert svc.report_batch_done(14) is True
    assert mc.reported == [(1, ""), (2, "")]
    assert svc.pending_count == 0


def test_batch_covers_multiple_whole_shards():
    mc = FakeMC(make_tasks([8, 8, 8]))
    svc = DataShardService(mc, batch_size=24)
    for _ in range(3):
        svc.fetch_task()
    assert svc.report_batch_done(24) is True
    assert [t for t, _ in mc.reported] == [1, 2, 3]


def test_failed_task_removed_from_pending():
    mc = FakeMC(make_tasks([16, 16]))
    svc = DataShardService(mc, batch_size=16)
    svc.fetch_task()
    svc.fetch_task()
    svc.report_task_failed(1, "boom")
    assert mc.reported == [(1, "boom")]
    assert svc.pending_count == 1
    svc.report_batch_done(16)
    assert mc.reported[-1] == (2, "")
