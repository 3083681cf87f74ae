"""Client-side dynamic sharding.

Mirrors elasticai_api/common/data_shard_service.py:46-212:

- DataShardService: fetch tasks from the master, keep a pending deque,
  count consumed records per batch and report a task complete once all its
  records are consumed (batches may straddle shards);
- RecordIndexService: background thread turning shards into a per-record
  index queue (torch Dataset __getitem__ support).
"""

import queue
import threading
from collections import deque
from typing import Optional

from elasticdl_amd.common.task import Task, TaskType


class DataShardService:
    def __init__(self, master_client, batch_size: int):
        self._mc = master_client
        self._batch_size = batch_size
        self._lock = threading.Lock()
        self._pending_tasks = deque()  # (task, remaining_records)
        self.current_task: Optional[Task] = None

    def fetch_task(self, task_type: Optional[str] = None) -> Optional[Task]:
        task = self._mc.get_task()
        # only TRAINING tasks complete via record counting; eval/predict
        # tasks are reported explicitly after processing
        if task.type == TaskType.TRAINING:
            with self._lock:
                self._pending_tasks.append([task, task.shard.size])
                self.current_task = task
        return task

    def report_batch_done(self, batch_size: Optional[int] = None) -> bool:
        """Consume records from the pending task queue front-to-back;
        report each task whose records are exhausted. Returns True if at
        least one task completed (reference: :111-148)."""
        remaining = batch_size or self._batch_size
        completed = False
        with self._lock:
            while remaining > 0 and self._pending_tasks:
                entry = self._pending_tasks[0]
                take = min(entry[1], remaining)
                entry[1] -= take
                remaining -= take
                if entry[1] == 0:
                    self._pending_tasks.popleft()
                    self._mc.report_task_result(entry[0].task_id)
                    completed = True
        return completed

    def report_task_failed(self, task_id: int, err: str) -> None:
        with self._lock:
            self._pending_tasks = deque(
                e for e in self._pending_tasks if e[0].task_id != task_id
            )
        self._mc.report_task_result(task_id, err_message=err or "failed")

    @property
    def pending_count(self) -> int:
        with self._lock:
            return len(self._pending_tasks)


class RecordIndexService:
    """Streams individual record indices from fetched shards — lets a
    map-style torch Dataset train elastically (reference: :161-212)."""

    def __init__(self, master_client, batch_size: int, maxsize: int = 1 << 16):
        self.shard_service = DataShardService(master_client, batch_size)
        self._queue: "queue.Queue[int]" = queue.Queue(maxsize=maxsize)
        self._stopped = threading.Event()
        self._thread = threading.Thread(
            target=self._fill, name="record-index", daemon=True
        )

    def start(self) -> "RecordIndexService":
        self._thread.start()
        return self

    def _fill(self) -> None:
        import time

        while not self._stopped.is_set():
            task = self.shard_service.fetch_task()
            if task.type == TaskType.WAIT:
                time.sleep(2)
                continue
            if task.type != TaskType.TRAINING:
                break
            indices = task.shard.indices or range(task.shard.start, task.shard.end)
            for i in indices:
                self._queue.put(i)

    def next_index(self, timeout: float = 60.0) -> Optional[int]:
        try:
            return self._queue.get(timeout=timeout)
        except queue.Empty:
            return None

    def report_batch_done(self, n: Optional[int] = None) -> None:
        self.shard_service.report_batch_done(n)

    def stop(self) -> None:
        self._stopped.set()
