"""Dynamic sharding semantics (mirrors reference task_manager_test.py)."""

from elasticdl_amd.common.task import TaskType
from elasticdl_amd.master.task_manager import TaskManager


def make_tm(**kw):
    defaults = dict(
        training_shards=[("f0", 0, 100), ("f1", 0, 50)],
        records_per_task=30,
        num_epochs=1,
    )
    defaults.update(kw)
    return TaskManager(**defaults)


def test_shard_math():
    tm = make_tm()
    c = tm.counts()
    # f0: 0-30,30-60,60-90,90-100 ; f1: 0-30,30-50
    assert c["todo"] == 6
    assert tm.total_records == 150


def test_get_report_cycle():
    tm = make_tm()
    seen = 0
    while True:
        t = tm.get(worker_id=0)
        if t.type != TaskType.TRAINING:
            break
        seen += t.shard.size
        tm.report(t.task_id, success=True, worker_id=0)
    assert seen == 150
    assert tm.finished()
    assert tm.completed_steps == 6


def test_failed_task_requeued_max3():
    tm = TaskManager(training_shards=[("f", 0, 10)], records_per_task=10)
    t = tm.get(0)
    for _ in range(3):
        tm.report(t.task_id, success=False, worker_id=0)
        t2 = tm.get(0)
        assert t2.task_id == t.task_id
        t = t2
    tm.report(t.task_id, success=False, worker_id=0)
    t = tm.get(0)
    assert t.type == TaskType.NONE
    assert tm.failed_records == 10


def test_recover_tasks_from_dead_worker():
    tm = make_tm()
    t1 = tm.get(1)
    t2 = tm.get(1)
    t3 = tm.get(2)
    assert tm.counts()["doing"] == 3
    n = tm.recover_tasks(1)
    assert n == 2
    assert tm.counts()["doing"] == 1
    # recovered tasks dispatchable again
    ids = set()
    while True:
        t = tm.get(3)
        if t.type != TaskType.TRAINING:
            break
        ids.add(t.task_id)
    assert t1.task_id in ids and t2.task_id in ids and t3.task_id not in ids


def test_multiple_epochs():
    tm = TaskManager(
        training_shards=[("f", 0, 20)], records_per_task=10, num_epochs=3
    )
    count = 0
    while True:
        t = tm.get(0)
        if t.type != TaskType.TRAINING:
            break
        count += 1
        tm.report(t.task_id, success=True, worker_id=0)
    assert count == 6  # 2 tasks x 3 epochs


def test_max_step_stop():
    tm = TaskManager(
        training_shards=[("f", 0, 100)], records_per_task=10, max_step=3
    )
    for _ in range(3):
        t = tm.get(0)
        tm.report(t.task_id, success=True, worker_id=0)
    t = tm.get(0)
    assert t.type == TaskType.NONE
    assert tm.finished()


def test_wait_while_tasks_in_flight():
    tm = TaskManager(training_shards=[("f", 0, 10)], records_per_task=10)
    t = tm.get(0)
    w = tm.get(1)
    assert w.type == TaskType.WAIT
    tm.report(t.task_id, success=True, worker_id=0)
    assert tm.get(1).type == TaskType.NONE


def test_shuffle_records():
    tm = TaskManager(
        training_shards=[("f", 0, 100)], records_per_task=25, shuffle=True
    )
    t = tm.get(0)
    assert t.shard.indices is not None
    assert len(t.shard.indices) == 25
    all_idx = list(t.shard.indices)
    while True:
        tm.report(t.task_id, True, 0)
        t = tm.get(0)
        if t.type != TaskType.TRAINING:
            break
        all_idx.extend(t.shard.indices)
    assert sorted(all_idx) == list(range(100))


def test_train_end_callback_task():
    tm = TaskManager(training_shards=[("f", 0, 10)], records_per_task=10)
    tm.enable_train_end_callback()
    t = tm.get(0)
    tm.report(t.task_id, True, 0)
    t = tm.get(0)
    assert t.type == TaskType.TRAIN_END_CALLBACK
    assert not tm.finished()
    tm.report(t.task_id, True, 0)
    assert tm.finished()


def test_evaluation_tasks_interleave():
    tm = make_tm(evaluation_shards=[("ev", 0, 20)])
    tm.create_evaluation_tasks(model_version=5)
    t = tm.get(0)
    assert t.type == TaskType.EVALUATION
    assert t.model_version == 5


def test_timeout_reassignment():
    tm = TaskManager(
        training_shards=[("f", 0, 10)],
        records_per_task=10,
        task_timeout_sec=0.01,
    )
    hung = []
    tm.register_task_timeout_callback(hung.append)
    t = tm.get(7)
    import time

    time.sleep(0.05)
    tm._reassign_timeout_tasks()
    assert hung == [7]
    t2 = tm.get(8)
    assert t2.task_id == t.task_id


def test_worker_driven_params():
    tm = TaskManager()
    tm.set_training_params(
        dataset_size=100, batch_size=10, num_epochs=2, num_minibatches_per_shard=5
    )
    c = tm.counts()
    assert c["todo"] == 2  # 100 / (10*5)


def test_strict_mode_disables_recovery_and_requeue():
    """--task_fault_tolerance false (reference task_manager.py:126,
    390, 467, 546): no dead-worker recovery, failed tasks drop on first
    failure, no watchdog thread."""
    from elasticdl_amd.common.task import TaskType

    tm = TaskManager(training_shards=[("s", 0, 64)], records_per_task=32,
                     task_fault_tolerance=False)
    tm.start()
    assert tm._watchdog is None
    t1 = tm.get(0)
    assert t1.type == TaskType.TRAINING
    assert tm.recover_tasks(0) == 0          # strict: no recovery
    tm.report(t1.task_id, False, 0)          # strict: dropped, not requeued
    t2 = tm.get(1)
    assert t2.task_id != t1.task_id
    tm.report(t2.task_id, True, 1)
    assert tm.get(1).type == TaskType.NONE
    assert tm.failed_records == 32
