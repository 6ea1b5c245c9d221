"""Pull-queue + fault-tolerance state machine (single-process, LocalStore,
injected clock/probe — the reference tests its timeout machinery the same
way: back-dated heartbeats + fake probe_worker, tests/test_job_timeout.py)."""

from comfyui_distributed_amd.parallel.tile_queue import (
    LocalStore,
    TileQueue,
    TileScheduler,
)


def make_queue(rank=0, job="j1"):
    store = LocalStore()
    q = TileQueue(store, job, rank)
    return store, q


def test_pull_exhaustion():
    store, q = make_queue()
    q.init_job(5)
    got = []
    while (i := q.pop()) is not None:
        got.append(i)
    assert sorted(got) == [0, 1, 2, 3, 4]
    assert q.pop() is None


def test_two_participants_disjoint():
    store = LocalStore()
    q0 = TileQueue(store, "j", 0)
    q1 = TileQueue(store, "j", 1)
    q0.init_job(10)
    got0, got1 = [], []
    while True:
        a = q0.pop()
        if a is not None:
            got0.append(a)
        b = q1.pop()
        if b is not None:
            got1.append(b)
        if a is None and b is None:
            break
    assert sorted(got0 + got1) == list(range(10))
    assert not (set(got0) & set(got1))


def test_completion_tracking_idempotent():
    store, q = make_queue()
    q.init_job(3)
    for _ in range(3):
        q.pop()
    q.mark_done(0)
    q.mark_done(0)  # double completion (requeue race) counts once
    q.mark_done(1)
    assert q.completed() == 2
    assert not q.is_complete()
    q.mark_done(2)
    assert q.is_complete()


def test_requeue_priority():
    store, q = make_queue()
    q.init_job(3)
    assert q.pop() == 0
    q.requeue_tasks([0])
    # requeued task comes before the untouched cursor tasks
    assert q.pop() == 0
    assert q.pop() == 1


def test_scheduler_timeout_requeues_and_drops():
    store = LocalStore()
    master = TileQueue(store, "j", 0)
    worker = TileQueue(store, "j", 1)
    master.init_job(4)

    now = [1000.0]
    worker.heartbeat(now[0])
    # worker pulls 2 tasks, completes one
    a, b = worker.pop(), worker.pop()
    worker.mark_done(a)

    sched = TileScheduler(master, [1], timeout=60, probe=lambda r: False,
                          clock=lambda: now[0])
    assert sched.check_and_requeue() == []  # fresh heartbeat
    now[0] += 120.0
    requeued = sched.check_and_requeue()
    assert requeued == [b]
    assert sched.no_active_workers()
    assert 1 in sched.dropped
    # master takeover: the requeued task is pullable again
    assert master.pop() == b


def test_scheduler_probe_grace():
    store = LocalStore()
    master = TileQueue(store, "j", 0)
    worker = TileQueue(store, "j", 1)
    master.init_job(2)
    now = [0.0]
    worker.heartbeat(0.0)
    worker.pop()
    sched = TileScheduler(master, [1], timeout=10, probe=lambda r: True,
                          clock=lambda: now[0])
    now[0] = 100.0
    assert sched.check_and_requeue() == []  # busy probe -> grace
    assert not sched.no_active_workers()
    # grace refreshed the heartbeat: age is now ~0
    assert master.heartbeat_age(1, now[0]) == 0.0


def test_scheduler_no_heartbeat_worker_never_started():
    store = LocalStore()
    master = TileQueue(store, "j", 0)
    master.init_job(2)
    sched = TileScheduler(master, [1], timeout=10, probe=lambda r: False,
                          clock=lambda: 100.0)
    # no heartbeat recorded at all -> worker kept (may still be starting;
    # reference gives workers a job-ready grace window)
    assert sched.check_and_requeue() == []


def test_concurrent_pull_exactly_once():
    """8 threads hammer one queue: every task is delivered exactly once."""
    import threading

    store = LocalStore()
    master = TileQueue(store, "soak", 0)
    master.init_job(500)
    seen = []
    lock = threading.Lock()

    def puller(rank):
        q = TileQueue(store, "soak", rank)
        got = []
        while True:
            i = q.pop()
            if i is None:
                break
            got.append(i)
            q.mark_done(i)
        with lock:
            seen.extend(got)

    threads = [threading.Thread(target=puller, args=(r,)) for r in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert sorted(seen) == list(range(500))
    assert master.completed() == 500 and master.is_complete()


def test_concurrent_requeue_no_duplication():
    """Requeued tasks are also delivered exactly once under contention."""
    import threading

    store = LocalStore()
    master = TileQueue(store, "rq", 0)
    master.init_job(100)
    # pre-assign and requeue 20 tasks as if a worker died
    dead = [master.pop() for _ in range(20)]
    master.requeue_tasks(dead)
    seen = []
    lock = threading.Lock()

    def puller(rank):
        q = TileQueue(store, "rq", rank)
        got = []
        while True:
            i = q.pop()
            if i is None:
                break
            got.append(i)
        with lock:
            seen.extend(got)

    threads = [threading.Thread(target=puller, args=(r,)) for r in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert sorted(seen) == sorted(dead + [i for i in range(100) if i not in dead])


def test_assigned_incomplete_store_ops_bounded_at_1024_tiles():
    """Round-1 advisor: assigned_incomplete did one store round-trip per
    task in the JOB (1024 at 8K canvases). Now it is two list reads plus
    one done-probe per *candidate* — count the ops to pin the contract."""

    class CountingStore(LocalStore):
        def __init__(self):
            super().__init__()
            self.ops = 0

        def get(self, key):
            self.ops += 1
            return super().get(key)

        def add(self, key, amount):
            self.ops += 1
            return super().add(key, amount)

        def compare_set(self, key, expected, desired):
            self.ops += 1
            return super().compare_set(key, expected, desired)

        def set(self, key, value):
            self.ops += 1
            return super().set(key, value)

    store = CountingStore()
    master = TileQueue(store, "big", 0)
    master.init_job(1024)
    worker = TileQueue(store, "big", 1)
    pulled = [worker.pop() for _ in range(64)]
    for idx in pulled[:32]:
        worker.mark_done(idx)
    store.ops = 0
    incomplete = master.assigned_incomplete(1)
    assert sorted(incomplete) == sorted(pulled[32:])
    # 2 list reads + <= 1 done-probe per incomplete candidate (32) — far
    # from the old 2*1024 scan
    assert store.ops <= 2 + 40, store.ops
