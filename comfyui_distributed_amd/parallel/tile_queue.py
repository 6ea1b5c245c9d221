"""Pull-queue tile scheduler over a shared KV store.

This is the intra-node replacement for the reference's HTTP pull scheduler
(`POST /distributed/request_image` popping ``pending_tasks`` with a 0.1 s
wait — api/usdu_routes.py:168-215) and its heartbeat/timeout/requeue state
machine (upscale/job_timeout.py:17-150). The store is a torch.distributed
TCPStore in production (atomic ``add`` = the pull counter; microseconds per
pop instead of an HTTP round trip) or an in-memory LocalStore in tests.

State per job (namespaced by job id):
  total            number of tasks
  next             atomic pull cursor (store.add)
  requeue          JSON list of task ids pushed back by the monitor
  alist:<rank>     JSON list of task ids rank pulled (written only by rank)
  dlist:<rank>     JSON list of task ids rank completed
  done:<idx>       completion marker (idempotence across requeue races)
  completed        atomic completion counter
  hb:<rank>        wall-clock heartbeat stamp
"""

from __future__ import annotations

import json
import threading
import time
from typing import Protocol

from ..utils import constants
from ..utils.logging import debug_log, log


class KVStore(Protocol):
    def set(self, key: str, value: str): ...
    def get(self, key: str) -> bytes: ...
    def add(self, key: str, amount: int) -> int: ...
    def compare_set(self, key: str, expected: str, desired: str) -> bytes: ...


class LocalStore:
    """In-memory KVStore with TCPStore semantics (for tests / 1-process)."""

    def __init__(self):
        self._d: dict[str, bytes] = {}
        self._lock = threading.Lock()

    def set(self, key, value):
        with self._lock:
            self._d[key] = value.encode() if isinstance(value, str) else bytes(value)

    def get(self, key):
        with self._lock:
            return self._d.get(key, b"")

    def add(self, key, amount):
        with self._lock:
            cur = int(self._d.get(key, b"0") or b"0")
            cur += amount
            self._d[key] = str(cur).encode()
            return cur

    def compare_set(self, key, expected, desired):
        exp = expected.encode() if isinstance(expected, str) else expected
        des = desired.encode() if isinstance(desired, str) else desired
        with self._lock:
            cur = self._d.get(key)
            if cur is None:
                if exp == b"":
                    self._d[key] = des
                    return des
                return b""
            if cur == exp:
                self._d[key] = des
                return des
            return cur


class TileQueue:
    """One job's pull queue, usable from any participant."""

    def __init__(self, store: KVStore, job_id: str, rank: int):
        self.store = store
        self.job_id = job_id
        self.rank = rank

    def _k(self, name: str) -> str:
        return f"tq:{self.job_id}:{name}"

    def _get(self, key: str) -> bytes:
        """Non-blocking read that works on a real TCPStore (whose ``get``
        blocks on missing keys): compare_set with empty expected+desired
        returns the current value, or b"" after creating the empty key."""
        return self.store.compare_set(key, "", "")

    # -- master-side init --------------------------------------------------

    def init_job(self, n_tasks: int) -> None:
        self.store.set(self._k("total"), str(n_tasks))
        self.store.set(self._k("next"), "0")
        self.store.set(self._k("requeue"), "[]")
        self.store.set(self._k("completed"), "0")
        self.store.set(self._k("ready"), "1")

    def is_ready(self) -> bool:
        return self._get(self._k("ready")) == b"1"

    def total(self) -> int:
        raw = self._get(self._k("total"))
        return int(raw) if raw else 0

    # -- worker side -------------------------------------------------------

    def pop(self) -> int | None:
        """Next task id, or None when the queue is exhausted. Requeued tasks
        take priority (reference job_store drains requeue first)."""
        idx = self._pop_requeue()
        if idx is None:
            total = self.total()
            cand = self.store.add(self._k("next"), 1) - 1
            if cand >= total:
                idx = self._pop_requeue()  # late requeues
                if idx is None:
                    return None
            else:
                idx = cand
        self._append_own(f"alist:{self.rank}", idx)
        return idx

    def _append_own(self, name: str, idx: int) -> None:
        """Append to a per-rank JSON list. Only THIS rank ever writes its
        own list (sequentially), so a plain read-modify-write is safe — no
        CAS loop, and the monitor's assigned_incomplete becomes two store
        reads instead of one per task in the job (round-1 O(total) scan)."""
        key = self._k(name)
        cur = self._get(key)
        lst = json.loads(cur) if cur else []
        lst.append(int(idx))
        self.store.set(key, json.dumps(lst))

    def _pop_requeue(self) -> int | None:
        key = self._k("requeue")
        while True:
            cur = self._get(key)
            if not cur:
                return None
            lst = json.loads(cur)
            if not lst:
                return None
            head, rest = lst[0], lst[1:]
            if self.store.compare_set(key, cur.decode(), json.dumps(rest)) == json.dumps(rest).encode():
                return int(head)

    def mark_done(self, idx: int) -> int:
        """Mark a task complete; returns the completion count. Idempotent
        across requeue races (only the first completion counts)."""
        self._append_own(f"dlist:{self.rank}", idx)
        n = self.store.add(self._k(f"done:{idx}"), 1)
        if n == 1:
            return self.store.add(self._k("completed"), 1)
        return self.completed()

    def completed(self) -> int:
        return int(self._get(self._k("completed")) or 0)

    def is_complete(self) -> bool:
        return self.completed() >= self.total()

    def heartbeat(self, now: float | None = None) -> None:
        self.store.set(self._k(f"hb:{self.rank}"), repr(now if now is not None else time.time()))

    # -- monitor (master) --------------------------------------------------

    def requeue_tasks(self, indices: list[int]) -> None:
        if not indices:
            return
        key = self._k("requeue")
        while True:
            cur = self._get(key) or b"[]"
            lst = json.loads(cur)
            lst.extend(int(i) for i in indices)
            desired = json.dumps(lst)
            if self.store.compare_set(key, cur.decode(), desired) == desired.encode():
                return

    def heartbeat_age(self, rank: int, now: float | None = None) -> float | None:
        raw = self._get(self._k(f"hb:{rank}"))
        if not raw:
            return None
        now = now if now is not None else time.time()
        return now - float(raw)

    def assigned_incomplete(self, rank: int) -> list[int]:
        """Task ids ``rank`` pulled but never completed. Two store reads
        plus one done-check per *candidate* (a requeued task the rank lost
        may have been finished elsewhere — skip those)."""
        raw_a = self._get(self._k(f"alist:{rank}"))
        raw_d = self._get(self._k(f"dlist:{rank}"))
        assigned = json.loads(raw_a) if raw_a else []
        done = set(json.loads(raw_d)) if raw_d else set()
        out = []
        for idx in assigned:
            if idx in done:
                continue
            # add(key, 0) is the non-creating-as-"" existence probe: _get's
            # compare_set would create the key as an EMPTY string, which a
            # later mark_done add() rejects server-side (stoll) and the
            # TCPStore then drops the connection
            if self.store.add(self._k(f"done:{idx}"), 0) > 0:
                continue
            out.append(idx)
        return out


class TileScheduler:
    """Master-side monitor reproducing the reference fault-tolerance state
    machine (upscale/job_timeout.py): heartbeat age > timeout -> probe the
    worker -> grace when it reports active work -> otherwise requeue its
    incomplete tasks and drop it."""

    def __init__(
        self,
        queue: TileQueue,
        worker_ranks: list[int],
        timeout: float | None = None,
        probe=None,
        clock=time.time,
    ):
        self.queue = queue
        self.active = set(worker_ranks)
        self.dropped: set[int] = set()
        self.timeout = timeout if timeout is not None else constants.ACTOR_HEARTBEAT_TIMEOUT
        self.probe = probe  # callable rank -> bool ("is the worker busy/alive")
        self.clock = clock
        #: job-start grace origin: a rank that NEVER heartbeats (crashed
        #: before its first chunk) is aged from here instead of being
        #: skipped forever (round-1 advisor finding)
        self._started = clock()

    def force_drop(self, rank: int) -> list[int]:
        """Immediately requeue a rank's incomplete work and drop it (used
        when its result transfer failed — no probe/grace)."""
        if rank not in self.active:
            return []
        tasks = self.queue.assigned_incomplete(rank)
        if tasks:
            self.queue.requeue_tasks(tasks)
        self.active.discard(rank)
        self.dropped.add(rank)
        log(f"scheduler: rank {rank} force-dropped — requeued {len(tasks)} tasks")
        return tasks

    def check_and_requeue(self) -> list[int]:
        """Run one monitor pass; returns task ids requeued this pass."""
        now = self.clock()
        requeued: list[int] = []
        for rank in sorted(self.active):
            age = self.queue.heartbeat_age(rank, now)
            if age is None:
                age = now - self._started
            if age <= self.timeout:
                continue
            # probe outside any lock (reference job_timeout.py:53-56)
            if self.probe is not None and self.probe(rank):
                # grace: refresh the heartbeat, keep the worker
                self.queue.store.set(
                    self.queue._k(f"hb:{rank}"), repr(now)
                )
                debug_log(f"scheduler: rank {rank} slow but busy — grace")
                continue
            tasks = self.queue.assigned_incomplete(rank)
            if tasks:
                self.queue.requeue_tasks(tasks)
                requeued.extend(tasks)
            self.active.discard(rank)
            self.dropped.add(rank)
            log(f"scheduler: rank {rank} timed out — requeued {len(tasks)} tasks")
        return requeued

    def no_active_workers(self) -> bool:
        return not self.active
