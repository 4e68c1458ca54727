"""Background task queue: the Celery/Redis analog.

SQLite-persisted tasks + an asyncio worker pool in-process.  Keeps the
reference's reliability semantics (reference: celery_config.py:66-148,
chat/background/task.py:439-499): dedup keys (SETNX analog via a UNIQUE
column), acks-late (status flips to done only after the handler returns),
bounded retries with backoff, per-task time limits, delayed dispatch
(countdown), and beat schedules.
"""
from __future__ import annotations

import asyncio
import json
import sqlite3
import time
import traceback
import uuid
from dataclasses import dataclass
from typing import Any, Awaitable, Callable, Dict, List, Optional

from .store import Database


@dataclass
class Beat:
    name: str
    interval_s: float
    args: dict
    next_run: float = 0.0


class TaskQueue:
    def __init__(self, db: Database, concurrency: int = 4,
                 default_time_limit_s: float = 1800.0):
        self.db = db
        self.concurrency = concurrency
        self.default_time_limit = default_time_limit_s
        self.handlers: Dict[str, Callable[[dict], Awaitable[Any]]] = {}
        self.beats: List[Beat] = []
        self._workers: List[asyncio.Task] = []
        self._beat_task: Optional[asyncio.Task] = None
        self._stop = asyncio.Event()
        self._wake = asyncio.Event()

    # ---- registration ----
    def task(self, name: str):
        def deco(fn):
            self.handlers[name] = fn
            return fn
        return deco

    def register(self, name: str, fn: Callable[[dict], Awaitable[Any]]) -> None:
        self.handlers[name] = fn

    def add_beat(self, name: str, interval_s: float, args: Optional[dict] = None) -> None:
        self.beats.append(Beat(name, interval_s, args or {}))

    # ---- dispatch ----
    def enqueue(self, name: str, args: dict, countdown_s: float = 0.0,
                dedup_key: Optional[str] = None, max_attempts: int = 3,
                org_id: Optional[str] = None) -> Optional[str]:
        tid = f"task-{uuid.uuid4().hex[:16]}"
        t = time.time()
        try:
            self.db.execute(
                "INSERT INTO task_queue (id,name,org_id,args,status,dedup_key,eta,"
                "max_attempts,created_at,updated_at) VALUES (?,?,?,?,?,?,?,?,?,?)",
                [tid, name, org_id, json.dumps(args), "pending", dedup_key,
                 t + countdown_s, max_attempts, t, t])
        except sqlite3.IntegrityError:
            return None  # dedup hit (UNIQUE dedup_key) — reference task.py:490
            # (other DB errors must RAISE: swallowing them would silently
            # drop the task)
        self._wake.set()
        return tid

    def get_task(self, task_id: str) -> Optional[dict]:
        rows = self.db.query("SELECT * FROM task_queue WHERE id=?", [task_id])
        return dict(rows[0]) if rows else None

    def cancel(self, task_id: str) -> bool:
        cur = self.db.execute(
            "UPDATE task_queue SET status='cancelled', updated_at=? "
            "WHERE id=? AND status='pending'", [time.time(), task_id])
        return cur.rowcount > 0

    def pending_count(self) -> int:
        return self.db.query(
            "SELECT COUNT(*) c FROM task_queue WHERE status IN ('pending','running')"
        )[0]["c"]

    # ---- execution ----
    def _claim(self) -> Optional[dict]:
        now = time.time()
        rows = self.db.query(
            "SELECT * FROM task_queue WHERE status='pending' AND eta<=? "
            "ORDER BY eta LIMIT 1", [now])
        if not rows:
            return None
        row = dict(rows[0])
        cur = self.db.execute(
            "UPDATE task_queue SET status='running', attempts=attempts+1, "
            "updated_at=? WHERE id=? AND status='pending'", [now, row["id"]])
        if cur.rowcount == 0:
            return None  # lost the race to another worker
        row["attempts"] += 1
        return row

    async def _run_one(self, row: dict) -> None:
        name = row["name"]
        fn = self.handlers.get(name)
        t = time.time()
        if fn is None:
            self.db.execute("UPDATE task_queue SET status='failed', result=?, "
                            "updated_at=? WHERE id=?",
                            [f"no handler for {name}", t, row["id"]])
            return
        args = json.loads(row["args"] or "{}")
        # handlers that re-raise transient faults for retry can see whether
        # this is the LAST attempt and fail terminally instead of stranding
        # their domain state (e.g. an incident left 'pending' forever)
        args["_task_meta"] = {"attempt": row["attempts"],
                              "max_attempts": row["max_attempts"]}
        try:
            result = await asyncio.wait_for(
                fn(args), timeout=self.default_time_limit)
            self.db.execute(
                "UPDATE task_queue SET status='done', result=?, updated_at=? WHERE id=?",
                [json.dumps(result)[:4000] if result is not None else None,
                 time.time(), row["id"]])
        except Exception as e:  # noqa: BLE001
            err = f"{type(e).__name__}: {e}\n{traceback.format_exc()[-1500:]}"
            if row["attempts"] < row["max_attempts"]:
                backoff = min(60.0, 2.0 ** row["attempts"])
                self.db.execute(
                    "UPDATE task_queue SET status='pending', eta=?, result=?, "
                    "updated_at=? WHERE id=?",
                    [time.time() + backoff, err[:2000], time.time(), row["id"]])
            else:
                self.db.execute(
                    "UPDATE task_queue SET status='failed', result=?, updated_at=? "
                    "WHERE id=?", [err[:2000], time.time(), row["id"]])

    async def _worker(self) -> None:
        while not self._stop.is_set():
            row = self._claim()
            if row is None:
                try:
                    await asyncio.wait_for(self._wake.wait(), timeout=0.25)
                except asyncio.TimeoutError:
                    pass
                self._wake.clear()
                continue
            await self._run_one(row)

    async def _beat_loop(self) -> None:
        while not self._stop.is_set():
            now = time.time()
            for b in self.beats:
                if now >= b.next_run:
                    b.next_run = now + b.interval_s
                    self.enqueue(b.name, b.args,
                                 dedup_key=f"beat:{b.name}:{int(now / b.interval_s)}")
            await asyncio.sleep(0.5)

    def start(self) -> None:
        loop = asyncio.get_running_loop()
        # acks-late redelivery: rows left 'running' by a previous process
        # (crash mid-task) are requeued — the handler never acked, so it
        # must run again (Celery acks_late + visibility-timeout analog,
        # reference: celery_config.py task_acks_late)
        self.db.execute(
            "UPDATE task_queue SET status='pending' WHERE status='running'")
        self._stop.clear()
        self._workers = [loop.create_task(self._worker())
                         for _ in range(self.concurrency)]
        if self.beats:
            self._beat_task = loop.create_task(self._beat_loop())

    async def stop(self) -> None:
        self._stop.set()
        self._wake.set()
        for w in self._workers:
            w.cancel()
        if self._beat_task:
            self._beat_task.cancel()
        await asyncio.gather(*self._workers, return_exceptions=True)

    async def drain(self, timeout_s: float = 30.0) -> bool:
        t0 = time.time()
        while time.time() - t0 < timeout_s:
            if self.pending_count() == 0:
                return True
            await asyncio.sleep(0.05)
        return False
