"""Async front-end for the engine: one background stepping thread, many
concurrent awaitable generations.

This is the in-process replacement for the reference's provider HTTP
boundary *and* its Celery worker concurrency: every concurrent
investigation/chat turn is just another sequence in the same continuously
batched engine (SURVEY.md §2.12 row 'Concurrent investigations').

Delivery path: the stepping thread groups each step's outputs by consumer
event loop and posts ONE ``call_soon_threadsafe`` batch per loop per step
onto per-request ``asyncio.Queue``s. No executor threads are parked in
blocking reads, so hundreds of concurrent streams (the flagship B=256
investigation batch) cost one self-pipe write per step instead of
saturating the default 32-thread pool."""
from __future__ import annotations

import asyncio
import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import AsyncIterator, Dict, List, Optional, Tuple

from .engine import EngineConfig, LLMEngine, StepOutput
from .sampling_params import SamplingParams


@dataclass
class GenResult:
    token_ids: List[int] = field(default_factory=list)
    text: str = ""
    finish_reason: Optional[str] = None
    prompt_tokens: int = 0
    elapsed_ms: float = 0.0


class AsyncEngineServer:
    def __init__(self, config: EngineConfig, engine: Optional[LLMEngine] = None):
        self.engine = engine or LLMEngine(config)
        # rid -> (consumer loop, asyncio queue); mutated only under _lock
        self._subs: Dict[str, Tuple[asyncio.AbstractEventLoop, asyncio.Queue]] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._wake = threading.Event()
        self._last_progress = time.monotonic()
        self.watchdog_timeout_s = 300.0
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="engine-step-loop")
        self._thread.start()

    # ------------------------------------------------------------- lifecycle

    def _run(self) -> None:
        while not self._stop.is_set():
            with self._lock:
                busy = self.engine.has_work()
            if not busy:
                self._wake.wait(timeout=0.02)
                self._wake.clear()
                continue
            try:
                with self._lock:
                    outs = self.engine.step()
                self._last_progress = time.monotonic()
            except Exception as e:  # noqa: BLE001 - a step failure must not
                # kill the loop: fail every in-flight request loudly instead
                import traceback
                err = f"engine step failed: {e}"
                traceback.print_exc()
                with self._lock:
                    rids = list(self._subs)
                    for rid in rids:
                        try:
                            self.engine.abort_request(rid)
                        except Exception:  # noqa: BLE001
                            pass
                self._deliver([StepOutput(rid, [], "", True, err) for rid in rids])
                continue
            self._deliver(outs)
        # shutdown: fail any remaining consumers so awaiters wake up
        with self._lock:
            rids = list(self._subs)
        self._deliver([StepOutput(rid, [], "", True, "engine shutdown")
                       for rid in rids])

    def _deliver(self, outs: List[StepOutput]) -> None:
        """Post step outputs to their consumers, one loop wakeup per batch."""
        if not outs:
            return
        by_loop: Dict[asyncio.AbstractEventLoop, List[Tuple[asyncio.Queue, StepOutput]]] = {}
        for out in outs:
            sub = self._subs.get(out.request_id)
            if sub is not None:
                by_loop.setdefault(sub[0], []).append((sub[1], out))
        for loop, items in by_loop.items():
            try:
                loop.call_soon_threadsafe(_post_batch, items)
            except RuntimeError:
                # consumer loop already closed; its finally-block cleanup
                # (or abort) will free the sequences
                pass

    def shutdown(self) -> None:
        self._stop.set()
        self._wake.set()
        self._thread.join(timeout=5)

    # ------------------------------------------------------------------- API

    def tokenizer(self):
        return self.engine.tokenizer

    async def stream(self, prompt_token_ids: List[int], params: SamplingParams,
                     session_id: Optional[str] = None,
                     request_id: Optional[str] = None) -> AsyncIterator[StepOutput]:
        rid = request_id or f"req-{uuid.uuid4().hex[:12]}"
        if self._stop.is_set():
            yield StepOutput(rid, [], "", True, "engine shutdown")
            return
        loop = asyncio.get_running_loop()
        q: asyncio.Queue = asyncio.Queue()
        with self._lock:
            self._subs[rid] = (loop, q)
            try:
                self.engine.add_request(rid, prompt_token_ids, params,
                                        session_id=session_id)
            except BaseException:  # rejection at the API boundary (empty /
                # out-of-vocab prompt): drop the subscription we just made
                # or it leaks until shutdown
                self._subs.pop(rid, None)
                raise
        self._wake.set()
        finished = False
        try:
            while True:
                out = await q.get()
                yield out
                if out.finished:
                    finished = True
                    return
        finally:
            with self._lock:
                self._subs.pop(rid, None)
                if not finished:  # consumer bailed (cancel / WS drop)
                    self.engine.abort_request(rid)

    async def generate(self, prompt_token_ids: List[int], params: SamplingParams,
                       session_id: Optional[str] = None,
                       raise_on_fault: bool = True) -> GenResult:
        t0 = time.monotonic()
        res = GenResult(prompt_tokens=len(prompt_token_ids))
        async for out in self.stream(prompt_token_ids, params, session_id):
            res.token_ids.extend(out.new_token_ids)
            res.text += out.new_text
            if out.finished:
                res.finish_reason = out.finish_reason
        res.elapsed_ms = (time.monotonic() - t0) * 1000.0
        # engine faults must be EXCEPTIONS to await-ers (so LLMManager's
        # transient-fault retry can fire), not silent empty generations;
        # "aborted" stays a normal result (deliberate cancellation)
        fr = res.finish_reason or ""
        if raise_on_fault and (fr.startswith("engine step failed")
                               or fr == "engine shutdown"):
            raise RuntimeError(fr)
        return res

    async def complete_text(self, prompt: str, params: SamplingParams,
                            session_id: Optional[str] = None) -> GenResult:
        tok = self.engine.tokenizer
        ids = tok.encode(prompt, add_bos=True)
        return await self.generate(ids, params, session_id=session_id)

    def abort(self, request_id: str) -> None:
        with self._lock:
            self.engine.abort_request(request_id)
            sub = self._subs.pop(request_id, None)
        if sub is not None:
            loop, q = sub
            try:
                loop.call_soon_threadsafe(
                    q.put_nowait, StepOutput(request_id, [], "", True, "aborted"))
            except RuntimeError:
                pass

    def counters(self) -> Dict[str, int]:
        return dict(self.engine.counters)

    def healthy(self) -> bool:
        """Engine watchdog (SURVEY.md §5.3 'engine watchdog'): unhealthy when
        a step has been stuck (hung kernel / collective) past the timeout."""
        with self._lock:
            busy = self.engine.has_work()
        return (not busy) or (time.monotonic() - self._last_progress
                              < self.watchdog_timeout_s)


def _post_batch(items: List[Tuple[asyncio.Queue, StepOutput]]) -> None:
    for q, out in items:
        q.put_nowait(out)
