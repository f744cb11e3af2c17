"""Async wrapper around LLMEngine for the HTTP server.

The engine loop runs in a dedicated thread (the GPU step loop must never
block the asyncio event loop); requests/aborts flow in through thread-safe
queues, streaming outputs flow out through per-request asyncio queues via
call_soon_threadsafe.
"""

from __future__ import annotations

import asyncio
import logging
import queue
import threading
import time
from typing import AsyncIterator, Dict, List, Optional, Union

from production_stack_amd.engine.engine import LLMEngine
from production_stack_amd.engine.sampling import SamplingParams
from production_stack_amd.engine.sequence import RequestOutput

logger = logging.getLogger("engine.async")


class AsyncEngine:
    def __init__(self, engine: LLMEngine) -> None:
        self.engine = engine
        self._pending: "queue.Queue" = queue.Queue()
        self._aborts: "queue.Queue" = queue.Queue()
        self._streams: Dict[str, asyncio.Queue] = {}
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------------
    def start(self, loop: Optional[asyncio.AbstractEventLoop] = None) -> None:
        self._loop = loop or asyncio.get_running_loop()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        self._wake.set()
        if self._thread:
            self._thread.join(timeout=5)

    def _drain_control(self) -> None:
        while True:
            try:
                (rid, prompt, params, arrival,
                 mm) = self._pending.get_nowait()
            except queue.Empty:
                break
            try:
                self.engine.add_request(rid, prompt, params, arrival,
                                        mm_embeds=mm)
            except ValueError as e:
                self._emit(
                    RequestOutput(
                        request_id=rid,
                        new_token_ids=[],
                        text_delta="",
                        finished=True,
                        finish_reason=f"error: {e}",
                    )
                )
        while True:
            try:
                rid = self._aborts.get_nowait()
            except queue.Empty:
                break
            self.engine.abort_request(rid)
            self._emit(
                RequestOutput(
                    request_id=rid,
                    new_token_ids=[],
                    text_delta="",
                    finished=True,
                    finish_reason="abort",
                )
            )

    def _emit(self, out: RequestOutput) -> None:
        q = self._streams.get(out.request_id)
        if q is None or self._loop is None:
            return
        self._loop.call_soon_threadsafe(q.put_nowait, out)

    def _run(self) -> None:
        logger.info("engine loop thread started")
        while not self._stop.is_set():
            self._drain_control()
            if self.engine.is_sleeping or not self.engine.has_unfinished():
                self._wake.wait(timeout=0.005)
                self._wake.clear()
                continue
            try:
                for out in self.engine.step():
                    self._emit(out)
            except Exception:
                logger.exception("engine step failed")
                time.sleep(0.1)

    # ------------------------------------------------------------------
    async def generate(
        self,
        request_id: str,
        prompt: Union[str, List[int]],
        params: SamplingParams,
        mm_embeds=None,
    ) -> AsyncIterator[RequestOutput]:
        q: asyncio.Queue = asyncio.Queue()
        self._streams[request_id] = q
        self._pending.put(
            (request_id, prompt, params, time.time(), mm_embeds))
        self._wake.set()
        try:
            while True:
                out = await q.get()
                yield out
                if out.finished:
                    break
        finally:
            self._streams.pop(request_id, None)

    def abort(self, request_id: str) -> None:
        self._aborts.put(request_id)
        self._wake.set()
