"""AsyncLLMEngine: asyncio front-end over the synchronous step loop.

The engine loop runs in a dedicated thread (GPU-synchronous steps must not
block the server's event loop); outputs stream back through per-request
asyncio queues. Replaces the reference's vLLM AsyncLLM client
(vllm_model.py:83-228).
"""

from __future__ import annotations

import asyncio
import queue
import threading
import uuid
from typing import AsyncIterator, Dict, List, Optional, Union

from kserve_amd.engine.config import EngineConfig
from kserve_amd.engine.engine import LLMEngine
from kserve_amd.engine.request import RequestOutput
from kserve_amd.engine.sampling_params import SamplingParams
from kserve_amd.errors import EngineDead
from kserve_amd.logging import logger

_STOP = object()


def _publish_batch(items):
    """Runs on the server's event loop: deliver one step's outputs."""
    for q, out in items:
        q.put_nowait(out)


class AsyncLLMEngine:
    def __init__(self, config: EngineConfig, tokenizer=None, lora_modules=None):
        self.config = config
        self.tokenizer = tokenizer
        self.lora_modules = dict(lora_modules or {})  # name -> adapter path
        self.engine: Optional[LLMEngine] = None
        self._submit_q: "queue.Queue" = queue.Queue()
        self._streams: Dict[str, asyncio.Queue] = {}
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._dead: Optional[BaseException] = None
        self._started = threading.Event()
        self._stopping = False

    # -- lifecycle ----------------------------------------------------------
    async def start(self):
        """Build the engine (heavy: weights + KV + graphs) off-loop, then
        start the step thread."""
        self._loop = asyncio.get_running_loop()
        await self._loop.run_in_executor(None, self._build)
        self._thread = threading.Thread(
            target=self._run_loop, name="llm-engine-loop", daemon=True
        )
        self._thread.start()
        self._started.set()

    def _build(self):
        self.engine = LLMEngine(self.config, tokenizer=self.tokenizer)
        for name, path in self.lora_modules.items():
            self.engine.register_lora(name, path)

    def stop(self):
        self._stopping = True
        self._submit_q.put(_STOP)
        if self._thread is not None:
            self._thread.join(timeout=10)

    @property
    def is_running(self) -> bool:
        return (
            self._thread is not None and self._thread.is_alive() and self._dead is None
        )

    def _check_health(self):
        if self._dead is not None:
            raise EngineDead(str(self._dead))
        if self.engine is None:
            raise EngineDead("engine not started")

    # -- engine thread ---------------------------------------------------------
    def _run_loop(self):
        import os

        if os.environ.get("KS_ENGINE_PROFILE") == "1":
            import cProfile

            prof = cProfile.Profile()
            try:
                prof.runcall(self._run_loop_inner)
            finally:
                import pstats
                import sys

                print("==== engine thread profile ====", file=sys.stderr)
                pstats.Stats(prof, stream=sys.stderr).sort_stats(
                    "cumulative"
                ).print_stats(30)
            return
        self._run_loop_inner()

    def _run_loop_inner(self):
        try:
            while not self._stopping:
                # drain submissions; block briefly when idle
                block = not self.engine.has_unfinished()
                try:
                    while True:
                        item = self._submit_q.get(block=block, timeout=0.02)
                        if item is _STOP:
                            return
                        rid, prompt, sp = item
                        if prompt is None:  # abort sentinel
                            self.engine.abort_request(rid)
                            # if anyone is still consuming this stream,
                            # terminate it instead of leaving it hanging
                            entry = self._streams.pop(rid, None)
                            if entry is not None:
                                from kserve_amd.engine.request import (
                                    RequestOutput,
                                )

                                q, loop = entry
                                loop.call_soon_threadsafe(
                                    q.put_nowait,
                                    RequestOutput(
                                        request_id=rid,
                                        new_token_ids=[],
                                        finished=True,
                                        finish_reason="abort",
                                        output_token_ids=[],
                                        num_prompt_tokens=0,
                                    ),
                                )
                        else:
                            try:
                                self.engine.add_request(
                                    prompt, sp, request_id=rid
                                )
                            except ValueError as e:
                                # invalid request (e.g. over-long prompt):
                                # surface to ITS stream, keep the loop alive
                                entry = self._streams.pop(rid, None)
                                if entry is not None:
                                    q, loop = entry
                                    loop.call_soon_threadsafe(q.put_nowait, e)
                        block = False
                except queue.Empty:
                    pass
                if self.engine.has_unfinished():
                    outputs = self.engine.step()
                    # ONE cross-thread callback per event loop per step (a
                    # per-output call_soon_threadsafe wakes the server loop
                    # hundreds of times per step at high concurrency)
                    by_loop: Dict[object, list] = {}
                    for out in outputs:
                        entry = self._streams.get(out.request_id)
                        if entry is None:
                            continue
                        q, loop = entry
                        by_loop.setdefault(loop, []).append((q, out))
                        if out.finished:
                            self._streams.pop(out.request_id, None)
                    for loop, items in by_loop.items():
                        loop.call_soon_threadsafe(_publish_batch, items)
        except BaseException as e:  # engine loop must fail loudly
            logger.exception("Engine loop died")
            self._dead = e
            for q, loop in list(self._streams.values()):
                loop.call_soon_threadsafe(q.put_nowait, e)
            self._streams.clear()

    # -- request API -------------------------------------------------------------
    async def generate(
        self,
        prompt: Union[str, List[int]],
        sampling_params: Optional[SamplingParams] = None,
        request_id: Optional[str] = None,
    ) -> AsyncIterator[RequestOutput]:
        """Async stream of per-token RequestOutputs."""
        self._check_health()
        rid = request_id or str(uuid.uuid4())
        q: asyncio.Queue = asyncio.Queue()
        # bind the stream to the caller's loop: the engine thread publishes
        # via call_soon_threadsafe on THIS loop (server may run several)
        self._streams[rid] = (q, asyncio.get_running_loop())
        self._submit_q.put((rid, prompt, sampling_params or SamplingParams()))
        try:
            while True:
                item = await q.get()
                if isinstance(item, ValueError):
                    raise item  # per-request validation error
                if isinstance(item, BaseException):
                    raise EngineDead(str(item))
                yield item
                if item.finished:
                    return
        finally:
            if rid in self._streams:
                # client disconnected mid-stream: abort in engine thread
                self._streams.pop(rid, None)
                self._submit_abort(rid)

    async def abort(self, rid: str) -> None:
        """Abort a request by id; any consumer of its stream receives a
        final finished output with reason "abort"."""
        self._submit_abort(rid)

    def _submit_abort(self, rid: str):
        if self.engine is not None:
            # abort is thread-safe enough: scheduler mutation happens in the
            # engine thread via a sentinel
            self._submit_q.put((rid, None, None))

    async def generate_full(
        self,
        prompt: Union[str, List[int]],
        sampling_params: Optional[SamplingParams] = None,
        request_id: Optional[str] = None,
    ) -> RequestOutput:
        last = None
        async for out in self.generate(prompt, sampling_params, request_id):
            last = out
        return last
