"""Background engine loop for concurrent serving (continuous batching).

The reference handles HTTP concurrency in gin only (SURVEY.md §2b
"Continuous batching / concurrent request scheduler" is vacant). Here a
single loop thread owns the LLMEngine (which is not thread-safe); callers
submit prompts from any thread and get a Future. Requests that arrive while
a batch is decoding join the running batch at the next engine step — one
hipGraph replay serves the whole batch.
"""

from __future__ import annotations

import queue
import threading
from concurrent.futures import Future
from typing import Dict, List, Optional

from opsagent_amd.engine.engine import LLMEngine, SamplingParams
from opsagent_amd.utils.logging import get_logger

log = get_logger("serving")


class EngineLoop:
    """Owns the engine thread; also the engine watchdog (SURVEY.md §5
    "failure detection ... engine watchdog for hung kernels/collectives"):
    a monitor thread flags the loop unhealthy when one engine step exceeds
    `watchdog_s` (a hung kernel or collective blocks the step forever —
    /api/health then reports engine: "unhealthy" so orchestration can
    restart the pod)."""

    def __init__(self, engine: LLMEngine, watchdog_s: float = 120.0):
        self.engine = engine
        self.watchdog_s = watchdog_s
        self.healthy = True
        self.last_step_ms = 0.0
        self._step_started: float = 0.0
        self._submit_q: "queue.Queue[Tuple[List[int], SamplingParams, Future]]" = queue.Queue()
        self._futures: Dict[int, Future] = {}
        self._wake = threading.Event()
        self._stop = False
        # TP>1 followers never accept requests: rank 0's EngineLoop drives
        # them through the engine's step broadcast (engine.follower_loop
        # is their run loop — see cli serve / bench). No threads here.
        self.is_follower = engine.request_bcast and engine.tp_rank != 0
        if self.is_follower:
            return
        self._thread = threading.Thread(target=self._run, name="engine-loop", daemon=True)
        self._thread.start()
        self._watchdog = threading.Thread(target=self._watch, name="engine-watchdog", daemon=True)
        self._watchdog.start()

    def _watch(self) -> None:
        import time as _t

        while not self._stop:
            started = self._step_started
            if started and (_t.monotonic() - started) > self.watchdog_s:
                if self.healthy:
                    log.error(
                        "engine step stuck for >%.0fs — marking engine unhealthy "
                        "(hung kernel or collective)",
                        self.watchdog_s,
                    )
                self.healthy = False
            _t.sleep(min(5.0, self.watchdog_s / 4))

    def submit(self, prompt_ids: List[int], params: Optional[SamplingParams] = None) -> Future:
        """Thread-safe. Future resolves to (output_ids, finish_reason)."""
        if self.is_follower:
            raise RuntimeError(
                "TP follower ranks do not accept requests — submit to rank 0"
            )
        fut: Future = Future()
        self._submit_q.put((list(prompt_ids), params or SamplingParams(), fut, None))
        self._wake.set()
        return fut

    def generate(self, prompt_ids: List[int], params: Optional[SamplingParams] = None):
        return self.submit(prompt_ids, params).result()

    def submit_stream(self, prompt_ids: List[int], params: Optional[SamplingParams] = None):
        """Returns (token_iterator, future). Tokens arrive as sampled; the
        future resolves to (output_ids, finish_reason) at completion."""
        tq: "queue.Queue" = queue.Queue()
        fut: Future = Future()
        self._submit_q.put((list(prompt_ids), params or SamplingParams(), fut, tq))
        self._wake.set()

        def _iter():
            while True:
                t = tq.get()
                if t is None:
                    return
                yield t

        return _iter(), fut

    def shutdown(self) -> None:
        self._stop = True
        if self.is_follower:
            return
        self._wake.set()
        self._thread.join(timeout=30)
        try:
            self.engine.shutdown_followers()
        except Exception:  # noqa: BLE001 — best effort; followers may be gone
            pass

    # -- loop thread -----------------------------------------------------
    def _drain_submissions(self) -> None:
        while True:
            try:
                ids, params, fut, tq = self._submit_q.get_nowait()
            except queue.Empty:
                return
            try:
                rid = self.engine.add_request(ids, params)
                if tq is not None:
                    self.engine.requests[rid].stream_queue = tq
                self._futures[rid] = fut
            except Exception as e:  # noqa: BLE001 — admission failure resolves the future
                fut.set_exception(e)
                if tq is not None:
                    tq.put(None)

    def _run(self) -> None:
        eng = self.engine
        while not self._stop:
            self._drain_submissions()
            if not eng.running and not eng.waiting:
                self._wake.wait(timeout=0.05)
                self._wake.clear()
                continue
            import time as _t

            self._step_started = _t.monotonic()
            try:
                eng.step()
                self.last_step_ms = (_t.monotonic() - self._step_started) * 1000.0
                self._step_started = 0.0
                self.healthy = True
            except Exception as e:  # noqa: BLE001 — engine fault fails all in-flight requests
                self._step_started = 0.0
                log.exception("engine step failed")
                if eng.request_bcast:
                    # followers mirror rank 0's state; a rank-0 fault would
                    # desynchronize them — stop the TP group (best effort;
                    # a fault inside a collective leaves them to the
                    # watchdog) and stay unhealthy for /api/health
                    self.healthy = False
                    try:
                        eng.shutdown_followers()
                    except Exception:  # noqa: BLE001
                        pass
                for rid, fut in list(self._futures.items()):
                    if not fut.done():
                        fut.set_exception(e)
                    req = eng.requests.pop(rid, None)
                    if req is not None and req.stream_queue is not None:
                        req.stream_queue.put(None)  # unblock stream consumers
                    self._futures.pop(rid, None)
                # release the failed requests' KV blocks — without this a
                # recurring fault leaks the cache dry
                for r in eng.running:
                    if r.seq is not None:
                        try:
                            r.seq.free()
                        except Exception:  # noqa: BLE001 — best effort
                            pass
                eng.running.clear()
                eng.waiting.clear()
                continue
            # resolve finished requests
            done = [rid for rid, r in eng.requests.items() if r.finished]
            for rid in done:
                req = eng.requests.pop(rid)
                fut = self._futures.pop(rid, None)
                if fut is not None and not fut.done():
                    # side-channels for OpenAI logprobs + truncation notice
                    # (result stays a 2-tuple for every existing consumer)
                    fut.logprob_content = req.logprob_content
                    fut.truncated_prompt_tokens = req.truncated_prompt_tokens
                    fut.set_result((req.output_ids, req.finish_reason))
