"""Cross-request dynamic batching for the serving surface.

The engine batches *within* one ``generate`` call (continuous-batching
admission, n-candidate fan-out). For serving, concurrent requests should
share one decode wave too: each ``generate`` call refreshes the fused
decode weights and (on GPU) captures/replays a hipGraph sized to the
batch, so 16 one-prompt calls cost ~16x one 16-prompt call. The reference
has no serving surface at all (docs/ROADMAP.md #10); this is native.

One background thread owns the engine. ``submit`` blocks the calling
(FastAPI threadpool) thread until its slice of the merged batch is done.
Requests are grouped by identical (SamplingParams, eos) since the engine
takes one SamplingParams per call; different groups run back-to-back in
arrival order.
"""

from __future__ import annotations

import queue
import threading
import time
from typing import List, Optional

from ..config import SamplingParams

_SHUTDOWN = object()


class _Request:
    __slots__ = ("prompts", "sp", "eos", "event", "result", "error",
                 "stream_q", "cancel_event")

    def __init__(self, prompts, sp, eos, stream=False, cancel_event=None):
        self.prompts = prompts
        self.sp = sp
        self.eos = eos
        self.event = threading.Event()
        self.result = None
        self.error = None
        # streaming requests get an event queue the submitter drains:
        # ("tok", prompt_i, cand_i, [ids...]) then ("done", result) or
        # ("err", exception)
        self.stream_q = queue.Queue() if stream else None
        # optional abort signal (e.g. SSE client disconnect): once set,
        # the engine retires this request's lanes with partial output
        self.cancel_event = cancel_event


class DynamicBatcher:
    """Thread-owning wrapper that merges concurrent generate calls.

    max_wait_ms: after the first waiting request is picked up, how long
    to keep draining the queue for co-batchable requests before running.
    Zero still batches whatever has already queued up while the previous
    wave was running (the common steady-state case).
    """

    def __init__(self, engine, max_wait_ms: float = 2.0):
        self.engine = engine
        self.max_wait_ms = max_wait_ms
        self.calls = 0  # engine.generate invocations (for tests/metrics)
        self._closed = False
        self._q: "queue.Queue" = queue.Queue()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="distrl-batcher")
        self._thread.start()

    # ---------------------------------------------------------------- API

    def submit(self, prompts: List[List[int]], sp: SamplingParams,
               eos_token_id: Optional[int] = None,
               cancel_event: Optional[threading.Event] = None
               ) -> List[List[List[int]]]:
        """Blocking: returns the engine.generate result for ``prompts``.
        A set ``cancel_event`` aborts the request (partial outputs)."""
        if self._closed:
            raise RuntimeError("DynamicBatcher is closed")
        req = _Request(prompts, sp, eos_token_id, cancel_event=cancel_event)
        self._q.put(req)
        req.event.wait()
        if req.error is not None:
            raise req.error
        return req.result

    def submit_stream(self, prompts: List[List[int]], sp: SamplingParams,
                      eos_token_id: Optional[int] = None,
                      cancel_event: Optional[threading.Event] = None
                      ) -> "queue.Queue":
        """Non-blocking: enqueue a streaming request and return its event
        queue. Items: ("tok", prompt_i, cand_i, [token_ids]) deltas, then
        a final ("done", full_result) or ("err", exception). Setting
        ``cancel_event`` (client disconnect) aborts generation within one
        decode chunk."""
        if self._closed:
            raise RuntimeError("DynamicBatcher is closed")
        req = _Request(prompts, sp, eos_token_id, stream=True,
                       cancel_event=cancel_event)
        self._q.put(req)
        return req.stream_q

    def close(self):
        self._closed = True
        self._q.put(_SHUTDOWN)
        self._thread.join(timeout=10)

    # --------------------------------------------------------------- loop

    @staticmethod
    def _key(req: _Request):
        sp = req.sp
        # seeded requests must run alone (their reproducibility contract
        # is over the exact call); streaming requests too (the stream_cb
        # belongs to one caller)
        nonce = (id(req) if sp.seed is not None or req.stream_q is not None
                 else None)
        # max_tokens is NOT part of the key: requests with different
        # output caps batch together via the engine's per-sequence token
        # limits (in-wave retirement frees a short request's lanes while
        # longer ones keep decoding)
        return (sp.temperature, sp.top_p, sp.top_k, sp.n, sp.geom_len_mean,
                req.eos, nonce)

    def _loop(self):
        shutdown = False
        while not shutdown:
            first = self._q.get()
            if first is _SHUTDOWN:
                break
            batch = [first]
            deadline = time.monotonic() + self.max_wait_ms / 1e3
            while True:
                timeout = deadline - time.monotonic()
                try:
                    nxt = self._q.get(timeout=max(timeout, 0.0))
                except queue.Empty:
                    break
                if nxt is _SHUTDOWN:
                    shutdown = True
                    break
                batch.append(nxt)
            groups: dict = {}
            for req in batch:
                groups.setdefault(self._key(req), []).append(req)
            for reqs in groups.values():
                self._run_group(reqs)
        # fail any submitters that raced the shutdown (never strand a
        # blocked caller)
        while True:
            try:
                it = self._q.get_nowait()
            except queue.Empty:
                break
            if it is not _SHUTDOWN:
                it.error = RuntimeError("DynamicBatcher is closed")
                if it.stream_q is not None:
                    it.stream_q.put(("err", it.error))
                it.event.set()

    def _run_group(self, reqs: List[_Request]):
        merged = [p for r in reqs for p in r.prompts]
        stream_q = reqs[0].stream_q  # streaming groups are singletons
        cb = None
        if stream_q is not None:
            cb = lambda pi, ci, toks: stream_q.put(("tok", pi, ci, toks))
        try:
            self.calls += 1
            sp = reqs[0].sp
            limits = None
            caps = [r.sp.max_tokens for r in reqs]
            if len(set(caps)) > 1:
                import dataclasses
                sp = dataclasses.replace(sp, max_tokens=max(caps))
                limits = [[r.sp.max_tokens] * sp.n
                          for r in reqs for _ in r.prompts]
            cancel_check = None
            if any(r.cancel_event is not None for r in reqs):
                # map merged prompt index -> owning request's abort event
                events = [r.cancel_event for r in reqs for _ in r.prompts]
                cancel_check = lambda pi: (events[pi] is not None
                                           and events[pi].is_set())
            outs = self.engine.generate(merged, sp,
                                        eos_token_id=reqs[0].eos,
                                        stream_cb=cb,
                                        token_limits=limits,
                                        cancel_check=cancel_check)
            off = 0
            for r in reqs:
                r.result = outs[off:off + len(r.prompts)]
                off += len(r.prompts)
            if stream_q is not None:
                stream_q.put(("done", reqs[0].result))
        except Exception as e:  # deliver the failure to every waiter
            for r in reqs:
                r.error = e
            if stream_q is not None:
                stream_q.put(("err", e))
        finally:
            for r in reqs:
                r.event.set()
