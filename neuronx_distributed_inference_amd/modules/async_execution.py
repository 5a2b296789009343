"""Asynchronous execution: overlap host-side pre/post-processing with
device execution.

Analog of the reference's async runtime mode (`modules/async_execution.py`,
343 LoC; `NEURON_RT_ASYNC_EXEC_MAX_INFLIGHT_REQUESTS=2`,
ref model_base.py:3044-3046): there, the Neuron runtime keeps up to N
in-flight requests and returns async tensor handles. On MI355X the same
overlap falls out of HIP's stream model — a forward launched on a
dedicated stream runs asynchronously while the host prepares the next
step's inputs; an event per submission gives the completion handle.

Double buffering: each in-flight slot owns its own input buffers, so the
host can fill slot (i+1) while slot i executes. `max_inflight` bounds the
queue exactly like the reference's env knob.
"""

from __future__ import annotations

import collections
from typing import Any, Callable, Deque, Optional

import torch


class AsyncResult:
    """Handle for one submitted forward (ref AsyncTensorWrapper)."""

    def __init__(self, outputs: Any, event: Optional[torch.cuda.Event]):
        self._outputs = outputs
        self._event = event
        self._done = event is None

    def done(self) -> bool:
        if not self._done and self._event.query():
            self._done = True
        return self._done

    def result(self) -> Any:
        """Block until the forward finished; returns the output tensors
        (device-resident — no copy is forced)."""
        if not self._done:
            self._event.synchronize()
            self._done = True
        return self._outputs


class AsyncExecutor:
    """Run model forwards on a side stream with up to `max_inflight`
    outstanding submissions.

    submit() enqueues the callable's kernels on the executor stream and
    returns immediately with an AsyncResult; the host thread is free to
    prepare the next inputs. When `max_inflight` submissions are pending,
    submit() first waits for the oldest — the same back-pressure rule as
    the reference's inflight cap.

    On CPU (tests, no device) everything degrades to synchronous calls
    with the same API.
    """

    def __init__(self, max_inflight: int = 2, device: Optional[torch.device] = None):
        self.max_inflight = max(1, int(max_inflight))
        self._use_cuda = torch.cuda.is_available() and (
            device is None or device.type == "cuda"
        )
        self._stream = torch.cuda.Stream() if self._use_cuda else None
        self._pending: Deque[AsyncResult] = collections.deque()

    # ------------------------------------------------------------------
    def submit(self, fn: Callable[..., Any], *args, **kwargs) -> AsyncResult:
        while len(self._pending) >= self.max_inflight:
            self._pending.popleft().result()
        if not self._use_cuda:
            res = AsyncResult(fn(*args, **kwargs), None)
            self._pending.append(res)
            return res
        cur = torch.cuda.current_stream()
        self._stream.wait_stream(cur)  # inputs written on the main stream
        with torch.cuda.stream(self._stream):
            outputs = fn(*args, **kwargs)
            event = torch.cuda.Event()
            event.record(self._stream)
        # downstream main-stream work ordering is the caller's choice —
        # reading outputs through .result() is always safe
        res = AsyncResult(outputs, event)
        self._pending.append(res)
        return res

    # ------------------------------------------------------------------
    def drain(self):
        """Wait for every outstanding submission."""
        while self._pending:
            self._pending.popleft().result()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.drain()
        return False
