"""In-process typed event buses feeding sync-waiters, SSE and WebSocket
streams (reference parity: internal/events/event_bus.go — generic bus with
buffered per-subscriber queues and non-blocking publish)."""
from __future__ import annotations

import asyncio
import itertools
from typing import Any


class EventBus:
    def __init__(self, buffer: int = 64):
        self._subs: dict[int, asyncio.Queue] = {}
        self._buffer = buffer
        self._ids = itertools.count()
        self.dropped = 0

    def subscribe(self) -> tuple[int, asyncio.Queue]:
        sid = next(self._ids)
        q: asyncio.Queue = asyncio.Queue(self._buffer)
        self._subs[sid] = q
        return sid, q

    def unsubscribe(self, sid: int) -> None:
        self._subs.pop(sid, None)

    def publish(self, event: Any) -> None:
        """Non-blocking: slow subscribers drop events rather than stall."""
        for q in list(self._subs.values()):
            try:
                q.put_nowait(event)
            except asyncio.QueueFull:
                self.dropped += 1

    @property
    def n_subscribers(self) -> int:
        return len(self._subs)


class Buses:
    """The control plane's bus set (execution / node / reasoner / memory).

    Sync-execute waiters do NOT ride the broadcast bus: with C concurrent
    waiters a broadcast is O(C) per event / O(C²) per batch and bounded
    subscriber queues drop the very terminal event a waiter needs (measured:
    p99 = the full 90 s sync timeout under concurrency 64).  Waiters are a
    targeted {execution_id -> [Future]} registry resolved in O(1); the bus
    stays for SSE/WS consumers that want the whole stream."""

    def __init__(self):
        self.execution = EventBus()
        self.node = EventBus()
        self.reasoner = EventBus()
        self.memory = EventBus(buffer=256)
        self._exec_waiters: dict[str, list[asyncio.Future]] = {}

    def register_waiter(self, execution_id: str) -> asyncio.Future:
        """Register BEFORE dispatching to the agent so a fast callback
        cannot race past the waiter."""
        fut = asyncio.get_running_loop().create_future()
        self._exec_waiters.setdefault(execution_id, []).append(fut)
        return fut

    def discard_waiter(self, execution_id: str, fut: asyncio.Future) -> None:
        lst = self._exec_waiters.get(execution_id)
        if lst is not None:
            try:
                lst.remove(fut)
            except ValueError:
                pass
            if not lst:
                self._exec_waiters.pop(execution_id, None)

    def publish_execution(self, event: dict) -> None:
        """Publish to SSE subscribers and resolve any sync waiters."""
        self.execution.publish(event)
        if event.get("terminal"):
            for fut in self._exec_waiters.pop(event["execution_id"], ()):
                if not fut.done():
                    fut.set_result(event)

    async def wait_for_execution(self, execution_id: str, timeout: float,
                                 fut: asyncio.Future | None = None):
        """Targeted wait for an execution's terminal event (reference:
        waitForExecutionCompletion, execute.go:568-629)."""
        if fut is None:
            fut = self.register_waiter(execution_id)
        try:
            return await asyncio.wait_for(fut, timeout)
        except asyncio.TimeoutError:
            return None
        finally:
            self.discard_waiter(execution_id, fut)
