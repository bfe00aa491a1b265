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
    """The control plane's bus set (execution / node / reasoner / memory)."""

    def __init__(self):
        self.execution = EventBus()
        self.node = EventBus()
        self.reasoner = EventBus()
        self.memory = EventBus(buffer=256)

    async def wait_for_execution(self, execution_id: str, timeout: float,
                                 queue: asyncio.Queue | None = None):
        """Event-bus wait used by the sync execute path (reference:
        waitForExecutionCompletion, execute.go:568-629).  Pass a queue from
        an earlier subscribe() to close the subscribe-after-dispatch race:
        the subscription must exist BEFORE the agent is called, or a fast
        callback can fire the terminal event with no listener."""
        sid = None
        if queue is None:
            sid, queue = self.execution.subscribe()
        try:
            loop = asyncio.get_running_loop()
            deadline = loop.time() + timeout
            while True:
                left = deadline - loop.time()
                if left <= 0:
                    return None
                try:
                    ev = await asyncio.wait_for(queue.get(), left)
                except asyncio.TimeoutError:
                    return None
                if ev.get("execution_id") == execution_id and ev.get("terminal"):
                    return ev
        finally:
            if sid is not None:
                self.execution.unsubscribe(sid)
