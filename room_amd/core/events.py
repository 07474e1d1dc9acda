"""In-process pub/sub event bus with channel + wildcard handlers
(reference: src/server/event-bus.ts). Feeds the WS fan-out."""
from __future__ import annotations

import time
from collections import defaultdict
from typing import Any, Callable

Handler = Callable[[str, Any], None]


class EventBus:
    def __init__(self) -> None:
        self._handlers: dict[str, list[Handler]] = defaultdict(list)
        self._wildcard: list[Handler] = []

    def on(self, channel: str, handler: Handler) -> Callable[[], None]:
        if channel == "*":
            self._wildcard.append(handler)
            return lambda: self._wildcard.remove(handler)
        self._handlers[channel].append(handler)
        return lambda: self._handlers[channel].remove(handler)

    def emit(self, channel: str, event_type: str, data: Any = None) -> dict:
        event = {"type": event_type, "channel": channel, "data": data,
                 "timestamp": int(time.time() * 1000)}
        for h in list(self._handlers.get(channel, [])):
            try:
                h(channel, event)
            except Exception:
                pass
        for h in list(self._wildcard):
            try:
                h(channel, event)
            except Exception:
                pass
        return event


# Global bus instance (the server creates its own; this one serves tests and
# embedded use).
bus = EventBus()
