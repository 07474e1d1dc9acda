"""Rate-limit detection and abortable sleep (reference: src/shared/rate-limit.ts).

Regex detection of rate/usage-limit failure text, reset-time parsing
("reset at 2:30 PM", "in 5 minutes", unix ts), wait clamped 30s–60min.
"""
from __future__ import annotations

import asyncio
import re
import time
from dataclasses import dataclass

from .constants import RATE_LIMIT_MAX_WAIT_MS, RATE_LIMIT_MIN_WAIT_MS

RATE_LIMIT_PATTERNS = [
    re.compile(r"rate.?limit", re.I),
    re.compile(r"usage.?limit", re.I),
    re.compile(r"too many requests", re.I),
    re.compile(r"\b429\b"),
    re.compile(r"quota exceeded", re.I),
    re.compile(r"overloaded", re.I),
]

_RESET_AT = re.compile(r"reset(?:s)? at (\d{1,2}):(\d{2})\s*(AM|PM)?", re.I)
_IN_MINUTES = re.compile(r"in (\d+)\s*min", re.I)
_IN_SECONDS = re.compile(r"in (\d+)\s*sec", re.I)
_UNIX_TS = re.compile(r"reset(?:s)?(?: at)? (\d{10})(?:\b|$)")


class RateLimitError(Exception):
    def __init__(self, message: str, wait_ms: int):
        super().__init__(message)
        self.wait_ms = wait_ms


@dataclass
class RateLimitInfo:
    detected: bool
    wait_ms: int = 0


def clamp_wait(ms: int) -> int:
    return max(RATE_LIMIT_MIN_WAIT_MS, min(RATE_LIMIT_MAX_WAIT_MS, ms))


def detect_rate_limit(text: str | None) -> RateLimitInfo:
    if not text:
        return RateLimitInfo(False)
    if not any(p.search(text) for p in RATE_LIMIT_PATTERNS):
        return RateLimitInfo(False)
    wait_ms = RATE_LIMIT_MIN_WAIT_MS

    m = _IN_MINUTES.search(text)
    if m:
        wait_ms = int(m.group(1)) * 60_000
    else:
        m = _IN_SECONDS.search(text)
        if m:
            wait_ms = int(m.group(1)) * 1000
        else:
            m = _UNIX_TS.search(text)
            if m:
                wait_ms = int((int(m.group(1)) - time.time()) * 1000)
            else:
                m = _RESET_AT.search(text)
                if m:
                    hour, minute = int(m.group(1)), int(m.group(2))
                    ampm = (m.group(3) or "").upper()
                    if ampm == "PM" and hour < 12:
                        hour += 12
                    if ampm == "AM" and hour == 12:
                        hour = 0
                    now = time.localtime()
                    target = time.mktime((now.tm_year, now.tm_mon, now.tm_mday,
                                          hour, minute, 0, 0, 0, -1))
                    if target < time.time():
                        target += 86400
                    wait_ms = int((target - time.time()) * 1000)
    return RateLimitInfo(True, clamp_wait(wait_ms))


async def abortable_sleep(ms: int, abort_event: asyncio.Event | None = None) -> bool:
    """Sleep that can be cut short by an abort event (the triggerAgent wake
    semantics, reference agent-loop.ts:266-287). Returns True if aborted."""
    if abort_event is None:
        await asyncio.sleep(ms / 1000)
        return False
    try:
        await asyncio.wait_for(abort_event.wait(), timeout=ms / 1000)
        return True
    except asyncio.TimeoutError:
        return False
