"""5-field cron parser/matcher (the reference uses node-cron; same semantics:
minute hour day-of-month month day-of-week, with *, */n, a-b, lists)."""
from __future__ import annotations

from datetime import datetime, timedelta

FIELD_RANGES = [(0, 59), (0, 23), (1, 31), (1, 12), (0, 6)]


def _parse_field(spec: str, lo: int, hi: int) -> set[int]:
    out: set[int] = set()
    for part in spec.split(","):
        part = part.strip()
        step = 1
        if "/" in part:
            part, step_s = part.split("/", 1)
            step = int(step_s)
        if part in ("*", ""):
            rng = range(lo, hi + 1)
        elif "-" in part:
            a, b = part.split("-", 1)
            rng = range(int(a), int(b) + 1)
        else:
            v = int(part)
            if v < lo or v > hi:
                raise ValueError(f"cron field value {v} out of range [{lo},{hi}]")
            rng = range(v, v + 1)
        for v in rng:
            if (v - rng.start) % step == 0 and lo <= v <= hi:
                out.add(v)
    if not out:
        raise ValueError(f"empty cron field: {spec}")
    return out


class CronExpression:
    def __init__(self, expr: str):
        fields = expr.split()
        if len(fields) != 5:
            raise ValueError(f"cron needs 5 fields, got {len(fields)}: {expr!r}")
        self.minute, self.hour, self.dom, self.month, self.dow = (
            _parse_field(f, lo, hi) for f, (lo, hi) in zip(fields, FIELD_RANGES))
        # standard cron (and node-cron): when BOTH day-of-month and
        # day-of-week are restricted, a time matches if EITHER matches
        # ('0 0 13 * 5' = every Friday AND every 13th, not Friday-the-13th)
        self._dom_restricted = fields[2].strip() != "*"
        self._dow_restricted = fields[4].strip() != "*"

    def matches(self, dt: datetime) -> bool:
        return self._full_match(dt)

    def _dow_match(self, dt: datetime) -> bool:
        cron_dow = (dt.weekday() + 1) % 7  # python Mon=0 → cron Sun=0
        return cron_dow in self.dow

    def _day_match(self, dt: datetime) -> bool:
        if self._dom_restricted and self._dow_restricted:
            return dt.day in self.dom or self._dow_match(dt)
        return dt.day in self.dom and self._dow_match(dt)

    def _full_match(self, dt: datetime) -> bool:
        return (dt.minute in self.minute and dt.hour in self.hour
                and dt.month in self.month and self._day_match(dt))

    def next_after(self, dt: datetime, limit_days: int = 366) -> datetime | None:
        t = dt.replace(second=0, microsecond=0) + timedelta(minutes=1)
        end = dt + timedelta(days=limit_days)
        while t <= end:
            if self._full_match(t):
                return t
            t += timedelta(minutes=1)
        return None


def validate_cron(expr: str) -> bool:
    try:
        CronExpression(expr)
        return True
    except (ValueError, TypeError):
        return False
