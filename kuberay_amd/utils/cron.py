"""Minimal 5-field cron parser + next-fire computation (robfig/cron analog,
used by the RayCronJob reconciler; reference: raycronjob_controller.go)."""
from __future__ import annotations

import datetime as dt
from typing import Optional, Set

_FIELDS = [
    ("minute", 0, 59),
    ("hour", 0, 23),
    ("dom", 1, 31),
    ("month", 1, 12),
    ("dow", 0, 6),  # 0 = Sunday
]

_MACROS = {
    "@hourly": "0 * * * *",
    "@daily": "0 0 * * *",
    "@midnight": "0 0 * * *",
    "@weekly": "0 0 * * 0",
    "@monthly": "0 0 1 * *",
    "@yearly": "0 0 1 1 *",
    "@annually": "0 0 1 1 *",
}


def _parse_field(expr: str, lo: int, hi: int, name: str) -> Set[int]:
    values: Set[int] = set()
    for part in expr.split(","):
        step = 1
        if "/" in part:
            part, step_s = part.split("/", 1)
            try:
                step = int(step_s)
            except ValueError:
                raise ValueError(f"{name}: bad step '{step_s}'")
            if step <= 0:
                raise ValueError(f"{name}: step must be positive")
        if part in ("*", ""):
            lo2, hi2 = lo, hi
        elif "-" in part:
            a, b = part.split("-", 1)
            lo2, hi2 = int(a), int(b)
        else:
            lo2 = hi2 = int(part)
        if lo2 < lo or hi2 > hi or lo2 > hi2:
            raise ValueError(f"{name}: value out of range in '{expr}'")
        values.update(range(lo2, hi2 + 1, step))
    return values


class CronSchedule:
    def __init__(self, minute, hour, dom, month, dow, dom_star, dow_star):
        self.minute, self.hour, self.dom = minute, hour, dom
        self.month, self.dow = month, dow
        self.dom_star, self.dow_star = dom_star, dow_star

    def matches(self, t: dt.datetime) -> bool:
        if t.minute not in self.minute or t.hour not in self.hour or t.month not in self.month:
            return False
        dow = (t.weekday() + 1) % 7  # python Mon=0 -> cron Sun=0
        dom_ok = t.day in self.dom
        dow_ok = dow in self.dow
        # standard cron: if both dom and dow are restricted, match either
        if not self.dom_star and not self.dow_star:
            return dom_ok or dow_ok
        return dom_ok and dow_ok

    def next_after(self, t: dt.datetime, limit_days: int = 366 * 5) -> Optional[dt.datetime]:
        t = (t.replace(second=0, microsecond=0) + dt.timedelta(minutes=1))
        end = t + dt.timedelta(days=limit_days)
        while t < end:
            if t.month not in self.month:
                if t.month == 12:
                    t = t.replace(year=t.year + 1, month=1, day=1, hour=0, minute=0)
                else:
                    t = t.replace(month=t.month + 1, day=1, hour=0, minute=0)
                continue
            if self.matches(t):
                return t
            t += dt.timedelta(minutes=1)
        return None


def parse_cron(expr: str) -> CronSchedule:
    expr = expr.strip()
    expr = _MACROS.get(expr, expr)
    parts = expr.split()
    if len(parts) != 5:
        raise ValueError(f"cron expression must have 5 fields, got {len(parts)}")
    parsed = []
    for (name, lo, hi), part in zip(_FIELDS, parts):
        field_part = part
        if name == "dow":
            # allow 7 == Sunday
            field_part = field_part.replace("7", "0") if field_part == "7" else field_part
        parsed.append(_parse_field(field_part, lo, hi, name))
    return CronSchedule(parsed[0], parsed[1], parsed[2], parsed[3], parsed[4],
                        dom_star=parts[2] == "*", dow_star=parts[4] == "*")
