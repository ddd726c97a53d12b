"""Cypher temporal values: DATE / TIME / DATETIME / DURATION.

Parity: the reference supports Neo4j temporal literals through its Cypher
layer and APOC date helpers (apoc/date, apoc/temporal — SURVEY.md §2 row
"APOC library"). Implemented here as thin wrappers over Python's datetime
with Neo4j accessor names (``d.year``, ``dt.epochMillis``,
``dur.seconds`` …), ISO-8601 parsing/printing, and the Cypher arithmetic
rules (temporal ± duration, duration ± duration, duration × number).

Property access goes through ``component()`` which the executor's Prop
evaluator calls for these types (they are not dicts or graph entities).
"""

from __future__ import annotations

import datetime as _dt
import re
from typing import Any, Optional

_DUR_RE = re.compile(
    r"^P(?:(-?\d+(?:\.\d+)?)Y)?(?:(-?\d+(?:\.\d+)?)M)?(?:(-?\d+(?:\.\d+)?)W)?"
    r"(?:(-?\d+(?:\.\d+)?)D)?"
    r"(?:T(?:(-?\d+(?:\.\d+)?)H)?(?:(-?\d+(?:\.\d+)?)M)?(?:(-?\d+(?:\.\d+)?)S)?)?$"
)


class CypherDuration:
    """Months / days / seconds / nanoseconds quad (Neo4j's duration model:
    calendar components don't collapse into each other)."""

    __slots__ = ("months", "days", "seconds", "nanoseconds")

    def __init__(self, months=0, days=0, seconds=0, nanoseconds=0):
        extra, self.nanoseconds = divmod(int(nanoseconds), 1_000_000_000)
        self.months = int(months)
        self.days = int(days)
        self.seconds = int(seconds) + extra

    # ---- constructors ----
    @classmethod
    def parse(cls, s: str) -> "CypherDuration":
        m = _DUR_RE.match(s.strip())
        if not m or s.strip() in ("P", "PT"):
            raise ValueError(f"invalid duration literal {s!r}")
        y, mo, w, d, h, mi, sec = (float(g) if g else 0.0 for g in m.groups())
        months = y * 12 + mo
        days = w * 7 + d
        seconds = h * 3600 + mi * 60 + sec
        frac = seconds - int(seconds)
        return cls(months, days, int(seconds), round(frac * 1e9))

    @classmethod
    def from_map(cls, m: dict) -> "CypherDuration":
        months = m.get("years", 0) * 12 + m.get("months", 0)
        days = m.get("weeks", 0) * 7 + m.get("days", 0)
        seconds = (m.get("hours", 0) * 3600 + m.get("minutes", 0) * 60
                   + m.get("seconds", 0))
        ns = (m.get("milliseconds", 0) * 1_000_000
              + m.get("microseconds", 0) * 1_000 + m.get("nanoseconds", 0))
        frac = seconds - int(seconds)
        return cls(int(months), int(days), int(seconds), round(frac * 1e9) + ns)

    # ---- accessors (Neo4j names) ----
    def component(self, key: str):
        k = key.lower()
        total_s = self.seconds + self.nanoseconds / 1e9
        table = {
            "years": self.months // 12, "months": self.months,
            "monthsofyear": self.months % 12,
            "weeks": self.days // 7, "days": self.days,
            "daysofweek": self.days % 7,
            "hours": self.seconds // 3600,
            "minutes": self.seconds // 60,
            "minutesofhour": (self.seconds // 60) % 60,
            "seconds": self.seconds, "secondsofminute": self.seconds % 60,
            "milliseconds": self.seconds * 1000 + self.nanoseconds // 1_000_000,
            "microseconds": self.seconds * 1_000_000 + self.nanoseconds // 1000,
            "nanoseconds": self.seconds * 1_000_000_000 + self.nanoseconds,
        }
        if k in table:
            return table[k]
        raise KeyError(key)

    def total_seconds_approx(self) -> float:
        """For comparison only: months ≈ 30.4375 days (Neo4j AVG_DAYS_PER_MONTH)."""
        return (self.months * 30.4375 * 86400 + self.days * 86400
                + self.seconds + self.nanoseconds / 1e9)

    # ---- arithmetic ----
    def __add__(self, o):
        if isinstance(o, CypherDuration):
            return CypherDuration(self.months + o.months, self.days + o.days,
                                  self.seconds + o.seconds,
                                  self.nanoseconds + o.nanoseconds)
        if isinstance(o, (CypherDate, CypherDateTime, CypherTime)):
            # duration + temporal commutes (Neo4j: duration('P7D') + date(...))
            return o + self
        return NotImplemented

    def __radd__(self, o):
        # sum() starts from 0
        if o == 0:
            return self
        return NotImplemented

    def __sub__(self, o):
        if isinstance(o, CypherDuration):
            return CypherDuration(self.months - o.months, self.days - o.days,
                                  self.seconds - o.seconds,
                                  self.nanoseconds - o.nanoseconds)
        return NotImplemented

    def __mul__(self, k):
        if isinstance(k, (int, float)) and not isinstance(k, bool):
            sec = self.seconds * k
            return CypherDuration(round(self.months * k), round(self.days * k),
                                  int(sec),
                                  round((sec - int(sec)) * 1e9
                                        + self.nanoseconds * k))
        return NotImplemented

    __rmul__ = __mul__

    def __truediv__(self, k):
        return self.__mul__(1.0 / k)

    def __neg__(self):
        return self * -1

    def __eq__(self, o):
        return (isinstance(o, CypherDuration)
                and (self.months, self.days, self.seconds, self.nanoseconds)
                == (o.months, o.days, o.seconds, o.nanoseconds))

    def __hash__(self):
        return hash(("dur", self.months, self.days, self.seconds,
                     self.nanoseconds))

    def __lt__(self, o):
        if not isinstance(o, CypherDuration):
            return NotImplemented
        return self.total_seconds_approx() < o.total_seconds_approx()

    def to_timedelta(self) -> _dt.timedelta:
        return _dt.timedelta(days=self.months * 30 + self.days,
                             seconds=self.seconds,
                             microseconds=self.nanoseconds / 1000)

    def __str__(self):
        out = "P"
        if self.months:
            y, m = divmod(self.months, 12)
            if y:
                out += f"{y}Y"
            if m:
                out += f"{m}M"
        if self.days:
            out += f"{self.days}D"
        if self.seconds or self.nanoseconds or out == "P":
            out += "T"
            s = self.seconds
            h, s = divmod(s, 3600)
            mi, s = divmod(s, 60)
            if h:
                out += f"{h}H"
            if mi:
                out += f"{mi}M"
            if s or self.nanoseconds or out.endswith("T"):
                if self.nanoseconds:
                    out += f"{s + self.nanoseconds / 1e9:.9f}".rstrip("0") + "S"
                else:
                    out += f"{s}S"
        return out

    __repr__ = __str__


class _TemporalBase:
    """Shared accessor plumbing for date/time/datetime wrappers around a
    datetime.datetime `_v` (tz-aware for DATETIME/TIME, naive otherwise)."""

    __slots__ = ("_v",)

    def component(self, key: str):
        v = self._v
        k = key.lower()
        iso = v.isocalendar()
        table = {
            "year": v.year, "month": v.month, "day": v.day,
            "quarter": (v.month - 1) // 3 + 1,
            "week": iso[1], "weekyear": iso[0],
            "dayofweek": v.isoweekday(), "dayofquarter":
                (v.date() - _dt.date(v.year, ((v.month - 1) // 3) * 3 + 1, 1)).days + 1,
            "ordinalday": v.timetuple().tm_yday, "dayofyear": v.timetuple().tm_yday,
            "hour": v.hour, "minute": v.minute, "second": v.second,
            "millisecond": v.microsecond // 1000,
            "microsecond": v.microsecond,
            "nanosecond": v.microsecond * 1000,
            "epochseconds": int(self._epoch()),
            "epochmillis": int(self._epoch() * 1000),
            "timezone": str(v.tzinfo) if v.tzinfo else None,
            "offset": (v.strftime("%z") or None) if v.tzinfo else None,
        }
        if k in table:
            return table[k]
        raise KeyError(key)

    def _epoch(self) -> float:
        v = self._v
        if v.tzinfo is None:
            v = v.replace(tzinfo=_dt.timezone.utc)
        return v.timestamp()

    def __eq__(self, o):
        return type(o) is type(self) and self._v == o._v

    def __hash__(self):
        return hash((type(self).__name__, self._v))

    def __lt__(self, o):
        if type(o) is not type(self):
            return NotImplemented
        return self._v < o._v

    def __le__(self, o):
        return self == o or self < o

    def __gt__(self, o):
        if type(o) is not type(self):
            return NotImplemented
        return o < self

    def __ge__(self, o):
        return self == o or o < self


class CypherDate(_TemporalBase):
    def __init__(self, v: _dt.date):
        self._v = _dt.datetime(v.year, v.month, v.day)

    @property
    def date(self) -> _dt.date:
        return self._v.date()

    def __add__(self, o):
        if isinstance(o, CypherDuration):
            return CypherDate(_shift(self._v, o).date())
        return NotImplemented

    def __sub__(self, o):
        if isinstance(o, CypherDuration):
            return CypherDate(_shift(self._v, -o).date())
        if isinstance(o, CypherDate):
            d = self._v - o._v
            return CypherDuration(0, d.days, 0, 0)
        return NotImplemented

    def __str__(self):
        return self._v.date().isoformat()

    __repr__ = __str__


class CypherDateTime(_TemporalBase):
    def __init__(self, v: _dt.datetime):
        self._v = v

    def __add__(self, o):
        if isinstance(o, CypherDuration):
            return CypherDateTime(_shift(self._v, o))
        return NotImplemented

    def __sub__(self, o):
        if isinstance(o, CypherDuration):
            return CypherDateTime(_shift(self._v, -o))
        if isinstance(o, CypherDateTime):
            d = self._v - o._v
            return CypherDuration(0, 0, int(d.total_seconds()),
                                  d.microseconds % 1_000_000 * 1000
                                  if d.total_seconds() >= 0 else 0)
        return NotImplemented

    def __str__(self):
        return self._v.isoformat()

    __repr__ = __str__


class CypherTime(_TemporalBase):
    def __init__(self, v: _dt.time):
        self._v = _dt.datetime(1970, 1, 1, v.hour, v.minute, v.second,
                               v.microsecond, tzinfo=v.tzinfo)

    def __add__(self, o):
        if isinstance(o, CypherDuration):
            nv = self._v + _dt.timedelta(seconds=o.seconds,
                                         microseconds=o.nanoseconds / 1000)
            return CypherTime(nv.timetz())
        return NotImplemented

    def __sub__(self, o):
        if isinstance(o, CypherDuration):
            return self.__add__(-o)
        return NotImplemented

    def __str__(self):
        return self._v.timetz().isoformat()

    __repr__ = __str__


def _shift(v: _dt.datetime, d: CypherDuration) -> _dt.datetime:
    """Calendar-correct shift: months first (clamping the day), then
    days/seconds — Neo4j's temporal arithmetic order."""
    if d.months:
        total = v.year * 12 + (v.month - 1) + d.months
        year, month0 = divmod(total, 12)
        day = min(v.day, _days_in_month(year, month0 + 1))
        v = v.replace(year=year, month=month0 + 1, day=day)
    return v + _dt.timedelta(days=d.days, seconds=d.seconds,
                             microseconds=d.nanoseconds / 1000)


def _days_in_month(y, m):
    if m == 12:
        return 31
    return (_dt.date(y, m + 1, 1) - _dt.date(y, m, 1)).days


# ---- constructor helpers used by the function registry ----
def _parse_tz(tz: Optional[str]):
    if tz is None or tz.upper() in ("Z", "UTC"):
        return _dt.timezone.utc
    m = re.match(r"^([+-])(\d{2}):?(\d{2})$", tz)
    if m:
        sign = 1 if m.group(1) == "+" else -1
        return _dt.timezone(sign * _dt.timedelta(hours=int(m.group(2)),
                                                 minutes=int(m.group(3))))
    return _dt.timezone.utc


def make_date(arg: Any = None) -> CypherDate:
    if arg is None:
        return CypherDate(_dt.date.today())
    if isinstance(arg, CypherDate):
        return arg
    if isinstance(arg, CypherDateTime):
        return CypherDate(arg._v.date())
    if isinstance(arg, str):
        return CypherDate(_dt.date.fromisoformat(arg))
    if isinstance(arg, dict):
        return CypherDate(_dt.date(int(arg.get("year", 1970)),
                                   int(arg.get("month", 1)),
                                   int(arg.get("day", 1))))
    raise ValueError(f"cannot build date from {arg!r}")


def make_datetime(arg: Any = None, *, local=False) -> CypherDateTime:
    tz = None if local else _dt.timezone.utc
    if arg is None:
        now = _dt.datetime.now(tz) if tz else _dt.datetime.now()
        return CypherDateTime(now)
    if isinstance(arg, CypherDateTime):
        return arg
    if isinstance(arg, CypherDate):
        return CypherDateTime(arg._v.replace(tzinfo=tz))
    if isinstance(arg, (int, float)):
        # epochMillis convention (reference stores ms timestamps)
        return CypherDateTime(_dt.datetime.fromtimestamp(arg / 1000.0, tz))
    if isinstance(arg, str):
        s = arg.strip().replace("Z", "+00:00")
        v = _dt.datetime.fromisoformat(s)
        if not local and v.tzinfo is None:
            v = v.replace(tzinfo=_dt.timezone.utc)
        if local:
            v = v.replace(tzinfo=None)
        return CypherDateTime(v)
    if isinstance(arg, dict):
        if "epochmillis" in {k.lower() for k in arg}:
            ms = next(v for k, v in arg.items() if k.lower() == "epochmillis")
            return CypherDateTime(_dt.datetime.fromtimestamp(ms / 1000.0, tz))
        if "epochseconds" in {k.lower() for k in arg}:
            s = next(v for k, v in arg.items() if k.lower() == "epochseconds")
            return CypherDateTime(_dt.datetime.fromtimestamp(float(s), tz))
        v = _dt.datetime(int(arg.get("year", 1970)), int(arg.get("month", 1)),
                         int(arg.get("day", 1)), int(arg.get("hour", 0)),
                         int(arg.get("minute", 0)), int(arg.get("second", 0)),
                         int(arg.get("millisecond", 0)) * 1000
                         + int(arg.get("microsecond", 0)),
                         tzinfo=None if local else
                         _parse_tz(arg.get("timezone")))
        return CypherDateTime(v)
    raise ValueError(f"cannot build datetime from {arg!r}")


def make_time(arg: Any = None, *, local=False) -> CypherTime:
    tz = None if local else _dt.timezone.utc
    if arg is None:
        now = _dt.datetime.now(_dt.timezone.utc)
        t = now.timetz() if not local else now.time()
        return CypherTime(t)
    if isinstance(arg, CypherTime):
        return arg
    if isinstance(arg, CypherDateTime):
        return CypherTime(arg._v.timetz() if not local else arg._v.time())
    if isinstance(arg, str):
        s = arg.strip().replace("Z", "+00:00")
        t = _dt.time.fromisoformat(s)
        if not local and t.tzinfo is None:
            t = t.replace(tzinfo=_dt.timezone.utc)
        if local:
            t = t.replace(tzinfo=None)
        return CypherTime(t)
    if isinstance(arg, dict):
        return CypherTime(_dt.time(int(arg.get("hour", 0)),
                                   int(arg.get("minute", 0)),
                                   int(arg.get("second", 0)),
                                   int(arg.get("millisecond", 0)) * 1000
                                   + int(arg.get("microsecond", 0)),
                                   tzinfo=tz))
    raise ValueError(f"cannot build time from {arg!r}")


def make_duration(arg: Any) -> CypherDuration:
    if isinstance(arg, CypherDuration):
        return arg
    if isinstance(arg, str):
        return CypherDuration.parse(arg)
    if isinstance(arg, dict):
        return CypherDuration.from_map(arg)
    raise ValueError(f"cannot build duration from {arg!r}")


def duration_between(a, b) -> CypherDuration:
    av = make_datetime(a) if not isinstance(a, CypherDateTime) else a
    bv = make_datetime(b) if not isinstance(b, CypherDateTime) else b
    x, y = av._v, bv._v
    if (x.tzinfo is None) != (y.tzinfo is None):
        x = x.replace(tzinfo=_dt.timezone.utc) if x.tzinfo is None else x
        y = y.replace(tzinfo=_dt.timezone.utc) if y.tzinfo is None else y
    d = y - x
    return CypherDuration(0, d.days, d.seconds, d.microseconds * 1000)


def truncate(unit: str, value, *, kind="datetime"):
    """datetime.truncate('day', dt) family."""
    v = value._v if isinstance(value, (_TemporalBase,)) else make_datetime(value)._v
    u = unit.lower()
    if u == "year":
        v = v.replace(month=1, day=1, hour=0, minute=0, second=0, microsecond=0)
    elif u == "quarter":
        v = v.replace(month=((v.month - 1) // 3) * 3 + 1, day=1, hour=0,
                      minute=0, second=0, microsecond=0)
    elif u == "month":
        v = v.replace(day=1, hour=0, minute=0, second=0, microsecond=0)
    elif u == "week":
        v = (v - _dt.timedelta(days=v.isoweekday() - 1)).replace(
            hour=0, minute=0, second=0, microsecond=0)
    elif u == "day":
        v = v.replace(hour=0, minute=0, second=0, microsecond=0)
    elif u == "hour":
        v = v.replace(minute=0, second=0, microsecond=0)
    elif u == "minute":
        v = v.replace(second=0, microsecond=0)
    elif u == "second":
        v = v.replace(microsecond=0)
    else:
        raise ValueError(f"unknown truncation unit {unit!r}")
    if kind == "date":
        return CypherDate(v.date())
    return CypherDateTime(v)
