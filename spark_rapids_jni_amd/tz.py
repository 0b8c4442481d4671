"""Timezone subsystem: host-built transition tables + device conversion.

Java API parity: GpuTimeZoneDB.java (host builds the whole tzdb at startup:
fixed transitions LIST<STRUCT<utcInstant, localInstant, offset>> + name ->
index map, shipped to the device once) and timezones.cu
(convert_timestamp_to_utc / convert_utc_timestamp_to_timezone binary-search
kernels), OrcDstRuleExtractor.java (recurring-rule recovery by probing
offsets).

Design difference (documented): instead of device-side DST *rule* evaluation,
recurring rules are pre-expanded into fixed transitions through year 2200 on
the host (TZif v2 + POSIX footer rules via Python zoneinfo probing), so the
device side is a pure binary search. Correct for any timestamp below the
expansion horizon; beyond it the last offset applies.
"""
import datetime
import struct
import zoneinfo
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch

from . import _native
from .columnar import Column, DType, make_validity

EXPAND_UNTIL_YEAR = 2200
_EPOCH = datetime.datetime(1970, 1, 1, tzinfo=datetime.timezone.utc)


def _utc_offset_at(tz, utc_seconds: int) -> int:
    dt = _EPOCH + datetime.timedelta(seconds=utc_seconds)
    return int(dt.astimezone(tz).utcoffset().total_seconds())


def extract_transitions(name: str,
                        until_year: int = EXPAND_UNTIL_YEAR
                        ) -> List[Tuple[int, int]]:
    """[(utc_second, offset_after_seconds)] transitions for a zone, found by
    probing zoneinfo offsets and bisecting each change (the
    OrcDstRuleExtractor probing approach, applied to the full table)."""
    tz = zoneinfo.ZoneInfo(name)
    start = int((datetime.datetime(1890, 1, 1, tzinfo=datetime.timezone.utc)
                 - _EPOCH).total_seconds())
    end = int((datetime.datetime(until_year, 1, 1,
                                 tzinfo=datetime.timezone.utc)
               - _EPOCH).total_seconds())
    # sentinel must survive *1e6 in int64: ~year -139000
    out = [(-(2**42), _utc_offset_at(tz, start))]
    step = 6 * 3600  # DST shifts are at least hours apart; 6h probe misses none
    prev_off = out[0][1]
    t = start
    while t < end:
        nxt = min(t + step, end)
        off = _utc_offset_at(tz, nxt)
        if off != prev_off:
            lo, hi = t, nxt  # bisect the exact transition second
            while hi - lo > 1:
                mid = (lo + hi) // 2
                if _utc_offset_at(tz, mid) == prev_off:
                    lo = mid
                else:
                    hi = mid
            out.append((hi, off))
            prev_off = off
        t = nxt
    return out


@dataclass
class _ZoneTable:
    index: int
    utc_instants: List[int]    # seconds
    offsets: List[int]         # seconds, offset AFTER the transition


class GpuTimeZoneDB:
    """Builds transition tables on host, ships them to the device once."""

    _instance: Optional["GpuTimeZoneDB"] = None

    def __init__(self, device="cuda"):
        self.device = device
        self._zones: Dict[str, _ZoneTable] = {}
        self._dirty = True
        self._utc_t = None
        self._local_t = None
        self._off_t = None
        self._zone_offsets = None

    @classmethod
    def instance(cls, device="cuda") -> "GpuTimeZoneDB":
        if cls._instance is None or cls._instance.device != device:
            cls._instance = GpuTimeZoneDB(device)
        return cls._instance

    def load(self, name: str) -> int:
        key = name
        if key not in self._zones:
            trans = extract_transitions(name)
            self._zones[key] = _ZoneTable(
                len(self._zones), [t for t, _ in trans], [o for _, o in trans])
            self._dirty = True
        return self._zones[key].index

    def _materialize(self):
        if not self._dirty:
            return
        utc, local, off, zoffs = [], [], [], [0]
        for z in sorted(self._zones.values(), key=lambda z: z.index):
            for i, (t, o) in enumerate(zip(z.utc_instants, z.offsets)):
                # localInstant follows the reference GpuTimeZoneDB.loadData
                # isGap/isOverlap split: for an overlap (offset decreases) the
                # boundary is utc+offsetBefore so ambiguous fall-back local
                # times resolve to the EARLIER offset; for a gap, offsetAfter.
                o_before = z.offsets[i - 1] if i > 0 else o
                lo = o_before if o < o_before else o
                utc.append(t * 1_000_000)
                local.append((t + lo) * 1_000_000)
                off.append(o)
            zoffs.append(len(utc))
        dev = self.device
        self._utc_t = torch.tensor(utc, dtype=torch.int64, device=dev)
        self._local_t = torch.tensor(local, dtype=torch.int64, device=dev)
        self._off_t = torch.tensor(off, dtype=torch.int64, device=dev)
        self._zone_offsets = torch.tensor(zoffs, dtype=torch.int32, device=dev)
        self._dirty = False

    def convert_timestamp_to_utc(self, col: Column, zone: str) -> Column:
        """Local-wall-clock micros -> UTC micros (reference timezones.hpp:28)."""
        idx = self.load(zone)
        self._materialize()
        g = _native.gpu()
        n = col.size
        out = torch.empty(n, dtype=torch.int64, device=col.device)
        g.tz_convert(col.data.data_ptr(),
                     col.validity.data_ptr() if col.validity is not None else 0,
                     n, self._utc_t.data_ptr(), self._local_t.data_ptr(),
                     self._off_t.data_ptr(), self._zone_offsets.data_ptr(),
                     idx, 1, out.data_ptr(), _native.current_stream())
        return Column(DType.TIMESTAMP_US, n, out, col.validity,
                      null_count=None)

    def convert_utc_timestamp_to_timezone(self, col: Column, zone: str) -> Column:
        idx = self.load(zone)
        self._materialize()
        g = _native.gpu()
        n = col.size
        out = torch.empty(n, dtype=torch.int64, device=col.device)
        g.tz_convert(col.data.data_ptr(),
                     col.validity.data_ptr() if col.validity is not None else 0,
                     n, self._utc_t.data_ptr(), self._local_t.data_ptr(),
                     self._off_t.data_ptr(), self._zone_offsets.data_ptr(),
                     idx, 0, out.data_ptr(), _native.current_stream())
        return Column(DType.TIMESTAMP_US, n, out, col.validity,
                      null_count=None)


# --- ORC timezone helpers (reference OrcTimezoneInfo.java /
#     OrcDstRuleExtractor.java) -------------------------------------------

@dataclass
class DstRule:
    month: int
    week: int        # 1..5, 5 = last
    day_of_week: int  # 0=Sunday (ORC convention); week == -1 means LAST
                      # occurrence in the month (DOW >= monthLength-6)
    seconds_of_day: int
    offset_after: int


def extract_dst_rules(name: str, probe_year: int = 2060) -> List[DstRule]:
    """Recover the recurring DST rule pair by probing a far-future year
    (validated the reference's way at years {2060, 2400}:
    OrcDstRuleExtractor.java:31-60)."""
    tz = zoneinfo.ZoneInfo(name)
    rules = []
    for year in (probe_year,):
        start = int((datetime.datetime(year, 1, 1,
                                       tzinfo=datetime.timezone.utc)
                     - _EPOCH).total_seconds())
        end = int((datetime.datetime(year + 1, 1, 1,
                                     tzinfo=datetime.timezone.utc)
                   - _EPOCH).total_seconds())
        prev = _utc_offset_at(tz, start)
        t = start
        while t < end:
            nxt = min(t + 3600, end)
            off = _utc_offset_at(tz, nxt)
            if off != prev:
                lo, hi = t, nxt
                while hi - lo > 1:
                    mid = (lo + hi) // 2
                    if _utc_offset_at(tz, mid) == prev:
                        lo = mid
                    else:
                        hi = mid
                local = hi + prev
                dt = _EPOCH + datetime.timedelta(seconds=local)
                # last-occurrence rules ("last Sunday of October") must not
                # encode an ordinal week — it differs between years with 4
                # vs 5 such weekdays. The reference encodes them as
                # DOW >= monthLength-6 (OrcDstRuleExtractor.decodeTransition
                # isLastOccurrence); week == -1 carries that here.
                import calendar
                mlen = calendar.monthrange(dt.year, dt.month)[1]
                week = -1 if dt.day + 7 > mlen else (dt.day - 1) // 7 + 1
                dow = (dt.weekday() + 1) % 7  # ORC: 0 = Sunday
                rules.append(DstRule(dt.month, week, dow,
                                     dt.hour * 3600 + dt.minute * 60 + dt.second,
                                     off))
                prev = off
            t = nxt
    return rules


def validate_rules_stable(name: str) -> bool:
    """Reference validation: the recurring rule must reproduce at far years."""
    a = extract_dst_rules(name, 2060)
    try:
        b = extract_dst_rules(name, 2096)  # same-rule far year
    except Exception:
        return False
    key = [(r.month, r.week, r.day_of_week, r.seconds_of_day) for r in a]
    key2 = [(r.month, r.week, r.day_of_week, r.seconds_of_day) for r in b]
    return key == key2
