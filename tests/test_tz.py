"""Timezone subsystem tests (transition extraction CPU + device conversion)."""
import datetime
import zoneinfo

import pytest
import torch

from spark_rapids_jni_amd.columnar import Column, DType

LA = "America/Los_Angeles"
EPOCH = datetime.datetime(1970, 1, 1, tzinfo=datetime.timezone.utc)


def _us(y, mo, d, h=0, mi=0, s=0):
    t = datetime.datetime(y, mo, d, h, mi, s, tzinfo=datetime.timezone.utc)
    return int((t - EPOCH).total_seconds()) * 1_000_000


def test_extract_transitions_la():
    from spark_rapids_jni_amd.tz import extract_transitions
    trans = extract_transitions(LA, until_year=2030)
    # 2021 DST: Mar 14 10:00 UTC (-> PDT), Nov 7 09:00 UTC (-> PST)
    secs = {t for t, _ in trans}
    assert _us(2021, 3, 14, 10) // 10**6 in secs
    assert _us(2021, 11, 7, 9) // 10**6 in secs
    offs = dict(trans)
    assert offs[_us(2021, 3, 14, 10) // 10**6] == -7 * 3600
    assert offs[_us(2021, 11, 7, 9) // 10**6] == -8 * 3600


def test_extract_dst_rules():
    from spark_rapids_jni_amd.tz import extract_dst_rules, validate_rules_stable
    rules = extract_dst_rules(LA)
    assert len(rules) == 2
    months = sorted(r.month for r in rules)
    assert months == [3, 11]
    assert validate_rules_stable(LA)


@pytest.mark.gpu
def test_tz_convert_device():
    from spark_rapids_jni_amd.tz import GpuTimeZoneDB
    db = GpuTimeZoneDB.instance()
    tz = zoneinfo.ZoneInfo(LA)
    cases_utc = [
        _us(2021, 1, 15, 12), _us(2021, 7, 15, 12),
        _us(2021, 3, 14, 9, 59), _us(2021, 3, 14, 10, 1),
        _us(1999, 12, 31, 23, 59), _us(2085, 6, 1),
    ]
    col = Column.from_pylist(cases_utc, DType.TIMESTAMP_US, "cuda")
    local = db.convert_utc_timestamp_to_timezone(col, LA).to_pylist()
    for u, lv in zip(cases_utc, local):
        dt = EPOCH + datetime.timedelta(microseconds=u)
        exp = u + int(dt.astimezone(tz).utcoffset().total_seconds()) * 10**6
        assert lv == exp, datetime.datetime.utcfromtimestamp(u / 1e6)
    # roundtrip local -> utc (unambiguous times)
    lcol = Column.from_pylist(local, DType.TIMESTAMP_US, "cuda")
    back = db.convert_timestamp_to_utc(lcol, LA).to_pylist()
    assert back == cases_utc


def test_tzdb_more_zones_against_zoneinfo():
    """Cross-check the built transition tables against zoneinfo for zones
    spanning both hemispheres, half-hour offsets and historic changes."""
    import bisect
    import datetime
    from zoneinfo import ZoneInfo
    from spark_rapids_jni_amd.tz import extract_transitions
    zones = ["Australia/Sydney", "America/Sao_Paulo", "Asia/Kathmandu",
             "Africa/Cairo", "Pacific/Auckland"]
    probes = [datetime.datetime(y, m, d, h) for (y, m, d, h) in
              [(1999, 6, 1, 12), (2005, 12, 31, 23), (2012, 3, 11, 2),
               (2019, 10, 6, 3), (2024, 1, 15, 0)]]
    for z in zones:
        try:
            zi = ZoneInfo(z)
        except Exception:
            continue
        trans = extract_transitions(z)  # [(utc_second, offset_after)]
        instants = [t[0] for t in trans]
        for dt in probes:
            utc = dt.replace(tzinfo=datetime.timezone.utc)
            ts = int(utc.timestamp())
            i = bisect.bisect_right(instants, ts) - 1
            expect = utc.astimezone(zi).utcoffset().total_seconds()
            assert trans[max(i, 0)][1] == expect, (z, dt)


def test_overlap_local_instant_uses_offset_before():
    # ADVICE fix: reference GpuTimeZoneDB.loadData stores local = utc +
    # offsetBefore for overlap transitions so ambiguous fall-back local
    # times resolve to the earlier offset; gaps keep offsetAfter.
    from spark_rapids_jni_amd.tz import GpuTimeZoneDB
    db = GpuTimeZoneDB(device="cpu")
    db.load("America/New_York")
    db._materialize()
    utc = db._utc_t.tolist()
    local = db._local_t.tolist()
    off = db._off_t.tolist()
    # 2024-11-03 06:00 UTC: EDT(-4h) -> EST(-5h), overlap
    fall = int(datetime.datetime(2024, 11, 3, 6, tzinfo=datetime.timezone.utc)
               .timestamp()) * 1_000_000
    i = utc.index(fall)
    assert off[i] == -5 * 3600
    assert off[i - 1] == -4 * 3600
    assert local[i] == fall + off[i - 1] * 1_000_000  # offsetBefore
    # 2024-03-10 07:00 UTC: EST(-5h) -> EDT(-4h), gap
    spring = int(datetime.datetime(2024, 3, 10, 7, tzinfo=datetime.timezone.utc)
                 .timestamp()) * 1_000_000
    j = utc.index(spring)
    assert off[j] == -4 * 3600
    assert local[j] == spring + off[j] * 1_000_000  # offsetAfter


def test_ambiguous_local_time_resolution_matches_java():
    # During the repeated 1:00-2:00 AM hour on 2024-11-03 in New York the
    # earlier offset (EDT, -4h) must win, matching java.time and Spark.
    import bisect
    from spark_rapids_jni_amd.tz import GpuTimeZoneDB
    db = GpuTimeZoneDB(device="cpu")
    db.load("America/New_York")
    db._materialize()
    local = db._local_t.tolist()
    off = db._off_t.tolist()

    def to_utc(local_us):
        k = bisect.bisect_right(local, local_us) - 1
        return local_us - off[k] * 1_000_000

    # 01:30 local on 2024-11-03 is ambiguous; java resolves to EDT (-4h)
    amb = int(datetime.datetime(2024, 11, 3, 1, 30).replace(
        tzinfo=datetime.timezone.utc).timestamp()) * 1_000_000
    expect = amb + 4 * 3600 * 1_000_000  # EDT: utc = local + 4h
    assert to_utc(amb) == expect


def test_extract_dst_rules_hemispheres_and_no_dst():
    """Reference OrcDstRuleExtractor tests: northern (Europe/London) and
    southern (Australia/Sydney) hemisphere rules, and zones without DST
    return no rules; rules stay stable at the reference's probe years."""
    from spark_rapids_jni_amd.tz import extract_dst_rules, validate_rules_stable
    lon = extract_dst_rules("Europe/London")
    assert len(lon) == 2 and sorted(r.month for r in lon) == [3, 10]
    # both London transitions are LAST-Sunday rules -> week -1
    assert all(r.week == -1 and r.day_of_week == 0 for r in lon)
    syd = extract_dst_rules("Australia/Sydney")
    assert len(syd) == 2 and sorted(r.month for r in syd) == [4, 10]
    assert extract_dst_rules("Asia/Shanghai") == []
    assert extract_dst_rules("UTC") == []
    for z in ("Europe/London", "Australia/Sydney"):
        assert validate_rules_stable(z)
