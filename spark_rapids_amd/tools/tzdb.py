"""Timezone transition database (reference analogue: spark-rapids-jni
GpuTimeZoneDB — SURVEY.md §2.8B: "transition table on device for non-UTC
timestamps").

Parses the binary TZif v2/v3 files shipped with the python `tzdata`
package (or /usr/share/zoneinfo when present) into two arrays per zone:
transition instants (UTC seconds, int64) and the UTC offset in seconds
that applies FROM each instant (int32). Conversion is then a binary
search per row — numpy searchsorted on the CPU backend, the k_tz_convert
kernel on device.
"""
from __future__ import annotations

import os
import struct
from typing import Dict, Tuple

import numpy as np

_CACHE: Dict[str, Tuple[np.ndarray, np.ndarray]] = {}


def _tzif_path(zone: str) -> str:
    if "/" in zone and ".." in zone:
        raise ValueError(f"bad zone name {zone!r}")
    sys_path = os.path.join("/usr/share/zoneinfo", *zone.split("/"))
    if os.path.exists(sys_path):
        return sys_path
    import tzdata

    p = os.path.join(os.path.dirname(tzdata.__file__), "zoneinfo",
                     *zone.split("/"))
    if not os.path.exists(p):
        raise ValueError(f"unknown timezone {zone!r}")
    return p


def load(zone: str) -> Tuple[np.ndarray, np.ndarray]:
    """(transitions int64 seconds-UTC, offsets int32 seconds). offsets[i]
    applies from transitions[i] (transitions[0] = -inf sentinel)."""
    hit = _CACHE.get(zone)
    if hit is not None:
        return hit
    raw = open(_tzif_path(zone), "rb").read()
    if raw[:4] != b"TZif":
        raise ValueError(f"{zone}: not a TZif file")
    version = raw[4:5]

    def parse_header(buf, off):
        (isutcnt, isstdcnt, leapcnt, timecnt, typecnt,
         charcnt) = struct.unpack_from(">6I", buf, off + 20)
        return isutcnt, isstdcnt, leapcnt, timecnt, typecnt, charcnt

    h1 = parse_header(raw, 0)
    # v1 data block size (32-bit times)
    v1_size = 44 + h1[3] * 5 + h1[4] * 6 + h1[5] + h1[2] * 8 \
        + h1[1] + h1[0]
    if version in (b"2", b"3"):
        off = v1_size
        h2 = parse_header(raw, off)
        off += 44
        timecnt, typecnt = h2[3], h2[4]
        trans = np.frombuffer(raw, dtype=">i8", count=timecnt,
                              offset=off).astype(np.int64)
        off += timecnt * 8
        idx = np.frombuffer(raw, dtype=np.uint8, count=timecnt, offset=off)
        off += timecnt
        utoffs = np.empty(typecnt, dtype=np.int32)
        for t in range(typecnt):
            (uo,) = struct.unpack_from(">i", raw, off + t * 6)
            utoffs[t] = uo
        # the offset before the first transition is type 0 by convention
        transitions = np.concatenate(
            [np.array([np.iinfo(np.int64).min], dtype=np.int64), trans])
        offsets = np.concatenate(
            [utoffs[:1], utoffs[idx]]).astype(np.int32)
    else:
        timecnt, typecnt = h1[3], h1[4]
        off = 44
        trans = np.frombuffer(raw, dtype=">i4", count=timecnt,
                              offset=off).astype(np.int64)
        off += timecnt * 4
        idx = np.frombuffer(raw, dtype=np.uint8, count=timecnt, offset=off)
        off += timecnt
        utoffs = np.empty(typecnt, dtype=np.int32)
        for t in range(typecnt):
            (uo,) = struct.unpack_from(">i", raw, off + t * 6)
            utoffs[t] = uo
        transitions = np.concatenate(
            [np.array([np.iinfo(np.int64).min], dtype=np.int64), trans])
        offsets = np.concatenate(
            [utoffs[:1], utoffs[idx]]).astype(np.int32)
    # "slim" tzdata: recurring DST lives in the POSIX TZ footer (e.g.
    # "EST5EDT,M3.2.0,M11.1.0"); expand it into explicit transitions
    if version in (b"2", b"3"):
        nl = raw.rfind(b"\n", 0, len(raw) - 1)
        footer = raw[nl + 1:-1].decode("ascii", "ignore").strip()
        if footer:
            last = int(transitions[-1]) if len(transitions) > 1 else 0
            ext_t, ext_o = _expand_posix_tz(footer, last)
            if len(ext_t):
                transitions = np.concatenate([transitions, ext_t])
            if len(ext_o):
                offsets = np.concatenate([offsets, ext_o]).astype(np.int32)
    _CACHE[zone] = (transitions, offsets)
    return _CACHE[zone]


def _parse_posix_offset(s: str, i: int):
    """[+-]hh[:mm[:ss]] -> (seconds WEST-positive reversed to utoff, next i)."""
    sign = 1
    if i < len(s) and s[i] in "+-":
        sign = -1 if s[i] == "-" else 1
        i += 1
    parts = [0, 0, 0]
    for k in range(3):
        j = i
        while j < len(s) and s[j].isdigit():
            j += 1
        if j == i:
            break
        parts[k] = int(s[i:j])
        i = j
        if i < len(s) and s[i] == ":":
            i += 1
        else:
            break
    secs = parts[0] * 3600 + parts[1] * 60 + parts[2]
    # POSIX offsets are west-positive; utoff is east-positive
    return -sign * secs, i


def _mwd_to_days(year: int, m: int, w: int, d: int) -> int:
    """days-since-epoch of the w-th (5=last) weekday-d of month m."""
    import calendar
    from datetime import date

    first = date(year, m, 1)
    shift = (d - first.weekday() - 1) % 7  # date.weekday(): Mon=0; d: Sun=0
    day = 1 + shift
    if w == 5:
        last_dom = calendar.monthrange(year, m)[1]
        while day + 7 <= last_dom:
            day += 7
    else:
        day += 7 * (w - 1)
    return (date(year, m, day) - date(1970, 1, 1)).days


def _expand_posix_tz(tz: str, after: int, until_year: int = 2100):
    """Expand 'STDoff[DST[off]],Mm.w.d[/t],Mm.w.d[/t]' into transitions
    strictly after `after`. Unsupported forms expand to nothing (the
    explicit table still covers history)."""
    import re

    name = r"(?:<[^>]+>|[A-Za-z]+)"
    m = re.match(
        rf"^{name}([+-]?\d+(?::\d+){{0,2}})"
        rf"(?:{name}([+-]?\d+(?::\d+){{0,2}})?"
        r",(M\d+\.\d+\.\d+)(?:/(-?\d+(?::\d+){0,2}))?"
        r",(M\d+\.\d+\.\d+)(?:/(-?\d+(?::\d+){0,2}))?)?$", tz)
    if not m:
        return np.zeros(0, dtype=np.int64), np.zeros(0, dtype=np.int32)
    std_off, _ = _parse_posix_offset(m.group(1), 0)
    if not m.group(3):
        return np.zeros(0, dtype=np.int64), np.zeros(0, dtype=np.int32)
    dst_off = std_off + 3600
    if m.group(2):
        dst_off, _ = _parse_posix_offset(m.group(2), 0)

    def rule_time(g):
        if not g:
            return 2 * 3600
        neg = g.startswith("-")
        hms = [int(x) for x in g.lstrip("-").split(":")] + [0, 0]
        t = hms[0] * 3600 + hms[1] * 60 + hms[2]
        return -t if neg else t

    def rule_parts(g):
        mm, ww, dd = (int(x) for x in g[1:].split("."))
        return mm, ww, dd

    sm, sw, sd = rule_parts(m.group(3))
    st = rule_time(m.group(4))
    em, ew, ed = rule_parts(m.group(5))
    et = rule_time(m.group(6))
    trans, offs = [], []
    from datetime import datetime, timezone

    y0 = max(1970, datetime.fromtimestamp(max(after, 0),
                                          tz=timezone.utc).year)
    for year in range(y0, until_year):
        start_utc = _mwd_to_days(year, sm, sw, sd) * 86400 + st - std_off
        end_utc = _mwd_to_days(year, em, ew, ed) * 86400 + et - dst_off
        for t, o in sorted([(start_utc, dst_off), (end_utc, std_off)]):
            if t > after:
                trans.append(t)
                offs.append(o)
    return (np.array(trans, dtype=np.int64),
            np.array(offs, dtype=np.int32))


def offset_at(zone: str, utc_seconds: np.ndarray) -> np.ndarray:
    """UTC offset (seconds) in force at each UTC instant."""
    trans, offs = load(zone)
    pos = np.searchsorted(trans, utc_seconds, side="right") - 1
    return offs[np.clip(pos, 0, len(offs) - 1)]
