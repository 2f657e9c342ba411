"""Structured log protocol + k-way timestamp merge.

Parity target: /root/reference/metaflow/mflog/mflog.py (decorate :62,
parse :80, merge_logs :124). Line format:

    [MFX|1|<utc-iso>|<source>|<logical_counter>]<message>

The scheduler decorates captured child output before persisting, so logs
from many tasks can be merged chronologically by the client/CLI.
"""

import heapq
import re
from collections import namedtuple
from datetime import datetime, timezone

VERSION = "1"
_LINE_RE = re.compile(
    rb"^\[MFX\|(?P<version>[^|]+)\|(?P<ts>[^|]+)\|(?P<source>[^|]*)\|"
    rb"(?P<counter>\d+)\](?P<msg>.*)$", re.DOTALL)

MFLogline = namedtuple("MFLogline", ["version", "ts", "source", "counter",
                                     "msg"])


def utc_now_str():
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%S.%f")


def decorate(source, msg, counter=0, ts=None):
    """Wrap one message (str or bytes) as a structured line (bytes, no
    trailing newline)."""
    if isinstance(msg, str):
        msg = msg.encode("utf-8", "replace")
    head = "[MFX|%s|%s|%s|%d]" % (VERSION, ts or utc_now_str(), source,
                                  counter)
    return head.encode() + msg


def parse(line):
    """Parse a structured line (bytes); returns MFLogline or None."""
    if isinstance(line, str):
        line = line.encode("utf-8", "replace")
    m = _LINE_RE.match(line.rstrip(b"\n"))
    if not m:
        return None
    return MFLogline(
        m.group("version").decode(),
        m.group("ts").decode(),
        m.group("source").decode(),
        int(m.group("counter")),
        m.group("msg").decode("utf-8", "replace"),
    )


def decorate_stream(source, data):
    """Decorate every line of a raw byte blob; returns bytes."""
    out = []
    for i, line in enumerate(data.split(b"\n")):
        if line:
            out.append(decorate(source, line, counter=i))
    return b"\n".join(out) + (b"\n" if out else b"")


def merge_logs(streams):
    """k-way merge of structured log byte-streams by (ts, counter).

    streams: iterable of bytes. Yields MFLogline in chronological order;
    unparseable lines get ts='' and sort first within their stream order.
    """
    iters = []
    for data in streams:
        lines = []
        for raw in data.split(b"\n"):
            if not raw:
                continue
            parsed = parse(raw)
            if parsed is None:
                parsed = MFLogline(VERSION, "", "?", 0,
                                   raw.decode("utf-8", "replace"))
            lines.append(parsed)
        iters.append(iter(lines))
    return heapq.merge(*iters, key=lambda l: (l.ts, l.source, l.counter))
