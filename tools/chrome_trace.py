#!/usr/bin/env python3
"""Convert a rocprofv3 kernel-trace DB into a chrome://tracing JSON of the
stream schedule.

Usage: python tools/chrome_trace.py <results.db> <out.json> [--limit N]
"""

import json
import re
import sqlite3
import sys


def table(db, like):
    for (t,) in db.execute("SELECT name FROM sqlite_master WHERE type='table'"):
        if like in t:
            return t
    return None


def main():
    path, out = sys.argv[1], sys.argv[2]
    limit = 200000
    if "--limit" in sys.argv:
        limit = int(sys.argv[sys.argv.index("--limit") + 1])
    db = sqlite3.connect(path)
    sym = table(db, "kernel_symbol")
    disp = table(db, "kernel_dispatch")
    q = (f"SELECT s.display_name, d.start, d.end, d.stream_id, d.queue_id "
         f"FROM {disp} d JOIN {sym} s ON d.kernel_id = s.id "
         f"ORDER BY d.start LIMIT {limit}")
    events = []
    t0 = None
    for name, start, end, stream, queue in db.execute(q):
        if t0 is None:
            t0 = start
        events.append({
            "name": re.sub(r"void |dvgg::|\(.*", "", name)[:80],
            "ph": "X",
            "ts": (start - t0) / 1e3,           # us
            "dur": (end - start) / 1e3,
            "pid": 0,
            "tid": int(stream if stream is not None else queue or 0),
            "cat": "kernel",
        })
    with open(out, "w") as fh:
        json.dump({"traceEvents": events,
                   "displayTimeUnit": "ms"}, fh)
    print(f"wrote {out} ({len(events)} events)")


if __name__ == "__main__":
    main()
