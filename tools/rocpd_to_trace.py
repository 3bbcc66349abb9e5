#!/usr/bin/env python3
"""Convert a rocprofv3 result database into a Chrome/Perfetto trace.

Counterpart of the reference's timeline tooling (ref: xpu_timer/py_xpu_timer/
gen_trace_timeline.py:519 — perfetto conversion of xpu_timer dumps). On
MI355X the kernel timeline comes from rocprofv3's rocpd SQLite output
(`rocprofv3 --kernel-trace -d out -- cmd` -> out/<host>/<pid>_results.db);
this tool emits chrome://tracing / ui.perfetto.dev compatible JSON.

Usage: python tools/rocpd_to_trace.py results.db trace.json [--top 0]
"""

import argparse
import json
import sqlite3


def convert(db_path: str, out_path: str, top: int = 0):
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    rows = cur.execute(
        "SELECT name, start, end, queue_id, stream_id, grid_x*grid_y*grid_z, "
        "workgroup_x*workgroup_y*workgroup_z, vgpr_count, lds_size "
        "FROM kernels ORDER BY start"
    ).fetchall()
    if not rows:
        raise SystemExit("no kernel dispatches in the database")
    t0 = min(r[1] for r in rows)
    events = []
    for name, start, end, queue, stream, grid, wg, vgpr, lds in rows:
        events.append(
            {
                "name": name.split("(")[0][:120],
                "ph": "X",
                "ts": (start - t0) / 1e3,  # ns -> us
                "dur": max((end - start) / 1e3, 0.01),
                "pid": 0,
                "tid": int(stream or queue or 0),
                "args": {
                    "grid": grid,
                    "workgroup": wg,
                    "vgpr": vgpr,
                    "lds": lds,
                },
            }
        )
    if top:
        events.sort(key=lambda e: -e["dur"])
        events = events[:top]
        events.sort(key=lambda e: e["ts"])
    with open(out_path, "w") as f:
        json.dump(
            {
                "traceEvents": events,
                "displayTimeUnit": "ms",
                "metadata": {"source": db_path},
            },
            f,
        )
    total_ms = sum(e["dur"] for e in events) / 1e3
    print(
        f"{len(events)} kernel events -> {out_path} "
        f"(sum of durations {total_ms:.1f} ms)"
    )


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("db")
    p.add_argument("out")
    p.add_argument("--top", type=int, default=0, help="keep only the N longest")
    a = p.parse_args()
    convert(a.db, a.out, a.top)
