#!/usr/bin/env python3
"""Summarize a rocprofv3 rocpd SQLite trace into a compact two-stream
overlap report (kept under profiles/; the raw DBs are too large to
ship).  Usage: python tools/summarize_overlap.py <results.db> [out.txt]

Reports, per stream: kernel count and busy time; then computes, for
every kernel on the busiest non-main stream (the comm stream), how
much of its span overlaps kernels on the main stream — the evidence
that bucket all-reduces ran concurrently with backward compute."""

import sqlite3
import sys


def main():
    db_path = sys.argv[1]
    out_path = sys.argv[2] if len(sys.argv) > 2 else None
    db = sqlite3.connect(db_path)
    cur = db.cursor()
    tabs = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type='table'")
        if r[0].startswith('rocpd_kernel_dispatch')]
    suff = tabs[0][len('rocpd_kernel_dispatch_'):]
    rows = list(cur.execute(
        f"SELECT k.start, k.end, ks.display_name, k.stream_id "
        f"FROM rocpd_kernel_dispatch_{suff} k "
        f"JOIN rocpd_info_kernel_symbol_{suff} ks ON k.kernel_id=ks.id "
        f"ORDER BY k.start"))
    lines = []
    streams = {}
    for s, e, name, st in rows:
        streams.setdefault(st, []).append((s, e, name))
    lines.append(f"total kernels: {len(rows)}, streams: "
                 f"{sorted(streams)}")
    for st in sorted(streams):
        ks = streams[st]
        busy = sum(e - s for s, e, _ in ks) / 1e6
        names = {}
        for _, _, n in ks:
            key = n.split('(')[0][:48]
            names[key] = names.get(key, 0) + 1
        top = sorted(names.items(), key=lambda kv: -kv[1])[:6]
        lines.append(f"stream {st}: {len(ks)} kernels, busy {busy:.2f} ms"
                     f"; top: {top}")
    # overlap of EVERY side stream against the main (compute) stream
    main_st = max(streams, key=lambda st: len(streams[st]))
    mains = streams[main_st]
    t0 = rows[0][0]
    for comm_st in sorted(st for st in streams if st != main_st):
        tot, ovl = 0, 0
        per = []
        for s, e, name in streams[comm_st]:
            tot += e - s
            o = 0
            for ms, me, _ in mains:
                if me <= s:
                    continue
                if ms >= e:
                    break
                o += min(e, me) - max(s, ms)
            ovl += o
            per.append((s, e, name.split('(')[0][:40],
                        100.0 * o / max(e - s, 1)))
        lines.append(
            f"stream {comm_st} vs main {main_st}: {tot/1e6:.3f} ms "
            f"total, {ovl/1e6:.3f} ms overlapped "
            f"({100.0*ovl/max(tot,1):.1f}%)")
        if len(per) >= 10:
            lines.append(f"  last 15 kernels on stream {comm_st} "
                         "(start us, dur us, %overlapped):")
            for s, e, n, pct in per[-15:]:
                lines.append(f"  {(s-t0)/1e3:12.1f} {(e-s)/1e3:9.1f} "
                             f"{pct:5.1f}%  {n}")
    text = "\n".join(lines) + "\n"
    if out_path:
        with open(out_path, "w") as f:
            f.write(text)
    print(text)


if __name__ == "__main__":
    main()
