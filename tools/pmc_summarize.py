#!/usr/bin/env python3
"""Aggregate rocprofv3 output into small per-kernel summaries.

    python tools/pmc_summarize.py <rocprof_out_dir> <summary.json>

Handles both --stats kernel-stats CSVs and --pmc counter-collection CSVs
(any column-name variant rocprofv3 emits): groups by kernel name, sums
counter values / durations, counts dispatches.  Raw per-dispatch CSVs
can then be deleted (they exceed the artifact budget).
"""

import csv
import glob
import json
import os
import sys
from collections import defaultdict


def norm(row):
    return {k.strip().lower().replace("-", "_").replace(" ", "_"): v
            for k, v in row.items() if k}


def kname(d):
    for k in ("kernel_name", "name", "kernel"):
        if k in d:
            return d[k].strip().strip('"')
    return None


def main(out_dir, dest):
    summary = {"counters": {}, "kernel_stats": {}}
    for path in glob.glob(os.path.join(out_dir, "**", "*.csv"),
                          recursive=True):
        base = os.path.basename(path).lower()
        try:
            with open(path, newline="") as f:
                rows = [norm(r) for r in csv.DictReader(f)]
        except Exception:
            continue
        if not rows:
            continue
        if "counter" in base:
            agg = defaultdict(lambda: defaultdict(float))
            disp = defaultdict(set)
            for d in rows:
                k = kname(d)
                cn = d.get("counter_name")
                cv = d.get("counter_value")
                if not k or not cn:
                    continue
                try:
                    agg[k][cn.strip()] += float(cv)
                except (TypeError, ValueError):
                    continue
                did = d.get("dispatch_id") or d.get("correlation_id")
                if did:
                    disp[k].add(did)
            for k, cs in agg.items():
                ent = summary["counters"].setdefault(
                    k, {"dispatches": 0, "sums": defaultdict(float)}
                )
                ent["dispatches"] += len(disp.get(k, ()))
                for cn, v in cs.items():
                    ent["sums"][cn] = ent["sums"].get(cn, 0.0) + v
        elif "kernel_trace" in base:
            # per-dispatch timestamps: steady-state stats from the second
            # half of the trace window (excludes MIOpen find-mode warmup)
            byk = defaultdict(list)
            t0, t1 = None, None
            for d in rows:
                k = kname(d)
                try:
                    s0 = float(d.get("start_timestamp") or d.get("start") or 0)
                    e0 = float(d.get("end_timestamp") or d.get("end") or 0)
                except (TypeError, ValueError):
                    continue
                if not k or e0 <= s0:
                    continue
                byk[k].append((s0, e0 - s0))
                t0 = s0 if t0 is None else min(t0, s0)
                t1 = e0 if t1 is None else max(t1, e0)
            if t0 is None:
                continue
            mid = t0 + 0.5 * (t1 - t0)
            ss = summary.setdefault("steady_state", {})
            for k, evs in byk.items():
                late = sorted(dur for s0, dur in evs if s0 >= mid)
                if not late:
                    continue
                n = len(late)
                ss[k] = {
                    "dispatches": n,
                    "total_us": sum(late) / 1e3,
                    "p50_us": late[n // 2] / 1e3,
                    "p95_us": late[min(n - 1, int(n * 0.95))] / 1e3,
                }
        elif "kernel_stats" in base or ("stats" in base and "domain" not in base):
            for d in rows:
                k = kname(d)
                if not k:
                    continue
                ent = summary["kernel_stats"].setdefault(
                    k, {"calls": 0, "total_ns": 0.0}
                )
                try:
                    ent["calls"] += int(float(d.get("calls", 0) or 0))
                    ent["total_ns"] += float(
                        d.get("totaldurationns")
                        or d.get("total_duration_ns")
                        or d.get("durationns")
                        or d.get("total_time")
                        or 0
                    )
                except (TypeError, ValueError):
                    continue
    # plain dicts for json
    for k, ent in summary["counters"].items():
        ent["sums"] = dict(ent["sums"])
    os.makedirs(os.path.dirname(dest) or ".", exist_ok=True)
    with open(dest, "w") as f:
        json.dump(summary, f, indent=1)
    print(
        f"wrote {dest}: {len(summary['counters'])} counter kernels, "
        f"{len(summary['kernel_stats'])} stats kernels"
    )
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1], sys.argv[2]))
