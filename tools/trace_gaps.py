#!/usr/bin/env python3
"""GPU busy/idle analysis of a rocprofv3 kernel trace (trace-union).

Re-derives the inter-kernel idle measurement behind ROADMAP item 5
(hipGraph capture) from a committed artifact, so the number is
reproducible:

    cd /tmp && export TMPDIR=/tmp
    rocprofv3 --kernel-trace -d out -- python bench.py --steps 15
    python tools/trace_gaps.py out/*/*_kernel_trace.csv

The busy time is the measure of the UNION of [start, end) kernel
intervals (concurrent kernels don't double-count); idle is the rest of
the [first start, last end] span.  ``--tail F`` restricts the analysis
to the last F fraction of the span (default 0.5) so bench warmup /
MIOpen find-mode noise is excluded and the number reflects the steady
state the hipGraph stepper targets.

Accepts rocprofv3 CSV kernel traces; the timestamp columns are matched
by name ("Start_Timestamp"/"End_Timestamp", case-insensitive, or any
pair containing "start"/"end" with integer values).
"""

import argparse
import csv
import sys


def _find_columns(header):
    lower = [h.strip().lower() for h in header]
    start = end = name = None
    for i, h in enumerate(lower):
        if "start" in h and "timestamp" in h:
            start = i
        elif "end" in h and "timestamp" in h:
            end = i
        elif "kernel" in h and "name" in h:
            name = i
    if start is None or end is None:
        for i, h in enumerate(lower):
            if start is None and "start" in h:
                start = i
            elif end is None and "end" in h:
                end = i
    if start is None or end is None:
        raise SystemExit("no start/end timestamp columns in: "
                         + ", ".join(header))
    return start, end, name


def _is_comm(kernel_name):
    """RCCL collective kernels (ncclDevKernel_* / rccl* on ROCm)."""
    low = kernel_name.lower()
    return "nccl" in low or "rccl" in low


def load_intervals(paths):
    """[(start_ns, end_ns, kernel_name)] across the kernel-trace CSVs."""
    intervals = []
    for path in paths:
        with open(path, newline="") as f:
            reader = csv.reader(f)
            header = next(reader)
            si, ei, ni = _find_columns(header)
            for row in reader:
                try:
                    s, e = int(row[si]), int(row[ei])
                except (ValueError, IndexError):
                    continue
                if e > s:
                    name = row[ni] if ni is not None and ni < len(row) \
                        else ""
                    intervals.append((s, e, name))
    return intervals


def union_busy(intervals):
    """Total measure of the union of [s, e) intervals."""
    busy = 0
    cur_s = cur_e = None
    for s, e in sorted(intervals):
        if cur_e is None or s > cur_e:
            if cur_e is not None:
                busy += cur_e - cur_s
            cur_s, cur_e = s, e
        else:
            cur_e = max(cur_e, e)
    if cur_e is not None:
        busy += cur_e - cur_s
    return busy


def analyze(intervals, tail=0.5, top=10):
    if not intervals:
        raise SystemExit("no kernel intervals found")
    t0 = min(s for s, _, _ in intervals)
    t1 = max(e for _, e, _ in intervals)
    cut = t1 - (t1 - t0) * tail
    window = [(max(s, cut), e, n) for s, e, n in intervals if e > cut]
    span = t1 - cut
    plain = [(s, e) for s, e, _ in window]
    busy = union_busy(plain)
    # Comm/compute overlap: how much of the RCCL collectives' wall time
    # ran concurrently with compute kernels (the xGMI backward-overlap
    # design goal; VERDICT r1 task 8).
    comm = [(s, e) for s, e, n in window if _is_comm(n)]
    comp = [(s, e) for s, e, n in window if not _is_comm(n)]
    comm_busy = union_busy(comm)
    overlap = comm_busy + union_busy(comp) - busy  # |comm ∩ comp|
    # Largest gaps inside the window (between consecutive union spans).
    gaps = []
    cur_e = None
    for s, e in sorted(plain):
        if cur_e is not None and s > cur_e:
            gaps.append((s - cur_e, cur_e))
        cur_e = e if cur_e is None else max(cur_e, e)
    gaps.sort(reverse=True)
    return {
        "span_ms": span / 1e6,
        "busy_ms": busy / 1e6,
        "idle_ms": (span - busy) / 1e6,
        "idle_pct": 100.0 * (span - busy) / span,
        "kernels": len(window),
        "comm_kernels": len(comm),
        "comm_busy_ms": comm_busy / 1e6,
        "comm_overlap_ms": overlap / 1e6,
        "comm_overlap_pct": (100.0 * overlap / comm_busy
                             if comm_busy else None),
        "top_gaps_us": [round(g / 1e3, 1) for g, _ in gaps[:top]],
    }


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("traces", nargs="+",
                    help="rocprofv3 *_kernel_trace.csv files")
    ap.add_argument("--tail", type=float, default=0.5,
                    help="analyze only the trailing fraction of the "
                         "span (steady state)")
    ap.add_argument("--top", type=int, default=10)
    args = ap.parse_args(argv)
    out = analyze(load_intervals(args.traces), args.tail, args.top)
    print("span    {span_ms:10.2f} ms  (tail window)".format(**out))
    print("busy    {busy_ms:10.2f} ms".format(**out))
    print("idle    {idle_ms:10.2f} ms  = {idle_pct:.2f}%".format(**out))
    print("kernels {kernels:10d}".format(**out))
    if out["comm_kernels"]:
        print("comm    {comm_busy_ms:10.2f} ms in {comm_kernels} RCCL "
              "kernels; {comm_overlap_ms:.2f} ms "
              "({comm_overlap_pct:.1f}%) overlapped with compute"
              .format(**out))
    print("top gaps (us):", out["top_gaps_us"])
    return out


if __name__ == "__main__":
    main()
