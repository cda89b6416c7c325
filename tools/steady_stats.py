"""Aggregate a rocprofv3 kernel trace over only its FINAL time window
(default last 25%), excluding the MIOpen/cudnn-benchmark autotune
find-phase that pollutes whole-run stats (round-1 VERDICT weak #7).

Usage: python tools/steady_stats.py <kernel_trace.csv> [frac] [out.md]
"""
import collections
import csv
import sys


def main():
    path = sys.argv[1]
    anchor = sys.argv[2] if len(sys.argv) > 2 else "conv1_fwd"
    n_steps = int(sys.argv[3]) if len(sys.argv) > 3 else 30
    out_md = sys.argv[4] if len(sys.argv) > 4 else None
    rows = list(csv.DictReader(open(path)))
    skey = next(k for k in rows[0] if "Start" in k)
    ekey = next(k for k in rows[0] if "End" in k)
    nkey = next(k for k in rows[0] if "Kernel_Name" in k or "Name" in k)
    # anchor the steady window on the last n_steps dispatches of a
    # once-per-step kernel: immune to autotune kernels interleaved
    # anywhere else in the trace
    anchors = sorted(
        int(r[skey]) for r in rows if anchor in r[nkey]
    )
    if len(anchors) < n_steps + 1:
        raise SystemExit(f"only {len(anchors)} {anchor!r} dispatches")
    cut = anchors[-n_steps]
    t1 = max(int(r[ekey]) for r in rows)
    t0 = cut
    frac = n_steps
    agg = collections.defaultdict(lambda: [0, 0.0])
    total = 0.0
    for r in rows:
        if int(r[skey]) < cut:
            continue
        dur = (int(r[ekey]) - int(r[skey])) / 1e6
        name = r[nkey].split("(")[0].strip()[:80]
        agg[name][0] += 1
        agg[name][1] += dur
        total += dur
    lines = [
        f"# Steady-state kernel profile: window = last {n_steps} "
        f"dispatches of '{anchor}' ({(t1 - t0) / 1e6:.0f} ms; MIOpen "
        f"autotune/find phases outside the window are EXCLUDED)",
        "",
        "| % | total ms | calls | avg ms | kernel |",
        "|---|---|---|---|---|",
    ]
    for name, (calls, ms) in sorted(agg.items(), key=lambda kv: -kv[1][1])[:25]:
        lines.append(
            f"| {ms / total * 100:5.1f} | {ms:9.2f} | {calls:5d} | "
            f"{ms / calls:7.3f} | `{name}` |"
        )
    text = "\n".join(lines) + "\n"
    print(text)
    if out_md:
        with open(out_md, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
