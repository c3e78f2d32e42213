"""Summarize an axon chrome-trace timeline (AXON_TIMELINE / hvd.start_timeline).

Per phase (NEGOTIATE / QUEUE_GPU / CPU_OP / RCCL): span counts, total and
mean durations; per-tensor top consumers; and the comm-busy fraction
(union of data-phase spans over the traced wall time) — the number that
tells you whether collectives are overlapping compute or serializing.

    python tools/timeline_report.py /tmp/tl.0.json [/tmp/tl.1.json ...]
    python tools/timeline_report.py --json /tmp/tl.0.json
"""
import argparse
import json
import sys
from collections import defaultdict


def load_events(path):
    text = open(path).read().rstrip().rstrip(",")
    if not text.endswith("]"):
        text += "]"
    return json.loads(text)


def analyze(events):
    lanes = {}     # (pid, tid) -> tensor name
    open_spans = {}
    phases = defaultdict(lambda: [0, 0.0])   # phase -> [count, total_us]
    per_tensor = defaultdict(float)          # (tensor, phase) -> usec
    data_spans = []
    t_min, t_max = None, None
    for e in events:
        ph = e.get("ph")
        if ph == "M" and e.get("name") == "thread_name":
            lanes[(e["pid"], e["tid"])] = e["args"]["name"]
            continue
        if ph not in ("B", "E"):
            continue
        ts = e["ts"]
        t_min = ts if t_min is None else min(t_min, ts)
        t_max = ts if t_max is None else max(t_max, ts)
        key = (e["pid"], e["tid"])
        if ph == "B":
            open_spans[key] = (e["name"], ts)
        elif key in open_spans:
            name, t0 = open_spans.pop(key)
            dur = ts - t0
            phases[name][0] += 1
            phases[name][1] += dur
            per_tensor[(lanes.get(key, f"lane{key}"), name)] += dur
            if name not in ("NEGOTIATE", "QUEUE_GPU"):
                data_spans.append((t0, ts))
    # union of data-phase spans
    busy = 0.0
    cur_end = None
    for a, b in sorted(data_spans):
        if cur_end is None or a > cur_end:
            busy += b - a
            cur_end = b
        elif b > cur_end:
            busy += b - cur_end
            cur_end = b
    wall = (t_max - t_min) if (t_min is not None and t_max is not None) else 0
    return {
        "phases": {k: {"count": v[0], "total_us": round(v[1], 1),
                       "mean_us": round(v[1] / v[0], 1) if v[0] else 0.0}
                   for k, v in phases.items()},
        "top_tensors": sorted(
            ({"tensor": t, "phase": p, "total_us": round(us, 1)}
             for (t, p), us in per_tensor.items()),
            key=lambda r: -r["total_us"])[:10],
        "wall_us": wall,
        "comm_busy_us": round(busy, 1),
        "comm_busy_frac": round(busy / wall, 4) if wall else 0.0,
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("files", nargs="+")
    ap.add_argument("--json", action="store_true")
    args = ap.parse_args()
    for path in args.files:
        rep = analyze(load_events(path))
        if args.json:
            print(json.dumps({"file": path, **rep}))
            continue
        print(f"== {path} (wall {rep['wall_us'] / 1e3:.2f} ms, "
              f"comm busy {rep['comm_busy_frac'] * 100:.1f}%)")
        for ph, st in sorted(rep["phases"].items()):
            print(f"  {ph:<10} n={st['count']:<6} total={st['total_us']:>10.1f}us"
                  f"  mean={st['mean_us']:>8.1f}us")
        print("  top tensors:")
        for r in rep["top_tensors"][:5]:
            print(f"    {r['tensor'][:48]:<50} {r['phase']:<10} "
                  f"{r['total_us']:>10.1f}us")


if __name__ == "__main__":
    main()
