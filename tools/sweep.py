"""Fusion-threshold / cycle-time sweep harness (the xGMI
re-tuning deliverable). Runs bench.py across a grid of AXON_FUSION_THRESHOLD
and AXON_CYCLE_TIME values and prints a ranked table.

    python tools/sweep.py --model resnet50 --gpus 1 --steps 10 --warmup 4
    python tools/sweep.py --thresholds 32,64,128,256 --cycles 0.5,1,2
"""
import argparse
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_one(args, thr_mb, cyc_ms):
    env = dict(os.environ)
    env["AXON_FUSION_THRESHOLD"] = str(thr_mb << 20)
    env["AXON_CYCLE_TIME"] = str(cyc_ms)
    if args.gpus > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
               "--master-port", str(29500 + os.getpid() % 500), os.path.join(REPO, "bench.py")]
    else:
        cmd = [sys.executable, os.path.join(REPO, "bench.py")]
    cmd += ["--gpus", str(args.gpus), "--steps", str(args.steps),
            "--warmup", str(args.warmup), "--model", args.model]
    if args.batch:
        cmd += ["--batch", str(args.batch)]
    r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                       timeout=args.timeout)
    for line in r.stdout.splitlines():
        if line.startswith("{"):
            try:
                return json.loads(line)
            except ValueError:
                pass
    print(f"  [sweep] thr={thr_mb}MiB cyc={cyc_ms}ms FAILED: "
          f"{r.stderr.strip().splitlines()[-1] if r.stderr.strip() else r.returncode}",
          file=sys.stderr)
    return None


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="resnet50")
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=4)
    p.add_argument("--batch", type=int, default=None)
    p.add_argument("--timeout", type=int, default=600)
    p.add_argument("--thresholds", default="32,64,128,256",
                   help="fusion thresholds to try (MiB)")
    p.add_argument("--cycles", default="0.5,1,2", help="cycle times (ms)")
    p.add_argument("--out", default=None, help="write results JSON here")
    p.add_argument("--curve", action="store_true",
                   help="scaling mode: run N=1,2,4,8 at fixed knobs and "
                        "print the whole-job curve + efficiency (VERDICT r1 "
                        "item 1b); knobs come from the first values of "
                        "--thresholds/--cycles")
    args = p.parse_args()

    if args.curve:
        thr = int(args.thresholds.split(",")[0])
        cyc = float(args.cycles.split(",")[0])
        import torch
        ndev = torch.cuda.device_count()
        curve = []
        for n in (1, 2, 4, 8):
            if n > ndev:
                print(f"  n={n}: skipped ({ndev} GPUs visible)")
                continue
            args.gpus = n
            r = run_one(args, thr, cyc)
            if r:
                curve.append({"n": n, "value": r["value"],
                              "ms_per_step": r["ms_per_step"]})
                base = curve[0]["value"]
                eff = r["value"] / (base * n) if base else 0.0
                print(f"  n={n}: {r['value']:.1f} {r['unit']} "
                      f"(eff {eff * 100:.1f}%)")
        if args.out:
            with open(args.out, "w") as f:
                json.dump(curve, f, indent=1)
        return

    results = []
    for thr in [int(x) for x in args.thresholds.split(",")]:
        for cyc in [float(x) for x in args.cycles.split(",")]:
            r = run_one(args, thr, cyc)
            if r:
                results.append({"threshold_mb": thr, "cycle_ms": cyc,
                                "value": r["value"], "unit": r["unit"],
                                "ms_per_step": r["ms_per_step"]})
                print(f"  thr={thr:4d}MiB cyc={cyc:4.1f}ms -> "
                      f"{r['value']:.1f} {r['unit']}")
    results.sort(key=lambda x: -x["value"])
    if results:
        best = results[0]
        print(f"best: thr={best['threshold_mb']}MiB cyc={best['cycle_ms']}ms "
              f"-> {best['value']:.1f} {best['unit']}")
    if args.out:
        with open(args.out, "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
