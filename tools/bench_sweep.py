#!/usr/bin/env python3
"""Serving bench sweep driver (avoids shell-quoting hazards on gpurun).

Usage: python tools/bench_sweep.py [--out gpurun_out/sweep.log]
Runs bench.py across session counts / engine counts and prints one line
per config: value, p50/p99 RTT, batches.
"""
import argparse
import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def run(flags):
    cmd = [sys.executable, str(REPO / "bench.py")] + flags
    p = subprocess.run(cmd, capture_output=True, text=True, timeout=300,
                       cwd=str(REPO))
    line = p.stdout.strip().splitlines()[-1] if p.stdout.strip() else ""
    try:
        d = json.loads(line)
    except Exception:
        return f"FAIL rc={p.returncode}: {p.stderr.strip()[-200:]}"
    c = d["config"]
    rtt = (f' p50={c.get("p50_rtt_ms")}ms p99={c.get("p99_rtt_ms")}ms'
           if "p50_rtt_ms" in c else f' step_p50={c.get("ms_per_step_p50")}ms')
    es = c.get("engine_stats", {})
    return (f'{d["value"]:,.0f} req/s{rtt} batches={es.get("batches")}'
            f' fallbacks={es.get("hostFallbacks")}')


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="")
    ap.add_argument("--steps", default="100")
    args = ap.parse_args()
    lines = []

    def log(s):
        print(s, flush=True)
        lines.append(s)

    base = ["--steps", args.steps, "--warmup", "20"]
    for b, t in ((256, 8), (512, 8), (1024, 16), (2048, 16), (4096, 32)):
        log(f"sessions={b}: " + run(base + ["--batch", str(b),
                                            "--client-threads", str(t)]))
    for s in (2, 4, 6, 8):
        log(f"streams={s} (1024 sessions): " + run(base + ["--streams", str(s)]))
    if args.out:
        Path(args.out).write_text("\n".join(lines) + "\n")


if __name__ == "__main__":
    main()
