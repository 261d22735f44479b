#!/usr/bin/env python3
"""Integrity sweep: every BASELINE workload once, WITH validation, at a
moderate size — the one-command correctness check for a fresh MI355X box.

  python scripts/validate_all.py [gb_per_gpu]
"""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

RUNS = [
    ["--workload", "terasort", "--record-bytes", "100", "--validate"],
    ["--workload", "terasort", "--record-bytes", "16", "--validate"],
    ["--workload", "terasort", "--mode", "rccl", "--record-bytes", "100",
     "--validate"],
    ["--workload", "join", "--validate"],
    ["--workload", "reducebykey", "--validate"],
    ["--workload", "pagerank"],      # validated vs dense in tests
    ["--workload", "groupby"],
]


def main():
    gb = sys.argv[1] if len(sys.argv) > 1 else "2"
    failures = []
    for extra in RUNS:
        cmd = [sys.executable, os.path.join(REPO, "bench.py"),
               "--steps", "1", "--warmup", "0", "--gb-per-gpu", gb] + extra
        print("==", " ".join(extra), flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True, timeout=900)
        line = next((l for l in r.stdout.splitlines()
                     if l.startswith("{")), "")
        if r.returncode != 0 or not line:
            failures.append((extra, r.stderr[-1500:]))
            print("   FAILED", flush=True)
        else:
            print("   ok:", line[:120], flush=True)
    if failures:
        for extra, err in failures:
            print("FAILURE", extra, err, file=sys.stderr)
        sys.exit(1)
    print(f"validate_all: {len(RUNS)} workload runs validated")


if __name__ == "__main__":
    main()
