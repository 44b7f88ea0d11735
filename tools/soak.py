#!/usr/bin/env python3
"""Flake-hunting soak: run the hermetic suite N times (optionally with
randomized hypothesis seeds) and report any failures.

Usage: python tools/soak.py [--runs 10] [--random-seeds]
Exit code 0 only if every run passes.
"""

from __future__ import annotations

import argparse
import subprocess
import sys


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--runs", type=int, default=10)
    ap.add_argument("--random-seeds", action="store_true",
                    help="randomize hypothesis seeds each run")
    args = ap.parse_args()

    fails = 0
    for i in range(1, args.runs + 1):
        cmd = [sys.executable, "-m", "pytest", "tests", "-q",
               "-m", "not gpu", "-p", "no:cacheprovider"]
        if args.random_seeds:
            cmd.append("--hypothesis-seed=random")
        r = subprocess.run(cmd, capture_output=True, text=True)
        last = r.stdout.strip().splitlines()[-1] if r.stdout.strip() else "?"
        status = "ok" if r.returncode == 0 else "FAIL"
        print(f"run {i}/{args.runs}: {status} — {last}", flush=True)
        if r.returncode != 0:
            fails += 1
            print(r.stdout[-3000:])
    print(f"soak: {args.runs - fails}/{args.runs} clean")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
