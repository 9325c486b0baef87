#!/usr/bin/env python3
"""Flaky-test reproducer (reference tools/flakiness_checker.py): run one
pytest node many times and report the failure rate.

    python tools/flakiness_checker.py tests/test_elastic.py::test_x -n 20
"""
import argparse
import os
import subprocess
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("test", help="pytest node id (file[::test])")
    ap.add_argument("-n", "--trials", type=int, default=10)
    ap.add_argument("--seed-env", default="DTMX_TEST_SEED",
                    help="env var set to the trial index (seeded tests)")
    args = ap.parse_args()
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    fails = 0
    for i in range(args.trials):
        env = dict(os.environ, **{args.seed_env: str(i)})
        r = subprocess.run(
            [sys.executable, "-m", "pytest", "-x", "-q", args.test],
            cwd=root, env=env, capture_output=True)
        ok = r.returncode == 0
        fails += 0 if ok else 1
        print(f"trial {i}: {'PASS' if ok else 'FAIL'}")
        if not ok:
            tail = r.stdout.decode()[-1500:]
            print(tail)
    print(f"\n{fails}/{args.trials} failures")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
