"""Run every test file in its own process with a timeout (reference
``tests/run_all.py:29-70`` — process isolation avoids GPU-memory conflicts
between files).

  python tests/run_all.py [-m "not gpu"] [--timeout 600]
"""
import argparse
import glob
import os
import subprocess
import sys


def main():
    p = argparse.ArgumentParser()
    p.add_argument("-m", "--marker", default="not gpu")
    p.add_argument("--timeout", type=int, default=900)
    p.add_argument("-k", default=None)
    args = p.parse_args()
    here = os.path.dirname(os.path.abspath(__file__))
    files = sorted(glob.glob(os.path.join(here, "test_*.py")))
    failed = []
    for f in files:
        cmd = [sys.executable, "-m", "pytest", f, "-q", "-m", args.marker]
        if args.k:
            cmd += ["-k", args.k]
        print(f"=== {os.path.basename(f)} ===", flush=True)
        try:
            r = subprocess.run(cmd, timeout=args.timeout)
            if r.returncode not in (0, 5):  # 5 = no tests collected
                failed.append(os.path.basename(f))
        except subprocess.TimeoutExpired:
            print(f"TIMEOUT {f}")
            failed.append(os.path.basename(f))
    if failed:
        print("FAILED:", ", ".join(failed))
        sys.exit(1)
    print("all test files passed")


if __name__ == "__main__":
    main()
