#!/usr/bin/env python3
"""Daily regression driver (the reference's
curvine-tests/regression/daily_regression_test.sh analog): one command
runs the full CPU test suite, the fio conformance sweep over a real
mount, the metadata QPS bench and the paged-namespace bench, and emits
a single JSON verdict (plus junit XML for CI consumption).

Usage: python scripts/daily_regression.py [--quick] [--out FILE]
Exit code 0 = every stage passed.
"""
import argparse
import json
import os
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_stage(name: str, cmd: list[str], timeout: int, results: dict,
              parse_json: bool = False) -> bool:
    t0 = time.time()
    try:
        r = subprocess.run(cmd, cwd=REPO, capture_output=True, text=True,
                           timeout=timeout)
        ok = r.returncode == 0
        stage = {"ok": ok, "seconds": round(time.time() - t0, 1)}
        if parse_json:
            for line in reversed(r.stdout.strip().split("\n")):
                line = line.strip()
                if line.startswith("{"):
                    try:
                        stage["result"] = json.loads(line)
                        break
                    except json.JSONDecodeError:
                        continue
        if not ok:
            stage["tail"] = (r.stdout + r.stderr)[-2000:]
    except subprocess.TimeoutExpired:
        stage = {"ok": False, "seconds": round(time.time() - t0, 1),
                 "tail": "TIMEOUT"}
        ok = False
    results["stages"][name] = stage
    print(f"[{name}] {'PASS' if ok else 'FAIL'} "
          f"({stage['seconds']}s)", flush=True)
    return ok


def main() -> int:
    p = argparse.ArgumentParser()
    p.add_argument("--quick", action="store_true",
                   help="smaller bench sizes")
    p.add_argument("--out", default="")
    p.add_argument("--junit", default="")
    args = p.parse_args()

    results = {"started": time.strftime("%Y-%m-%dT%H:%M:%S"),
               "stages": {}}
    ok = True
    pytest_cmd = [sys.executable, "-m", "pytest", "tests/", "-q",
                  "-m", "not gpu", "--timeout", "600"]
    if args.junit:
        pytest_cmd += ["--junitxml", args.junit]
    ok &= run_stage("pytest_cpu", pytest_cmd, 1200, results)

    fio = [sys.executable, "scripts/fio_sweep.py",
           "--files", "2" if args.quick else "4",
           "--file-size", str((64 if args.quick else 128) << 20),
           "--rand4k", "4000" if args.quick else "20000"]
    ok &= run_stage("fio_sweep", fio, 900, results, parse_json=True)

    meta = [sys.executable, "scripts/meta_bench.py",
            "--n", "5000" if args.quick else "20000",
            "--concurrency", "40", "--procs", "4"]
    ok &= run_stage("meta_bench", meta, 900, results, parse_json=True)

    paged = [sys.executable, "scripts/paged_meta_bench.py",
             "--n", "20000" if args.quick else "100000",
             "--resident", "2000" if args.quick else "10000",
             "--k", "4000" if args.quick else "10000", "--paged-only"]
    ok &= run_stage("paged_namespace", paged, 900, results,
                    parse_json=True)

    results["ok"] = bool(ok)
    out = json.dumps(results)
    print(out)
    if args.out:
        with open(args.out, "w") as f:
            f.write(out + "\n")
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
