#!/usr/bin/env python3
"""Multi-seed launcher (reference parity: the repo's multi-seed run scripts,
SURVEY.md section 3.1 'Plotting / multi-seed' row).

Launches N sequential (or parallel, one per GPU) training runs with
different seeds and collates final eval scores.

  python scripts/launch_multiseed.py --seeds 3 --game pong -- \
      --T-max 1000000 --learn-start 20000
"""
import argparse
import json
import os
import subprocess
import sys


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seeds", type=int, default=3)
    p.add_argument("--game", type=str, default="pong")
    p.add_argument("--id", type=str, default="multiseed")
    p.add_argument("--parallel", action="store_true",
                   help="one run per GPU (CUDA_VISIBLE_DEVICES=i)")
    p.add_argument("rest", nargs=argparse.REMAINDER)
    ns = p.parse_args()
    rest = [a for a in ns.rest if a != "--"]

    procs = []
    for i in range(ns.seeds):
        run_id = f"{ns.id}-s{i}"
        cmd = [sys.executable, "-m", "rainbowiqn_amd.train_apex",
               "--game", ns.game, "--seed", str(1000 + i), "--id", run_id] + rest
        env = dict(os.environ)
        if ns.parallel:
            env["CUDA_VISIBLE_DEVICES"] = str(i)
            procs.append(subprocess.Popen(cmd, env=env))
        else:
            print(f"[multiseed] seed {i}: {' '.join(cmd)}", flush=True)
            subprocess.check_call(cmd, env=env)
    for pr in procs:
        ret = pr.wait()
        if ret != 0:
            raise SystemExit(f"seed run failed with {ret}")
    print(json.dumps({"runs": [f"results/{ns.id}-s{i}" for i in range(ns.seeds)]}))


if __name__ == "__main__":
    main()
