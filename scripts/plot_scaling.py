#!/usr/bin/env python3
"""Render the multi-GPU scaling curve from bench JSON lines (e.g. the
driver's SCALE_rNN.json or saved bench outputs).

  python scripts/plot_scaling.py out.html bench_n1.json bench_n2.json ...

Each input file holds one bench.py JSON line; output is a self-contained
HTML/SVG chart of aggregate frames/s and grad-steps/s vs GPU count.
"""
import json
import sys


def main():
    out, files = sys.argv[1], sys.argv[2:]
    pts = []
    for f in files:
        with open(f) as fh:
            for line in fh:
                line = line.strip()
                if line.startswith("{"):
                    d = json.loads(line)
                    pts.append((d["n_gpus"], d["value"],
                                d.get("learner_grad_steps_per_sec", 0.0)))
    pts.sort()
    from rainbowiqn_amd.utils.logging import _plot_rewards_svg

    # reuse the SVG plotter: one "episode" per N with [frames/s, grads*1000]
    _plot_rewards_svg([p[0] for p in pts], [[p[1]] for p in pts], out,
                      "aggregate env-frames/s vs GPUs")
    print(f"wrote {out}: " + ", ".join(
        f"N={n}: {v:.3g} f/s, {g:.0f} grad/s" for n, v, g in pts))


if __name__ == "__main__":
    main()
