"""stdout logging + reward-curve plotting (reference parity: log() prints and
plotly HTML reward curves written at eval time -- SURVEY.md section 6.5)."""

from __future__ import annotations

import os
import time
from typing import List


def log(msg: str) -> None:
    print(f"[{time.strftime('%Y-%m-%dT%H:%M:%S')}] {msg}", flush=True)


def plot_rewards(steps: List[int], rewards: List[List[float]], path: str,
                 title: str = "Reward") -> None:
    """Write an HTML reward curve (mean +- std band over eval episodes).

    Uses plotly when importable (reference behavior); otherwise falls back to
    a self-contained inline-SVG HTML file so eval always leaves a curve."""
    try:
        import plotly.graph_objects as go
    except ImportError:
        _plot_rewards_svg(steps, rewards, path, title)
        return
    import numpy as np

    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    r = np.asarray([[float(x) for x in ep] for ep in rewards], dtype=float)
    mean, std = r.mean(axis=1), r.std(axis=1)
    fig = go.Figure()
    fig.add_trace(go.Scatter(x=steps, y=mean, name="mean", line=dict(color="#d62728")))
    fig.add_trace(
        go.Scatter(
            x=list(steps) + list(steps)[::-1],
            y=list(mean + std) + list(mean - std)[::-1],
            fill="toself",
            fillcolor="rgba(214,39,40,0.2)",
            line=dict(width=0),
            name="+-std",
        )
    )
    fig.update_layout(title=title, xaxis_title="step", yaxis_title="reward")
    fig.write_html(path, include_plotlyjs="cdn")


def _plot_rewards_svg(steps: List[int], rewards: List[List[float]], path: str,
                      title: str) -> None:
    """Dependency-free reward curve: mean line + std band as inline SVG."""
    import numpy as np

    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    r = np.asarray([[float(x) for x in ep] for ep in rewards], dtype=float)
    mean, std = r.mean(axis=1), r.std(axis=1)
    xs = np.asarray(steps, dtype=float)
    w, h, pad = 720.0, 360.0, 50.0
    x0, x1 = float(xs.min()), float(max(xs.max(), xs.min() + 1))
    lo, hi = float((mean - std).min()), float((mean + std).max())
    if hi <= lo:
        hi = lo + 1.0

    def sx(v):
        return pad + (v - x0) / (x1 - x0) * (w - 2 * pad)

    def sy(v):
        return h - pad - (v - lo) / (hi - lo) * (h - 2 * pad)

    pts = " ".join(f"{sx(x):.1f},{sy(y):.1f}" for x, y in zip(xs, mean))
    band = " ".join(f"{sx(x):.1f},{sy(y):.1f}" for x, y in zip(xs, mean + std))
    band += " " + " ".join(
        f"{sx(x):.1f},{sy(y):.1f}" for x, y in zip(xs[::-1], (mean - std)[::-1])
    )
    svg = (
        f'<svg xmlns="http://www.w3.org/2000/svg" width="{w:.0f}" height="{h:.0f}">'
        f'<rect width="100%" height="100%" fill="white"/>'
        f'<text x="{w/2:.0f}" y="20" text-anchor="middle" font-size="15">{title}</text>'
        f'<polygon points="{band}" fill="rgba(214,39,40,0.2)" stroke="none"/>'
        f'<polyline points="{pts}" fill="none" stroke="#d62728" stroke-width="2"/>'
        f'<text x="{w/2:.0f}" y="{h-8:.0f}" text-anchor="middle" font-size="12">step</text>'
        f'<text x="12" y="{h/2:.0f}" font-size="12" transform="rotate(-90 12 {h/2:.0f})">reward</text>'
        f'<text x="{pad:.0f}" y="{h-pad+16:.0f}" font-size="11">{x0:.0f}</text>'
        f'<text x="{w-pad:.0f}" y="{h-pad+16:.0f}" text-anchor="end" font-size="11">{x1:.0f}</text>'
        f'<text x="{pad-4:.0f}" y="{sy(lo):.0f}" text-anchor="end" font-size="11">{lo:.1f}</text>'
        f'<text x="{pad-4:.0f}" y="{sy(hi)+4:.0f}" text-anchor="end" font-size="11">{hi:.1f}</text>'
        f"</svg>"
    )
    with open(path, "w") as f:
        f.write(f"<!DOCTYPE html><html><head><title>{title}</title></head>"
                f"<body>{svg}</body></html>")
