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
    """Write an HTML reward curve (mean +- std band over eval episodes)."""
    try:
        import plotly.graph_objects as go
    except ImportError:  # pragma: no cover
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
