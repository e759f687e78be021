"""Optional Prometheus metrics exporter (the reference has
stdout prints only; we add native steps/sec + loss + replay gauges).

Enabled with --metrics-port; no-op if prometheus_client is missing.
"""

from __future__ import annotations

from typing import Optional


class Metrics:
    def __init__(self, port: Optional[int] = None):
        self.enabled = False
        if not port:
            return
        try:
            from prometheus_client import (CollectorRegistry, Counter, Gauge,
                                           start_http_server)
        except ImportError:  # pragma: no cover
            return
        # private registry: repeated Metrics construction in one process
        # (tests, notebooks) must not collide in the global REGISTRY
        reg = CollectorRegistry()
        start_http_server(port, registry=reg)
        self.env_steps = Counter("riqn_env_steps_total",
                                 "agent env steps on this rank", registry=reg)
        self.grad_steps = Counter("riqn_grad_steps_total",
                                  "learner gradient steps", registry=reg)
        self.loss = Gauge("riqn_loss", "latest quantile-Huber loss",
                          registry=reg)
        self.replay_size = Gauge("riqn_replay_transitions",
                                 "transitions resident in replay",
                                 registry=reg)
        self.steps_per_sec = Gauge("riqn_env_steps_per_sec",
                                   "recent env steps/sec (this rank)",
                                   registry=reg)
        self.recoveries = Gauge("riqn_nan_recoveries_total",
                                "divergence rollbacks (utils/failsafe.py)",
                                registry=reg)
        self.enabled = True

    def update(self, d_env_steps: int = 0, d_grad_steps: int = 0,
               loss: Optional[float] = None,
               replay_size: Optional[int] = None,
               steps_per_sec: Optional[float] = None,
               recoveries: Optional[int] = None) -> None:
        if not self.enabled:
            return
        if recoveries is not None:
            self.recoveries.set(recoveries)
        if d_env_steps:
            self.env_steps.inc(d_env_steps)
        if d_grad_steps:
            self.grad_steps.inc(d_grad_steps)
        if loss is not None:
            self.loss.set(loss)
        if replay_size is not None:
            self.replay_size.set(replay_size)
        if steps_per_sec is not None:
            self.steps_per_sec.set(steps_per_sec)
