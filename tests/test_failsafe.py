"""Aux subsystems: round-phase tracing and failure
detection / recovery -- CPU path."""

import json
import os
import signal
import threading

import pytest
import torch


@pytest.fixture
def cfg():
    from rainbowiqn_amd.config import Config

    return Config(device="cpu", dtype="fp32", memory_capacity=2048,
                  learn_start=64, batch_size=8, learn_per_round=1,
                  T_max=100000)


def test_round_tracer_cpu(tmp_path):
    from rainbowiqn_amd.utils.trace import RoundTracer

    t = RoundTracer(str(tmp_path / "t.json"), use_cuda=False, pid=3)
    with t.span("actor"):
        pass
    with t.span("learn"):
        with t.span("inner"):
            pass
    path = t.dump()
    ev = json.load(open(path))["traceEvents"]
    assert {e["name"] for e in ev} == {"actor", "learn", "inner"}
    assert all(e["ph"] == "X" and e["pid"] == 3 and e["dur"] >= 0
               for e in ev)


def test_round_tracer_span_cap(tmp_path):
    from rainbowiqn_amd.utils.trace import RoundTracer

    t = RoundTracer(str(tmp_path / "t.json"), use_cuda=False, max_spans=2)
    for _ in range(5):
        with t.span("x"):
            pass
    assert len(t.events()) == 2  # silently stops recording past the cap


def test_train_apex_writes_trace(tmp_path, cfg):
    from rainbowiqn_amd.parallel.comm import Communicator
    from rainbowiqn_amd.train_apex import train_apex

    trace = str(tmp_path / "round_trace.json")
    train_apex(cfg, Communicator(backend="auto"), n_envs=4,
               results_dir=str(tmp_path), max_rounds=6, trace_file=trace)
    ev = json.load(open(trace))["traceEvents"]
    names = {e["name"] for e in ev}
    assert "actor" in names and "learn" in names
    # 6 rounds of actor spans at 4 envs
    assert sum(1 for e in ev if e["name"] == "actor") == 6


def test_graceful_shutdown_flag():
    from rainbowiqn_amd.utils.failsafe import GracefulShutdown

    g = GracefulShutdown(signals=(signal.SIGUSR1,))
    assert not g.requested
    os.kill(os.getpid(), signal.SIGUSR1)
    # the handler runs synchronously in the main thread on the next check
    assert g.requested
    g.restore()


def test_train_apex_sigterm_checkpoints(tmp_path, cfg):
    """SIGTERM mid-training -> clean break at a round boundary + model
    checkpoint on disk (reference parity: learner restart resumes from
    checkpoint)."""
    from rainbowiqn_amd.parallel.comm import Communicator
    from rainbowiqn_amd.train_apex import train_apex

    cfg.learn_start = 10**9  # no learning: rounds are fast
    timer = threading.Timer(0.25, os.kill, (os.getpid(), signal.SIGTERM))
    timer.start()
    try:
        runner = train_apex(cfg, Communicator(backend="auto"),
                            n_envs=2, results_dir=str(tmp_path),
                            max_rounds=100000)
    finally:
        timer.cancel()
    assert runner is not None
    assert os.path.exists(tmp_path / "model.pth")
    # the handler was restored: a later SIGTERM must not be swallowed
    assert signal.getsignal(signal.SIGTERM) in (
        signal.SIG_DFL, signal.default_int_handler) or callable(
        signal.getsignal(signal.SIGTERM))


def test_nan_guard_eager_rollback(cfg):
    """Corrupt weights + non-finite loss -> NanGuard restores the last
    healthy snapshot and resets the optimizer."""
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator
    from rainbowiqn_amd.utils.failsafe import NanGuard

    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=4)
    guard = NanGuard(runner, interval=1)
    assert not guard.check(0.5)  # healthy round: snapshot taken
    p = next(runner.agent.online_net.parameters())
    healthy = p.detach().clone()
    with torch.no_grad():
        p.add_(1e6)
    assert guard.check(float("nan"))  # divergence -> rollback
    assert torch.equal(p.detach(), healthy)
    assert guard.recoveries == 1
    # training continues: next healthy round re-snapshots
    assert not guard.check(0.4)


def test_nan_guard_before_first_snapshot(cfg):
    from rainbowiqn_amd.parallel.apex import ApexRunner
    from rainbowiqn_amd.parallel.comm import Communicator
    from rainbowiqn_amd.utils.failsafe import NanGuard

    runner = ApexRunner(cfg, Communicator(backend="auto"), n_envs=4)
    guard = NanGuard(runner)
    assert not guard.check(float("inf"))  # nothing to restore: no crash
    assert guard.recoveries == 0


def test_metrics_exporter(tmp_path):
    """Prometheus exporter serves the gauges when enabled and
    no-ops cleanly when disabled."""
    import socket
    import urllib.request

    from rainbowiqn_amd.utils.metrics import Metrics

    off = Metrics(None)
    off.update(d_env_steps=10, loss=1.0)  # disabled: no-op, no crash
    assert not off.enabled

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    m = Metrics(port)
    if not m.enabled:  # prometheus_client missing on this box
        return
    m.update(d_env_steps=123, d_grad_steps=7, loss=0.5, replay_size=42,
             steps_per_sec=1000.0, recoveries=1)
    body = urllib.request.urlopen(
        f"http://127.0.0.1:{port}/metrics", timeout=10).read().decode()
    assert "riqn_env_steps_total 123" in body
    assert "riqn_loss 0.5" in body
    assert "riqn_nan_recoveries_total 1" in body
