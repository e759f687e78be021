"""Distributed Ape-X training entry point (MI355X-native).

This is the rebuild of the reference's learner/actor launch scripts
(rainbow-iqn-apex root-level entry scripts, SURVEY.md sections 4.1/4.2): the
reference starts one learner process and N actor processes that meet at a
Redis server; here ONE command launches W ranks over torch.distributed
(RCCL/xGMI on GPU, gloo on CPU), rank 0 = learner, ranks 1..W-1 = actors.
World size 1 colocates both roles in a single process on one GPU.

  # 1 learner + 7 actors, one 8-GPU MI355X node:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 -m rainbowiqn_amd.train_apex --game pong

  # single process (colocated roles):
  python -m rainbowiqn_amd.train_apex --game pong --T-max 1000000

The loop runs fixed-cadence Ape-X rounds (ApexRunner.run_round): actors
produce chunk_steps x n_envs transitions, an RCCL all-gather lands them in
the learner replay, the learner takes learn_per_round grad steps, and a
broadcast returns fresh weights -- bounded one-round staleness in place of
the reference's Redis polling.  The learner side carries the reference's
full training schedule: learn_start warmup, beta annealing to 1, hard target
update every target_update agent steps, periodic evaluation with best-model
checkpointing, periodic model + replay checkpoints, and resume.
"""

from __future__ import annotations

import argparse
import os
from typing import Optional

import torch

from .config import Config, parse_args
from .env import n_actions_for
from .eval import Evaluator
from .parallel.apex import ApexRunner
from .parallel.comm import Communicator
from .utils.logging import log


def _anneal_beta(runner: ApexRunner, T: int, cfg: Config) -> None:
    """beta: priority_weight -> 1 linearly over [learn_start, T_max]."""
    span = max(1, cfg.T_max - cfg.learn_start)
    beta = min(1.0, cfg.priority_weight + (1.0 - cfg.priority_weight)
               * max(0, T - cfg.learn_start) / span)
    if runner.glearner is not None:
        runner.glearner.anneal_beta(beta)
    elif runner.replay is not None:
        runner.replay.priority_weight = beta


def _learner_state_dict(runner: ApexRunner) -> dict:
    if runner.glearner is not None:
        return runner.glearner.state_dict()
    return runner.agent.online_net.state_dict()


def _save_model(runner: ApexRunner, path: str, name: str = "model.pth") -> None:
    os.makedirs(path, exist_ok=True)
    torch.save(_learner_state_dict(runner), os.path.join(path, name))


def _save_memory(runner: ApexRunner, path: str) -> None:
    """Replay persistence (reference: Redis RDB snapshot / BGSAVE)."""
    rep = runner.replay
    if rep is None:
        return
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    torch.save(rep.state_dict(), path)


def _load_memory(runner: ApexRunner, path: str) -> None:
    rep = runner.replay
    if rep is None or not os.path.exists(path):
        return
    rep.load_state_dict(torch.load(path, map_location="cpu", weights_only=False))
    log(f"replay restored from {path} ({len(rep)} transitions)")


class _EvalProxy:
    """Adapts learner weights (fast or eager path) to the eager eval Agent."""

    def __init__(self, cfg: Config, runner: ApexRunner, device: str):
        from .agent import Agent

        ecfg = Config(**{**cfg.__dict__, "model": None})
        self.agent = Agent(ecfg, n_actions_for(cfg.game), device)
        self.runner = runner

    def refresh(self) -> "_EvalProxy.agent":
        sd = _learner_state_dict(self.runner)
        self.agent.online_net.load_state_dict(
            {k: v.float() for k, v in sd.items()}
        )
        return self.agent


def train_apex(cfg: Config, comm: Optional[Communicator] = None,
               n_envs: int = 16, results_dir: Optional[str] = None,
               max_rounds: Optional[int] = None, log_interval: int = 50,
               metrics_port: Optional[int] = None,
               chunk_steps: Optional[int] = None,
               trace_file: Optional[str] = None):
    """Run Ape-X training to cfg.T_max agent steps.  Returns the runner."""
    device = cfg.resolve_device()
    if device == "cuda":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    comm = comm or Communicator(backend=cfg.backend)
    torch.manual_seed(cfg.seed + comm.rank * 9973)

    runner = ApexRunner(cfg, comm, n_envs=n_envs, device=device,
                        chunk_steps=chunk_steps)
    tracer = None
    if trace_file:
        from .utils.trace import RoundTracer

        path = (trace_file if comm.world_size == 1
                else f"{trace_file}.rank{comm.rank}")
        tracer = RoundTracer(path, use_cuda=(device == "cuda"),
                             pid=comm.rank)
        runner.set_tracer(tracer)
    if cfg.model and comm.is_learner:
        _load_model(runner, cfg.model)
    if cfg.memory_save_path:
        if runner.sharded and comm.is_actor:
            # sharded mode: each actor owns (and persists) its shard
            p = f"{cfg.memory_save_path}.rank{comm.rank}"
            if os.path.exists(p):
                runner.shard.replay.load_state_dict(
                    torch.load(p, map_location="cpu", weights_only=False))
                log(f"replay shard restored from {p}")
        elif comm.is_learner:
            _load_memory(runner, cfg.memory_save_path)

    from .utils.metrics import Metrics

    metrics = Metrics((metrics_port + comm.rank) if metrics_port else None)
    evaluator = None
    proxy = None
    if comm.is_learner and cfg.evaluation_interval <= cfg.T_max:
        evaluator = Evaluator(cfg, device="cpu")
        proxy = _EvalProxy(cfg, runner, device)

    # failure handling: SIGTERM/SIGINT -> checkpoint at the
    # round boundary; non-finite loss -> roll back to last healthy snapshot
    from .utils.failsafe import GracefulShutdown, NanGuard

    shutdown = GracefulShutdown()
    guard = NanGuard(runner) if comm.is_learner else None

    # env steps generated per round across ALL actor loops: n_actors ranks
    # x n_local_actors loops per rank (--nb-actor, BASELINE config 3).
    # Undercounting here would stretch beta annealing and the eval/target/
    # checkpoint cadences by that factor.
    steps_per_round = (runner.chunk_steps * runner.n_envs * comm.n_actors
                       * runner.n_local_actors)
    T, rounds = 0, 0
    next_eval = cfg.evaluation_interval
    next_ckpt = cfg.checkpoint_interval or float("inf")
    import time

    t_log = time.perf_counter()
    T_log = 0
    g_log = 0
    while T < cfg.T_max:
        runner.run_round()
        rounds += 1
        T += steps_per_round
        if shutdown.requested:
            if comm.is_learner and results_dir:
                _save_model(runner, results_dir)
                if cfg.memory_save_path and not runner.sharded:
                    _save_memory(runner, cfg.memory_save_path)
            break
        if comm.is_learner:
            if guard.check(runner.losses[-1] if runner.losses else None):
                metrics.update(recoveries=guard.recoveries)
            _anneal_beta(runner, T, cfg)
            if evaluator is not None and T >= next_eval:
                next_eval += cfg.evaluation_interval
                evaluator.evaluate(proxy.refresh(), T, results_dir)
            if T >= next_ckpt and results_dir:
                next_ckpt += cfg.checkpoint_interval
                _save_model(runner, results_dir)
                if cfg.memory_save_path:
                    _save_memory(runner, cfg.memory_save_path)
            if rounds % log_interval == 0:
                if device == "cuda":
                    # drain queued rounds so steps/s is an execution rate
                    # (pre-learn_start rounds have no host sync point)
                    torch.cuda.synchronize()
                dt = time.perf_counter() - t_log
                sps = (T - T_log) / max(dt, 1e-9)
                loss = runner.losses[-1] if runner.losses else float("nan")
                log(f"T={T} rounds={rounds} grad_steps={runner.grad_steps} "
                    f"loss={loss:.4f} steps/s={sps:.0f} "
                    f"frames/s={sps * 4:.0f}")
                metrics.update(d_env_steps=T - T_log,
                               d_grad_steps=runner.grad_steps - g_log,
                               loss=loss if loss == loss else None,
                               replay_size=(len(runner.replay)
                                            if runner.replay else None),
                               steps_per_sec=sps)
                t_log, T_log = time.perf_counter(), T
                g_log = runner.grad_steps
        if max_rounds is not None and rounds >= max_rounds:
            break

    if comm.is_learner and results_dir:
        _save_model(runner, results_dir)
        if cfg.memory_save_path and not runner.sharded:
            _save_memory(runner, cfg.memory_save_path)
    if runner.sharded and comm.is_actor and cfg.memory_save_path:
        torch.save(runner.shard.replay.state_dict(),
                   f"{cfg.memory_save_path}.rank{comm.rank}")
    if tracer is not None:
        log(f"round-phase trace written to {tracer.dump()}")
    shutdown.restore()
    return runner


def _load_model(runner: ApexRunner, path: str) -> None:
    if os.path.isdir(path):
        path = os.path.join(path, "model.pth")
    sd = torch.load(path, map_location="cpu", weights_only=True)
    if runner.glearner is not None:
        runner.glearner.load_state_dict_ref(sd)
    else:
        from .utils.checkpoint import remap_reference_state_dict

        net = runner.agent.online_net
        net.load_state_dict(remap_reference_state_dict(sd, net))
        runner.agent.update_target_net()
    log(f"model restored from {path}")


def main(argv=None):
    p = argparse.ArgumentParser(add_help=False)
    p.add_argument("--n-envs", type=int, default=16,
                   help="vectorized env streams per actor rank")
    p.add_argument("--max-rounds", type=int, default=None)
    p.add_argument("--metrics-port", type=int, default=None,
                   help="Prometheus exporter port (rank offset added)")
    p.add_argument("--chunk-steps", type=int, default=8,
                   help="env steps per Ape-X exchange round")
    p.add_argument("--trace-file", type=str, default=None,
                   help="write a Chrome-trace round-phase timeline here")
    ns, rest = p.parse_known_args(argv)
    cfg = parse_args(rest)
    comm = Communicator(backend=cfg.backend)
    results_dir = os.path.join("results", cfg.id)
    runner = train_apex(cfg, comm, n_envs=ns.n_envs, results_dir=results_dir,
                        max_rounds=ns.max_rounds,
                        metrics_port=ns.metrics_port,
                        chunk_steps=ns.chunk_steps,
                        trace_file=ns.trace_file)
    if comm.is_learner:
        log(f"done: grad_steps={runner.grad_steps}")
    comm.close()


if __name__ == "__main__":
    main()
