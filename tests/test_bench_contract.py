"""bench.py driver-contract regression test: one JSON line with the exact
field set the round driver parses (run tiny on CPU)."""

import json
import subprocess
import sys


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "0",
         "--n-envs", "4", "--chunk-steps", "4", "--learn-per-round", "1",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=600, check=True,
    )
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    for k in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
              "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
              "dtype", "data", "config"):
        assert k in d, k
    assert d["n_gpus"] == 1 and d["steps"] == 1 and d["warmup"] == 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["unit"] == "frames/s" and d["value"] > 0
    assert d["data"] == "synthetic"
    assert "learner_grad_steps_per_sec" in d
    cfgd = d["config"]
    for k in ("model", "global_batch", "seq_len", "parallelism"):
        assert k in cfgd, k


def test_bench_learner_only_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "0",
         "--learn-per-round", "1", "--device", "cpu", "--learner-only"],
        capture_output=True, text=True, timeout=600, check=True,
    )
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["unit"] == "grad-steps/s" and d["value"] > 0
    assert d["env_steps_per_sec"] == 0.0
    assert "learner-only" in d["config"]["parallelism"]


def test_bench_torchrun_world2_contract():
    """The driver's exact multi-rank launch shape (torchrun --nnodes=1):
    rank 0 emits ONE JSON line aggregating both ranks' work."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29653", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--n-envs", "4",
         "--chunk-steps", "4", "--learn-per-round", "1", "--device", "cpu"],
        capture_output=True, text=True, timeout=900, check=True,
    )
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert "1 learner + 1 actor" in d["config"]["parallelism"]
    assert d["value"] > 0 and d["learner_grad_steps_per_sec"] > 0
