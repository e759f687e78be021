"""Flag surface of the framework.

Mirrors the argparse surface of the reference (rainbow-iqn-apex ``args.py``,
Kaixhin/Rainbow lineage -- see SURVEY.md section 3.4; /root/reference was empty
at survey time so flag names follow the surveyed [MED]-confidence inventory).

Every hyperparameter of the reference exists here under the same flag name;
MI355X-native additions (RCCL topology, GPU replay, hipGraph toggles) are
grouped at the end and default to values that reproduce reference behavior.
"""

from __future__ import annotations

import argparse
import dataclasses
from typing import Optional


@dataclasses.dataclass
class Config:
    # -- run identity ------------------------------------------------------
    id: str = "default"
    game: str = "pong"  # Atari id; synthetic source mimics its frame shape
    seed: int = 123
    disable_cuda: bool = False

    # -- agent / model (reference defaults) --------------------
    T_max: int = int(50e6)  # total agent steps (= 200M frames at repeat 4)
    max_episode_length: int = int(108e3)  # SABER 30-min cap, in frames
    history_length: int = 4
    hidden_size: int = 512
    noisy_std: float = 0.1
    atoms: int = 51  # unused by IQN head; kept for flag parity
    # IQN quantile machinery (Dabney et al. 2018 defaults)
    quantile_embedding_dim: int = 64
    num_tau_samples: int = 8  # N   (online-net taus in the loss)
    num_tau_prime_samples: int = 8  # N'  (target-net taus in the loss)
    num_quantile_samples: int = 32  # K   (taus for action selection)
    kappa: float = 1.0  # Huber threshold

    # -- replay ------------------------------------------------------------
    memory_capacity: int = int(1e6)
    replay_frequency: int = 4  # env steps per grad step (single-proc lineage)
    priority_exponent: float = 0.5  # omega
    priority_weight: float = 0.4  # beta, annealed -> 1
    multi_step: int = 3
    learn_start: int = int(80e3)

    # -- optimisation ------------------------------------------------------
    discount: float = 0.99
    target_update: int = int(32e3)  # agent steps between hard target copies
    reward_clip: float = 1.0  # clip to [-r, r]; 0 disables
    lr: float = 0.0000625
    adam_eps: float = 1.5e-4
    batch_size: int = 32

    # -- evaluation --------------------------------------------------------
    evaluation_interval: int = int(100e3)
    evaluation_episodes: int = 10
    evaluation_size: int = 500  # states held out for avg-Q tracking
    render: bool = False

    # -- checkpointing -----------------------------------------------------
    model: Optional[str] = None  # path to load
    checkpoint_interval: int = 0  # 0 disables periodic checkpoints
    memory_save_path: Optional[str] = None  # replay checkpoint (reference: redis RDB)
    recorded_env: Optional[str] = None  # .npz trajectory tape: run the real
    # AtariEnv wrapper over recorded frames (env_recorded.py; no ALE needed)

    # -- Ape-X topology (reference: redis flags; here: RCCL over xGMI) -----
    nb_actor: int = 1
    id_actor: int = 0
    transition_chunk_size: int = 64  # transitions pushed per exchange
    weight_sync_period: int = 25  # learner grad-steps between weight publishes
    # Reference's redis host/port kept for CLI compatibility; ignored (we use
    # torch.distributed rendezvous instead of a Redis server).
    host_redis: str = "localhost"
    port_redis: int = 6379

    # -- MI355X-native knobs ----------------------------------------------
    device: str = "auto"  # auto|cpu|cuda
    dtype: str = "bf16"  # compute dtype on GPU: bf16|fp32
    gpu_replay: bool = True  # on-GPU sum-tree replay when on GPU
    use_hip_kernels: bool = True  # hand-written CDNA4 kernels for hot ops
    hipgraph: bool = True  # capture act/learn steps in hip graphs
    fused_forward: bool = True  # hand-written fused model-forward kernels
    fused_backward: bool = True  # hand-written learn-step backward kernels
    fp8_heads: bool = False  # MX-fp8 NoisyLinear heads GEMM (config 5 path)
    overlap_colocated: bool = True  # actor/learner stream concurrency on 1 GPU
    cu_split: int = 96  # colocated overlap: CUs [0,N) -> learner queue,
    # rest -> actor queue (per-queue CU mask; 96 = 3 XCDs, measured best;
    # 0 disables the partition)
    sharded_replay: bool = False  # replay shards live on the actor GPUs (config 5)
    backend: str = "auto"  # dist backend: auto|nccl|gloo
    learn_per_round: int = 8  # learner grad-steps per exchange round (Ape-X)

    # ----------------------------------------------------------------------
    @property
    def reward_clip_enabled(self) -> bool:
        return self.reward_clip > 0

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        import torch

        if not self.disable_cuda and torch.cuda.is_available():
            return "cuda"
        return "cpu"


def _add_args(p: argparse.ArgumentParser) -> None:
    d = Config()
    p.add_argument("--id", type=str, default=d.id, help="experiment id")
    p.add_argument("--game", type=str, default=d.game, help="Atari game id")
    p.add_argument("--seed", type=int, default=d.seed)
    p.add_argument("--disable-cuda", action="store_true")
    p.add_argument("--T-max", type=int, default=d.T_max, metavar="STEPS")
    p.add_argument("--max-episode-length", type=int, default=d.max_episode_length)
    p.add_argument("--history-length", type=int, default=d.history_length)
    p.add_argument("--hidden-size", type=int, default=d.hidden_size)
    p.add_argument("--noisy-std", type=float, default=d.noisy_std)
    p.add_argument("--atoms", type=int, default=d.atoms)
    p.add_argument("--quantile-embedding-dim", type=int, default=d.quantile_embedding_dim)
    p.add_argument("--num-tau-samples", type=int, default=d.num_tau_samples)
    p.add_argument("--num-tau-prime-samples", type=int, default=d.num_tau_prime_samples)
    p.add_argument("--num-quantile-samples", type=int, default=d.num_quantile_samples)
    p.add_argument("--kappa", type=float, default=d.kappa)
    p.add_argument("--memory-capacity", type=int, default=d.memory_capacity)
    p.add_argument("--replay-frequency", type=int, default=d.replay_frequency)
    p.add_argument("--priority-exponent", type=float, default=d.priority_exponent)
    p.add_argument("--priority-weight", type=float, default=d.priority_weight)
    p.add_argument("--multi-step", type=int, default=d.multi_step)
    p.add_argument("--learn-start", type=int, default=d.learn_start)
    p.add_argument("--discount", type=float, default=d.discount)
    p.add_argument("--target-update", type=int, default=d.target_update)
    p.add_argument("--reward-clip", type=float, default=d.reward_clip)
    p.add_argument("--lr", type=float, default=d.lr)
    p.add_argument("--adam-eps", type=float, default=d.adam_eps)
    p.add_argument("--batch-size", type=int, default=d.batch_size)
    p.add_argument("--evaluation-interval", type=int, default=d.evaluation_interval)
    p.add_argument("--evaluation-episodes", type=int, default=d.evaluation_episodes)
    p.add_argument("--evaluation-size", type=int, default=d.evaluation_size)
    p.add_argument("--render", action="store_true")
    p.add_argument("--model", type=str, default=None, help="checkpoint to load")
    p.add_argument("--checkpoint-interval", type=int, default=d.checkpoint_interval)
    p.add_argument("--memory-save-path", type=str, default=None)
    p.add_argument("--recorded-env", type=str, default=None,
                   help=".npz trajectory tape for the recorded-frame env")
    p.add_argument("--nb-actor", type=int, default=d.nb_actor)
    p.add_argument("--id-actor", type=int, default=d.id_actor)
    p.add_argument("--transition-chunk-size", type=int, default=d.transition_chunk_size)
    p.add_argument("--weight-sync-period", type=int, default=d.weight_sync_period)
    p.add_argument("--host-redis", type=str, default=d.host_redis)
    p.add_argument("--port-redis", type=int, default=d.port_redis)
    p.add_argument("--device", type=str, default=d.device, choices=["auto", "cpu", "cuda"])
    p.add_argument("--dtype", type=str, default=d.dtype, choices=["bf16", "fp32"])
    p.add_argument("--no-gpu-replay", dest="gpu_replay", action="store_false")
    p.add_argument("--no-hip-kernels", dest="use_hip_kernels", action="store_false")
    p.add_argument("--no-hipgraph", dest="hipgraph", action="store_false")
    p.add_argument("--no-fused-forward", dest="fused_forward", action="store_false")
    p.add_argument("--no-fused-backward", dest="fused_backward", action="store_false")
    p.add_argument("--fp8-heads", action="store_true")
    p.add_argument("--no-overlap", dest="overlap_colocated", action="store_false")
    p.add_argument("--cu-split", dest="cu_split", type=int, default=d.cu_split)
    p.add_argument("--sharded-replay", action="store_true")
    p.add_argument("--backend", type=str, default=d.backend, choices=["auto", "nccl", "gloo"])
    p.add_argument("--learn-per-round", type=int, default=d.learn_per_round)


def parse_args(argv=None) -> Config:
    p = argparse.ArgumentParser(description="rainbow-iqn-apex, MI355X-native")
    _add_args(p)
    ns = p.parse_args(argv)
    fields = {f.name for f in dataclasses.fields(Config)}
    return Config(**{k: v for k, v in vars(ns).items() if k in fields})
