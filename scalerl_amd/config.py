"""Dataclass configuration for every algorithm, plus a tiny dataclass→argparse CLI.

Capability parity with the reference's config layer
(``scalerl/algorithms/rl_args.py:8-362`` — RLArguments / DQNArguments /
A3CArguments) plus the IMPALA fields the reference reads but never declares
(SURVEY.md "Broken-as-shipped": ``num_buffers``, ``use_lstm``,
``total_steps``, ``reward_clipping``, ``discounting``, ``baseline_cost``,
``entropy_cost`` …).  The reference parses with ``tyro.cli``; tyro is not in
this image, so :func:`parse_cli` provides the same UX from argparse.
"""

from __future__ import annotations

import argparse
import dataclasses
from dataclasses import dataclass, field, fields
from typing import Optional, Sequence, Type, TypeVar

T = TypeVar("T")


def _add_dataclass_args(parser: argparse.ArgumentParser, cls: Type) -> None:
    for f in fields(cls):
        if not f.init:
            continue
        name = "--" + f.name.replace("_", "-")
        default = f.default if f.default is not dataclasses.MISSING else (
            f.default_factory() if f.default_factory is not dataclasses.MISSING else None)
        help_str = f.metadata.get("help", "") if f.metadata else ""
        ftype = f.type if isinstance(f.type, type) else None
        # Resolve string annotations ("int", "float", ...) from __future__ annotations.
        if ftype is None:
            tname = str(f.type)
            ftype = {"int": int, "float": float, "str": str, "bool": bool,
                     "Optional[int]": int, "Optional[float]": float,
                     "Optional[str]": str}.get(tname, str)
        if ftype is bool:
            parser.add_argument(name, type=lambda s: s.lower() in ("1", "true", "yes"),
                                default=default, help=help_str, metavar="BOOL")
        else:
            parser.add_argument(name, type=ftype, default=default, help=help_str)


def parse_cli(cls: Type[T], argv: Optional[Sequence[str]] = None) -> T:
    """Parse ``cls`` (a dataclass) from the command line (tyro-style UX)."""
    parser = argparse.ArgumentParser(description=cls.__doc__)
    _add_dataclass_args(parser, cls)
    ns = parser.parse_args(argv)
    kwargs = {f.name: getattr(ns, f.name) for f in fields(cls) if f.init}
    return cls(**kwargs)


def _h(s: str, **kw):
    return field(metadata={"help": s}, **kw)


@dataclass
class RLArguments:
    """Common knobs shared by every algorithm (reference: rl_args.py:8-159)."""

    project: str = _h("project name for run-dir layout", default="scalerl-amd")
    algo_name: str = _h("algorithm name", default="dqn")
    env_id: str = _h("environment id", default="CartPole-v1")
    seed: int = _h("random seed", default=42)
    device: str = _h("compute device: cuda | cpu | auto", default="auto")

    num_envs: int = _h("vectorized envs per actor process", default=8)
    num_actors: int = _h("actor (env-worker) processes", default=2)
    num_learners: int = _h("learner ranks (GPUs)", default=1)

    buffer_size: int = _h("replay capacity (transitions)", default=100_000)
    batch_size: int = _h("learner batch size", default=64)
    warmup_learn_steps: int = _h("transitions before learning starts", default=1_000)
    train_frequency: int = _h("env steps between learner updates", default=4)
    learner_update_times: int = _h("gradient steps per update", default=1)

    gamma: float = _h("discount factor", default=0.99)
    learning_rate: float = _h("optimizer learning rate", default=1e-4)
    min_learning_rate: float = _h("floor for lr decay", default=1e-5)
    max_grad_norm: float = _h("gradient clipping norm (0 disables)", default=40.0)

    max_train_steps: int = _h("total env steps to train for", default=100_000)
    eval_episodes: int = _h("episodes per evaluation", default=3)
    train_log_interval: int = _h("episodes between train logs", default=5)
    test_log_interval: int = _h("episodes between evals", default=20)

    work_dir: str = _h("output root", default="work_dirs")
    logger_backend: str = _h("metrics backend: jsonl | tensorboard | wandb", default="jsonl")
    save_model: bool = _h("save checkpoint at end of training", default=True)
    checkpoint_interval_s: float = _h("seconds between periodic checkpoints", default=600.0)


@dataclass
class DQNArguments(RLArguments):
    """DQN family (reference: rl_args.py:163-315)."""

    algo_name: str = "dqn"
    double_dqn: bool = _h("double-DQN target selection", default=True)
    dueling_dqn: bool = _h("dueling heads", default=False)
    noisy_dqn: bool = _h("noisy linear layers", default=False)
    categorical_dqn: bool = _h("C51 distributional head", default=False)
    n_steps: int = _h("n-step returns (1 = vanilla)", default=1)
    use_per: bool = _h("prioritized replay", default=False)
    per_alpha: float = _h("PER priority exponent", default=0.6)
    per_beta: float = _h("PER IS-weight exponent (annealed→1)", default=0.4)
    v_min: float = _h("C51 value-support min", default=-10.0)
    v_max: float = _h("C51 value-support max", default=10.0)
    num_atoms: int = _h("C51 atoms", default=51)
    eps_greedy_start: float = _h("initial exploration epsilon", default=1.0)
    eps_greedy_end: float = _h("final exploration epsilon", default=0.1)
    eps_decay_steps: int = _h("steps for linear epsilon decay", default=50_000)
    target_update_frequency: int = _h("learner steps between target syncs", default=100)
    soft_update_tau: float = _h("polyak τ (0 = hard update)", default=0.05)
    hidden_dim: int = _h("MLP hidden width", default=128)


@dataclass
class A3CArguments(RLArguments):
    """A3C / A2C (reference: rl_args.py:319-362)."""

    algo_name: str = "a3c"
    env_id: str = "CartPole-v1"
    num_workers: int = _h("hogwild worker processes", default=4)
    rollout_steps: int = _h("steps per update rollout", default=20)
    gae_lambda: float = _h("GAE λ", default=1.0)
    value_loss_coef: float = _h("critic loss coefficient", default=0.5)
    entropy_coef: float = _h("entropy bonus coefficient", default=0.01)
    no_shared: bool = _h("per-worker (non-shared) optimizer state", default=False)
    max_episode_steps: int = _h("episode step cap in workers", default=200)


@dataclass
class A3CGpuArguments(A3CArguments):
    """A3C at GPU scale (BASELINE config 2: Pong 42×42, 16 CPU actors +
    1 MI355X learner) — batched synchronous A2C on the shared
    actor-learner runtime (reference counterpart: parallel_a3c.py:71-513
    scaled past hogwild)."""

    algo_name: str = "a3c-gpu"
    env_id: str = "synthetic-atari"  # ALE Pong when gymnasium is present
    num_actors: int = _h("CPU actor processes", default=16)
    envs_per_actor: int = _h("vectorized envs per actor", default=8)
    rollout_steps: int = 20
    slots_per_batch: int = _h("rollout slots per learner batch", default=4)
    gae_lambda: float = 1.0
    learning_rate: float = 1e-4
    max_grad_norm: float = 50.0
    dtype: str = _h("learner compute dtype: bf16 | fp32", default="bf16")
    total_steps: int = _h("total env steps", default=1_000_000)
    disable_checkpoint: bool = _h("skip periodic checkpoints", default=False)
    output_dir: str = _h("checkpoint directory", default="work_dirs/a3c_gpu")


@dataclass
class ImpalaArguments(RLArguments):
    """IMPALA (fields the reference reads in impala_atari.py but never
    declares — SURVEY.md 'Broken-as-shipped' list)."""

    algo_name: str = "impala"
    env_id: str = "synthetic-atari"
    total_steps: int = _h("total env steps", default=1_000_000)
    rollout_length: int = _h("T: env steps per rollout slot", default=80)
    batch_size: int = _h("B: rollout columns per learner batch", default=32)
    num_buffers: int = _h("rollout slots in the trajectory store", default=0)  # 0 → auto
    num_actors: int = _h("actor processes", default=4)
    envs_per_actor: int = _h("vectorized envs per actor process", default=16)
    use_lstm: bool = _h("LSTM core in AtariNet", default=True)
    reward_clipping: str = _h("abs_one | none", default="abs_one")
    discounting: float = _h("γ", default=0.99)
    baseline_cost: float = _h("value-loss coefficient", default=0.5)
    entropy_cost: float = _h("entropy-bonus coefficient", default=0.0006)
    clip_rho_threshold: float = _h("V-trace ρ̄", default=1.0)
    clip_pg_rho_threshold: float = _h("V-trace ρ̄_pg", default=1.0)
    learning_rate: float = 6e-4
    rmsprop_alpha: float = _h("RMSProp smoothing α", default=0.99)
    rmsprop_eps: float = _h("RMSProp ε", default=0.01)
    rmsprop_momentum: float = _h("RMSProp momentum", default=0.0)
    disable_checkpoint: bool = _h("skip periodic checkpoints", default=False)
    output_dir: str = _h("checkpoint/log dir", default="work_dirs/impala")
    checkpoint_path: str = _h("explicit checkpoint file (empty → auto)", default="")
    inference: str = _h("actor inference placement: cpu | gpu", default="cpu")
    inference_worker: str = _h(
        "gpu-inference execution: 'process' = dedicated process (default; "
        "a same-process 'thread' variant exists but aborts with HSA "
        "exceptions on ROCm 7.2 — see profiles/README.md)",
        default="process")
    remote_actor_slots: int = _h("store slots reserved for remote-node "
                                 "actors (0 disables the TCP server)", default=0)
    remote_port: int = _h("TCP port for remote actor nodes (0 = ephemeral)", default=0)
    remote_publish_interval: int = _h("learn steps between TCP weight "
                                      "publications", default=10)
    dtype: str = _h("learner compute dtype: bf16 | fp32", default="bf16")
    use_graph: bool = _h("hipGraph-capture the learner step (capture "
                         "happens before the inference worker starts)",
                         default=False)
    actor_double_buffer: bool = _h(
        "two env groups per actor, interleaved to hide the GPU-inference "
        "round latency (SEED-style; gpu inference only)", default=False)


@dataclass
class ApexArguments(DQNArguments):
    """Ape-X distributed prioritized DQN (reference sketch: algorithms/apex/)."""

    algo_name: str = "apex"
    env_id: str = "synthetic-atari"
    num_actors: int = 8
    envs_per_actor: int = _h("vectorized envs per actor process", default=16)
    use_per: bool = True
    n_steps: int = 3
    per_beta_anneal_steps: int = _h("learner steps to anneal β→1", default=1_000_000)
    eps_base: float = _h("Ape-X per-actor ε = base^(1+i/(N-1)·alpha)", default=0.4)
    eps_alpha: float = _h("Ape-X ε exponent spread", default=7.0)
    buffer_size: int = 1_000_000
    publish_interval: int = _h("learner steps between weight publications", default=50)


@dataclass
class DDPPOArguments(RLArguments):
    """DD-PPO (decentralized distributed PPO; not in the reference code —
    README citation only — implemented fresh per SURVEY.md §7 step 9)."""

    algo_name: str = "ddppo"
    env_id: str = "synthetic-pointgoal"
    rollout_length: int = _h("T: steps per PPO rollout", default=128)
    num_envs: int = _h("envs per learner rank", default=16)
    ppo_epochs: int = _h("PPO epochs per rollout", default=2)
    num_minibatches: int = _h("minibatches per epoch", default=2)
    clip_eps: float = _h("PPO clip ε", default=0.2)
    gae_lambda: float = _h("GAE λ", default=0.95)
    value_loss_coef: float = 0.5
    entropy_coef: float = 0.01
    preemption_threshold: float = _h("DD-PPO straggler preemption fraction", default=0.6)
    learning_rate: float = 2.5e-4
