"""ScaleRL-MI355X: an MI355X-native distributed deep-RL engine.

Brand-new implementation of the capability surface of jianzhnie/ScaleRL
(DQN / A3C / IMPALA / Ape-X / DD-PPO), designed AMD-first:

- PyTorch-ROCm for autograd/optimizer scaffolding and orchestration.
- Hand-written HIP/CDNA4 kernels (``scalerl_amd/ops/csrc``) for the learner
  hot path: V-trace, fused IMPALA losses, GAE / n-step scans, prioritized
  replay segment trees, fused TD losses, fused optimizers, LSTM cell.
- RCCL (``torch.distributed`` backend "nccl" on ROCm) over xGMI for learner
  data-parallelism and weight publication.

Layer map (mirrors SURVEY.md §1, re-homed on a single actor-learner
runtime instead of per-algorithm process topologies):

- :mod:`scalerl_amd.config`     — dataclass configs + CLI parsing
- :mod:`scalerl_amd.envs`       — env API, Atari wrappers, vectorized + synthetic envs
- :mod:`scalerl_amd.data`       — replay buffers (uniform / n-step / prioritized)
- :mod:`scalerl_amd.models`     — policy/value networks
- :mod:`scalerl_amd.ops`        — HIP kernels with pure-PyTorch CPU references
- :mod:`scalerl_amd.parallel`   — process groups, bucketed collectives, rollout transport
- :mod:`scalerl_amd.runtime`    — per-algorithm trainers on the shared runtime
- :mod:`scalerl_amd.trainer`    — generic on/off-policy training loops
- :mod:`scalerl_amd.utils`      — logging, schedulers, timing, checkpoints
"""

__version__ = "0.1.0"
