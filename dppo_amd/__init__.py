"""dppo_amd — an MI355X-native Distributed PPO (DPPO) training framework.

A from-scratch rebuild of the capability set of oswsnqc/Tensorflow-DPPO
(reference: /root/reference) designed MI355X-first:

- PyTorch-ROCm framework layer, one worker process per GPU
  (the reference's threads-in-one-tf.Session multi-tower scheme,
  reference main.py:31-58, becomes torch.distributed over RCCL/xGMI).
- Hand-written CDNA4 HIP kernels (gfx950) for the PPO hot path:
  fused rollout step (MLP forward + sampling + synthetic env),
  segmented GAE reverse scan + whitening, fused PPO loss fwd/bwd,
  fused flat-bucket Adam.
- RCCL collectives: flat-bucket gradient all-reduce(AVG), rank-0
  parameter broadcast, all-gather of the per-rank 9-float stats vector
  that carries the reference's Chief semantics (batch score sort,
  shared l_mul, global stop rule — reference Chief.py:51,80-87).

Public surface mirrors the reference:
  Chief(scope, parameter_dict, ...)   with .check(), .act(s)   (Chief.py:9,19,89)
  Worker(scope, parameter_dict, ...)  with .work(), .act(s)    (Worker.py:9,29,140)
  make_pdtype(action_space)                                    (Others/distributions.py:231)
  Model().FC(scope, ...)                                       (Model.py:7)
"""

__version__ = "0.1.0"

from .config import DPPOConfig, REFERENCE_DEFAULTS
from . import spaces
from .distributions import make_pdtype, validate_probtype
from .models.mlp import Model, PolicyValueMLP

__all__ = [
    "DPPOConfig",
    "REFERENCE_DEFAULTS",
    "spaces",
    "make_pdtype",
    "validate_probtype",
    "PPO",
    "Model",
    "PolicyValueMLP",
    "Chief",
    "Worker",
    "DPPOEngine",
    "__version__",
]


def __getattr__(name):
    # Chief/Worker pull in torch.distributed; import lazily so light-weight
    # users (e.g. kernels-only) do not pay for it.
    if name == "Chief":
        from .chief import Chief
        return Chief
    if name == "Worker":
        from .worker import Worker
        return Worker
    if name == "PPO":
        from .ppo import PPO
        return PPO
    if name == "DPPOEngine":
        from .trainer import DPPOEngine
        return DPPOEngine
    raise AttributeError(f"module 'dppo_amd' has no attribute {name!r}")
