"""Configuration for DPPO training.

Mirrors the reference's single literal config dict (reference main.py:12-29):
same keys, same defaults, consumed by the same components — plus the
MI355X-native extension keys (env batch size, model widths, device/dtype,
kernel toggles) the rebuild needs.

Reference key -> consumer map:
  GAME            Chief.py:10, Worker.py:10, main.py:67
  LEARNING_RATE   PPO.py:9,20
  ENTCOEFF        PPO.py:11,35
  VCOEFF          PPO.py:12,39
  CLIP_PARAM      PPO.py:10,19
  GAMMA/LAM       Worker.py:24-25,89-90
  SCHEDULE        Worker.py:23,77-80
  MAX/MIN_AC_EXP_RATE, AC_EXP_PERCENTAGE   Worker.py:19-22,140-144
  UPDATE_STEPS    Chief.py:15,64
  MAX_EPOCH_STEPS Worker.py:18,39
  EPOCH_MAX       Worker.py:17,22,80
  NUM_WORKERS     main.py:27,40, Chief.py:17,58, PPO.py:13,50
  LOG_FILE_PATH   main.py:28,46, Worker.py:27
  ENV_SAMPLE_ITERATIONS  phantom key (read Worker.py:26, never used) — kept
                         for dict-compat, ignored.
"""

from __future__ import annotations

import dataclasses
import json
import multiprocessing
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Tuple


def _default_num_workers() -> int:
    # reference main.py:27: multiprocessing.cpu_count(); in the rebuild a
    # "worker" is one GPU rank, so the launcher overrides this with
    # WORLD_SIZE. Kept for dict parity.
    return multiprocessing.cpu_count()


#: The reference's literal config dict, verbatim keys and defaults
#: (reference main.py:12-29).
REFERENCE_DEFAULTS: Dict[str, Any] = {
    "GAME": "CartPole-v0",
    "LEARNING_RATE": 2e-5,
    "ENTCOEFF": 0.01,
    "VCOEFF": 0.5,
    "CLIP_PARAM": 0.2,
    "GAMMA": 0.99,
    "LAM": 0.95,
    "SCHEDULE": "linear",
    "MAX_AC_EXP_RATE": 0.4,
    "MIN_AC_EXP_RATE": 0.15,
    "AC_EXP_PERCENTAGE": 1,
    "UPDATE_STEPS": 4,
    "MAX_EPOCH_STEPS": 100,
    "EPOCH_MAX": 500,
    "LOG_FILE_PATH": "./logs",
}


@dataclass
class DPPOConfig:
    # ---- reference keys (main.py:12-29), same names/defaults ----
    GAME: str = "CartPole-v0"
    LEARNING_RATE: float = 2e-5
    ENTCOEFF: float = 0.01
    VCOEFF: float = 0.5
    CLIP_PARAM: float = 0.2
    GAMMA: float = 0.99
    LAM: float = 0.95
    SCHEDULE: str = "linear"          # 'linear' | 'constant' (Worker.py:77-80)
    MAX_AC_EXP_RATE: float = 0.4
    MIN_AC_EXP_RATE: float = 0.15
    AC_EXP_PERCENTAGE: float = 1.0
    UPDATE_STEPS: int = 4
    MAX_EPOCH_STEPS: int = 100        # rollout length T per iteration
    EPOCH_MAX: int = 500
    NUM_WORKERS: int = field(default_factory=_default_num_workers)
    LOG_FILE_PATH: str = "./logs"
    ENV_SAMPLE_ITERATIONS: int = 1    # phantom reference key (Worker.py:26); unused

    # ---- hardcoded magic in the reference, surfaced as config ----
    STOP_EPOCH: int = 500             # stop threshold (Chief.py:86, hardcoded 500)
    HIDDEN_SIZES: Tuple[int, ...] = (16,)  # hidden width 16 (Model.py:12)
    INIT_STD: float = 0.01            # normc init std (Model.py:10)
    ACTIVATION: str = "relu"          # Model.py:12 relu; humanoid configs use tanh

    # ---- MI355X-native extensions ----
    NUM_ENVS: int = 64                # batched synthetic envs per rank
    SEED: int = 0
    DEVICE: str = "auto"              # 'auto' | 'cpu' | 'cuda'
    DTYPE: str = "float32"            # compute dtype: 'float32' | 'bfloat16'
    USE_HIP_KERNELS: str = "auto"     # 'auto' | 'always' | 'never'
    USE_GRAPHS: bool = True           # hipGraph-capture the rollout/update inner loops
    BROADCAST_INTERVAL: int = 64      # drift-guard param broadcast every N rounds
                                      # (replaces per-round assigns, Chief.py:67-70)
    ADV_EPS: float = 1e-8             # guard for the reference's unguarded
                                      # whitening divide (Worker.py:92)
    MAX_ROLLOUT_RETRIES: int = 16     # rollouts per round before a rank reports
                                      # an invalid batch (no completed episode,
                                      # Worker.py:135 push-guard analog)
    MINIBATCH_SIZE: int = 0           # 0 = full-batch updates (the reference's
                                      # scheme, Chief.py:64); >0 = sequential
                                      # minibatch chunks per update step
                                      # (BASELINE config 4's 4096-minibatch)
    BATCH_CURATION: bool = False      # reference Chief.py:33-53 parity: sort
                                      # every rank's batch by best episode
                                      # reward (logs[2]) and assign the top
                                      # batches one-per-rank, so a rank can
                                      # train on a BETTER rank's data; off =
                                      # rank-local batches (documented
                                      # deviation, PARITY.md §7)

    def __post_init__(self) -> None:
        if isinstance(self.HIDDEN_SIZES, list):
            self.HIDDEN_SIZES = tuple(self.HIDDEN_SIZES)
        if self.SCHEDULE not in ("linear", "constant"):
            raise ValueError(f"SCHEDULE must be 'linear' or 'constant', got {self.SCHEDULE!r}")
        if self.ACTIVATION not in ("relu", "tanh"):
            raise ValueError(f"ACTIVATION must be 'relu' or 'tanh', got {self.ACTIVATION!r}")
        if self.USE_HIP_KERNELS not in ("auto", "always", "never"):
            raise ValueError("USE_HIP_KERNELS must be 'auto'|'always'|'never'")

    # -- construction --------------------------------------------------
    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "DPPOConfig":
        """Build from a reference-style parameter_dict; unknown keys rejected."""
        names = {f.name for f in dataclasses.fields(cls)}
        unknown = set(d) - names
        if unknown:
            raise KeyError(f"unknown config keys: {sorted(unknown)}")
        return cls(**d)

    @classmethod
    def from_json(cls, path: str) -> "DPPOConfig":
        with open(path) as f:
            return cls.from_dict(json.load(f))

    def to_dict(self) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        d["HIDDEN_SIZES"] = list(self.HIDDEN_SIZES)
        return d

    def replace(self, **kw: Any) -> "DPPOConfig":
        return dataclasses.replace(self, **kw)

    # -- derived -------------------------------------------------------
    def resolve_device(self) -> str:
        if self.DEVICE != "auto":
            return self.DEVICE
        import torch

        return "cuda" if torch.cuda.is_available() else "cpu"

    def torch_dtype(self):
        import torch

        return {"float32": torch.float32, "bfloat16": torch.bfloat16}[self.DTYPE]


# ---------------------------------------------------------------------------
# Named environment shape presets.  There is no network access for real
# gym/mujoco; GAME names map to synthetic env shapes matching the named
# task's observation/action spaces (BASELINE.json configs).
# ---------------------------------------------------------------------------

#: name -> (obs_dim, kind, act_dim) ; kind 'discrete' -> Discrete(act_dim),
#: 'box' -> Box(act_dim), 'multidiscrete' -> MultiDiscrete(act_dim tuple),
#: 'multibinary' -> MultiBinary(act_dim) — the reference trains whatever
#: gym action space the env exposes (make_pdtype dispatch, reference
#: Others/distributions.py:231-243), so every family is trainable here
GAME_SHAPES: Dict[str, Tuple[int, str, object]] = {
    "CartPole-v0": (4, "discrete", 2),
    "Pendulum-v1": (3, "box", 1),
    "HalfCheetah-v4": (17, "box", 6),
    "Humanoid-v4": (376, "box", 17),
    "Wide-4096": (4096, "box", 256),
    "MultiLever-v0": (12, "multidiscrete", (3, 3, 4)),
    "BitFlipper-v0": (16, "multibinary", 8),
}


def game_spaces(game: str):
    """Observation/action spaces for a GAME preset (synthetic shapes)."""
    from . import spaces

    if game not in GAME_SHAPES:
        raise KeyError(f"unknown GAME {game!r}; known: {sorted(GAME_SHAPES)}")
    obs_dim, kind, act_dim = GAME_SHAPES[game]
    obs_space = spaces.Box(low=-float("inf"), high=float("inf"), shape=(obs_dim,))
    if kind == "discrete":
        act_space = spaces.Discrete(act_dim)
    elif kind == "multidiscrete":
        act_space = spaces.MultiDiscrete(list(act_dim))
    elif kind == "multibinary":
        act_space = spaces.MultiBinary(act_dim)
    else:
        act_space = spaces.Box(low=-1.0, high=1.0, shape=(act_dim,))
    return obs_space, act_space
