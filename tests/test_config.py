import pytest

from dppo_amd.config import DPPOConfig, REFERENCE_DEFAULTS, game_spaces
from dppo_amd import spaces


def test_reference_defaults_match():
    """Defaults mirror the reference's literal dict (reference main.py:12-29)."""
    cfg = DPPOConfig()
    for k, v in REFERENCE_DEFAULTS.items():
        assert getattr(cfg, k) == v, k


def test_from_dict_rejects_unknown():
    with pytest.raises(KeyError):
        DPPOConfig.from_dict({"NOT_A_KEY": 1})


def test_from_dict_accepts_reference_dict():
    d = dict(REFERENCE_DEFAULTS)
    d["NUM_WORKERS"] = 4
    d["ENV_SAMPLE_ITERATIONS"] = 7  # phantom reference key must be accepted
    cfg = DPPOConfig.from_dict(d)
    assert cfg.NUM_WORKERS == 4


def test_roundtrip():
    cfg = DPPOConfig(HIDDEN_SIZES=(64, 64), GAME="Humanoid-v4")
    cfg2 = DPPOConfig.from_dict(cfg.to_dict())
    assert cfg2.HIDDEN_SIZES == (64, 64)
    assert cfg2 == cfg


def test_schedule_validation():
    with pytest.raises(ValueError):
        DPPOConfig(SCHEDULE="cosine")


def test_game_spaces():
    obs, act = game_spaces("CartPole-v0")
    assert obs.shape == (4,) and isinstance(act, spaces.Discrete) and act.n == 2
    obs, act = game_spaces("Humanoid-v4")
    assert obs.shape == (376,) and isinstance(act, spaces.Box) and act.shape == (17,)
    with pytest.raises(KeyError):
        game_spaces("Doom")
