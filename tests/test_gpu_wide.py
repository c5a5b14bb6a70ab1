"""Wide bf16 config (BASELINE #5): hand MFMA path vs the eager autocast
reference — forward tolerance, gradient agreement, end-to-end training."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine


def _cfg(**kw):
    base = dict(
        GAME="Wide-4096", HIDDEN_SIZES=(4096, 4096, 4096, 4096),
        ACTIVATION="tanh", DTYPE="bfloat16", NUM_ENVS=256,
        MAX_EPOCH_STEPS=4, EPOCH_MAX=1000, STOP_EPOCH=1000,
        LEARNING_RATE=3e-4, NUM_WORKERS=1,
        LOG_FILE_PATH="/tmp/dppo_gpu_test_logs", DEVICE="cuda",
    )
    base.update(kw)
    return DPPOConfig(**base)


def test_wide_path_eligible():
    eng = DPPOEngine(_cfg(), comm=Comm(device="cuda:0"))
    assert eng._can_wide_bf16()
    assert not eng._can_fuse_update() and not eng._can_fuse_rollout()


def test_wide_forward_matches_autocast():
    torch.manual_seed(0)
    eng = DPPOEngine(_cfg(SEED=3), comm=Comm(device="cuda:0"))
    obs = torch.randn(256, 4096, device="cuda") * 0.5
    v1, pd1 = eng._wide().forward(obs)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        v2, pd2 = eng.pi(obs)
    v2, pd2 = v2.float(), pd2.float()
    # both paths are bf16 GEMMs (different reassociation); bf16 tolerance
    torch.testing.assert_close(pd1, pd2, atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(v1, v2, atol=3e-2, rtol=3e-2)


def test_wide_update_gradient_matches_autograd():
    """lr=0 wide update vs the eager autocast autograd gradient on the
    same batch: gradients agree in direction and magnitude (both are
    bf16-GEMM-noisy; cosine + relative-L2 check)."""
    torch.manual_seed(0)
    eng = DPPOEngine(_cfg(SEED=5, UPDATE_STEPS=1), comm=Comm(device="cuda:0"))
    batch, _ = eng.rollout_once()
    assert batch.valid

    # eager autograd reference gradient (autocast bf16 GEMMs, f32 loss)
    eng.flat_pi.zero_grad()
    losses = eng._losses(batch, l_mul=0.5)
    losses["total_loss"].backward()
    ref_grad = eng.flat_pi.flat_grad.detach().clone()

    # wide path, lr=0 so parameters stay put and the grad is inspectable
    for g in eng.optimizer.param_groups:
        g["lr"] = 0.0
    p0 = eng.flat_pi.flat_param.detach().clone()
    eng._wide().update(batch, l_mul=0.5)
    torch.cuda.synchronize()
    wide_grad = eng.flat_pi.flat_grad.detach().clone()
    torch.testing.assert_close(eng.flat_pi.flat_param, p0)  # lr=0

    cos = torch.nn.functional.cosine_similarity(
        wide_grad.unsqueeze(0), ref_grad.unsqueeze(0)).item()
    rel = (wide_grad - ref_grad).norm().item() / (ref_grad.norm().item() + 1e-12)
    assert cos > 0.99, f"cosine {cos}"
    assert rel < 0.1, f"relative L2 {rel}"


def test_wide_training_rounds():
    """Full rounds through the wide path: finite stats, params move,
    losses stay sane over several rounds."""
    eng = DPPOEngine(_cfg(SEED=7), comm=Comm(device="cuda:0"))
    p0 = eng.flat_pi.flat_param.detach().clone()
    for _ in range(3):
        stats, stop = eng.train_round()
    torch.cuda.synchronize()
    assert not stop
    assert all(math.isfinite(v) for v in stats.values())
    assert not torch.equal(p0, eng.flat_pi.flat_param.detach())
