"""End-to-end GPU training tests (single MI355X)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from dppo_amd.config import DPPOConfig
from dppo_amd.parallel.comm import Comm
from dppo_amd.trainer import DPPOEngine, RolloutBatch


def _cfg(**kw):
    base = dict(
        GAME="Humanoid-v4", HIDDEN_SIZES=(64, 64), ACTIVATION="tanh",
        NUM_ENVS=256, MAX_EPOCH_STEPS=16, EPOCH_MAX=1000, STOP_EPOCH=1000,
        LEARNING_RATE=3e-4, NUM_WORKERS=1,
        LOG_FILE_PATH="/tmp/dppo_gpu_test_logs", DEVICE="cuda",
    )
    base.update(kw)
    return DPPOConfig(**base)


def test_gpu_round_runs():
    eng = DPPOEngine(_cfg(), comm=Comm(device="cuda:0"))
    p0 = eng.flat_pi.flat_param.detach().clone()
    stats, stop = eng.train_round()
    torch.cuda.synchronize()
    assert not stop
    assert all(math.isfinite(v) for v in stats.values())
    assert not torch.equal(p0, eng.flat_pi.flat_param.detach())


def test_gpu_uses_fused_adam():
    from dppo_amd.ops.adam import FusedFlatAdam

    eng = DPPOEngine(_cfg(), comm=Comm(device="cuda:0"))
    assert isinstance(eng.optimizer, FusedFlatAdam)


def test_hip_vs_eager_training_close():
    """Several rounds with the HIP loss/GAE/Adam kernels vs the eager path
    from identical init and IDENTICAL rollouts (both engines pinned to the
    eager rollout so torch RNG streams match) stay numerically close —
    isolates the update-path kernels."""
    torch.manual_seed(0)
    e1 = DPPOEngine(_cfg(USE_HIP_KERNELS="always", SEED=7), comm=Comm(device="cuda:0"))
    e1._can_fuse_rollout = lambda: False  # eager rollout, HIP update path
    torch.manual_seed(0)
    e2 = DPPOEngine(_cfg(USE_HIP_KERNELS="never", SEED=7), comm=Comm(device="cuda:0"))
    torch.testing.assert_close(e1.flat_pi.flat_param, e2.flat_pi.flat_param)

    for _ in range(3):
        torch.manual_seed(123)
        e1.env._noise_gen.manual_seed(99)
        s1, _ = e1.train_round()
        torch.manual_seed(123)
        e2.env._noise_gen.manual_seed(99)
        s2, _ = e2.train_round()
    # identical seeds -> identical rollouts -> near-identical updates
    torch.testing.assert_close(
        e1.flat_pi.flat_param, e2.flat_pi.flat_param, atol=5e-4, rtol=1e-3
    )
    assert abs(s1["total_loss"] - s2["total_loss"]) < 1e-2


def test_discrete_game_on_gpu():
    eng = DPPOEngine(
        _cfg(GAME="CartPole-v0", HIDDEN_SIZES=(16,), ACTIVATION="relu",
             NUM_ENVS=128),
        comm=Comm(device="cuda:0"),
    )
    stats, _ = eng.train_round()
    assert math.isfinite(stats["total_loss"])


def test_checkpoint_roundtrip_gpu(tmp_path):
    from dppo_amd.checkpoint import load_state, save_state

    eng = DPPOEngine(_cfg(), comm=Comm(device="cuda:0"))
    eng.train_round()
    path = str(tmp_path / "gpu_ckpt.pt")
    save_state(path, eng)
    eng2 = DPPOEngine(_cfg(SEED=55), comm=Comm(device="cuda:0"))
    load_state(path, eng2)
    torch.testing.assert_close(eng2.flat_pi.flat_param, eng.flat_pi.flat_param)
    # fused Adam moments restored
    torch.testing.assert_close(eng2.optimizer.exp_avg, eng.optimizer.exp_avg)


def test_minibatched_graphed_matches_uncaptured():
    """The hipGraph-captured minibatch chunk walk (BASELINE config 4)
    replays the exact kernels of the uncaptured fused loop: identical
    rollouts (same seeds) must produce bit-close parameters."""
    kw = dict(MINIBATCH_SIZE=512, NUM_ENVS=128, MAX_EPOCH_STEPS=32, SEED=11)
    torch.manual_seed(0)
    e1 = DPPOEngine(_cfg(USE_GRAPHS=True, **kw), comm=Comm(device="cuda:0"))
    torch.manual_seed(0)
    e2 = DPPOEngine(_cfg(USE_GRAPHS=False, **kw), comm=Comm(device="cuda:0"))
    for _ in range(3):
        e1.train_round()
        e2.train_round()
    torch.cuda.synchronize()
    assert getattr(e1, "_mb_graphs", None), "graphed minibatch path not taken"
    assert not getattr(e1, "_mb_graph_failed", False)
    torch.testing.assert_close(
        e1.flat_pi.flat_param, e2.flat_pi.flat_param, atol=1e-6, rtol=1e-6
    )


def test_minibatched_graphed_tail_chunk():
    """B % mb != 0: the tail chunk runs uncaptured after each epoch's
    replays and the round still steps Adam B//mb*... + tail times."""
    kw = dict(MINIBATCH_SIZE=768, NUM_ENVS=128, MAX_EPOCH_STEPS=32, SEED=13)
    eng = DPPOEngine(_cfg(USE_GRAPHS=True, **kw), comm=Comm(device="cuda:0"))
    s0 = eng.optimizer.step_count
    eng.train_round()
    torch.cuda.synchronize()
    B = 128 * 32
    n_chunks = (B + 768 - 1) // 768
    assert eng.optimizer.step_count == s0 + eng.cfg.UPDATE_STEPS * n_chunks


def test_chunk_kernel_update_matches_eager():
    """The fused single-kernel chunk step (ops/hip/mlp_train.hip) applied
    UPDATE_STEPS times on one recorded batch matches the eager
    autograd + torch.optim.Adam path from identical init."""
    torch.manual_seed(0)
    e1 = DPPOEngine(_cfg(USE_HIP_KERNELS="always", USE_GRAPHS=False, SEED=5),
                    comm=Comm(device="cuda:0"))
    e1.CHUNK_KERNEL_MAX_B = 1 << 30  # route ANY batch size to the chunk
    # kernel (production default routes batches <= CHUNK_KERNEL_MAX_B=8192)
    torch.manual_seed(0)
    e2 = DPPOEngine(_cfg(USE_HIP_KERNELS="never", SEED=5),
                    comm=Comm(device="cuda:0"))
    torch.testing.assert_close(e1.flat_pi.flat_param, e2.flat_pi.flat_param)
    batch, _ = e2.rollout_once()
    assert e1._can_chunk_kernel(batch.states.shape[0])
    e1.update(batch, 0.7)
    e2.update(batch, 0.7)
    torch.cuda.synchronize()
    torch.testing.assert_close(
        e1.flat_pi.flat_param, e2.flat_pi.flat_param, atol=5e-5, rtol=1e-4
    )


def test_chunk_kernel_single_hidden_layer():
    """Reference-default architecture (one hidden layer, width 16) through
    the chunk kernel: finite losses, params move."""
    eng = DPPOEngine(
        _cfg(GAME="Pendulum-v1", HIDDEN_SIZES=(16,), NUM_ENVS=128,
             MAX_EPOCH_STEPS=32, USE_GRAPHS=False),
        comm=Comm(device="cuda:0"))
    eng.CHUNK_KERNEL_MAX_B = 1 << 30  # route any batch size to it
    assert eng._can_chunk_kernel(128 * 32)
    p0 = eng.flat_pi.flat_param.detach().clone()
    stats, _ = eng.train_round()
    torch.cuda.synchronize()
    assert all(math.isfinite(v) for v in stats.values())
    assert not torch.equal(p0, eng.flat_pi.flat_param.detach())


def test_chunk_kernel_grad_mode_matches_autograd():
    """The chunk kernel's grad-only mode (the distributed path: summed
    gradient into flat_grad, Adam applied separately) matches autograd
    gradients on the same batch."""
    from dppo_amd.ops import hip_ext

    torch.manual_seed(0)
    eng = DPPOEngine(_cfg(USE_HIP_KERNELS="always", USE_GRAPHS=False, SEED=9),
                     comm=Comm(device="cuda:0"))
    eng.CHUNK_KERNEL_MAX_B = 1 << 30
    batch, _ = eng.rollout_once()
    n = 2048
    assert eng._can_chunk_kernel(n)

    # autograd reference on the same sub-batch
    sub = RolloutBatch(
        states=batch.states.narrow(0, 0, n),
        actions=batch.actions.narrow(0, 0, n),
        adv=batch.adv.narrow(0, 0, n), etr=batch.etr.narrow(0, 0, n),
        oldflat=batch.oldflat.narrow(0, 0, n),
        oldv=batch.oldv.narrow(0, 0, n), cur_lr=1.0,
        ep_count=batch.ep_count, ep_sum=batch.ep_sum,
        ep_sumsq=batch.ep_sumsq, ep_min=batch.ep_min, ep_max=batch.ep_max,
        valid=True,
    )
    eng.flat_pi.zero_grad()
    losses = eng._losses(sub, l_mul=1.0)
    losses["total_loss"].backward()
    ref_grad = eng.flat_pi.flat_grad.detach().clone()

    # kernel, grad-only mode (fuse_adam=False)
    c = eng.cfg
    opt = eng.optimizer
    eng.flat_pi.zero_grad()
    hip_ext().mlp_chunk_train(
        eng.flat_pi.flat_param.data,
        sub.states, sub.actions, sub.adv, sub.etr, sub.oldflat, sub.oldv,
        [sl.start for sl in eng.flat_pi.slices],
        [eng.obs_space.shape[0], *c.HIDDEN_SIZES],
        1 if c.ACTIVATION == "tanh" else 0,
        torch.empty(0, device=eng.device), c.CLIP_PARAM, c.ENTCOEFF,
        c.VCOEFF, eng._chunk_scratch(), opt.exp_avg, opt.exp_avg_sq,
        opt.step_dev, opt.lr_dev, opt.coef, eng.flat_pi.flat_grad,
        False, opt.betas[0], opt.betas[1], opt.eps)
    torch.cuda.synchronize()
    torch.testing.assert_close(
        eng.flat_pi.flat_grad, ref_grad, atol=2e-5, rtol=1e-4)
