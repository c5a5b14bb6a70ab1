"""torch_util parity surface (reference Others/tf_util.py live pieces)."""

import torch

from dppo_amd.utils import torch_util as U


def test_clip_mean_sum_max_argmax():
    x = torch.tensor([[-2.0, 0.5], [3.0, -1.0]])
    torch.testing.assert_close(U.clip(x, -1, 1),
                               torch.tensor([[-1.0, 0.5], [1.0, -1.0]]))
    torch.testing.assert_close(U.mean(x), x.mean())
    torch.testing.assert_close(U.mean(x, axis=0), x.mean(0))
    torch.testing.assert_close(U.sum(x, axis=1), x.sum(1))
    torch.testing.assert_close(U.max(x, axis=1), x.max(1).values)
    assert U.argmax(x, axis=1).tolist() == [1, 0]


def test_normc_initializer():
    w = torch.empty(8, 20)
    U.normc_initializer(0.5)(w)
    torch.testing.assert_close(w.pow(2).sum(1).sqrt(), torch.full((8,), 0.5),
                               atol=1e-6, rtol=1e-5)


def test_get_set_flat_roundtrip():
    lin = torch.nn.Linear(4, 3)
    flat = U.get_flat(lin.parameters())
    assert flat.numel() == 4 * 3 + 3
    flat2 = torch.randn_like(flat)
    U.set_from_flat(lin.parameters(), flat2)
    torch.testing.assert_close(U.get_flat(lin.parameters()), flat2)


def test_flatgrad():
    lin = torch.nn.Linear(4, 2)
    x = torch.randn(8, 4)
    loss = lin(x).pow(2).sum()
    fg = U.flatgrad(loss, list(lin.parameters()))
    assert fg.numel() == 4 * 2 + 2
    # matches autograd
    gs = torch.autograd.grad(lin(x).pow(2).sum(), list(lin.parameters()))
    torch.testing.assert_close(fg, torch.cat([g.reshape(-1) for g in gs]))


def test_flatgrad_clip_norm():
    p = torch.nn.Parameter(torch.tensor([3.0, 4.0]))
    loss = (p * torch.tensor([30.0, 40.0])).sum()
    fg = U.flatgrad(loss, [p], clip_norm=1.0)
    assert abs(float(fg.norm()) - 1.0) < 1e-6


def test_save_load_reexported():
    from dppo_amd.checkpoint import save_state, load_state

    assert U.save_state is save_state and U.load_state is load_state
