import torch

from dppo_amd import spaces
from dppo_amd.models.mlp import Model, PolicyValueMLP, normc_init_


def test_normc_init_rows_normalized():
    w = torch.empty(32, 100)
    normc_init_(w, std=0.01)
    norms = w.pow(2).sum(dim=1).sqrt()
    torch.testing.assert_close(norms, torch.full((32,), 0.01), atol=1e-6, rtol=1e-5)


def test_forward_shapes_box():
    net = PolicyValueMLP(17, spaces.Box(-1, 1, (6,)), hidden_sizes=(64, 64),
                         activation="tanh")
    s = torch.randn(10, 17)
    v, flat = net(s)
    assert v.shape == (10,)
    assert flat.shape == (10, 12)  # DiagGaussian param = 2*act_dim
    vv, pd = net.pd(s)
    assert pd.sample().shape == (10, 6)


def test_forward_shapes_discrete():
    net = PolicyValueMLP(4, spaces.Discrete(2))  # reference default: hidden 16 relu
    assert len(net.hidden) == 1 and net.hidden[0].out_features == 16
    v, flat = net(torch.randn(7, 4))
    assert flat.shape == (7, 2)


def test_model_fc_api():
    """Model().FC returns (net, pdtype, params) — the reference's
    (predv, pd, para) analog (Model.py:7-18)."""
    m = Model()
    net, pdtype, params = m.FC(
        "Chiefpi", spaces.Box(-1, 1, (3,)), spaces.Box(-1, 1, (1,))
    )
    assert net.scope == "Chiefpi"
    assert pdtype.param_shape() == [2]
    assert len(params) == len(list(net.parameters()))
    assert all(p.requires_grad for p in params)


def test_biases_zero_init():
    net = PolicyValueMLP(8, spaces.Discrete(3))
    for m in [*net.hidden, net.vf, net.pi]:
        assert torch.all(m.bias == 0)


def test_deterministic_under_seed():
    torch.manual_seed(7)
    n1 = PolicyValueMLP(8, spaces.Discrete(3))
    torch.manual_seed(7)
    n2 = PolicyValueMLP(8, spaces.Discrete(3))
    for a, b in zip(n1.parameters(), n2.parameters()):
        torch.testing.assert_close(a, b)
