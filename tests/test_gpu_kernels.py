"""HIP kernel numerics vs plain-PyTorch fp32 references (gfx950).

Every kernel is compared against the eager reference path of the same op
(kernel-vs-torch tolerance tests).
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from dppo_amd.distributions import CategoricalPdType, DiagGaussianPdType
from dppo_amd.ops import hip_ext, require_hip_ext
from dppo_amd.ops.gae import gae_advantages_ref
from dppo_amd.ops.ppo_loss import PPOLossCoeffs, ppo_losses_ref


@pytest.fixture(scope="module")
def ext():
    return require_hip_ext()


def _e():
    return torch.empty(0, device="cuda")


def test_gae_scan_matches_ref(ext):
    T, E = 128, 512
    g = torch.Generator(device="cuda").manual_seed(0)
    r = torch.randn(T, E, device="cuda", generator=g)
    v = torch.randn(T, E, device="cuda", generator=g)
    d = (torch.rand(T, E, device="cuda", generator=g) < 0.05).float()
    boot = torch.randn(E, device="cuda", generator=g)
    adv, etr = ext.gae_scan(r, v, d, boot, 0.99, 0.95, True, 1e-8, _e(), _e())
    adv_ref, etr_ref = gae_advantages_ref(r, v, d, boot, 0.99, 0.95, True, 1e-8)
    torch.testing.assert_close(adv, adv_ref, atol=2e-4, rtol=2e-4)
    torch.testing.assert_close(etr, etr_ref, atol=1e-4, rtol=1e-4)


def test_gae_scan_long_rollout(ext):
    """BASELINE config 4 shape: 65k-step scans."""
    T, E = 65536, 64
    r = torch.randn(T, E, device="cuda")
    v = torch.randn(T, E, device="cuda")
    d = (torch.rand(T, E, device="cuda") < 0.01).float()
    boot = torch.randn(E, device="cuda")
    adv, etr = ext.gae_scan(r, v, d, boot, 0.99, 0.95, False, 1e-8, _e(), _e())
    adv_ref, etr_ref = gae_advantages_ref(r, v, d, boot, 0.99, 0.95, False)
    torch.testing.assert_close(adv, adv_ref, atol=5e-4, rtol=5e-4)


def test_gae_whiten_constant_guard(ext):
    T, E = 8, 64
    r = torch.ones(T, E, device="cuda")
    v = torch.zeros(T, E, device="cuda")
    d = torch.ones(T, E, device="cuda")
    boot = torch.zeros(E, device="cuda")
    adv, _ = ext.gae_scan(r, v, d, boot, 0.99, 0.95, True, 1e-8, _e(), _e())
    assert torch.isfinite(adv).all()


def _loss_case(B=8192, A=17, seed=0, clip=0.2):
    g = torch.Generator(device="cuda").manual_seed(seed)
    pdflat = torch.randn(B, 2 * A, device="cuda", generator=g) * 0.5
    oldflat = pdflat + 0.1 * torch.randn(B, 2 * A, device="cuda", generator=g)
    v = torch.randn(B, device="cuda", generator=g)
    oldv = v + 0.3 * torch.randn(B, device="cuda", generator=g)
    pdt = DiagGaussianPdType(A)
    with torch.no_grad():
        actions = pdt.pdfromflat(oldflat).sample()
    adv = torch.randn(B, device="cuda", generator=g)
    etr = torch.randn(B, device="cuda", generator=g)
    return pdt, pdflat, oldflat, v, oldv, actions, adv, etr


def test_ppo_loss_fwd_matches_ref(ext):
    pdt, pdflat, oldflat, v, oldv, a, adv, etr = _loss_case()
    clip, entc, vc = 0.2, 0.01, 0.5
    losses = ext.ppo_loss_gauss_fwd(pdflat, oldflat, v, oldv, a, adv, etr,
                                    clip, entc, vc)
    ref = ppo_losses_ref(pdt.pdfromflat(pdflat), pdt.pdfromflat(oldflat),
                         v, oldv, a, adv, etr, PPOLossCoeffs(clip, entc, vc))
    torch.testing.assert_close(losses[0], ref["policyLoss"], atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(losses[1], ref["entropyLoss"], atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(losses[2], ref["valueLoss"], atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(losses[3], ref["total_loss"], atol=2e-5, rtol=1e-4)


def test_ppo_loss_bwd_matches_autograd(ext):
    pdt, pdflat, oldflat, v, oldv, a, adv, etr = _loss_case(B=4096, seed=3)
    clip, entc, vc = 0.2, 0.01, 0.5
    pdflat_r = pdflat.clone().requires_grad_(True)
    v_r = v.clone().requires_grad_(True)
    ref = ppo_losses_ref(pdt.pdfromflat(pdflat_r), pdt.pdfromflat(oldflat),
                         v_r, oldv, a, adv, etr, PPOLossCoeffs(clip, entc, vc))
    ref["total_loss"].backward()
    gt = torch.ones((), device="cuda")
    g_pdflat, g_v = ext.ppo_loss_gauss_bwd(pdflat, oldflat, v, oldv, a, adv,
                                           etr, clip, entc, vc, gt)
    torch.testing.assert_close(g_pdflat, pdflat_r.grad, atol=1e-6, rtol=1e-4)
    torch.testing.assert_close(g_v, v_r.grad, atol=1e-6, rtol=1e-4)


def test_ppo_loss_autograd_function_end_to_end(ext):
    """The autograd.Function wrapper routes gradients like the eager path."""
    from dppo_amd.ops.ppo_loss import ppo_losses

    pdt, pdflat, oldflat, v, oldv, a, adv, etr = _loss_case(B=2048, seed=5)
    coeffs = PPOLossCoeffs(0.2, 0.01, 0.5)

    p1 = pdflat.clone().requires_grad_(True)
    v1 = v.clone().requires_grad_(True)
    out = ppo_losses(pdt.pdfromflat(p1), pdt.pdfromflat(oldflat), v1, oldv,
                     a, adv, etr, coeffs, policy="always")
    out["total_loss"].backward()

    p2 = pdflat.clone().requires_grad_(True)
    v2 = v.clone().requires_grad_(True)
    ref = ppo_losses_ref(pdt.pdfromflat(p2), pdt.pdfromflat(oldflat), v2, oldv,
                         a, adv, etr, coeffs)
    ref["total_loss"].backward()

    torch.testing.assert_close(out["total_loss"], ref["total_loss"],
                               atol=2e-5, rtol=1e-4)
    torch.testing.assert_close(p1.grad, p2.grad, atol=1e-6, rtol=1e-4)
    torch.testing.assert_close(v1.grad, v2.grad, atol=1e-6, rtol=1e-4)


def _cat_case(B=8192, K=6, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    lpi = torch.randn(B, K, device="cuda", generator=g)
    lold = lpi + 0.1 * torch.randn(B, K, device="cuda", generator=g)
    v = torch.randn(B, device="cuda", generator=g)
    oldv = v + 0.3 * torch.randn(B, device="cuda", generator=g)
    pdt = CategoricalPdType(K)
    with torch.no_grad():
        actions = pdt.pdfromflat(lold).sample()
    adv = torch.randn(B, device="cuda", generator=g)
    etr = torch.randn(B, device="cuda", generator=g)
    return pdt, lpi, lold, v, oldv, actions, adv, etr


def test_ppo_cat_loss_fwd_matches_ref(ext):
    pdt, lpi, lold, v, oldv, a, adv, etr = _cat_case()
    clip, entc, vc = 0.2, 0.01, 0.5
    losses = ext.ppo_loss_cat_fwd(lpi, lold, v, oldv, a, adv, etr,
                                  clip, entc, vc)
    ref = ppo_losses_ref(pdt.pdfromflat(lpi), pdt.pdfromflat(lold),
                         v, oldv, a, adv, etr, PPOLossCoeffs(clip, entc, vc))
    torch.testing.assert_close(losses[0], ref["policyLoss"], atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(losses[1], ref["entropyLoss"], atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(losses[2], ref["valueLoss"], atol=1e-5, rtol=1e-4)
    torch.testing.assert_close(losses[3], ref["total_loss"], atol=2e-5, rtol=1e-4)


def test_ppo_cat_loss_bwd_matches_autograd(ext):
    pdt, lpi, lold, v, oldv, a, adv, etr = _cat_case(B=4096, K=9, seed=3)
    clip, entc, vc = 0.2, 0.01, 0.5
    lpi_r = lpi.clone().requires_grad_(True)
    v_r = v.clone().requires_grad_(True)
    ref = ppo_losses_ref(pdt.pdfromflat(lpi_r), pdt.pdfromflat(lold),
                         v_r, oldv, a, adv, etr, PPOLossCoeffs(clip, entc, vc))
    ref["total_loss"].backward()
    gt = torch.ones((), device="cuda")
    g_logits, g_v = ext.ppo_loss_cat_bwd(lpi, lold, v, oldv, a, adv, etr,
                                         clip, entc, vc, gt)
    torch.testing.assert_close(g_logits, lpi_r.grad, atol=1e-6, rtol=1e-4)
    torch.testing.assert_close(g_v, v_r.grad, atol=1e-6, rtol=1e-4)


def test_ppo_cat_autograd_function_end_to_end(ext):
    """ppo_losses dispatches CategoricalPd on GPU to the fused kernel and
    its gradients match the eager path (VERDICT r01 missing #4)."""
    from dppo_amd.ops.ppo_loss import ppo_losses

    pdt, lpi, lold, v, oldv, a, adv, etr = _cat_case(B=2048, K=2, seed=5)
    coeffs = PPOLossCoeffs(0.2, 0.01, 0.5)
    p1 = lpi.clone().requires_grad_(True)
    v1 = v.clone().requires_grad_(True)
    out = ppo_losses(pdt.pdfromflat(p1), pdt.pdfromflat(lold), v1, oldv,
                     a, adv, etr, coeffs, policy="always")
    out["total_loss"].backward()
    p2 = lpi.clone().requires_grad_(True)
    v2 = v.clone().requires_grad_(True)
    ref = ppo_losses_ref(pdt.pdfromflat(p2), pdt.pdfromflat(lold), v2, oldv,
                         a, adv, etr, coeffs)
    ref["total_loss"].backward()
    torch.testing.assert_close(out["total_loss"], ref["total_loss"],
                               atol=2e-5, rtol=1e-4)
    torch.testing.assert_close(p1.grad, p2.grad, atol=1e-6, rtol=1e-4)
    torch.testing.assert_close(v1.grad, v2.grad, atol=1e-6, rtol=1e-4)


def test_cat_sample_distribution(ext):
    """Gumbel-max sample kernel: empirical frequencies match softmax(logits)
    within Monte-Carlo error (the reference's statistical-identity test
    pattern, distributions.py:269-295)."""
    K, N = 5, 200_000
    logits = torch.tensor([0.2, -1.0, 0.5, 1.3, -0.3], device="cuda")
    rows = logits.expand(N, K).contiguous()
    a = ext.cat_sample(rows, 1234, 1)
    assert a.dtype == torch.int64 and a.shape == (N,)
    freq = torch.bincount(a, minlength=K).float() / N
    p = torch.softmax(logits, dim=-1)
    torch.testing.assert_close(freq, p, atol=5e-3, rtol=0.05)
    # different counters decorrelate
    a2 = ext.cat_sample(rows, 1234, 2)
    assert not torch.equal(a, a2)


def test_adam_matches_torch(ext):
    n = 100_003  # odd size exercises the float4 tail
    p = torch.randn(n, device="cuda")
    gref = torch.randn(n, device="cuda")

    p1 = p.clone().requires_grad_(True)
    p1.grad = gref.clone()
    opt = torch.optim.Adam([p1], lr=1e-3)

    p2 = p.clone()
    m = torch.zeros(n, device="cuda")
    vv = torch.zeros(n, device="cuda")
    for step in range(1, 4):
        opt.step()
        ext.adam_step(p2, gref, m, vv, step, 1e-3, 0.9, 0.999, 1e-8)
    torch.testing.assert_close(p2, p1.detach(), atol=2e-6, rtol=2e-5)
    st = opt.state_dict()["state"][0]
    # fp32 fma-vs-separate rounding differences in the moment updates
    torch.testing.assert_close(m, st["exp_avg"], atol=2e-6, rtol=2e-5)
    torch.testing.assert_close(vv, st["exp_avg_sq"], atol=2e-6, rtol=2e-5)


def test_native_extension_is_loaded():
    """The ops layer must be running the native path on GPU, not a silent
    eager fallback."""
    import _dppo_hip  # built in-tree; import must succeed on a GPU box

    assert hip_ext() is not None
    assert hasattr(_dppo_hip, "gae_scan")


def test_adam_dev_matches_torch(ext):
    """The graph-replayable Adam (device step/lr/bias-corrections) matches
    torch.optim.Adam like the host-arg variant."""
    n = 50_001
    p = torch.randn(n, device="cuda")
    gref = torch.randn(n, device="cuda")
    p1 = p.clone().requires_grad_(True)
    p1.grad = gref.clone()
    opt = torch.optim.Adam([p1], lr=2e-3)
    p2 = p.clone()
    m = torch.zeros(n, device="cuda")
    vv = torch.zeros(n, device="cuda")
    step_dev = torch.zeros(1, device="cuda", dtype=torch.int32)
    lr_dev = torch.tensor([2e-3], device="cuda")
    coef = torch.zeros(3, device="cuda")
    for _ in range(3):
        opt.step()
        ext.adam_step_dev(p2, gref, m, vv, step_dev, lr_dev, coef,
                          0.9, 0.999, 1e-8)
    assert int(step_dev.item()) == 3
    torch.testing.assert_close(p2, p1.detach(), atol=2e-6, rtol=2e-5)


def test_ppo_loss_fwd_tile_matches_legacy(ext, monkeypatch):
    """The LDS-tiled forward and the legacy wave-per-row kernel agree
    (same math, different reduction order; DPPO_GH_TILE picks the path
    per call)."""
    _, pdflat, oldflat, v, oldv, a, adv, etr = _loss_case(B=8192 + 37, seed=7)
    monkeypatch.setenv("DPPO_GH_TILE", "1")
    lt = ext.ppo_loss_gauss_fwd(pdflat, oldflat, v, oldv, a, adv, etr,
                                0.2, 0.01, 0.5)
    monkeypatch.setenv("DPPO_GH_TILE", "0")
    ll = ext.ppo_loss_gauss_fwd(pdflat, oldflat, v, oldv, a, adv, etr,
                                0.2, 0.01, 0.5)
    torch.testing.assert_close(lt, ll, atol=2e-6, rtol=1e-5)


def test_gh_tile_matches_legacy(ext, monkeypatch):
    """ppo_gh_tile_kernel vs the wave-per-row gh kernel: per-row grads
    are bitwise-comparable up to fp reassociation of the A-dim sums."""
    _, pdflat, oldflat, v, oldv, a, adv, etr = _loss_case(B=4096 + 13, seed=9)
    cde = torch.empty(0, device="cuda")
    monkeypatch.setenv("DPPO_GH_TILE", "1")
    gt = ext.ppo_loss_gauss_gh(pdflat, oldflat, v, oldv, a, adv, etr,
                               0.2, 0.01, 0.5, cde)
    monkeypatch.setenv("DPPO_GH_TILE", "0")
    gl = ext.ppo_loss_gauss_gh(pdflat, oldflat, v, oldv, a, adv, etr,
                               0.2, 0.01, 0.5, cde)
    torch.testing.assert_close(gt, gl, atol=1e-6, rtol=1e-5)
