"""bf16 MFMA GEMM kernels (bf16_gemm.hip) vs fp32 torch references.

These kernels are the hand-written hot path of the wide config
(BASELINE #5): forward with fused tanh+bias, dgrad with fused dtanh,
dW into the f32 flat gradient, transpose+colsum, and the wide-policy
PPO gh kernel.  Random (asymmetric) operands everywhere so output/operand
transposes are caught (guide §5.4 rule 16).
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from dppo_amd.ops import require_hip_ext
from dppo_amd.ops.ppo_loss import PPOLossCoeffs, ppo_losses_ref
from dppo_amd.distributions import DiagGaussianPdType


@pytest.fixture(scope="module")
def ext():
    return require_hip_ext()


def _e(dtype=torch.float32):
    return torch.empty(0, device="cuda", dtype=dtype)


def _eb():
    return _e(torch.bfloat16)


def _mm_case(M, N, K, seed=0, scale=1.0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    A = (torch.randn(M, K, device="cuda", generator=g) * scale).bfloat16()
    B = (torch.randn(N, K, device="cuda", generator=g) * scale).bfloat16()
    return A, B


def _close(out_f32, ref_f32, K):
    # bf16 inputs + f32 MFMA accumulation vs f32 torch on the bf16-rounded
    # inputs: per-element error ~ bf16 eps * sqrt(K) * |row||col|
    tol = 4e-3 * max(out_f32.abs().max().item(), 1.0)
    torch.testing.assert_close(out_f32, ref_f32, atol=tol, rtol=2e-2)


def test_mm256_raw(ext):
    M, N, K = 512, 512, 320
    A, B = _mm_case(M, N, K, seed=1, scale=0.3)
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    ext.bf16_mm256(A, B, C, 0, _e(), _eb(), _e(), 0, _eb(), 0, _e(), 0)
    ref = A.float() @ B.float().t()
    _close(C.float(), ref, K)


def test_mm256_large_k(ext):
    # dW-shaped: big contraction dim
    M, N, K = 256, 256, 8192
    A, B = _mm_case(M, N, K, seed=2, scale=0.1)
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    ext.bf16_mm256(A, B, C, 0, _e(), _eb(), _e(), 0, _eb(), 0, _e(), 0)
    ref = A.float() @ B.float().t()
    _close(C.float(), ref, K)


def test_mm256_tanh_bias(ext):
    M, N, K = 256, 512, 128
    A, B = _mm_case(M, N, K, seed=3, scale=0.3)
    bias = torch.randn(N, device="cuda")
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    ext.bf16_mm256(A, B, C, 1, bias, _eb(), _e(), 0, _eb(), 0, _e(), 0)
    ref = torch.tanh(A.float() @ B.float().t() + bias)
    torch.testing.assert_close(C.float(), ref, atol=1e-2, rtol=2e-2)


def test_mm256_dtanh(ext):
    M, N, K = 256, 256, 192
    A, B = _mm_case(M, N, K, seed=4, scale=0.3)
    h = torch.tanh(torch.randn(M, N, device="cuda")).bfloat16()
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    ext.bf16_mm256(A, B, C, 2, _e(), h, _e(), 0, _eb(), 0, _e(), 0)
    ref = (A.float() @ B.float().t()) * (1.0 - h.float() ** 2)
    _close(C.float(), ref, K)


def test_mm256_grad_accum(ext):
    M, N, K = 256, 512, 256
    A, B = _mm_case(M, N, K, seed=5, scale=0.2)
    off = 128
    grad = torch.zeros(off + M * N + 16, device="cuda")
    ext.bf16_mm256(A, B, _eb(), 3, _e(), _eb(), grad, off, _eb(), 0, _e(), 0)
    ref = A.float() @ B.float().t()
    _close(grad[off:off + M * N].view(M, N), ref, K)
    assert grad[:off].abs().sum() == 0


def test_mm256_dual_write_transpose_colsum(ext):
    """The epilogue's transposed dual-write + fused column sums (the dW
    operand and bias-grad producers) match the straight output."""
    M, N, K = 512, 256, 128
    A, B = _mm_case(M, N, K, seed=12, scale=0.3)
    h = torch.tanh(torch.randn(M, N, device="cuda")).bfloat16()
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    ldt = M + 64
    CT = torch.zeros(N, ldt, device="cuda", dtype=torch.bfloat16)
    sums = torch.zeros(N, device="cuda")
    ext.bf16_mm256(A, B, C, 2, _e(), h, _e(), 0, CT, ldt, sums, 0)
    ref = (A.float() @ B.float().t()) * (1.0 - h.float() ** 2)
    _close(C.float(), ref, K)
    torch.testing.assert_close(CT[:, :M].float(), C.float().t())
    torch.testing.assert_close(sums, ref.sum(dim=0), atol=0.3, rtol=1e-2)


def test_mm_small_ragged(ext):
    M, N, K = 100, 513, 544
    A, B = _mm_case(M, N, K, seed=6, scale=0.3)
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    ext.bf16_mm_small(A, B, C, _eb(), _eb(), _e(), 0, 0, 0, 0, M, N, N, _e())
    ref = A.float() @ B.float().t()
    _close(C.float(), ref, K)


def test_mm_small_heads_split(ext):
    # heads forward: N = P+1, last column is the value head
    M, P, K = 200, 512, 4096
    A, B = _mm_case(M, P + 1, K, seed=7, scale=0.05)
    B_heads = B  # [P+1][K]
    pdflat = torch.empty(M, P, device="cuda", dtype=torch.bfloat16)
    v = torch.empty(M, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(P + 1, device="cuda")
    ext.bf16_mm_small(A, B_heads, pdflat, v, _eb(), _e(), 0, 0, 0, 4,
                      M, P + 1, P, bias)
    ref = A.float() @ B_heads.float().t() + bias
    _close(pdflat.float(), ref[:, :P], K)
    _close(v.float(), ref[:, P], K)


def test_mm_small_grad_split(ext):
    # heads dW: rows < srow -> Wp grad, row == srow -> Wv grad
    M, N, K = 513, 256, 2048
    A, B = _mm_case(M, N, K, seed=8, scale=0.1)
    g1_off, g2_off = 64, 64 + 512 * N
    grad = torch.zeros(g2_off + N + 8, device="cuda")
    ext.bf16_mm_small(A, B, _eb(), _eb(), _eb(), grad, g1_off, g2_off,
                      512, 5, M, N, 0, _e())
    ref = A.float() @ B.float().t()
    _close(grad[g1_off:g1_off + 512 * N].view(512, N), ref[:512], K)
    _close(grad[g2_off:g2_off + N], ref[512], K)


def test_transpose_colsum(ext):
    R, C = 1000, 513
    g = torch.Generator(device="cuda").manual_seed(9)
    x = torch.randn(R, C, device="cuda", generator=g).bfloat16()
    ldo = 1024
    out = torch.zeros(C, ldo, device="cuda", dtype=torch.bfloat16)
    sums = torch.zeros(C + 8, device="cuda")
    ext.bf16_transpose(x, out, sums, 4, R, C, C, ldo)
    torch.testing.assert_close(out[:, :R].float(), x.float().t())
    torch.testing.assert_close(sums[4:4 + C], x.float().sum(dim=0),
                               atol=1e-2, rtol=1e-3)
    assert sums[:4].abs().sum() == 0


def test_gauss_gh_wide_matches_autograd(ext):
    B, A = 2048, 256
    g = torch.Generator(device="cuda").manual_seed(10)
    pdflat = torch.randn(B, 2 * A, device="cuda", generator=g) * 0.5
    pdflat_bf = pdflat.bfloat16()
    pdflat_f = pdflat_bf.float()  # the values the kernel actually sees
    oldflat = pdflat_f + 0.05 * torch.randn(B, 2 * A, device="cuda", generator=g)
    v = torch.randn(B, device="cuda", generator=g)
    v_bf = v.bfloat16()
    oldv = v_bf.float() + 0.2 * torch.randn(B, device="cuda", generator=g)
    pdt = DiagGaussianPdType(A)
    with torch.no_grad():
        act = pdt.pdfromflat(oldflat).sample()
    adv = torch.randn(B, device="cuda", generator=g)
    etr = torch.randn(B, device="cuda", generator=g)
    clip, entc, vc = 0.2, 0.01, 0.5

    gh = torch.zeros(B, 2 * A + 1 + 63, device="cuda", dtype=torch.bfloat16)
    ext.gauss_gh_wide(pdflat_bf, oldflat, v_bf, oldv, act, adv, etr, gh,
                      _e(), clip, entc, vc)

    p_r = pdflat_f.clone().requires_grad_(True)
    v_r = v_bf.float().clone().requires_grad_(True)
    ref = ppo_losses_ref(pdt.pdfromflat(p_r), pdt.pdfromflat(oldflat),
                         v_r, oldv, act, adv, etr,
                         PPOLossCoeffs(clip, entc, vc))
    ref["total_loss"].backward()
    scale = p_r.grad.abs().max().item()
    torch.testing.assert_close(gh[:, :2 * A].float(), p_r.grad,
                               atol=2e-2 * scale, rtol=2e-2)
    vscale = v_r.grad.abs().max().item()
    torch.testing.assert_close(gh[:, 2 * A].float(), v_r.grad,
                               atol=2e-2 * vscale, rtol=2e-2)
    assert gh[:, 2 * A + 1:].abs().sum() == 0  # padding untouched


def test_gauss_fwd_wide_action_dim(ext):
    """A=256 (wide config) must accumulate EVERY action dim — the A<=64
    wave kernel silently dropped dims past 64 in round 1."""
    B, A = 4096, 256
    g = torch.Generator(device="cuda").manual_seed(11)
    pdflat = torch.randn(B, 2 * A, device="cuda", generator=g) * 0.5
    oldflat = pdflat + 0.05 * torch.randn(B, 2 * A, device="cuda", generator=g)
    v = torch.randn(B, device="cuda", generator=g)
    oldv = v + 0.2 * torch.randn(B, device="cuda", generator=g)
    pdt = DiagGaussianPdType(A)
    with torch.no_grad():
        act = pdt.pdfromflat(oldflat).sample()
    adv = torch.randn(B, device="cuda", generator=g)
    etr = torch.randn(B, device="cuda", generator=g)
    losses = ext.ppo_loss_gauss_fwd(pdflat, oldflat, v, oldv, act, adv, etr,
                                    0.2, 0.01, 0.5)
    ref = ppo_losses_ref(pdt.pdfromflat(pdflat), pdt.pdfromflat(oldflat),
                         v, oldv, act, adv, etr, PPOLossCoeffs(0.2, 0.01, 0.5))
    torch.testing.assert_close(losses[3], ref["total_loss"], atol=1e-4,
                               rtol=1e-3)
    torch.testing.assert_close(losses[1], ref["entropyLoss"], atol=1e-4,
                               rtol=1e-3)
