// Fused PPO clipped-surrogate loss for DiagGaussian policies (gfx950).
//
// One forward kernel fuses everything the reference's graph does per
// update step between the network outputs and the scalar losses
// (reference PPO.py:29-40 + Others/distributions.py:195-203): both
// Gaussian log-probs (pi, oldpi), the importance ratio, the surrogate
// clip & min, the clipped value loss, the entropy, and the three
// block-reduced means (double accumulators, one atomic per wave).  One
// backward kernel recomputes the cheap per-sample quantities and writes
// analytic gradients for pdflat_pi = [mean, logstd] and vpred only
// (oldpi gets none — compute_gradients(total_loss, pipara), PPO.py:46).
//
// Memory shape: the forward and the gh gradient kernel are wave-per-row
// (lanes cooperate on the 2A distribution columns with coalesced row
// loads and wave-reduced log-prob sums); the row-per-thread backward
// keeps L1-resident rows.  All paths are tested against the eager
// reference to tolerance.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ppo_math.h"

namespace {

// Stage a [rows][width] contiguous global stream into padded LDS rows.
// Global side is float4 where width >= 8 (the caller guarantees the
// base is 16-B aligned: tile_base*width is a multiple of 4), LDS side
// is 4 scalar writes with an at-most-one row wrap per group.  row/col
// kept by increment — one div/mod per stream, not per element.
DEV_INLINE void stage_tile(const float* __restrict__ src,
                           float* __restrict__ dst, int width, int stride,
                           int rows, int tid) {
  const int n = rows * width;
  if (width >= 8) {
    const float4* s4 = reinterpret_cast<const float4*>(src);
    const int n4 = n >> 2;
    const int dr = 1024 / width, dc = 1024 % width;
    int r = (tid * 4) / width, c = (tid * 4) % width;
    for (int q = tid; q < n4; q += 256) {
      const float4 v = s4[q];
      int rr = r, cc = c;
      const float vv[4] = {v.x, v.y, v.z, v.w};
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        dst[rr * stride + cc] = vv[k];
        if (++cc == width) { cc = 0; ++rr; }
      }
      r += dr; c += dc;
      if (c >= width) { c -= width; ++r; }
    }
    for (int g = (n4 << 2) + tid; g < n; g += 256)
      dst[(g / width) * stride + g % width] = src[g];
  } else {
    const int dr = 256 / width, dc = 256 % width;
    for (int g = tid, r = tid / width, c = tid % width; g < n; g += 256) {
      dst[r * stride + c] = src[g];
      r += dr; c += dc;
      if (c >= width) { c -= width; r += 1; }
    }
  }
}

DEV_INLINE GaussRow gauss_row(const float* __restrict__ pdpi,
                              const float* __restrict__ pdold,
                              const float* __restrict__ act,
                              int64_t b, int A) {
  return ppo_gauss_row(pdpi, pdold, act, b, A);
}

__launch_bounds__(256)
__global__ void ppo_gauss_fwd_kernel(
    const float* __restrict__ pdpi, const float* __restrict__ pdold,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const float* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr,
    double* __restrict__ acc,  // [3] {policy_min_sum, ent_sum, value_max_sum}
    int64_t B, int A, float clip) {
  // wave-per-row (lanes cooperate on the 2A distribution columns with
  // coalesced row loads + wave reductions): the row-per-thread version
  // thrashed L1 across 64 concurrent 2A-float rows and ran ~10x slower.
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int P = 2 * A;
  const int64_t waves_total = (int64_t)gridDim.x * 4;
  const int64_t wid = (int64_t)blockIdx.x * 4 + wave;
  // CONTIGUOUS row chunk per wave (not wave-stride): wave-stride put the
  // resident 8192 waves ~1.1 MB apart in 5 arrays at once — tens of
  // thousands of concurrent DRAM streams, no row-buffer locality.
  const int64_t per = (B + waves_total - 1) / waves_total;
  const int64_t rb0 = wid * per;
  const int64_t rb1 = rb0 + per < B ? rb0 + per : B;

  float pol = 0.f, ent = 0.f, val = 0.f;  // lane 0 accumulates rows
  // three rows of operands in flight (same ping-pong as ppo_gh_kernel)
  struct Row { float mu, ls, aj, mo, lso, vp, ov, ad, et; };
  auto load_row = [&](int64_t b, Row& r) {
    if (lane < A) {
      r.mu = pdpi[b * P + lane];
      r.ls = pdpi[b * P + A + lane];
      r.aj = act[b * A + lane];
      r.mo = pdold[b * P + lane];
      r.lso = pdold[b * P + A + lane];
    }
    r.vp = vpred[b];
    r.ov = oldv[b];
    r.ad = adv[b];
    r.et = etr[b];
  };
  auto compute_row = [&](const Row& r) {
    float lp_part = 0.f, lo_part = 0.f, ent_part = 0.f;
    if (lane < A) {
      const float zp = (r.aj - r.mu) * __expf(-r.ls);
      lp_part = -0.5f * zp * zp - r.ls;
      const float zo = (r.aj - r.mo) * __expf(-r.lso);
      lo_part = -0.5f * zo * zo - r.lso;
      ent_part = r.ls;
    }
    const float lp = wave_reduce_sum(lp_part);
    const float lo = wave_reduce_sum(lo_part);
    const float es = wave_reduce_sum(ent_part);
    if (lane == 0) {
      const float ratio = __expf(lp - lo);  // the logp constants cancel
      const float ab = r.ad;
      const float surr1 = ratio * ab;
      const float rc = fminf(fmaxf(ratio, 1.f - clip), 1.f + clip);
      pol += fminf(surr1, rc * ab);
      ent += es + 0.5f * (PPO_LOG_2PI + 1.f) * A;
      const float d1 = r.vp - r.et;
      const float dc = fminf(fmaxf(r.vp - r.ov, -clip), clip);
      const float d2 = r.ov + dc - r.et;
      val += fmaxf(d1 * d1, d2 * d2);
    }
  };
  Row ra, rc_, re;
  int64_t b = rb0;
  if (b < rb1) load_row(b, ra);
  if (b + 1 < rb1) load_row(b + 1, rc_);
  if (b + 2 < rb1) load_row(b + 2, re);
  #pragma unroll 1
  for (; b + 5 < rb1; b += 3) {
    compute_row(ra);
    load_row(b + 3, ra);
    compute_row(rc_);
    load_row(b + 4, rc_);
    compute_row(re);
    load_row(b + 5, re);
  }
  if (b < rb1) { compute_row(ra); ++b; }
  if (b < rb1) { compute_row(rc_); ++b; }
  if (b < rb1) { compute_row(re); ++b; }
  for (; b < rb1; ++b) {
    load_row(b, ra);
    compute_row(ra);
  }
  if (lane == 0) {
    atomicAdd(&acc[0], static_cast<double>(pol));
    atomicAdd(&acc[1], static_cast<double>(ent));
    atomicAdd(&acc[2], static_cast<double>(val));
  }
}

// Wide-policy (A > 64) forward: same math, lane-strided column loop so
// every action dim is accumulated (the pipelined kernel above maps one
// column per lane and is dispatched only for A <= 64).
__launch_bounds__(256)
__global__ void ppo_gauss_fwd_wide_kernel(
    const float* __restrict__ pdpi, const float* __restrict__ pdold,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const float* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr, double* __restrict__ acc,
    int64_t B, int A, float clip) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int P = 2 * A;
  const int64_t waves_total = (int64_t)gridDim.x * 4;
  const int64_t wid = (int64_t)blockIdx.x * 4 + wave;
  const int64_t per = (B + waves_total - 1) / waves_total;
  const int64_t rb0 = wid * per;
  const int64_t rb1 = rb0 + per < B ? rb0 + per : B;

  float pol = 0.f, ent = 0.f, val = 0.f;
  for (int64_t b = rb0; b < rb1; ++b) {
    float lp_part = 0.f, lo_part = 0.f, ent_part = 0.f;
    const float* mu = pdpi + b * P;
    const float* mo = pdold + b * P;
    const float* aj = act + b * (int64_t)A;
    for (int j = lane; j < A; j += WAVE) {
      const float ls = mu[A + j];
      const float zp = (aj[j] - mu[j]) * __expf(-ls);
      lp_part += -0.5f * zp * zp - ls;
      const float lso = mo[A + j];
      const float zo = (aj[j] - mo[j]) * __expf(-lso);
      lo_part += -0.5f * zo * zo - lso;
      ent_part += ls;
    }
    const float lp = wave_reduce_sum(lp_part);
    const float lo = wave_reduce_sum(lo_part);
    const float es = wave_reduce_sum(ent_part);
    if (lane == 0) {
      const float ratio = __expf(lp - lo);
      const float ab = adv[b];
      const float surr1 = ratio * ab;
      const float rc = fminf(fmaxf(ratio, 1.f - clip), 1.f + clip);
      pol += fminf(surr1, rc * ab);
      ent += es + 0.5f * (PPO_LOG_2PI + 1.f) * A;
      const float d1 = vpred[b] - etr[b];
      const float dc = fminf(fmaxf(vpred[b] - oldv[b], -clip), clip);
      const float d2 = oldv[b] + dc - etr[b];
      val += fmaxf(d1 * d1, d2 * d2);
    }
  }
  if (lane == 0) {
    atomicAdd(&acc[0], static_cast<double>(pol));
    atomicAdd(&acc[1], static_cast<double>(ent));
    atomicAdd(&acc[2], static_cast<double>(val));
  }
}

// LDS-tiled row-per-lane forward (same staging pattern as
// ppo_gh_tile_kernel below: the wave-per-row kernel above keeps only A
// of 64 lanes loading and measured 0.88 TB/s).  A 256-thread block
// stages a 128-row tile coalesced, each of 128 lanes computes its row's
// surrogate/entropy/value terms serially, then one block reduction
// feeds the three double accumulators (one atomicAdd triple per block).
__launch_bounds__(256)
__global__ void ppo_gauss_fwd_tile_kernel(
    const float* __restrict__ pdpi, const float* __restrict__ pdold,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const float* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr, double* __restrict__ acc,
    int64_t B, int A, float clip) {
  constexpr int TILE = 128;
  extern __shared__ float lds[];
  const int P = 2 * A;
  const int sp = P + 1;
  const int sa = A | 1;
  float* l_pd = lds;
  float* l_og = l_pd + TILE * sp;
  float* l_ac = l_og + TILE * sp;
  float* l_sc = l_ac + TILE * sa;  // [4][TILE]: vpred|oldv|adv|etr
  const int tid = threadIdx.x;
  const int64_t ntiles = (B + TILE - 1) / TILE;

  // grid-stride over tiles, per-thread accumulators, ONE atomic triple
  // per block at the end: one atomic per TILE serialized 3*(B/128)
  // same-address f64 atomics at L2 and dominated the kernel
  float pol = 0.f, ent = 0.f, val = 0.f;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
  const int64_t tb = tile * TILE;
  const int rows = (int)min((int64_t)TILE, B - tb);

  stage_tile(pdpi + tb * P, l_pd, P, sp, rows, tid);
  stage_tile(pdold + tb * P, l_og, P, sp, rows, tid);
  stage_tile(act + tb * A, l_ac, A, sa, rows, tid);
  for (int idx = tid; idx < 4 * TILE; idx += 256) {
    const int s = idx / TILE, t = idx % TILE;
    if (t < rows) {
      const float* src = s == 0 ? vpred : s == 1 ? oldv : s == 2 ? adv : etr;
      l_sc[s * TILE + t] = src[tb + t];
    }
  }
  __syncthreads();

  if (tid < rows) {
    const float* pd = l_pd + tid * sp;
    const float* og = l_og + tid * sp;
    const float* ac = l_ac + tid * sa;
    float lp = 0.f, lo_ = 0.f, es = 0.f;
    for (int j = 0; j < A; ++j) {
      const float aj = ac[j];
      const float ls = pd[A + j];
      const float zp = (aj - pd[j]) * __expf(-ls);
      lp += -0.5f * zp * zp - ls;
      const float lso = og[A + j];
      const float zo = (aj - og[j]) * __expf(-lso);
      lo_ += -0.5f * zo * zo - lso;
      es += ls;
    }
    const float ratio = __expf(lp - lo_);  // the logp constants cancel
    const float ab = l_sc[2 * TILE + tid];
    const float surr1 = ratio * ab;
    const float rc = fminf(fmaxf(ratio, 1.f - clip), 1.f + clip);
    pol += fminf(surr1, rc * ab);
    ent += es + 0.5f * (PPO_LOG_2PI + 1.f) * A;
    const float vp = l_sc[tid], ov = l_sc[TILE + tid];
    const float et = l_sc[3 * TILE + tid];
    const float d1 = vp - et;
    const float dc = fminf(fmaxf(vp - ov, -clip), clip);
    const float d2 = ov + dc - et;
    val += fmaxf(d1 * d1, d2 * d2);
  }
  __syncthreads();  // lanes done reading before the next tile restages
  }
  // block reduce (wave sums -> LDS -> wave 0) then one atomic triple
  __syncthreads();  // staging regions reused for the 4x3 wave partials
  const int lane = tid & (WAVE - 1);
  const int wave = tid / WAVE;
  pol = wave_reduce_sum(pol);
  ent = wave_reduce_sum(ent);
  val = wave_reduce_sum(val);
  if (lane == 0) {
    lds[wave * 3 + 0] = pol;
    lds[wave * 3 + 1] = ent;
    lds[wave * 3 + 2] = val;
  }
  __syncthreads();
  if (tid == 0) {
    double p = 0.0, e = 0.0, v = 0.0;
    #pragma unroll
    for (int w = 0; w < 4; ++w) {
      p += lds[w * 3 + 0];
      e += lds[w * 3 + 1];
      v += lds[w * 3 + 2];
    }
    atomicAdd(&acc[0], p);
    atomicAdd(&acc[1], e);
    atomicAdd(&acc[2], v);
  }
}

__global__ void ppo_gauss_finalize_kernel(const double* __restrict__ acc,
                                          float* __restrict__ losses,  // [4]
                                          int64_t B, float entcoeff,
                                          float vcoeff) {
  const double ib = 1.0 / static_cast<double>(B);
  const float pol = static_cast<float>(-acc[0] * ib);
  const float ent = static_cast<float>(-entcoeff * acc[1] * ib);
  const float val = static_cast<float>(vcoeff * acc[2] * ib);
  losses[0] = pol;
  losses[1] = ent;
  losses[2] = val;
  losses[3] = pol + ent + val;
}

__global__ void ppo_gauss_bwd_kernel(
    const float* __restrict__ pdpi, const float* __restrict__ pdold,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const float* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr,
    const float* __restrict__ gtotal,   // [1] upstream d(total_loss)
    float* __restrict__ g_pdflat,       // [B, 2A]
    float* __restrict__ g_v,            // [B]
    int64_t B, int A, float clip, float entcoeff, float vcoeff) {
  const float g = gtotal[0];
  for (int64_t b = gidx(); b < B; b += gstride()) {
    const GaussRow r = gauss_row(pdpi, pdold, act, b, A);
    const float ratio = __expf(r.logp_pi - r.logp_old);
    const float ab = adv[b];
    const float surr1 = ratio * ab;
    const float lo = 1.f - clip, hi = 1.f + clip;
    const float rc = fminf(fmaxf(ratio, lo), hi);
    const float surr2 = rc * ab;
    // d policyLoss / d logp_pi.  min() routes the gradient to the branch
    // that attains the min; when the clip branch wins, clamp passes
    // gradient only strictly inside [lo, hi] — matching torch's
    // min/clamp subgradients on the measure-one set (ties are split by
    // torch but occur on a measure-zero set; tolerance tests cover it).
    float flow;
    if (surr1 <= surr2) {
      flow = 1.f;
    } else {
      flow = (ratio >= lo && ratio <= hi) ? 1.f : 0.f;
    }
    const float g_logp = -g / static_cast<float>(B) * ab * ratio * flow;
    const float g_ent = -g * entcoeff / static_cast<float>(B);

    const float* mu_pi = pdpi + (int64_t)b * 2 * A;
    const float* ls_pi = mu_pi + A;
    const float* a = act + (int64_t)b * A;
    float* gm = g_pdflat + (int64_t)b * 2 * A;
    float* gs = gm + A;
    for (int j = 0; j < A; ++j) {
      const float lsp = ls_pi[j];
      const float inv_s = __expf(-lsp);
      const float z = (a[j] - mu_pi[j]) * inv_s;
      // dlogp/dmu = z/std ; dlogp/dlogstd = z^2 - 1 ; dent/dlogstd = 1
      gm[j] = g_logp * z * inv_s;
      gs[j] = g_logp * (z * z - 1.f) + g_ent;
    }
    // value branch (PPO.py:36-39): max(vf1, vf2) with vf2 clipped.
    const float vb = vpred[b], ob = oldv[b], eb = etr[b];
    const float d1 = vb - eb;
    const float diff = vb - ob;
    const bool inside = (diff >= -clip && diff <= clip);
    const float dc = fminf(fmaxf(diff, -clip), clip);
    const float d2 = ob + dc - eb;
    const float vf1 = d1 * d1, vf2 = d2 * d2;
    float gv;
    if (inside) {
      // vclip == v: both branches equal; torch splits 0.5/0.5 — the sum
      // is the same single gradient 2*(v - etr).
      gv = 2.f * d1;
    } else if (vf1 >= vf2) {
      gv = 2.f * d1;
    } else {
      gv = 0.f;  // clipped branch is constant in v outside the clip window
    }
    g_v[b] = g * vcoeff / static_cast<float>(B) * gv;
  }
}

}  // namespace

torch::Tensor ppo_loss_gauss_fwd(torch::Tensor pdpi, torch::Tensor pdold,
                                 torch::Tensor vpred, torch::Tensor oldv,
                                 torch::Tensor act, torch::Tensor adv,
                                 torch::Tensor etr, double clip,
                                 double entcoeff, double vcoeff) {
  TORCH_CHECK(pdpi.is_cuda() && pdpi.dtype() == torch::kFloat32);
  TORCH_CHECK(pdpi.dim() == 2 && pdpi.size(1) % 2 == 0);
  const int64_t B = pdpi.size(0);
  const int A = static_cast<int>(pdpi.size(1) / 2);
  TORCH_CHECK(act.sizes() == torch::IntArrayRef({B, A}));
  TORCH_CHECK(vpred.numel() == B && adv.numel() == B && etr.numel() == B);

  auto acc = torch::zeros({3}, pdpi.options().dtype(torch::kFloat64));
  auto losses = torch::empty({4}, pdpi.options());
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int block = 256;
  const char* fte = getenv("DPPO_GH_TILE");  // same knob as the gh kernel
  const int fwd_tile_env = fte ? atoi(fte) : 1;
  const int fwd_lds =
      128 * (2 * (2 * A + 1) + (A | 1) + 4) * (int)sizeof(float);
  if (fwd_tile_env && A <= WAVE && fwd_lds <= 65536) {
    const int64_t grid = std::min<int64_t>(1024, (B + 127) / 128);
    hipLaunchKernelGGL(ppo_gauss_fwd_tile_kernel, dim3((unsigned)grid),
                       dim3(block), fwd_lds, stream, pdpi.data_ptr<float>(),
                       pdold.data_ptr<float>(), vpred.data_ptr<float>(),
                       oldv.data_ptr<float>(), act.data_ptr<float>(),
                       adv.data_ptr<float>(), etr.data_ptr<float>(),
                       acc.data_ptr<double>(), B, A, (float)clip);
    hipLaunchKernelGGL(ppo_gauss_finalize_kernel, dim3(1), dim3(1), 0,
                       stream, acc.data_ptr<double>(),
                       losses.data_ptr<float>(), B, (float)entcoeff,
                       (float)vcoeff);
    return losses;
  }
  auto* fwd = (A <= WAVE) ? &ppo_gauss_fwd_kernel : &ppo_gauss_fwd_wide_kernel;
  hipLaunchKernelGGL(fwd, dim3(2048),
                     dim3(block), 0, stream, pdpi.data_ptr<float>(),
                     pdold.data_ptr<float>(), vpred.data_ptr<float>(),
                     oldv.data_ptr<float>(), act.data_ptr<float>(),
                     adv.data_ptr<float>(), etr.data_ptr<float>(),
                     acc.data_ptr<double>(), B, A, (float)clip);
  hipLaunchKernelGGL(ppo_gauss_finalize_kernel, dim3(1), dim3(1), 0, stream,
                     acc.data_ptr<double>(), losses.data_ptr<float>(), B,
                     (float)entcoeff, (float)vcoeff);
  return losses;
}

std::vector<torch::Tensor> ppo_loss_gauss_bwd(
    torch::Tensor pdpi, torch::Tensor pdold, torch::Tensor vpred,
    torch::Tensor oldv, torch::Tensor act, torch::Tensor adv,
    torch::Tensor etr, double clip, double entcoeff, double vcoeff,
    torch::Tensor gtotal) {
  const int64_t B = pdpi.size(0);
  const int A = static_cast<int>(pdpi.size(1) / 2);
  auto g_pdflat = torch::empty_like(pdpi);
  auto g_v = torch::empty_like(vpred);
  auto gt = gtotal.to(pdpi.options()).contiguous();
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int block = 256;
  hipLaunchKernelGGL(ppo_gauss_bwd_kernel, dim3(elementwise_grid(B, block)),
                     dim3(block), 0, stream, pdpi.data_ptr<float>(),
                     pdold.data_ptr<float>(), vpred.data_ptr<float>(),
                     oldv.data_ptr<float>(), act.data_ptr<float>(),
                     adv.data_ptr<float>(), etr.data_ptr<float>(),
                     gt.data_ptr<float>(), g_pdflat.data_ptr<float>(),
                     g_v.data_ptr<float>(), B, A, (float)clip, (float)entcoeff,
                     (float)vcoeff);
  return {g_pdflat, g_v};
}

// ---------------------------------------------------------------------------
// Wave-per-row loss gradient: gh[b] = [dL/d pdflat | dL/d vpred].
// Lanes j < 2A cooperate on one row (coalesced row loads, wave-reduced
// log-prob sums), so there is no per-thread row loop to thrash L1 — the
// input to the GEMM-based backward chain (engine _update_fused).
// ---------------------------------------------------------------------------

namespace {

__launch_bounds__(256)
__global__ void ppo_gh_kernel(
    const float* __restrict__ pdflat, const float* __restrict__ oldflat,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const float* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr, float* __restrict__ gh,  // [B][ldgh]
    const float* __restrict__ clip_dev,  // nullptr -> use `clip` arg
    int64_t B, int A, int ldgh, float clip, float entcoeff, float vcoeff) {
  if (clip_dev != nullptr) clip = clip_dev[0];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int P = 2 * A;
  const int64_t waves_total = (int64_t)gridDim.x * 4;
  const int64_t wid = (int64_t)blockIdx.x * 4 + wave;
  // contiguous chunk per wave — see the fwd kernel's comment
  const int64_t per = (B + waves_total - 1) / waves_total;
  const int64_t rb0 = wid * per;
  const int64_t rb1 = rb0 + per < B ? rb0 + per : B;

  // Three rows of operands in flight (dw_mfma-style ping-pong register
  // sets, no copies): one row is a ~6-deep dependent load chain, and the
  // single-row loop measured load-latency-bound at 1.5 TB/s.
  const int jj = (lane < A) ? lane : (lane < P ? lane - A : 0);
  struct Row { float mu, ls, aj, mo, lso, vp, ov, ad, et; };
  auto load_row = [&](int64_t b, Row& r) {
    if (lane < P) {
      r.mu = pdflat[b * P + jj];
      r.ls = pdflat[b * P + A + jj];
      r.aj = act[b * A + jj];
      if (lane < A) {
        r.mo = oldflat[b * P + jj];
        r.lso = oldflat[b * P + A + jj];
      }
    }
    r.vp = vpred[b];
    r.ov = oldv[b];
    r.ad = adv[b];
    r.et = etr[b];
  };
  auto compute_row = [&](int64_t b, const Row& r) {
    float lp_part = 0.f, lo_part = 0.f, ent_part = 0.f;
    float z = 0.f, inv_s = 0.f;
    if (lane < P) {
      inv_s = __expf(-r.ls);
      z = (r.aj - r.mu) * inv_s;
      if (lane < A) {
        lp_part = -0.5f * z * z - r.ls;
        const float zo = (r.aj - r.mo) * __expf(-r.lso);
        lo_part = -0.5f * zo * zo - r.lso;
        ent_part = r.ls;
      }
    }
    const float c = 0.5f * PPO_LOG_2PI * A;
    GaussRow row;
    row.logp_pi = __shfl(wave_reduce_sum(lp_part), 0, WAVE) - c;
    row.logp_old = __shfl(wave_reduce_sum(lo_part), 0, WAVE) - c;
    row.ent = __shfl(wave_reduce_sum(ent_part), 0, WAVE) +
              0.5f * (PPO_LOG_2PI + 1.f) * A;
    const PPORowGrads g = ppo_row_grads(row, r.vp, r.ov, r.ad, r.et, B, clip,
                                        entcoeff, vcoeff, 1.f);
    if (lane < P) {
      gh[b * ldgh + lane] =
          (lane < A) ? g.g_logp * z * inv_s
                     : g.g_logp * (z * z - 1.f) + g.g_ent;
    }
    if (lane == 0) gh[b * ldgh + P] = g.g_v;
    // zero the pad columns (P+1..ldgh): downstream GEMM/dW treat gh as
    // a [B][ldgh] operand and the pad must contribute exactly nothing
    if (lane > P && lane < ldgh) gh[b * ldgh + lane] = 0.f;
  };

  Row ra, rc, re;
  int64_t b = rb0;
  if (b < rb1) load_row(b, ra);
  if (b + 1 < rb1) load_row(b + 1, rc);
  if (b + 2 < rb1) load_row(b + 2, re);
  #pragma unroll 1
  for (; b + 5 < rb1; b += 3) {
    compute_row(b, ra);
    load_row(b + 3, ra);
    compute_row(b + 1, rc);
    load_row(b + 4, rc);
    compute_row(b + 2, re);
    load_row(b + 5, re);
  }
  if (b < rb1) { compute_row(b, ra); ++b; }
  if (b < rb1) { compute_row(b, rc); ++b; }
  if (b < rb1) { compute_row(b, re); ++b; }
  for (; b < rb1; ++b) {
    load_row(b, ra);
    compute_row(b, ra);
  }
}

// LDS-tiled row-per-lane variant.  The wave-per-row kernel above keeps
// only 2A+2 of 64 lanes active (36/64 on the flagship shape) and loads
// pdflat twice per row — measured 0.8 TB/s.  Here a 256-thread block
// stages a GH_TILE-row tile of every operand with full-width coalesced
// loads into odd-stride (bank-conflict-free) LDS rows, then each of
// GH_TILE lanes computes one row serially — no idle-lane guards and no
// wave reductions (the serial j order matches ppo_gauss_row exactly).
// gh rows are built in LDS, overwriting the lane's own oldflat row
// (oldflat is consumed in the first pass only), and stored coalesced.
constexpr int GH_TILE = 128;


__launch_bounds__(256)
__global__ void ppo_gh_tile_kernel(
    const float* __restrict__ pdflat, const float* __restrict__ oldflat,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const float* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr, float* __restrict__ gh,  // [B][ldgh]
    const float* __restrict__ clip_dev,  // nullptr -> use `clip` arg
    int64_t B, int A, int ldgh, float clip, float entcoeff, float vcoeff) {
  if (clip_dev != nullptr) clip = clip_dev[0];
  extern __shared__ float lds[];
  const int P = 2 * A;
  const int sp = P + 1;     // pdflat row stride (odd: P even)
  const int sg = ldgh | 1;  // oldflat/gh shared row stride (>= P, odd)
  const int sa = A | 1;     // act row stride (odd)
  float* l_pd = lds;
  float* l_og = l_pd + GH_TILE * sp;  // oldflat, later this tile's gh
  float* l_ac = l_og + GH_TILE * sg;
  float* l_sc = l_ac + GH_TILE * sa;  // [4][GH_TILE]: vpred|oldv|adv|etr
  const int tid = threadIdx.x;
  const int64_t tb = (int64_t)blockIdx.x * GH_TILE;
  const int rows = (int)min((int64_t)GH_TILE, B - tb);

  stage_tile(pdflat + tb * P, l_pd, P, sp, rows, tid);
  stage_tile(oldflat + tb * P, l_og, P, sg, rows, tid);
  stage_tile(act + tb * A, l_ac, A, sa, rows, tid);
  for (int idx = tid; idx < 4 * GH_TILE; idx += 256) {
    const int s = idx / GH_TILE, t = idx % GH_TILE;
    if (t < rows) {
      const float* src = s == 0 ? vpred : s == 1 ? oldv : s == 2 ? adv : etr;
      l_sc[s * GH_TILE + t] = src[tb + t];
    }
  }
  __syncthreads();

  // compute: one row per lane (rows <= 128 of 256 threads — the kernel
  // is memory-bound, the idle compute lanes cost nothing)
  if (tid < rows) {
    const float* pd = l_pd + tid * sp;
    const float* og = l_og + tid * sg;
    const float* ac = l_ac + tid * sa;
    float lp = 0.f, lo = 0.f, ent = 0.f;
    for (int j = 0; j < A; ++j) {
      const float aj = ac[j];
      const float lsp = pd[A + j];
      const float zp = (aj - pd[j]) * __expf(-lsp);
      lp += -0.5f * zp * zp - lsp;
      const float lso = og[A + j];
      const float zo = (aj - og[j]) * __expf(-lso);
      lo += -0.5f * zo * zo - lso;
      ent += lsp;
    }
    const float cc = 0.5f * PPO_LOG_2PI * A;
    GaussRow row;
    row.logp_pi = lp - cc;
    row.logp_old = lo - cc;
    row.ent = ent + 0.5f * (PPO_LOG_2PI + 1.f) * A;
    const PPORowGrads g =
        ppo_row_grads(row, l_sc[tid], l_sc[GH_TILE + tid],
                      l_sc[2 * GH_TILE + tid], l_sc[3 * GH_TILE + tid], B,
                      clip, entcoeff, vcoeff, 1.f);
    // overwrite this lane's own oldflat row (read only in the loop
    // above, and only by this lane); z/inv_s recomputed from pd/ac so
    // no runtime-indexed register arrays are kept (scratch-spill rule)
    float* ghr = l_og + tid * sg;
    for (int j = 0; j < A; ++j) {
      const float lsp = pd[A + j];
      const float inv_s = __expf(-lsp);
      const float z = (ac[j] - pd[j]) * inv_s;
      ghr[j] = g.g_logp * z * inv_s;
      ghr[A + j] = g.g_logp * (z * z - 1.f) + g.g_ent;
    }
    ghr[P] = g.g_v;
    for (int j = P + 1; j < ldgh; ++j) ghr[j] = 0.f;
  }
  __syncthreads();

  // store: padded LDS gh rows -> contiguous [B][ldgh], float4 on the
  // global side (ldgh is a multiple of 4, so n is too; wrap handling
  // mirrors stage() — ldgh >= 4, so at most one wrap per group)
  {
    const int n4 = (rows * ldgh) >> 2;
    float4* d4 = reinterpret_cast<float4*>(gh + tb * ldgh);
    const int dr = 1024 / ldgh, dc = 1024 % ldgh;
    int r = (tid * 4) / ldgh, c = (tid * 4) % ldgh;
    for (int q = tid; q < n4; q += 256) {
      float vv[4];
      int rr = r, cc = c;
      #pragma unroll
      for (int k = 0; k < 4; ++k) {
        vv[k] = l_og[rr * sg + cc];
        if (++cc == ldgh) { cc = 0; ++rr; }
      }
      d4[q] = make_float4(vv[0], vv[1], vv[2], vv[3]);
      r += dr; c += dc;
      if (c >= ldgh) { c -= ldgh; ++r; }
    }
  }
}

}  // namespace

torch::Tensor ppo_loss_gauss_gh(torch::Tensor pdflat, torch::Tensor oldflat,
                                torch::Tensor vpred, torch::Tensor oldv,
                                torch::Tensor act, torch::Tensor adv,
                                torch::Tensor etr, double clip,
                                double entcoeff, double vcoeff,
                                torch::Tensor clip_dev) {
  const int64_t B = vpred.numel();
  const int A = static_cast<int>(pdflat.size(1) / 2);
  TORCH_CHECK(2 * A + 2 <= WAVE,
              "ppo_loss_gauss_gh is wave-per-row (2A+1 <= 63 lanes); "
              "wide policies use the bf16 gh kernel");
  // row stride padded to a float4 multiple: 16-B-aligned rows let the
  // dgrad GEMM and the glds dW kernel consume gh directly (the pad
  // columns are zeroed in-kernel and contribute nothing downstream)
  const int ldgh = (2 * A + 1 + 3) & ~3;
  auto gh = torch::empty({B, (int64_t)ldgh}, pdflat.options());
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const float* cd =
      (clip_dev.numel() > 0) ? clip_dev.data_ptr<float>() : nullptr;
  const char* ghe = getenv("DPPO_GH_TILE");  // re-read: testable per call
  const int gh_tile_env = ghe ? atoi(ghe) : 1;
  const int lds_bytes =
      GH_TILE * (2 * A + 1 + (ldgh | 1) + (A | 1) + 4) * (int)sizeof(float);
  if (gh_tile_env && lds_bytes <= 65536) {
    const int64_t grid = (B + GH_TILE - 1) / GH_TILE;
    hipLaunchKernelGGL(ppo_gh_tile_kernel, dim3((unsigned)grid), dim3(256),
                       lds_bytes, stream, pdflat.data_ptr<float>(),
                       oldflat.data_ptr<float>(), vpred.data_ptr<float>(),
                       oldv.data_ptr<float>(), act.data_ptr<float>(),
                       adv.data_ptr<float>(), etr.data_ptr<float>(),
                       gh.data_ptr<float>(), cd, B, A, ldgh, (float)clip,
                       (float)entcoeff, (float)vcoeff);
    return gh;
  }
  hipLaunchKernelGGL(ppo_gh_kernel, dim3(2048), dim3(256), 0, stream,
                     pdflat.data_ptr<float>(), oldflat.data_ptr<float>(),
                     vpred.data_ptr<float>(), oldv.data_ptr<float>(),
                     act.data_ptr<float>(), adv.data_ptr<float>(),
                     etr.data_ptr<float>(), gh.data_ptr<float>(), cd, B, A,
                     ldgh, (float)clip, (float)entcoeff, (float)vcoeff);
  return gh;
}
