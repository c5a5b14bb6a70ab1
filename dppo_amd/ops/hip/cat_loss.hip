// Fused PPO clipped-surrogate loss + Gumbel-max sampling for Categorical
// policies (gfx950).
//
// The reference's DEFAULT action family is Categorical (CartPole,
// reference main.py:13 -> distributions.py:124-159).  These kernels fuse
// what its graph launches per row: the stable logsumexp log-prob of pi and
// oldpi (softmax-CE against a one-hot target, distributions.py:131-138),
// the shifted-logit entropy (:148-153), the importance ratio with both
// PPO clips and the three block-reduced means (reference PPO.py:29-40),
// plus Gumbel-max sampling argmax(logits - log(-log U)) (:154-156) with
// the same counter-based RNG family as rollout.hip.
//
// Layout: wave-per-row for the loss kernels (lane j owns logit column j,
// K <= 64; coalesced row loads, butterfly wave reductions so every lane
// holds the row max/sum), row-per-thread for the sampler (K-deep local
// loop, B-way parallel).  Tested against the eager torch path.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ppo_math.h"

namespace {

constexpr float NEG_INF = -3.0e38f;

DEV_INLINE float wave_allreduce_sum(float x) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

DEV_INLINE float wave_allreduce_max(float x) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// counter-based RNG (same lowbias32 family as rollout.hip)
DEV_INLINE unsigned cat_lowbias32(unsigned x) {
  x ^= x >> 16;
  x *= 0x7feb352dU;
  x ^= x >> 15;
  x *= 0x846ca68bU;
  x ^= x >> 16;
  return x;
}

// uniform in (0, 1]
DEV_INLINE float cat_uniform(unsigned seed, unsigned row, unsigned ctr,
                             unsigned slot) {
  unsigned h = cat_lowbias32(seed ^ row * 0x9E3779B9U ^ ctr * 0x85EBCA6BU ^
                             slot * 0xC2B2AE35U);
  return ((h >> 8) + 1) * (1.0f / 16777217.0f);
}

// Per-row categorical quantities every lane ends up holding.
struct CatRow {
  float lp;      // logp_pi(a)   = logit_pi[a]  - lse(logits_pi)
  float lo;      // logp_old(a)  = logit_old[a] - lse(logits_old)
  float ent;     // H(pi) (shifted-logit stable form)
  float p;       // this lane's softmax prob p_j (lanes j < K)
  float logp_j;  // this lane's log p_j
};

DEV_INLINE CatRow cat_row(const float* __restrict__ lpi,
                          const float* __restrict__ lold,
                          int64_t a, int lane, int K) {
  const float lg = (lane < K) ? lpi[lane] : NEG_INF;
  const float lgo = (lane < K) ? lold[lane] : NEG_INF;
  const float m = wave_allreduce_max(lg);
  const float mo = wave_allreduce_max(lgo);
  const float a0 = (lane < K) ? lg - m : NEG_INF;
  const float e = (lane < K) ? __expf(a0) : 0.f;
  const float eo = (lane < K) ? __expf(lgo - mo) : 0.f;
  const float z = wave_allreduce_sum(e);
  const float zo = wave_allreduce_sum(eo);
  // entropy = log z0 - sum p * a0  (reference distributions.py:148-153)
  const float sa = wave_allreduce_sum((lane < K) ? e * a0 : 0.f);
  const float logz = __logf(z);
  CatRow r;
  r.ent = logz - sa / z;
  r.p = e / z;
  r.logp_j = a0 - logz;
  const float picked = wave_allreduce_sum(lane == (int)a ? lg : 0.f);
  const float pickedo = wave_allreduce_sum(lane == (int)a ? lgo : 0.f);
  r.lp = picked - (m + logz);
  r.lo = pickedo - (mo + __logf(zo));
  return r;
}

__launch_bounds__(256)
__global__ void ppo_cat_fwd_kernel(
    const float* __restrict__ lpi, const float* __restrict__ lold,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const int64_t* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr,
    double* __restrict__ acc,  // [3] {policy_min_sum, ent_sum, value_max_sum}
    int64_t B, int K, float clip) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t waves_total = (int64_t)gridDim.x * 4;
  const int64_t wid = (int64_t)blockIdx.x * 4 + wave;
  const int64_t per = (B + waves_total - 1) / waves_total;
  const int64_t rb0 = wid * per;
  const int64_t rb1 = rb0 + per < B ? rb0 + per : B;

  float pol = 0.f, ent = 0.f, val = 0.f;  // lane 0 accumulates rows
  for (int64_t b = rb0; b < rb1; ++b) {
    const CatRow r = cat_row(lpi + b * K, lold + b * K, act[b], lane, K);
    if (lane == 0) {
      const float ratio = __expf(r.lp - r.lo);
      const float ab = adv[b];
      const float surr1 = ratio * ab;
      const float rc = fminf(fmaxf(ratio, 1.f - clip), 1.f + clip);
      pol += fminf(surr1, rc * ab);
      ent += r.ent;
      const float vb = vpred[b], ob = oldv[b], eb = etr[b];
      const float d1 = vb - eb;
      const float dc = fminf(fmaxf(vb - ob, -clip), clip);
      const float d2 = ob + dc - eb;
      val += fmaxf(d1 * d1, d2 * d2);
    }
  }
  if (lane == 0) {
    atomicAdd(&acc[0], static_cast<double>(pol));
    atomicAdd(&acc[1], static_cast<double>(ent));
    atomicAdd(&acc[2], static_cast<double>(val));
  }
}

__global__ void ppo_cat_finalize_kernel(const double* __restrict__ acc,
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

__launch_bounds__(256)
__global__ void ppo_cat_bwd_kernel(
    const float* __restrict__ lpi, const float* __restrict__ lold,
    const float* __restrict__ vpred, const float* __restrict__ oldv,
    const int64_t* __restrict__ act, const float* __restrict__ adv,
    const float* __restrict__ etr,
    const float* __restrict__ gtotal,   // [1] upstream d(total_loss)
    float* __restrict__ g_logits,       // [B, K]
    float* __restrict__ g_v,            // [B]
    int64_t B, int K, float clip, float entcoeff, float vcoeff) {
  const float g = gtotal[0];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int64_t waves_total = (int64_t)gridDim.x * 4;
  const int64_t wid = (int64_t)blockIdx.x * 4 + wave;
  const int64_t per = (B + waves_total - 1) / waves_total;
  const int64_t rb0 = wid * per;
  const int64_t rb1 = rb0 + per < B ? rb0 + per : B;

  for (int64_t b = rb0; b < rb1; ++b) {
    const int64_t a = act[b];
    const CatRow r = cat_row(lpi + b * K, lold + b * K, a, lane, K);
    GaussRow rr;  // ppo_row_grads only reads logp_pi / logp_old
    rr.logp_pi = r.lp;
    rr.logp_old = r.lo;
    rr.ent = r.ent;
    const PPORowGrads gr = ppo_row_grads(rr, vpred[b], oldv[b], adv[b],
                                         etr[b], B, clip, entcoeff, vcoeff, g);
    if (lane < K) {
      // d logp(a)/d logit_j = 1[j==a] - p_j  (softmax CE);
      // d H/d logit_j = -p_j (log p_j + H)   (shifted-logit entropy)
      const float d_lp = (lane == (int)a ? 1.f : 0.f) - r.p;
      const float d_ent = -r.p * (r.logp_j + r.ent);
      g_logits[b * K + lane] = gr.g_logp * d_lp + gr.g_ent * d_ent;
    }
    if (lane == 0) g_v[b] = gr.g_v;
  }
}

__global__ void cat_sample_kernel(const float* __restrict__ logits,
                                  int64_t* __restrict__ out, int64_t B, int K,
                                  unsigned seed, unsigned ctr) {
  for (int64_t b = gidx(); b < B; b += gstride()) {
    const float* row = logits + b * K;
    float best = NEG_INF;
    int arg = 0;
    for (int j = 0; j < K; ++j) {
      // Gumbel-max: argmax(logits - log(-log U)) (distributions.py:154-156)
      const float u = cat_uniform(seed, (unsigned)b, ctr, (unsigned)j);
      const float gv = row[j] - __logf(-__logf(u));
      if (gv > best) {
        best = gv;
        arg = j;
      }
    }
    out[b] = arg;
  }
}

}  // namespace

torch::Tensor ppo_loss_cat_fwd(torch::Tensor lpi, torch::Tensor lold,
                               torch::Tensor vpred, torch::Tensor oldv,
                               torch::Tensor act, torch::Tensor adv,
                               torch::Tensor etr, double clip,
                               double entcoeff, double vcoeff) {
  TORCH_CHECK(lpi.is_cuda() && lpi.dtype() == torch::kFloat32);
  TORCH_CHECK(lpi.dim() == 2 && lpi.size(1) <= WAVE,
              "categorical fused loss supports K <= 64");
  TORCH_CHECK(act.dtype() == torch::kInt64);
  const int64_t B = lpi.size(0);
  const int K = static_cast<int>(lpi.size(1));
  TORCH_CHECK(act.numel() == B && vpred.numel() == B && adv.numel() == B &&
              etr.numel() == B);

  auto acc = torch::zeros({3}, lpi.options().dtype(torch::kFloat64));
  auto losses = torch::empty({4}, lpi.options());
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(ppo_cat_fwd_kernel, dim3(2048), dim3(256), 0, stream,
                     lpi.data_ptr<float>(), lold.data_ptr<float>(),
                     vpred.data_ptr<float>(), oldv.data_ptr<float>(),
                     act.data_ptr<int64_t>(), adv.data_ptr<float>(),
                     etr.data_ptr<float>(), acc.data_ptr<double>(), B, K,
                     (float)clip);
  hipLaunchKernelGGL(ppo_cat_finalize_kernel, dim3(1), dim3(1), 0, stream,
                     acc.data_ptr<double>(), losses.data_ptr<float>(), B,
                     (float)entcoeff, (float)vcoeff);
  return losses;
}

std::vector<torch::Tensor> ppo_loss_cat_bwd(
    torch::Tensor lpi, torch::Tensor lold, torch::Tensor vpred,
    torch::Tensor oldv, torch::Tensor act, torch::Tensor adv,
    torch::Tensor etr, double clip, double entcoeff, double vcoeff,
    torch::Tensor gtotal) {
  const int64_t B = lpi.size(0);
  const int K = static_cast<int>(lpi.size(1));
  auto g_logits = torch::empty_like(lpi);
  auto g_v = torch::empty_like(vpred);
  auto gt = gtotal.to(lpi.options()).contiguous();
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(ppo_cat_bwd_kernel, dim3(2048), dim3(256), 0, stream,
                     lpi.data_ptr<float>(), lold.data_ptr<float>(),
                     vpred.data_ptr<float>(), oldv.data_ptr<float>(),
                     act.data_ptr<int64_t>(), adv.data_ptr<float>(),
                     etr.data_ptr<float>(), gt.data_ptr<float>(),
                     g_logits.data_ptr<float>(), g_v.data_ptr<float>(), B, K,
                     (float)clip, (float)entcoeff, (float)vcoeff);
  return {g_logits, g_v};
}

torch::Tensor cat_sample(torch::Tensor logits, int64_t seed, int64_t ctr) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(logits.dim() == 2);
  const int64_t B = logits.size(0);
  const int K = static_cast<int>(logits.size(1));
  auto out = torch::empty({B}, logits.options().dtype(torch::kInt64));
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int block = 256;
  hipLaunchKernelGGL(cat_sample_kernel, dim3(elementwise_grid(B, block)),
                     dim3(block), 0, stream,
                     logits.contiguous().data_ptr<float>(),
                     out.data_ptr<int64_t>(), B, K,
                     (unsigned)(seed & 0xFFFFFFFF), (unsigned)(ctr & 0xFFFFFFFF));
  return out;
}
