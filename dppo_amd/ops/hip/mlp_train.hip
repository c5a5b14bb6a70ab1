// Fused small-MLP training chunk step (gfx950): ONE kernel computes the
// whole forward + PPO loss gradient + backward + per-block dW partials
// for a minibatch chunk, and a second kernel reduces the partials and
// applies Adam.  Together they replace the ~15-kernel chunk chain
// (gemm_fwd x3 + transposes + gh + dgrad x2 + dw_mfma x3 + zero + Adam)
// whose per-kernel latency (~258 us per 4096-sample chunk measured at
// the BASELINE config-4 shapes) dominates the minibatched update and the
// tiny-batch full-batch update (BASELINE config 2).
//
// Reference semantics: one chunk step == one Chief train application
// (reference Chief.py:64 + PPO.py:29-53) on a (mini)batch; loss math is
// ppo_math.h (== reference PPO.py:29-40 / distributions.py:195-203);
// Adam matches torch.optim.Adam exactly (adam.hip).
//
// MI355X design (why this shape; the measured ladder 693 -> 182 us per
// 4096-sample chunk is in profiles/r01_chunk_kernel_notes.md):
//   - ALL weights live in LDS, transposed at kernel start ([k][u] images,
//     16-B-aligned regions, odd row strides where rows are read
//     column-wise), so the per-tile phase chain touches only LDS + VALU:
//     no L2 latency inside the sample loop.  At the Humanoid shapes
//     (D=376, H=64, A=17) the image is ~144 KB of the 160 KB LDS ->
//     1 block/CU of 8 waves (2/SIMD, ~206 VGPRs/lane of the 512-deep
//     unified file).
//   - dW accumulates in REGISTERS: each of the 512 threads owns a fixed
//     (unit, k-chunk) slice of every weight matrix (KC1-template for the
//     input layer).  Per sample that is a broadcast LDS read per k plus
//     one FMA per owned element; PMC showed the kernel issue-bound, so
//     the slices are read float4 with per-group (not per-element) guards.
//   - each chunk spreads over up to 256 blocks (spb = ceil(B/256)): the
//     per-block serial tile loop is the latency term, so fill the chip.
//   - Per-block partial gradients go to a slab (flat-grad layout); the
//     reduce kernel sums NB slabs and applies Adam (device-state
//     step/lr/bias-correction, so the pair is hipGraph-replayable), or
//     writes the summed gradient for the distributed path (all-reduce +
//     adam_step_dev on the host side).
//
// Limits (host-checked): Box/DiagGaussian policy, fp32, 1..2 hidden
// layers of equal width H in {16,32,64}, D <= 512, A <= 32, LDS image
// <= 158 KB.  Larger problems stay on the MFMA GEMM path (mfma_gemm.hip)
// which wins when compute, not latency, is the bound.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <vector>

#include "common.h"
#include "ppo_math.h"

namespace {

constexpr int CT = 512;   // threads per block (8 waves, 2/SIMD): halves
                          // the per-thread dW slice (fewer accumulator
                          // VGPRs, fewer instructions) and doubles the
                          // latency-hiding occupancy vs the 256-thread
                          // first cut
constexpr int CNW = 8;    // waves
constexpr int ST = 4;     // samples per tile (LDS budget at CT=512)
constexpr int KH_MAX = 9;   // dWh rows per thread: ceil((P+1)/(CT/H)) <= 9
constexpr int KC2_MAX = 8;  // dW2 k's per thread: H/(CT/H) <= 8

struct ChunkLds {
  int w1t, w2t, wht, b1, b2, bh;
  int x, h1, h2, gh, dh1, dh2, act, oldf, sc, part, dbias;
  int dp, ws1, ws2, wsh, pc, ap;
  int total;
};

// shared host/device LDS layout (floats)
__host__ __device__ inline ChunkLds chunk_lds_map(int D, int H, int A,
                                                  int n_hidden) {
  auto r4 = [](int v) { return (v + 3) & ~3; };
  const int P = 2 * A;
  ChunkLds m;
  m.dp = r4(D);
  m.ws1 = H + 1;           // w1t row stride: odd -> both the transpose
                           // staging writes (k-consecutive per thread)
                           // and nothing else conflict; fwd reads are
                           // lane==u, stride-agnostic
  m.ws2 = H + 1;           // w2t row stride (odd for H even: conflict-free
                           // column reads in the dgrad phase)
  m.wsh = P + 1;           // wht row stride (P even -> odd stride)
  m.pc = r4(H > P + 1 ? H : P + 1);
  m.ap = r4(A);
  // every region start is rounded to 4 floats: the x/h tiles are read
  // as float4 (ds_read_b128 needs 16-B alignment; a misaligned image
  // measured ~10x slower, not wrong)
  int o = 0;
  m.w1t = o; o += r4(D * m.ws1);
  m.w2t = o; o += (n_hidden == 2) ? r4(H * m.ws2) : 0;
  m.wht = o; o += r4(H * m.wsh);
  m.b1 = o; o += r4(H);
  m.b2 = o; o += (n_hidden == 2) ? r4(H) : 0;
  m.bh = o; o += r4(P + 1);
  m.x = o; o += ST * m.dp;
  m.h1 = o; o += ST * H;
  m.h2 = o; o += (n_hidden == 2) ? ST * H : 0;
  m.gh = o; o += r4(ST * (P + 1));
  m.dh1 = o; o += ST * H;
  m.dh2 = o; o += (n_hidden == 2) ? ST * H : 0;
  m.act = o; o += ST * m.ap;
  m.oldf = o; o += r4(ST * P);
  m.sc = o; o += 4 * ST;  // adv | etr | oldv | (spare)
  m.part = o; o += CNW * ST * m.pc;
  m.dbias = o; o += r4(2 * H + P + 1);
  m.total = o;
  return m;
}

struct ChunkArgs {
  const float* states;   // [B][D]
  const float* actions;  // [B][A]
  const float* adv;      // [B]
  const float* etr;      // [B]
  const float* oldflat;  // [B][P]
  const float* oldv;     // [B]
  const float* params;   // flat torch-layout parameter buffer
  const float* clip_dev; // [1]; nullptr -> clip_host
  float* slabs;          // [NB][ppad] per-block gradient partials
  int* step_dev;         // fused-Adam prep (block 0 bumps step, fills coef)
  const float* lr_dev;
  float* coef;           // [3] lr | bc1 | bc2
  int off_W0, off_b0, off_W1, off_b1;
  int off_Wv, off_bv, off_Wp, off_bp;
  int n_hidden, D, H, A, activation, spb, ppad;
  int64_t B;
  float clip_host, entcoeff, vcoeff, beta1, beta2;
  int fuse_adam;
};

DEV_INLINE float dact(float h, int activation) {
  return activation ? (1.f - h * h) : (h > 0.f ? 1.f : 0.f);
}

// TD/TH/TA/TNH = 0 -> runtime shapes (the generic ladder); nonzero ->
// EXACT compile-time shapes: the LDS map, every i/H-i%H index map, the
// float4 tail guards and the per-thread slice bounds all constant-fold,
// removing the address-arithmetic / exec-mask / SGPR-spill classes the
// static ISA audit blamed for the issue-bound profile
// (profiles/r01_chunk_kernel_notes.md round-2 target list).  TACT: 1 =
// tanh folded, -1 = relu folded, 0 = runtime.
template <int KC1, int TD = 0, int TH = 0, int TA = 0, int TNH = 0,
          int TACT = 0>
__launch_bounds__(CT, 1)
__global__ void mlp_chunk_kernel(ChunkArgs a) {
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid / WAVE;
  const int D = TD ? TD : a.D;
  const int H = TH ? TH : a.H;
  const int A = TA ? TA : a.A;
  const int P = 2 * A;
  const int NH = TNH ? TNH : a.n_hidden;
  const int activation = TACT ? (TACT > 0 ? 1 : 0) : a.activation;
  const int CH = CT / H;        // k-chunks (16/8/4 for H=16/32/64)
  const int uo = tid % H;       // owned unit (dW phases)
  const int co = tid / H;       // owned k-chunk

  extern __shared__ __attribute__((aligned(16))) float lds[];
  const ChunkLds m = chunk_lds_map(D, H, A, NH);

  // fused-Adam prep: step/lr/bias-corrections into coef[3] (the reduce
  // kernel reads them; the A->B kernel boundary publishes)
  if (a.fuse_adam && blockIdx.x == 0 && tid == 0) {
    const int t = ++a.step_dev[0];
    a.coef[0] = a.lr_dev[0];
    a.coef[1] = 1.f - powf(a.beta1, (float)t);
    a.coef[2] = 1.f - powf(a.beta2, (float)t);
  }

  // ---- stage weights (transposed) + biases into LDS; zero dbias ----
  for (int i = tid; i < D * H; i += CT) {
    const int u = i / D, k = i - u * D;  // coalesced param read
    lds[m.w1t + k * m.ws1 + u] = a.params[a.off_W0 + i];
  }
  if (NH == 2) {
    for (int i = tid; i < H * H; i += CT) {
      const int u = i / H, k = i - u * H;
      lds[m.w2t + k * m.ws2 + u] = a.params[a.off_W1 + i];
    }
  }
  for (int i = tid; i < H * P; i += CT) {
    const int u = i / H, k = i - u * H;
    lds[m.wht + k * m.wsh + u] = a.params[a.off_Wp + i];
  }
  for (int k = tid; k < H; k += CT)
    lds[m.wht + k * m.wsh + P] = a.params[a.off_Wv + k];
  for (int i = tid; i < H; i += CT) lds[m.b1 + i] = a.params[a.off_b0 + i];
  if (NH == 2)
    for (int i = tid; i < H; i += CT) lds[m.b2 + i] = a.params[a.off_b1 + i];
  for (int i = tid; i < P; i += CT) lds[m.bh + i] = a.params[a.off_bp + i];
  if (tid == 0) lds[m.bh + P] = a.params[a.off_bv];
  for (int i = tid; i < 2 * H + P + 1; i += CT) lds[m.dbias + i] = 0.f;
  __syncthreads();

  const float clip = a.clip_dev ? a.clip_dev[0] : a.clip_host;

  // ---- register dW accumulators ----
  float acc1[KC1];
  #pragma unroll
  for (int j = 0; j < KC1; ++j) acc1[j] = 0.f;
  float acc2[KC2_MAX];
  #pragma unroll
  for (int j = 0; j < KC2_MAX; ++j) acc2[j] = 0.f;
  float acch[KH_MAX];
  #pragma unroll
  for (int j = 0; j < KH_MAX; ++j) acch[j] = 0.f;
  const int kc2 = (H + CH - 1) / CH;  // dW2 k's per thread

  // k-split bounds for the layer-forward phases (rollout.hip pattern)
  auto kpart = [&](int in_dim, int in_off, int in_stride, int w_off,
                   int w_stride, int out_dim, int st) {
    const int kq = (((in_dim + CNW * 4 - 1) / (CNW * 4)) * 4);
    const int k0 = wv * kq;
    const int k1 = min(in_dim, k0 + kq);
    for (int u = lane; u < out_dim; u += WAVE) {
      float acc[ST];
      #pragma unroll
      for (int s = 0; s < ST; ++s) acc[s] = 0.f;
      int k = k0;
      for (; k + 4 <= k1; k += 4) {
        const float w0 = lds[w_off + (k + 0) * w_stride + u];
        const float w1 = lds[w_off + (k + 1) * w_stride + u];
        const float w2 = lds[w_off + (k + 2) * w_stride + u];
        const float w3 = lds[w_off + (k + 3) * w_stride + u];
        #pragma unroll
        for (int s = 0; s < ST; ++s) {
          if (s < st) {
            const float4 x4 = *reinterpret_cast<const float4*>(
              __builtin_assume_aligned(&lds[in_off + s * in_stride + k], 16));
            acc[s] += w0 * x4.x + w1 * x4.y + w2 * x4.z + w3 * x4.w;
          }
        }
      }
      for (; k < k1; ++k) {
        const float w = lds[w_off + k * w_stride + u];
        #pragma unroll
        for (int s = 0; s < ST; ++s)
          if (s < st) acc[s] += w * lds[in_off + s * in_stride + k];
      }
      #pragma unroll
      for (int s = 0; s < ST; ++s)
        if (s < st) lds[m.part + (wv * ST + s) * m.pc + u] = acc[s];
    }
  };

  // ---- sample-tile loop ----
  const int64_t s0 = (int64_t)blockIdx.x * a.spb;
  const int64_t send = min(a.B, s0 + a.spb);
  for (int64_t t0 = s0; t0 < send; t0 += ST) {
    const int st = (int)min((int64_t)ST, send - t0);

    // P0: stage the sample tile
    for (int i = tid; i < st * D; i += CT) {
      const int s = i / D, k = i - s * D;
      lds[m.x + s * m.dp + k] = a.states[(t0 + s) * D + k];
    }
    for (int i = tid; i < st * A; i += CT) {
      const int s = i / A, j = i - s * A;
      lds[m.act + s * m.ap + j] = a.actions[(t0 + s) * A + j];
    }
    for (int i = tid; i < st * P; i += CT) {
      const int s = i / P, j = i - s * P;
      lds[m.oldf + s * P + j] = a.oldflat[(t0 + s) * P + j];
    }
    if (tid < st) {
      lds[m.sc + tid] = a.adv[t0 + tid];
      lds[m.sc + ST + tid] = a.etr[t0 + tid];
      lds[m.sc + 2 * ST + tid] = a.oldv[t0 + tid];
    }
    __syncthreads();

    // P1: layer 1 forward
    kpart(D, m.x, m.dp, m.w1t, m.ws1, H, st);
    __syncthreads();
    for (int i = tid; i < st * H; i += CT) {
      const int s = i / H, u = i - s * H;
      float sum = lds[m.b1 + u];
      #pragma unroll
      for (int w = 0; w < CNW; ++w) sum += lds[m.part + (w * ST + s) * m.pc + u];
      lds[m.h1 + i] = activation ? fast_tanhf(sum) : fmaxf(sum, 0.f);
    }
    __syncthreads();

    // P2: layer 2 forward
    const int hlast = (NH == 2) ? m.h2 : m.h1;
    if (NH == 2) {
      kpart(H, m.h1, H, m.w2t, m.ws2, H, st);
      __syncthreads();
      for (int i = tid; i < st * H; i += CT) {
        const int s = i / H, u = i - s * H;
        float sum = lds[m.b2 + u];
        #pragma unroll
        for (int w = 0; w < CNW; ++w)
          sum += lds[m.part + (w * ST + s) * m.pc + u];
        lds[m.h2 + i] = activation ? fast_tanhf(sum) : fmaxf(sum, 0.f);
      }
      __syncthreads();
    }

    // P3: heads forward -> gh tile holds [pdflat | v]
    kpart(H, hlast, H, m.wht, m.wsh, P + 1, st);
    __syncthreads();
    for (int i = tid; i < st * (P + 1); i += CT) {
      const int s = i / (P + 1), u = i - s * (P + 1);
      float sum = lds[m.bh + u];
      #pragma unroll
      for (int w = 0; w < CNW; ++w) sum += lds[m.part + (w * ST + s) * m.pc + u];
      lds[m.gh + s * (P + 1) + u] = sum;
    }
    __syncthreads();

    // P4: PPO loss gradient rows (ppo_gh_kernel math, in place over gh)
    for (int s = wv; s < st; s += CNW) {
      const int jj = (lane < A) ? lane : (lane < P ? lane - A : 0);
      float mu = 0.f, ls = 0.f, aj = 0.f, mo = 0.f, lso = 0.f;
      if (lane < P) {
        mu = lds[m.gh + s * (P + 1) + jj];
        ls = lds[m.gh + s * (P + 1) + A + jj];
        aj = lds[m.act + s * m.ap + jj];
        if (lane < A) {
          mo = lds[m.oldf + s * P + jj];
          lso = lds[m.oldf + s * P + A + jj];
        }
      }
      const float vp = lds[m.gh + s * (P + 1) + P];
      const float ov = lds[m.sc + 2 * ST + s];
      const float ad = lds[m.sc + s];
      const float et = lds[m.sc + ST + s];
      float lp_part = 0.f, lo_part = 0.f, ent_part = 0.f;
      float z = 0.f, inv_s = 0.f;
      if (lane < P) {
        inv_s = __expf(-ls);
        z = (aj - mu) * inv_s;
        if (lane < A) {
          lp_part = -0.5f * z * z - ls;
          const float zo = (aj - mo) * __expf(-lso);
          lo_part = -0.5f * zo * zo - lso;
          ent_part = ls;
        }
      }
      const float c = 0.5f * PPO_LOG_2PI * A;
      GaussRow row;
      row.logp_pi = __shfl(wave_reduce_sum(lp_part), 0, WAVE) - c;
      row.logp_old = __shfl(wave_reduce_sum(lo_part), 0, WAVE) - c;
      row.ent = __shfl(wave_reduce_sum(ent_part), 0, WAVE) +
                0.5f * (PPO_LOG_2PI + 1.f) * A;
      const PPORowGrads g = ppo_row_grads(row, vp, ov, ad, et, a.B, clip,
                                          a.entcoeff, a.vcoeff, 1.f);
      if (lane < P) {
        lds[m.gh + s * (P + 1) + lane] =
            (lane < A) ? g.g_logp * z * inv_s
                       : g.g_logp * (z * z - 1.f) + g.g_ent;
      }
      if (lane == 0) lds[m.gh + s * (P + 1) + P] = g.g_v;
    }
    __syncthreads();

    // P5: dgrad through the heads -> dh_last
    const int dlast = (NH == 2) ? m.dh2 : m.dh1;
    for (int i = tid; i < st * H; i += CT) {
      const int s = i / H, k = i - s * H;
      // 4 interleaved partial sums: the single-accumulator chain pays
      // an LDS latency per iteration
      float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
      int u = 0;
      for (; u + 4 <= P + 1; u += 4) {
        a0 += lds[m.wht + k * m.wsh + u] * lds[m.gh + s * (P + 1) + u];
        a1 += lds[m.wht + k * m.wsh + u + 1] * lds[m.gh + s * (P + 1) + u + 1];
        a2 += lds[m.wht + k * m.wsh + u + 2] * lds[m.gh + s * (P + 1) + u + 2];
        a3 += lds[m.wht + k * m.wsh + u + 3] * lds[m.gh + s * (P + 1) + u + 3];
      }
      for (; u < P + 1; ++u)
        a0 += lds[m.wht + k * m.wsh + u] * lds[m.gh + s * (P + 1) + u];
      lds[dlast + i] =
          dact(lds[hlast + i], activation) * ((a0 + a1) + (a2 + a3));
    }
    __syncthreads();

    // P6: dgrad through layer 2 -> dh1
    if (NH == 2) {
      for (int i = tid; i < st * H; i += CT) {
        const int s = i / H, k = i - s * H;
        float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
        for (int u = 0; u < H; u += 4) {
          a0 += lds[m.w2t + k * m.ws2 + u] * lds[m.dh2 + s * H + u];
          a1 += lds[m.w2t + k * m.ws2 + u + 1] * lds[m.dh2 + s * H + u + 1];
          a2 += lds[m.w2t + k * m.ws2 + u + 2] * lds[m.dh2 + s * H + u + 2];
          a3 += lds[m.w2t + k * m.ws2 + u + 3] * lds[m.dh2 + s * H + u + 3];
        }
        lds[m.dh1 + i] =
            dact(lds[m.h1 + i], activation) * ((a0 + a1) + (a2 + a3));
      }
      __syncthreads();
    }

    // P7: dW/bias accumulation (register slices; disjoint LDS bias slots)
    for (int s = 0; s < st; ++s) {
      const float d1 = lds[m.dh1 + s * H + uo];
      // float4 x reads + per-group guards: the per-element guarded form
      // tripled the instruction count of the dominant dW1 term (PMC:
      // the kernel is issue-bound, profiles/r01_chunk_kernel_notes.md)
      #pragma unroll
      for (int j = 0; j < KC1; j += 4) {
        const int k = co * KC1 + j;
        if (k + 3 < D) {
          const float4 x4 =
              *reinterpret_cast<const float4*>(
              __builtin_assume_aligned(&lds[m.x + s * m.dp + k], 16));
          acc1[j] += d1 * x4.x;
          acc1[j + 1] += d1 * x4.y;
          acc1[j + 2] += d1 * x4.z;
          acc1[j + 3] += d1 * x4.w;
        } else if (k < D) {
          #pragma unroll
          for (int t = 0; t < 4; ++t)
            if (k + t < D) acc1[j + t] += d1 * lds[m.x + s * m.dp + k + t];
        }
      }
      if (NH == 2) {
        const float d2 = lds[m.dh2 + s * H + uo];
        #pragma unroll
        for (int j = 0; j < KC2_MAX; ++j) {
          const int k = co * kc2 + j;
          if (j < kc2 && k < H) acc2[j] += d2 * lds[m.h1 + s * H + k];
        }
      }
      const float hl = lds[hlast + s * H + uo];
      #pragma unroll
      for (int j = 0; j < KH_MAX; ++j) {
        const int r = co + CH * j;
        if (r < P + 1) acch[j] += lds[m.gh + s * (P + 1) + r] * hl;
      }
    }
    if (tid < H) {
      float acc = 0.f;
      for (int s = 0; s < st; ++s) acc += lds[m.dh1 + s * H + tid];
      lds[m.dbias + tid] += acc;
    } else if (NH == 2 && tid < 2 * H) {
      float acc = 0.f;
      for (int s = 0; s < st; ++s) acc += lds[m.dh2 + s * H + tid - H];
      lds[m.dbias + tid] += acc;
    }
    if (tid >= 2 * H && tid < 2 * H + P + 1) {
      const int u = tid - 2 * H;
      float acc = 0.f;
      for (int s = 0; s < st; ++s) acc += lds[m.gh + s * (P + 1) + u];
      lds[m.dbias + tid] += acc;
    }
    __syncthreads();
  }

  // ---- write the block's partial gradient slab (flat-grad layout) ----
  float* slab = a.slabs + (int64_t)blockIdx.x * a.ppad;
  #pragma unroll
  for (int j = 0; j < KC1; ++j) {
    const int k = co * KC1 + j;
    if (k < D) slab[a.off_W0 + uo * D + k] = acc1[j];
  }
  if (NH == 2) {
    #pragma unroll
    for (int j = 0; j < KC2_MAX; ++j) {
      const int k = co * kc2 + j;
      if (j < kc2 && k < H) slab[a.off_W1 + uo * H + k] = acc2[j];
    }
  }
  #pragma unroll
  for (int j = 0; j < KH_MAX; ++j) {
    const int r = co + CH * j;
    if (r < P) slab[a.off_Wp + r * H + uo] = acch[j];
    else if (r == P) slab[a.off_Wv + uo] = acch[j];
  }
  if (tid < H) slab[a.off_b0 + tid] = lds[m.dbias + tid];
  else if (NH == 2 && tid < 2 * H) slab[a.off_b1 + tid - H] = lds[m.dbias + tid];
  if (tid >= 2 * H && tid < 2 * H + P + 1) {
    const int u = tid - 2 * H;
    if (u < P) slab[a.off_bp + u] = lds[m.dbias + tid];
    else slab[a.off_bv] = lds[m.dbias + tid];
  }
}

// reduce NB slabs; fuse_adam: torch-exact Adam with device coef; else
// write the summed (already 1/B-scaled) gradient to flat_grad
__global__ void mlp_chunk_reduce_kernel(const float* __restrict__ slabs,
                                        int nb, int64_t ppad, int64_t n,
                                        float* __restrict__ p,
                                        float* __restrict__ mom,
                                        float* __restrict__ vel,
                                        const float* __restrict__ coef,
                                        float beta1, float beta2, float eps,
                                        float* __restrict__ grad_out,
                                        int fuse_adam) {
  for (int64_t f = gidx(); f < n; f += gstride()) {
    float g = 0.f;
    for (int b = 0; b < nb; ++b) g += slabs[(int64_t)b * ppad + f];
    if (fuse_adam) {
      const float lr = coef[0], bc1 = coef[1], bc2 = coef[2];
      const float mm = beta1 * mom[f] + (1.f - beta1) * g;
      const float vv = beta2 * vel[f] + (1.f - beta2) * g * g;
      mom[f] = mm;
      vel[f] = vv;
      p[f] -= lr * (mm / bc1) / (sqrtf(vv / bc2) + eps);
    } else {
      grad_out[f] = g;
    }
  }
}

static bool chunk_lds_attr_ok() {
  static const bool ok = []() {
    bool r = true;
    for (const void* f :
         {reinterpret_cast<const void*>(&(*mlp_chunk_kernel<8>)),
          reinterpret_cast<const void*>(&(*mlp_chunk_kernel<16>)),
          reinterpret_cast<const void*>(&(*mlp_chunk_kernel<32>)),
          reinterpret_cast<const void*>(&(*mlp_chunk_kernel<48>)),
          reinterpret_cast<const void*>(&(*mlp_chunk_kernel<64>)),
          reinterpret_cast<const void*>(
              &(*mlp_chunk_kernel<48, 376, 64, 17, 2, 1>)),
          reinterpret_cast<const void*>(
              &(*mlp_chunk_kernel<4, 17, 64, 6, 2, 1>)),
          reinterpret_cast<const void*>(
              &(*mlp_chunk_kernel<1, 3, 16, 1, 1, -1>))})
      r &= hipFuncSetAttribute(f, hipFuncAttributeMaxDynamicSharedMemorySize,
                               159 * 1024) == hipSuccess;
    return r;
  }();
  return ok;
}

}  // namespace

// host-visible feasibility check (mirrors the kernel gates)
bool mlp_chunk_supported(int64_t D, int64_t H, int64_t A, int64_t n_hidden) {
  if (n_hidden < 1 || n_hidden > 2) return false;
  if (H != 16 && H != 32 && H != 64) return false;
  if (D < 1 || D > 512 || A < 1 || A > 32) return false;
  const int CH = CT / (int)H;
  if ((D + CH - 1) / CH > 64) return false;  // KC1 template ceiling
  const ChunkLds m = chunk_lds_map((int)D, (int)H, (int)A, (int)n_hidden);
  if ((size_t)m.total * sizeof(float) > 158 * 1024) return false;
  return chunk_lds_attr_ok();
}

void mlp_chunk_train(
    torch::Tensor params, torch::Tensor states, torch::Tensor actions,
    torch::Tensor adv, torch::Tensor etr, torch::Tensor oldflat,
    torch::Tensor oldv, std::vector<int64_t> offsets,
    std::vector<int64_t> dims, int64_t activation, torch::Tensor clip_dev,
    double clip, double entcoeff, double vcoeff, torch::Tensor slabs,
    torch::Tensor mom, torch::Tensor vel, torch::Tensor step_dev,
    torch::Tensor lr_dev, torch::Tensor coef, torch::Tensor flat_grad,
    bool fuse_adam, double beta1, double beta2, double eps) {
  const int64_t B = states.size(0);
  const int64_t D = states.size(1);
  const int64_t A = actions.size(1);
  const int n_hidden = (int)dims.size() - 1;
  const int64_t H = dims[1];
  TORCH_CHECK(states.is_cuda() && states.dtype() == torch::kFloat32 &&
              states.is_contiguous());
  TORCH_CHECK(actions.is_contiguous() && adv.is_contiguous() &&
              etr.is_contiguous() && oldflat.is_contiguous() &&
              oldv.is_contiguous() && params.is_contiguous());
  TORCH_CHECK(dims[0] == D, "dims[0] must be obs_dim");
  for (int l = 1; l <= n_hidden; ++l)
    TORCH_CHECK(dims[l] == H, "equal hidden widths required");
  TORCH_CHECK(mlp_chunk_supported(D, H, A, n_hidden),
              "shape outside mlp_chunk limits");
  TORCH_CHECK((int)offsets.size() == 2 * n_hidden + 4);

  const int64_t ptotal = params.numel();
  const int64_t ppad = (ptotal + 3) & ~3LL;
  // fill the chip: one block per CU until samples run out (the serial
  // tile loop per block is the latency term; 64-sample blocks measured
  // 322 us vs 16-sample blocks at the same 4096-sample chunk)
  int64_t spb = (B + N_CU - 1) / N_CU;
  if (spb < ST) spb = ST;
  spb = (spb + ST - 1) / ST * ST;
  int64_t nb = (B + spb - 1) / spb;
  TORCH_CHECK(slabs.numel() >= nb * ppad, "slab scratch too small");

  ChunkArgs a{};
  a.states = states.data_ptr<float>();
  a.actions = actions.data_ptr<float>();
  a.adv = adv.data_ptr<float>();
  a.etr = etr.data_ptr<float>();
  a.oldflat = oldflat.data_ptr<float>();
  a.oldv = oldv.data_ptr<float>();
  a.params = params.data_ptr<float>();
  a.clip_dev = clip_dev.numel() > 0 ? clip_dev.data_ptr<float>() : nullptr;
  a.slabs = slabs.data_ptr<float>();
  a.step_dev = step_dev.data_ptr<int>();
  a.lr_dev = lr_dev.data_ptr<float>();
  a.coef = coef.data_ptr<float>();
  a.off_W0 = (int)offsets[0];
  a.off_b0 = (int)offsets[1];
  a.off_W1 = n_hidden == 2 ? (int)offsets[2] : 0;
  a.off_b1 = n_hidden == 2 ? (int)offsets[3] : 0;
  a.off_Wv = (int)offsets[2 * n_hidden];
  a.off_bv = (int)offsets[2 * n_hidden + 1];
  a.off_Wp = (int)offsets[2 * n_hidden + 2];
  a.off_bp = (int)offsets[2 * n_hidden + 3];
  a.n_hidden = n_hidden;
  a.D = (int)D;
  a.H = (int)H;
  a.A = (int)A;
  a.activation = (int)activation;
  a.spb = (int)spb;
  a.ppad = (int)ppad;
  a.B = B;
  a.clip_host = (float)clip;
  a.entcoeff = (float)entcoeff;
  a.vcoeff = (float)vcoeff;
  a.beta1 = (float)beta1;
  a.beta2 = (float)beta2;
  a.fuse_adam = fuse_adam ? 1 : 0;

  const ChunkLds m = chunk_lds_map((int)D, (int)H, (int)A, n_hidden);
  const size_t lds_bytes = (size_t)m.total * sizeof(float);
  const int CH = CT / (int)H;
  const int kc1 = (int)((D + CH - 1) / CH);
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  // exact-trip-count specializations for the BASELINE shape families
  if (D == 376 && H == 64 && A == 17 && n_hidden == 2 && activation == 1)
    hipLaunchKernelGGL((mlp_chunk_kernel<48, 376, 64, 17, 2, 1>),
                       dim3((uint32_t)nb), dim3(CT), lds_bytes, stream, a);
  else if (D == 17 && H == 64 && A == 6 && n_hidden == 2 && activation == 1)
    hipLaunchKernelGGL((mlp_chunk_kernel<4, 17, 64, 6, 2, 1>),
                       dim3((uint32_t)nb), dim3(CT), lds_bytes, stream, a);
  else if (D == 3 && H == 16 && A == 1 && n_hidden == 1 && activation == 0)
    hipLaunchKernelGGL((mlp_chunk_kernel<1, 3, 16, 1, 1, -1>),
                       dim3((uint32_t)nb), dim3(CT), lds_bytes, stream, a);
  else if (kc1 <= 8)
    hipLaunchKernelGGL(mlp_chunk_kernel<8>, dim3((uint32_t)nb), dim3(CT),
                       lds_bytes, stream, a);
  else if (kc1 <= 16)
    hipLaunchKernelGGL(mlp_chunk_kernel<16>, dim3((uint32_t)nb), dim3(CT),
                       lds_bytes, stream, a);
  else if (kc1 <= 32)
    hipLaunchKernelGGL(mlp_chunk_kernel<32>, dim3((uint32_t)nb), dim3(CT),
                       lds_bytes, stream, a);
  else if (kc1 <= 48)
    hipLaunchKernelGGL(mlp_chunk_kernel<48>, dim3((uint32_t)nb), dim3(CT),
                       lds_bytes, stream, a);
  else
    hipLaunchKernelGGL(mlp_chunk_kernel<64>, dim3((uint32_t)nb), dim3(CT),
                       lds_bytes, stream, a);

  hipLaunchKernelGGL(
      mlp_chunk_reduce_kernel, dim3(elementwise_grid(ptotal, 256)), dim3(256),
      0, stream, slabs.data_ptr<float>(), (int)nb, ppad, ptotal,
      params.data_ptr<float>(), mom.data_ptr<float>(), vel.data_ptr<float>(),
      coef.data_ptr<float>(), (float)beta1, (float)beta2, (float)eps,
      fuse_adam ? params.data_ptr<float>() : flat_grad.data_ptr<float>(),
      fuse_adam ? 1 : 0);
}
