// Fused DPPO rollout engine (gfx950): the ENTIRE T-step rollout in one
// kernel launch.
//
// The reference steps one env per sess.run (reference Worker.py:49,146 —
// per-step launch overhead dominates); the first eager
// rebuild still launches ~30 kernels per batched env step.  This kernel
// replaces all of it: per step it runs the policy MLP forward (K1-K3 in
// SURVEY.md §2.4), DiagGaussian sampling (K4), the epsilon-greedy overlay
// (Worker.py:149-152), the synthetic env dynamics + reward + done/reset
// (envs/synthetic.py), and the episode bookkeeping (Worker.py:57-65) —
// T times, with env state resident in LDS, so a whole rollout costs ONE
// launch and all policy/env weights stream from L1/L2.
//
// Decomposition: one block of FOUR waves (256 threads) owns ENV_TILE=8
// envs for the full T-step loop (E/8 blocks ≈ 4096 at the bench config).
// The four waves K-SPLIT each layer's dot products (wave w accumulates
// the k-quarter of every unit, partials combined through LDS), so the
// chip runs 4096 waves (≈4 waves/SIMD) at UNCHANGED weight traffic —
// the single-wave variant measured latency-bound at 1 wave/SIMD, every
// W-row L2 access exposed.  No inter-block communication exists (each
// block's envs are private, output rows disjoint), so the in-kernel
// T-loop needs no grid sync by construction.
//   - Layer phase: lane u computes output unit u for all 4 envs
//     (register-blocked: one W float4 load feeds 16 FMAs; W rows stream
//     per-lane from L1/L2; inputs broadcast from LDS via ds_read_b128 —
//     LDS is indexed through ONE shared array so the address space is
//     provable and no flat loads are emitted).
//   - Env phase: lanes split the obs dimension d; low-rank dynamics read
//     transposed Vt/Ut/Bt so every per-lane stream is contiguous.
//   - RNG: counter-based 32-bit (lowbias32 hash -> Box-Muller):
//     stateless, deterministic per launch, cheap (no 64-bit mults).
//   - Kernarg pressure: SEVEN pointers total (policy weights are offsets
//     into the flat parameter buffer; env matrices are one packed blob;
//     all outputs are one blob carved by the python side) — the previous
//     23-pointer struct spilled 80 SGPRs to VGPR lanes.
//
// Limits (checked in the binding): obs_dim <= 512, hidden <= 128,
// <= 3 hidden layers, act_dim <= 32, rank <= 32, Box/DiagGaussian only.
// The wide config (BASELINE #5) takes the GEMM path instead.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int ENV_TILE = 8;
constexpr int MAX_D = 512;
constexpr int MAX_H = 128;
constexpr int MAX_A = 32;
constexpr int MAX_R = 32;
constexpr int MAX_HIDDEN = 3;

constexpr int NWAVES = 4;  // waves per block

// The LDS image is dynamically sized to the ACTUAL dims (obs, widest
// hidden, act): a MAX-dims static image (44 KB at the flagship shapes
// needing only 27 KB) capped residency at 3 blocks/CU and left the
// kernel stall-bound.  All strides are rounded to 4 floats so float4
// LDS accesses stay 16-B aligned.
struct LdsMap {
  int x, h0, h1, pd, act, xv, part, val, rsum, epr, racc;  // float offsets
  int xs, hs, total;  // x row stride, h row stride, total floats
};

DEV_INLINE LdsMap lds_map(int D, int Hmax, int A, int rank, int et) {
  auto r4 = [](int v) { return (v + 3) & ~3; };
  LdsMap m;
  m.xs = r4(D);
  m.hs = r4(Hmax);
  int o = 0;
  m.x = o; o += et * m.xs;
  m.h0 = o; o += et * m.hs;
  m.h1 = o; o += et * m.hs;
  m.pd = o; o += et * r4(2 * A);
  m.act = o; o += et * r4(A);
  m.xv = o; o += et * r4(rank);
  m.part = o; o += NWAVES * et * m.hs;
  m.val = o; o += et;
  m.rsum = o; o += et;
  m.epr = o; o += et;
  m.racc = o; o += et;
  m.total = o;
  return m;
}

struct RolloutArgs {
  const float* params;   // rollout weight blob: TRANSPOSED layer weights
                         // (Wt[in][out]) + biases + Wv[in] + Wpt[in][2A] —
                         // per-k row reads are then fully coalesced
                         // (lane == unit); the torch-layout row-per-lane
                         // reads thrashed L1 (64 lines per instruction,
                         // 96 KB working set)
  const float* envblob;  // d | Vt[r][D] | Ut[D][r] | Bt[D][A]
  const int* horizons;   // [E]
  float* x;              // [E][D] persistent env state
  int* t;                // [E]
  float* epr;            // [E]
  float* out;            // states|pdflats|actions|values|rewards|dones|boot|moments
  int off_W[MAX_HIDDEN], off_b[MAX_HIDDEN];
  int dims[MAX_HIDDEN + 1];
  int off_Wv, off_bv, off_Wp, off_bp;
  int n_hidden, act_dim, activation, rank, h_max;
  float noise, act_low, act_high, eps_explore;
  int T, E, D;
  unsigned seed;
  int ablate;  // perf-ablation bitmask (0 in production): 1=trunk,
               // 2=heads, 4=sampling, 8=env dynamics, 16=buffer writes,
               // 32=env noise only, 64=low/ain inner products only
};

// ---- counter-based 32-bit RNG --------------------------------------------

DEV_INLINE unsigned lowbias32(unsigned x) {
  x ^= x >> 16;
  x *= 0x7feb352dU;
  x ^= x >> 15;
  x *= 0x846ca68bU;
  x ^= x >> 16;
  return x;
}

// uniform in (0, 1]
DEV_INLINE float rng_uniform(unsigned seed, int env, int step, int slot) {
  unsigned h = lowbias32(seed ^ (unsigned)env * 0x9E3779B9U ^
                         (unsigned)step * 0x85EBCA6BU ^
                         (unsigned)slot * 0xC2B2AE35U);
  return ((h >> 8) + 1) * (1.0f / 16777217.0f);
}

// One hash -> two 16-bit uniforms -> one Box-Muller pair.  The rollout
// draws ~1500 normals per block-step (env noise dominates); the naive
// 2-hash + log + cos per normal made the whole kernel transcendental-
// bound.  16-bit resolution truncates the tails at ~4.7 sigma, which is
// statistically invisible for synthetic-env noise and recorded actions.
DEV_INLINE float2 rng_normal2(unsigned seed, int env, int step, int slot) {
  const unsigned h = lowbias32(seed ^ (unsigned)env * 0x9E3779B9U ^
                               (unsigned)step * 0x85EBCA6BU ^
                               (unsigned)slot * 0xC2B2AE35U);
  const float u1 = ((h >> 16) + 1) * (1.0f / 65537.0f);
  const float u2 = (h & 0xFFFFu) * (1.0f / 65536.0f);
  const float r = sqrtf(-2.0f * __logf(u1));
  float sn, cs;
  __sincosf(6.2831853071795865f * u2, &sn, &cs);
  return make_float2(r * cs, r * sn);
}

DEV_INLINE float rng_normal(unsigned seed, int env, int step, int slot) {
  return rng_normal2(seed, env, step, slot).x;
}

// monotone float<->uint encoding for atomic max (min via -x); raw 0 is the
// neutral "below every float" element.
DEV_INLINE unsigned f2u_mono(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

DEV_INLINE void atomic_max_float(float* addr, float val) {
  atomicMax(reinterpret_cast<unsigned*>(addr), f2u_mono(val));
}

DEV_INLINE float u2f_mono(unsigned u) {
  return __uint_as_float((u & 0x80000000u) ? (u & 0x7fffffffu) : ~u);
}

// ---- the rollout kernel ---------------------------------------------------

// MINWAVES: waves/SIMD forced on the allocator.  4 caps VGPRs at 128 and
// SPILLS (29 VGPR + 185 SGPR, 96 B scratch/lane measured on this source) —
// both instantiated, dispatch measures/chooses.
// TD/TH/TA/TNH/TRANK/TACT = 0 -> runtime shapes; nonzero -> exact
// compile-time shapes (constant-folds the LDS map, index maps and trip
// counts — the same instruction-count program as mlp_chunk_kernel;
// profiles/r01_chunk_kernel_notes.md round-2 target list).
template <int MINWAVES, int ET = ENV_TILE, int TD = 0, int TH = 0,
          int TA = 0, int TNH = 0, int TRANK = 0, int TACT = 0>
__launch_bounds__(NWAVES * 64, MINWAVES)
__global__ void rollout_kernel(RolloutArgs a) {
  const int tid = threadIdx.x;      // block = 4 waves of 64
  const int lane = tid & (WAVE - 1);
  const int wv = tid / WAVE;        // k-split wave index
  const int e0 = blockIdx.x * ET;
  // specialized variants are dispatched only when E % ET == 0:
  // nE folds and every per-env tail guard disappears
  const int nE = TD ? ET : min(ET, a.E - e0);
  const int D = TD ? TD : a.D;
  const int A = TA ? TA : a.act_dim;
  const int P = 2 * A;
  const int T = a.T, E = a.E;
  const int RNK = TRANK ? TRANK : a.rank;
  const int NHID = TNH ? TNH : a.n_hidden;
  const int HMAX = TH ? TH : a.h_max;
  const int ACT = TACT ? (TACT > 0 ? 1 : 0) : a.activation;

  extern __shared__ __attribute__((aligned(16))) float lds[];
  const LdsMap lm = lds_map(D, HMAX, A, RNK, ET);
  const int X_OFF = lm.x, H0_OFF = lm.h0, H1_OFF = lm.h1, PD_OFF = lm.pd;
  const int ACT_OFF = lm.act, XV_OFF = lm.xv, PART_OFF = lm.part;
  const int MAX_D_S = lm.xs, MAX_H_S = lm.hs;
  const int PD_S = ((2 * A) + 3) & ~3, ACT_S = (A + 3) & ~3,
            XV_S = (RNK + 3) & ~3;
  float* val_lds = &lds[lm.val];
  float* rsum_lds = &lds[lm.rsum];
  float* epr_lds = &lds[lm.epr];
  float* racc_lds = &lds[lm.racc];
  __shared__ int tc_lds[ET];
  __shared__ int done_lds[ET];

  // env blob offsets (d | Vt[r][D] | U[r][D] | B[A][D])
  const float* env_d = a.envblob;
  const float* env_Vt = env_d + D;
  const float* env_U = env_Vt + (int64_t)RNK * D;
  const float* env_B = env_U + (int64_t)RNK * D;
  // output blob offsets
  float* out_states = a.out;
  float* out_pdflats = out_states + (int64_t)T * E * D;
  float* out_actions = out_pdflats + (int64_t)T * E * P;
  float* out_values = out_actions + (int64_t)T * E * A;
  float* out_rewards = out_values + (int64_t)T * E;
  float* out_dones = out_rewards + (int64_t)T * E;
  float* out_boot = out_dones + (int64_t)T * E;

  // ---- load persistent state; zero tail envs ----
  for (int e = 0; e < ET; ++e) {
    for (int d = tid; d < D; d += NWAVES * WAVE)
      lds[X_OFF + e * MAX_D_S + d] =
          (e < nE) ? a.x[(int64_t)(e0 + e) * D + d] : 0.f;
  }
  if (tid < nE) {
    epr_lds[tid] = a.epr[e0 + tid];
    tc_lds[tid] = a.t[e0 + tid];
  }
  __syncthreads();

  // K-split layer forward: wave wv accumulates its k-quarter of every
  // unit into the partial slab; a combine pass sums the 4 partials,
  // adds bias and applies the activation.
  // W is TRANSPOSED ([in_dim][w_cols]); lane u reads column u of k-rows:
  // one coalesced 256 B wave-read per k, streaming in k.
  auto layer_kpart = [&](const float* W, int w_cols, int off_Wv_u, int in_off,
                         int in_stride, int in_dim, int out_dim,
                         bool heads) {
    const int kq = (((in_dim + NWAVES * 4 - 1) / (NWAVES * 4)) * 4);
    const int k0 = wv * kq;
    const int k1 = min(in_dim, k0 + kq);
    for (int u = lane; u < out_dim; u += WAVE) {
      float acc[ET];
      #pragma unroll
      for (int e = 0; e < ET; ++e) acc[e] = 0.f;
      const bool is_v = heads && (u == out_dim - 1);
      const float* Wcol = is_v ? (a.params + off_Wv_u) : (W + u);
      const int stride = is_v ? 1 : w_cols;
      // 4 k-steps per iteration: one float4 LDS broadcast per env feeds
      // 4 FMAs (scalar b32 broadcasts measured as a main trunk cost),
      // and the 4 independent W loads pipeline.
      int k = k0;
      #pragma unroll 2
      for (; k + 4 <= k1; k += 4) {
        const float w0 = Wcol[(int64_t)(k + 0) * stride];
        const float w1 = Wcol[(int64_t)(k + 1) * stride];
        const float w2 = Wcol[(int64_t)(k + 2) * stride];
        const float w3 = Wcol[(int64_t)(k + 3) * stride];
        #pragma unroll
        for (int e = 0; e < ET; ++e) {
          const float4 x4 = *reinterpret_cast<const float4*>(
              __builtin_assume_aligned(&lds[in_off + e * in_stride + k], 16));
          acc[e] += w0 * x4.x + w1 * x4.y + w2 * x4.z + w3 * x4.w;
        }
      }
      for (; k < k1; ++k) {
        const float w = Wcol[(int64_t)k * stride];
        #pragma unroll
        for (int e = 0; e < ET; ++e)
          acc[e] += w * lds[in_off + e * in_stride + k];
      }
      #pragma unroll
      for (int e = 0; e < ET; ++e)
        lds[PART_OFF + (wv * ET + e) * MAX_H_S + u] = acc[e];
    }
  };

  for (int step = 0; step < T; ++step) {
    // ---- write current obs ----
    if (!(a.ablate & 16)) {
      for (int e = 0; e < nE; ++e) {
        const int64_t base = ((int64_t)step * E + e0 + e) * D;
        for (int d = tid; d < D; d += NWAVES * WAVE)
          out_states[base + d] = lds[X_OFF + e * MAX_D_S + d];
      }
    }

    // ---- env low-rank projection, merged into the trunk phase: both
    // only READ x, and the trunk's first barrier publishes the xv slab
    // long before the env phase consumes it ----
    // ---- env low-rank projection: wave wv handles env wv; lanes split
    if (!(a.ablate & 8))
    // as (rr, k-quarter) so all 64 lanes stream and the per-lane load
    // chain is 4x shorter (the 16-lane serial version stalled the whole
    // block at the next barrier) ----
    for (int e = wv; e < nE; e += NWAVES) {
      const int rr = lane & 15;
      const int kq = lane >> 4;  // 4 k-quarters
      const int kq_len = ((D / 4 + 3) & ~3);
      const int k0q = kq * kq_len;
      const int k1q = min(D, k0q + kq_len);
      float accv = 0.f;
      if (rr < RNK) {
        const float* Vrow = env_Vt + (int64_t)rr * D;
        int k = k0q;
        #pragma unroll 2
        for (; k + 4 <= k1q; k += 4) {
          const float4 v4 = *reinterpret_cast<const float4*>(Vrow + k);
          const float4 x4 =
              *reinterpret_cast<const float4*>(
              __builtin_assume_aligned(&lds[X_OFF + e * MAX_D_S + k], 16));
          accv += v4.x * x4.x + v4.y * x4.y + v4.z * x4.z + v4.w * x4.w;
        }
        for (; k < k1q; ++k) accv += Vrow[k] * lds[X_OFF + e * MAX_D_S + k];
      }
      // butterfly-reduce over the k-quarter lanes (bits 4 and 5)
      accv += __shfl_xor(accv, 16, WAVE);
      accv += __shfl_xor(accv, 32, WAVE);
      if (kq == 0 && rr < RNK) lds[XV_OFF + e * XV_S + rr] = accv;
    }

    // ---- policy MLP forward (K-split + combine per layer) ----
    int in_off = X_OFF, in_stride = MAX_D_S, in_dim = D;
    if (!(a.ablate & 1))
    for (int l = 0; l < NHID; ++l) {
      const int out_dim = TH ? TH : a.dims[l + 1];
      const int out_off = (l & 1) ? H1_OFF : H0_OFF;
      layer_kpart(a.params + a.off_W[l], out_dim, 0, in_off, in_stride,
                  in_dim, out_dim, false);
      __syncthreads();
      const float* bias = a.params + a.off_b[l];
      for (int idx = tid; idx < ET * out_dim; idx += NWAVES * WAVE) {
        const int e = idx / out_dim, u = idx % out_dim;
        float sum = bias[u];
        #pragma unroll
        for (int w = 0; w < NWAVES; ++w)
          sum += lds[PART_OFF + (w * ET + e) * MAX_H_S + u];
        lds[out_off + e * MAX_H_S + u] =
            ACT ? fast_tanhf(sum) : fmaxf(sum, 0.f);
      }
      __syncthreads();
      in_off = out_off;
      in_stride = MAX_H_S;
      in_dim = out_dim;
    }
    if (a.ablate & 1) {
      in_off = (NHID & 1) ? H0_OFF : H1_OFF;
      in_stride = MAX_H_S;
      in_dim = TH ? TH : a.dims[NHID];
    }

    // ---- heads (u < P: pd params; u == P: value), no activation ----
    if (!(a.ablate & 2)) {
      layer_kpart(a.params + a.off_Wp, P, a.off_Wv, in_off, in_stride, in_dim,
                  P + 1, true);
      __syncthreads();
      for (int idx = tid; idx < ET * (P + 1); idx += NWAVES * WAVE) {
        const int e = idx / (P + 1), u = idx % (P + 1);
        float sum = a.params[(u == P) ? a.off_bv : a.off_bp + u];
        #pragma unroll
        for (int w = 0; w < NWAVES; ++w)
          sum += lds[PART_OFF + (w * ET + e) * MAX_H_S + u];
        if (u == P) val_lds[e] = sum;
        else lds[PD_OFF + e * PD_S + u] = sum;
      }
      __syncthreads();
    }

    // ---- sample actions (threads = (env, dim) pairs) ----
    if (!(a.ablate & 4)) {
      const int e = tid / MAX_A;   // 8 groups x 32 dims
      const int j = tid % MAX_A;
      if (e < nE && j < A) {
        const int ge = e0 + e;
        const float mean = lds[PD_OFF + e * PD_S + j];
        const float logstd = lds[PD_OFF + e * PD_S + A + j];
        float act = mean + __expf(logstd) * rng_normal(a.seed, ge, step, j);
        // epsilon-greedy overlay (Worker.py:149-152)
        const float u_dec = rng_uniform(a.seed, ge, step, 90001);
        if (u_dec < a.eps_explore) {
          const float u = rng_uniform(a.seed, ge, step, 90010 + j);
          act = a.act_low + (a.act_high - a.act_low) * u;
        }
        lds[ACT_OFF + e * ACT_S + j] = act;
      }
    }
    if (tid < ET) racc_lds[tid] = 0.f;
    __syncthreads();

    // ---- env state update + reward partials (threads split d) ----
    // ---- pdflat/action/value buffer writes, folded into the env phase
    // (their inputs were published by the sampling barrier) ----
    if (!(a.ablate & 16)) {
      for (int e = 0; e < nE; ++e) {
        const int64_t row = (int64_t)step * E + e0 + e;
        if (tid < P) out_pdflats[row * P + tid] = lds[PD_OFF + e * PD_S + tid];
        if (tid < A) out_actions[row * A + tid] = lds[ACT_OFF + e * ACT_S + tid];
      }
      if (tid < nE) out_values[(int64_t)step * E + e0 + tid] = val_lds[tid];
    }
    float racc[ET];
    #pragma unroll
    for (int e = 0; e < ET; ++e) racc[e] = 0.f;
    if (!(a.ablate & 8))
    for (int d = tid; d < D; d += NWAVES * WAVE) {
      const float dd = env_d[d];
      // one Box-Muller pair per (even env, d) feeds two envs
      float nz[ET];
      #pragma unroll
      for (int q = 0; q < ET; ++q) nz[q] = 0.f;
      if (!(a.ablate & 32)) {
        #pragma unroll
        for (int q = 0; q < ET / 2; ++q) {
          const float2 p = rng_normal2(a.seed, e0 + 2 * q, step, 1000 + d);
          nz[2 * q] = p.x;
          nz[2 * q + 1] = p.y;
        }
      }
      // U[rr][d] / B[j][d] row loads: lane == d, fully coalesced, ONE
      // load per rr/j shared by all four envs (the previous transposed
      // per-d rows cost 33 uncoalesced loads per env per d — the kernel
      // measured VMEM-wait bound, SQ_WAIT_ANY 27x SQ_BUSY).
      float low[ET], ain[ET];
      #pragma unroll
      for (int e = 0; e < ET; ++e) low[e] = ain[e] = 0.f;
      if (!(a.ablate & 64)) {
        // float4 LDS reads: 4x fewer LDS instructions than scalar
        // broadcasts (the scalar form measured 2.5 ms of the rollout)
        int rr = 0;
        for (; rr + 4 <= RNK; rr += 4) {
          const float u0 = env_U[(int64_t)(rr + 0) * D + d];
          const float u1 = env_U[(int64_t)(rr + 1) * D + d];
          const float u2 = env_U[(int64_t)(rr + 2) * D + d];
          const float u3 = env_U[(int64_t)(rr + 3) * D + d];
          #pragma unroll
          for (int e = 0; e < ET; ++e) {
            const float4 xv4 =
                *reinterpret_cast<const float4*>(
              __builtin_assume_aligned(&lds[XV_OFF + e * XV_S + rr], 16));
            low[e] += xv4.x * u0 + xv4.y * u1 + xv4.z * u2 + xv4.w * u3;
          }
        }
        for (; rr < RNK; ++rr) {
          const float uv = env_U[(int64_t)rr * D + d];
          #pragma unroll
          for (int e = 0; e < ET; ++e)
            low[e] += lds[XV_OFF + e * XV_S + rr] * uv;
        }
        int j = 0;
        for (; j + 4 <= A; j += 4) {
          const float b0 = env_B[(int64_t)(j + 0) * D + d];
          const float b1 = env_B[(int64_t)(j + 1) * D + d];
          const float b2 = env_B[(int64_t)(j + 2) * D + d];
          const float b3 = env_B[(int64_t)(j + 3) * D + d];
          #pragma unroll
          for (int e = 0; e < ET; ++e) {
            const float4 a4 =
                *reinterpret_cast<const float4*>(
              __builtin_assume_aligned(&lds[ACT_OFF + e * ACT_S + j], 16));
            ain[e] += a4.x * b0 + a4.y * b1 + a4.z * b2 + a4.w * b3;
          }
        }
        for (; j < A; ++j) {
          const float bvv = env_B[(int64_t)j * D + d];
          #pragma unroll
          for (int e = 0; e < ET; ++e)
            ain[e] += lds[ACT_OFF + e * ACT_S + j] * bvv;
        }
      }
      for (int e = 0; e < nE; ++e) {
        const float xn = fast_tanhf(lds[X_OFF + e * MAX_D_S + d] * dd +
                                    low[e] + ain[e] + a.noise * nz[e]);
        lds[X_OFF + e * MAX_D_S + d] = xn;
        racc[e] += xn * xn;
      }
    }
    #pragma unroll
    for (int e = 0; e < ET; ++e) {
      const float w = wave_reduce_sum(racc[e]);
      if (lane == 0 && e < nE) atomicAdd(&racc_lds[e], w);
    }
    __syncthreads();

    // ---- reward, done, episode bookkeeping (thread e handles env e) ----
    if (tid < nE) {
      const int e = tid;
      const float r = 1.0f - racc_lds[e] / D;
      out_rewards[(int64_t)step * E + e0 + e] = r;
      epr_lds[e] += r;
      int tc = tc_lds[e] + 1;
      const int done = (tc >= a.horizons[e0 + e]) ? 1 : 0;
      out_dones[(int64_t)step * E + e0 + e] = (float)done;
      if (done) {
        epr_lds[e] = 0.f;
        tc = 0;
      }
      tc_lds[e] = tc;
      done_lds[e] = done;
    }
    __syncthreads();

    // ---- reset finished envs ----
    for (int e = 0; e < nE; ++e) {
      if (done_lds[e]) {
        for (int d = tid; d < D; d += NWAVES * WAVE)
          lds[X_OFF + e * MAX_D_S + d] =
              0.1f * rng_normal(a.seed, e0 + e, step, 5000 + d);
      }
    }
    __syncthreads();
  }

  // ---- bootstrap value V(x_T): trunk + value head ----
  {
    int in_off = X_OFF, in_stride = MAX_D_S, in_dim = D;
    for (int l = 0; l < NHID; ++l) {
      const int out_dim = TH ? TH : a.dims[l + 1];
      const int out_off = (l & 1) ? H1_OFF : H0_OFF;
      layer_kpart(a.params + a.off_W[l], out_dim, 0, in_off, in_stride,
                  in_dim, out_dim, false);
      __syncthreads();
      const float* bias = a.params + a.off_b[l];
      for (int idx = tid; idx < ET * out_dim; idx += NWAVES * WAVE) {
        const int e = idx / out_dim, u = idx % out_dim;
        float sum = bias[u];
        #pragma unroll
        for (int w = 0; w < NWAVES; ++w)
          sum += lds[PART_OFF + (w * ET + e) * MAX_H_S + u];
        lds[out_off + e * MAX_H_S + u] =
            ACT ? fast_tanhf(sum) : fmaxf(sum, 0.f);
      }
      __syncthreads();
      in_off = out_off;
      in_stride = MAX_H_S;
      in_dim = out_dim;
    }
    // value head: wave wv reduces env wv over lanes
    for (int e = wv; e < nE; e += NWAVES) {
      const float* Wv = a.params + a.off_Wv;
      float acc = 0.f;
      for (int k = lane; k < in_dim; k += WAVE)
        acc += Wv[k] * lds[in_off + e * in_stride + k];
      const float total = wave_reduce_sum(acc);
      if (lane == 0) out_boot[e0 + e] = total + a.params[a.off_bv];
    }
  }

  // ---- persist env state ----
  for (int e = 0; e < nE; ++e) {
    for (int d = tid; d < D; d += NWAVES * WAVE)
      a.x[(int64_t)(e0 + e) * D + d] = lds[X_OFF + e * MAX_D_S + d];
  }
  if (tid < nE) {
    a.epr[e0 + tid] = epr_lds[tid];
    a.t[e0 + tid] = tc_lds[tid];
  }
}

// Episode moments recomputed from the rewards/dones streams (identical
// math to the in-rollout epr updates); min/max go through the monotone
// atomicMax encoding and a decode kernel.
__global__ void ep_moments_kernel(const float* __restrict__ rewards,
                                  const float* __restrict__ dones,
                                  const float* __restrict__ epr_in,
                                  float* __restrict__ out,  // [5]
                                  int64_t T, int64_t E) {
  const int64_t e = gidx();
  float c = 0.f, s = 0.f, ss = 0.f, mn = 3.0e38f, mx = -3.0e38f;
  if (e < E) {
    float epr = epr_in[e];
    for (int64_t t = 0; t < T; ++t) {
      const int64_t i = t * E + e;
      epr += rewards[i];
      if (dones[i] != 0.f) {
        c += 1.f;
        s += epr;
        ss += epr * epr;
        mn = fminf(mn, epr);
        mx = fmaxf(mx, epr);
        epr = 0.f;
      }
    }
  }
  atomicAdd(&out[0], c);
  atomicAdd(&out[1], s);
  atomicAdd(&out[2], ss);
  if (c > 0.f) {
    atomic_max_float(&out[3], -mn);  // min via negated max
    atomic_max_float(&out[4], mx);
  }
}

__global__ void ep_moments_decode_kernel(float* __restrict__ out) {
  const unsigned u3 = __float_as_uint(out[3]);
  const unsigned u4 = __float_as_uint(out[4]);
  out[3] = (u3 == 0u) ? 0.f : -u2f_mono(u3);
  out[4] = (u4 == 0u) ? 0.f : u2f_mono(u4);
}

// ---------------------------------------------------------------------------
// Per-step rollout path (v3): the T-step loop runs as a hipGraph of
// pipelined MFMA GEMMs (mfma_gemm.hip does the MLP forward AND the env
// dynamics' low-rank/action matmuls as [E][36] @ [36][D] panels); the two
// kernels below supply the non-GEMM pieces.  RNG slots/order are IDENTICAL
// to rollout_kernel, so both paths sample the same actions/noise/resets
// for a given seed.  seed/eps live in DEVICE memory so one captured graph
// replays across rounds (they change per round/rollout).
// ---------------------------------------------------------------------------

// thread per (env, action-dim): sample + epsilon-greedy; also writes the
// action into the [E][xva_w] concat buffer consumed by the dynamics GEMM.
__global__ void sample_kernel(const float* __restrict__ pdflat,  // [E][2A]
                              float* __restrict__ actions,       // [E][A]
                              float* __restrict__ xva,           // [E][xva_w]
                              const long* __restrict__ seed_dev,
                              const float* __restrict__ eps_dev,
                              int step, int64_t E, int A, int xva_w,
                              int va_off, float act_low, float act_high) {
  const unsigned seed = (unsigned)*seed_dev;
  const float eps = *eps_dev;
  const int P = 2 * A;
  for (int64_t i = gidx(); i < E * A; i += gstride()) {
    const int64_t e = i / A;
    const int j = (int)(i % A);
    const float mean = pdflat[e * P + j];
    const float logstd = pdflat[e * P + A + j];
    float act = mean + __expf(logstd) * rng_normal(seed, (int)e, step, j);
    const float u_dec = rng_uniform(seed, (int)e, step, 90001);
    if (u_dec < eps) {
      const float u = rng_uniform(seed, (int)e, step, 90010 + j);
      act = act_low + (act_high - act_low) * u;
    }
    actions[e * A + j] = act;
    xva[e * xva_w + va_off + j] = act;
  }
}

// wave per env: x' = tanh(x*d + G + sigma*noise), reward, done/reset and
// episode bookkeeping.  G = [x@V | act] @ [U; B] comes from the GEMM.
__global__ void env_finish_kernel(
    const float* __restrict__ xin,     // [E][D] state at `step`
    float* __restrict__ xout,          // [E][D] state at `step`+1 (the v3
                                       // loop passes the states-blob slot
                                       // directly — one stream, no env.x
                                       // mirror write; xout may alias xin)
    const float* __restrict__ G,       // [E][D]
    const float* __restrict__ envd,    // [D] diagonal
    const int* __restrict__ horizons,  // [E]
    int* __restrict__ t,               // [E]
    float* __restrict__ epr,           // [E]
    float* __restrict__ rewards,       // [E] out_rewards[step]
    float* __restrict__ dones,         // [E] out_dones[step]
    const long* __restrict__ seed_dev, float sigma, int step, int64_t E,
    int D) {
  const unsigned seed = (unsigned)*seed_dev;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid =
      ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  if (wid >= E) return;
  const int64_t e = wid;
  // float4 per lane (the v3 gate guarantees D%4==0): the scalar version
  // kept only ~4 B/lane in flight and ran 2x its streaming floor.  Noise
  // slots differ from the fused kernel (one Box-Muller pair feeds d and
  // d+1 of the SAME env, slot 1000+d/2) — each path is deterministic per
  // seed on its own; nothing requires the two paths to share trajectories.
  constexpr int CMAX = (MAX_D / 4 + WAVE - 1) / WAVE;  // float4 chunks/lane
  const int nc4 = D / 4;
  float4 xn[CMAX];
  float ss = 0.f;
  int it = 0;
  for (int c = lane; c < nc4; c += WAVE, ++it) {
    const int d = 4 * c;
    const float4 xv = *reinterpret_cast<const float4*>(&xin[e * D + d]);
    const float4 gv = *reinterpret_cast<const float4*>(&G[e * D + d]);
    const float4 dv = *reinterpret_cast<const float4*>(&envd[d]);
    float nz[4] = {0.f, 0.f, 0.f, 0.f};
    if (sigma != 0.f) {
      const float2 p0 = rng_normal2(seed, (int)e, step, 1000 + 2 * c);
      const float2 p1 = rng_normal2(seed, (int)e, step, 1000 + 2 * c + 1);
      nz[0] = p0.x; nz[1] = p0.y; nz[2] = p1.x; nz[3] = p1.y;
    }
    float4 v;
    v.x = fast_tanhf(xv.x * dv.x + gv.x + sigma * nz[0]);
    v.y = fast_tanhf(xv.y * dv.y + gv.y + sigma * nz[1]);
    v.z = fast_tanhf(xv.z * dv.z + gv.z + sigma * nz[2]);
    v.w = fast_tanhf(xv.w * dv.w + gv.w + sigma * nz[3]);
    xn[it] = v;
    ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
  }
  ss = wave_reduce_sum(ss);
  int done;
  if (lane == 0) {
    const float r = 1.0f - ss / D;
    rewards[e] = r;
    float ep = epr[e] + r;
    int tc = t[e] + 1;
    done = (tc >= horizons[e]) ? 1 : 0;
    dones[e] = (float)done;
    if (done) {
      ep = 0.f;
      tc = 0;
    }
    epr[e] = ep;
    t[e] = tc;
  }
  done = __shfl(done, 0, WAVE);
  it = 0;
  for (int c = lane; c < nc4; c += WAVE, ++it) {
    const int d = 4 * c;
    float4 v = xn[it];
    if (done) {
      v.x = 0.1f * rng_normal(seed, (int)e, step, 5000 + d);
      v.y = 0.1f * rng_normal(seed, (int)e, step, 5000 + d + 1);
      v.z = 0.1f * rng_normal(seed, (int)e, step, 5000 + d + 2);
      v.w = 0.1f * rng_normal(seed, (int)e, step, 5000 + d + 3);
    }
    *reinterpret_cast<float4*>(&xout[e * D + d]) = v;
  }
}

}  // namespace

std::vector<torch::Tensor> rollout_run(
    torch::Tensor params, std::vector<int64_t> offsets,
    std::vector<int64_t> dims, int64_t activation,
    torch::Tensor envblob, int64_t rank, torch::Tensor horizons,
    double noise, double act_low, double act_high, double eps_explore,
    torch::Tensor x, torch::Tensor t, torch::Tensor epr,
    int64_t T, int64_t act_dim, int64_t seed, torch::Tensor out_buf,
    int64_t ablate) {
  // offsets: [W0, b0, W1, b1, ..., Wv, bv, Wp, bp] into params
  // dims: [obs, h1, ..., hN]
  const int64_t E = x.size(0);
  const int64_t D = x.size(1);
  const int A = static_cast<int>(act_dim);
  const int n_hidden = static_cast<int>(dims.size()) - 1;
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kFloat32 && x.is_contiguous());
  TORCH_CHECK(params.is_contiguous() && params.dtype() == torch::kFloat32);
  TORCH_CHECK(n_hidden >= 1 && n_hidden <= MAX_HIDDEN, "1..3 hidden layers");
  TORCH_CHECK(D <= MAX_D && A <= MAX_A && rank <= MAX_R,
              "dims exceed rollout-kernel limits");
  TORCH_CHECK(dims[0] == D, "first layer must consume obs_dim");
  TORCH_CHECK(horizons.dtype() == torch::kInt32 && t.dtype() == torch::kInt32);
  TORCH_CHECK(static_cast<int>(offsets.size()) == 2 * n_hidden + 4);
  for (int l = 0; l < n_hidden; ++l)
    TORCH_CHECK(dims[l + 1] <= MAX_H, "hidden width exceeds kernel limit");

  RolloutArgs a{};
  a.params = params.data_ptr<float>();
  a.envblob = envblob.data_ptr<float>();
  a.horizons = horizons.data_ptr<int>();
  a.x = x.data_ptr<float>();
  a.t = t.data_ptr<int>();
  a.epr = epr.data_ptr<float>();
  for (int l = 0; l < n_hidden; ++l) {
    a.off_W[l] = static_cast<int>(offsets[2 * l]);
    a.off_b[l] = static_cast<int>(offsets[2 * l + 1]);
    a.dims[l] = static_cast<int>(dims[l]);
  }
  a.dims[n_hidden] = static_cast<int>(dims[n_hidden]);
  a.off_Wv = static_cast<int>(offsets[2 * n_hidden]);
  a.off_bv = static_cast<int>(offsets[2 * n_hidden + 1]);
  a.off_Wp = static_cast<int>(offsets[2 * n_hidden + 2]);
  a.off_bp = static_cast<int>(offsets[2 * n_hidden + 3]);
  a.n_hidden = n_hidden;
  int h_max = 0;
  for (int l = 1; l <= n_hidden; ++l)
    h_max = std::max(h_max, static_cast<int>(dims[l]));
  a.h_max = h_max;
  a.act_dim = A;
  a.activation = static_cast<int>(activation);
  a.rank = static_cast<int>(rank);
  a.noise = static_cast<float>(noise);
  a.act_low = static_cast<float>(act_low);
  a.act_high = static_cast<float>(act_high);
  a.eps_explore = static_cast<float>(eps_explore);
  a.T = static_cast<int>(T);
  a.E = static_cast<int>(E);
  a.D = static_cast<int>(D);
  a.seed = static_cast<unsigned>(seed & 0xFFFFFFFFULL);
  a.ablate = static_cast<int>(ablate);

  const int P = 2 * A;
  const int64_t n_out = T * E * (D + P + A + 3) + E + 5;
  // caller may provide a persistent output blob (stable addresses for
  // hipGraph capture of the downstream update)
  torch::Tensor out = (out_buf.numel() == n_out)
                          ? out_buf
                          : torch::empty({n_out}, x.options());
  auto epr_in = epr.clone();
  a.out = out.data_ptr<float>();

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  // tiny-E: 2 envs per block -> 4x the blocks (E=64 fills 32 CUs instead
  // of 8) AND a ~4x shorter per-phase issue chain (the per-lane env
  // unrolls shrink); large E keeps the 8-env tile (ILP beats occupancy
  // once the grid fills the chip)
  const int et = ((E + ENV_TILE - 1) / ENV_TILE < 128 && E % 2 == 0) ? 2
                                                                     : ENV_TILE;
  const int grid = static_cast<int>((E + et - 1) / et);
  auto r4 = [](int v) { return (v + 3) & ~3; };
  const size_t lds_bytes =
      (et * (r4((int)D) + 2 * r4(h_max) + r4(2 * A) + r4(A) +
             r4((int)rank) + 4) +
       NWAVES * et * r4(h_max)) *
      sizeof(float);
  // MINWAVES=4 measured 24.8 ms vs 27.7 ms at MINWAVES=3 (65k envs):
  // the 4th wave/SIMD buys more than the 29-VGPR spill costs.  Below
  // ~1 block/CU (tiny-E configs: BASELINE #2/#4) occupancy is grid-bound
  // anyway, so MINWAVES=1 lifts the 128-VGPR cap — no spills, no
  // per-lane scratch on the latency-critical per-step chain.
  // DPPO_ROLLOUT_MW: absent -> auto (small grids get the unconstrained
  // 1-wave/SIMD minimum, big grids the 4-wave one); explicit values pick
  // the __launch_bounds__ min-waves instantiation: >=4 -> <4>, 2..3 ->
  // <3>, 0..1 -> <1> (0 = "no minimum" = the 128-VGPR-cap-lifted kernel).
  static const int mw_env = []() {
    const char* e = getenv("DPPO_ROLLOUT_MW");
    return e ? atoi(e) : -1;  // -1 = unset, distinct from explicit 0
  }();
  const int mw = mw_env >= 0 ? mw_env : (grid <= N_CU ? 1 : 4);
  // exact-shape specializations for the BASELINE families (index maps,
  // trip counts and the LDS map constant-folded)
  const bool full_tiles = (a.E % et) == 0;  // nE folds to the tile
  const bool hc_shape = full_tiles && (a.D == 17 && a.h_max == 64 &&
                         a.act_dim == 6 && a.n_hidden == 2 &&
                         a.rank == 16 && a.activation == 1);
  const bool hum_shape = full_tiles && (a.D == 376 && a.h_max == 64 &&
                          a.act_dim == 17 && a.n_hidden == 2 &&
                          a.rank == 16 && a.activation == 1);
  if (mw >= 4) {
    if (hc_shape && et == 8)
      hipLaunchKernelGGL((rollout_kernel<4, 8, 17, 64, 6, 2, 16, 1>),
                         dim3(grid), dim3(NWAVES * WAVE), lds_bytes, stream,
                         a);
    else if (hum_shape && et == 8)
      hipLaunchKernelGGL((rollout_kernel<4, 8, 376, 64, 17, 2, 16, 1>),
                         dim3(grid), dim3(NWAVES * WAVE), lds_bytes, stream,
                         a);
    else if (et == 2)
      hipLaunchKernelGGL((rollout_kernel<4, 2>), dim3(grid),
                         dim3(NWAVES * WAVE), lds_bytes, stream, a);
    else
      hipLaunchKernelGGL(rollout_kernel<4>, dim3(grid), dim3(NWAVES * WAVE),
                         lds_bytes, stream, a);
  } else if (mw >= 2) {
    if (et == 2)
      hipLaunchKernelGGL((rollout_kernel<3, 2>), dim3(grid),
                         dim3(NWAVES * WAVE), lds_bytes, stream, a);
    else
      hipLaunchKernelGGL(rollout_kernel<3>, dim3(grid), dim3(NWAVES * WAVE),
                         lds_bytes, stream, a);
  } else {
    if (hc_shape && et == 2)
      hipLaunchKernelGGL((rollout_kernel<1, 2, 17, 64, 6, 2, 16, 1>),
                         dim3(grid), dim3(NWAVES * WAVE), lds_bytes, stream,
                         a);
    else if (hum_shape && et == 2)
      hipLaunchKernelGGL((rollout_kernel<1, 2, 376, 64, 17, 2, 16, 1>),
                         dim3(grid), dim3(NWAVES * WAVE), lds_bytes, stream,
                         a);
    else if (hc_shape)
      hipLaunchKernelGGL((rollout_kernel<1, 8, 17, 64, 6, 2, 16, 1>),
                         dim3(grid), dim3(NWAVES * WAVE), lds_bytes, stream,
                         a);
    else if (hum_shape)
      hipLaunchKernelGGL((rollout_kernel<1, 8, 376, 64, 17, 2, 16, 1>),
                         dim3(grid), dim3(NWAVES * WAVE), lds_bytes, stream,
                         a);
    else if (et == 2)
      hipLaunchKernelGGL((rollout_kernel<1, 2>), dim3(grid),
                         dim3(NWAVES * WAVE), lds_bytes, stream, a);
    else
      hipLaunchKernelGGL(rollout_kernel<1>, dim3(grid), dim3(NWAVES * WAVE),
                         lds_bytes, stream, a);
  }

  // carve views out of the blob
  int64_t o = 0;
  auto take = [&](std::vector<int64_t> shape) {
    int64_t n = 1;
    for (auto sdim : shape) n *= sdim;
    auto v = out.narrow(0, o, n).view(shape);
    o += n;
    return v;
  };
  auto states = take({T, E, D});
  auto pdflats = take({T, E, (int64_t)P});
  auto actions = take({T, E, (int64_t)A});
  auto values = take({T, E});
  auto rewards = take({T, E});
  auto dones = take({T, E});
  auto boot_v = take({E});
  auto moments = take({5});
  moments.zero_();

  hipLaunchKernelGGL(ep_moments_kernel,
                     dim3(static_cast<int>((E + 255) / 256)), dim3(256), 0,
                     stream, rewards.data_ptr<float>(), dones.data_ptr<float>(),
                     epr_in.data_ptr<float>(), moments.data_ptr<float>(), T, E);
  hipLaunchKernelGGL(ep_moments_decode_kernel, dim3(1), dim3(1), 0, stream,
                     moments.data_ptr<float>());
  return {states, pdflats, actions, values, rewards, dones, boot_v, moments};
}

void rollout_sample(torch::Tensor pdflat, torch::Tensor actions,
                    torch::Tensor xva, torch::Tensor seed_dev,
                    torch::Tensor eps_dev, int64_t step, int64_t va_off,
                    double act_low, double act_high) {
  const int64_t E = pdflat.size(0);
  const int A = static_cast<int>(pdflat.size(1) / 2);
  TORCH_CHECK(pdflat.is_cuda() && actions.numel() == E * A);
  TORCH_CHECK(seed_dev.dtype() == torch::kInt64 && eps_dev.dtype() == torch::kFloat32);
  const int xva_w = static_cast<int>(xva.size(1));
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(sample_kernel,
                     dim3(elementwise_grid(E * A, 256)), dim3(256), 0, stream,
                     pdflat.data_ptr<float>(), actions.data_ptr<float>(),
                     xva.data_ptr<float>(), seed_dev.data_ptr<int64_t>(),
                     eps_dev.data_ptr<float>(), (int)step, E, A, xva_w,
                     (int)va_off, (float)act_low, (float)act_high);
}

void rollout_env_step(torch::Tensor xin, torch::Tensor xout, torch::Tensor G,
                      torch::Tensor envd, torch::Tensor horizons,
                      torch::Tensor t, torch::Tensor epr,
                      torch::Tensor rewards, torch::Tensor dones,
                      torch::Tensor seed_dev, double sigma, int64_t step) {
  const int64_t E = xin.size(0);
  const int D = static_cast<int>(xin.size(1));
  TORCH_CHECK(xin.is_cuda() && xin.is_contiguous() && G.numel() >= E * D);
  TORCH_CHECK(xout.is_contiguous() && xout.numel() == E * D);
  TORCH_CHECK(D <= MAX_D && D % 4 == 0, "env_finish_kernel is float4-wide");
  TORCH_CHECK(horizons.dtype() == torch::kInt32 && t.dtype() == torch::kInt32);
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int64_t waves = E;
  const int64_t blocks = (waves * WAVE + 255) / 256;
  hipLaunchKernelGGL(env_finish_kernel,
                     dim3((unsigned)std::min<int64_t>(blocks, 1 << 26)),
                     dim3(256), 0, stream, xin.data_ptr<float>(),
                     xout.data_ptr<float>(), G.data_ptr<float>(),
                     envd.data_ptr<float>(), horizons.data_ptr<int>(),
                     t.data_ptr<int>(), epr.data_ptr<float>(),
                     rewards.data_ptr<float>(), dones.data_ptr<float>(),
                     seed_dev.data_ptr<int64_t>(), (float)sigma, (int)step, E,
                     D);
}

torch::Tensor rollout_moments(torch::Tensor rewards, torch::Tensor dones,
                              torch::Tensor epr_in, int64_t T, int64_t E) {
  auto out = torch::zeros({5}, rewards.options());
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(ep_moments_kernel, dim3(elementwise_grid(E, 256)),
                     dim3(256), 0, stream, rewards.data_ptr<float>(),
                     dones.data_ptr<float>(), epr_in.data_ptr<float>(),
                     out.data_ptr<float>(), T, E);
  hipLaunchKernelGGL(ep_moments_decode_kernel, dim3(1), dim3(1), 0, stream,
                     out.data_ptr<float>());
  return out;
}
