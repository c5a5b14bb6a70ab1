// Fused DPPO rollout engine (gfx950): the ENTIRE T-step rollout in one
// kernel launch.
//
// The reference steps one env per sess.run (reference Worker.py:49,146 —
// per-step launch overhead dominates); the first eager
// rebuild still launches ~30 kernels per batched env step.  This kernel
// replaces all of it: per step it runs the policy MLP forward (K1-K3 in
// SURVEY.md §2.4), DiagGaussian sampling (K4), the epsilon-greedy overlay
// (Worker.py:149-152), the synthetic env dynamics + reward + done/reset
// (envs/synthetic.py), and the episode-reward moment bookkeeping
// (Worker.py:57-65) — T times, with env state resident in LDS, so a whole
// rollout costs ONE launch and all policy/env weights stream from L2.
//
// Decomposition: one wave (64 lanes) per block owns ENV_TILE=8 envs for
// the full T-step loop.  No inter-block communication exists (each block's
// envs are private; output buffers are disjoint), so no grid sync is
// needed — the T-loop is safe inside the launch by construction.
//   - Layer phase: lane u computes output unit u for all 8 envs
//     (register-blocked: one W element load feeds 8 FMAs; W rows stream
//     per-lane from L1/L2; inputs broadcast from LDS as float4).
//   - Env phase: lanes split the obs dimension d; low-rank dynamics read
//     transposed Vt/Ut/Bt so every per-lane stream is contiguous.
//   - RNG: counter-based (splitmix64 hash -> Box-Muller), keyed by
//     (seed, env, step, slot): stateless, deterministic per launch,
//     no generator state to carry.
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

struct RolloutArgs {
  // policy (torch Linear layout: W[out][in] row-major)
  const float* W[MAX_HIDDEN];
  const float* b[MAX_HIDDEN];
  int dims[MAX_HIDDEN + 1];  // dims[0]=obs_dim, dims[i]=hidden_i
  int n_hidden;
  const float* Wv;  // [1][H]
  const float* bv;  // [1]
  const float* Wp;  // [2A][H]
  const float* bp;  // [2A]
  int act_dim;
  int activation;  // 0=relu, 1=tanh
  // env (transposed layouts: Vt[r][D], Ut[D][r], Bt[D][A])
  const float* env_d;   // [D]
  const float* env_Vt;  // [r][D]
  const float* env_Ut;  // [D][r]
  const float* env_Bt;  // [D][A]
  const int* horizons;  // [E]
  int rank;
  float noise;
  float act_low, act_high;
  float eps_explore;
  // persistent env state
  float* x;    // [E][D]
  int* t;      // [E]
  float* epr;  // [E]
  // outputs
  float* states;   // [T][E][D]
  float* pdflats;  // [T][E][2A]
  float* actions;  // [T][E][A]
  float* values;   // [T][E]
  float* rewards;  // [T][E]
  float* dones;    // [T][E]
  float* boot_v;   // [E]
  float* ep_moments;  // [5]: count, sum, sumsq, min(neg-encoded), max
  int T, E, D;
  unsigned long long seed;
};

// ---- counter-based RNG ----------------------------------------------------

DEV_INLINE unsigned long long mix64(unsigned long long x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return x;
}

// uniform in (0, 1]
DEV_INLINE float rng_uniform(unsigned long long seed, int env, int step,
                             int slot) {
  unsigned long long k = seed;
  k = mix64(k ^ (static_cast<unsigned long long>(env) << 1 | 1));
  k = mix64(k ^ (static_cast<unsigned long long>(step) << 20) ^
            static_cast<unsigned long long>(slot));
  const unsigned u = static_cast<unsigned>(k >> 40);  // 24 bits
  return (u + 1) * (1.0f / 16777217.0f);
}

DEV_INLINE float rng_normal(unsigned long long seed, int env, int step,
                            int slot) {
  const float u1 = rng_uniform(seed, env, step, 2 * slot + 100000);
  const float u2 = rng_uniform(seed, env, step, 2 * slot + 100001);
  return sqrtf(-2.0f * __logf(u1)) * __cosf(6.2831853071795865f * u2);
}

// monotonic float<->uint encoding for atomic max (works for min via -x)
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

__launch_bounds__(64)
__global__ void rollout_kernel(RolloutArgs a) {
  const int lane = threadIdx.x;  // block = one wave of 64
  const int block = blockIdx.x;
  const int e0 = block * ENV_TILE;  // first global env of this tile
  const int nE = min(ENV_TILE, a.E - e0);
  const int D = a.D;
  const int A = a.act_dim;
  const int P = 2 * A;

  __shared__ float x_lds[ENV_TILE][MAX_D];       // env state
  __shared__ float h0[ENV_TILE][MAX_H];
  __shared__ float h1[ENV_TILE][MAX_H];
  __shared__ float pd_lds[ENV_TILE][2 * MAX_A];  // mean | logstd
  __shared__ float act_lds[ENV_TILE][MAX_A];
  __shared__ float xv_lds[ENV_TILE][MAX_R];
  __shared__ float val_lds[ENV_TILE];
  __shared__ float rsum_lds[ENV_TILE];
  __shared__ float epr_lds[ENV_TILE];
  __shared__ int tc_lds[ENV_TILE];
  __shared__ int done_lds[ENV_TILE];

  // ---- load persistent state ----
  for (int e = 0; e < nE; ++e) {
    for (int d = lane; d < D; d += WAVE) x_lds[e][d] = a.x[(int64_t)(e0 + e) * D + d];
  }
  if (lane < nE) {
    epr_lds[lane] = a.epr[e0 + lane];
    tc_lds[lane] = a.t[e0 + lane];
  }
  // zero the tail envs' state so register-blocked loops over the full
  // ENV_TILE never read uninitialized LDS
  for (int e = nE; e < ENV_TILE; ++e) {
    for (int d = lane; d < D; d += WAVE) x_lds[e][d] = 0.f;
  }
  __syncthreads();

  for (int step = 0; step < a.T; ++step) {
    // ---- write current obs to the rollout buffer ----
    for (int e = 0; e < nE; ++e) {
      const int64_t base = ((int64_t)step * a.E + e0 + e) * D;
      for (int d = lane; d < D; d += WAVE) a.states[base + d] = x_lds[e][d];
    }

    // ---- policy MLP forward: unit-per-lane, envs register-blocked ----
    const float* in = &x_lds[0][0];
    int in_dim = D, in_stride = MAX_D;
    for (int l = 0; l < a.n_hidden; ++l) {
      const int out_dim = a.dims[l + 1];
      const float* W = a.W[l];
      float* out = (l & 1) ? &h1[0][0] : &h0[0][0];
      for (int u = lane; u < out_dim; u += WAVE) {
        float acc[ENV_TILE];
        const float bias = a.b[l][u];
        #pragma unroll
        for (int e = 0; e < ENV_TILE; ++e) acc[e] = bias;
        const float* Wrow = W + (int64_t)u * in_dim;
        int k = 0;
        for (; k + 4 <= in_dim; k += 4) {
          const float4 w4 = *reinterpret_cast<const float4*>(Wrow + k);
          #pragma unroll
          for (int e = 0; e < ENV_TILE; ++e) {
            const float4 i4 = *reinterpret_cast<const float4*>(in + e * in_stride + k);
            acc[e] += w4.x * i4.x + w4.y * i4.y + w4.z * i4.z + w4.w * i4.w;
          }
        }
        for (; k < in_dim; ++k) {
          const float w = Wrow[k];
          #pragma unroll
          for (int e = 0; e < ENV_TILE; ++e) acc[e] += w * in[e * in_stride + k];
        }
        #pragma unroll
        for (int e = 0; e < ENV_TILE; ++e) {
          float v = acc[e];
          v = a.activation ? tanhf(v) : fmaxf(v, 0.f);
          out[e * MAX_H + u] = v;
        }
      }
      __syncthreads();
      in = out;
      in_dim = out_dim;
      in_stride = MAX_H;
    }

    // ---- heads: value (unit 0 handled by lane 0..nE) + pd params ----
    for (int u = lane; u < P + 1; u += WAVE) {
      const bool is_v = (u == P);
      const float* Wrow = (is_v ? a.Wv : a.Wp + (int64_t)u * in_dim);
      const float bias = is_v ? a.bv[0] : a.bp[u];
      float acc[ENV_TILE];
      #pragma unroll
      for (int e = 0; e < ENV_TILE; ++e) acc[e] = bias;
      for (int k = 0; k < in_dim; ++k) {
        const float w = Wrow[k];
        #pragma unroll
        for (int e = 0; e < ENV_TILE; ++e) acc[e] += w * in[e * in_stride + k];
      }
      #pragma unroll
      for (int e = 0; e < ENV_TILE; ++e) {
        if (is_v) val_lds[e] = acc[e];
        else pd_lds[e][u] = acc[e];
      }
    }
    __syncthreads();

    // ---- sample actions: lanes = (env, action-dim) pairs ----
    {
      const int e = lane / MAX_A;     // 2 envs x 32 dims per pass
      const int j = lane % MAX_A;
      for (int ee = e; ee < nE; ee += WAVE / MAX_A) {
        if (j < A) {
          const int ge = e0 + ee;
          const float mean = pd_lds[ee][j];
          const float logstd = pd_lds[ee][A + j];
          float act = mean + __expf(logstd) * rng_normal(a.seed, ge, step, j);
          // epsilon-greedy overlay (Worker.py:149-152): one decision draw
          // per env, uniform replacement per dim
          const float u_dec = rng_uniform(a.seed, ge, step, 90001);
          if (u_dec < a.eps_explore) {
            const float u = rng_uniform(a.seed, ge, step, 90010 + j);
            act = a.act_low + (a.act_high - a.act_low) * u;
          }
          act_lds[ee][j] = act;
        }
      }
    }
    __syncthreads();

    // ---- write pdflat / action / value buffers ----
    for (int e = 0; e < nE; ++e) {
      const int64_t row = (int64_t)step * a.E + e0 + e;
      for (int j = lane; j < P; j += WAVE) a.pdflats[row * P + j] = pd_lds[e][j];
      for (int j = lane; j < A; j += WAVE) a.actions[row * A + j] = act_lds[e][j];
    }
    if (lane < nE) a.values[(int64_t)step * a.E + e0 + lane] = val_lds[lane];

    // ---- env low-rank projection: xv = x @ V  (lanes = (env, r)) ----
    {
      const int per = WAVE / ENV_TILE;  // 8 lanes per env
      const int e = lane / per;
      for (int rr = lane % per; rr < a.rank; rr += per) {
        const float* Vrow = a.env_Vt + (int64_t)rr * D;  // Vt[r][D]
        float acc = 0.f;
        for (int d = 0; d < D; ++d) acc += x_lds[e][d] * Vrow[d];
        if (e < nE) xv_lds[e][rr] = acc;
      }
    }
    __syncthreads();

    // ---- env state update + reward partial sums (lanes split d) ----
    float racc[ENV_TILE];
    #pragma unroll
    for (int e = 0; e < ENV_TILE; ++e) racc[e] = 0.f;
    for (int d = lane; d < D; d += WAVE) {
      const float dd = a.env_d[d];
      const float* Ut_row = a.env_Ut + (int64_t)d * a.rank;  // Ut[D][r]
      const float* Bt_row = a.env_Bt + (int64_t)d * A;       // Bt[D][A]
      for (int e = 0; e < nE; ++e) {
        float low = 0.f;
        for (int rr = 0; rr < a.rank; ++rr) low += xv_lds[e][rr] * Ut_row[rr];
        float ain = 0.f;
        for (int j = 0; j < A; ++j) ain += act_lds[e][j] * Bt_row[j];
        const float n = a.noise * rng_normal(a.seed, e0 + e, step, 1000 + d);
        const float xn = tanhf(x_lds[e][d] * dd + low + ain + n);
        x_lds[e][d] = xn;
        racc[e] += xn * xn;
      }
    }
    // cross-lane reduce rewards per env
    #pragma unroll
    for (int e = 0; e < ENV_TILE; ++e) {
      racc[e] = wave_reduce_sum(racc[e]);
    }
    if (lane == 0) {
      for (int e = 0; e < nE; ++e) rsum_lds[e] = 1.0f - racc[e] / D;
    }
    __syncthreads();

    // ---- reward, done, episode bookkeeping (lane e handles env e) ----
    if (lane < nE) {
      const int e = lane;
      const float r = rsum_lds[e];
      a.rewards[(int64_t)step * a.E + e0 + e] = r;
      epr_lds[e] += r;
      int tc = tc_lds[e] + 1;
      const int done = (tc >= a.horizons[e0 + e]) ? 1 : 0;
      a.dones[(int64_t)step * a.E + e0 + e] = (float)done;
      if (done) {
        epr_lds[e] = 0.f;
        tc = 0;
      }
      tc_lds[e] = tc;
      done_lds[e] = done;
    }
    __syncthreads();

    // ---- reset finished envs (all lanes split d) ----
    for (int e = 0; e < nE; ++e) {
      if (done_lds[e]) {
        for (int d = lane; d < D; d += WAVE) {
          x_lds[e][d] = 0.1f * rng_normal(a.seed, e0 + e, step, 5000 + d);
        }
      }
    }
    __syncthreads();
  }

  // ---- bootstrap value V(x_T): rerun trunk + value head ----
  {
    const float* in = &x_lds[0][0];
    int in_dim = D, in_stride = MAX_D;
    for (int l = 0; l < a.n_hidden; ++l) {
      const int out_dim = a.dims[l + 1];
      const float* W = a.W[l];
      float* out = (l & 1) ? &h1[0][0] : &h0[0][0];
      for (int u = lane; u < out_dim; u += WAVE) {
        float acc[ENV_TILE];
        const float bias = a.b[l][u];
        #pragma unroll
        for (int e = 0; e < ENV_TILE; ++e) acc[e] = bias;
        const float* Wrow = W + (int64_t)u * in_dim;
        for (int k = 0; k < in_dim; ++k) {
          const float w = Wrow[k];
          #pragma unroll
          for (int e = 0; e < ENV_TILE; ++e) acc[e] += w * in[e * in_stride + k];
        }
        #pragma unroll
        for (int e = 0; e < ENV_TILE; ++e) {
          float v = acc[e];
          out[e * MAX_H + u] = a.activation ? tanhf(v) : fmaxf(v, 0.f);
        }
      }
      __syncthreads();
      in = out;
      in_dim = out_dim;
      in_stride = MAX_H;
    }
    if (lane < nE) {
      float acc = a.bv[0];
      for (int k = 0; k < in_dim; ++k) acc += a.Wv[k] * in[lane * in_stride + k];
      a.boot_v[e0 + lane] = acc;
    }
  }

  // ---- persist env state + flush episode moments ----
  for (int e = 0; e < nE; ++e) {
    for (int d = lane; d < D; d += WAVE) a.x[(int64_t)(e0 + e) * D + d] = x_lds[e][d];
  }
  if (lane < nE) {
    a.epr[e0 + lane] = epr_lds[lane];
    a.t[e0 + lane] = tc_lds[lane];
  }
}

}  // namespace

// Episode moments are recomputed from the rewards/dones buffers by one
// cheap per-env kernel (identical math to the in-rollout epr updates);
// min/max go through the monotone-uint atomicMax encoding and a finalize
// kernel decodes them.

namespace {

__global__ void ep_moments_kernel(const float* __restrict__ rewards,
                                  const float* __restrict__ dones,
                                  const float* __restrict__ epr_in,  // [E] pre-rollout
                                  float* __restrict__ out,  // [5]
                                  int64_t T, int64_t E) {
  // one thread per env: walk its column, reconstruct episode sums
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
  // out[3]/out[4] hold monotone-encoded uints (0 = neutral/none).
  const unsigned u3 = __float_as_uint(out[3]);
  const unsigned u4 = __float_as_uint(out[4]);
  out[3] = (u3 == 0u) ? 0.f : -u2f_mono(u3);
  out[4] = (u4 == 0u) ? 0.f : u2f_mono(u4);
}

}  // namespace

std::vector<torch::Tensor> rollout_run(
    std::vector<torch::Tensor> Ws, std::vector<torch::Tensor> bs,
    torch::Tensor Wv, torch::Tensor bv, torch::Tensor Wp, torch::Tensor bp,
    int64_t activation,
    torch::Tensor env_d, torch::Tensor env_Vt, torch::Tensor env_Ut,
    torch::Tensor env_Bt, torch::Tensor horizons, double noise,
    double act_low, double act_high, double eps_explore,
    torch::Tensor x, torch::Tensor t, torch::Tensor epr,
    int64_t T, int64_t seed) {
  const int64_t E = x.size(0);
  const int64_t D = x.size(1);
  const int A = static_cast<int>(Wp.size(0) / 2);
  const int n_hidden = static_cast<int>(Ws.size());
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kFloat32 && x.is_contiguous());
  TORCH_CHECK(n_hidden >= 1 && n_hidden <= MAX_HIDDEN, "1..3 hidden layers");
  TORCH_CHECK(D <= MAX_D && A <= MAX_A, "obs/act dims exceed kernel limits");
  TORCH_CHECK(env_Vt.size(0) <= MAX_R, "env rank exceeds kernel limit");
  TORCH_CHECK(horizons.dtype() == torch::kInt32 && t.dtype() == torch::kInt32);

  RolloutArgs a{};
  for (int l = 0; l < n_hidden; ++l) {
    TORCH_CHECK(Ws[l].is_contiguous() && bs[l].is_contiguous());
    TORCH_CHECK(Ws[l].size(0) <= MAX_H, "hidden width exceeds kernel limit");
    a.W[l] = Ws[l].data_ptr<float>();
    a.b[l] = bs[l].data_ptr<float>();
    a.dims[l] = static_cast<int>(Ws[l].size(1));
    a.dims[l + 1] = static_cast<int>(Ws[l].size(0));
  }
  TORCH_CHECK(a.dims[0] == D, "first layer must consume obs_dim");
  a.n_hidden = n_hidden;
  a.Wv = Wv.data_ptr<float>();
  a.bv = bv.data_ptr<float>();
  a.Wp = Wp.data_ptr<float>();
  a.bp = bp.data_ptr<float>();
  a.act_dim = A;
  a.activation = static_cast<int>(activation);
  a.env_d = env_d.data_ptr<float>();
  a.env_Vt = env_Vt.data_ptr<float>();
  a.env_Ut = env_Ut.data_ptr<float>();
  a.env_Bt = env_Bt.data_ptr<float>();
  a.horizons = horizons.data_ptr<int>();
  a.rank = static_cast<int>(env_Vt.size(0));
  a.noise = static_cast<float>(noise);
  a.act_low = static_cast<float>(act_low);
  a.act_high = static_cast<float>(act_high);
  a.eps_explore = static_cast<float>(eps_explore);
  a.x = x.data_ptr<float>();
  a.t = t.data_ptr<int>();
  a.epr = epr.data_ptr<float>();

  auto opt = x.options();
  auto states = torch::empty({T, E, D}, opt);
  auto pdflats = torch::empty({T, E, 2 * (int64_t)A}, opt);
  auto actions = torch::empty({T, E, (int64_t)A}, opt);
  auto values = torch::empty({T, E}, opt);
  auto rewards = torch::empty({T, E}, opt);
  auto dones = torch::empty({T, E}, opt);
  auto boot_v = torch::empty({E}, opt);
  auto moments = torch::zeros({5}, opt);
  auto epr_in = epr.clone();

  a.states = states.data_ptr<float>();
  a.pdflats = pdflats.data_ptr<float>();
  a.actions = actions.data_ptr<float>();
  a.values = values.data_ptr<float>();
  a.rewards = rewards.data_ptr<float>();
  a.dones = dones.data_ptr<float>();
  a.boot_v = boot_v.data_ptr<float>();
  a.ep_moments = moments.data_ptr<float>();
  a.T = static_cast<int>(T);
  a.E = static_cast<int>(E);
  a.D = static_cast<int>(D);
  a.seed = static_cast<unsigned long long>(seed);

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = static_cast<int>((E + ENV_TILE - 1) / ENV_TILE);
  hipLaunchKernelGGL(rollout_kernel, dim3(grid), dim3(WAVE), 0, stream, a);

  // moments[3]/moments[4] accumulate in the monotone-uint encoding where
  // raw 0 bits are the neutral "less than any encoded float" element, so
  // the zeros() init above is correct; the decode kernel converts back.
  hipLaunchKernelGGL(ep_moments_kernel,
                     dim3(static_cast<int>((E + 255) / 256)), dim3(256), 0,
                     stream, rewards.data_ptr<float>(), dones.data_ptr<float>(),
                     epr_in.data_ptr<float>(), moments.data_ptr<float>(),
                     T, E);
  hipLaunchKernelGGL(ep_moments_decode_kernel, dim3(1), dim3(1), 0, stream,
                     moments.data_ptr<float>());
  return {states, pdflats, actions, values, rewards, dones, boot_v, moments};
}
