// Fused MLP update-path kernels (gfx950): forward, per-row backward
// (fused with the analytic PPO loss gradient), and dW accumulation.
//
// These replace the reference's per-update-step graph execution
// (reference Chief.py:64 running K1-K3/K8 via TF's GEMMs — SURVEY.md
// §2.4) and the eager rebuild's rocBLAS path, whose Tensile fp32 kernels
// run the tall-skinny [B, 376] x [376, 64] shapes at <1 TB/s effective.
// One fused forward reads the observation batch ONCE for the whole
// network (obs -> hidden stack -> value + pd heads); the backward is
// three kernel families instead of ~25 launches:
//
//   mlp_fwd:       saves every activation a_l; writes v, pdflat.
//   mlp_bwd_rows:  per sample row, recomputes the PPO loss gradients
//                  (ppo_math.h — no autograd, upstream g = 1) and chains
//                  them through the heads and hidden layers to produce
//                  every pre-activation gradient dz_l, plus g_pd/g_v for
//                  the head weight gradients.  One wave per row: lane u
//                  owns unit u, so H-sized accumulators live one-per-lane
//                  in registers (a per-thread H-array would spill to
//                  scratch — cdna guide §5.4 rule 20).
//   dw_accum:      dW += delta^T @ acts and db += sum(delta), column-
//                  chunked, per-thread register tiles, fp32 atomics into
//                  the flat gradient buffer.
//
// Gradients land directly in the flat grad bucket at the offsets the
// fused Adam and the RCCL all-reduce consume — one update step is ~7
// launches end to end.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ppo_math.h"

namespace {

constexpr int MAX_H = 128;
constexpr int MAX_A = 32;
constexpr int MAX_HIDDEN = 3;
constexpr int ROWS_PER_WAVE = 8;
constexpr int WAVES_PER_BLOCK = 4;
constexpr int ROW_TILE = ROWS_PER_WAVE * WAVES_PER_BLOCK;  // 32
constexpr int WBUF_FLOATS = 24576;  // 96 KiB staged weights cap (bwd)

struct MLPArgs {
  const float* params;
  const float* states;  // [B][D]
  float* acts;          // a_1 | a_2 | ... (each [B][H_l])
  float* v;             // [B]
  float* pdflat;        // [B][2A]
  int off_W[MAX_HIDDEN], off_b[MAX_HIDDEN];
  int dims[MAX_HIDDEN + 1];
  int off_Wv, off_bv, off_Wp, off_bp;
  int n_hidden, act_dim, activation;
  int64_t B;
};

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------

__launch_bounds__(WAVES_PER_BLOCK * 64)
__global__ void mlp_fwd_kernel(MLPArgs a) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  __shared__ float h_lds[2][ROW_TILE][MAX_H];

  const int D = a.dims[0];
  const int A = a.act_dim;
  const int P = 2 * A;
  const int row0 = wave * ROWS_PER_WAVE;

  for (int64_t tile = blockIdx.x; tile * ROW_TILE < a.B; tile += gridDim.x) {
    const int64_t r0 = tile * ROW_TILE + row0;
    const int nR = static_cast<int>(
        min((int64_t)ROWS_PER_WAVE, max((int64_t)0, a.B - r0)));
    if (nR <= 0) continue;

    int in_dim = D;
    int64_t act_base = 0;
    for (int l = 0; l < a.n_hidden; ++l) {
      const int out_dim = a.dims[l + 1];
      const float* W = a.params + a.off_W[l];
      const float* bias = a.params + a.off_b[l];
      const int cur = l & 1;
      const int prev = (l - 1) & 1;
      for (int u = lane; u < out_dim; u += WAVE) {
        float acc[ROWS_PER_WAVE];
        const float bu = bias[u];
        #pragma unroll
        for (int e = 0; e < ROWS_PER_WAVE; ++e) acc[e] = bu;
        const float* Wrow = W + (int64_t)u * in_dim;
        if (l == 0) {
          #pragma unroll 2
          for (int k = 0; k + 4 <= in_dim; k += 4) {
            const float4 w4 = *reinterpret_cast<const float4*>(Wrow + k);
            for (int e = 0; e < nR; ++e) {
              const float4 i4 = *reinterpret_cast<const float4*>(
                  a.states + (r0 + e) * in_dim + k);
              acc[e] += w4.x * i4.x + w4.y * i4.y + w4.z * i4.z + w4.w * i4.w;
            }
          }
          for (int k = in_dim & ~3; k < in_dim; ++k) {
            const float w = Wrow[k];
            for (int e = 0; e < nR; ++e)
              acc[e] += w * a.states[(r0 + e) * in_dim + k];
          }
        } else {
          #pragma unroll 2
          for (int k = 0; k + 4 <= in_dim; k += 4) {
            const float4 w4 = *reinterpret_cast<const float4*>(Wrow + k);
            #pragma unroll
            for (int e = 0; e < ROWS_PER_WAVE; ++e) {
              const float4 i4 = *reinterpret_cast<const float4*>(
                  &h_lds[prev][row0 + e][k]);
              acc[e] += w4.x * i4.x + w4.y * i4.y + w4.z * i4.z + w4.w * i4.w;
            }
          }
        }
        #pragma unroll
        for (int e = 0; e < ROWS_PER_WAVE; ++e) {
          const float val = a.activation ? tanhf(acc[e]) : fmaxf(acc[e], 0.f);
          h_lds[cur][row0 + e][u] = val;
          if (e < nR) a.acts[act_base + (r0 + e) * out_dim + u] = val;
        }
      }
      // waves touch only their own LDS rows; in-wave LDS ordering is
      // enforced by the compiler's lgkmcnt waits — no block barrier
      // (waves run independent row ranges).
      __builtin_amdgcn_wave_barrier();
      in_dim = out_dim;
      act_base += a.B * out_dim;
    }

    const int last = (a.n_hidden - 1) & 1;
    for (int u = lane; u < P + 1; u += WAVE) {
      const bool is_v = (u == P);
      const float* Wrow =
          a.params + (is_v ? (int64_t)a.off_Wv : a.off_Wp + (int64_t)u * in_dim);
      float acc[ROWS_PER_WAVE];
      const float bu = a.params[is_v ? a.off_bv : a.off_bp + u];
      #pragma unroll
      for (int e = 0; e < ROWS_PER_WAVE; ++e) acc[e] = bu;
      for (int k = 0; k + 4 <= in_dim; k += 4) {
        const float4 w4 = *reinterpret_cast<const float4*>(Wrow + k);
        #pragma unroll
        for (int e = 0; e < ROWS_PER_WAVE; ++e) {
          const float4 i4 =
              *reinterpret_cast<const float4*>(&h_lds[last][row0 + e][k]);
          acc[e] += w4.x * i4.x + w4.y * i4.y + w4.z * i4.z + w4.w * i4.w;
        }
      }
      for (int e = 0; e < nR; ++e) {
        if (is_v) a.v[r0 + e] = acc[e];
        else a.pdflat[(r0 + e) * P + u] = acc[e];
      }
    }
    __builtin_amdgcn_wave_barrier();
  }
}

// ---------------------------------------------------------------------------
// Backward rows
// ---------------------------------------------------------------------------

struct BwdArgs {
  const float* params;
  const float* pdflat;
  const float* oldflat;
  const float* v;
  const float* oldv;
  const float* actions;
  const float* adv;
  const float* etr;
  const float* acts;  // a_1|a_2|...
  float* dz;          // dz_1|dz_2|...
  float* g_pd;        // [B][2A]
  float* g_v;         // [B]
  int off_W[MAX_HIDDEN];
  int dims[MAX_HIDDEN + 1];
  int off_Wv, off_Wp;
  int n_hidden, act_dim, activation;
  float clip, entcoeff, vcoeff;
  int64_t B;
};

__launch_bounds__(WAVES_PER_BLOCK * 64)
__global__ void mlp_bwd_rows_kernel(BwdArgs a) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int A = a.act_dim;
  const int P = 2 * A;
  const int HL = a.dims[a.n_hidden];

  // dynamic LDS sized to the ACTUAL staged weights (a fixed 96 KiB
  // static buffer capped occupancy at 1 block/CU and left every L2 load
  // latency exposed)
  extern __shared__ __attribute__((aligned(16))) float dynlds[];
  float* wbuf = dynlds;

  // stage W_2..W_n (hidden-to-hidden), then Wp [P][HL], then Wv [HL]
  int hsz = 0;
  for (int l = 1; l < a.n_hidden; ++l) hsz += a.dims[l + 1] * a.dims[l];
  {
    const int total = hsz + P * HL + HL;
    for (int i = threadIdx.x; i < total; i += blockDim.x) {
      float val;
      if (i < hsz) {
        int rem = i, l = 1;
        for (; l < a.n_hidden; ++l) {
          const int sz = a.dims[l + 1] * a.dims[l];
          if (rem < sz) break;
          rem -= sz;
        }
        val = a.params[a.off_W[l] + rem];
      } else if (i < hsz + P * HL) {
        val = a.params[a.off_Wp + (i - hsz)];
      } else {
        val = a.params[a.off_Wv + (i - hsz - P * HL)];
      }
      wbuf[i] = val;
    }
    __syncthreads();
  }
  const float* Wp_l = &wbuf[hsz];
  const float* Wv_l = &wbuf[hsz + P * HL];
  const int wtotal = ((hsz + P * HL + HL) + 3) & ~3;
  float* scratch = &wbuf[wtotal];  // [WAVES_PER_BLOCK][2*MAX_A + 2*MAX_H]
  float* gpd_s = scratch + wave * (2 * MAX_A + 2 * MAX_H);
  float* dz_s = gpd_s + 2 * MAX_A;  // [MAX_H]
  float* red_s = dz_s + MAX_H;      // [MAX_H]

  int64_t act_base_last = 0;
  for (int l = 0; l < a.n_hidden - 1; ++l) act_base_last += a.B * a.dims[l + 1];

  const int64_t waves_total = (int64_t)gridDim.x * WAVES_PER_BLOCK;
  const int64_t wid = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wave;

  for (int64_t b = wid; b < a.B; b += waves_total) {
    // ---- per-row PPO loss gradients (lanes < 2A cooperate) ----
    float lp_part = 0.f, lo_part = 0.f, ent_part = 0.f;
    float z = 0.f, inv_s = 0.f;
    if (lane < P) {
      const int jj = (lane < A) ? lane : lane - A;
      const float mu = a.pdflat[b * P + jj];
      const float ls = a.pdflat[b * P + A + jj];
      const float aj = a.actions[b * A + jj];
      inv_s = __expf(-ls);
      z = (aj - mu) * inv_s;
      if (lane < A) {
        lp_part = -0.5f * z * z - ls;
        const float mo = a.oldflat[b * P + jj];
        const float lso = a.oldflat[b * P + A + jj];
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

    const float vb = a.v[b], ob = a.oldv[b], ab = a.adv[b], eb = a.etr[b];
    const PPORowGrads g = ppo_row_grads(row, vb, ob, ab, eb, a.B, a.clip,
                                        a.entcoeff, a.vcoeff, 1.f);

    if (lane < P) {
      const float gj = (lane < A)
                           ? g.g_logp * z * inv_s
                           : g.g_logp * (z * z - 1.f) + g.g_ent;
      gpd_s[lane] = gj;
      a.g_pd[b * P + lane] = gj;
    }
    if (lane == 0) a.g_v[b] = g.g_v;
    __builtin_amdgcn_wave_barrier();

    // ---- dz_last = (g_pd @ Wp + g_v * Wv) * act'(a_last) ----
    for (int u = lane; u < HL; u += WAVE) {
      float acc = g.g_v * Wv_l[u];
      #pragma unroll 4
      for (int j = 0; j < P; ++j) acc += gpd_s[j] * Wp_l[j * HL + u];
      const float h = a.acts[act_base_last + b * HL + u];
      const float dact = a.activation ? (1.f - h * h) : (h > 0.f ? 1.f : 0.f);
      const float dzv = acc * dact;
      dz_s[u] = dzv;
      a.dz[act_base_last + b * HL + u] = dzv;
    }
    __builtin_amdgcn_wave_barrier();

    // ---- chain: dz_{l-1} = (dz_l @ W_l) * act'(a_{l-1}) ----
    int64_t act_base = act_base_last;
    int wl_off = hsz;
    for (int l = a.n_hidden - 1; l >= 1; --l) {
      const int out_dim = a.dims[l + 1];
      const int in_dim = a.dims[l];
      wl_off -= out_dim * in_dim;
      const float* Wl = &wbuf[wl_off];
      act_base -= a.B * in_dim;
      for (int u = lane; u < in_dim; u += WAVE) {
        float acc = 0.f;
        #pragma unroll 4
        for (int k = 0; k < out_dim; ++k) acc += dz_s[k] * Wl[k * in_dim + u];
        const float h = a.acts[act_base + b * in_dim + u];
        const float dact = a.activation ? (1.f - h * h) : (h > 0.f ? 1.f : 0.f);
        red_s[u] = acc * dact;
        a.dz[act_base + b * in_dim + u] = red_s[u];
      }
      __builtin_amdgcn_wave_barrier();
      for (int u = lane; u < in_dim; u += WAVE) dz_s[u] = red_s[u];
      __builtin_amdgcn_wave_barrier();
    }
  }
}

// ---------------------------------------------------------------------------
// dW accumulation: dW[out][in] += delta^T @ acts ; db[out] += sum(delta)
// ---------------------------------------------------------------------------

constexpr int DW_BLOCK = 256;
constexpr int DW_ROWS = 8;     // rows staged per iteration
constexpr int DW_ACC = 32;     // register accumulators per thread

__launch_bounds__(DW_BLOCK)
__global__ void dw_accum_kernel(const float* __restrict__ delta,  // [B][out]
                                const float* __restrict__ acts,   // [B][in]
                                float* __restrict__ dW,   // [out][in] (+atomics)
                                float* __restrict__ db,   // [out] or nullptr
                                int64_t B, int out_dim, int in_dim,
                                int col0, int cols, int row_splits) {
  // thread -> (u, igroup): u = tid % out_dim_r, igroup = tid / out_dim_r
  // each thread accumulates up to DW_ACC columns for its u.
  __shared__ float d_lds[DW_ROWS][MAX_H];       // delta rows
  __shared__ float a_lds[DW_ROWS][DW_ACC * 8];  // acts rows (<=256 cols)

  const int split = blockIdx.x % row_splits;
  // rows handled by this block
  const int64_t rows_per = (B + row_splits - 1) / row_splits;
  const int64_t rb0 = split * rows_per;
  const int64_t rb1 = min(B, rb0 + rows_per);

  const int ngroups = DW_BLOCK / out_dim;           // out_dim <= 256, pow2-ish
  const int u = threadIdx.x % out_dim;
  const int grp = threadIdx.x / out_dim;
  const int cols_per_grp = (cols + ngroups - 1) / ngroups;
  const int i_start = grp * cols_per_grp;
  const int n_i = max(0, min(cols_per_grp, cols - i_start));

  float acc[DW_ACC];
  #pragma unroll
  for (int i = 0; i < DW_ACC; ++i) acc[i] = 0.f;
  float bacc = 0.f;

  for (int64_t r = rb0; r < rb1; r += DW_ROWS) {
    const int nr = static_cast<int>(min((int64_t)DW_ROWS, rb1 - r));
    // stage delta rows [nr][out_dim] and acts rows [nr][cols]
    for (int i = threadIdx.x; i < nr * out_dim; i += DW_BLOCK)
      d_lds[i / out_dim][i % out_dim] = delta[(r + i / out_dim) * out_dim + i % out_dim];
    for (int i = threadIdx.x; i < nr * cols; i += DW_BLOCK)
      a_lds[i / cols][i % cols] = acts[(r + i / cols) * in_dim + col0 + i % cols];
    __syncthreads();
    for (int e = 0; e < nr; ++e) {
      const float d = d_lds[e][u];
      if (grp == 0) bacc += d;
      #pragma unroll
      for (int ii = 0; ii < DW_ACC; ++ii) {
        if (ii < n_i) acc[ii] += d * a_lds[e][i_start + ii];
      }
    }
    __syncthreads();
  }
  #pragma unroll
  for (int ii = 0; ii < DW_ACC; ++ii) {
    if (ii < n_i) atomicAdd(&dW[(int64_t)u * in_dim + col0 + i_start + ii], acc[ii]);
  }
  if (db != nullptr && grp == 0) atomicAdd(&db[u], bacc);
}

}  // namespace

// ---------------------------------------------------------------------------
// Bindings
// ---------------------------------------------------------------------------

static void fill_dims(const std::vector<int64_t>& dims, int* out) {
  for (size_t i = 0; i < dims.size(); ++i) out[i] = static_cast<int>(dims[i]);
}

std::vector<torch::Tensor> mlp_fwd(torch::Tensor params,
                                   std::vector<int64_t> offsets,
                                   std::vector<int64_t> dims,
                                   int64_t activation, torch::Tensor states,
                                   int64_t act_dim) {
  const int64_t B = states.size(0);
  const int n_hidden = static_cast<int>(dims.size()) - 1;
  TORCH_CHECK(states.is_cuda() && states.dtype() == torch::kFloat32 &&
              states.is_contiguous());
  TORCH_CHECK(n_hidden >= 1 && n_hidden <= MAX_HIDDEN);
  TORCH_CHECK(act_dim <= MAX_A);
  for (int l = 1; l <= n_hidden; ++l) TORCH_CHECK(dims[l] <= MAX_H);
  TORCH_CHECK(states.size(1) == dims[0]);

  int64_t acts_floats = 0;
  for (int l = 1; l <= n_hidden; ++l) acts_floats += B * dims[l];
  auto acts = torch::empty({acts_floats}, states.options());
  auto v = torch::empty({B}, states.options());
  auto pdflat = torch::empty({B, 2 * act_dim}, states.options());

  MLPArgs a{};
  a.params = params.data_ptr<float>();
  a.states = states.data_ptr<float>();
  a.acts = acts.data_ptr<float>();
  a.v = v.data_ptr<float>();
  a.pdflat = pdflat.data_ptr<float>();
  const int nh = n_hidden;
  for (int l = 0; l < nh; ++l) {
    a.off_W[l] = static_cast<int>(offsets[2 * l]);
    a.off_b[l] = static_cast<int>(offsets[2 * l + 1]);
  }
  fill_dims(dims, a.dims);
  a.off_Wv = static_cast<int>(offsets[2 * nh]);
  a.off_bv = static_cast<int>(offsets[2 * nh + 1]);
  a.off_Wp = static_cast<int>(offsets[2 * nh + 2]);
  a.off_bp = static_cast<int>(offsets[2 * nh + 3]);
  a.n_hidden = nh;
  a.act_dim = static_cast<int>(act_dim);
  a.activation = static_cast<int>(activation);
  a.B = B;

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int64_t tiles = (B + ROW_TILE - 1) / ROW_TILE;
  const int grid = static_cast<int>(std::min<int64_t>(tiles, 8192));
  hipLaunchKernelGGL(mlp_fwd_kernel, dim3(grid), dim3(WAVES_PER_BLOCK * 64), 0,
                     stream, a);
  return {acts, v, pdflat};
}

std::vector<torch::Tensor> mlp_bwd_rows(
    torch::Tensor params, std::vector<int64_t> offsets,
    std::vector<int64_t> dims, int64_t activation, torch::Tensor acts,
    torch::Tensor pdflat, torch::Tensor oldflat, torch::Tensor v,
    torch::Tensor oldv, torch::Tensor actions, torch::Tensor adv,
    torch::Tensor etr, double clip, double entcoeff, double vcoeff) {
  const int64_t B = v.numel();
  const int n_hidden = static_cast<int>(dims.size()) - 1;
  const int A = static_cast<int>(pdflat.size(1) / 2);
  const int P = 2 * A;
  const int HL = static_cast<int>(dims[n_hidden]);
  int hsz = 0;
  for (int l = 1; l < n_hidden; ++l)
    hsz += static_cast<int>(dims[l + 1] * dims[l]);
  TORCH_CHECK(hsz + P * HL + HL <= WBUF_FLOATS,
              "staged weights exceed the bwd LDS budget");

  auto dz = torch::empty_like(acts);
  auto g_pd = torch::empty_like(pdflat);
  auto g_v = torch::empty_like(v);

  BwdArgs a{};
  a.params = params.data_ptr<float>();
  a.pdflat = pdflat.data_ptr<float>();
  a.oldflat = oldflat.data_ptr<float>();
  a.v = v.data_ptr<float>();
  a.oldv = oldv.data_ptr<float>();
  a.actions = actions.data_ptr<float>();
  a.adv = adv.data_ptr<float>();
  a.etr = etr.data_ptr<float>();
  a.acts = acts.data_ptr<float>();
  a.dz = dz.data_ptr<float>();
  a.g_pd = g_pd.data_ptr<float>();
  a.g_v = g_v.data_ptr<float>();
  for (int l = 0; l < n_hidden; ++l)
    a.off_W[l] = static_cast<int>(offsets[2 * l]);
  fill_dims(dims, a.dims);
  a.off_Wv = static_cast<int>(offsets[2 * n_hidden]);
  a.off_Wp = static_cast<int>(offsets[2 * n_hidden + 2]);
  a.n_hidden = n_hidden;
  a.act_dim = A;
  a.activation = static_cast<int>(activation);
  a.clip = static_cast<float>(clip);
  a.entcoeff = static_cast<float>(entcoeff);
  a.vcoeff = static_cast<float>(vcoeff);
  a.B = B;

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int grid = 2048;
  const int wtotal = ((hsz + P * HL + HL) + 3) & ~3;
  const size_t lds_bytes =
      (wtotal + WAVES_PER_BLOCK * (2 * MAX_A + 2 * MAX_H)) * sizeof(float);
  hipLaunchKernelGGL(mlp_bwd_rows_kernel, dim3(grid),
                     dim3(WAVES_PER_BLOCK * 64), lds_bytes, stream, a);
  return {dz, g_pd, g_v};
}

void dw_accum(torch::Tensor delta, torch::Tensor acts, torch::Tensor grad_buf,
              int64_t w_off, int64_t b_off) {
  // delta [B, out], acts [B, in]; accumulates into grad_buf[w_off:...] and
  // (if b_off >= 0) grad_buf[b_off:...]
  const int64_t B = delta.size(0);
  const int out_dim = static_cast<int>(delta.size(1));
  const int in_dim = static_cast<int>(acts.size(1));
  TORCH_CHECK(out_dim <= DW_BLOCK && out_dim <= MAX_H);
  TORCH_CHECK(delta.is_contiguous() && acts.is_contiguous());

  const int ngroups = DW_BLOCK / out_dim;
  const int max_cols =
      std::min(DW_ACC * ngroups, DW_ACC * 8);  // a_lds capacity per row
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  float* dW = grad_buf.data_ptr<float>() + w_off;
  float* db = (b_off >= 0) ? grad_buf.data_ptr<float>() + b_off : nullptr;

  for (int col0 = 0; col0 < in_dim; col0 += max_cols) {
    const int cols = std::min(max_cols, in_dim - col0);
    // enough blocks to fill the chip; each handles a row split
    const int row_splits = 512;
    hipLaunchKernelGGL(dw_accum_kernel, dim3(row_splits), dim3(DW_BLOCK), 0,
                       stream, delta.data_ptr<float>(), acts.data_ptr<float>(),
                       dW, (col0 == 0) ? db : nullptr, B, out_dim, in_dim,
                       col0, cols, row_splits);
  }
}
