// MFMA f32 GEMM kernels for the MLP update path (gfx950).
//
// gfx950 has exact fp32-input MFMA (v_mfma_f32_32x32x2_f32: bitwise a
// k-ordered fmaf chain, 157 TF chip peak = the f32 vector peak — cdna
// guide §3 "FP32-input MFMA") — ~2.4x a VALU f32 GEMM at identical
// numerics, and an order of magnitude over rocBLAS/Tensile's fp32
// tall-skinny picks on these shapes (<1 TB/s effective measured).
//
//   gemm_fwd:   C[B,N] = act(X[B,K] @ Wt[K,N] + bias) — one hidden layer,
//               fused activation (tanh/relu), activations ARE the saved
//               forward outputs.  X tiles are LDS-staged (the MFMA A
//               fragment is a column read, lane = row); Wt streams from
//               L2 coalesced (reused by every row block).
//   heads mode: same, N = 2A+1, no activation; epilogue splits columns
//               into pdflat[B,2A] and v[B].
//   dgrad modes (activation 3/4): C = act'(aux) * (X @ Wt) — the
//               backward chain reuses torch weight layouts directly.
//   dw_mfma:    dW[out,in] += delta^T @ acts, split-K over row blocks
//               into per-split slabs (atomics measured 8.4M adds/launch
//               and dominated), reduced by dw_reduce/db_reduce; fused
//               db += sum(delta); optional row split routes combined
//               [g_pd | g_v] head deltas to Wp/bp and Wv/bv.
//
// Tile-size experiments are in the commit history with measurements:
// MT=2 row fragments and NT>2 column tiles LOSE on these shapes (64-128
// AGPR accumulators drop waves/SIMD; occupancy beats per-wave tile size
// every time it was tried), as did T14 register staging and dual-buffer
// staging — BK=16 single-buffer with 5-6 blocks/CU is the measured
// optimum.
//
// Fragment layout (cdna guide §3, v_mfma_f32_32x32x2_f32):
//   lane l: A[i = l&31][k = l>>5], B[k = l>>5][j = l&31]
//   C/D reg r (of 16): row = (r&3) + 8*(r>>2) + 4*(l>>5), col = l&31.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

using f32x16 = __attribute__((ext_vector_type(16))) float;

constexpr int BK = 16;        // K-step per stage
constexpr int M_WAVE = 32;    // rows per wave tile
constexpr int FWD_WAVES = 4;  // waves per block (each owns 32 rows)
constexpr int FWD_M = FWD_WAVES * M_WAVE;  // 128 rows per block
constexpr int MAX_NT = 4;     // 32-col accumulator tiles per wave

DEV_INLINE int cd_row(int reg, int lane) {
  return (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
}

// ---------------------------------------------------------------------------
// Forward layer: C = act(X @ Wt + b).  heads_mode splits the epilogue.
// ---------------------------------------------------------------------------

struct FwdArgs {
  const float* X;     // [B][K]
  const float* Wt;    // [K][N] (wt_layout 0) or [N][K] torch layout (1)
  const float* bias;  // [N]
  const float* aux;   // [B][N] forward activations (dgrad modes)
  float* C;           // [B][N]      (heads: pdflat [B][N-1])
  float* v;           // [B] heads only
  int64_t B;
  int K, N;
  int activation;     // 0 relu, 1 tanh, 2 none;
                      // 3 tanh-grad: C = (1-aux^2)*acc (no bias)
                      // 4 relu-grad: C = (aux>0)*acc   (no bias)
  int heads;          // if 1: last column -> v, rest -> C (pdflat)
  int wt_layout;      // 0: Wt[K][N]; 1: torch W[N][K] (staged transposed)
  int ldc;            // C row stride (>= N; heads: >= N-1) — lets C land
                      // in a column block of a wider buffer
  int ablate;         // perf diagnosis only (wrong results when nonzero):
                      // 1 skip X stage, 2 skip W stage, 4 skip MFMA,
                      // 8 skip epilogue stores
  // activation 5 (fused env step, v3 rollout): the G = [XV|a] @ [U;B]
  // GEMM's epilogue computes the env transition directly —
  // pred = tanh(x*d + acc + sigma*noise), out = done ? fresh : pred —
  // writing the new env state and the next recorded state, and
  // accumulating per-env pred^2 partials for the reward.  Kills the
  // 2x[E][D] G round trip and most of env_finish_kernel (round-2 lever
  // #3, profiles/r01_final_v3_69M.txt).  RNG slots == env_finish's.
  const float* env_xin;     // [E][D] state at `step`
  float* env_x;             // [E][D] out: state at `step`+1 (the v3 loop
                            // passes the states-blob slot directly; may
                            // alias env_xin)
  const float* envd;        // [D]
  const int* horizons;      // [E]
  const int* tcount;        // [E] (read; env_finish2 increments)
  const long long* seed_dev;
  float* env_rsum;          // [npanels][E] per-panel pred^2 partials
                            // (plain stores — deterministic, no zeroing)
  float sigma;
  int step;
};

// counter-based RNG — formulas identical to rollout.hip's env path
DEV_INLINE unsigned g_lowbias32(unsigned x) {
  x ^= x >> 16;
  x *= 0x7feb352dU;
  x ^= x >> 15;
  x *= 0x846ca68bU;
  x ^= x >> 16;
  return x;
}
DEV_INLINE float2 g_rng_normal2(unsigned seed, int env, int step, int slot) {
  const unsigned h = g_lowbias32(seed ^ (unsigned)env * 0x9E3779B9U ^
                                 (unsigned)step * 0x85EBCA6BU ^
                                 (unsigned)slot * 0xC2B2AE35U);
  const float u1 = ((h >> 16) + 1) * (1.0f / 65537.0f);
  const float u2 = (h & 0xFFFFu) * (1.0f / 65536.0f);
  const float r = sqrtf(-2.0f * __logf(u1));
  float sn, cs;
  __sincosf(6.2831853071795865f * u2, &sn, &cs);
  return make_float2(r * cs, r * sn);
}
DEV_INLINE float g_rng_normal(unsigned seed, int env, int step, int slot) {
  return g_rng_normal2(seed, env, step, slot).x;
}

template <int NT>
__launch_bounds__(FWD_WAVES * 64)
__global__ void gemm_fwd_kernel(FwdArgs a) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;

  // X tile [FWD_M][BK] (+1 pad: the A-fragment read is a column read);
  // Wt tile [BK][NT*32] staged coalesced so the MFMA B-operand comes
  // from LDS instead of a fresh L2 round trip per k-step.
  __shared__ float xs[FWD_M][BK + 1];
  __shared__ float ws[BK][NT * M_WAVE];

  const int i_l = lane & 31;   // A row within wave tile
  const int k_l = lane >> 5;   // A k within pair
  const int NW = NT * M_WAVE;

  for (int64_t tile = blockIdx.x; tile * FWD_M < a.B; tile += gridDim.x) {
    const int64_t b0 = tile * FWD_M;
    f32x16 acc[NT];
    #pragma unroll
    for (int t = 0; t < NT; ++t)
      #pragma unroll
      for (int r = 0; r < 16; ++r) acc[t][r] = 0.f;

    for (int kb = 0; kb < a.K; kb += BK) {
      __syncthreads();
      // float4 staging (scalar element staging measured as the dominant
      // per-call cost: 32 scalar dword loads per thread per stage)
      constexpr int BK4 = BK / 4;
      if (!(a.ablate & 1))
      for (int idx = threadIdx.x; idx < FWD_M * BK4; idx += FWD_WAVES * 64) {
        const int r = idx / BK4, c4 = (idx % BK4) * 4;
        const int64_t row = b0 + r;
        const int col = kb + c4;
        float4 val = make_float4(0.f, 0.f, 0.f, 0.f);
        if (row < a.B) {
          if (col + 3 < a.K) {
            val = *reinterpret_cast<const float4*>(&a.X[row * a.K + col]);
          } else {
            float tmp[4] = {0.f, 0.f, 0.f, 0.f};
            for (int q = 0; q < 4; ++q)
              if (col + q < a.K) tmp[q] = a.X[row * a.K + col + q];
            val = make_float4(tmp[0], tmp[1], tmp[2], tmp[3]);
          }
        }
        xs[r][c4] = val.x;
        xs[r][c4 + 1] = val.y;
        xs[r][c4 + 2] = val.z;
        xs[r][c4 + 3] = val.w;
      }
      if (a.ablate & 2) {
        // skip W stage
      } else if (a.wt_layout == 0) {
        const int NW4 = NW / 4;
        for (int idx = threadIdx.x; idx < BK * NW4; idx += FWD_WAVES * 64) {
          const int r = idx / NW4, c4 = (idx % NW4) * 4;
          const int krow = kb + r;
          float4 val = make_float4(0.f, 0.f, 0.f, 0.f);
          if (krow < a.K) {
            if (c4 + 3 < a.N) {
              val = *reinterpret_cast<const float4*>(
                  &a.Wt[(int64_t)krow * a.N + c4]);
            } else {
              float tmp[4] = {0.f, 0.f, 0.f, 0.f};
              for (int q = 0; q < 4; ++q)
                if (c4 + q < a.N) tmp[q] = a.Wt[(int64_t)krow * a.N + c4 + q];
              val = make_float4(tmp[0], tmp[1], tmp[2], tmp[3]);
            }
          }
          *reinterpret_cast<float4*>(&ws[r][c4]) = val;
        }
      } else {
        // torch W[N][K]: c-major mapping keeps the row-segment reads
        // coalesced and transposes into ws on the fly — no host-side
        // W.t().contiguous() per update step
        constexpr int BK4 = BK / 4;
        for (int idx = threadIdx.x; idx < NW * BK4; idx += FWD_WAVES * 64) {
          const int c = idx / BK4, r4 = (idx % BK4) * 4;
          const int krow = kb + r4;
          float4 val = make_float4(0.f, 0.f, 0.f, 0.f);
          if (c < a.N) {
            if (krow + 3 < a.K) {
              val = *reinterpret_cast<const float4*>(
                  &a.Wt[(int64_t)c * a.K + krow]);
            } else {
              float tmp[4] = {0.f, 0.f, 0.f, 0.f};
              for (int q = 0; q < 4; ++q)
                if (krow + q < a.K) tmp[q] = a.Wt[(int64_t)c * a.K + krow + q];
              val = make_float4(tmp[0], tmp[1], tmp[2], tmp[3]);
            }
          }
          ws[r4][c] = val.x;
          ws[r4 + 1][c] = val.y;
          ws[r4 + 2][c] = val.z;
          ws[r4 + 3][c] = val.w;
        }
      }
      __syncthreads();

      const int ksteps = (a.ablate & 4) ? 0 : min(BK, a.K - kb);
      #pragma unroll 4
      for (int k2 = 0; k2 < ksteps; k2 += 2) {
        const float av = xs[wave * M_WAVE + i_l][k2 + k_l];
        #pragma unroll
        for (int t = 0; t < NT; ++t) {
          const float bv = ws[k2 + k_l][t * M_WAVE + i_l];
          acc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(av, bv, acc[t], 0, 0, 0);
        }
      }
    }

    // ---- epilogue: bias + activation + store ----
    #pragma unroll
    for (int t = 0; t < NT; ++t) {
      const int col = t * M_WAVE + i_l;
      if (col < a.N && !(a.ablate & 8)) {
        const float bv = (a.activation >= 3) ? 0.f : a.bias[col];
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int64_t row = b0 + wave * M_WAVE + cd_row(r, lane);
          if (row < a.B) {
            float x;
            if (a.activation >= 3) {
              const float h = a.aux[row * a.N + col];
              x = acc[t][r] * ((a.activation == 3) ? (1.f - h * h)
                                                   : (h > 0.f ? 1.f : 0.f));
            } else {
              x = acc[t][r] + bv;
              if (a.activation == 0) x = fmaxf(x, 0.f);
              else if (a.activation == 1) x = fast_tanhf(x);
            }
            if (a.heads == 2) {
              // padded heads: N includes pad columns so N%4==0 shapes
              // can take the glds kernel; v sits at N-2, cols >= N-1
              // are the zero-weight pad (discarded)
              if (col == a.N - 2) a.v[row] = x;
              else if (col < a.N - 2) a.C[row * a.ldc + col] = x;
            } else if (a.heads) {
              if (col == a.N - 1) a.v[row] = x;
              else a.C[row * a.ldc + col] = x;
            } else {
              a.C[row * a.ldc + col] = x;
            }
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Pipelined forward (T14 register staging, cdna guide "write tile t+1 AFTER
// the barrier"): the runtime-ablation split measured the plain kernel's
// stage / MFMA / epilogue phases strictly ADDITIVE (L1 886 us = 420 + 321
// + 145; profiles/r01_gemm_ablation.txt) — no cross-phase overlap at 4
// waves/SIMD.  Here each stage issues the NEXT tile's global loads into
// registers before the barrier (plain loads survive s_barrier; guide §5),
// so HBM latency hides under the current stage's MFMAs.
//
// Loads are BRANCHLESS (clamped addresses, whole-float4 validity, zero-fix
// by select at LDS-write time — NO global access in the write phase): a
// per-element guard branch or a write-time refetch makes hipcc emit
// load;s_waitcnt vmcnt(0);ds_write per ELEMENT and serializes the whole
// pipeline (first attempt measured 1.7x SLOWER than the plain kernel).
// Whole-float4 validity requires K%4==0 (and N%4==0 for wt_layout 0);
// other shapes take the plain kernel (host dispatch).
// ---------------------------------------------------------------------------

// NT<=2 (every shape this model family hits): force the allocation to 128
// VGPRs = 4 waves/SIMD — the un-hinted build lands at 132 and loses a
// whole wave to 4 registers.  NT>=3 would spill catastrophically; leave it.
template <int NT, int PBK>
__launch_bounds__(FWD_WAVES * 64, (NT <= 2 && PBK == 16) ? 4 : 1)
__global__ void gemm_fwd_pipe_kernel(FwdArgs a) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;

  // PBK: K-step per stage.  32 gives the prefetch twice the MFMA cycles
  // to hide under but costs 177 VGPR (2 waves/SIMD); 16 keeps 4
  // waves/SIMD.  Both instantiated; dispatch measures/chooses per shape.
  // X rows padded +4 (not +1): a 17-float row stride leaves the float4
  // staging writes 16B-MISALIGNED, so hipcc splits them into scalar
  // ds_writes (PMC: 5.7e8 bank-conflict cycles on 3.2e8 LDS ops).  +4
  // keeps b128 writes; the A-fragment column reads pick up a 4-way bank
  // conflict but there are only 8 of them per wave-stage.
  __shared__ float xs[FWD_M][PBK + 4];
  __shared__ float ws[PBK][NT * M_WAVE];

  const int i_l = lane & 31;
  const int k_l = lane >> 5;
  constexpr int NW = NT * M_WAVE;
  constexpr int PBK4 = PBK / 4;
  constexpr int XIT = (FWD_M * PBK4 + FWD_WAVES * 64 - 1) / (FWD_WAVES * 64);
  constexpr int WIT = (PBK * NW / 4 + FWD_WAVES * 64 - 1) / (FWD_WAVES * 64);
  constexpr int W_ITEMS = PBK * NW / 4;  // float4 slots in the W tile
  // column panel (grid.y): this block computes columns [n0, n0+NW) of N —
  // wide-N GEMMs (the env-dynamics [E][36] @ [36][D] call) run as panels
  const int n0 = blockIdx.y * NW;

  float4 xreg[XIT];
  float4 wreg[WIT];

  // ---- branchless load issue for one k-stage (clamped addresses) ----
  auto issue_x = [&](int64_t b0, int kb) {
    #pragma unroll
    for (int it = 0; it < XIT; ++it) {
      const int idx = threadIdx.x + it * FWD_WAVES * 64;
      const int r = idx / PBK4, c4 = (idx % PBK4) * 4;
      int64_t row = b0 + r;
      row = row < a.B ? row : a.B - 1;
      int col = kb + c4;
      col = col < a.K ? col : a.K - 4;   // K%4==0: col<K => col+3<K
      xreg[it] = *reinterpret_cast<const float4*>(&a.X[row * a.K + col]);
    }
  };
  auto issue_w = [&](int kb) {
    if (a.wt_layout == 0) {
      constexpr int NW4 = NW / 4;
      #pragma unroll
      for (int it = 0; it < WIT; ++it) {
        const int idx = threadIdx.x + it * FWD_WAVES * 64;
        if (idx >= W_ITEMS) break;
        const int r = idx / NW4, c4 = (idx % NW4) * 4;
        int krow = kb + r;
        krow = krow < a.K ? krow : a.K - 1;
        const int c = n0 + c4 < a.N ? n0 + c4 : a.N - 4;  // N%4==0
        wreg[it] = *reinterpret_cast<const float4*>(&a.Wt[(int64_t)krow * a.N + c]);
      }
    } else {
      #pragma unroll
      for (int it = 0; it < WIT; ++it) {
        const int idx = threadIdx.x + it * FWD_WAVES * 64;
        if (idx >= W_ITEMS) break;
        const int c0 = idx / PBK4, r4 = (idx % PBK4) * 4;
        const int c = n0 + c0 < a.N ? n0 + c0 : a.N - 1;
        int krow = kb + r4;
        krow = krow < a.K ? krow : a.K - 4;  // K%4==0
        wreg[it] = *reinterpret_cast<const float4*>(&a.Wt[(int64_t)c * a.K + krow]);
      }
    }
  };
  // ---- LDS write of the staged registers (selects only, no loads) ----
  auto write_x = [&](int64_t b0, int kb) {
    #pragma unroll
    for (int it = 0; it < XIT; ++it) {
      const int idx = threadIdx.x + it * FWD_WAVES * 64;
      const int r = idx / PBK4, c4 = (idx % PBK4) * 4;
      const bool ok = (b0 + r < a.B) & (kb + c4 < a.K);
      const float* v = reinterpret_cast<const float*>(&xreg[it]);
      #pragma unroll
      for (int q = 0; q < 4; ++q) xs[r][c4 + q] = ok ? v[q] : 0.f;
    }
  };
  auto write_w = [&](int kb) {
    if (a.wt_layout == 0) {
      constexpr int NW4 = NW / 4;
      #pragma unroll
      for (int it = 0; it < WIT; ++it) {
        const int idx = threadIdx.x + it * FWD_WAVES * 64;
        if (idx >= W_ITEMS) break;
        const int r = idx / NW4, c4 = (idx % NW4) * 4;
        const bool ok = (kb + r < a.K) & (n0 + c4 < a.N);
        const float* v = reinterpret_cast<const float*>(&wreg[it]);
        #pragma unroll
        for (int q = 0; q < 4; ++q) ws[r][c4 + q] = ok ? v[q] : 0.f;
      }
    } else {
      #pragma unroll
      for (int it = 0; it < WIT; ++it) {
        const int idx = threadIdx.x + it * FWD_WAVES * 64;
        if (idx >= W_ITEMS) break;
        const int c0 = idx / PBK4, r4 = (idx % PBK4) * 4;  // c0 < NW always
        const bool cok = n0 + c0 < a.N;
        const bool kok = kb + r4 < a.K;  // K%4==0: whole-float4 validity
        const float* v = reinterpret_cast<const float*>(&wreg[it]);
        #pragma unroll
        for (int q = 0; q < 4; ++q)
          ws[r4 + q][c0] = (cok & kok) ? v[q] : 0.f;
      }
    }
  };

  // One TILE per block (host launches grid == tile count): tile-boundary
  // cold starts are hidden by BLOCK scheduling — a freshly placed block's
  // prologue loads overlap the other resident blocks' MFMA phases — which
  // costs zero registers, where an explicit cross-tile prefetch kept
  // xreg/wreg live across the epilogue and measured +29 VGPR (137: 3
  // waves/SIMD) and gave the occupancy back.
  for (int64_t tile = blockIdx.x; tile * FWD_M < a.B; tile += gridDim.x) {
    const int64_t b0 = tile * FWD_M;
    f32x16 acc[NT];
    #pragma unroll
    for (int t = 0; t < NT; ++t)
      #pragma unroll
      for (int r = 0; r < 16; ++r) acc[t][r] = 0.f;

    issue_x(b0, 0);
    issue_w(0);
    for (int kb = 0; kb < a.K; kb += PBK) {
      write_x(b0, kb);
      write_w(kb);
      if (kb + PBK < a.K) {  // prefetch the NEXT stage before the barrier
        issue_x(b0, kb + PBK);
        issue_w(kb + PBK);
      }
      __syncthreads();

      const int ksteps = min(PBK, a.K - kb);
      #pragma unroll 4
      for (int k2 = 0; k2 < ksteps; k2 += 2) {
        const float av = xs[wave * M_WAVE + i_l][k2 + k_l];
        #pragma unroll
        for (int t = 0; t < NT; ++t) {
          const float bv = ws[k2 + k_l][t * M_WAVE + i_l];
          acc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(av, bv, acc[t], 0, 0, 0);
        }
      }
      __syncthreads();
    }

    #pragma unroll
    for (int t = 0; t < NT; ++t) {
      const int col = n0 + t * M_WAVE + i_l;
      if (col < a.N) {
        const float bv = (a.activation >= 3) ? 0.f : a.bias[col];
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int64_t row = b0 + wave * M_WAVE + cd_row(r, lane);
          if (row < a.B) {
            float x;
            if (a.activation >= 3) {
              const float h = a.aux[row * a.N + col];
              x = acc[t][r] * ((a.activation == 3) ? (1.f - h * h)
                                                   : (h > 0.f ? 1.f : 0.f));
            } else {
              x = acc[t][r] + bv;
              if (a.activation == 0) x = fmaxf(x, 0.f);
              else if (a.activation == 1) x = fast_tanhf(x);
            }
            if (a.heads == 2) {
              // padded heads: N includes pad columns so N%4==0 shapes
              // can take the glds kernel; v sits at N-2, cols >= N-1
              // are the zero-weight pad (discarded)
              if (col == a.N - 2) a.v[row] = x;
              else if (col < a.N - 2) a.C[row * a.ldc + col] = x;
            } else if (a.heads) {
              if (col == a.N - 1) a.v[row] = x;
              else a.C[row * a.ldc + col] = x;
            } else {
              a.C[row * a.ldc + col] = x;
            }
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// glds 3-buffer streaming forward (round-2 lever #1, profiles/
// r01_final_v3_69M.txt): the register-staged pipe kernel measured 2.5
// TB/s effective on the tall-skinny update GEMMs vs ~6 TB/s streaming
// capability — its per-stage issue->vmcnt(0)-at-write->ds_write chain
// plus two __syncthreads serialize the stream.  Here the X/W tiles go
// straight to LDS by global_load_lds into a 3-deep ring with COUNTED
// vmcnt across raw barriers (guide §5 "3-buf span +83%"), so two stages
// stay in flight while one computes and the wave stream carries only
// MFMAs + fragment reads.
//
// Shape contract: NT==2 (one W piece per thread keeps the vmcnt count
// wave-uniform), wt_layout 0.  K tails (K % 16 != 0) are handled by
// clamped sources + zeroing the ws rows >= K before the last stage's
// MFMAs (garbage X columns then multiply zero W rows).  The xs image is
// UNPADDED [128][16] (glds needs a lane-linear dest); the 4-way
// A-column bank spread that the pipe kernel's +4 row pad bought is kept
// by XOR-swizzling the float4 piece index with row bits on the SOURCE
// address (guide rule 21) — same 64 B cacheline, zero coalescing cost.
// ---------------------------------------------------------------------------

template <int RING>  // ring depth: 2 (24 KB, 6 blocks/CU — overlap via
                     // MORE co-resident blocks), 3 (36 KB, 4 blocks/CU,
                     // default), 4 (48 KB, 3 blocks/CU — deeper
                     // in-flight staging; measured slower)
__launch_bounds__(FWD_WAVES * 64, RING == 2 ? 6 : RING == 3 ? 4 : 3)
__global__ void gemm_fwd_glds_kernel(FwdArgs a) {
  constexpr int NT = 2;
  constexpr int PBK = 16;
  constexpr int NW = NT * M_WAVE;          // 64
  constexpr int XB = FWD_M * PBK * 4;      // 8192 B
  constexpr int WB = PBK * NW * 4;         // 4096 B
  constexpr int SLOT = XB + WB;
  __shared__ __attribute__((aligned(16))) char smem[RING * SLOT];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid / WAVE;
  const int i_l = lane & 31;
  const int k_l = lane >> 5;
  const int n0 = blockIdx.y * NW;

  const int64_t b0 = (int64_t)blockIdx.x * FWD_M;
  if (b0 >= a.B) return;

  // per-thread glds sources (constant; advance by kb each stage)
  // X: two 16-B pieces per thread; dest piece q = tid + p*256 ->
  // row = q/4, c4 = (q%4) with the source float4 swizzled by row bits
  const float* srcX[2];
  int xcol[2];
  int64_t xrow[2];
  #pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int q = tid + p * FWD_WAVES * 64;
    const int r = q >> 2;
    const int c4 = (q & 3) ^ (r & 3);  // source-side st swizzle
    int64_t row = b0 + r;
    row = row < a.B ? row : a.B - 1;
    xrow[p] = row;
    xcol[p] = c4 * 4;
    srcX[p] = a.X + row * a.K;
  }
  // W: one 16-B piece per thread; dest q = tid -> k = q/(NW/4), c4 = q%(NW/4)
  const int wk = tid / (NW / 4);
  int wc4 = (tid % (NW / 4)) * 4;
  {
    int c = n0 + wc4;
    c = c + 3 < a.N ? c : a.N - 4;  // clamped (garbage cols discarded)
    wc4 = c;
  }

  // per-block K-phase rotation: co-resident blocks otherwise hit their
  // stage barriers in lockstep (PMC: MFMA pipe 51% busy, 34% of wave
  // time parked on vmcnt/barrier) — rotating the stage order by
  // blockIdx desynchronizes the convoy at unchanged traffic.  fp32
  // accumulation order changes per block (still deterministic per
  // shape; numerics tests compare against eager to tolerance).
  const int S_ = (a.K + PBK - 1) / PBK;
  const int phase = (int)(blockIdx.x % (unsigned)S_);
  auto smap = [&](int s) {
    const int t = s + phase;
    return t >= S_ ? t - S_ : t;
  };
  auto issue = [&](int s) {
    const int kb = smap(s) * PBK;
    char* slot = smem + (s % RING) * SLOT;
    #pragma unroll
    for (int p = 0; p < 2; ++p) {
      int col = kb + xcol[p];
      col = col + 3 < a.K ? col : a.K - 4;  // clamp; tail fixed via ws zeros
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned*)(srcX[p] + col),
          (unsigned*)(slot + p * 4096 + wave * 1024), 16, 0, 0);
    }
    int krow = kb + wk;
    krow = krow < a.K ? krow : a.K - 1;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned*)(
            a.Wt + (int64_t)krow * a.N + wc4),
        (unsigned*)(slot + XB + wave * 1024), 16, 0, 0);
  };

  const int S = (a.K + PBK - 1) / PBK;
  #pragma unroll
  for (int p = 0; p < RING - 1; ++p)
    if (p < S) issue(p);

  f32x16 acc[NT];
  #pragma unroll
  for (int t = 0; t < NT; ++t)
    #pragma unroll
    for (int r = 0; r < 16; ++r) acc[t][r] = 0.f;

  for (int s = 0; s < S; ++s) {
    // stage s landed when only the later in-flight stages' glds (3 each)
    // remain outstanding: rem = how many stages beyond s are in flight
    const int rem = min(S - 1 - s, RING - 2);
    if (rem >= RING - 2)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(3 * (RING - 2)) : "memory");
    else if (RING == 4 && rem == 1)
      asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    char* slot = smem + (s % RING) * SLOT;
    float* xs = (float*)slot;
    float* ws = (float*)(slot + XB);
    const int kb = smap(s) * PBK;
    if (kb + PBK > a.K) {
      // zero the W rows past K so the clamped/garbage X columns of the
      // tail stage contribute nothing
      for (int idx = tid; idx < (kb + PBK - a.K) * NW; idx += FWD_WAVES * 64)
        ws[(a.K - kb) * NW + idx] = 0.f;
      __builtin_amdgcn_s_barrier();
    }
    // buf (s+RING-1)%RING == (s-1)%RING, freed by the barrier above
    if (s + RING - 1 < S) issue(s + RING - 1);
    #pragma unroll 4
    for (int k2 = 0; k2 < PBK; k2 += 2) {
      const int k = k2 + k_l;
      const int row = wave * M_WAVE + i_l;
      const float av = xs[row * PBK + (((k >> 2) ^ (row & 3)) << 2) + (k & 3)];
      #pragma unroll
      for (int t = 0; t < NT; ++t) {
        const float bv = ws[k * NW + t * M_WAVE + i_l];
        acc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(av, bv, acc[t], 0, 0, 0);
      }
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  if (a.activation == 5) {
    // fused env step: pred = tanh(x*d + acc + sigma*noise);
    // out = done ? fresh-seed : pred; reward partials via one atomic per
    // (row, 32-lane half).  RNG slots match env_finish_kernel exactly.
    const unsigned seed = (unsigned)(*a.seed_dev);
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int64_t row = b0 + wave * M_WAVE + cd_row(r, lane);
      const bool rok = row < a.B;
      const int done =
          rok ? (a.tcount[row] + 1 >= a.horizons[row] ? 1 : 0) : 0;
      float part = 0.f;
      #pragma unroll
      for (int t = 0; t < NT; ++t) {
        const int col = n0 + t * M_WAVE + i_l;
        if (rok && col < a.N) {
          const float dv = a.envd[col];
          const float xv = a.env_xin[row * a.N + col];
          float nz = 0.f;
          if (a.sigma != 0.f) {
            const float2 pr =
                g_rng_normal2(seed, (int)row, a.step, 1000 + (col >> 1));
            nz = (col & 1) ? pr.y : pr.x;
          }
          const float pred = fast_tanhf(xv * dv + acc[t][r] + a.sigma * nz);
          part += pred * pred;
          const float out =
              done ? 0.1f * g_rng_normal(seed, (int)row, a.step, 5000 + col)
                   : pred;
          a.env_x[row * a.N + col] = out;
        }
      }
      #pragma unroll
      for (int off = 16; off > 0; off >>= 1)
        part += __shfl_down(part, off, 32);
      // exactly one (wave, r, half) owns each (panel, row): plain store
      // keeps the reduction deterministic (atomicAdd order is not)
      if (i_l == 0 && rok)
        a.env_rsum[(int64_t)blockIdx.y * a.B + row] = part;
    }
    return;
  }

  #pragma unroll
  for (int t = 0; t < NT; ++t) {
    const int col = n0 + t * M_WAVE + i_l;
    if (col < a.N) {
      const float bv = (a.activation >= 3) ? 0.f : a.bias[col];
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int64_t row = b0 + wave * M_WAVE + cd_row(r, lane);
        if (row < a.B) {
          float x;
          if (a.activation >= 3) {
            const float h = a.aux[row * a.N + col];
            x = acc[t][r] * ((a.activation == 3) ? (1.f - h * h)
                                                 : (h > 0.f ? 1.f : 0.f));
          } else {
            x = acc[t][r] + bv;
            if (a.activation == 0) x = fmaxf(x, 0.f);
            else if (a.activation == 1) x = fast_tanhf(x);
          }
          if (a.heads == 2) {
            if (col == a.N - 2) a.v[row] = x;
            else if (col < a.N - 2) a.C[row * a.ldc + col] = x;
          } else if (a.heads) {
            if (col == a.N - 1) a.v[row] = x;
            else a.C[row * a.ldc + col] = x;
          } else {
            a.C[row * a.ldc + col] = x;
          }
        }
      }
    }
  }
}

// per-env finish after the fused env-step GEMM: reward from the pred^2
// partials, episode bookkeeping, and re-zero rsum for the next step
__global__ void env_finish2_kernel(const float* __restrict__ rsum,
                                   int* __restrict__ t,
                                   const int* __restrict__ horizons,
                                   float* __restrict__ epr,
                                   float* __restrict__ rewards,
                                   float* __restrict__ dones, int64_t E,
                                   int D, int npanels) {
  for (int64_t e = gidx(); e < E; e += gstride()) {
    float ss = 0.f;
    for (int p = 0; p < npanels; ++p) ss += rsum[(int64_t)p * E + e];
    const float r = 1.0f - ss / (float)D;
    rewards[e] = r;
    float ep = epr[e] + r;
    int tc = t[e] + 1;
    const int done = tc >= horizons[e] ? 1 : 0;
    dones[e] = (float)done;
    if (done) {
      ep = 0.f;
      tc = 0;
    }
    epr[e] = ep;
    t[e] = tc;
  }
}

// ---------------------------------------------------------------------------
// dW accumulation: dW[out][in] += delta^T @ acts (+ db += sum delta)
// grid = (m_tiles, n_tiles, splits); one wave per block.
// ---------------------------------------------------------------------------

struct DwArgs {
  const float* delta;  // [B][out]
  const float* acts;   // [B][in]
  float* slab;         // [splits][out][in] per-split partials (plain stores
                       // — atomics measured 8.4M adds/launch and dominated)
  float* db_slab;      // [splits][out]
  int64_t B;
  int out_dim, in_dim;
  int nt;       // 32-col tiles per wave (<= MAX_NT)
  int splits;   // K splits
  int ablate;   // perf diagnosis only: 4 skip MFMA, 8 skip slab stores
};

template <int NT, int MT>  // MT row fragments of 32 (1 or 2)
// the hot <2,2> instantiation needs the allocator pressured to keep 3
// waves/SIMD with the two-pair pipeline (unforced: 200+ VGPR, 2 waves)
__launch_bounds__(64, (NT == 2 && MT == 2) ? 3 : 1)
__global__ void dw_mfma_kernel(DwArgs a) {
  const int lane = threadIdx.x;
  const int m0 = blockIdx.x * (M_WAVE * MT);
  const int n_base = blockIdx.y * a.nt * M_WAVE;
  const int split = blockIdx.z;

  const int64_t rows_per = (a.B + a.splits - 1) / a.splits;
  const int64_t k0 = split * rows_per;
  const int64_t k1 = min(a.B, k0 + rows_per);

  const int i_l = lane & 31;
  const int k_l = lane >> 5;
  int mcol[MT];
  bool m_ok[MT];
  #pragma unroll
  for (int m = 0; m < MT; ++m) {
    mcol[m] = m0 + m * M_WAVE + i_l;
    m_ok[m] = mcol[m] < a.out_dim;
  }

  f32x16 acc[MT][NT];
  #pragma unroll
  for (int m = 0; m < MT; ++m)
    #pragma unroll
    for (int t = 0; t < NT; ++t)
      #pragma unroll
      for (int r = 0; r < 16; ++r) acc[m][t][r] = 0.f;
  float dbacc[MT];
  #pragma unroll
  for (int m = 0; m < MT; ++m) dbacc[m] = 0.f;

  // column guards hoisted out of the K loop (tail tiles load col 0 and
  // discard at scatter time — a harmless re-read keeps the loop branchless)
  int bcol[NT];
  bool col_ok[NT];
  #pragma unroll
  for (int t = 0; t < NT; ++t) {
    const int col = n_base + t * M_WAVE + i_l;
    col_ok[t] = col < a.in_dim;
    bcol[t] = col_ok[t] ? col : 0;
  }

  // two-pair ping-pong pipeline: TWO register sets alternate with no
  // copies, keeping two pairs of streamed loads in flight ahead of the
  // MFMAs (the one-pair version kept only ~16 B/wave outstanding and
  // measured load-phase-bound: 858 of 1079 us).  MFMA order stays k-
  // ascending -> results bitwise identical.
  const int64_t kend = k0 + ((k1 - k0) & ~1);
  int64_t k = k0;
  float av_A[MT], av_B[MT];
  float bv_A[NT], bv_B[NT];
  auto load_pair = [&](int64_t kk, float (&av)[MT], float (&bv)[NT]) {
    #pragma unroll
    for (int m = 0; m < MT; ++m)
      av[m] = m_ok[m] ? a.delta[(kk + k_l) * a.out_dim + mcol[m]] : 0.f;
    #pragma unroll
    for (int t = 0; t < NT; ++t)
      bv[t] = a.acts[(kk + k_l) * a.in_dim + bcol[t]];
  };
  auto fma_pair = [&](const float (&av)[MT], const float (&bv)[NT]) {
    if (!(a.ablate & 4)) {
      #pragma unroll
      for (int m = 0; m < MT; ++m) {
        dbacc[m] += av[m];
        #pragma unroll
        for (int t = 0; t < NT; ++t)
          acc[m][t] = __builtin_amdgcn_mfma_f32_32x32x2f32(av[m], bv[t],
                                                           acc[m][t], 0, 0, 0);
      }
    } else {
      #pragma unroll
      for (int m = 0; m < MT; ++m) dbacc[m] += av[m];
      #pragma unroll
      for (int t = 0; t < NT; ++t) dbacc[0] += bv[t] * 1e-38f;
    }
  };
  if (k < kend) load_pair(k, av_A, bv_A);
  if (k + 2 < kend) load_pair(k + 2, av_B, bv_B);
  #pragma unroll 1
  for (; k + 6 < kend; k += 4) {
    fma_pair(av_A, bv_A);
    load_pair(k + 4, av_A, bv_A);
    fma_pair(av_B, bv_B);
    load_pair(k + 6, av_B, bv_B);
  }
  if (k < kend) { fma_pair(av_A, bv_A); k += 2; }
  if (k < kend) { fma_pair(av_B, bv_B); k += 2; }
  #pragma unroll 1
  for (; k < kend; k += 2) {
    load_pair(k, av_A, bv_A);
    fma_pair(av_A, bv_A);
  }
  for (; k < k1; ++k) {
    #pragma unroll
    for (int m = 0; m < MT; ++m) {
      const float av =
          (m_ok[m] && k_l == 0) ? a.delta[k * a.out_dim + mcol[m]] : 0.f;
      if (k_l == 0) dbacc[m] += av;
      #pragma unroll
      for (int t = 0; t < NT; ++t) {
        const float bv = (k_l == 0) ? a.acts[k * a.in_dim + bcol[t]] : 0.f;
        acc[m][t] =
            __builtin_amdgcn_mfma_f32_32x32x2f32(av, bv, acc[m][t], 0, 0, 0);
      }
    }
  }

  // ---- store accumulators to this split's slab (each (row,col) of a
  // split is owned by exactly one block: no atomics, no zero-init) ----
  float* slab = a.slab + (int64_t)split * a.out_dim * a.in_dim;
  if (a.ablate & 8) return;
  #pragma unroll
  for (int m = 0; m < MT; ++m) {
    #pragma unroll
    for (int t = 0; t < NT; ++t) {
      if (col_ok[t]) {
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int mrow = m0 + m * M_WAVE + cd_row(r, lane);
          if (mrow < a.out_dim)
            slab[(int64_t)mrow * a.in_dim + bcol[t]] = acc[m][t][r];
        }
      }
    }
  }
  if (blockIdx.y == 0) {
    #pragma unroll
    for (int m = 0; m < MT; ++m) {
      const float other = __shfl(dbacc[m], lane ^ 32, WAVE);
      if (lane < 32 && m_ok[m])
        a.db_slab[(int64_t)split * a.out_dim + mcol[m]] = dbacc[m] + other;
    }
  }
}

// ---------------------------------------------------------------------------
// dW glds kernel (round-2 lever #2): one 8-wave block owns the WHOLE
// [out<=64][in<=384] dW tile for one K split, with both operands staged
// row-major by global_load_lds into a 3-deep ring with counted vmcnt
// (same pipeline as gemm_fwd_glds_kernel).  Replaces the 1-wave
// dw_mfma_kernel's 4-B-per-lane streams for the big flagship dW calls:
// delta is read ONCE (the old n-tile grid re-read it 6x) and the block
// keeps two full stages in flight.
// Both MFMA operands are row-major-k, column-per-lane LDS reads
// (conflict-free); K-split tails zero the delta rows past the split so
// clamped/garbage rows contribute nothing.
// ---------------------------------------------------------------------------

constexpr int DWG_K = 16;        // k rows per stage

// Generic over the block geometry: MW m-wave-tiles x NWV n-wave-groups
// of NTW 32-col tiles (threads = MW*NWV*64).  <2,4,3> covers the wide
// dW1 tile [64][384]; <2,2,1> covers the narrow [64][64] shapes (dW2,
// padded heads dW) without idling 3/4 of the block.
template <int MW, int NWV, int NTW, int RING = 3>
__launch_bounds__(MW * NWV * 64, 2)
__global__ void dw_glds_kernel(DwArgs a) {
  constexpr int THREADS = MW * NWV * 64;
  constexpr int DWG_M = MW * 32;
  constexpr int DWG_N = NWV * NTW * 32;
  constexpr int DWG_DB = DWG_K * DWG_M * 4;
  constexpr int DWG_AB = DWG_K * DWG_N * 4;
  constexpr int DWG_SLOT = DWG_DB + DWG_AB;
  constexpr int DWG_PIECES = DWG_SLOT / 16;
  constexpr int DWG_PPT = (DWG_PIECES + THREADS - 1) / THREADS;
  // RING: the wide <2,4,3> tile is 28 KB/stage -> 1 block/CU at ANY ring
  // depth, so a 5-deep ring (4 stages = 112 KB in flight) is free and
  // covers loaded-chip HBM latency the 3-deep ring stalled on
  __shared__ __attribute__((aligned(16))) char smem[RING * DWG_SLOT + 4096];
  char* const pad = smem + RING * DWG_SLOT;  // dummy glds target (uniformity)
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wave = tid >> 6;
  const int i_l = lane & 31;
  const int k_l = lane >> 5;
  const int wm = wave / NWV;              // m-tile of 32
  const int wn = wave % NWV;              // n-group of NTW*32
  const int split = blockIdx.x;
  // n-panel (grid.y): narrow tiles with small per-stage LDS let 2 blocks
  // co-reside per CU so stage-barrier bubbles overlap across blocks; the
  // (small) delta stage is duplicated per panel
  const int nb0 = (int)blockIdx.y * DWG_N;

  const int64_t rows_per = (a.B + a.splits - 1) / a.splits;
  const int64_t k0 = (int64_t)split * rows_per;
  const int64_t k1 = min(a.B, k0 + rows_per);
  if (k0 >= a.B) return;
  const int S = (int)((k1 - k0 + DWG_K - 1) / DWG_K);

  // per-thread glds sources: piece q = tid + p*THREADS; rows past B
  // clamp to B-1 (garbage — the delta-row zeroing kills them)
  const float* base[DWG_PPT];
  int krow_p[DWG_PPT];
  int ld_p[DWG_PPT];
  unsigned dsto[DWG_PPT];
  #pragma unroll
  for (int p = 0; p < DWG_PPT; ++p) {
    const int q = tid + p * THREADS;
    if (q < DWG_DB / 16) {               // delta piece
      const int kq = q / (DWG_M / 4);
      int c4 = (q % (DWG_M / 4)) * 4;
      c4 = c4 + 3 < a.out_dim ? c4 : (a.out_dim > 4 ? a.out_dim - 4 : 0);
      base[p] = a.delta + c4;
      krow_p[p] = kq;
      ld_p[p] = a.out_dim;
      dsto[p] = q * 16;
    } else if (q < DWG_PIECES) {         // acts piece
      const int qq = q - DWG_DB / 16;
      const int kq = qq / (DWG_N / 4);
      int c4 = nb0 + (qq % (DWG_N / 4)) * 4;
      c4 = c4 + 3 < a.in_dim ? c4 : (a.in_dim > 4 ? a.in_dim - 4 : 0);
      base[p] = a.acts + c4;
      krow_p[p] = kq;
      ld_p[p] = a.in_dim;
      dsto[p] = q * 16;
    } else {                             // pad piece: dummy, uniform vmcnt
      base[p] = a.delta;
      krow_p[p] = 0;
      ld_p[p] = 0;
      dsto[p] = 0xFFFFFFFFu;
    }
  }

  auto issue = [&](int s) {
    char* slot = smem + (s % RING) * DWG_SLOT;
    #pragma unroll
    for (int p = 0; p < DWG_PPT; ++p) {
      int64_t row = k0 + (int64_t)s * DWG_K + krow_p[p];
      row = row < a.B ? row : a.B - 1;
      const float* g = base[p] + row * ld_p[p];
      char* dst = dsto[p] == 0xFFFFFFFFu ? pad + (tid & 255) * 16
                                         : slot + dsto[p];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned*)g,
          (unsigned*)dst, 16, 0, 0);
    }
  };

  #pragma unroll
  for (int p = 0; p < RING - 1; ++p)
    if (p < S) issue(p);

  f32x16 acc[NTW];
  #pragma unroll
  for (int t = 0; t < NTW; ++t)
    #pragma unroll
    for (int r = 0; r < 16; ++r) acc[t][r] = 0.f;
  float dbacc = 0.f;
  const int mcol = wm * M_WAVE + i_l;

  for (int s = 0; s < S; ++s) {
    // stage s landed when only the rem in-flight later stages' loads
    // (DWG_PPT each) remain outstanding
    const int rem = min(S - 1 - s, RING - 2);
    if (RING >= 5 && rem >= 3)
      asm volatile("s_waitcnt vmcnt(%0)" :: "i"(3 * DWG_PPT) : "memory");
    else if (RING >= 4 && rem == 2)
      asm volatile("s_waitcnt vmcnt(%0)" :: "i"(2 * DWG_PPT) : "memory");
    else if (rem == 1)
      asm volatile("s_waitcnt vmcnt(%0)" :: "i"(DWG_PPT) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    char* slot = smem + (s % RING) * DWG_SLOT;
    float* ds = (float*)slot;                    // [16][DWG_M] delta
    float* as = (float*)(slot + DWG_DB);         // [16][DWG_N] acts
    const int64_t kb = k0 + (int64_t)s * DWG_K;
    const int kval = (int)((k1 - kb) < DWG_K ? (k1 - kb) : DWG_K);
    if (kval < DWG_K) {
      // zero the delta rows past the split (kills garbage contributions)
      for (int idx = tid; idx < (DWG_K - kval) * DWG_M; idx += THREADS)
        ds[kval * DWG_M + idx] = 0.f;
      __builtin_amdgcn_s_barrier();
    }
    // buf (s+RING-1)%RING == (s-1)%RING, freed by the barrier above
    if (s + RING - 1 < S) issue(s + RING - 1);
    #pragma unroll 4
    for (int k2 = 0; k2 < DWG_K; k2 += 2) {
      const float av = ds[(k2 + k_l) * DWG_M + mcol];
      dbacc += av;
      #pragma unroll
      for (int t = 0; t < NTW; ++t) {
        const float bv =
            as[(k2 + k_l) * DWG_N + wn * (NTW * M_WAVE) + t * M_WAVE + i_l];
        acc[t] = __builtin_amdgcn_mfma_f32_32x32x2f32(av, bv, acc[t], 0, 0, 0);
      }
    }
  }

  // ---- store to this split's slab ----
  float* slab = a.slab + (int64_t)split * a.out_dim * a.in_dim;
  #pragma unroll
  for (int t = 0; t < NTW; ++t) {
    const int col = nb0 + wn * (NTW * M_WAVE) + t * M_WAVE + i_l;
    if (col < a.in_dim) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int mrow = wm * M_WAVE + cd_row(r, lane);
        if (mrow < a.out_dim)
          slab[(int64_t)mrow * a.in_dim + col] = acc[t][r];
      }
    }
  }
  if (wn == 0 && blockIdx.y == 0) {
    const float other = __shfl(dbacc, lane ^ 32, WAVE);
    if (lane < 32 && mcol < a.out_dim)
      a.db_slab[(int64_t)split * a.out_dim + mcol] = dbacc + other;
  }
}

// Split-slab reduction into the flat grad, handling the optional
// combined-heads row split (rows < split_row -> dW/db, rest -> dW2/db2).
// Parallel over (element, split-chunk): a serial full-splits loop per
// element left the chip at <1 wave/SIMD and 480 us per call.
constexpr int DW_RED_CHUNK = 32;

__global__ void dw_reduce_kernel(const float* __restrict__ slab,
                                 const float* __restrict__ db_slab,
                                 float* __restrict__ dW, float* __restrict__ db,
                                 float* __restrict__ dW2,
                                 float* __restrict__ db2, int64_t out_dim,
                                 int64_t in_dim, int splits, int split_row) {
  const int64_t n = out_dim * in_dim;
  const int s0 = blockIdx.y * DW_RED_CHUNK;
  const int s1 = min(splits, s0 + DW_RED_CHUNK);
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int s = s0; s < s1; ++s) acc += slab[(int64_t)s * n + i];
    const int64_t mrow = i / in_dim;
    float* dst = (split_row >= 0 && mrow >= split_row)
                     ? &dW2[(mrow - split_row) * in_dim + i % in_dim]
                     : &dW[i];
    atomicAdd(dst, acc);
  }
}

__global__ void db_reduce_kernel(const float* __restrict__ db_slab,
                                 float* __restrict__ db,
                                 float* __restrict__ db2, int64_t out_dim,
                                 int splits, int split_row) {
  // one wave per output column; lanes stride the splits axis
  const int m = blockIdx.x;
  if (m >= out_dim) return;
  const int lane = threadIdx.x;
  float acc = 0.f;
  for (int s = lane; s < splits; s += WAVE)
    acc += db_slab[(int64_t)s * out_dim + m];
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    if (split_row >= 0 && m >= split_row) db2[m - split_row] += acc;
    else if (db != nullptr) db[m] += acc;
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// Bindings
// ---------------------------------------------------------------------------

void gemm_fwd(torch::Tensor X, torch::Tensor Wt, torch::Tensor bias,
              int64_t activation, int64_t heads, torch::Tensor C,
              torch::Tensor v, torch::Tensor aux, int64_t wt_layout,
              int64_t ablate, int64_t ldc) {
  // C and (for heads) v are caller-allocated so activations can land
  // directly in the backward's blob layout.  ldc (0 = natural width) lets
  // C land in a column block of a wider row-major buffer; N > 128 runs as
  // column panels on the pipelined kernel (grid.y).
  const int64_t B = X.size(0);
  const int K = static_cast<int>(X.size(1));
  const int N = static_cast<int>(wt_layout ? Wt.size(0) : Wt.size(1));
  TORCH_CHECK(X.is_cuda() && X.is_contiguous() && Wt.is_contiguous());
  TORCH_CHECK(C.is_contiguous());
  TORCH_CHECK((wt_layout ? Wt.size(1) : Wt.size(0)) == K);
  TORCH_CHECK(activation >= 3 || bias.numel() == N);
  const int ncols = static_cast<int>(heads == 2 ? N - 2 : heads ? N - 1 : N);
  const int ldc_eff = static_cast<int>(ldc > 0 ? ldc : ncols);
  TORCH_CHECK(ldc_eff >= ncols);
  if (heads) {
    TORCH_CHECK(C.numel() >= (B - 1) * ldc_eff + ncols && v.numel() == B);
  } else {
    TORCH_CHECK(C.numel() >= (B - 1) * ldc_eff + ncols);
  }

  FwdArgs a{};
  a.X = X.data_ptr<float>();
  a.Wt = Wt.data_ptr<float>();
  a.bias = bias.data_ptr<float>();
  a.B = B;
  a.K = K;
  a.N = N;
  a.activation = static_cast<int>(activation);
  a.heads = static_cast<int>(heads);
  a.wt_layout = static_cast<int>(wt_layout);
  a.ablate = static_cast<int>(ablate);
  a.ldc = ldc_eff;
  a.C = C.data_ptr<float>();
  a.v = heads ? v.data_ptr<float>() : nullptr;
  if (activation >= 3) {
    TORCH_CHECK(aux.numel() == B * N, "dgrad mode needs aux = fwd activations");
    a.aux = aux.data_ptr<float>();
  }

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int64_t tiles = (B + FWD_M - 1) / FWD_M;
  // plain kernel: capped grid + tile loop.  pipe kernel: one block per
  // tile — block scheduling hides the per-tile staging prologue.
  const int grid = static_cast<int>(std::min<int64_t>(tiles, 4096));
  const int grid_pipe = static_cast<int>(std::min<int64_t>(tiles, 1 << 22));
  const int NT = (N + M_WAVE - 1) / M_WAVE;
  // T14 register-pipelined variant (see gemm_fwd_pipe_kernel).  Its
  // branchless clamped loads need K>=4 (X / layout-1 W rows) and, for
  // layout 0, N>=4 (W rows); tiny shapes take the plain kernel.
  static const int pipe_env = []() {
    const char* e = getenv("DPPO_GEMM_PIPE");
    return e ? atoi(e) : 1;
  }();
  static const int pbk_env = []() {
    const char* e = getenv("DPPO_GEMM_PBK");
    return e ? atoi(e) : 0;  // 0 = per-shape heuristic
  }();
  // K%4!=0 is allowed for the dgrad-from-gh call (activation>=3,
  // wt_layout 0): its X is the gh buffer, allocated with a float4 of
  // slack, and the overhang columns multiply zeroed W rows >= K.
  const bool pipe = pipe_env && !a.ablate && K >= 4 &&
                    (K % 4 == 0 || (wt_layout == 0 && activation >= 3)) &&
                    (wt_layout == 1 || (N >= 4 && N % 4 == 0));
  const int pbk = pbk_env ? pbk_env : 16;
  static const int glds_env = []() {
    const char* e = getenv("DPPO_GEMM_GLDS");
    return e ? atoi(e) : 1;
  }();
  // glds 3-buffer streaming variant: NT==2 only (wave-uniform vmcnt),
  // layout-0 W, 16-B-aligned rows (K%4==0, N%4==0), no ablation hooks
  const bool glds_ok = glds_env && !a.ablate && wt_layout == 0 &&
                       K % 4 == 0 && K >= 4 && N % 4 == 0 && N >= 4 &&
                       a.activation != 9999;
  static const int ring_env = []() {
    const char* e = getenv("DPPO_FWD_RING");
    return e ? atoi(e) : 0;  // 0 = by shape (below); 2/3/4 force
  }();
  // by shape: short-K GEMMs (few stages) run the 2-deep ring — 24 KB
  // LDS lets 6 blocks/CU co-reside and prologue/drain is a large
  // fraction of S (L2t K=64: 179 -> 157 us at B=1M); long-K streams
  // keep the 3-deep ring (L1t K=376: wash; ring 4 measured slower)
  const int ring_sel = ring_env ? ring_env : (a.K <= 128 ? 2 : 3);
  #define DISPATCH_FWD(NTV, NPANEL)                                          \
    if (glds_ok && NTV == 2 && ring_sel == 2)                                \
      hipLaunchKernelGGL(gemm_fwd_glds_kernel<2>,                            \
                         dim3((unsigned)grid_pipe, NPANEL),                  \
                         dim3(FWD_WAVES * 64), 0, stream, a);                \
    else if (glds_ok && NTV == 2 && ring_sel == 4)                           \
      hipLaunchKernelGGL(gemm_fwd_glds_kernel<4>,                            \
                         dim3((unsigned)grid_pipe, NPANEL),                  \
                         dim3(FWD_WAVES * 64), 0, stream, a);                \
    else if (glds_ok && NTV == 2)                                            \
      hipLaunchKernelGGL(gemm_fwd_glds_kernel<3>,                            \
                         dim3((unsigned)grid_pipe, NPANEL),                  \
                         dim3(FWD_WAVES * 64), 0, stream, a);                \
    else if (pipe && pbk == 32)                                              \
      hipLaunchKernelGGL((gemm_fwd_pipe_kernel<NTV, 32>),                    \
                         dim3(grid_pipe, NPANEL), dim3(FWD_WAVES * 64), 0,   \
                         stream, a);                                         \
    else if (pipe)                                                           \
      hipLaunchKernelGGL((gemm_fwd_pipe_kernel<NTV, 16>),                    \
                         dim3(grid_pipe, NPANEL), dim3(FWD_WAVES * 64), 0,   \
                         stream, a);                                         \
    else                                                                     \
      hipLaunchKernelGGL(gemm_fwd_kernel<NTV>, dim3(grid),                   \
                         dim3(FWD_WAVES * 64), 0, stream, a)
  if (N > MAX_NT * M_WAVE) {
    // wide N: NT=2 column panels (the NT=2 pipe schedule holds 4
    // waves/SIMD; NT=4 would drop to 2) — pipe-capable shapes only
    TORCH_CHECK(pipe, "N > 128 needs the pipelined path (K%4==0 etc.)");
    const int panels = (N + 2 * M_WAVE - 1) / (2 * M_WAVE);
    DISPATCH_FWD(2, panels);
  } else {
    switch (NT) {
      case 1: DISPATCH_FWD(1, 1); break;
      case 2: DISPATCH_FWD(2, 1); break;
      case 3: DISPATCH_FWD(3, 1); break;
      default: DISPATCH_FWD(4, 1); break;
    }
  }
  #undef DISPATCH_FWD
}

void gemm_env_step(torch::Tensor xva, torch::Tensor M, torch::Tensor xin,
                   torch::Tensor xout, torch::Tensor envd,
                   torch::Tensor horizons, torch::Tensor t,
                   torch::Tensor epr,
                   torch::Tensor rewards, torch::Tensor dones,
                   torch::Tensor rsum, torch::Tensor seed_dev, double sigma,
                   int64_t step) {
  const int64_t E = xva.size(0);
  const int K = static_cast<int>(xva.size(1));
  const int D = static_cast<int>(M.size(1));
  TORCH_CHECK(xva.is_contiguous() && M.is_contiguous() &&
              xin.is_contiguous() && xout.is_contiguous());
  TORCH_CHECK(M.size(0) == K && xin.size(0) == E && xin.size(1) == D);
  TORCH_CHECK(xout.numel() == E * D);
  TORCH_CHECK(K % 4 == 0 && D % 4 == 0, "fused env step needs 16B rows");
  TORCH_CHECK(horizons.dtype() == torch::kInt32 && t.dtype() == torch::kInt32);
  const int panels_chk = (D + 63) / 64;
  TORCH_CHECK(rsum.numel() >= panels_chk * E);
  FwdArgs a{};
  a.X = xva.data_ptr<float>();
  a.Wt = M.data_ptr<float>();
  a.B = E;
  a.K = K;
  a.N = D;
  a.activation = 5;
  a.ldc = D;
  a.env_xin = xin.data_ptr<float>();
  a.env_x = xout.data_ptr<float>();
  a.envd = envd.data_ptr<float>();
  a.horizons = horizons.data_ptr<int>();
  a.tcount = t.data_ptr<int>();
  a.seed_dev = reinterpret_cast<const long long*>(seed_dev.data_ptr<int64_t>());
  a.env_rsum = rsum.data_ptr<float>();
  a.sigma = (float)sigma;
  a.step = (int)step;
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int64_t tiles = (E + FWD_M - 1) / FWD_M;
  const int panels = (D + 63) / 64;
  hipLaunchKernelGGL(gemm_fwd_glds_kernel<3>,
                     dim3((unsigned)tiles, (unsigned)panels),
                     dim3(FWD_WAVES * 64), 0, stream, a);
  hipLaunchKernelGGL(env_finish2_kernel, dim3(elementwise_grid(E, 256)),
                     dim3(256), 0, stream, rsum.data_ptr<float>(),
                     t.data_ptr<int>(), horizons.data_ptr<int>(),
                     epr.data_ptr<float>(), rewards.data_ptr<float>(),
                     dones.data_ptr<float>(), E, D, panels);
}

// Grow-only scratch cache for the dW split slabs.  Allocating them per
// call fragmented the caching allocator against hipGraph private pools
// (tests with several captured engines measured SECONDS per eager dW
// call in hipMalloc/hipFree); a persistent buffer also gives captures a
// stable address.  Correctness: every (split, element) of the used
// region is overwritten each call (splits are clamped so k0 < B).
static torch::Tensor& dw_scratch(int idx, int64_t need,
                                 const torch::TensorOptions& opt) {
  static torch::Tensor cache[2];
  static std::vector<torch::Tensor> retired;  // graphs captured against an
                                              // old buffer must keep it
                                              // alive (grow-only, bounded
                                              // by the few distinct sizes)
  torch::Tensor& t = cache[idx];
  if (!t.defined() || t.numel() < need) {
    if (t.defined()) retired.push_back(t);
    t = torch::empty({need}, opt);
  }
  return t;
}

void dw_mfma(torch::Tensor delta, torch::Tensor acts, torch::Tensor grad_buf,
             int64_t w_off, int64_t b_off, int64_t split_row, int64_t w_off2,
             int64_t b_off2, int64_t ablate) {
  const int64_t B = delta.size(0);
  const int out_dim = static_cast<int>(delta.size(1));
  const int in_dim = static_cast<int>(acts.size(1));
  TORCH_CHECK(delta.is_contiguous() && acts.is_contiguous());

  DwArgs a{};
  a.delta = delta.data_ptr<float>();
  a.acts = acts.data_ptr<float>();
  a.B = B;
  a.out_dim = out_dim;
  a.in_dim = in_dim;
  a.ablate = static_cast<int>(ablate);
  // tile shape: maximize N per wave to avoid re-reading delta; wide
  // out_dim also takes 2 row fragments per wave (halves acts re-reads)
  // Wide out_dim: 2 row fragments with NT capped at 2 keeps the
  // accumulators at 64 AGPR (4 waves/SIMD) while halving the streamed
  // acts re-reads (MT=2 with NT=4 = 128 AGPR measured 17x stalls).
  const int mt = (out_dim >= 64) ? 2 : 1;
  a.nt = std::min(mt == 2 ? 2 : MAX_NT, (in_dim + M_WAVE - 1) / M_WAVE);
  const int m_tiles = (out_dim + M_WAVE * mt - 1) / (M_WAVE * mt);
  const int n_tiles = (in_dim + a.nt * M_WAVE - 1) / (a.nt * M_WAVE);
  // enough waves to hide the streamed-operand latency; slab stores make
  // extra splits nearly free (the reduce is split-chunk parallel)
  const int target_blocks = 4096;
  a.splits = std::max(1, target_blocks / std::max(1, m_tiles * n_tiles));
  a.splits = static_cast<int>(
      std::min<int64_t>(a.splits, std::max<int64_t>(1, B / 256)));

  auto& slab = dw_scratch(0, (int64_t)a.splits * out_dim * in_dim,
                          delta.options());
  auto& db_slab = dw_scratch(1, (int64_t)a.splits * out_dim,
                             delta.options());
  // the old kernels' split grid also overwrites every used element, but
  // a db tail beyond this call's splits must not leak: zero the region
  db_slab.narrow(0, 0, (int64_t)a.splits * out_dim).zero_();
  a.slab = slab.data_ptr<float>();
  a.db_slab = db_slab.data_ptr<float>();

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  static const int dwglds_env = []() {
    const char* e = getenv("DPPO_DW_GLDS");
    return e ? atoi(e) : 1;
  }();
  // 8-wave LDS-staged variant: whole dW tile per block, delta read once
  // <2,4,3> covers wide-acts tiles (dW1: in=obs); <2,2,1> covers the
  // narrow [<=64][<=64] shapes (dW2, padded heads dW)
  const bool glds_wide = in_dim > 192 && in_dim <= 384;
  const bool glds_narrow = in_dim <= 64 && out_dim <= 64;
  if (dwglds_env && !a.ablate && out_dim <= 64 &&
      (glds_wide || glds_narrow) && out_dim % 4 == 0 && in_dim % 4 == 0 &&
      B >= 4096) {
    const char* sc_env = getenv("DPPO_DW_SPLITS");  // re-read: sweepable
    const int64_t split_cap = sc_env ? (int64_t)atoll(sc_env) : (int64_t)1024;
    a.splits = static_cast<int>(
        std::min<int64_t>(split_cap, std::max<int64_t>(256, B / 4096)));
    a.splits = static_cast<int>(
        std::min<int64_t>(a.splits, std::max<int64_t>(1, B / DWG_K)));
    auto& slab2 = dw_scratch(0, (int64_t)a.splits * out_dim * in_dim,
                             delta.options());
    auto& db_slab2 = dw_scratch(1, (int64_t)a.splits * out_dim,
                                delta.options());
    a.slab = slab2.data_ptr<float>();
    a.db_slab = db_slab2.data_ptr<float>();
    if (glds_wide)
      // single 384-wide tile, 5-deep ring (measured: 2-panel <2,2,3,4>
      // grids with 2 blocks/CU ran 6% SLOWER — the duplicated delta
      // stage and shorter MFMA chains cost more than the cross-block
      // barrier overlap bought; both are ~60% of the f32-MFMA floor)
      hipLaunchKernelGGL((dw_glds_kernel<2, 4, 3, 5>),
                         dim3((unsigned)a.splits, 1), dim3(512), 0,
                         stream, a);
    else
      hipLaunchKernelGGL((dw_glds_kernel<2, 2, 1>), dim3((unsigned)a.splits),
                         dim3(256), 0, stream, a);
    float* dW = grad_buf.data_ptr<float>() + w_off;
    float* db = (b_off >= 0) ? grad_buf.data_ptr<float>() + b_off : nullptr;
    float* dW2 =
        (split_row >= 0) ? grad_buf.data_ptr<float>() + w_off2 : nullptr;
    float* db2 = (split_row >= 0 && b_off2 >= 0)
                     ? grad_buf.data_ptr<float>() + b_off2
                     : nullptr;
    const int n_chunks = (a.splits + DW_RED_CHUNK - 1) / DW_RED_CHUNK;
    const dim3 rgrid(elementwise_grid((int64_t)out_dim * in_dim, 256),
                     n_chunks);
    hipLaunchKernelGGL(dw_reduce_kernel, rgrid, dim3(256), 0, stream, a.slab,
                       a.db_slab, dW, db, dW2, db2, out_dim, in_dim, a.splits,
                       static_cast<int>(split_row));
    hipLaunchKernelGGL(db_reduce_kernel, dim3(out_dim), dim3(WAVE), 0,
                       stream, a.db_slab, db, db2, out_dim, a.splits,
                       static_cast<int>(split_row));
    return;
  }
  const dim3 grid(m_tiles, n_tiles, a.splits);
  const int key = a.nt * 10 + mt;
  switch (key) {
    case 11: hipLaunchKernelGGL((dw_mfma_kernel<1, 1>), grid, dim3(WAVE), 0, stream, a); break;
    case 21: hipLaunchKernelGGL((dw_mfma_kernel<2, 1>), grid, dim3(WAVE), 0, stream, a); break;
    case 31: hipLaunchKernelGGL((dw_mfma_kernel<3, 1>), grid, dim3(WAVE), 0, stream, a); break;
    case 41: hipLaunchKernelGGL((dw_mfma_kernel<4, 1>), grid, dim3(WAVE), 0, stream, a); break;
    case 12: hipLaunchKernelGGL((dw_mfma_kernel<1, 2>), grid, dim3(WAVE), 0, stream, a); break;
    case 22: hipLaunchKernelGGL((dw_mfma_kernel<2, 2>), grid, dim3(WAVE), 0, stream, a); break;
    case 32: hipLaunchKernelGGL((dw_mfma_kernel<3, 2>), grid, dim3(WAVE), 0, stream, a); break;
    default: hipLaunchKernelGGL((dw_mfma_kernel<4, 2>), grid, dim3(WAVE), 0, stream, a); break;
  }
  float* dW = grad_buf.data_ptr<float>() + w_off;
  float* db = (b_off >= 0) ? grad_buf.data_ptr<float>() + b_off : nullptr;
  float* dW2 = (split_row >= 0) ? grad_buf.data_ptr<float>() + w_off2 : nullptr;
  float* db2 = (split_row >= 0 && b_off2 >= 0)
                   ? grad_buf.data_ptr<float>() + b_off2
                   : nullptr;
  {
    const int n_chunks = (a.splits + DW_RED_CHUNK - 1) / DW_RED_CHUNK;
    const dim3 rgrid(elementwise_grid((int64_t)out_dim * in_dim, 256),
                     n_chunks);
    hipLaunchKernelGGL(dw_reduce_kernel, rgrid, dim3(256), 0, stream, a.slab,
                       a.db_slab, dW, db, dW2, db2, out_dim, in_dim, a.splits,
                       static_cast<int>(split_row));
    hipLaunchKernelGGL(db_reduce_kernel, dim3(out_dim), dim3(WAVE), 0,
                       stream, a.db_slab, db, db2, out_dim, a.splits,
                       static_cast<int>(split_row));
  }
}

