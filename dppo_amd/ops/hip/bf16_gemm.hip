// Hand-written bf16 MFMA GEMM path for the wide MLP config (gfx950).
//
// Replaces the round-1 rocBLAS/autocast hot path of BASELINE config 5
// (obs=4096, 4x4096 hidden, bf16) with CDNA4-native kernels:
//
//   bf16_mm256      C[M,N] = epi(A[M,K] @ B[N,K]^T)  — 256x256 tile,
//                   BK=64, 8 waves, v_mfma_f32_16x16x32_bf16, 4-phase
//                   pipelined global_load_lds staging with counted vmcnt
//                   across raw barriers, st_16x32 LDS swizzle, XCD-aware
//                   bijective block remap, LDS-repacked coalesced
//                   epilogue with fused bias+tanh / dtanh / f32-grad
//                   stores (the cdna guide's 256^2 8-phase template,
//                   restructured as 4 phases per K-tile x 2-tile
//                   software pipeline with identical in-flight counts).
//   bf16_mm_small   guarded 64x64-tile variant for the ragged heads
//                   GEMMs (N or M = P+1 = 513), incl. heads-split and
//                   split-row grad epilogues.
//   bf16_transpose  64x64 LDS tile transpose with optional fused column
//                   sums (the bias gradients ride along for free).
//   gauss_gh_wide   wave-per-row PPO loss gradient for wide policies
//                   (A > 32; lane-strided column loops), bf16 output
//                   feeding the dgrad GEMM chain directly.
//
// The torch weight layout [out][in] IS the MFMA B-operand layout (both
// fragments read 8 contiguous bf16 along K), so forward needs no weight
// transposes; dgrad uses W^T copies and dW uses transposed activations/
// deltas produced by bf16_transpose.
//
// Reference semantics: these kernels implement the same per-layer math
// the reference's tf.layers.dense graph launches (reference Model.py:12-14,
// PPO.py:46 backward) at the wide config's shapes.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"
#include "ppo_math.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) unsigned short ushort4;

DEV_INLINE float bf2f(unsigned short s) {
  return __uint_as_float(((unsigned)s) << 16);
}

DEV_INLINE unsigned short f2bf(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);  // round-to-nearest-even
  return *reinterpret_cast<unsigned short*>(&h);
}

// ---------------------------------------------------------------------------
// bf16_mm256: 256x256x(K)  C = epi(A @ B^T)
// ---------------------------------------------------------------------------

constexpr int BK = 64;             // K elements per tile
constexpr int IMG = 32 * 1024;     // one [256][64] bf16 image
constexpr int HALFB = 16 * 1024;   // [128][64] half image
// LDS: A images at 0/IMG, B images at 2*IMG/3*IMG = 128 KiB total
constexpr int LDS_BYTES = 4 * IMG;

enum { EPI_RAW = 0, EPI_TANH_BIAS = 1, EPI_DTANH = 2, EPI_GRAD = 3 };

// XOR swizzle: involution on byte offsets within an image.  Rows are
// 128 B, so byte bits 8/9/10 are row-index bits 1/2/3; XOR-ing them into
// the three 16-B slot-select bits (4/5/6) gives every one of the 16 rows
// a fragment read touches a distinct 16-B slot of the 256-B bank row —
// conflict-FREE ds_read_b128 for the [row=base+(lane&15)] access pattern
// (the guide's 2-bit st_16x32 left a residual 2-way conflict here).
DEV_INLINE unsigned swz(unsigned o) {
  return o ^ (((o >> 9) & 1u) << 5) ^ (((o >> 8) & 1u) << 4) ^
         (((o >> 10) & 1u) << 6);
}

struct MM256Args {
  const unsigned short* A;  // [M][K]
  const unsigned short* B;  // [N][K]
  unsigned short* C;        // [M][N] bf16 (epi 0/1/2)
  const float* bias;        // [N]     (epi 1)
  const unsigned short* aux;  // [M][N] (epi 2: dtanh factor source)
  float* grad;              // epi 3: f32 out at grad[m*N + n]
  unsigned short* CT;       // optional transposed dual-write [N][ldt]
  float* sums;              // optional [N] f32 += column sums (with CT)
  long M, N, K, ldt;
  int nbn;                  // N / 256
  int epi;
};

// Per-thread glds source pointer for one (half, piece): linear LDS dest,
// st_16x32 pre-swizzle applied to the SOURCE address (guide rule 21:
// glds dest must stay lane-linear).  Row/col are constant per thread, so
// the 64-bit address math happens ONCE; the K-loop just adds k0.
DEV_INLINE const unsigned short* stage_src(const unsigned short* g, long row0,
                                           long ld, int half, int p, int tid) {
  const unsigned o = (unsigned)(half * HALFB + p * 8192 + tid * 16);
  const unsigned so = swz(o);  // flips only bit 5 within the image
  const unsigned row = so >> 7;        // 128 B per row
  const unsigned colb = so & 127u;
  return g + (row0 + row) * ld + (colb >> 1);
}

DEV_INLINE void stage_piece(const unsigned short* src, long k0, char* img,
                            int half, int p, int wave) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned*)(src + k0),
      (unsigned*)(img + half * HALFB + p * 8192 + wave * 1024), 16, 0, 0);
}

DEV_INLINE short8 frag_read(const char* img, int row, int colb) {
  unsigned byte = (unsigned)(row * 128 + colb);
  byte = swz(byte);
  return *(const short8*)(img + byte);
}

template <int NPH, bool TWO_BARRIERS, bool STATIC_PRIO = false,
          int NIMG = 2>
__launch_bounds__(512, 2)
__global__ void bf16_mm256_kernel(MM256Args a) {
  constexpr int FMPP = 8 / NPH;  // fm-blocks computed per phase
  // NIMG == 3: a third A image (5 x 32 KB = the full 160 KB CU; still
  // 1 block/CU so occupancy is unchanged) lets ph0 stage A TWO tiles
  // ahead — the tile-seam wait then only covers the t+2 stages
  // (vmcnt(8)), never tile t+1's A, which landed a full tile earlier.
  __shared__ __attribute__((aligned(16))) char smem[(NIMG + 2) * IMG];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2;   // 2 M-halves
  const int wc = wave & 3;    // 4 N-quarters

  // bijective XCD-aware remap (8 XCDs; block b runs on XCD b%8)
  const long nwg = (long)gridDim.x;
  const long q = nwg >> 3, r8 = nwg & 7;
  const long xcd = blockIdx.x & 7, idx = blockIdx.x >> 3;
  const long sw = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q)
                  + idx;
  const int bm = (int)(sw / a.nbn);
  const int bn = (int)(sw % a.nbn);

  char* const imgA0 = smem;
  char* const imgA1 = smem + IMG;
  char* const imgA2 = smem + 2 * IMG;  // NIMG == 3 only
  char* const imgB0 = smem + NIMG * IMG;
  char* const imgB1 = smem + (NIMG + 1) * IMG;
  auto imgA = [&](int t) {
    return NIMG == 2 ? ((t & 1) ? imgA1 : imgA0)
                     : (t % 3 == 0 ? imgA0 : t % 3 == 1 ? imgA1 : imgA2);
  };
  const long rowA = (long)bm * 256;
  const long rowB = (long)bn * 256;
  const int NT = (int)(a.K / BK);

  f32x4 acc[8][4] = {};

  // T5 static form: the second-dispatched half (waves 4-7) loses VALU
  // arbitration to the older half on every segment; one wave-uniform
  // s_setprio(1) for it, no per-cluster flips
  if (STATIC_PRIO &&
      __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  // per-thread glds source pointers (constant rows/cols; advance by k0)
  const unsigned short* sA00 = stage_src(a.A, rowA, a.K, 0, 0, tid);
  const unsigned short* sA01 = stage_src(a.A, rowA, a.K, 0, 1, tid);
  const unsigned short* sA10 = stage_src(a.A, rowA, a.K, 1, 0, tid);
  const unsigned short* sA11 = stage_src(a.A, rowA, a.K, 1, 1, tid);
  const unsigned short* sB00 = stage_src(a.B, rowB, a.K, 0, 0, tid);
  const unsigned short* sB01 = stage_src(a.B, rowB, a.K, 0, 1, tid);
  const unsigned short* sB10 = stage_src(a.B, rowB, a.K, 1, 0, tid);
  const unsigned short* sB11 = stage_src(a.B, rowB, a.K, 1, 1, tid);

  // Prologue: tile0 A+B, tile1 B; leave tile1's B in flight (vmcnt(4)).
  stage_piece(sA00, 0, imgA0, 0, 0, wave);
  stage_piece(sA01, 0, imgA0, 0, 1, wave);
  stage_piece(sA10, 0, imgA0, 1, 0, wave);
  stage_piece(sA11, 0, imgA0, 1, 1, wave);
  stage_piece(sB00, 0, imgB0, 0, 0, wave);
  stage_piece(sB01, 0, imgB0, 0, 1, wave);
  stage_piece(sB10, 0, imgB0, 1, 0, wave);
  stage_piece(sB11, 0, imgB0, 1, 1, wave);
  if (NT > 1) {
    if (NIMG == 3) {  // A(1) rides the prologue; ph0 then stages t+2
      stage_piece(sA00, BK, imgA1, 0, 0, wave);
      stage_piece(sA01, BK, imgA1, 0, 1, wave);
      stage_piece(sA10, BK, imgA1, 1, 0, wave);
      stage_piece(sA11, BK, imgA1, 1, 1, wave);
    }
    stage_piece(sB00, BK, imgB1, 0, 0, wave);
    stage_piece(sB01, BK, imgB1, 0, 1, wave);
    stage_piece(sB10, BK, imgB1, 1, 0, wave);
    stage_piece(sB11, BK, imgB1, 1, 1, wave);
    asm volatile("s_waitcnt vmcnt(%0)" ::"i"(4 * (NIMG - 1)) : "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < NT; ++t) {
    const char* iA = imgA(t);
    const char* iB = (t & 1) ? imgB1 : imgB0;
    // B fragments: one read pass, live for the whole tile
    short8 bfr[4][2];
    #pragma unroll
    for (int fn = 0; fn < 4; ++fn)
      #pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[fn][ks] =
            frag_read(iB, wc * 64 + fn * 16 + (lane & 15),
                      ks * 64 + ((lane >> 4) & 3) * 16);

    #pragma unroll
    for (int ph = 0; ph < NPH; ++ph) {
      // A fragments for this phase's FMPP row-blocks
      short8 afr[FMPP][2];
      #pragma unroll
      for (int i = 0; i < FMPP; ++i)
        #pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          afr[i][ks] =
              frag_read(iA, wr * 128 + (FMPP * ph + i) * 16 + (lane & 15),
                        ks * 64 + ((lane >> 4) & 3) * 16);
      // Staged prefetch — every piece gets >= 1 full phase of flight
      // before its consumer's seam wait:
      //   first phase: BOTH A halves of t+1 (their buffer is idle
      //   during t); later phase(s): B halves of t+2 (B(t)'s buffer is
      //   free after the first phase's B-fragment reads, sealed by the
      //   phase-end barrier).
      if (ph == 0) {
        const int ta = t + NIMG - 1;  // NIMG=2: t+1; NIMG=3: t+2
        if (ta < NT) {
          char* img = imgA(ta);
          stage_piece(sA00, (long)ta * BK, img, 0, 0, wave);
          stage_piece(sA01, (long)ta * BK, img, 0, 1, wave);
          stage_piece(sA10, (long)ta * BK, img, 1, 0, wave);
          stage_piece(sA11, (long)ta * BK, img, 1, 1, wave);
        }
      } else if (ph == 1) {
        if (t + 2 < NT) {
          char* img = (t & 1) ? imgB1 : imgB0;
          stage_piece(sB00, (long)(t + 2) * BK, img, 0, 0, wave);
          stage_piece(sB01, (long)(t + 2) * BK, img, 0, 1, wave);
          if (NPH == 2) {
            stage_piece(sB10, (long)(t + 2) * BK, img, 1, 0, wave);
            stage_piece(sB11, (long)(t + 2) * BK, img, 1, 1, wave);
          }
        }
      } else if (ph == 2) {
        if (t + 2 < NT) {
          char* img = (t & 1) ? imgB1 : imgB0;
          stage_piece(sB10, (long)(t + 2) * BK, img, 1, 0, wave);
          stage_piece(sB11, (long)(t + 2) * BK, img, 1, 1, wave);
        }
      }
      // TWO_BARRIERS aligns every wave's MFMA cluster (matrix||matrix on
      // each SIMD); the single-barrier form lets a wave's reads/stages
      // overlap its SIMD partner's MFMA segment (matrix||memory pairing,
      // microarch §two-waves-per-SIMD item 5) — measured faster, kept as
      // the default; correctness needs only the end-of-phase barrier
      // (read-before-overwrite is sealed one phase ahead either way).
      if (TWO_BARRIERS) __builtin_amdgcn_s_barrier();
      if (!STATIC_PRIO) __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int i = 0; i < FMPP; ++i) {
        #pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          acc[FMPP * ph + i][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][0], bfr[fn][0], acc[FMPP * ph + i][fn], 0, 0, 0);
          acc[FMPP * ph + i][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][1], bfr[fn][1], acc[FMPP * ph + i][fn], 0, 0, 0);
        }
      }
      if (!STATIC_PRIO) __builtin_amdgcn_s_setprio(0);
      if (ph == NPH - 1) {
        // seam: tile t+1's halves must be LANDED in every wave before
        // anyone reads them after this barrier.  Steady state leaves the
        // two B halves of t+2 in flight (4 glds); at the tail drain all.
        if (t + 2 < NT)
          asm volatile("s_waitcnt vmcnt(%0)" ::"i"(4 * (NIMG - 1))
                       : "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue: LDS-repack each 16-row stripe to coalesced rows ----
  float* stripe = (float*)(smem + wave * 4096);  // [16][64] f32, wave-private
  const long cb = (long)bn * 256 + wc * 64;
  // full unroll: keeps acc[] indices compile-time so the accumulator
  // array stays in registers (guide §5.4 rule 20 — a runtime index sends
  // the whole array to scratch)
  #pragma unroll
  for (int fm = 0; fm < 8; ++fm) {
    #pragma unroll
    for (int fn = 0; fn < 4; ++fn)
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        stripe[(((lane >> 4) & 3) * 4 + rr) * 64 + fn * 16 + (lane & 15)] =
            acc[fm][fn][rr];
    const long rb = (long)bm * 256 + wr * 128 + fm * 16;
    #pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int row = pass * 4 + ((lane >> 4) & 3);
      const int col0 = (lane & 15) * 4;
      f32x4 v = *(const f32x4*)(stripe + row * 64 + col0);
      const long gr = rb + row;
      const long gc = cb + col0;
      if (a.epi == EPI_TANH_BIAS) {
        const f32x4 b4 = *(const f32x4*)(a.bias + gc);
        #pragma unroll
        for (int j = 0; j < 4; ++j) v[j] = fast_tanhf(v[j] + b4[j]);
      } else if (a.epi == EPI_DTANH) {
        const ushort4 h4 = *(const ushort4*)(a.aux + gr * a.N + gc);
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float h = bf2f(h4[j]);
          v[j] *= 1.f - h * h;
        }
      }
      if (a.CT != nullptr && a.epi != EPI_RAW)
        *(f32x4*)(stripe + row * 64 + col0) = v;  // transformed write-back
      if (a.epi == EPI_GRAD) {
        *(f32x4*)(a.grad + gr * a.N + gc) = v;
      } else {
        ushort4 o;
        #pragma unroll
        for (int j = 0; j < 4; ++j) o[j] = f2bf(v[j]);
        *(ushort4*)(a.C + gr * a.N + gc) = o;
      }
    }
    if (a.CT != nullptr) {
      // transposed dual-write: lane = column; its 16 transformed rows
      // are one conflict-free b32 column walk of the stripe, stored as
      // 32 contiguous bytes of CT[col][rb..rb+15].  Fused column sums
      // (the bias gradient of a dgrad output) ride along.
      float csum = 0.f;
      ushort4 p[4];
      #pragma unroll
      for (int qq = 0; qq < 4; ++qq)
        #pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          const float x = stripe[(qq * 4 + rr) * 64 + lane];
          csum += x;
          p[qq][rr] = f2bf(x);
        }
      unsigned short* ct = a.CT + (cb + lane) * a.ldt + rb;
      *(ushort4*)(ct) = p[0];
      *(ushort4*)(ct + 4) = p[1];
      *(ushort4*)(ct + 8) = p[2];
      *(ushort4*)(ct + 12) = p[3];
      if (a.sums != nullptr) atomicAdd(&a.sums[cb + lane], csum);
    }
  }
}

// ---------------------------------------------------------------------------
// 32x32x16-MFMA variant of the same pipeline: half the MFMA instruction
// count at a higher ceiling (2382 vs 2075 TF µbench), same glds staging,
// swizzle and phase/barrier structure.  Fragment map (guide §3):
// A/B lane l holds 8 bf16 at [i = l&31][k = (l>>5)*8..]; C/D lane l reg r
// covers [row = (r&3)+8*(r>>2)+4*(l>>5)][col = l&31].
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(16))) float f32x16;

template <int NPH>
__launch_bounds__(512, 2)
__global__ void bf16_mm256_kernel32(MM256Args a) {
  constexpr int FMPP = 4 / NPH;  // 32-row fm-blocks per phase
  __shared__ __attribute__((aligned(16))) char smem[LDS_BYTES];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2;
  const int wc = wave & 3;

  const long nwg = (long)gridDim.x;
  const long q = nwg >> 3, r8 = nwg & 7;
  const long xcd = blockIdx.x & 7, idx = blockIdx.x >> 3;
  const long sw = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q)
                  + idx;
  const int bm = (int)(sw / a.nbn);
  const int bn = (int)(sw % a.nbn);

  char* const imgA0 = smem;
  char* const imgA1 = smem + IMG;
  char* const imgB0 = smem + 2 * IMG;
  char* const imgB1 = smem + 3 * IMG;
  const long rowA = (long)bm * 256;
  const long rowB = (long)bn * 256;
  const int NT = (int)(a.K / BK);

  f32x16 acc[4][2] = {};

  const unsigned short* sA00 = stage_src(a.A, rowA, a.K, 0, 0, tid);
  const unsigned short* sA01 = stage_src(a.A, rowA, a.K, 0, 1, tid);
  const unsigned short* sA10 = stage_src(a.A, rowA, a.K, 1, 0, tid);
  const unsigned short* sA11 = stage_src(a.A, rowA, a.K, 1, 1, tid);
  const unsigned short* sB00 = stage_src(a.B, rowB, a.K, 0, 0, tid);
  const unsigned short* sB01 = stage_src(a.B, rowB, a.K, 0, 1, tid);
  const unsigned short* sB10 = stage_src(a.B, rowB, a.K, 1, 0, tid);
  const unsigned short* sB11 = stage_src(a.B, rowB, a.K, 1, 1, tid);

  stage_piece(sA00, 0, imgA0, 0, 0, wave);
  stage_piece(sA01, 0, imgA0, 0, 1, wave);
  stage_piece(sA10, 0, imgA0, 1, 0, wave);
  stage_piece(sA11, 0, imgA0, 1, 1, wave);
  stage_piece(sB00, 0, imgB0, 0, 0, wave);
  stage_piece(sB01, 0, imgB0, 0, 1, wave);
  stage_piece(sB10, 0, imgB0, 1, 0, wave);
  stage_piece(sB11, 0, imgB0, 1, 1, wave);
  if (NT > 1) {
    stage_piece(sB00, BK, imgB1, 0, 0, wave);
    stage_piece(sB01, BK, imgB1, 0, 1, wave);
    stage_piece(sB10, BK, imgB1, 1, 0, wave);
    stage_piece(sB11, BK, imgB1, 1, 1, wave);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < NT; ++t) {
    const char* iA = (t & 1) ? imgA1 : imgA0;
    const char* iB = (t & 1) ? imgB1 : imgB0;
    short8 bfr[2][4];  // 2 fn-blocks x 4 k-steps of 16
    #pragma unroll
    for (int fn = 0; fn < 2; ++fn)
      #pragma unroll
      for (int ks = 0; ks < 4; ++ks)
        bfr[fn][ks] =
            frag_read(iB, wc * 64 + fn * 32 + (lane & 31),
                      ks * 32 + (lane >> 5) * 16);

    #pragma unroll
    for (int ph = 0; ph < NPH; ++ph) {
      short8 afr[FMPP][4];
      #pragma unroll
      for (int i = 0; i < FMPP; ++i)
        #pragma unroll
        for (int ks = 0; ks < 4; ++ks)
          afr[i][ks] =
              frag_read(iA, wr * 128 + (FMPP * ph + i) * 32 + (lane & 31),
                        ks * 32 + (lane >> 5) * 16);
      if (ph == 0) {
        if (t + 1 < NT) {
          char* img = ((t + 1) & 1) ? imgA1 : imgA0;
          stage_piece(sA00, (long)(t + 1) * BK, img, 0, 0, wave);
          stage_piece(sA01, (long)(t + 1) * BK, img, 0, 1, wave);
          stage_piece(sA10, (long)(t + 1) * BK, img, 1, 0, wave);
          stage_piece(sA11, (long)(t + 1) * BK, img, 1, 1, wave);
        }
      } else if (ph == 1) {
        if (t + 2 < NT) {
          char* img = (t & 1) ? imgB1 : imgB0;
          stage_piece(sB00, (long)(t + 2) * BK, img, 0, 0, wave);
          stage_piece(sB01, (long)(t + 2) * BK, img, 0, 1, wave);
          if (NPH == 2) {
            stage_piece(sB10, (long)(t + 2) * BK, img, 1, 0, wave);
            stage_piece(sB11, (long)(t + 2) * BK, img, 1, 1, wave);
          }
        }
      } else if (ph == 2) {
        if (t + 2 < NT) {
          char* img = (t & 1) ? imgB1 : imgB0;
          stage_piece(sB10, (long)(t + 2) * BK, img, 1, 0, wave);
          stage_piece(sB11, (long)(t + 2) * BK, img, 1, 1, wave);
        }
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int i = 0; i < FMPP; ++i) {
        #pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          #pragma unroll
          for (int ks = 0; ks < 4; ++ks)
            acc[FMPP * ph + i][fn] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                afr[i][ks], bfr[fn][ks], acc[FMPP * ph + i][fn], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
      if (ph == NPH - 1) {
        if (t + 2 < NT)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: 32-row stripes through LDS (wave-private 8 KB regions)
  float* stripe = (float*)(smem + wave * 8192);  // [32][64] f32
  const long cb = (long)bn * 256 + wc * 64;
  #pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
    #pragma unroll
    for (int fn = 0; fn < 2; ++fn)
      #pragma unroll
      for (int rg = 0; rg < 16; ++rg)
        stripe[((rg & 3) + 8 * (rg >> 2) + 4 * (lane >> 5)) * 64 + fn * 32 +
               (lane & 31)] = acc[fm][fn][rg];
    const long rb = (long)bm * 256 + wr * 128 + fm * 32;
    #pragma unroll
    for (int pass = 0; pass < 8; ++pass) {
      const int row = pass * 4 + ((lane >> 4) & 3);
      const int col0 = (lane & 15) * 4;
      f32x4 v = *(const f32x4*)(stripe + row * 64 + col0);
      const long gr = rb + row;
      const long gc = cb + col0;
      if (a.epi == EPI_TANH_BIAS) {
        const f32x4 b4 = *(const f32x4*)(a.bias + gc);
        #pragma unroll
        for (int j = 0; j < 4; ++j) v[j] = fast_tanhf(v[j] + b4[j]);
      } else if (a.epi == EPI_DTANH) {
        const ushort4 h4 = *(const ushort4*)(a.aux + gr * a.N + gc);
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float h = bf2f(h4[j]);
          v[j] *= 1.f - h * h;
        }
      }
      if (a.epi == EPI_GRAD) {
        *(f32x4*)(a.grad + gr * a.N + gc) = v;
      } else {
        ushort4 o;
        #pragma unroll
        for (int j = 0; j < 4; ++j) o[j] = f2bf(v[j]);
        *(ushort4*)(a.C + gr * a.N + gc) = o;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// bf16_mm_small: guarded 64x64-tile GEMM for ragged heads shapes.
//   epi 0: C[M][ldc] bf16          epi 4: heads split (C=pdflat [M][N-1],
//   epi 2: dtanh via aux [M][N]            C2=v [M]; no activation)
//   epi 5: grad split-row (f32 grad: rows < srow -> g1 + m*N + n,
//          row == srow -> g2 + n; rows > srow skipped)
// Requires K % 32 == 0 and 16-B-aligned rows (callers pad K).
// ---------------------------------------------------------------------------

struct MMSmallArgs {
  const unsigned short* A;  // [M][K]
  const unsigned short* B;  // [N][K]
  unsigned short* C;
  unsigned short* C2;       // epi 4: value column
  const unsigned short* aux;
  const float* bias;        // optional [Nreal] f32 (epi 0/4)
  float* g1;
  float* g2;
  long M, N, K, Mreal, Nreal, ldc;
  int nbn, epi, srow, nsk;  // nsk: split-K slices (grid.y); epi-5 grads
                            // become atomicAdd when nsk > 1
};

constexpr int SM_STRIDE = 80;  // padded row stride (bytes) of the LDS images

__launch_bounds__(256)
__global__ void bf16_mm_small_kernel(MMSmallArgs a) {
  __shared__ __attribute__((aligned(16))) char smem[2 * 2 * 64 * SM_STRIDE];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int bm = blockIdx.x / a.nbn;
  const int bn = blockIdx.x % a.nbn;
  const long r0 = (long)bm * 64;   // A rows
  const long c0 = (long)bn * 64;   // B rows (= C cols)
  const int row = tid >> 2;
  const int piece = tid & 3;
  const int NC_all = (int)(a.K / 32);
  const int per_sk = (NC_all + a.nsk - 1) / a.nsk;
  const int kc0 = (int)blockIdx.y * per_sk;
  const int kc1 = min(kc0 + per_sk, NC_all);

  char* const bufA0 = smem;
  char* const bufB0 = smem + 64 * SM_STRIDE;
  char* const bufA1 = smem + 2 * 64 * SM_STRIDE;
  char* const bufB1 = smem + 3 * 64 * SM_STRIDE;

  auto load_guarded = [&](const unsigned short* g, long grow, long rows,
                          long k) -> short8 {
    short8 z = {};
    if (grow < rows) z = *(const short8*)(g + grow * a.K + k);
    return z;
  };

  f32x4 acc[4] = {};
  short8 ra = load_guarded(a.A, r0 + row, a.Mreal, (long)kc0 * 32 + piece * 8);
  short8 rb = load_guarded(a.B, c0 + row, a.Nreal, (long)kc0 * 32 + piece * 8);
  for (int kc = kc0; kc < kc1; ++kc) {
    char* wA = (kc & 1) ? bufA1 : bufA0;
    char* wB = (kc & 1) ? bufB1 : bufB0;
    *(short8*)(wA + row * SM_STRIDE + piece * 16) = ra;
    *(short8*)(wB + row * SM_STRIDE + piece * 16) = rb;
    if (kc + 1 < kc1) {
      ra = load_guarded(a.A, r0 + row, a.Mreal, (long)(kc + 1) * 32 + piece * 8);
      rb = load_guarded(a.B, c0 + row, a.Nreal, (long)(kc + 1) * 32 + piece * 8);
    }
    __syncthreads();
    const char* iA = wA;
    const char* iB = wB;
    short8 bf = *(const short8*)(iB + (wave * 16 + (lane & 15)) * SM_STRIDE +
                                 ((lane >> 4) & 3) * 16);
    #pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
      short8 af = *(const short8*)(iA + (fm * 16 + (lane & 15)) * SM_STRIDE +
                                   ((lane >> 4) & 3) * 16);
      acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[fm], 0, 0, 0);
    }
    __syncthreads();
  }

  // guarded scalar epilogue
  #pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
    #pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const long m = r0 + fm * 16 + ((lane >> 4) & 3) * 4 + rr;
      const long n = c0 + wave * 16 + (lane & 15);
      if (m >= a.Mreal || n >= a.Nreal) continue;
      float v = acc[fm][rr];
      if (a.bias != nullptr) v += a.bias[n];
      if (a.epi == EPI_DTANH) {
        const float h = bf2f(a.aux[m * a.Nreal + n]);
        v *= 1.f - h * h;
      }
      if (a.epi == 4) {  // heads split: col < N-1 -> pdflat, col N-1 -> v
        if (n < a.Nreal - 1)
          a.C[m * a.ldc + n] = f2bf(v);
        else
          a.C2[m] = f2bf(v);
      } else if (a.epi == 5) {  // grad split-row
        if (a.nsk > 1) {
          if (m < a.srow)
            atomicAdd(&a.g1[m * a.Nreal + n], v);
          else if (m == a.srow)
            atomicAdd(&a.g2[n], v);
        } else if (m < a.srow) {
          a.g1[m * a.Nreal + n] = v;
        } else if (m == a.srow) {
          a.g2[n] = v;
        }
      } else {
        a.C[m * a.ldc + n] = f2bf(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// bf16_transpose (+ fused column sums): in [R][ld_in] -> out [C][ld_out]
// ---------------------------------------------------------------------------

struct TransArgs {
  const unsigned short* in;
  unsigned short* out;
  float* sums;  // optional [C] f32, += column sums of `in`
  long R, C, ld_in, ld_out;
  int nbc;
};

__launch_bounds__(256)
__global__ void bf16_transpose_kernel(TransArgs a) {
  __shared__ __attribute__((aligned(16))) unsigned short tile[64][72];
  const int tid = threadIdx.x;
  const long br = (long)(blockIdx.x / a.nbc) * 64;
  const long bc = (long)(blockIdx.x % a.nbc) * 64;
  const int row = tid >> 2;           // 0..63
  const int seg = tid & 3;            // 16-element column segment

  // load [64][64] tile (2 x short8 per thread), scatter-transpose into LDS
  #pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int c0 = seg * 16 + p * 8;
    short8 v = {};
    if (br + row < a.R) {
      if (bc + c0 + 7 < a.C) {
        v = *(const short8*)(a.in + (br + row) * a.ld_in + bc + c0);
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          if (bc + c0 + j < a.C)
            v[j] = (short)a.in[(br + row) * a.ld_in + bc + c0 + j];
      }
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) tile[c0 + j][row] = (unsigned short)v[j];
  }
  __syncthreads();

  // write out rows of the transpose (out row = column of `in`)
  const int oc = row;  // output row index within tile (= in column)
  float csum = 0.f;
  #pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int r0 = seg * 16 + p * 8;
    short8 v = *(const short8*)(&tile[oc][r0]);
    if (a.out != nullptr && bc + oc < a.C) {
      if (br + r0 + 7 < a.R) {
        *(short8*)(a.out + (bc + oc) * a.ld_out + br + r0) = v;
      } else {
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          if (br + r0 + j < a.R)
            a.out[(bc + oc) * a.ld_out + br + r0 + j] = (unsigned short)v[j];
      }
    }
    if (a.sums != nullptr) {
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        if (br + r0 + j < a.R) csum += bf2f((unsigned short)v[j]);
    }
  }
  if (a.sums != nullptr && bc + oc < a.C) {
    // 4 threads (seg 0..3) share one output column; lane-group reduce
    csum += __shfl_down(csum, 1, 4);
    csum += __shfl_down(csum, 2, 4);
    if (seg == 0) atomicAdd(&a.sums[bc + oc], csum);
  }
}

// ---------------------------------------------------------------------------
// gauss_gh_wide: wave-per-row PPO loss gradient, wide policies (A > 32)
// gh[b][j] = dtotal/d pdflat[b][j] (j < 2A), gh[b][2A] = dtotal/d v[b]
// ---------------------------------------------------------------------------

struct GhWideArgs {
  const unsigned short* pdflat;  // [B][2A] bf16 (pi outputs)
  const float* oldflat;          // [B][2A] f32 (recorded)
  const unsigned short* v;       // [B] bf16
  const float* oldv;             // [B]
  const float* act;              // [B][A]
  const float* adv;              // [B]
  const float* etr;              // [B]
  unsigned short* gh;            // [B][ldgh] bf16 (zero-padded tail)
  const float* clip_dev;         // optional device clip override
  long B, ldgh;
  int A;
  float clip, entcoeff, vcoeff;
};

__launch_bounds__(256)
__global__ void gauss_gh_wide_kernel(GhWideArgs a) {
  const float clip = (a.clip_dev != nullptr) ? a.clip_dev[0] : a.clip;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int P = 2 * a.A;
  const int64_t wid = (int64_t)blockIdx.x * 4 + wave;
  const int64_t waves_total = (int64_t)gridDim.x * 4;
  const int64_t per = (a.B + waves_total - 1) / waves_total;
  const int64_t rb0 = wid * per;
  const int64_t rb1 = rb0 + per < a.B ? rb0 + per : a.B;

  for (int64_t b = rb0; b < rb1; ++b) {
    const unsigned short* mu = a.pdflat + b * P;
    const float* mo = a.oldflat + b * P;
    const float* aj = a.act + b * a.A;
    float lp = 0.f, lo = 0.f;
    for (int j = lane; j < a.A; j += WAVE) {
      const float ls = bf2f(mu[a.A + j]);
      const float zp = (aj[j] - bf2f(mu[j])) * __expf(-ls);
      lp += -0.5f * zp * zp - ls;
      const float lso = mo[a.A + j];
      const float zo = (aj[j] - mo[j]) * __expf(-lso);
      lo += -0.5f * zo * zo - lso;
    }
    const float c = 0.5f * PPO_LOG_2PI * a.A;
    GaussRow row;
    row.logp_pi = __shfl(wave_reduce_sum(lp), 0, WAVE) - c;
    row.logp_old = __shfl(wave_reduce_sum(lo), 0, WAVE) - c;
    row.ent = 0.f;  // unused by ppo_row_grads
    const PPORowGrads g =
        ppo_row_grads(row, bf2f(a.v[b]), a.oldv[b], a.adv[b], a.etr[b], a.B,
                      clip, a.entcoeff, a.vcoeff, 1.f);
    unsigned short* out = a.gh + b * a.ldgh;
    for (int j = lane; j < a.A; j += WAVE) {
      const float ls = bf2f(mu[a.A + j]);
      const float inv_s = __expf(-ls);
      const float z = (aj[j] - bf2f(mu[j])) * inv_s;
      out[j] = f2bf(g.g_logp * z * inv_s);
      out[a.A + j] = f2bf(g.g_logp * (z * z - 1.f) + g.g_ent);
    }
    if (lane == 0) out[P] = f2bf(g.g_v);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static const unsigned short* bf_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr<at::BFloat16>());
}
static unsigned short* bf_ptr_mut(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr<at::BFloat16>());
}

void bf16_mm256(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                int64_t epi, torch::Tensor bias, torch::Tensor aux,
                torch::Tensor grad, int64_t grad_off, torch::Tensor CT,
                int64_t ldt, torch::Tensor sums, int64_t sums_off) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16 && A.dim() == 2);
  TORCH_CHECK(B.dtype() == torch::kBFloat16 && B.dim() == 2);
  const long M = A.size(0), K = A.size(1), N = B.size(0);
  TORCH_CHECK(B.size(1) == K);
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0,
              "bf16_mm256 needs M,N % 256 == 0 and K % 64 == 0");
  TORCH_CHECK(A.is_contiguous() && B.is_contiguous());
  MM256Args a{};
  a.A = bf_ptr(A);
  a.B = bf_ptr(B);
  a.M = M;
  a.N = N;
  a.K = K;
  a.nbn = (int)(N / 256);
  a.epi = (int)epi;
  if (epi == EPI_GRAD) {
    TORCH_CHECK(grad.numel() >= grad_off + M * N);
    a.grad = grad.data_ptr<float>() + grad_off;
  } else {
    TORCH_CHECK(C.dtype() == torch::kBFloat16 && C.is_contiguous() &&
                C.numel() == M * N);
    a.C = bf_ptr_mut(C);
  }
  if (epi == EPI_TANH_BIAS) {
    TORCH_CHECK(bias.numel() == N && bias.dtype() == torch::kFloat32);
    a.bias = bias.data_ptr<float>();
  }
  if (epi == EPI_DTANH) {
    TORCH_CHECK(aux.dtype() == torch::kBFloat16 && aux.numel() == M * N);
    a.aux = bf_ptr(aux);
  }
  if (CT.numel() > 0) {
    TORCH_CHECK(CT.dtype() == torch::kBFloat16 && CT.numel() >= M * N);
    a.CT = bf_ptr_mut(CT);
    a.ldt = ldt > 0 ? ldt : M;
    TORCH_CHECK(a.ldt % 2 == 0);
    if (sums.numel() > 0) {
      TORCH_CHECK(sums.dtype() == torch::kFloat32 &&
                  sums.numel() >= sums_off + N);
      a.sums = sums.data_ptr<float>() + sums_off;
    }
  }
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const long grid = (M / 256) * (N / 256);
  static const bool two_bar = []() {
    const char* e = getenv("DPPO_MM256_2BAR");
    return e != nullptr && e[0] == '1';
  }();
  static const int nph = []() {
    const char* e = getenv("DPPO_MM256_PH");
    return e ? atoi(e) : 2;  // 2 phases/tile measured fastest
  }();
  static const int mfma = []() {
    const char* e = getenv("DPPO_MM256_MFMA");
    // 16x16x32 measured FASTER than 32x32x16 here (1160-1199 vs
    // 1020-1047 TF/s) despite the larger µbench ceiling — the 32x32
    // form's 4-deep chained accumulator + 16-reg C/D operands lose more
    // than the halved instruction count buys.  Kept for ablation.
    return e ? atoi(e) : 16;
  }();
  const bool want32 = (mfma == 32) && CT.numel() == 0;  // 32-variant has
                                                        // no dual-write
  if (two_bar)
    hipLaunchKernelGGL((bf16_mm256_kernel<4, true>), dim3((unsigned)grid),
                       dim3(512), 0, stream, a);
  else if (want32 && nph == 2)
    hipLaunchKernelGGL((bf16_mm256_kernel32<2>), dim3((unsigned)grid),
                       dim3(512), 0, stream, a);
  else if (want32)
    hipLaunchKernelGGL((bf16_mm256_kernel32<4>), dim3((unsigned)grid),
                       dim3(512), 0, stream, a);
  else if (nph == 2) {
    static const bool sprio = []() {
      const char* e = getenv("DPPO_MM256_SPRIO");
      return e != nullptr && e[0] == '1';
    }();
    static const int nimg = []() {
      const char* e = getenv("DPPO_MM256_IMG");
      return e ? atoi(e) : 2;  // 3 = third A image (UNVALIDATED on
                               // hardware this round — ROADMAP #1a)
    }();
    if (nimg == 3)
      hipLaunchKernelGGL((bf16_mm256_kernel<2, false, false, 3>),
                         dim3((unsigned)grid), dim3(512), 0, stream, a);
    else if (sprio)
      hipLaunchKernelGGL((bf16_mm256_kernel<2, false, true>),
                         dim3((unsigned)grid), dim3(512), 0, stream, a);
    else
      hipLaunchKernelGGL((bf16_mm256_kernel<2, false>), dim3((unsigned)grid),
                         dim3(512), 0, stream, a);
  } else
    hipLaunchKernelGGL((bf16_mm256_kernel<4, false>), dim3((unsigned)grid),
                       dim3(512), 0, stream, a);
}

void bf16_mm_small(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                   torch::Tensor C2, torch::Tensor aux, torch::Tensor grad,
                   int64_t g1_off, int64_t g2_off, int64_t srow, int64_t epi,
                   int64_t m_real, int64_t n_real, int64_t ldc,
                   torch::Tensor bias) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16 && A.dim() == 2);
  const bool splitk_ok = (epi == 5);
  const long K = A.size(1);
  TORCH_CHECK(B.size(1) == K && K % 32 == 0 && K % 8 == 0);
  MMSmallArgs a{};
  a.A = bf_ptr(A);
  a.B = bf_ptr(B);
  a.K = K;
  a.Mreal = m_real > 0 ? m_real : A.size(0);
  a.Nreal = n_real > 0 ? n_real : B.size(0);
  a.M = (a.Mreal + 63) & ~63L;
  a.N = (a.Nreal + 63) & ~63L;
  a.ldc = ldc > 0 ? ldc : a.Nreal;
  a.nbn = (int)(a.N / 64);
  a.epi = (int)epi;
  a.srow = (int)srow;
  if (bias.numel() > 0) {
    TORCH_CHECK(bias.dtype() == torch::kFloat32 && bias.numel() >= a.Nreal);
    a.bias = bias.data_ptr<float>();
  }
  if (epi == 5) {
    a.g1 = grad.data_ptr<float>() + g1_off;
    a.g2 = grad.data_ptr<float>() + g2_off;
    a.Nreal = n_real;  // grads indexed by Nreal row stride
  } else {
    a.C = bf_ptr_mut(C);
    if (epi == 4) a.C2 = bf_ptr_mut(C2);
    if (epi == EPI_DTANH) a.aux = bf_ptr(aux);
  }
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const long grid = (a.M / 64) * (a.N / 64);
  // split K across grid.y until the launch has ~4 blocks/CU (grad epilogue
  // accumulates with atomics, so only epi 5 splits)
  long nsk = 1;
  if (splitk_ok) {
    while (grid * nsk < 1024 && nsk < 32 && (a.K / 32) / (nsk * 2) >= 8)
      nsk *= 2;
  }
  a.nsk = (int)nsk;
  hipLaunchKernelGGL(bf16_mm_small_kernel, dim3((unsigned)grid, (unsigned)nsk),
                     dim3(256), 0, stream, a);
}

void bf16_transpose(torch::Tensor in, torch::Tensor out, torch::Tensor sums,
                    int64_t sums_off, int64_t R, int64_t C, int64_t ld_in,
                    int64_t ld_out) {
  TORCH_CHECK(in.is_cuda() && in.dtype() == torch::kBFloat16);
  TransArgs a{};
  a.in = bf_ptr(in);
  a.out = out.numel() > 0 ? bf_ptr_mut(out) : nullptr;
  a.sums = sums.numel() > 0 ? sums.data_ptr<float>() + sums_off : nullptr;
  a.R = R;
  a.C = C;
  a.ld_in = ld_in > 0 ? ld_in : C;
  a.ld_out = ld_out > 0 ? ld_out : R;
  const long nbr = (R + 63) / 64, nbc = (C + 63) / 64;
  a.nbc = (int)nbc;
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(bf16_transpose_kernel, dim3((unsigned)(nbr * nbc)),
                     dim3(256), 0, stream, a);
}

void gauss_gh_wide(torch::Tensor pdflat_bf, torch::Tensor oldflat,
                   torch::Tensor v_bf, torch::Tensor oldv, torch::Tensor act,
                   torch::Tensor adv, torch::Tensor etr, torch::Tensor gh,
                   torch::Tensor clip_dev, double clip, double entcoeff,
                   double vcoeff) {
  TORCH_CHECK(pdflat_bf.is_cuda() && pdflat_bf.dtype() == torch::kBFloat16);
  TORCH_CHECK(gh.dtype() == torch::kBFloat16 && gh.dim() == 2);
  const long B = pdflat_bf.size(0);
  const int A = (int)(pdflat_bf.size(1) / 2);
  TORCH_CHECK(gh.size(0) == B && gh.size(1) >= 2 * A + 1);
  GhWideArgs a{};
  a.pdflat = bf_ptr(pdflat_bf);
  a.oldflat = oldflat.data_ptr<float>();
  a.v = bf_ptr(v_bf);
  a.oldv = oldv.data_ptr<float>();
  a.act = act.data_ptr<float>();
  a.adv = adv.data_ptr<float>();
  a.etr = etr.data_ptr<float>();
  a.gh = bf_ptr_mut(gh);
  a.clip_dev = clip_dev.numel() > 0 ? clip_dev.data_ptr<float>() : nullptr;
  a.B = B;
  a.ldgh = gh.size(1);
  a.A = A;
  a.clip = (float)clip;
  a.entcoeff = (float)entcoeff;
  a.vcoeff = (float)vcoeff;
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(gauss_gh_wide_kernel, dim3(1024), dim3(256), 0, stream, a);
}
