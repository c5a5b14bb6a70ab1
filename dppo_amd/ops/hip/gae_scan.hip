// Segmented GAE reverse scan + fused whitening statistics (gfx950).
//
// Device form of the reference's host-side numpy loop (reference
// Worker.py:84-92): one thread owns one env column of the [T, E] rollout
// and runs the affine reverse recurrence serially over T while the wave's
// 64 adjacent threads stream 64 adjacent env columns — every global access
// is a fully-coalesced 256-B wave transaction at each timestep.  The scan
// also accumulates sum / sum-of-squares of the advantages (one double
// atomic per wave) so whitening (Worker.py:92) needs no extra pass over
// the data; a finalize kernel turns the accumulators into (mean, 1/(std+eps))
// and an elementwise kernel applies them.  E >= a few hundred keeps the
// chip busy; per-env serialization over T is the memory-optimal shape here
// (3 reads + 2 writes per cell, nothing recomputed).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

__global__ void gae_scan_kernel(
    const float* __restrict__ rewards,   // [T, E]
    const float* __restrict__ values,    // [T, E]
    const float* __restrict__ dones,     // [T, E] (0/1)
    const float* __restrict__ boot,      // [E]
    float* __restrict__ adv,             // [T, E]
    float* __restrict__ etr,             // [T, E]
    double* __restrict__ stats,          // [2] {sum, sumsq}
    int64_t T, int64_t E, float gamma, float lam) {
  const int64_t e = gidx();
  const bool active = e < E;
  float lastg = 0.f, nextv = 0.f, sum = 0.f, sumsq = 0.f;
  if (active) nextv = boot[e];
  for (int64_t t = T - 1; t >= 0; --t) {
    if (active) {
      const int64_t i = t * E + e;
      const float nonterm = 1.f - dones[i];
      const float vt = values[i];
      const float delta = rewards[i] + gamma * nextv * nonterm - vt;
      lastg = delta + gamma * lam * nonterm * lastg;
      adv[i] = lastg;
      etr[i] = lastg + vt;
      nextv = vt;
      sum += lastg;
      sumsq += lastg * lastg;
    }
  }
  wave_atomic_add(&stats[0], sum);
  wave_atomic_add(&stats[1], sumsq);
}

__global__ void gae_finalize_kernel(const double* __restrict__ stats,
                                    float* __restrict__ mean_inv,  // [2]
                                    int64_t n, float eps) {
  const double mean = stats[0] / static_cast<double>(n);
  double var = stats[1] / static_cast<double>(n) - mean * mean;
  if (var < 0.0) var = 0.0;
  mean_inv[0] = static_cast<float>(mean);
  // reference Worker.py:92 divides by std; eps guards the constant case
  mean_inv[1] = static_cast<float>(1.0 / (sqrt(var) + static_cast<double>(eps)));
}

__global__ void gae_whiten_kernel(float* __restrict__ adv,
                                  const float* __restrict__ mean_inv,
                                  int64_t n) {
  const float mean = mean_inv[0];
  const float inv = mean_inv[1];
  const int64_t n4 = n / 4;
  float4* a4 = reinterpret_cast<float4*>(adv);
  for (int64_t i = gidx(); i < n4; i += gstride()) {
    float4 x = a4[i];
    x.x = (x.x - mean) * inv;
    x.y = (x.y - mean) * inv;
    x.z = (x.z - mean) * inv;
    x.w = (x.w - mean) * inv;
    a4[i] = x;
  }
  // tail
  for (int64_t i = n4 * 4 + gidx(); i < n; i += gstride()) {
    adv[i] = (adv[i] - mean) * inv;
  }
}

}  // namespace

std::vector<torch::Tensor> gae_scan(torch::Tensor rewards, torch::Tensor values,
                                    torch::Tensor dones, torch::Tensor boot,
                                    double gamma, double lam, bool whiten,
                                    double eps, torch::Tensor adv_out,
                                    torch::Tensor etr_out) {
  TORCH_CHECK(rewards.is_cuda() && rewards.dtype() == torch::kFloat32,
              "gae_scan: rewards must be fp32 CUDA");
  TORCH_CHECK(rewards.dim() == 2, "gae_scan: rewards must be [T, E]");
  const int64_t T = rewards.size(0), E = rewards.size(1);
  TORCH_CHECK(values.sizes() == rewards.sizes() && dones.sizes() == rewards.sizes());
  TORCH_CHECK(boot.numel() == E);

  auto adv = (adv_out.numel() == rewards.numel())
                 ? adv_out.view_as(rewards)
                 : torch::empty_like(rewards);
  auto etr = (etr_out.numel() == rewards.numel())
                 ? etr_out.view_as(rewards)
                 : torch::empty_like(rewards);
  auto stats = torch::zeros({2}, rewards.options().dtype(torch::kFloat64));
  auto mean_inv = torch::empty({2}, rewards.options());

  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  const int block = 256;
  const int grid = static_cast<int>((E + block - 1) / block);
  hipLaunchKernelGGL(gae_scan_kernel, dim3(grid), dim3(block), 0, stream,
                     rewards.data_ptr<float>(), values.data_ptr<float>(),
                     dones.data_ptr<float>(), boot.data_ptr<float>(),
                     adv.data_ptr<float>(), etr.data_ptr<float>(),
                     stats.data_ptr<double>(), T, E, (float)gamma, (float)lam);
  if (whiten) {
    const int64_t n = T * E;
    hipLaunchKernelGGL(gae_finalize_kernel, dim3(1), dim3(1), 0, stream,
                       stats.data_ptr<double>(), mean_inv.data_ptr<float>(), n,
                       (float)eps);
    hipLaunchKernelGGL(gae_whiten_kernel, dim3(elementwise_grid(n / 4, 256)),
                       dim3(256), 0, stream, adv.data_ptr<float>(),
                       mean_inv.data_ptr<float>(), n);
  }
  return {adv, etr};
}
