// Fused flat-bucket Adam (gfx950).
//
// The reference applies Adam through TF's per-variable apply_gradients on
// the Chief (reference PPO.py:20,53 — K10 in SURVEY.md §2.4).  Here the
// whole parameter set lives in ONE flat fp32 buffer (parallel/comm.py
// FlatBuffers), so the update is a single vectorized elementwise kernel:
// float4 loads/stores (16 B/lane — the coalescing sweet spot, guideline
// 13), grid-stride, with torch.optim.Adam's exact update rule (bias
// correction on both moments, eps added after the sqrt) so replicas
// trained with this kernel stay bit-identical to each other.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

__global__ void adam_kernel(float* __restrict__ p, const float* __restrict__ g,
                            float* __restrict__ m, float* __restrict__ v,
                            int64_t n, float lr, float beta1, float beta2,
                            float eps, float bc1, float bc2) {
  const int64_t n4 = n / 4;
  float4* p4 = reinterpret_cast<float4*>(p);
  const float4* g4 = reinterpret_cast<const float4*>(g);
  float4* m4 = reinterpret_cast<float4*>(m);
  float4* v4 = reinterpret_cast<float4*>(v);
  for (int64_t i = gidx(); i < n4; i += gstride()) {
    float4 pp = p4[i], gg = g4[i], mm = m4[i], vv = v4[i];
    #define ADAM1(c)                                                    \
      mm.c = beta1 * mm.c + (1.f - beta1) * gg.c;                       \
      vv.c = beta2 * vv.c + (1.f - beta2) * gg.c * gg.c;                \
      pp.c -= lr * (mm.c / bc1) / (sqrtf(vv.c / bc2) + eps);
    ADAM1(x) ADAM1(y) ADAM1(z) ADAM1(w)
    #undef ADAM1
    p4[i] = pp;
    m4[i] = mm;
    v4[i] = vv;
  }
  for (int64_t i = n4 * 4 + gidx(); i < n; i += gstride()) {
    float mm = beta1 * m[i] + (1.f - beta1) * g[i];
    float vv = beta2 * v[i] + (1.f - beta2) * g[i] * g[i];
    m[i] = mm;
    v[i] = vv;
    p[i] -= lr * (mm / bc1) / (sqrtf(vv / bc2) + eps);
  }
}

__global__ void adam_prep_kernel(int* __restrict__ step,
                                 const float* __restrict__ lr_dev,
                                 float* __restrict__ coef,  // [3] lr,bc1,bc2
                                 float beta1, float beta2) {
  const int t = ++step[0];
  coef[0] = lr_dev[0];
  coef[1] = 1.f - powf(beta1, (float)t);
  coef[2] = 1.f - powf(beta2, (float)t);
}

__global__ void adam_dev_kernel(float* __restrict__ p,
                                const float* __restrict__ g,
                                float* __restrict__ m, float* __restrict__ v,
                                const float* __restrict__ coef, int64_t n,
                                float beta1, float beta2, float eps) {
  const float lr = coef[0], bc1 = coef[1], bc2 = coef[2];
  const int64_t n4 = n / 4;
  float4* p4 = reinterpret_cast<float4*>(p);
  const float4* g4 = reinterpret_cast<const float4*>(g);
  float4* m4 = reinterpret_cast<float4*>(m);
  float4* v4 = reinterpret_cast<float4*>(v);
  for (int64_t i = gidx(); i < n4; i += gstride()) {
    float4 pp = p4[i], gg = g4[i], mm = m4[i], vv = v4[i];
    #define ADAM1(c)                                                    \
      mm.c = beta1 * mm.c + (1.f - beta1) * gg.c;                       \
      vv.c = beta2 * vv.c + (1.f - beta2) * gg.c * gg.c;                \
      pp.c -= lr * (mm.c / bc1) / (sqrtf(vv.c / bc2) + eps);
    ADAM1(x) ADAM1(y) ADAM1(z) ADAM1(w)
    #undef ADAM1
    p4[i] = pp;
    m4[i] = mm;
    v4[i] = vv;
  }
  for (int64_t i = n4 * 4 + gidx(); i < n; i += gstride()) {
    float mm = beta1 * m[i] + (1.f - beta1) * g[i];
    float vv = beta2 * v[i] + (1.f - beta2) * g[i] * g[i];
    m[i] = mm;
    v[i] = vv;
    p[i] -= lr * (mm / bc1) / (sqrtf(vv / bc2) + eps);
  }
}

}  // namespace

// Graph-replayable Adam: the step counter, learning rate and bias
// corrections live in device memory, so a captured update replays with
// fresh values each round (kernel args are frozen under replay).
void adam_step_dev(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                   torch::Tensor v, torch::Tensor step_dev,
                   torch::Tensor lr_dev, torch::Tensor coef, double beta1,
                   double beta2, double eps) {
  TORCH_CHECK(p.is_cuda() && p.dtype() == torch::kFloat32 && p.is_contiguous());
  const int64_t n = p.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  TORCH_CHECK(step_dev.dtype() == torch::kInt32 && coef.numel() >= 3);
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(adam_prep_kernel, dim3(1), dim3(1), 0, stream,
                     step_dev.data_ptr<int>(), lr_dev.data_ptr<float>(),
                     coef.data_ptr<float>(), (float)beta1, (float)beta2);
  hipLaunchKernelGGL(adam_dev_kernel, dim3(elementwise_grid(n / 4 + 1, 256)),
                     dim3(256), 0, stream, p.data_ptr<float>(),
                     g.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), coef.data_ptr<float>(), n,
                     (float)beta1, (float)beta2, (float)eps);
}

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, int64_t step, double lr, double beta1,
               double beta2, double eps) {
  TORCH_CHECK(p.is_cuda() && p.dtype() == torch::kFloat32 && p.is_contiguous());
  const int64_t n = p.numel();
  TORCH_CHECK(g.numel() == n && m.numel() == n && v.numel() == n);
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  hipStream_t stream = c10::hip::getCurrentHIPStream().stream();
  hipLaunchKernelGGL(adam_kernel, dim3(elementwise_grid(n / 4 + 1, 256)),
                     dim3(256), 0, stream, p.data_ptr<float>(),
                     g.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), n, (float)lr, (float)beta1,
                     (float)beta2, (float)eps, bc1, bc2);
}
