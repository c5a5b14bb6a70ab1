// Shared per-row DiagGaussian PPO loss math (forward quantities and
// analytic gradients), used by both the standalone fused loss kernels
// (ppo_loss.hip) and the fused MLP backward (mlp_train.hip).
//
// Implements exactly reference PPO.py:29-40 + Others/distributions.py:195-203.
#pragma once

#include "common.h"

constexpr float PPO_LOG_2PI = 1.8378770664093453f;

struct GaussRow {
  float logp_pi, logp_old, ent;
};

// Per-sample log-probs + entropy for one row (loops over action dim).
// pdflat layout: [mean(A) | logstd(A)].
DEV_INLINE GaussRow ppo_gauss_row(const float* __restrict__ pdpi,
                                  const float* __restrict__ pdold,
                                  const float* __restrict__ act,
                                  int64_t b, int A) {
  const float* mu_pi = pdpi + (int64_t)b * 2 * A;
  const float* ls_pi = mu_pi + A;
  const float* mu_old = pdold + (int64_t)b * 2 * A;
  const float* ls_old = mu_old + A;
  const float* a = act + (int64_t)b * A;
  float lp = 0.f, lo = 0.f, ent = 0.f;
  for (int j = 0; j < A; ++j) {
    const float aj = a[j];
    const float lsp = ls_pi[j];
    const float zp = (aj - mu_pi[j]) * __expf(-lsp);
    lp += -0.5f * zp * zp - lsp;
    const float lso = ls_old[j];
    const float zo = (aj - mu_old[j]) * __expf(-lso);
    lo += -0.5f * zo * zo - lso;
    ent += lsp;
  }
  const float c = 0.5f * PPO_LOG_2PI * A;
  GaussRow r;
  r.logp_pi = lp - c;
  r.logp_old = lo - c;
  r.ent = ent + 0.5f * (PPO_LOG_2PI + 1.f) * A;
  return r;
}

struct PPORowGrads {
  float g_logp;  // dL/d logp_pi for this row (policy term)
  float g_ent;   // dL/d logstd_j entropy contribution (same for all j)
  float g_v;     // dL/d vpred for this row
};

// Per-row gradient coefficients of total_loss (upstream grad `g`).
// min()/clamp()/max() subgradients follow torch on the measure-one set
// (ties split by torch are measure-zero; tolerance tests cover it).
DEV_INLINE PPORowGrads ppo_row_grads(const GaussRow& r, float vb, float ob,
                                     float ab, float eb, int64_t B,
                                     float clip, float entcoeff, float vcoeff,
                                     float g) {
  const float ratio = __expf(r.logp_pi - r.logp_old);
  const float surr1 = ratio * ab;
  const float lo = 1.f - clip, hi = 1.f + clip;
  const float rc = fminf(fmaxf(ratio, lo), hi);
  const float surr2 = rc * ab;
  float flow;
  if (surr1 <= surr2) {
    flow = 1.f;
  } else {
    flow = (ratio >= lo && ratio <= hi) ? 1.f : 0.f;
  }
  PPORowGrads o;
  o.g_logp = -g / static_cast<float>(B) * ab * ratio * flow;
  o.g_ent = -g * entcoeff / static_cast<float>(B);

  const float d1 = vb - eb;
  const float diff = vb - ob;
  const bool inside = (diff >= -clip && diff <= clip);
  const float dc = fminf(fmaxf(diff, -clip), clip);
  const float d2 = ob + dc - eb;
  float gv;
  if (inside) {
    gv = 2.f * d1;  // vclip == v: both max branches equal; torch's 0.5/0.5
                    // split sums to the same single gradient
  } else if (d1 * d1 >= d2 * d2) {
    gv = 2.f * d1;
  } else {
    gv = 0.f;
  }
  o.g_v = g * vcoeff / static_cast<float>(B) * gv;
  return o;
}
