// Python bindings for the dppo_amd gfx950 kernel library.
#include <torch/extension.h>

#include <vector>

std::vector<torch::Tensor> gae_scan(torch::Tensor rewards, torch::Tensor values,
                                    torch::Tensor dones, torch::Tensor boot,
                                    double gamma, double lam, bool whiten,
                                    double eps, torch::Tensor adv_out,
                                    torch::Tensor etr_out);

torch::Tensor ppo_loss_gauss_fwd(torch::Tensor pdpi, torch::Tensor pdold,
                                 torch::Tensor vpred, torch::Tensor oldv,
                                 torch::Tensor act, torch::Tensor adv,
                                 torch::Tensor etr, double clip,
                                 double entcoeff, double vcoeff);

torch::Tensor ppo_loss_gauss_gh(torch::Tensor pdflat, torch::Tensor oldflat,
                                torch::Tensor vpred, torch::Tensor oldv,
                                torch::Tensor act, torch::Tensor adv,
                                torch::Tensor etr, double clip,
                                double entcoeff, double vcoeff,
                                torch::Tensor clip_dev);

std::vector<torch::Tensor> ppo_loss_gauss_bwd(
    torch::Tensor pdpi, torch::Tensor pdold, torch::Tensor vpred,
    torch::Tensor oldv, torch::Tensor act, torch::Tensor adv,
    torch::Tensor etr, double clip, double entcoeff, double vcoeff,
    torch::Tensor gtotal);

torch::Tensor ppo_loss_cat_fwd(torch::Tensor lpi, torch::Tensor lold,
                               torch::Tensor vpred, torch::Tensor oldv,
                               torch::Tensor act, torch::Tensor adv,
                               torch::Tensor etr, double clip,
                               double entcoeff, double vcoeff);

std::vector<torch::Tensor> ppo_loss_cat_bwd(
    torch::Tensor lpi, torch::Tensor lold, torch::Tensor vpred,
    torch::Tensor oldv, torch::Tensor act, torch::Tensor adv,
    torch::Tensor etr, double clip, double entcoeff, double vcoeff,
    torch::Tensor gtotal);

torch::Tensor cat_sample(torch::Tensor logits, int64_t seed, int64_t ctr);

void bf16_mm256(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                int64_t epi, torch::Tensor bias, torch::Tensor aux,
                torch::Tensor grad, int64_t grad_off, torch::Tensor CT,
                int64_t ldt, torch::Tensor sums, int64_t sums_off);

void bf16_mm_small(torch::Tensor A, torch::Tensor B, torch::Tensor C,
                   torch::Tensor C2, torch::Tensor aux, torch::Tensor grad,
                   int64_t g1_off, int64_t g2_off, int64_t srow, int64_t epi,
                   int64_t m_real, int64_t n_real, int64_t ldc,
                   torch::Tensor bias);

void bf16_transpose(torch::Tensor in, torch::Tensor out, torch::Tensor sums,
                    int64_t sums_off, int64_t R, int64_t C, int64_t ld_in,
                    int64_t ld_out);

void gauss_gh_wide(torch::Tensor pdflat_bf, torch::Tensor oldflat,
                   torch::Tensor v_bf, torch::Tensor oldv, torch::Tensor act,
                   torch::Tensor adv, torch::Tensor etr, torch::Tensor gh,
                   torch::Tensor clip_dev, double clip, double entcoeff,
                   double vcoeff);

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, int64_t step, double lr, double beta1,
               double beta2, double eps);

void adam_step_dev(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                   torch::Tensor v, torch::Tensor step_dev,
                   torch::Tensor lr_dev, torch::Tensor coef, double beta1,
                   double beta2, double eps);

void gemm_fwd(torch::Tensor X, torch::Tensor Wt, torch::Tensor bias,
              int64_t activation, int64_t heads, torch::Tensor C,
              torch::Tensor v, torch::Tensor aux, int64_t wt_layout,
              int64_t ablate, int64_t ldc);

void dw_mfma(torch::Tensor delta, torch::Tensor acts, torch::Tensor grad_buf,
             int64_t w_off, int64_t b_off, int64_t split_row, int64_t w_off2,
             int64_t b_off2, int64_t ablate);

void rollout_sample(torch::Tensor pdflat, torch::Tensor actions,
                    torch::Tensor xva, torch::Tensor seed_dev,
                    torch::Tensor eps_dev, int64_t step, int64_t va_off,
                    double act_low, double act_high);

void gemm_env_step(torch::Tensor xva, torch::Tensor M, torch::Tensor xin,
                   torch::Tensor xout, torch::Tensor envd,
                   torch::Tensor horizons, torch::Tensor t,
                   torch::Tensor epr,
                   torch::Tensor rewards, torch::Tensor dones,
                   torch::Tensor rsum, torch::Tensor seed_dev, double sigma,
                   int64_t step);

void rollout_env_step(torch::Tensor xin, torch::Tensor xout, torch::Tensor G,
                      torch::Tensor envd, torch::Tensor horizons,
                      torch::Tensor t, torch::Tensor epr,
                      torch::Tensor rewards, torch::Tensor dones,
                      torch::Tensor seed_dev, double sigma, int64_t step);

torch::Tensor rollout_moments(torch::Tensor rewards, torch::Tensor dones,
                              torch::Tensor epr_in, int64_t T, int64_t E);

bool mlp_chunk_supported(int64_t D, int64_t H, int64_t A, int64_t n_hidden);

void mlp_chunk_train(
    torch::Tensor params, torch::Tensor states, torch::Tensor actions,
    torch::Tensor adv, torch::Tensor etr, torch::Tensor oldflat,
    torch::Tensor oldv, std::vector<int64_t> offsets,
    std::vector<int64_t> dims, int64_t activation, torch::Tensor clip_dev,
    double clip, double entcoeff, double vcoeff, torch::Tensor slabs,
    torch::Tensor mom, torch::Tensor vel, torch::Tensor step_dev,
    torch::Tensor lr_dev, torch::Tensor coef, torch::Tensor flat_grad,
    bool fuse_adam, double beta1, double beta2, double eps);

std::vector<torch::Tensor> rollout_run(
    torch::Tensor params, std::vector<int64_t> offsets,
    std::vector<int64_t> dims, int64_t activation,
    torch::Tensor envblob, int64_t rank, torch::Tensor horizons,
    double noise, double act_low, double act_high, double eps_explore,
    torch::Tensor x, torch::Tensor t, torch::Tensor epr,
    int64_t T, int64_t act_dim, int64_t seed, torch::Tensor out_buf,
    int64_t ablate);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("gae_scan", &gae_scan,
          "segmented GAE reverse scan + whitening (gfx950)");
  mod.def("ppo_loss_gauss_fwd", &ppo_loss_gauss_fwd,
          "fused DiagGaussian PPO loss forward (gfx950)");
  mod.def("ppo_loss_gauss_bwd", &ppo_loss_gauss_bwd,
          "fused DiagGaussian PPO loss backward (gfx950)");
  mod.def("ppo_loss_cat_fwd", &ppo_loss_cat_fwd,
          "fused Categorical PPO loss forward (gfx950)");
  mod.def("ppo_loss_cat_bwd", &ppo_loss_cat_bwd,
          "fused Categorical PPO loss backward (gfx950)");
  mod.def("cat_sample", &cat_sample,
          "Gumbel-max categorical sampling, counter-based RNG (gfx950)");
  mod.def("bf16_mm256", &bf16_mm256,
          "bf16 MFMA GEMM, 256^2 tile, pipelined glds, fused epilogues "
          "(gfx950)");
  mod.def("bf16_mm_small", &bf16_mm_small,
          "guarded bf16 MFMA GEMM for ragged heads shapes (gfx950)");
  mod.def("bf16_transpose", &bf16_transpose,
          "bf16 2D transpose with fused column sums (gfx950)");
  mod.def("gauss_gh_wide", &gauss_gh_wide,
          "wide-policy PPO loss gradient -> bf16 [g_pd | g_v] (gfx950)");
  mod.def("adam_step", &adam_step, "fused flat Adam step (gfx950)");
  mod.def("adam_step_dev", &adam_step_dev,
          "graph-replayable fused Adam (device step/lr) (gfx950)");
  mod.def("rollout_run", &rollout_run,
          "fused T-step rollout: MLP fwd + sample + synthetic env (gfx950)");
  mod.def("rollout_sample", &rollout_sample,
          "per-step action sampling + eps-greedy (v3 rollout) (gfx950)");
  mod.def("gemm_env_step", &gemm_env_step,
          "fused G-GEMM + env transition epilogue + per-env finish "
          "(v3 rollout; gfx950)");
  mod.def("rollout_env_step", &rollout_env_step,
          "per-step synthetic env finish: tanh/reward/done/reset (gfx950)");
  mod.def("rollout_moments", &rollout_moments,
          "episode-reward moments from reward/done streams (gfx950)");
  mod.def("ppo_loss_gauss_gh", &ppo_loss_gauss_gh,
          "wave-per-row PPO loss gradient -> [g_pd | g_v] (gfx950)");
  mod.def("gemm_fwd", &gemm_fwd,
          "MFMA f32 layer forward C=act(X@Wt+b), fused tanh (gfx950)");
  mod.def("dw_mfma", &dw_mfma,
          "MFMA f32 split-K dW += delta^T@acts into flat grad (gfx950)");
  mod.def("mlp_chunk_supported", &mlp_chunk_supported,
          "shape eligibility for the fused chunk-step kernel (gfx950)");
  mod.def("mlp_chunk_train", &mlp_chunk_train,
          "fused small-MLP training chunk step: fwd+PPO grad+bwd+dW "
          "partials + slab-reduce Adam (gfx950)");
}
