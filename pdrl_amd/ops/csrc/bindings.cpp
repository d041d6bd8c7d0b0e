// Python bindings for the pdrl_amd CDNA4 HIP kernels.
#include <torch/extension.h>

#include <vector>

std::vector<at::Tensor> seq_lstm_forward_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&);
std::vector<at::Tensor> seq_lstm_backward_core_hip(
    const at::Tensor&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const c10::optional<at::Tensor>&);
std::vector<at::Tensor> seq_lstm_wgrad_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const c10::optional<at::Tensor>&);
void seq_lstm_wgrad_out_hip(const at::Tensor&, const at::Tensor&,
                            const at::Tensor&, const at::Tensor&,
                            const at::Tensor&, const at::Tensor&, at::Tensor&,
                            at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
                            at::Tensor&, at::Tensor&,
                            const c10::optional<at::Tensor>&,
                            const c10::optional<at::Tensor>&,
                            const c10::optional<at::Tensor>&,
                            const c10::optional<at::Tensor>&);
void seq_lstm_forward_multi_hip(const at::Tensor&, const at::Tensor&,
                                const at::Tensor&, const at::Tensor&,
                                const at::Tensor&, long, long, long, long);
void seq_lstm_backward_multi_hip(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&, long,
                                 long, long, long, bool);
void seq_lstm_wgrad_multi_hip(const at::Tensor&, const at::Tensor&,
                              const at::Tensor&, long, long,
                              const c10::optional<at::Tensor>&, long, long);
bool megastep_onpolicy_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, long,
    double, double, double, double, double, double, double, double, double,
    double, double, double, double, double, double, bool, long);
void barrier_bench_hip(at::Tensor&, long, long);
void seq_lstm_fwd_loss_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, at::Tensor&, at::Tensor&,
    const c10::optional<at::Tensor>&, long, double, double, double, double,
    double, double, double, double, double, double, double,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&);
void seq_lstm_bwd_fin_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, long, double, double, double,
    double, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&);
bool vmpo_mid_hip(const at::Tensor&, const at::Tensor&, const at::Tensor&,
                  const at::Tensor&, const at::Tensor&, const at::Tensor&,
                  const at::Tensor&, const at::Tensor&, at::Tensor&,
                  at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
                  const c10::optional<at::Tensor>&, at::Tensor&, long,
                  double, double, double, double, double, double);
at::Tensor gae_hip(const at::Tensor&, double, double, const at::Tensor&);
std::vector<at::Tensor> vtrace_hip(const at::Tensor&, const at::Tensor&,
                                   const at::Tensor&, const at::Tensor&,
                                   const at::Tensor&, double, double, double,
                                   double, long, long, double);
std::vector<at::Tensor> cat_stats_hip(const at::Tensor&, const at::Tensor&,
                                      long);
std::vector<at::Tensor> ppo_td_gae_hip(const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, long, double, double,
                                       double);
void impala_loss_reduce_hip(const at::Tensor&, const at::Tensor&,
                            const at::Tensor&, long, const at::Tensor&,
                            const at::Tensor&, const at::Tensor&, at::Tensor&,
                            double, double, double, double);
void ppo_loss_reduce_hip(const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&, long,
                         const at::Tensor&, const at::Tensor&, at::Tensor&,
                         double, double, double, double, double);
at::Tensor impala_loss_bwd_hip(const at::Tensor&, long, const at::Tensor&,
                               const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, const at::Tensor&, double,
                               double, double, double);
at::Tensor ppo_loss_bwd_hip(const at::Tensor&, long, const at::Tensor&,
                            const at::Tensor&, const at::Tensor&,
                            const at::Tensor&, const at::Tensor&,
                            const at::Tensor&, const at::Tensor&, double,
                            double, double, double, double);
bool impala_loss_mega_hip(const at::Tensor&, const at::Tensor&,
                          const at::Tensor&, const at::Tensor&,
                          const at::Tensor&, at::Tensor&, at::Tensor&,
                          const c10::optional<at::Tensor>&, long, double,
                          double, double, double, double, double, double,
                          double, double);
bool ppo_loss_mega_hip(const at::Tensor&, const at::Tensor&, const at::Tensor&,
                       const at::Tensor&, const at::Tensor&, at::Tensor&,
                       at::Tensor&, const c10::optional<at::Tensor>&, long,
                       double, double, double, double, double, double,
                       double, double);
bool vmpo_loss_mega_hip(const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, at::Tensor&, at::Tensor&,
                        at::Tensor&, at::Tensor&,
                        const c10::optional<at::Tensor>&, at::Tensor&, long,
                        double, double, double, double, double, double,
                        double, double, double, long);
bool ppoc_loss_mega_hip(const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, at::Tensor&, at::Tensor&,
                        const c10::optional<at::Tensor>&, long, double,
                        double, double, double, double, double, double,
                        double);
void sac_actor_loss_hip(const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&, at::Tensor&,
                        at::Tensor&, at::Tensor&,
                        const c10::optional<at::Tensor>&,
                        const c10::optional<at::Tensor>&, double,
                        const c10::optional<at::Tensor>&, double, double);
void sac_critic_loss_hip(const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, at::Tensor&, at::Tensor&,
                         at::Tensor&, const c10::optional<at::Tensor>&,
                         double, double);
void sacc_sample_hip(const at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
                     at::Tensor&);
void sacc_actor_grad_hip(const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, at::Tensor&, at::Tensor&,
                         at::Tensor&, const c10::optional<at::Tensor>&,
                         const c10::optional<at::Tensor>&, double,
                         const c10::optional<at::Tensor>&, double, double);
void sacc_min_mask_hip(const at::Tensor&, const at::Tensor&, at::Tensor&,
                       at::Tensor&, at::Tensor&);
void sacc_critic_loss_hip(const at::Tensor&, const at::Tensor&,
                          const at::Tensor&, const at::Tensor&,
                          const at::Tensor&, const at::Tensor&,
                          const at::Tensor&, const at::Tensor&, at::Tensor&,
                          at::Tensor&, at::Tensor&,
                          const c10::optional<at::Tensor>&, double, double);
void l2norm_sq_hip(const at::Tensor&, at::Tensor&);
void rmsprop_step_hip(at::Tensor&, const at::Tensor&, at::Tensor&,
                      const at::Tensor&, double, double, double, double);
void adam_step_hip(at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
                   at::Tensor&, const at::Tensor&, double, double, double,
                   double, double, bool, const c10::optional<at::Tensor>&,
                   const c10::optional<at::Tensor>&, double,
                   const c10::optional<at::Tensor>&, double);
void sac_actor_bwd_hip(const at::Tensor&, const at::Tensor&,
                       const at::Tensor&, const at::Tensor&, at::Tensor&,
                       at::Tensor&, const c10::optional<at::Tensor>&,
                       const at::Tensor&, const at::Tensor&,
                       const at::Tensor&, const at::Tensor&,
                       const at::Tensor&, const at::Tensor&,
                       const at::Tensor&, at::Tensor&, at::Tensor&);
void sac_actor_wgrad_hip(const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, const at::Tensor&, at::Tensor&,
                         at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
                         at::Tensor&, at::Tensor&,
                         const c10::optional<at::Tensor>&, const at::Tensor&,
                         const at::Tensor&, at::Tensor&, at::Tensor&,
                         const c10::optional<at::Tensor>&,
                         const c10::optional<at::Tensor>&, double, double,
                         double);
void sac_fwd2_critic_loss_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, const c10::optional<at::Tensor>&, double, double);
void sacc_fwd_sample_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&,
    const c10::optional<at::Tensor>&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&);
void sacc_minmask_bwd_hip(
    const at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, at::Tensor&, long);
void sacc_actor_bwd_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, at::Tensor&, at::Tensor&,
    const c10::optional<at::Tensor>&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, at::Tensor&, at::Tensor&);
void sacc_critic_bwd_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
    at::Tensor&, const c10::optional<at::Tensor>&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, at::Tensor&,
    at::Tensor&, at::Tensor&, at::Tensor&, double, double);
void adam_prep_hip(at::Tensor&, double, double);
void adam_multi_hip(const at::Tensor&, const at::Tensor&, const at::Tensor&,
                    long, long, double, double, double);
void soft_update_hip(const std::vector<at::Tensor>&,
                     const std::vector<at::Tensor>&, double);
void soft_update_cached_hip(const at::Tensor&, const at::Tensor&,
                            const at::Tensor&, long, long, double);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("seq_lstm_forward", &seq_lstm_forward_hip,
        "fused body+LSTM+heads forward (gfx950); dual-body via x2/body2",
        py::arg("x"), py::arg("h0"), py::arg("c0"), py::arg("body_w"),
        py::arg("body_b"), py::arg("w_ih"), py::arg("w_hh"), py::arg("b_g"),
        py::arg("heads_w"), py::arg("heads_b"), py::arg("x2") = c10::nullopt,
        py::arg("body2_w") = c10::nullopt, py::arg("body2_b") = c10::nullopt);
  m.def("seq_lstm_backward_core", &seq_lstm_backward_core_hip,
        "fused BPTT backward core (gfx950); dual-body adds dx2",
        py::arg("gouts"), py::arg("ghS"), py::arg("gcS"), py::arg("stash"),
        py::arg("x"), py::arg("c0"), py::arg("body_w"), py::arg("w_ih"),
        py::arg("w_hh"), py::arg("heads_w"),
        py::arg("body2_w") = c10::nullopt);
  m.def("seq_lstm_wgrad", &seq_lstm_wgrad_hip,
        "MFMA weight-gradient GEMMs + wave-per-element small grads",
        py::arg("x"), py::arg("h0"), py::arg("stash"), py::arg("dgates"),
        py::arg("dxb"), py::arg("gouts"), py::arg("x2") = c10::nullopt);
  m.def("seq_lstm_wgrad_out", &seq_lstm_wgrad_out_hip,
        "wgrad writing into caller buffers (flat grad views)",
        py::arg("x"), py::arg("h0"), py::arg("stash"), py::arg("dgates"),
        py::arg("dxb"), py::arg("gouts"), py::arg("dw_ih"), py::arg("dw_hh"),
        py::arg("dbody_w"), py::arg("dbody_b"), py::arg("db_g"),
        py::arg("dheads_w"), py::arg("dheads_b"), py::arg("norm_sq"),
        py::arg("x2") = c10::nullopt, py::arg("dbody2_w") = c10::nullopt,
        py::arg("dbody2_b") = c10::nullopt);
  m.def("seq_lstm_forward_multi", &seq_lstm_forward_multi_hip,
        "multi-network fused forward (device pointer tables; dual-body "
        "rows carry body2/x2 pointers)",
        py::arg("x"), py::arg("h0"), py::arg("c0"), py::arg("core_tab"),
        py::arg("out_tab"), py::arg("C"), py::arg("D"), py::arg("F2") = 0,
        py::arg("half") = 0);
  m.def("seq_lstm_backward_multi", &seq_lstm_backward_multi_hip,
        "multi-network fused backward (leaf inputs; dgates/dxb [+ dx2])",
        py::arg("x"), py::arg("c0"), py::arg("in_tab"), py::arg("out_tab"),
        py::arg("C"), py::arg("D"), py::arg("F2") = 0, py::arg("half") = 0,
        py::arg("accum_dx2") = false);
  m.def("seq_lstm_wgrad_multi", &seq_lstm_wgrad_multi_hip,
        "multi-network MFMA weight grads (device pointer table; dual-body "
        "encoder grad segments)",
        py::arg("x"), py::arg("h0"), py::arg("tab"), py::arg("C"),
        py::arg("D"), py::arg("x2") = c10::nullopt, py::arg("F2") = 0,
        py::arg("half") = 0);
  m.def("megastep_onpolicy", &megastep_onpolicy_hip,
        "ENTIRE IMPALA/PPO training step in one launch (fwd+loss+bwd+wgrad"
        "+RMSprop between grid barriers); false => shape not co-resident");
  m.def("barrier_bench", &barrier_bench_hip,
        "grid-barrier microbenchmark (N back-to-back barriers)");
  m.def("seq_lstm_fwd_loss", &seq_lstm_fwd_loss_hip,
        "forward + row-local loss (IMPALA/PPO/PPO-C, V-MPO pre) in ONE "
        "launch",
        py::arg("x"), py::arg("h0"), py::arg("c0"), py::arg("body_w"),
        py::arg("body_b"), py::arg("w_ih"), py::arg("w_hh"), py::arg("b_g"),
        py::arg("heads_w"), py::arg("heads_b"), py::arg("outs"),
        py::arg("hS"), py::arg("cS"), py::arg("stash"), py::arg("act"),
        py::arg("behav"), py::arg("rew"), py::arg("fir"), py::arg("gouts"),
        py::arg("stats_part"), py::arg("norm_sq"), py::arg("algo"),
        py::arg("gamma"), py::arg("lmbda"), py::arg("rho_bar"),
        py::arg("rho_min"), py::arg("c_bar"), py::arg("rew_scale"),
        py::arg("cp"), py::arg("cv"), py::arg("ce"), py::arg("eps_clip"),
        py::arg("creg"), py::arg("vm_lse") = c10::nullopt,
        py::arg("vm_logp") = c10::nullopt, py::arg("vm_adv") = c10::nullopt,
        py::arg("vm_td") = c10::nullopt);
  m.def("seq_lstm_bwd_fin", &seq_lstm_bwd_fin_hip,
        "BPTT backward + stat finalization (or V-MPO grad emission)",
        py::arg("gouts"), py::arg("stash"), py::arg("x"), py::arg("c0"),
        py::arg("body_w"), py::arg("w_ih"), py::arg("w_hh"),
        py::arg("heads_w"), py::arg("dgates"), py::arg("dxb"),
        py::arg("stats"), py::arg("stats_part"), py::arg("algo"),
        py::arg("cp"), py::arg("cv"), py::arg("ce"), py::arg("creg"),
        py::arg("act") = c10::nullopt, py::arg("behav") = c10::nullopt,
        py::arg("vm_lse") = c10::nullopt, py::arg("vm_psi") = c10::nullopt,
        py::arg("vm_td") = c10::nullopt,
        py::arg("vm_scalars") = c10::nullopt,
        py::arg("vm_outs") = c10::nullopt);
  m.def("vmpo_mid", &vmpo_mid_hip,
        "V-MPO cross-row middle kernel: selection + psi + duals + stats");
  m.def("gae", &gae_hip, "GAE reverse scan");
  m.def("vtrace", &vtrace_hip, "fused V-trace scan",
        pybind11::arg("behav_lp"), pybind11::arg("target_lp"),
        pybind11::arg("is_fir"), pybind11::arg("rew"), pybind11::arg("val"),
        pybind11::arg("gamma"), pybind11::arg("rho_bar"),
        pybind11::arg("rho_min"), pybind11::arg("c_bar"),
        pybind11::arg("vD") = 1, pybind11::arg("val_off") = 0,
        pybind11::arg("rew_scale") = 1.0);
  m.def("cat_stats", &cat_stats_hip, "categorical log-softmax stats");
  m.def("ppo_td_gae", &ppo_td_gae_hip, "fused TD target + GAE scan");
  m.def("impala_loss_reduce", &impala_loss_reduce_hip, "IMPALA loss stats");
  m.def("ppo_loss_reduce", &ppo_loss_reduce_hip, "PPO loss stats");
  m.def("impala_loss_bwd", &impala_loss_bwd_hip, "analytic IMPALA loss grad");
  m.def("ppo_loss_bwd", &ppo_loss_bwd_hip, "analytic PPO loss grad");
  m.def("impala_loss_mega", &impala_loss_mega_hip,
        "single-launch IMPALA loss: stats+vtrace+reduce+bwd");
  m.def("ppo_loss_mega", &ppo_loss_mega_hip,
        "single-launch PPO loss: stats+gae+reduce+bwd");
  m.def("vmpo_loss_mega", &vmpo_loss_mega_hip,
        "single-launch V-MPO loss: stats+gae+topk+duals+bwd",
        py::arg("mo"), py::arg("act"), py::arg("behav"), py::arg("rew"),
        py::arg("fir"), py::arg("log_eta"), py::arg("log_alpha"),
        py::arg("gouts"), py::arg("g_eta"), py::arg("g_alpha"),
        py::arg("stats"), py::arg("norm_sq"), py::arg("rng_state"),
        py::arg("A"), py::arg("gamma"), py::arg("lmbda"),
        py::arg("rew_scale"), py::arg("cp"), py::arg("cv"), py::arg("creg"),
        py::arg("eps_eta"), py::arg("alpha_below"), py::arg("alpha_upper"),
        py::arg("max_phase") = 99);
  m.def("ppoc_loss_mega", &ppoc_loss_mega_hip,
        "single-launch Gaussian-policy PPO loss (K5)");
  m.def("sac_actor_loss", &sac_actor_loss_hip,
        "SAC discrete actor+alpha loss with analytic grads",
        py::arg("moA"), py::arg("q1"), py::arg("q2"), py::arg("log_alpha"),
        py::arg("gouts"), py::arg("g_alpha"), py::arg("stats"),
        py::arg("actor_norm"), py::arg("alpha_norm"),
        py::arg("target_entropy"), py::arg("clock") = c10::nullopt,
        py::arg("beta1") = 0.9, py::arg("beta2") = 0.999);
  m.def("sac_critic_loss", &sac_critic_loss_hip,
        "SAC discrete soft-Q target + twin critic loss grads");
  m.def("sac_actor_bwd", &sac_actor_bwd_hip,
        "SAC discrete actor loss (row-local) fused into the actor BPTT "
        "launch; per-row {ubar, entropy} partials to stats_part");
  m.def("sac_actor_wgrad", &sac_actor_wgrad_hip,
        "actor MFMA wgrad + actor-loss cross-row reduce (g_alpha, stats, "
        "alpha_norm, shared Adam clock prep) in one extra block",
        py::arg("x"), py::arg("h0"), py::arg("stash"), py::arg("dgates"),
        py::arg("dxb"), py::arg("gouts"), py::arg("dw_ih"), py::arg("dw_hh"),
        py::arg("dbody_w"), py::arg("dbody_b"), py::arg("db_g"),
        py::arg("dheads_w"), py::arg("dheads_b"), py::arg("norm_sq"),
        py::arg("stats_part"), py::arg("log_alpha"), py::arg("g_alpha"),
        py::arg("stats4"), py::arg("alpha_norm") = c10::nullopt,
        py::arg("clock") = c10::nullopt, py::arg("target_entropy") = 0.0,
        py::arg("beta1") = 0.9, py::arg("beta2") = 0.999);
  m.def("sac_fwd2_critic_loss", &sac_fwd2_critic_loss_hip,
        "post-update actor forward + SAC critic loss in one launch "
        "(fwd+loss pattern; vl partials reduced by the critic Adam)");
  m.def("sacc_sample", &sacc_sample_hip,
        "reparameterized tanh-Gaussian sample + log-prob (graph-safe RNG)");
  m.def("sacc_actor_grad", &sacc_actor_grad_hip,
        "SAC-continuous actor+alpha loss with analytic dmu/dlog_std",
        py::arg("moA"), py::arg("eps"), py::arg("act"), py::arg("g"),
        py::arg("q1"), py::arg("q2"), py::arg("log_alpha"), py::arg("dmoA"),
        py::arg("g_alpha"), py::arg("stats"), py::arg("actor_norm"),
        py::arg("alpha_norm"), py::arg("target_entropy"),
        py::arg("clock") = c10::nullopt, py::arg("beta1") = 0.9,
        py::arg("beta2") = 0.999);
  m.def("sacc_critic_loss", &sacc_critic_loss_hip,
        "SAC-continuous soft-Q target + twin critic loss grads");
  m.def("sacc_fwd_sample", &sacc_fwd_sample_hip,
        "actor forward + reparameterized tanh-Gaussian sample in one "
        "launch (optional dQ/da zeroing and behaviour-action staging)",
        py::arg("x"), py::arg("h0"), py::arg("c0"), py::arg("body_w"),
        py::arg("body_b"), py::arg("w_ih"), py::arg("w_hh"), py::arg("b_g"),
        py::arg("heads_w"), py::arg("heads_b"), py::arg("moA"),
        py::arg("hS"), py::arg("cS"), py::arg("stash"), py::arg("rng"),
        py::arg("eps"), py::arg("act"), py::arg("logpi"),
        py::arg("dact_zero") = c10::nullopt,
        py::arg("act_src") = c10::nullopt, py::arg("actb") = c10::nullopt);
  m.def("sacc_minmask_bwd", &sacc_minmask_bwd_hip,
        "min-critic selection + twin-critic input-grad backward in one "
        "launch (dminQ/da accumulated into the shared buffer)");
  m.def("sacc_actor_bwd", &sacc_actor_bwd_hip,
        "SAC-continuous analytic actor grad + actor BPTT in one launch; "
        "per-row partials reduced by sac_actor_wgrad");
  m.def("sacc_critic_bwd", &sacc_critic_bwd_hip,
        "SAC-continuous soft-Q critic loss + twin BPTT in one launch; "
        "huber partials reduced by the critic Adam");
  m.def("sacc_min_mask", &sacc_min_mask_hip,
        "min-critic selection masks for dE[-minQ]/dq + zero the dQ/da "
        "accumulator");
  m.def("l2norm_sq", &l2norm_sq_hip, "squared L2 norm into a device scalar");
  m.def("rmsprop_step", &rmsprop_step_hip, "fused clip+RMSprop on flat buffers");
  m.def("adam_step", &adam_step_hip,
        "fused clip+Adam on flat buffers (optional per-row stats reduce in "
        "block 0 and inline Polyak target tracking)",
        py::arg("p"), py::arg("g"), py::arg("m"), py::arg("v"),
        py::arg("state3"), py::arg("norm_sq"), py::arg("lr"), py::arg("beta1"),
        py::arg("beta2"), py::arg("eps"), py::arg("max_norm"),
        py::arg("do_prep") = true, py::arg("stats_part") = c10::nullopt,
        py::arg("stats_out") = c10::nullopt, py::arg("part_scale") = 1.0,
        py::arg("polyak") = c10::nullopt, py::arg("tau") = 0.0);
  m.def("adam_prep", &adam_prep_hip,
        "advance the shared device Adam step clock");
  m.def("adam_multi", &adam_multi_hip,
        "multi-group fused clip+Adam (shared step clock, pointer table)");
  m.def("soft_update", &soft_update_hip, "multi-tensor Polyak update");
  m.def("soft_update_cached", &soft_update_cached_hip,
        "Polyak update with prebuilt device pointer tables (graph-safe)");
}
