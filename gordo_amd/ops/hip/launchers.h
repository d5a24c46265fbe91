// Host launch-function declarations shared by bindings.cpp (host-only TU) and
// the .hip TUs that define them.
#pragma once
#include <torch/extension.h>
#include <vector>

void hg_train_step_launch(torch::Tensor X, torch::Tensor Y,
                          torch::Tensor counters, torch::Tensor w, torch::Tensor wt,
                          torch::Tensor bias, torch::Tensor dw, torch::Tensor db,
                          torch::Tensor loss, int64_t B, int64_t F_out,
                          std::vector<int64_t> acts, double gscale, bool y_is_x,
                          c10::optional<torch::Tensor> dx_out, int64_t wmax,
                          int64_t kbits, int64_t seed,
                          std::vector<int64_t> widths = {});

void hg_adam_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                    torch::Tensor dw, torch::Tensor db, torch::Tensor w,
                    torch::Tensor wt, torch::Tensor bias, torch::Tensor counters,
                    int64_t n_slabs, int64_t L, double lr, double beta1, double beta2,
                    double eps, int64_t wmax);

void hg_bump_launch(torch::Tensor counters, int64_t n_batches);

void hg_permute_fill_launch(torch::Tensor out, int64_t kbits, int64_t key);

void hg_forward_launch(torch::Tensor X, torch::Tensor w, torch::Tensor bias,
                       torch::Tensor xscale, torch::Tensor xshift, torch::Tensor yhat,
                       c10::optional<torch::Tensor> err, c10::optional<torch::Tensor> total,
                       c10::optional<torch::Tensor> inv_err_scale,
                       int64_t F_in, int64_t F_out, std::vector<int64_t> acts,
                       int64_t wmax);

void hg_grad_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor g,
                           int64_t n_slabs, int64_t L, int64_t wmax);

void hg_adam_flat_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                         torch::Tensor g, torch::Tensor w, torch::Tensor wt,
                         torch::Tensor bias, torch::Tensor counters, int64_t L,
                         double lr, double beta1, double beta2, double eps,
                         int64_t wmax);

void hg_slab_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor partial,
                           c10::optional<torch::Tensor> counters, int64_t n_slabs,
                           int64_t G, int64_t L, int64_t n_batches, int64_t wmax);

void hg_adam_p_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                      torch::Tensor partial, torch::Tensor w, torch::Tensor wt,
                      torch::Tensor bias, torch::Tensor counters, int64_t G,
                      int64_t L, double lr, double beta1, double beta2, double eps,
                      int64_t wmax);

void hg_reduce_adam_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor master,
                           torch::Tensor m, torch::Tensor v, torch::Tensor w,
                           torch::Tensor wt, torch::Tensor bias,
                           torch::Tensor counters_r, torch::Tensor counters_w,
                           int64_t n_slabs, int64_t L, int64_t n_batches,
                           double lr, double beta1, double beta2, double eps,
                           int64_t wmax);

void hg_train_step_w1_launch(torch::Tensor X, torch::Tensor Y,
                             torch::Tensor counters, torch::Tensor w, torch::Tensor wt,
                             torch::Tensor bias, torch::Tensor dw, torch::Tensor db,
                             torch::Tensor loss, int64_t B, int64_t F_out,
                             std::vector<int64_t> acts, double gscale, bool y_is_x,
                             c10::optional<torch::Tensor> dx_out, int64_t wmax,
                             int64_t kbits, int64_t seed, std::vector<int64_t> widths);

void hg_flat_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor g,
                           torch::Tensor counters_r, torch::Tensor counters_w,
                           int64_t n_slabs, int64_t L, int64_t n_batches,
                           int64_t wmax);

void hg_partial_flat_launch(torch::Tensor partial, torch::Tensor g, int64_t G,
                            int64_t L, int64_t wmax);

void lstm_fwd_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                     torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                     torch::Tensor gates_seq, torch::Tensor c_seq,
                     c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T);

void lstm_bwd_cell_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                          torch::Tensor c_seq, torch::Tensor wcat_t,
                          torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                          int64_t B, int64_t T, bool dh_last_only);

void lstm_dw_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                    torch::Tensor partial, torch::Tensor db_partial,
                    int64_t BT, int64_t T);

void lstm_fwd_wide_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                          torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                          torch::Tensor gates_seq, torch::Tensor c_seq,
                          c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T);

void lstm_bwd_wide_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                          torch::Tensor c_seq, torch::Tensor wcat_t,
                          torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                          int64_t B, int64_t T, bool dh_last_only);

void lstm_dw_wide_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                         torch::Tensor partial, torch::Tensor db_partial,
                         int64_t BT, int64_t T);

void lstm_fwd_wide256_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                             torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                             torch::Tensor gates_seq, torch::Tensor c_seq,
                             c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T);

void lstm_bwd_wide256_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                             torch::Tensor c_seq, torch::Tensor wcat_t,
                             torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                             int64_t B, int64_t T, bool dh_last_only);

void lstm_dw_wide256_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                            torch::Tensor partial, torch::Tensor db_partial,
                            int64_t BT, int64_t T);

void lstm_fwd_h32_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                         torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                         torch::Tensor gates_seq, torch::Tensor c_seq,
                         c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T);

void lstm_bwd_h32_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                         torch::Tensor c_seq, torch::Tensor wcat_t,
                         torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                         int64_t B, int64_t T, bool dh_last_only);

void lstm_dw_h32_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                        torch::Tensor partial, torch::Tensor db_partial,
                        int64_t BT, int64_t T);
