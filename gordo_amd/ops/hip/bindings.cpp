// Torch extension module registration (host-only TU; kernels + launchers live
// in the .hip TUs so they get HIP device compilation).
#include <torch/extension.h>

#include "launchers.h"

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("hg_train_step", &hg_train_step_launch, "fused hourglass train step",
          py::arg("X"), py::arg("Y"), py::arg("counters"), py::arg("w"),
          py::arg("wt"), py::arg("bias"), py::arg("dw"), py::arg("db"),
          py::arg("loss"), py::arg("B"), py::arg("F_out"), py::arg("acts"),
          py::arg("gscale"), py::arg("y_is_x"), py::arg("dx_out"),
          py::arg("wmax"), py::arg("kbits"), py::arg("seed"),
          py::arg("widths") = std::vector<int64_t>{});
    m.def("hg_train_step_w1", &hg_train_step_w1_launch,
          "wave-local hourglass train step (W=64, 16-row blocks, no barriers)",
          py::arg("X"), py::arg("Y"), py::arg("counters"), py::arg("w"),
          py::arg("wt"), py::arg("bias"), py::arg("dw"), py::arg("db"),
          py::arg("loss"), py::arg("B"), py::arg("F_out"), py::arg("acts"),
          py::arg("gscale"), py::arg("y_is_x"), py::arg("dx_out"),
          py::arg("wmax"), py::arg("kbits"), py::arg("seed"),
          py::arg("widths") = std::vector<int64_t>{});
    m.def("hg_adam", &hg_adam_launch, "slab-reduce + Adam + bf16 weight emit");
    m.def("hg_bump", &hg_bump_launch, "advance step counters");
    m.def("hg_permute_fill", &hg_permute_fill_launch, "materialize Feistel shuffle (tests)");
    m.def("hg_forward", &hg_forward_launch, "fused serving forward (+anomaly)");
    m.def("hg_grad_reduce", &hg_grad_reduce_launch, "slab reduce -> flat grad bucket");
    m.def("hg_adam_flat", &hg_adam_flat_launch, "Adam from reduced flat grad");
    m.def("hg_slab_reduce", &hg_slab_reduce_launch, "stage-A slab reduce (+counter bump)");
    m.def("hg_adam_p", &hg_adam_p_launch, "stage-B Adam over G partials");
    m.def("hg_partial_flat", &hg_partial_flat_launch, "partials -> flat DP bucket");
    m.def("hg_reduce_adam", &hg_reduce_adam_launch,
          "fused slab reduce + Adam (ping-pong counters)");
    m.def("hg_flat_reduce", &hg_flat_reduce_launch,
          "fused slabs -> flat DP bucket (ping-pong counters)");
    m.def("lstm_fwd", &lstm_fwd_launch, "persistent fused 4-gate LSTM forward");
    m.def("lstm_bwd_cell", &lstm_bwd_cell_launch, "reverse-time BPTT cell kernel (+fused dx)");
    m.def("lstm_dw", &lstm_dw_launch, "split-K dWcat + dbias from one dgates pass");
    m.def("lstm_fwd_wide", &lstm_fwd_wide_launch, "wide (W=128) persistent LSTM forward");
    m.def("lstm_bwd_wide", &lstm_bwd_wide_launch, "wide reverse-time BPTT cell (+fused dx)");
    m.def("lstm_dw_wide", &lstm_dw_wide_launch, "wide split-K dWcat quadrants + dbias");
    m.def("lstm_fwd_h32", &lstm_fwd_h32_launch, "narrow (W=32) persistent LSTM forward");
    m.def("lstm_bwd_h32", &lstm_bwd_h32_launch, "narrow reverse-time BPTT cell (+fused dx)");
    m.def("lstm_dw_h32", &lstm_dw_h32_launch, "narrow split-K dWcat + dbias");
    m.def("lstm_fwd_wide256", &lstm_fwd_wide256_launch, "W=256 persistent LSTM forward (column-half split)");
    m.def("lstm_bwd_wide256", &lstm_bwd_wide256_launch, "W=256 reverse-time BPTT cell (+fused dx)");
    m.def("lstm_dw_wide256", &lstm_dw_wide256_launch, "W=256 split-K dWcat 4x4 quadrants + dbias");
    m.attr("WMAX") = 64;
    m.attr("MAXL") = 8;
    m.attr("MT") = 32;   // hg_train_step batch rows per block/slab
    m.attr("MT_W1") = 16;  // hg_train_step_w1 (wave-local W=64) rows/slab
}
