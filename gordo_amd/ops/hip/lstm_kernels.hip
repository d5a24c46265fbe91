// Fused 4-gate LSTM cell kernels for MI355X (gfx950) — SURVEY.md §2.3 K4-K6.
//
// The reference trains Keras stacked LSTMs over (batch, lookback=144, tags)
// windows; here each layer's sequential time loop is ONE persistent kernel:
//
//  lstm_fwd:  per block = 64 batch rows, loops t = 0..T-1 on-chip.
//    gates[m][4H] = cat(x_t, h_{t-1})[m][128] x Wcat^T   (MFMA bf16 16x16x32,
//    x staged per-t into a swizzled LDS tile, h kept in a swizzled LDS tile
//    across the whole loop, weights streamed from L2)
//    -> i,f,g,o activations; c lives in REGISTERS (the 4 gate columns of one
//    (m,j) position land in the same lane's accumulators: jt, jt+4, jt+8,
//    jt+12), h -> LDS + global. Saves h/c/gates sequences for BPTT.
//    Layer 0 gathers windows straight from the resident dataset (K6).
//
//  lstm_bwd_cell: reverse-time persistent kernel: elementwise BPTT state
//    update (dc in registers) + dgates, and the sequential recurrent GEMM
//    dh_rec = dgates x W_hh per step. Writes dgates_seq; the non-sequential
//    heavy GEMMs (dW = in^T x dgates over K = B*T, dx = dgates x W_ih) are
//    plain library GEMMs done by the engine via torch.matmul (rocBLAS).
//
// Layout: H and F padded to 64; Wcat is [4H=256][128] bf16 (cols 0..63 =
// W_ih padded, 64..127 = W_hh padded); WcatT is [128][256].
#include "common.h"

#define LSTM_H GA_WMAX          // padded hidden width (64)
#define LSTM_4H (4 * GA_WMAX)   // 256
#define LSTM_K 128              // cat(x, h) padded
#define LSTM_NT 16              // 256/16 column tiles per wave row-stripe

struct LstmFwdArgs {
    const unsigned short* __restrict__ X;     // layer0: dataset [N][64]; else seq [B][T][64]
    const int* __restrict__ win;              // [B] window start rows (layer0) or null
    const unsigned short* __restrict__ wcat;  // [256][128] bf16
    const float* __restrict__ bias;           // [256] fp32 (b_ih + b_hh)
    unsigned short* __restrict__ h_seq;       // [B][T][64] out
    unsigned short* __restrict__ gates_seq;   // [B][T][256] out (activated i,f,g,o)
    unsigned short* __restrict__ c_seq;       // [B][T][64] out
    unsigned short* __restrict__ x_seq_out;   // [B][T][64] gathered x copy (layer0) or null
    int B, T, N;
};

extern __shared__ __attribute__((aligned(16))) char ga_smem_lstm[];

__launch_bounds__(256)
__global__ void lstm_fwd(LstmFwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * GA_MT;
    const int wrow = wid * 16;

    char* x_tile = ga_smem_lstm;                    // [64][64] swizzled bf16
    char* h_tile = ga_smem_lstm + GA_TILE_BYTES;    // [64][64] swizzled bf16
    float* s_bias = (float*)(ga_smem_lstm + 2 * GA_TILE_BYTES);  // [256]

    for (int i = tid; i < LSTM_4H; i += 256) s_bias[i] = a.bias[i];

    // zero h_tile (h_{-1} = 0)
    for (int c = tid; c < GA_MT * 8; c += 256) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        ga_tile_write8(h_tile, c >> 3, (c & 7) * 8, z);
    }

    // c state in registers: lane owns (rows wrow + (lane>>4)*4 + r, col jt*16 + (lane&15))
    float c_reg[4][4];  // [jt][r]
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int r = 0; r < 4; ++r) c_reg[jt][r] = 0.f;

    __syncthreads();

    for (int t = 0; t < a.T; ++t) {
        // ---- stage x_t tile ------------------------------------------------
        for (int c = tid; c < GA_MT * 8; c += 256) {
            int m = c >> 3, col0 = (c & 7) * 8;
            int gi = base + m;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gi < a.B) {
                if (a.win != nullptr) {
                    int row = a.win[gi] + t;
                    v = *(const ga_u16x8*)(a.X + (size_t)row * GA_WMAX + col0);
                } else {
                    v = *(const ga_u16x8*)(a.X + ((size_t)gi * a.T + t) * GA_WMAX + col0);
                }
                if (a.x_seq_out != nullptr)
                    *(ga_u16x8*)(a.x_seq_out + ((size_t)gi * a.T + t) * GA_WMAX + col0) = v;
            }
            ga_tile_write8(x_tile, m, col0, v);
        }
        __syncthreads();

        // ---- gates GEMM: [64][256] = cat(x,h)[64][128] x Wcat^T ----------
        ga_f32x4 acc[LSTM_NT];
#pragma unroll
        for (int nt = 0; nt < LSTM_NT; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < LSTM_K; kk += 32) {
            const char* src = (kk < 64) ? x_tile : h_tile;
            int klocal = kk & 63;
            ga_bf16x8 afrag = ga_tile_read8(src, wrow + (lane & 15), klocal + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < LSTM_NT; ++nt) {
                // B[k][j] = Wcat[j][k] row-read (16B, L2-resident)
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(a.wcat + (size_t)(nt * 16 + (lane & 15)) * LSTM_K + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }
        __syncthreads();  // x_tile/h_tile reads done; epilogue rewrites h_tile

        // ---- cell update (lane-local: gates i,f,g,o of one (m,j) live in
        //      acc[jt], acc[jt+4], acc[jt+8], acc[jt+12]) -------------------
#pragma unroll
        for (int jt = 0; jt < 4; ++jt) {
            int col = jt * 16 + (lane & 15);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int m = wrow + (lane >> 4) * 4 + r;
                int gi = base + m;
                float i_g = 1.f / (1.f + __expf(-(acc[jt][r] + s_bias[col])));
                float f_g = 1.f / (1.f + __expf(-(acc[jt + 4][r] + s_bias[GA_WMAX + col])));
                float g_g = tanhf(acc[jt + 8][r] + s_bias[2 * GA_WMAX + col]);
                float o_g = 1.f / (1.f + __expf(-(acc[jt + 12][r] + s_bias[3 * GA_WMAX + col])));
                float c_new = f_g * c_reg[jt][r] + i_g * g_g;
                c_reg[jt][r] = c_new;
                float h_new = o_g * tanhf(c_new);
                unsigned short hb = ga_f32_to_bf16(h_new);
                ga_tile_write1(h_tile, m, col, hb);
                if (gi < a.B) {
                    size_t bt = (size_t)gi * a.T + t;
                    a.h_seq[bt * GA_WMAX + col] = hb;
                    a.c_seq[bt * GA_WMAX + col] = ga_f32_to_bf16(c_new);
                    a.gates_seq[bt * LSTM_4H + col] = ga_f32_to_bf16(i_g);
                    a.gates_seq[bt * LSTM_4H + GA_WMAX + col] = ga_f32_to_bf16(f_g);
                    a.gates_seq[bt * LSTM_4H + 2 * GA_WMAX + col] = ga_f32_to_bf16(g_g);
                    a.gates_seq[bt * LSTM_4H + 3 * GA_WMAX + col] = ga_f32_to_bf16(o_g);
                }
            }
        }
        __syncthreads();
    }
}

// ===========================================================================
// Backward cell: reverse time, produce dgates_seq; dc carried in registers,
// dh_rec through LDS. dh_out is either a full sequence [B][T][64] or (for the
// top layer feeding a last-step head) a single [B][64] applied at t = T-1.
// ===========================================================================
struct LstmBwdArgs {
    const unsigned short* __restrict__ dh_out;     // [B][T][64] or [B][64]
    const unsigned short* __restrict__ gates_seq;  // [B][T][256]
    const unsigned short* __restrict__ c_seq;      // [B][T][64]
    const unsigned short* __restrict__ wcat_t;     // [128][256] bf16 (WcatT)
    unsigned short* __restrict__ dgates_seq;       // [B][T][256] out
    int B, T;
    int dh_last_only;  // 1: dh_out is [B][64] at the final step only
};

__launch_bounds__(256)
__global__ void lstm_bwd_cell(LstmBwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * GA_MT;
    const int wrow = wid * 16;

    char* dh_tile = ga_smem_lstm;                     // [64][64] dh_rec (swizzled)
    char* dg_tiles = ga_smem_lstm + GA_TILE_BYTES;    // 4 x [64][64] dgates tiles

    // zero dh_rec
    for (int c = tid; c < GA_MT * 8; c += 256) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        ga_tile_write8(dh_tile, c >> 3, (c & 7) * 8, z);
    }
    // dc state registers: elementwise thread mapping m = tid>>2, 16 cols each
    const int em = tid >> 2;          // 0..63 row
    const int eg = tid & 3;           // col group: cols eg*16 + e
    float dc_reg[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) dc_reg[e] = 0.f;
    __syncthreads();

    for (int t = a.T - 1; t >= 0; --t) {
        // ---- elementwise BPTT state update + dgates -> LDS + global -------
        {
            int gi = base + em;
            size_t bt = (size_t)gi * a.T + t;
#pragma unroll
            for (int e = 0; e < 16; ++e) {
                int col = eg * 16 + e;
                float dh = 0.f, i_g = 0.f, f_g = 0.f, g_g = 0.f, o_g = 0.f;
                float c_t = 0.f, c_prev = 0.f;
                if (gi < a.B) {
                    if (a.dh_last_only) {
                        dh = (t == a.T - 1)
                                 ? ga_bf16_to_f32(a.dh_out[(size_t)gi * GA_WMAX + col])
                                 : 0.f;
                    } else {
                        dh = ga_bf16_to_f32(a.dh_out[bt * GA_WMAX + col]);
                    }
                    i_g = ga_bf16_to_f32(a.gates_seq[bt * LSTM_4H + col]);
                    f_g = ga_bf16_to_f32(a.gates_seq[bt * LSTM_4H + GA_WMAX + col]);
                    g_g = ga_bf16_to_f32(a.gates_seq[bt * LSTM_4H + 2 * GA_WMAX + col]);
                    o_g = ga_bf16_to_f32(a.gates_seq[bt * LSTM_4H + 3 * GA_WMAX + col]);
                    c_t = ga_bf16_to_f32(a.c_seq[bt * GA_WMAX + col]);
                    c_prev = (t > 0) ? ga_bf16_to_f32(a.c_seq[(bt - 1) * GA_WMAX + col]) : 0.f;
                }
                dh += ga_bf16_to_f32(ga_tile_read1(dh_tile, em, col));
                float tanhc = tanhf(c_t);
                float dc = dc_reg[e] + dh * o_g * (1.f - tanhc * tanhc);
                float do_pre = dh * tanhc * o_g * (1.f - o_g);
                float di_pre = dc * g_g * i_g * (1.f - i_g);
                float df_pre = dc * c_prev * f_g * (1.f - f_g);
                float dg_pre = dc * i_g * (1.f - g_g * g_g);
                dc_reg[e] = dc * f_g;  // carry to t-1
                unsigned short di_b = ga_f32_to_bf16(di_pre);
                unsigned short df_b = ga_f32_to_bf16(df_pre);
                unsigned short dg_b = ga_f32_to_bf16(dg_pre);
                unsigned short do_b = ga_f32_to_bf16(do_pre);
                ga_tile_write1(dg_tiles, em, col, di_b);
                ga_tile_write1(dg_tiles + GA_TILE_BYTES, em, col, df_b);
                ga_tile_write1(dg_tiles + 2 * GA_TILE_BYTES, em, col, dg_b);
                ga_tile_write1(dg_tiles + 3 * GA_TILE_BYTES, em, col, do_b);
                if (gi < a.B) {
                    a.dgates_seq[bt * LSTM_4H + col] = di_b;
                    a.dgates_seq[bt * LSTM_4H + GA_WMAX + col] = df_b;
                    a.dgates_seq[bt * LSTM_4H + 2 * GA_WMAX + col] = dg_b;
                    a.dgates_seq[bt * LSTM_4H + 3 * GA_WMAX + col] = do_b;
                }
            }
        }
        __syncthreads();

        if (t == 0) break;  // no recurrent grad into h_{-1}

        // ---- dh_rec[m][k<H] = dgates[m][256] x W_hh[256->k] ---------------
        // B[k4h][j=k] = Wcat[k4h][64+j] = WcatT[64+j][k4h]: row-read of WcatT
        ga_f32x4 acc[4];
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < LSTM_4H; kk += 32) {
            const char* dgt = dg_tiles + (kk >> 6) * GA_TILE_BYTES;
            ga_bf16x8 afrag = ga_tile_read8(dgt, wrow + (lane & 15), (kk & 63) + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < 4; ++nt) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(a.wcat_t + (size_t)(GA_WMAX + nt * 16 + (lane & 15)) * LSTM_4H + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }
        __syncthreads();  // dg_tiles reads done before next iteration rewrites
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
            int col = nt * 16 + (lane & 15);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int m = wrow + (lane >> 4) * 4 + r;
                ga_tile_write1(dh_tile, m, col, ga_f32_to_bf16(acc[nt][r]));
            }
        }
        __syncthreads();
    }
}

// ===========================================================================
// Host launchers
// ===========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

void lstm_fwd_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                     torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                     torch::Tensor gates_seq, torch::Tensor c_seq,
                     c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T) {
    LstmFwdArgs a;
    a.X = (const unsigned short*)X.data_ptr();
    a.win = win.has_value() ? win->data_ptr<int>() : nullptr;
    a.wcat = (const unsigned short*)wcat.data_ptr();
    a.bias = bias.data_ptr<float>();
    a.h_seq = (unsigned short*)h_seq.data_ptr();
    a.gates_seq = (unsigned short*)gates_seq.data_ptr();
    a.c_seq = (unsigned short*)c_seq.data_ptr();
    a.x_seq_out = x_seq_out.has_value() ? (unsigned short*)x_seq_out->data_ptr() : nullptr;
    a.B = (int)B;
    a.T = (int)T;
    a.N = (int)X.size(0);
    int n_tiles = (int)((B + GA_MT - 1) / GA_MT);
    size_t smem = 2 * GA_TILE_BYTES + LSTM_4H * sizeof(float);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_fwd, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_bwd_cell_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                          torch::Tensor c_seq, torch::Tensor wcat_t,
                          torch::Tensor dgates_seq, int64_t B, int64_t T,
                          bool dh_last_only) {
    LstmBwdArgs a;
    a.dh_out = (const unsigned short*)dh_out.data_ptr();
    a.gates_seq = (const unsigned short*)gates_seq.data_ptr();
    a.c_seq = (const unsigned short*)c_seq.data_ptr();
    a.wcat_t = (const unsigned short*)wcat_t.data_ptr();
    a.dgates_seq = (unsigned short*)dgates_seq.data_ptr();
    a.B = (int)B;
    a.T = (int)T;
    a.dh_last_only = dh_last_only ? 1 : 0;
    int n_tiles = (int)((B + GA_MT - 1) / GA_MT);
    size_t smem = 5 * GA_TILE_BYTES;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_bwd_cell, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}
