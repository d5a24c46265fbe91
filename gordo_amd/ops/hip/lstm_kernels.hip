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
    unsigned char* __restrict__ gates_seq;    // [B][T][256] u8 out (activated
                                              // i,f,o as v*255; g as (g+1)/2*255)
    unsigned short* __restrict__ c_seq;       // [B][T][64] out
    unsigned short* __restrict__ x_seq_out;   // [B][T][64] gathered x copy (layer0) or null
    int B, T, N;
};

extern __shared__ __attribute__((aligned(16))) char ga_smem_lstm[];

// Gate-specialized design: wave w owns gate w. Its 64x128 weight slice lives
// entirely in REGISTERS (16 bf16x8 fragments/lane, loaded once) and its bias
// is pre-added into the MFMA accumulators — the per-timestep weight traffic
// (originally 74% of the kernel as L2 loads, then an LDS image) is zero.
// Per t: waves compute their gate pre-activation tile [64 m][64 j] -> LDS;
// a vectorized elementwise phase (thread = row x 16 cols) applies the cell
// nonlinearity on 16-B vectors and streams h/c/gates straight to HBM.
// LDS: x + h + 4 pre-act tiles = 48 KiB -> 2 blocks/CU at ~190 VGPRs.
__launch_bounds__(256, 2)
__global__ void lstm_fwd(LstmFwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;  // wave = gate
    const int base = blockIdx.x * GA_MT;

    char* x_tile = ga_smem_lstm;                     // [64][64] swizzled bf16
    char* h_tile = ga_smem_lstm + GA_TILE_BYTES;     // [64][64]
    char* pre_tiles = ga_smem_lstm + 2 * GA_TILE_BYTES;  // 4 x [64][64] pre+bias

    // ---- per-lane weight fragments (whole gate slice, registers) ---------
    // wfrag[jt][kk] = Wcat[wid*64 + jt*16 + (lane&15)][kk*32 + (lane>>4)*8 ..+8]
    ga_bf16x8 wfrag[4][4];
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int kk = 0; kk < 4; ++kk)
            wfrag[jt][kk] = *(const ga_bf16x8*)(a.wcat +
                (size_t)(wid * GA_WMAX + jt * 16 + (lane & 15)) * LSTM_K +
                kk * 32 + (lane >> 4) * 8);
    float bias_reg[4];
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
        bias_reg[jt] = a.bias[wid * GA_WMAX + jt * 16 + (lane & 15)];

    // zero h_tile (h_{-1} = 0)
    for (int c = tid; c < GA_MT * 8; c += 256) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        ga_tile_write8(h_tile, c >> 3, (c & 7) * 8, z);
    }

    // elementwise-phase mapping: row em = tid>>2, 16 cols from ec0
    const int em = tid >> 2;
    const int ec0 = (tid & 3) * 16;
    const int egi = base + em;
    float c_reg[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) c_reg[e] = 0.f;

    // x prefetch: thread owns two fixed (row, col) granules of the x tile
    const int px_m[2] = {tid >> 3, (tid >> 3) + 32};
    const int px_c = (tid & 7) * 8;
    ga_u16x8 px[2];
    auto px_load = [&](int t) {
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            int gi = base + px_m[i];
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gi < a.B) {
                if (a.win != nullptr) {
                    int row = a.win[gi] + t;
                    v = *(const ga_u16x8*)(a.X + (size_t)row * GA_WMAX + px_c);
                } else {
                    v = *(const ga_u16x8*)(a.X + ((size_t)gi * a.T + t) * GA_WMAX + px_c);
                }
            }
            px[i] = v;
        }
    };
    px_load(0);
    __syncthreads();

    for (int t = 0; t < a.T; ++t) {
        // ---- stage x_t from the prefetch registers -----------------------
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            int gi = base + px_m[i];
            if (gi < a.B && a.x_seq_out != nullptr)
                *(ga_u16x8*)(a.x_seq_out + ((size_t)gi * a.T + t) * GA_WMAX + px_c) = px[i];
            ga_tile_write8(x_tile, px_m[i], px_c, px[i]);
        }
        __syncthreads();
        if (t + 1 < a.T) px_load(t + 1);  // overlap next x load with the GEMM

        // ---- this wave's gate pre-activations: [64 m][64 j] --------------
        ga_f32x4 acc[4][4];  // [mt][jt], bias pre-added
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
            for (int jt = 0; jt < 4; ++jt)
                acc[mt][jt] = ga_f32x4{bias_reg[jt], bias_reg[jt], bias_reg[jt], bias_reg[jt]};
        for (int kk = 0; kk < 4; ++kk) {
            const char* src = (kk < 2) ? x_tile : h_tile;
            int klocal = (kk & 1) * 32;
#pragma unroll
            for (int mt = 0; mt < 4; ++mt) {
                ga_bf16x8 afrag = ga_tile_read8(src, mt * 16 + (lane & 15), klocal + (lane >> 4) * 8);
#pragma unroll
                for (int jt = 0; jt < 4; ++jt)
                    acc[mt][jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, wfrag[jt][kk], acc[mt][jt], 0, 0, 0);
            }
        }
        char* my_pre = pre_tiles + wid * GA_TILE_BYTES;
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
            for (int jt = 0; jt < 4; ++jt) {
                int col = jt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r)
                    ga_tile_write1(my_pre, mt * 16 + (lane >> 4) * 4 + r, col,
                                   ga_f32_to_bf16(acc[mt][jt][r]));
            }
        __syncthreads();

        // ---- vectorized cell update + direct HBM streaming ---------------
        {
            const bool live = egi < a.B;
            const size_t bt = (size_t)egi * a.T + t;
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                int cc = ec0 + hh * 8;
                ga_u16x8 vi = *(ga_u16x8*)&pre_tiles[ga_sw(em, cc)];
                ga_u16x8 vf = *(ga_u16x8*)&pre_tiles[GA_TILE_BYTES + ga_sw(em, cc)];
                ga_u16x8 vg = *(ga_u16x8*)&pre_tiles[2 * GA_TILE_BYTES + ga_sw(em, cc)];
                ga_u16x8 vo = *(ga_u16x8*)&pre_tiles[3 * GA_TILE_BYTES + ga_sw(em, cc)];
                ga_u16x8 oh, oc;
                float qi[8], qf[8], qg[8], qo[8];
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    float i_g = ga_sigmoid(ga_bf16_to_f32(vi[e]));
                    float f_g = ga_sigmoid(ga_bf16_to_f32(vf[e]));
                    float g_g = ga_tanh(ga_bf16_to_f32(vg[e]));
                    float o_g = ga_sigmoid(ga_bf16_to_f32(vo[e]));
                    int ei = hh * 8 + e;
                    float c_new = f_g * c_reg[ei] + i_g * g_g;
                    c_reg[ei] = c_new;
                    oh[e] = ga_f32_to_bf16(o_g * ga_tanh(c_new));
                    oc[e] = ga_f32_to_bf16(c_new);
                    qi[e] = i_g;
                    qf[e] = f_g;
                    qg[e] = g_g * 0.5f + 0.5f;
                    qo[e] = o_g;
                }
                *(ga_u16x8*)&h_tile[ga_sw(em, cc)] = oh;
                if (live) {
                    *(ga_u16x8*)(a.h_seq + bt * GA_WMAX + cc) = oh;
                    *(ga_u16x8*)(a.c_seq + bt * GA_WMAX + cc) = oc;
                    *(uint2*)(a.gates_seq + bt * LSTM_4H + cc) = ga_q8_pack8(qi);
                    *(uint2*)(a.gates_seq + bt * LSTM_4H + GA_WMAX + cc) = ga_q8_pack8(qf);
                    *(uint2*)(a.gates_seq + bt * LSTM_4H + 2 * GA_WMAX + cc) = ga_q8_pack8(qg);
                    *(uint2*)(a.gates_seq + bt * LSTM_4H + 3 * GA_WMAX + cc) = ga_q8_pack8(qo);
                }
            }
        }
        __syncthreads();  // h_tile/pre reads-writes settle before next t
    }
}

// ===========================================================================
// Backward cell: reverse time, produce dgates_seq; dc carried in registers,
// dh_rec through LDS. dh_out is either a full sequence [B][T][64] or (for the
// top layer feeding a last-step head) a single [B][64] applied at t = T-1.
// ===========================================================================
struct LstmBwdArgs {
    const unsigned short* __restrict__ dh_out;     // [B][T][64] or [B][64]
    const unsigned char* __restrict__ gates_seq;   // [B][T][256] u8 (fwd's layout)
    const unsigned short* __restrict__ c_seq;      // [B][T][64]
    const unsigned short* __restrict__ wcat_t;     // [128][256] bf16 (WcatT)
    unsigned short* __restrict__ dgates_seq;       // [B][T][256] out
    unsigned short* __restrict__ dx_seq;           // [B][T][64] out, or null:
                                                   // dx = dgates x W_ih fused
                                                   // (skips a rocBLAS pass
                                                   // re-reading dgates)
    int B, T;
    int dh_last_only;  // 1: dh_out is [B][64] at the final step only
};

// LDS: dh tile + 4 dgates tiles + the whole 64-KiB WcatT (staged once,
// swizzled — same L2-latency fix as lstm_fwd). Elementwise inputs for step
// t-1 are prefetched into registers during step t's GEMM.
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
    const int eg = tid & 3;           // col group: cols eg*16 .. eg*16+15
    const int c0 = eg * 16;
    float dc_reg[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) dc_reg[e] = 0.f;

    // stage WcatT into LDS once (rows 512 B -> swizzled 16-B slots)
    char* wT_lds = ga_smem_lstm + 5 * GA_TILE_BYTES;  // [128][256]
    for (int g = tid; g < LSTM_K * 32; g += 256) {
        int row = g >> 5, col0 = (g & 31) * 8;
        ga_u16x8 v = *(const ga_u16x8*)(a.wcat_t + (size_t)row * LSTM_4H + col0);
        *(ga_u16x8*)&wT_lds[ga_swW(row, col0 * 2, 512)] = v;
    }

    // register prefetch of the elementwise inputs (issued during the GEMM of
    // the following timestep so the HBM latency is off the critical path)
    const int pgi = base + em;
    const bool plive = pgi < a.B;
    uint2 pvi[2], pvf[2], pvg[2], pvo[2];
    ga_u16x8 pvc[2], pvcp[2], pvdh[2];
    // pvc (c_t) is only LOADED once at t = T-1; afterwards it's carried from
    // the previous iteration's pvcp — c_{t-1} would otherwise be read twice
    // (as c_prev at t, then as c_t at t-1), a full re-read of the c stream.
    auto preload = [&](int t) {
        const size_t bt = (size_t)pgi * a.T + t;
        const ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        const uint2 zq = {0, 0};
#pragma unroll
        for (int hh = 0; hh < 2; ++hh) {
            int cc = c0 + hh * 8;
            pvi[hh] = pvf[hh] = pvg[hh] = pvo[hh] = zq;
            pvcp[hh] = pvdh[hh] = z;
            if (plive) {
                pvi[hh] = *(const uint2*)(a.gates_seq + bt * LSTM_4H + cc);
                pvf[hh] = *(const uint2*)(a.gates_seq + bt * LSTM_4H + GA_WMAX + cc);
                pvg[hh] = *(const uint2*)(a.gates_seq + bt * LSTM_4H + 2 * GA_WMAX + cc);
                pvo[hh] = *(const uint2*)(a.gates_seq + bt * LSTM_4H + 3 * GA_WMAX + cc);
                if (t > 0) pvcp[hh] = *(const ga_u16x8*)(a.c_seq + (bt - 1) * GA_WMAX + cc);
                if (a.dh_last_only) {
                    if (t == a.T - 1)
                        pvdh[hh] = *(const ga_u16x8*)(a.dh_out + (size_t)pgi * GA_WMAX + cc);
                } else {
                    pvdh[hh] = *(const ga_u16x8*)(a.dh_out + bt * GA_WMAX + cc);
                }
            }
        }
    };
    {
        const ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        const size_t bt0 = (size_t)pgi * a.T + (a.T - 1);
#pragma unroll
        for (int hh = 0; hh < 2; ++hh) {
            pvc[hh] = z;
            if (plive)
                pvc[hh] = *(const ga_u16x8*)(a.c_seq + bt0 * GA_WMAX + c0 + hh * 8);
        }
    }
    preload(a.T - 1);
    __syncthreads();

    for (int t = a.T - 1; t >= 0; --t) {
        // ---- elementwise BPTT on the prefetched registers -----------------
        {
            const int gi = pgi;
            const size_t bt = (size_t)gi * a.T + t;
            const bool live = plive;
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                int cc = c0 + hh * 8;
                uint2 vi = pvi[hh], vf = pvf[hh], vg = pvg[hh], vo = pvo[hh];
                ga_u16x8 vc = pvc[hh], vcp = pvcp[hh], vdh = pvdh[hh];
                pvc[hh] = vcp;  // carry: c_{t-1} becomes next iteration's c_t
                ga_u16x8 vrec = *(ga_u16x8*)&dh_tile[ga_sw(em, cc)];
                ga_u16x8 odi, odf, odg, odo;
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    float dh = ga_bf16_to_f32(vdh[e]) + ga_bf16_to_f32(vrec[e]);
                    float i_g = ga_q8_at(vi, e);
                    float f_g = ga_q8_at(vf, e);
                    float g_g = ga_q8_at(vg, e) * 2.f - 1.f;
                    float o_g = ga_q8_at(vo, e);
                    float c_t = ga_bf16_to_f32(vc[e]);
                    float c_prev = ga_bf16_to_f32(vcp[e]);
                    float tanhc = ga_tanh(c_t);
                    int ei = hh * 8 + e;
                    float dc = dc_reg[ei] + dh * o_g * (1.f - tanhc * tanhc);
                    odo[e] = ga_f32_to_bf16(dh * tanhc * o_g * (1.f - o_g));
                    odi[e] = ga_f32_to_bf16(dc * g_g * i_g * (1.f - i_g));
                    odf[e] = ga_f32_to_bf16(dc * c_prev * f_g * (1.f - f_g));
                    odg[e] = ga_f32_to_bf16(dc * i_g * (1.f - g_g * g_g));
                    dc_reg[ei] = dc * f_g;  // carry to t-1
                }
                *(ga_u16x8*)&dg_tiles[ga_sw(em, cc)] = odi;
                *(ga_u16x8*)&dg_tiles[GA_TILE_BYTES + ga_sw(em, cc)] = odf;
                *(ga_u16x8*)&dg_tiles[2 * GA_TILE_BYTES + ga_sw(em, cc)] = odg;
                *(ga_u16x8*)&dg_tiles[3 * GA_TILE_BYTES + ga_sw(em, cc)] = odo;
                if (live) {
                    *(ga_u16x8*)(a.dgates_seq + bt * LSTM_4H + cc) = odi;
                    *(ga_u16x8*)(a.dgates_seq + bt * LSTM_4H + GA_WMAX + cc) = odf;
                    *(ga_u16x8*)(a.dgates_seq + bt * LSTM_4H + 2 * GA_WMAX + cc) = odg;
                    *(ga_u16x8*)(a.dgates_seq + bt * LSTM_4H + 3 * GA_WMAX + cc) = odo;
                }
            }
        }
        __syncthreads();

        const bool need_rec = (t > 0);
        const bool need_dx = (a.dx_seq != nullptr);
        if (!need_rec && !need_dx) break;
        if (t > 0) preload(t - 1);  // overlap next loads with the GEMM

        // ---- shared-A GEMMs over dgates tiles -----------------------------
        //   dh_rec[m][j<H] = dgates[m][256] x W_hh  (B row-read of WcatT 64+j)
        //   dx[m][j<64]    = dgates[m][256] x W_ih  (B row-read of WcatT j)
        ga_f32x4 acc[4], accx[4];
#pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
            acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            accx[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        }
        for (int kk = 0; kk < LSTM_4H; kk += 32) {
            const char* dgt = dg_tiles + (kk >> 6) * GA_TILE_BYTES;
            ga_bf16x8 afrag = ga_tile_read8(dgt, wrow + (lane & 15), (kk & 63) + (lane >> 4) * 8);
            if (need_rec)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                        GA_WMAX + nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 512)];
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            if (need_dx)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                        nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 512)];
                    accx[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, accx[nt], 0, 0, 0);
                }
        }
        // (no barrier needed here: nothing below touches dg_tiles, and the
        // end-of-iteration barrier fences the next elementwise rewrite)
        if (need_rec)
#pragma unroll
            for (int nt = 0; nt < 4; ++nt) {
                int col = nt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    ga_tile_write1(dh_tile, m, col, ga_f32_to_bf16(acc[nt][r]));
                }
            }
        if (need_dx)
#pragma unroll
            for (int nt = 0; nt < 4; ++nt) {
                int col = nt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    int gi = base + m;
                    if (gi < a.B)
                        a.dx_seq[((size_t)gi * a.T + t) * GA_WMAX + col] = ga_f32_to_bf16(accx[nt][r]);
                }
            }
        __syncthreads();
        if (!need_rec) break;
    }
}

// ===========================================================================
// lstm_dw: split-K weight-gradient kernel.
//   dWcat[4H=256][K=128] = dgates^T x [x_seq | h_{t-1}],  K-split over B*T
//   dbias[256]           = column sums of dgates        (same dgates pass)
// rocBLAS is hopeless at this shape (C is ONE 256x64 tile -> 4 workgroups for
// a K=B*T~2.4M reduction, measured 2.8 ms each); here NB blocks each reduce a
// grid-strided set of 32-row chunks with MFMA (operands staged k-contiguous
// in LDS) and write per-block partials [NB][256*128] summed by torch.sum —
// one pass over dgates replaces TWO rocBLAS GEMMs + a bf16 bias reduce + the
// hprev materialization (h_{t-1} is read with an in-kernel t-1 shift).
// ===========================================================================
#define DW_KC 32                // k-rows per staged chunk
// Row-major chunk images with padded strides: staging is all 16-B vector
// writes (the transposed scalar writes were 32-way bank-conflicted: col
// stride 80 B puts every lane of a half-wave on one bank — measured 80% of
// the kernel); the transpose happens at MFMA-fragment load time as scalar
// reads, whose bank pattern the pad spreads.
#define DW_DGS 264              // dg chunk row stride (u16): [32][264]
#define DW_ICS 136              // ic chunk row stride (u16): [32][136]

struct LstmDwArgs {
    const unsigned short* __restrict__ dgates;  // [BT][256]
    const unsigned short* __restrict__ x_seq;   // [BT][64]
    const unsigned short* __restrict__ h_seq;   // [BT][64] (read at t-1)
    float* __restrict__ partial;                // [NB][256*128]
    float* __restrict__ db_partial;             // [NB][256]
    int BT, T, NB;
};

__launch_bounds__(256)
__global__ void lstm_dw(LstmDwArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;  // wave = gate block (rows wid*64..wid*64+63)

    extern __shared__ __attribute__((aligned(16))) char smem[];
    // double-buffered: stage chunk c+1 into buf[~c&1] WHILE MFMAing buf[c&1]
    // — one barrier per chunk and the staging global-load latency hides
    // behind the MFMA phase instead of being exposed serially.
#define DW_BUF_U16 (DW_KC * (DW_DGS + DW_ICS))
    unsigned short* const smem_u16 = (unsigned short*)smem;
    float* s_db = (float*)(smem_u16 + 2 * DW_BUF_U16);  // [8][256]
#define DW_DGT(buf) (smem_u16 + (buf) * DW_BUF_U16)
#define DW_ICT(buf) (smem_u16 + (buf) * DW_BUF_U16 + DW_KC * DW_DGS)

    ga_f32x4 acc[4][8];  // [jt][ct] — C tile [64 rows of gate wid][128 cols]
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int ct = 0; ct < 8; ++ct) acc[jt][ct] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
    float db_reg[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) db_reg[e] = 0.f;

    // thread-fixed staging coordinates
    const int dg_row0 = tid >> 5, dg_col0 = (tid & 31) * 8;   // +8 rows per pass
    const int ic_row0 = tid >> 4, ic_col0 = (tid & 15) * 8;   // +16 rows per pass

    const int n_chunks = (a.BT + DW_KC - 1) / DW_KC;

    auto load_chunk = [&](int c, ga_u16x8* vdg, ga_u16x8* vic) {
        const int r0 = c * DW_KC;
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT) v = *(const ga_u16x8*)(a.dgates + (size_t)r * LSTM_4H + dg_col0);
            vdg[p] = v;
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT) {
                if (ic_col0 < GA_WMAX) {
                    v = *(const ga_u16x8*)(a.x_seq + (size_t)r * GA_WMAX + ic_col0);
                } else {
                    int t = r % a.T;
                    if (t > 0)
                        v = *(const ga_u16x8*)(a.h_seq + (size_t)(r - 1) * GA_WMAX + ic_col0 - GA_WMAX);
                }
            }
            vic[p] = v;
        }
    };
    auto write_chunk = [&](int buf, const ga_u16x8* vdg, const ga_u16x8* vic) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            *(ga_u16x8*)(DW_DGT(buf) + (size_t)row * DW_DGS + dg_col0) = vdg[p];
#pragma unroll
            for (int e = 0; e < 8; ++e) db_reg[e] += ga_bf16_to_f32(vdg[p][e]);
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            *(ga_u16x8*)(DW_ICT(buf) + (size_t)row * DW_ICS + ic_col0) = vic[p];
        }
    };

    int nb = 0;
    ga_u16x8 vdg[4], vic[2];
    load_chunk(blockIdx.x, vdg, vic);
    write_chunk(0, vdg, vic);
    __syncthreads();
    for (int c = blockIdx.x; c < n_chunks; c += a.NB) {
        const bool more = (c + a.NB) < n_chunks;
        if (more) load_chunk(c + a.NB, vdg, vic);  // loads overlap the MFMAs
        // ---- 32 MFMAs: C[64][128] += dg_chunk^T x ic_chunk ----------------
        // fragments gathered transposed: 8 scalar LDS reads each (row-major
        // image; the k rows are the slow dim)
        const int k0 = (lane >> 4) * 8;
        ga_bf16x8 bfr[8];
#pragma unroll
        for (int ct = 0; ct < 8; ++ct) {
            int colc = ct * 16 + (lane & 15);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = DW_ICT(nb)[(size_t)(k0 + e) * DW_ICS + colc];
                bfr[ct][e] = *(__bf16*)&u;
            }
        }
#pragma unroll
        for (int jt = 0; jt < 4; ++jt) {
            int j = wid * 64 + jt * 16 + (lane & 15);
            ga_bf16x8 afrag;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = DW_DGT(nb)[(size_t)(k0 + e) * DW_DGS + j];
                afrag[e] = *(__bf16*)&u;
            }
#pragma unroll
            for (int ct = 0; ct < 8; ++ct)
                acc[jt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfr[ct], acc[jt][ct], 0, 0, 0);
        }
        if (more) write_chunk(nb ^ 1, vdg, vic);  // waitcnt lands after MFMAs
        nb ^= 1;
        __syncthreads();
    }

    // ---- emit per-block partials ------------------------------------------
    float* out = a.partial + (size_t)blockIdx.x * LSTM_4H * LSTM_K;
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int ct = 0; ct < 8; ++ct)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int j = wid * 64 + jt * 16 + (lane >> 4) * 4 + r;
                int col = ct * 16 + (lane & 15);
                out[(size_t)j * LSTM_K + col] = acc[jt][ct][r];
            }
    // db: threads with the same (tid&31) cover the same 8 cols; 8 row-groups
#pragma unroll
    for (int e = 0; e < 8; ++e)
        s_db[(size_t)(tid >> 5) * LSTM_4H + (tid & 31) * 8 + e] = db_reg[e];
    __syncthreads();
    if (tid < LSTM_4H) {
        float s = 0.f;
#pragma unroll
        for (int gr = 0; gr < 8; ++gr) s += s_db[(size_t)gr * LSTM_4H + tid];
        a.db_partial[(size_t)blockIdx.x * LSTM_4H + tid] = s;
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
    TORCH_CHECK(gates_seq.scalar_type() == torch::kByte, "gates_seq must be u8");
    a.gates_seq = (unsigned char*)gates_seq.data_ptr();
    a.c_seq = (unsigned short*)c_seq.data_ptr();
    a.x_seq_out = x_seq_out.has_value() ? (unsigned short*)x_seq_out->data_ptr() : nullptr;
    a.B = (int)B;
    a.T = (int)T;
    a.N = (int)X.size(0);
    int n_tiles = (int)((B + GA_MT - 1) / GA_MT);
    size_t smem = 6 * GA_TILE_BYTES;
    static bool smem_cfg = [] {  // >64 KiB dynamic LDS needs an explicit opt-in
        (void)hipFuncSetAttribute((const void*)lstm_fwd,
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
        return true;
    }();
    (void)smem_cfg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_fwd, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_bwd_cell_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                          torch::Tensor c_seq, torch::Tensor wcat_t,
                          torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                          int64_t B, int64_t T, bool dh_last_only) {
    LstmBwdArgs a;
    a.dh_out = (const unsigned short*)dh_out.data_ptr();
    TORCH_CHECK(gates_seq.scalar_type() == torch::kByte, "gates_seq must be u8");
    a.gates_seq = (const unsigned char*)gates_seq.data_ptr();
    a.c_seq = (const unsigned short*)c_seq.data_ptr();
    a.wcat_t = (const unsigned short*)wcat_t.data_ptr();
    a.dgates_seq = (unsigned short*)dgates_seq.data_ptr();
    a.dx_seq = dx_seq.has_value() ? (unsigned short*)dx_seq->data_ptr() : nullptr;
    a.B = (int)B;
    a.T = (int)T;
    a.dh_last_only = dh_last_only ? 1 : 0;
    int n_tiles = (int)((B + GA_MT - 1) / GA_MT);
    size_t smem = 5 * GA_TILE_BYTES + LSTM_K * LSTM_4H * 2;
    static bool smem_cfg = [] {
        (void)hipFuncSetAttribute((const void*)lstm_bwd_cell,
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
        return true;
    }();
    (void)smem_cfg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_bwd_cell, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_dw_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                    torch::Tensor partial, torch::Tensor db_partial,
                    int64_t BT, int64_t T) {
    LstmDwArgs a;
    a.dgates = (const unsigned short*)dgates.data_ptr();
    a.x_seq = (const unsigned short*)x_seq.data_ptr();
    a.h_seq = (const unsigned short*)h_seq.data_ptr();
    a.partial = partial.data_ptr<float>();
    a.db_partial = db_partial.data_ptr<float>();
    a.BT = (int)BT;
    a.T = (int)T;
    a.NB = (int)partial.size(0);
    TORCH_CHECK(db_partial.size(0) == partial.size(0), "partial slab counts differ");
    size_t smem = (size_t)2 * DW_BUF_U16 * 2 + 8 * LSTM_4H * sizeof(float);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_dw, dim3(a.NB), dim3(256), smem, stream.stream(), a);
}
