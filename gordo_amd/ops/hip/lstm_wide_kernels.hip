// Wide (W=128) persistent LSTM cell kernels for MI355X (gfx950).
//
// Covers feature/hidden widths 65..128 (the ≤64 production kernels keep the
// whole gate weight chip-resident; at W=128 the packed weight is 256 KB per
// layer — past the register/LDS budget — so the MFMA B-fragments stream from
// L2, which the per-wave 256-MFMA/t compute hides). Feasibility measured in
// micro_lstm128.hip: fwd 3.75 ms at B=16384, T=144 → ~80x the GEMM-loop
// fallback for the full step.
//
// Layout: W=128 padded; Wcat [4W=512][2W=256] bf16 (cols 0..127 = W_ih,
// 128..255 = W_hh; rows gate-major i,f,g,o each padded to 128); WcatT
// [256][512]. Sequences: h/c [B][T][128], gates/dgates [B][T][512].
#include "common.h"

#define WW 128
#define W4H (4 * WW)   // 512
#define WK (2 * WW)    // 256
#define WTB (64 * WW * 2)  // [64][128] tile bytes (16 KiB)

// [64][128] bf16 tile, 256-B rows, swizzled 16-B slots
__device__ __forceinline__ unsigned int sw128(int row, int col) {
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 7u) << 4);
    return (unsigned int)row * (WW * 2) + byte;
}

extern __shared__ __attribute__((aligned(16))) char wsm[];

// ===========================================================================
// forward: wave = gate; pre-activations -> LDS; vectorized cell epilogue
// streams h/c/gates to HBM. Layer 0 gathers windows from the dataset.
// ===========================================================================
struct WFwdArgs {
    const unsigned short* __restrict__ X;     // layer0: dataset [N][128]; else [B][T][128]
    const int* __restrict__ win;              // [B] window starts (layer0) or null
    const unsigned short* __restrict__ wcat;  // [512][256]
    const float* __restrict__ bias;           // [512]
    unsigned short* __restrict__ h_seq;       // [B][T][128]
    unsigned char* __restrict__ gates_seq;    // [B][T][512] u8 (i,f,o v*255;
                                              // g (g+1)/2*255 — see common.h)
    unsigned short* __restrict__ c_seq;       // [B][T][128]
    unsigned short* __restrict__ x_seq_out;   // [B][T][128] or null
    int B, T;
};

__launch_bounds__(256)
__global__ void lstm_fwd_wide(WFwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;  // wave = gate
    const int base = blockIdx.x * 64;
    char* x_tile = wsm;
    char* h_tile = wsm + WTB;
    char* pre_tiles = wsm + 2 * WTB;  // 4 x [64][128]

    float bias_reg[8];
#pragma unroll
    for (int jt = 0; jt < 8; ++jt)
        bias_reg[jt] = a.bias[wid * WW + jt * 16 + (lane & 15)];

    for (int c = tid; c < 64 * 16; c += 256) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *(ga_u16x8*)&h_tile[sw128(c >> 4, (c & 15) * 8)] = z;
    }
    const int em = tid >> 2;
    const int ec0 = (tid & 3) * 32;
    const int egi = base + em;
    float c_reg[32];
#pragma unroll
    for (int e = 0; e < 32; ++e) c_reg[e] = 0.f;
    __syncthreads();

    const unsigned short* wcat_p = a.wcat;
    for (int t = 0; t < a.T; ++t) {
        // the weight addresses are loop-invariant; without this barrier the
        // compiler hoists ~100 B-fragments into registers across the t-loop
        // and spills (measured 20-96 spills/lane, occupancy 1 wave/SIMD)
        asm volatile("" : "+s"(wcat_p));
        // ---- stage x_t (window gather on layer 0) -------------------------
        for (int c = tid; c < 64 * 16; c += 256) {
            int m = c >> 4, col0 = (c & 15) * 8;
            int gi = base + m;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gi < a.B) {
                if (a.win != nullptr) {
                    int row = a.win[gi] + t;
                    v = *(const ga_u16x8*)(a.X + (size_t)row * WW + col0);
                } else {
                    v = *(const ga_u16x8*)(a.X + ((size_t)gi * a.T + t) * WW + col0);
                }
                if (a.x_seq_out != nullptr)
                    *(ga_u16x8*)(a.x_seq_out + ((size_t)gi * a.T + t) * WW + col0) = v;
            }
            *(ga_u16x8*)&x_tile[sw128(m, col0)] = v;
        }
        __syncthreads();

        // ---- gate pre-activations: C [64 m][128 j], K = 256, B-frags L2 ---
        ga_f32x4 acc[4][8];
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
            for (int jt = 0; jt < 8; ++jt)
                acc[mt][jt] = ga_f32x4{bias_reg[jt], bias_reg[jt], bias_reg[jt], bias_reg[jt]};
        for (int kk = 0; kk < 8; ++kk) {
            const char* src = (kk < 4) ? x_tile : h_tile;
            int klocal = (kk & 3) * 32;
#pragma unroll
            for (int mt = 0; mt < 4; ++mt) {
                ga_bf16x8 afrag = *(const ga_bf16x8*)&src[sw128(mt * 16 + (lane & 15), klocal + (lane >> 4) * 8)];
#pragma unroll
                for (int jt = 0; jt < 8; ++jt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)(wcat_p +
                        (size_t)(wid * WW + jt * 16 + (lane & 15)) * WK + kk * 32 + (lane >> 4) * 8);
                    acc[mt][jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mt][jt], 0, 0, 0);
                }
            }
        }
        char* my_pre = pre_tiles + wid * WTB;
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
            for (int jt = 0; jt < 8; ++jt) {
                int col = jt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r)
                    *(unsigned short*)&my_pre[sw128(mt * 16 + (lane >> 4) * 4 + r, col)] =
                        ga_f32_to_bf16(acc[mt][jt][r]);
            }
        __syncthreads();

        // ---- vectorized cell update + HBM streaming -----------------------
        {
            const bool live = egi < a.B;
            const size_t bt = (size_t)egi * a.T + t;
#pragma unroll
            for (int hh = 0; hh < 4; ++hh) {
                int cc = ec0 + hh * 8;
                ga_u16x8 vi = *(ga_u16x8*)&pre_tiles[sw128(em, cc)];
                ga_u16x8 vf = *(ga_u16x8*)&pre_tiles[WTB + sw128(em, cc)];
                ga_u16x8 vg = *(ga_u16x8*)&pre_tiles[2 * WTB + sw128(em, cc)];
                ga_u16x8 vo = *(ga_u16x8*)&pre_tiles[3 * WTB + sw128(em, cc)];
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
                *(ga_u16x8*)&h_tile[sw128(em, cc)] = oh;
                if (live) {
                    *(ga_u16x8*)(a.h_seq + bt * WW + cc) = oh;
                    *(ga_u16x8*)(a.c_seq + bt * WW + cc) = oc;
                    *(uint2*)(a.gates_seq + bt * W4H + cc) = ga_q8_pack8(qi);
                    *(uint2*)(a.gates_seq + bt * W4H + WW + cc) = ga_q8_pack8(qf);
                    *(uint2*)(a.gates_seq + bt * W4H + 2 * WW + cc) = ga_q8_pack8(qg);
                    *(uint2*)(a.gates_seq + bt * W4H + 3 * WW + cc) = ga_q8_pack8(qo);
                }
            }
        }
        __syncthreads();
    }
}

// ===========================================================================
// backward: reverse-time BPTT; dc in registers, dh_rec via LDS; dh_rec and
// fused-dx GEMMs stream WcatT fragments from L2.
// ===========================================================================
struct WBwdArgs {
    const unsigned short* __restrict__ dh_out;     // [B][T][128] or [B][128]
    const unsigned char* __restrict__ gates_seq;   // [B][T][512] u8
    const unsigned short* __restrict__ c_seq;      // [B][T][128]
    const unsigned short* __restrict__ wcat_t;     // [256][512]
    unsigned short* __restrict__ dgates_seq;       // [B][T][512] out
    unsigned short* __restrict__ dx_seq;           // [B][T][128] out or null
    int B, T;
    int dh_last_only;
};

__launch_bounds__(256)
__global__ void lstm_bwd_wide(WBwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * 64;
    const int wrow = wid * 16;  // this wave's 16-row stripe for the GEMMs
    char* dh_tile = wsm;                 // [64][128]
    char* dg_tiles = wsm + WTB;          // 4 x [64][128]

    for (int c = tid; c < 64 * 16; c += 256) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *(ga_u16x8*)&dh_tile[sw128(c >> 4, (c & 15) * 8)] = z;
    }
    const int em = tid >> 2;
    const int c0 = (tid & 3) * 32;
    float dc_reg[32];
#pragma unroll
    for (int e = 0; e < 32; ++e) dc_reg[e] = 0.f;
    __syncthreads();

    const unsigned short* wct_p = a.wcat_t;
    for (int t = a.T - 1; t >= 0; --t) {
        asm volatile("" : "+s"(wct_p));  // see lstm_fwd_wide: stop weight-LICM
        // ---- elementwise BPTT (16-B vector IO) ----------------------------
        {
            const int gi = base + em;
            const size_t bt = (size_t)gi * a.T + t;
            const bool live = gi < a.B;
            const ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
            for (int hh = 0; hh < 4; ++hh) {
                int cc = c0 + hh * 8;
                uint2 vi = {0, 0}, vf = {0, 0}, vg = {0, 0}, vo = {0, 0};
                ga_u16x8 vc = z, vcp = z, vdh = z;
                if (live) {
                    vi = *(const uint2*)(a.gates_seq + bt * W4H + cc);
                    vf = *(const uint2*)(a.gates_seq + bt * W4H + WW + cc);
                    vg = *(const uint2*)(a.gates_seq + bt * W4H + 2 * WW + cc);
                    vo = *(const uint2*)(a.gates_seq + bt * W4H + 3 * WW + cc);
                    vc = *(const ga_u16x8*)(a.c_seq + bt * WW + cc);
                    if (t > 0) vcp = *(const ga_u16x8*)(a.c_seq + (bt - 1) * WW + cc);
                    if (a.dh_last_only) {
                        if (t == a.T - 1)
                            vdh = *(const ga_u16x8*)(a.dh_out + (size_t)gi * WW + cc);
                    } else {
                        vdh = *(const ga_u16x8*)(a.dh_out + bt * WW + cc);
                    }
                }
                ga_u16x8 vrec = *(ga_u16x8*)&dh_tile[sw128(em, cc)];
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
                    dc_reg[ei] = dc * f_g;
                }
                *(ga_u16x8*)&dg_tiles[sw128(em, cc)] = odi;
                *(ga_u16x8*)&dg_tiles[WTB + sw128(em, cc)] = odf;
                *(ga_u16x8*)&dg_tiles[2 * WTB + sw128(em, cc)] = odg;
                *(ga_u16x8*)&dg_tiles[3 * WTB + sw128(em, cc)] = odo;
                if (live) {
                    *(ga_u16x8*)(a.dgates_seq + bt * W4H + cc) = odi;
                    *(ga_u16x8*)(a.dgates_seq + bt * W4H + WW + cc) = odf;
                    *(ga_u16x8*)(a.dgates_seq + bt * W4H + 2 * WW + cc) = odg;
                    *(ga_u16x8*)(a.dgates_seq + bt * W4H + 3 * WW + cc) = odo;
                }
            }
        }
        __syncthreads();

        const bool need_rec = (t > 0);
        const bool need_dx = (a.dx_seq != nullptr);
        if (!need_rec && !need_dx) break;

        // ---- GEMMs over dgates tiles (B-frags from L2) --------------------
        //   dh_rec[m][j<W] = dgates[m][512] x W_hh (WcatT rows W+j)
        //   dx[m][j<W]     = dgates[m][512] x W_ih (WcatT rows j)
        // When both outputs are needed, wave pairs specialize (0,1 -> rec;
        // 2,3 -> dx) and each wave covers TWO 16-row stripes per B-fragment
        // load — halving the L2 loads per MFMA vs every wave doing both.
        if (need_rec && need_dx) {
            const int otype = wid >> 1;   // 0 = rec, 1 = dx
            const int sub = wid & 1;      // row half: rows sub*32 .. +32
            const int brow0 = (otype == 0) ? WW : 0;
            ga_f32x4 acc2[2][8];
#pragma unroll
            for (int st2 = 0; st2 < 2; ++st2)
#pragma unroll
                for (int nt = 0; nt < 8; ++nt) acc2[st2][nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            for (int kk = 0; kk < W4H; kk += 32) {
                const char* dgt = dg_tiles + (kk >> 7) * WTB;
                ga_bf16x8 bfrag[8];
#pragma unroll
                for (int nt = 0; nt < 8; ++nt)
                    bfrag[nt] = *(const ga_bf16x8*)(wct_p +
                        (size_t)(brow0 + nt * 16 + (lane & 15)) * W4H + kk + (lane >> 4) * 8);
#pragma unroll
                for (int st2 = 0; st2 < 2; ++st2) {
                    ga_bf16x8 afrag = *(const ga_bf16x8*)&dgt[sw128(
                        sub * 32 + st2 * 16 + (lane & 15), (kk & 127) + (lane >> 4) * 8)];
#pragma unroll
                    for (int nt = 0; nt < 8; ++nt)
                        acc2[st2][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag[nt], acc2[st2][nt], 0, 0, 0);
                }
            }
#pragma unroll
            for (int st2 = 0; st2 < 2; ++st2)
#pragma unroll
                for (int nt = 0; nt < 8; ++nt) {
                    int col = nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = sub * 32 + st2 * 16 + (lane >> 4) * 4 + r;
                        if (otype == 0) {
                            *(unsigned short*)&dh_tile[sw128(m, col)] =
                                ga_f32_to_bf16(acc2[st2][nt][r]);
                        } else {
                            int gi = base + m;
                            if (gi < a.B)
                                a.dx_seq[((size_t)gi * a.T + t) * WW + col] =
                                    ga_f32_to_bf16(acc2[st2][nt][r]);
                        }
                    }
                }
        } else {
            // single output: all 4 waves, one 16-row stripe each
            const bool rec_mode = need_rec;
            const int brow0 = rec_mode ? WW : 0;
            ga_f32x4 acc[8];
#pragma unroll
            for (int nt = 0; nt < 8; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            for (int kk = 0; kk < W4H; kk += 32) {
                const char* dgt = dg_tiles + (kk >> 7) * WTB;
                ga_bf16x8 afrag = *(const ga_bf16x8*)&dgt[sw128(wrow + (lane & 15), (kk & 127) + (lane >> 4) * 8)];
#pragma unroll
                for (int nt = 0; nt < 8; ++nt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)(wct_p +
                        (size_t)(brow0 + nt * 16 + (lane & 15)) * W4H + kk + (lane >> 4) * 8);
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            }
#pragma unroll
            for (int nt = 0; nt < 8; ++nt) {
                int col = nt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    if (rec_mode) {
                        *(unsigned short*)&dh_tile[sw128(m, col)] = ga_f32_to_bf16(acc[nt][r]);
                    } else {
                        int gi = base + m;
                        if (gi < a.B)
                            a.dx_seq[((size_t)gi * a.T + t) * WW + col] = ga_f32_to_bf16(acc[nt][r]);
                    }
                }
            }
        }
        __syncthreads();
        if (!need_rec) break;
    }
}

// ===========================================================================
// dW: split-K over B*T in output quadrants. Quadrant (gh, ih) computes
// C[gh*256..+256][ih*128..+128] of dWcat[512][256]: dgates gate-half gh vs
// ic half ih (0 = x_seq, 1 = h_{t-1}). Same [256][128] C shape, double-
// buffered row-major staging and read-side transpose as the W=64 kernel.
// ===========================================================================
#define WDW_KC 32
#define WDW_DGS 264   // dg chunk row stride (u16): [32][264]
#define WDW_ICS 136   // ic chunk row stride (u16): [32][136]
#define WDW_BUF (WDW_KC * (WDW_DGS + WDW_ICS))

struct WDwArgs {
    const unsigned short* __restrict__ dgates;  // [BT][512]
    const unsigned short* __restrict__ x_seq;   // [BT][128]
    const unsigned short* __restrict__ h_seq;   // [BT][128] (read at t-1)
    float* __restrict__ partial;                // [2][2][NB][256*128]
    float* __restrict__ db_partial;             // [2][NB][256] (ih==0 only)
    int BT, T, NB;
};

__launch_bounds__(256)
__global__ void lstm_dw_wide(WDwArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int gh = blockIdx.y;   // dgates gate-half (rows gh*256..+256)
    const int ih = blockIdx.z;   // ic half (0 = x, 1 = hprev)
    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* const smem_u16 = (unsigned short*)smem;
    float* s_db = (float*)(smem_u16 + 2 * WDW_BUF);  // [8][256]
#define WDGT(buf) (smem_u16 + (buf) * WDW_BUF)
#define WICT(buf) (smem_u16 + (buf) * WDW_BUF + WDW_KC * WDW_DGS)

    ga_f32x4 acc[4][8];
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int ct = 0; ct < 8; ++ct) acc[jt][ct] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
    float db_reg[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) db_reg[e] = 0.f;

    const int dg_row0 = tid >> 5, dg_col0 = (tid & 31) * 8;
    const int ic_row0 = tid >> 4, ic_col0 = (tid & 15) * 8;
    const int n_chunks = (a.BT + WDW_KC - 1) / WDW_KC;

    auto load_chunk = [&](int c, ga_u16x8* vdg, ga_u16x8* vic) {
        const int r0 = c * WDW_KC;
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT)
                v = *(const ga_u16x8*)(a.dgates + (size_t)r * W4H + gh * 256 + dg_col0);
            vdg[p] = v;
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT) {
                if (ih == 0) {
                    v = *(const ga_u16x8*)(a.x_seq + (size_t)r * WW + ic_col0);
                } else {
                    int t = r % a.T;
                    if (t > 0)
                        v = *(const ga_u16x8*)(a.h_seq + (size_t)(r - 1) * WW + ic_col0);
                }
            }
            vic[p] = v;
        }
    };
    auto write_chunk = [&](int buf, const ga_u16x8* vdg, const ga_u16x8* vic) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            *(ga_u16x8*)(WDGT(buf) + (size_t)row * WDW_DGS + dg_col0) = vdg[p];
#pragma unroll
            for (int e = 0; e < 8; ++e) db_reg[e] += ga_bf16_to_f32(vdg[p][e]);
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            *(ga_u16x8*)(WICT(buf) + (size_t)row * WDW_ICS + ic_col0) = vic[p];
        }
    };

    int nb = 0;
    ga_u16x8 vdg[4], vic[2];
    load_chunk(blockIdx.x, vdg, vic);
    write_chunk(0, vdg, vic);
    __syncthreads();
    for (int c = blockIdx.x; c < n_chunks; c += a.NB) {
        const bool more = (c + a.NB) < n_chunks;
        if (more) load_chunk(c + a.NB, vdg, vic);
        const int k0 = (lane >> 4) * 8;
        ga_bf16x8 bfr[8];
#pragma unroll
        for (int ct = 0; ct < 8; ++ct) {
            int colc = ct * 16 + (lane & 15);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = WICT(nb)[(size_t)(k0 + e) * WDW_ICS + colc];
                bfr[ct][e] = *(__bf16*)&u;
            }
        }
#pragma unroll
        for (int jt = 0; jt < 4; ++jt) {
            int j = wid * 64 + jt * 16 + (lane & 15);
            ga_bf16x8 afrag;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = WDGT(nb)[(size_t)(k0 + e) * WDW_DGS + j];
                afrag[e] = *(__bf16*)&u;
            }
#pragma unroll
            for (int ct = 0; ct < 8; ++ct)
                acc[jt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfr[ct], acc[jt][ct], 0, 0, 0);
        }
        if (more) write_chunk(nb ^ 1, vdg, vic);
        nb ^= 1;
        __syncthreads();
    }

    // partial layout: [(gh*2 + ih)][NB][256*128]
    float* out = a.partial +
        ((size_t)(gh * 2 + ih) * a.NB + blockIdx.x) * 256 * WW;
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int ct = 0; ct < 8; ++ct)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int j = wid * 64 + jt * 16 + (lane >> 4) * 4 + r;
                int col = ct * 16 + (lane & 15);
                out[(size_t)j * WW + col] = acc[jt][ct][r];
            }
    if (ih == 0) {  // dbias counted once per gate-half
#pragma unroll
        for (int e = 0; e < 8; ++e)
            s_db[(size_t)(tid >> 5) * 256 + (tid & 31) * 8 + e] = db_reg[e];
        __syncthreads();
        if (tid < 256) {
            float ssum = 0.f;
#pragma unroll
            for (int gr = 0; gr < 8; ++gr) ssum += s_db[(size_t)gr * 256 + tid];
            a.db_partial[((size_t)gh * a.NB + blockIdx.x) * 256 + tid] = ssum;
        }
    }
}

// ===========================================================================
// Host launchers
// ===========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "launchers.h"

void lstm_fwd_wide_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                          torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                          torch::Tensor gates_seq, torch::Tensor c_seq,
                          c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T) {
    WFwdArgs a;
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
    int n_tiles = (int)((B + 63) / 64);
    size_t smem = 6 * WTB;  // 96 KiB
    static bool cfg = [] {
        (void)hipFuncSetAttribute((const void*)lstm_fwd_wide,
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
        return true;
    }();
    (void)cfg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_fwd_wide, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_bwd_wide_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                          torch::Tensor c_seq, torch::Tensor wcat_t,
                          torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                          int64_t B, int64_t T, bool dh_last_only) {
    WBwdArgs a;
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
    int n_tiles = (int)((B + 63) / 64);
    size_t smem = 5 * WTB;  // 80 KiB
    static bool cfg = [] {
        (void)hipFuncSetAttribute((const void*)lstm_bwd_wide,
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
        return true;
    }();
    (void)cfg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_bwd_wide, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_dw_wide_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                         torch::Tensor partial, torch::Tensor db_partial,
                         int64_t BT, int64_t T) {
    WDwArgs a;
    a.dgates = (const unsigned short*)dgates.data_ptr();
    a.x_seq = (const unsigned short*)x_seq.data_ptr();
    a.h_seq = (const unsigned short*)h_seq.data_ptr();
    a.partial = partial.data_ptr<float>();
    a.db_partial = db_partial.data_ptr<float>();
    a.BT = (int)BT;
    a.T = (int)T;
    a.NB = (int)partial.size(1);
    TORCH_CHECK(partial.size(0) == 4, "partial must be [4][NB][256*128]");
    TORCH_CHECK(db_partial.size(0) == 2 && db_partial.size(1) == partial.size(1),
                "db_partial must be [2][NB][256]");
    size_t smem = (size_t)2 * WDW_BUF * 2 + 8 * 256 * sizeof(float);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_dw_wide, dim3(a.NB, 2, 2), dim3(256), smem,
                       stream.stream(), a);
}
