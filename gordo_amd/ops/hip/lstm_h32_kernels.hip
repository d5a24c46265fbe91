// W=32 narrow LSTM cell kernels for MI355X (gfx950).
//
// Fourth width tier (real widths <= 32): the 50-tag headline stack's two
// 25-wide cells padded to W=64 spent half their elementwise VALU work and
// sequence bandwidth on padding. At H=32 the per-wave weight slice is
// [32][64] = 4 KiB (4 register fragments), the per-timestep GEMM is 16
// MFMAs/wave, every h/c/gates/dgates stream halves, and the per-thread
// elementwise vector count halves vs W=64. Structure is the W=64 family's
// (gate-specialized waves, register weights, register prefetch of x and of
// the BPTT inputs, c/dc in registers) at half the lane width; per-layer
// width tiers (lstm_engine) make it drop-in for mixed stacks.
#include "common.h"

#define L32_H 32
#define L32_4H 128
#define L32_K 64               // cat(x, h)
#define T32B (GA_MT * 32 * 2)  // [64][32] bf16 tile bytes (4 KiB)

extern __shared__ __attribute__((aligned(16))) char l32_smem[];

struct L32FwdArgs {
    const unsigned short* __restrict__ X;     // layer0: dataset [N][32]; else seq [B][T][32]
    const int* __restrict__ win;              // [B] window start rows (layer0) or null
    const unsigned short* __restrict__ wcat;  // [128][64] bf16
    const float* __restrict__ bias;           // [128] fp32
    unsigned short* __restrict__ h_seq;       // [B][T][32] out
    unsigned char* __restrict__ gates_seq;    // [B][T][128] u8 out
    unsigned short* __restrict__ c_seq;       // [B][T][32] out
    unsigned short* __restrict__ x_seq_out;   // [B][T][32] or null
    int B, T, N;
};

__launch_bounds__(256)
__global__ void lstm_fwd_h32(L32FwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;  // wave = gate
    const int base = blockIdx.x * GA_MT;

    char* x_tile = l32_smem;              // [64][32] swizzled bf16
    char* h_tile = l32_smem + T32B;       // [64][32]
    char* pre_tiles = l32_smem + 2 * T32B;  // 4 x [64][32]

    // whole gate weight slice [32 j][64 k] in registers: 4 fragments/lane
    ga_bf16x8 wfrag[2][2];
#pragma unroll
    for (int jt = 0; jt < 2; ++jt)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
            wfrag[jt][kk] = *(const ga_bf16x8*)(a.wcat +
                (size_t)(wid * L32_H + jt * 16 + (lane & 15)) * L32_K +
                kk * 32 + (lane >> 4) * 8);
    float bias_reg[2];
#pragma unroll
    for (int jt = 0; jt < 2; ++jt)
        bias_reg[jt] = a.bias[wid * L32_H + jt * 16 + (lane & 15)];

    // zero h_tile (h_{-1} = 0): [64][32] = 256 granules of 8
    {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *(ga_u16x8*)&h_tile[ga_sw32(tid >> 2, (tid & 3) * 8)] = z;
    }

    // elementwise mapping: row em = tid>>2, ONE 8-col vector from ec0
    const int em = tid >> 2;
    const int ec0 = (tid & 3) * 8;
    const int egi = base + em;
    float c_reg[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) c_reg[e] = 0.f;

    // x prefetch: thread owns one fixed (row, col) granule
    const int px_m = tid >> 2;
    const int px_c = (tid & 3) * 8;
    ga_u16x8 px;
    auto px_load = [&](int t) {
        int gi = base + px_m;
        ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gi < a.B) {
            if (a.win != nullptr) {
                int row = a.win[gi] + t;
                v = *(const ga_u16x8*)(a.X + (size_t)row * L32_H + px_c);
            } else {
                v = *(const ga_u16x8*)(a.X + ((size_t)gi * a.T + t) * L32_H + px_c);
            }
        }
        px = v;
    };
    px_load(0);
    __syncthreads();

    for (int t = 0; t < a.T; ++t) {
        // ---- stage x_t from the prefetch register -----------------------
        {
            int gi = base + px_m;
            if (gi < a.B && a.x_seq_out != nullptr)
                *(ga_u16x8*)(a.x_seq_out + ((size_t)gi * a.T + t) * L32_H + px_c) = px;
            *(ga_u16x8*)&x_tile[ga_sw32(px_m, px_c)] = px;
        }
        __syncthreads();
        if (t + 1 < a.T) px_load(t + 1);

        // ---- this wave's gate pre-activations: [64 m][32 j] --------------
        ga_f32x4 acc[4][2];
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
            for (int jt = 0; jt < 2; ++jt)
                acc[mt][jt] = ga_f32x4{bias_reg[jt], bias_reg[jt], bias_reg[jt], bias_reg[jt]};
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            const char* src = (kk == 0) ? x_tile : h_tile;
#pragma unroll
            for (int mt = 0; mt < 4; ++mt) {
                ga_bf16x8 afrag = ga_t32_read8(src, mt * 16 + (lane & 15), (lane >> 4) * 8);
#pragma unroll
                for (int jt = 0; jt < 2; ++jt)
                    acc[mt][jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, wfrag[jt][kk], acc[mt][jt], 0, 0, 0);
            }
        }
        char* my_pre = pre_tiles + wid * T32B;
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
            for (int jt = 0; jt < 2; ++jt) {
                int col = jt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r)
                    ga_t32_write1(my_pre, mt * 16 + (lane >> 4) * 4 + r, col,
                                  ga_f32_to_bf16(acc[mt][jt][r]));
            }
        __syncthreads();

        // ---- vectorized cell update + direct HBM streaming ---------------
        {
            const bool live = egi < a.B;
            const size_t bt = (size_t)egi * a.T + t;
            const int cc = ec0;
            ga_u16x8 vi = *(ga_u16x8*)&pre_tiles[ga_sw32(em, cc)];
            ga_u16x8 vf = *(ga_u16x8*)&pre_tiles[T32B + ga_sw32(em, cc)];
            ga_u16x8 vg = *(ga_u16x8*)&pre_tiles[2 * T32B + ga_sw32(em, cc)];
            ga_u16x8 vo = *(ga_u16x8*)&pre_tiles[3 * T32B + ga_sw32(em, cc)];
            ga_u16x8 oh, oc;
            float qi[8], qf[8], qg[8], qo[8];
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                float i_g = ga_sigmoid(ga_bf16_to_f32(vi[e]));
                float f_g = ga_sigmoid(ga_bf16_to_f32(vf[e]));
                float g_g = ga_tanh(ga_bf16_to_f32(vg[e]));
                float o_g = ga_sigmoid(ga_bf16_to_f32(vo[e]));
                float c_new = f_g * c_reg[e] + i_g * g_g;
                c_reg[e] = c_new;
                oh[e] = ga_f32_to_bf16(o_g * ga_tanh(c_new));
                oc[e] = ga_f32_to_bf16(c_new);
                qi[e] = i_g;
                qf[e] = f_g;
                qg[e] = g_g * 0.5f + 0.5f;
                qo[e] = o_g;
            }
            *(ga_u16x8*)&h_tile[ga_sw32(em, cc)] = oh;
            if (live) {
                *(ga_u16x8*)(a.h_seq + bt * L32_H + cc) = oh;
                *(ga_u16x8*)(a.c_seq + bt * L32_H + cc) = oc;
                *(uint2*)(a.gates_seq + bt * L32_4H + cc) = ga_q8_pack8(qi);
                *(uint2*)(a.gates_seq + bt * L32_4H + L32_H + cc) = ga_q8_pack8(qf);
                *(uint2*)(a.gates_seq + bt * L32_4H + 2 * L32_H + cc) = ga_q8_pack8(qg);
                *(uint2*)(a.gates_seq + bt * L32_4H + 3 * L32_H + cc) = ga_q8_pack8(qo);
            }
        }
        __syncthreads();
    }
}

// ===========================================================================
// Backward cell (W=32): reverse-time; dc in registers; WcatT [64][128] LDS.
// ===========================================================================
struct L32BwdArgs {
    const unsigned short* __restrict__ dh_out;     // [B][T][32] or [B][32]
    const unsigned char* __restrict__ gates_seq;   // [B][T][128] u8
    const unsigned short* __restrict__ c_seq;      // [B][T][32]
    const unsigned short* __restrict__ wcat_t;     // [64][128] bf16
    unsigned short* __restrict__ dgates_seq;       // [B][T][128] out
    unsigned short* __restrict__ dx_seq;           // [B][T][32] out or null
    int B, T;
    int dh_last_only;
};

__launch_bounds__(256)
__global__ void lstm_bwd_h32(L32BwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * GA_MT;
    const int wrow = wid * 16;

    char* dh_tile = l32_smem;              // [64][32]
    char* dg_tiles = l32_smem + T32B;      // 4 x [64][32]

    {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *(ga_u16x8*)&dh_tile[ga_sw32(tid >> 2, (tid & 3) * 8)] = z;
    }
    const int em = tid >> 2;
    const int c0 = (tid & 3) * 8;
    float dc_reg[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) dc_reg[e] = 0.f;

    // stage WcatT into LDS once: [64][128], 256-B rows, swizzled slots
    char* wT_lds = l32_smem + 5 * T32B;
    for (int g = tid; g < L32_K * 16; g += 256) {
        int row = g >> 4, col0 = (g & 15) * 8;
        ga_u16x8 v = *(const ga_u16x8*)(a.wcat_t + (size_t)row * L32_4H + col0);
        *(ga_u16x8*)&wT_lds[ga_swW(row, col0 * 2, 256)] = v;
    }

    const int pgi = base + em;
    const bool plive = pgi < a.B;
    uint2 pvi, pvf, pvg, pvo;
    ga_u16x8 pvc, pvcp, pvdh;
    auto preload = [&](int t) {
        const size_t bt = (size_t)pgi * a.T + t;
        const ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        const uint2 zq = {0, 0};
        pvi = pvf = pvg = pvo = zq;
        pvcp = pvdh = z;
        if (plive) {
            pvi = *(const uint2*)(a.gates_seq + bt * L32_4H + c0);
            pvf = *(const uint2*)(a.gates_seq + bt * L32_4H + L32_H + c0);
            pvg = *(const uint2*)(a.gates_seq + bt * L32_4H + 2 * L32_H + c0);
            pvo = *(const uint2*)(a.gates_seq + bt * L32_4H + 3 * L32_H + c0);
            if (t > 0) pvcp = *(const ga_u16x8*)(a.c_seq + (bt - 1) * L32_H + c0);
            if (a.dh_last_only) {
                if (t == a.T - 1)
                    pvdh = *(const ga_u16x8*)(a.dh_out + (size_t)pgi * L32_H + c0);
            } else {
                pvdh = *(const ga_u16x8*)(a.dh_out + bt * L32_H + c0);
            }
        }
    };
    {
        const ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        pvc = z;
        if (plive)
            pvc = *(const ga_u16x8*)(a.c_seq + ((size_t)pgi * a.T + a.T - 1) * L32_H + c0);
    }
    preload(a.T - 1);
    __syncthreads();

    for (int t = a.T - 1; t >= 0; --t) {
        // ---- elementwise BPTT on the prefetched registers -----------------
        {
            const size_t bt = (size_t)pgi * a.T + t;
            uint2 vi = pvi, vf = pvf, vg = pvg, vo = pvo;
            ga_u16x8 vc = pvc, vcp = pvcp, vdh = pvdh;
            pvc = vcp;  // carry: c_{t-1} becomes next iteration's c_t
            ga_u16x8 vrec = *(ga_u16x8*)&dh_tile[ga_sw32(em, c0)];
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
                float dc = dc_reg[e] + dh * o_g * (1.f - tanhc * tanhc);
                odo[e] = ga_f32_to_bf16(dh * tanhc * o_g * (1.f - o_g));
                odi[e] = ga_f32_to_bf16(dc * g_g * i_g * (1.f - i_g));
                odf[e] = ga_f32_to_bf16(dc * c_prev * f_g * (1.f - f_g));
                odg[e] = ga_f32_to_bf16(dc * i_g * (1.f - g_g * g_g));
                dc_reg[e] = dc * f_g;
            }
            *(ga_u16x8*)&dg_tiles[ga_sw32(em, c0)] = odi;
            *(ga_u16x8*)&dg_tiles[T32B + ga_sw32(em, c0)] = odf;
            *(ga_u16x8*)&dg_tiles[2 * T32B + ga_sw32(em, c0)] = odg;
            *(ga_u16x8*)&dg_tiles[3 * T32B + ga_sw32(em, c0)] = odo;
            if (plive) {
                *(ga_u16x8*)(a.dgates_seq + bt * L32_4H + c0) = odi;
                *(ga_u16x8*)(a.dgates_seq + bt * L32_4H + L32_H + c0) = odf;
                *(ga_u16x8*)(a.dgates_seq + bt * L32_4H + 2 * L32_H + c0) = odg;
                *(ga_u16x8*)(a.dgates_seq + bt * L32_4H + 3 * L32_H + c0) = odo;
            }
        }
        __syncthreads();

        const bool need_rec = (t > 0);
        const bool need_dx = (a.dx_seq != nullptr);
        if (!need_rec && !need_dx) break;
        if (t > 0) preload(t - 1);

        // ---- shared-A GEMMs over dgates tiles -----------------------------
        ga_f32x4 acc[2], accx[2];
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
            acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            accx[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        }
#pragma unroll
        for (int kk = 0; kk < L32_4H; kk += 32) {
            const char* dgt = dg_tiles + (kk >> 5) * T32B;
            ga_bf16x8 afrag = ga_t32_read8(dgt, wrow + (lane & 15), (lane >> 4) * 8);
            if (need_rec)
#pragma unroll
                for (int nt = 0; nt < 2; ++nt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                        L32_H + nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 256)];
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            if (need_dx)
#pragma unroll
                for (int nt = 0; nt < 2; ++nt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                        nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 256)];
                    accx[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, accx[nt], 0, 0, 0);
                }
        }
        if (need_rec)
#pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                int col = nt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    ga_t32_write1(dh_tile, m, col, ga_f32_to_bf16(acc[nt][r]));
                }
            }
        if (need_dx)
#pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                int col = nt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    int gi = base + m;
                    if (gi < a.B)
                        a.dx_seq[((size_t)gi * a.T + t) * L32_H + col] = ga_f32_to_bf16(accx[nt][r]);
                }
            }
        __syncthreads();
        if (!need_rec) break;
    }
}

// ===========================================================================
// lstm_dw_h32: split-K weight gradient, dWcat [128][64] + dbias [128].
// ===========================================================================
#define DW32_KC 32
#define DW32_DGS 136   // dg chunk row stride (u16): [32][128] padded
#define DW32_ICS 72    // ic chunk row stride (u16): [32][64] padded
#define DW32_BUF (DW32_KC * (DW32_DGS + DW32_ICS))

struct L32DwArgs {
    const unsigned short* __restrict__ dgates;  // [BT][128]
    const unsigned short* __restrict__ x_seq;   // [BT][32]
    const unsigned short* __restrict__ h_seq;   // [BT][32] (read at t-1)
    float* __restrict__ partial;                // [NB][128*64]
    float* __restrict__ db_partial;             // [NB][128]
    int BT, T, NB;
};

__launch_bounds__(256)
__global__ void lstm_dw_h32(L32DwArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;  // wave covers dWcat rows wid*32..+32

    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* const smem_u16 = (unsigned short*)smem;
    float* s_db = (float*)(smem_u16 + 2 * DW32_BUF);  // [16][128]
#define DW32_DGT(buf) (smem_u16 + (buf) * DW32_BUF)
#define DW32_ICT(buf) (smem_u16 + (buf) * DW32_BUF + DW32_KC * DW32_DGS)

    ga_f32x4 acc[2][4];  // [jt][ct] — C tile [32 rows of this wave][64 cols]
#pragma unroll
    for (int jt = 0; jt < 2; ++jt)
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) acc[jt][ct] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
    float db_reg[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) db_reg[e] = 0.f;

    const int dg_row0 = tid >> 4, dg_col0 = (tid & 15) * 8;  // [32][128]: 2 passes
    const int ic_row0 = tid >> 3, ic_col0 = (tid & 7) * 8;   // [32][64]: 1 pass
    const int n_chunks = (a.BT + DW32_KC - 1) / DW32_KC;

    auto load_chunk = [&](int c, ga_u16x8* vdg, ga_u16x8* vic) {
        const int r0 = c * DW32_KC;
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + dg_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT) v = *(const ga_u16x8*)(a.dgates + (size_t)r * L32_4H + dg_col0);
            vdg[p] = v;
        }
        {
            int r = r0 + ic_row0;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT) {
                if (ic_col0 < L32_H) {
                    v = *(const ga_u16x8*)(a.x_seq + (size_t)r * L32_H + ic_col0);
                } else {
                    int t = r % a.T;
                    if (t > 0)
                        v = *(const ga_u16x8*)(a.h_seq + (size_t)(r - 1) * L32_H + ic_col0 - L32_H);
                }
            }
            vic[0] = v;
        }
    };
    auto write_chunk = [&](int buf, const ga_u16x8* vdg, const ga_u16x8* vic) {
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + dg_row0;
            *(ga_u16x8*)(DW32_DGT(buf) + (size_t)row * DW32_DGS + dg_col0) = vdg[p];
#pragma unroll
            for (int e = 0; e < 8; ++e) db_reg[e] += ga_bf16_to_f32(vdg[p][e]);
        }
        *(ga_u16x8*)(DW32_ICT(buf) + (size_t)ic_row0 * DW32_ICS + ic_col0) = vic[0];
    };

    int nb = 0;
    ga_u16x8 vdg[2], vic[1];
    load_chunk(blockIdx.x, vdg, vic);
    write_chunk(0, vdg, vic);
    __syncthreads();
    for (int c = blockIdx.x; c < n_chunks; c += a.NB) {
        const bool more = (c + a.NB) < n_chunks;
        if (more) load_chunk(c + a.NB, vdg, vic);
        const int k0 = (lane >> 4) * 8;
        ga_bf16x8 bfr[4];
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
            int colc = ct * 16 + (lane & 15);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = DW32_ICT(nb)[(size_t)(k0 + e) * DW32_ICS + colc];
                bfr[ct][e] = *(__bf16*)&u;
            }
        }
#pragma unroll
        for (int jt = 0; jt < 2; ++jt) {
            int j = wid * 32 + jt * 16 + (lane & 15);
            ga_bf16x8 afrag;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = DW32_DGT(nb)[(size_t)(k0 + e) * DW32_DGS + j];
                afrag[e] = *(__bf16*)&u;
            }
#pragma unroll
            for (int ct = 0; ct < 4; ++ct)
                acc[jt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfr[ct], acc[jt][ct], 0, 0, 0);
        }
        if (more) write_chunk(nb ^ 1, vdg, vic);
        nb ^= 1;
        __syncthreads();
    }

    float* out = a.partial + (size_t)blockIdx.x * L32_4H * L32_K;
#pragma unroll
    for (int jt = 0; jt < 2; ++jt)
#pragma unroll
        for (int ct = 0; ct < 4; ++ct)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int j = wid * 32 + jt * 16 + (lane >> 4) * 4 + r;
                int col = ct * 16 + (lane & 15);
                out[(size_t)j * L32_K + col] = acc[jt][ct][r];
            }
    // db: threads with the same (tid&15) cover the same 8 cols; 16 groups
#pragma unroll
    for (int e = 0; e < 8; ++e)
        s_db[(size_t)(tid >> 4) * L32_4H + (tid & 15) * 8 + e] = db_reg[e];
    __syncthreads();
    if (tid < L32_4H) {
        float s = 0.f;
#pragma unroll
        for (int gr = 0; gr < 16; ++gr) s += s_db[(size_t)gr * L32_4H + tid];
        a.db_partial[(size_t)blockIdx.x * L32_4H + tid] = s;
    }
}

// ===========================================================================
// Host launchers
// ===========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "launchers.h"

void lstm_fwd_h32_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                         torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                         torch::Tensor gates_seq, torch::Tensor c_seq,
                         c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T) {
    L32FwdArgs a;
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
    size_t smem = 6 * T32B;  // 24 KiB
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_fwd_h32, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_bwd_h32_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                         torch::Tensor c_seq, torch::Tensor wcat_t,
                         torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                         int64_t B, int64_t T, bool dh_last_only) {
    L32BwdArgs a;
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
    size_t smem = 5 * T32B + L32_K * L32_4H * 2;  // 20 KiB + 16 KiB WcatT
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_bwd_h32, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void lstm_dw_h32_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                        torch::Tensor partial, torch::Tensor db_partial,
                        int64_t BT, int64_t T) {
    L32DwArgs a;
    a.dgates = (const unsigned short*)dgates.data_ptr();
    a.x_seq = (const unsigned short*)x_seq.data_ptr();
    a.h_seq = (const unsigned short*)h_seq.data_ptr();
    a.partial = partial.data_ptr<float>();
    a.db_partial = db_partial.data_ptr<float>();
    a.BT = (int)BT;
    a.T = (int)T;
    a.NB = (int)partial.size(0);
    TORCH_CHECK(db_partial.size(0) == partial.size(0), "partial slab counts differ");
    size_t smem = (size_t)2 * DW32_BUF * 2 + 16 * L32_4H * sizeof(float);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_dw_h32, dim3(a.NB), dim3(256), smem, stream.stream(), a);
}
