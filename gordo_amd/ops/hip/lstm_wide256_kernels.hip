// W=256 persistent LSTM cell kernels for MI355X (gfx950).
//
// Third width tier (129..256 features/hidden — the 200-tag asset class,
// SURVEY.md §2.1 "100s of tags"). The packed weight is
// 1 MiB/layer (Wcat [4W=1024][2W=512] bf16) — far past LDS — so, like the
// W=128 tier, MFMA B-fragments stream from L2 while per-wave MFMA work hides
// the latency. What changes at 256 is LDS pressure: the W=128 plan
// (x + h + 4 pre-activation tiles resident) would need 192 KiB. Design:
//
//  * 512-thread blocks (8 waves), 64 batch rows per block.
//  * The 256 cell columns are processed in TWO column-halves per timestep:
//    pre-activation tiles are 4 x [64][128] (64 KiB) for the CURRENT half
//    only; x_tile/h_tile stay full-K [64][256] (32 KiB each) => 128 KiB.
//  * h_t outputs are STASHED IN REGISTERS through both halves (h_tile must
//    keep h_{t-1} until the second half's GEMM has read it) and written to
//    LDS once per timestep.
//  * forward GEMM: wave = (gate, column-sub-half): each wave owns a
//    [64 m][64 j] C tile with K = 512 (x 256 | h 256), acc[4][4] f32x4.
//  * backward: dgates computed half-by-half (dg tiles 64 KiB), the dh_rec /
//    dx GEMMs accumulate across the two half-K passes; 8 waves = 4 dh_rec
//    stripes + 4 dx stripes of [16][256].
//  * dW: the W=128 quadrant scheme widened to a 4x4 grid of [256][128]
//    C tiles over dWcat [1024][512], same double-buffered split-K body.
#include "common.h"

#define QW 256
#define Q4H (4 * QW)    // 1024
#define QK (2 * QW)     // 512
#define QTB (64 * QW * 2)      // [64][256] bf16 tile bytes (32 KiB)
#define HTB (64 * 128 * 2)     // [64][128] half tile bytes (16 KiB)

// [64][256] bf16 tile, 512-B rows, swizzled 16-B slots (same conflict
// spreading as sw128: 8 distinct slots across row&7)
__device__ __forceinline__ unsigned int sw256(int row, int col) {
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 7u) << 4);
    return (unsigned int)row * (QW * 2) + byte;
}
// [64][128] half tiles reuse the W=128 swizzle
__device__ __forceinline__ unsigned int qsw128(int row, int col) {
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 7u) << 4);
    return (unsigned int)row * (128 * 2) + byte;
}

extern __shared__ __attribute__((aligned(16))) char qsm[];

// ===========================================================================
// forward
// ===========================================================================
struct QFwdArgs {
    const unsigned short* __restrict__ X;     // layer0: dataset [N][256]; else [B][T][256]
    const int* __restrict__ win;              // [B] window starts (layer0) or null
    const unsigned short* __restrict__ wcat;  // [1024][512]
    const float* __restrict__ bias;           // [1024]
    unsigned short* __restrict__ h_seq;       // [B][T][256]
    unsigned char* __restrict__ gates_seq;    // [B][T][1024] u8
    unsigned short* __restrict__ c_seq;       // [B][T][256]
    unsigned short* __restrict__ x_seq_out;   // [B][T][256] or null
    int B, T;
};

__launch_bounds__(512)
__global__ void lstm_fwd_wide256(QFwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;          // 0..7
    const int gate = wid >> 1;         // wave's gate
    const int sub = wid & 1;           // wave's 64-col sub-half
    const int base = blockIdx.x * 64;
    char* x_tile = qsm;                // [64][256]
    char* h_tile = qsm + QTB;          // [64][256]
    char* pre_tiles = qsm + 2 * QTB;   // 4 x [64][128] (current half)

    float bias_reg[2][4];
#pragma unroll
    for (int h = 0; h < 2; ++h)
#pragma unroll
        for (int jt = 0; jt < 4; ++jt)
            bias_reg[h][jt] =
                a.bias[gate * QW + h * 128 + sub * 64 + jt * 16 + (lane & 15)];

    for (int c = tid; c < 64 * 32; c += 512) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *(ga_u16x8*)&h_tile[sw256(c >> 5, (c & 31) * 8)] = z;
    }
    // epilogue ownership: em = tid>>3 (64 rows), (tid&7) -> 32 cols/thread
    // (16 per half: cols half*128 + (tid&7)*16 .. +16)
    const int em = tid >> 3;
    const int ec0 = (tid & 7) * 16;
    const int egi = base + em;
    // c/h state as TWO 16-element banks (one per column half). A single
    // c_reg[32] indexed by the runtime `half` lowered to ~500 v_cndmask
    // register-select instructions per half-pass (comparable to the 256
    // MFMAs!); selecting the bank once per access group costs ~100.
    float c_lo[16], c_hi[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) c_lo[e] = c_hi[e] = 0.f;
    unsigned short oh_lo[16], oh_hi[16];
    __syncthreads();

    const unsigned short* wcat_p = a.wcat;
    for (int t = 0; t < a.T; ++t) {
        asm volatile("" : "+s"(wcat_p));  // stop weight-LICM (see W=128 file)
        // ---- stage x_t ----------------------------------------------------
        for (int c = tid; c < 64 * 32; c += 512) {
            int m = c >> 5, col0 = (c & 31) * 8;
            int gi = base + m;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gi < a.B) {
                if (a.win != nullptr) {
                    int row = a.win[gi] + t;
                    v = *(const ga_u16x8*)(a.X + (size_t)row * QW + col0);
                } else {
                    v = *(const ga_u16x8*)(a.X + ((size_t)gi * a.T + t) * QW + col0);
                }
                if (a.x_seq_out != nullptr)
                    *(ga_u16x8*)(a.x_seq_out + ((size_t)gi * a.T + t) * QW + col0) = v;
            }
            *(ga_u16x8*)&x_tile[sw256(m, col0)] = v;
        }
        __syncthreads();

        // deliberately NOT unrolled: unrolling doubles the live accumulator
        // set (256 VGPRs + 28 spills); rolled, the half-dependent c/oh slice
        // costs 48 B/lane of scratch, negligible vs 1024 MFMAs/wave/t
        for (int half = 0; half < 2; ++half) {
            // ---- pre-activations for this half: C [64 m][64 j], K = 512 ---
            ga_f32x4 acc[4][4];
#pragma unroll
            for (int mt = 0; mt < 4; ++mt)
#pragma unroll
                for (int jt = 0; jt < 4; ++jt)
                    acc[mt][jt] = ga_f32x4{bias_reg[half][jt], bias_reg[half][jt],
                                           bias_reg[half][jt], bias_reg[half][jt]};
            for (int kk = 0; kk < 16; ++kk) {
                const char* src = (kk < 8) ? x_tile : h_tile;
                int klocal = (kk & 7) * 32;
#pragma unroll
                for (int mt = 0; mt < 4; ++mt) {
                    ga_bf16x8 afrag = *(const ga_bf16x8*)&src[sw256(
                        mt * 16 + (lane & 15), klocal + (lane >> 4) * 8)];
#pragma unroll
                    for (int jt = 0; jt < 4; ++jt) {
                        ga_bf16x8 bfrag = *(const ga_bf16x8*)(wcat_p +
                            (size_t)(gate * QW + half * 128 + sub * 64 + jt * 16 + (lane & 15)) * QK +
                            kk * 32 + (lane >> 4) * 8);
                        acc[mt][jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mt][jt], 0, 0, 0);
                    }
                }
            }
            char* my_pre = pre_tiles + gate * HTB;
#pragma unroll
            for (int mt = 0; mt < 4; ++mt)
#pragma unroll
                for (int jt = 0; jt < 4; ++jt) {
                    int col = sub * 64 + jt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r)
                        *(unsigned short*)&my_pre[qsw128(mt * 16 + (lane >> 4) * 4 + r, col)] =
                            ga_f32_to_bf16(acc[mt][jt][r]);
                }
            __syncthreads();

            // ---- cell update for this half's 16 cols per thread -----------
            {
                const bool live = egi < a.B;
                const size_t bt = (size_t)egi * a.T + t;
                float cur[16];
#pragma unroll
                for (int e2 = 0; e2 < 16; ++e2) cur[e2] = half ? c_hi[e2] : c_lo[e2];
                unsigned short ohv[16];
#pragma unroll
                for (int hh = 0; hh < 2; ++hh) {
                    int lc = ec0 + hh * 8;            // local col in half
                    int cc = half * 128 + lc;         // cell col
                    ga_u16x8 vi = *(ga_u16x8*)&pre_tiles[qsw128(em, lc)];
                    ga_u16x8 vf = *(ga_u16x8*)&pre_tiles[HTB + qsw128(em, lc)];
                    ga_u16x8 vg = *(ga_u16x8*)&pre_tiles[2 * HTB + qsw128(em, lc)];
                    ga_u16x8 vo = *(ga_u16x8*)&pre_tiles[3 * HTB + qsw128(em, lc)];
                    ga_u16x8 oh, oc;
                    float qi[8], qf[8], qg[8], qo[8];
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        float i_g = ga_sigmoid(ga_bf16_to_f32(vi[e]));
                        float f_g = ga_sigmoid(ga_bf16_to_f32(vf[e]));
                        float g_g = ga_tanh(ga_bf16_to_f32(vg[e]));
                        float o_g = ga_sigmoid(ga_bf16_to_f32(vo[e]));
                        float c_new = f_g * cur[hh * 8 + e] + i_g * g_g;
                        cur[hh * 8 + e] = c_new;
                        unsigned short hv = ga_f32_to_bf16(o_g * ga_tanh(c_new));
                        oh[e] = hv;
                        ohv[hh * 8 + e] = hv;
                        oc[e] = ga_f32_to_bf16(c_new);
                        qi[e] = i_g;
                        qf[e] = f_g;
                        qg[e] = g_g * 0.5f + 0.5f;
                        qo[e] = o_g;
                    }
                    if (live) {
                        *(ga_u16x8*)(a.h_seq + bt * QW + cc) = oh;
                        *(ga_u16x8*)(a.c_seq + bt * QW + cc) = oc;
                        *(uint2*)(a.gates_seq + bt * Q4H + cc) = ga_q8_pack8(qi);
                        *(uint2*)(a.gates_seq + bt * Q4H + QW + cc) = ga_q8_pack8(qf);
                        *(uint2*)(a.gates_seq + bt * Q4H + 2 * QW + cc) = ga_q8_pack8(qg);
                        *(uint2*)(a.gates_seq + bt * Q4H + 3 * QW + cc) = ga_q8_pack8(qo);
                    }
                }
#pragma unroll
                for (int e2 = 0; e2 < 16; ++e2) {
                    if (half) { c_hi[e2] = cur[e2]; oh_hi[e2] = ohv[e2]; }
                    else      { c_lo[e2] = cur[e2]; oh_lo[e2] = ohv[e2]; }
                }
            }
            __syncthreads();  // pre_tiles reused by the next half
        }

        // ---- commit h_t to LDS (h_{t-1} fully consumed by both halves) ----
#pragma unroll
        for (int hh = 0; hh < 2; ++hh) {
            ga_u16x8 vlo, vhi;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                vlo[e] = oh_lo[hh * 8 + e];
                vhi[e] = oh_hi[hh * 8 + e];
            }
            *(ga_u16x8*)&h_tile[sw256(em, ec0 + hh * 8)] = vlo;
            *(ga_u16x8*)&h_tile[sw256(em, 128 + ec0 + hh * 8)] = vhi;
        }
        __syncthreads();
    }
}

// ===========================================================================
// backward
// ===========================================================================
struct QBwdArgs {
    const unsigned short* __restrict__ dh_out;     // [B][T][256] or [B][256]
    const unsigned char* __restrict__ gates_seq;   // [B][T][1024] u8
    const unsigned short* __restrict__ c_seq;      // [B][T][256]
    const unsigned short* __restrict__ wcat_t;     // [512][1024]
    unsigned short* __restrict__ dgates_seq;       // [B][T][1024] out
    unsigned short* __restrict__ dx_seq;           // [B][T][256] out or null
    int B, T;
    int dh_last_only;
};

__launch_bounds__(512)
__global__ void lstm_bwd_wide256(QBwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;          // 0..7
    const int base = blockIdx.x * 64;
    char* dh_tile = qsm;               // [64][256]
    char* dg_tiles = qsm + QTB;        // 4 x [64][128] (current half)

    for (int c = tid; c < 64 * 32; c += 512) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *(ga_u16x8*)&dh_tile[sw256(c >> 5, (c & 31) * 8)] = z;
    }
    const int em = tid >> 3;
    const int ec0 = (tid & 7) * 16;
    float dc_reg[32];
#pragma unroll
    for (int e = 0; e < 32; ++e) dc_reg[e] = 0.f;
    __syncthreads();

    const unsigned short* wct_p = a.wcat_t;
    for (int t = a.T - 1; t >= 0; --t) {
        asm volatile("" : "+s"(wct_p));
        const bool need_rec = (t > 0);
        const bool need_dx = (a.dx_seq != nullptr);
        // wave role: (otype, output-col-QUARTER). Each wave covers ALL FOUR
        // 16-row stripes per B-fragment load: 8 waves = 2 outputs x 4 col
        // quarters with zero cross-wave B-fragment redundancy (the
        // wave-per-stripe layout was 4x redundant, the half/stripe-pair
        // layout 2x — each halving measured double-digit gains) at the same
        // register cost (acc[4][4] == acc[2][8]).
        const bool rec_wave = (wid >> 2) == 0;
        const bool active = rec_wave ? need_rec : need_dx;
        const int nq = wid & 3;          // output col quarter: nt 0..3
        const int brow0 = (rec_wave ? QW : 0) + nq * 64;
        ga_f32x4 acc[4][4];
#pragma unroll
        for (int st2 = 0; st2 < 4; ++st2)
#pragma unroll
            for (int nt = 0; nt < 4; ++nt) acc[st2][nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};

        for (int half = 0; half < 2; ++half) {
            // ---- elementwise BPTT for this half's cols --------------------
            {
                const int gi = base + em;
                const size_t bt = (size_t)gi * a.T + t;
                const bool live = gi < a.B;
                const ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
                for (int hh = 0; hh < 2; ++hh) {
                    int lc = ec0 + hh * 8;
                    int cc = half * 128 + lc;
                    uint2 vi = {0, 0}, vf = {0, 0}, vg = {0, 0}, vo = {0, 0};
                    ga_u16x8 vc = z, vcp = z, vdh = z;
                    if (live) {
                        vi = *(const uint2*)(a.gates_seq + bt * Q4H + cc);
                        vf = *(const uint2*)(a.gates_seq + bt * Q4H + QW + cc);
                        vg = *(const uint2*)(a.gates_seq + bt * Q4H + 2 * QW + cc);
                        vo = *(const uint2*)(a.gates_seq + bt * Q4H + 3 * QW + cc);
                        vc = *(const ga_u16x8*)(a.c_seq + bt * QW + cc);
                        if (t > 0) vcp = *(const ga_u16x8*)(a.c_seq + (bt - 1) * QW + cc);
                        if (a.dh_last_only) {
                            if (t == a.T - 1)
                                vdh = *(const ga_u16x8*)(a.dh_out + (size_t)gi * QW + cc);
                        } else {
                            vdh = *(const ga_u16x8*)(a.dh_out + bt * QW + cc);
                        }
                    }
                    ga_u16x8 vrec = *(ga_u16x8*)&dh_tile[sw256(em, cc)];
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
                        int ei = half * 16 + hh * 8 + e;
                        float dc = dc_reg[ei] + dh * o_g * (1.f - tanhc * tanhc);
                        odo[e] = ga_f32_to_bf16(dh * tanhc * o_g * (1.f - o_g));
                        odi[e] = ga_f32_to_bf16(dc * g_g * i_g * (1.f - i_g));
                        odf[e] = ga_f32_to_bf16(dc * c_prev * f_g * (1.f - f_g));
                        odg[e] = ga_f32_to_bf16(dc * i_g * (1.f - g_g * g_g));
                        dc_reg[ei] = dc * f_g;
                    }
                    *(ga_u16x8*)&dg_tiles[qsw128(em, lc)] = odi;
                    *(ga_u16x8*)&dg_tiles[HTB + qsw128(em, lc)] = odf;
                    *(ga_u16x8*)&dg_tiles[2 * HTB + qsw128(em, lc)] = odg;
                    *(ga_u16x8*)&dg_tiles[3 * HTB + qsw128(em, lc)] = odo;
                    if (live) {
                        *(ga_u16x8*)(a.dgates_seq + bt * Q4H + cc) = odi;
                        *(ga_u16x8*)(a.dgates_seq + bt * Q4H + QW + cc) = odf;
                        *(ga_u16x8*)(a.dgates_seq + bt * Q4H + 2 * QW + cc) = odg;
                        *(ga_u16x8*)(a.dgates_seq + bt * Q4H + 3 * QW + cc) = odo;
                    }
                }
            }
            __syncthreads();

            // ---- partial GEMM over this half's K (4 gates x 128) ----------
            //   dh_rec[m][j<256] += dgates_half x WcatT rows 256+j
            //   dx[m][j<256]     += dgates_half x WcatT rows j
            if (active) {
                for (int kk = 0; kk < 512; kk += 32) {
                    const char* dgt = dg_tiles + (kk >> 7) * HTB;
                    ga_bf16x8 afrag[4];
#pragma unroll
                    for (int st2 = 0; st2 < 4; ++st2)
                        afrag[st2] = *(const ga_bf16x8*)&dgt[qsw128(
                            st2 * 16 + (lane & 15), (kk & 127) + (lane >> 4) * 8)];
                    const size_t gc = (size_t)(kk >> 7) * QW + half * 128 + (kk & 127) + (lane >> 4) * 8;
#pragma unroll
                    for (int nt = 0; nt < 4; ++nt) {
                        ga_bf16x8 bfrag = *(const ga_bf16x8*)(wct_p +
                            (size_t)(brow0 + nt * 16 + (lane & 15)) * Q4H + gc);
#pragma unroll
                        for (int st2 = 0; st2 < 4; ++st2)
                            acc[st2][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[st2], bfrag, acc[st2][nt], 0, 0, 0);
                    }
                }
            }
            __syncthreads();  // dg_tiles reused by next half
        }

        // ---- write GEMM outputs ------------------------------------------
        if (active) {
#pragma unroll
            for (int st2 = 0; st2 < 4; ++st2)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt) {
                    int col = nq * 64 + nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = st2 * 16 + (lane >> 4) * 4 + r;
                        if (rec_wave) {
                            *(unsigned short*)&dh_tile[sw256(m, col)] = ga_f32_to_bf16(acc[st2][nt][r]);
                        } else {
                            int gi = base + m;
                            if (gi < a.B)
                                a.dx_seq[((size_t)gi * a.T + t) * QW + col] = ga_f32_to_bf16(acc[st2][nt][r]);
                        }
                    }
                }
        }
        __syncthreads();
        if (!need_rec) break;
    }
}

// ===========================================================================
// dW: the [256][128]-C-tile split-K body of the W=128 kernel on a 4x4
// quadrant grid over dWcat [1024][512]. Quadrant (gh, ih): dgates rows
// gh*256..+256 vs ic cols ih*128..+128 of [x(256) | h_{t-1}(256)].
// ===========================================================================
#define QDW_KC 32
#define QDW_DGS 264   // dg chunk row stride (u16): [32][264]
#define QDW_ICS 136   // ic chunk row stride (u16): [32][136]
#define QDW_BUF (QDW_KC * (QDW_DGS + QDW_ICS))

struct QDwArgs {
    const unsigned short* __restrict__ dgates;  // [BT][1024]
    const unsigned short* __restrict__ x_seq;   // [BT][256]
    const unsigned short* __restrict__ h_seq;   // [BT][256] (read at t-1)
    float* __restrict__ partial;                // [16][NB][256*128]
    float* __restrict__ db_partial;             // [4][NB][256] (ih==0 only)
    int BT, T, NB;
};

__launch_bounds__(256)
__global__ void lstm_dw_wide256(QDwArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int gh = blockIdx.y;   // dgates quarter (rows gh*256..+256)
    const int ih = blockIdx.z;   // ic quarter (cols ih*128 of [x | hprev])
    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* const smem_u16 = (unsigned short*)smem;
    float* s_db = (float*)(smem_u16 + 2 * QDW_BUF);  // [8][256]
#define QDGT(buf) (smem_u16 + (buf) * QDW_BUF)
#define QICT(buf) (smem_u16 + (buf) * QDW_BUF + QDW_KC * QDW_DGS)

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
    const int n_chunks = (a.BT + QDW_KC - 1) / QDW_KC;

    auto load_chunk = [&](int c, ga_u16x8* vdg, ga_u16x8* vic) {
        const int r0 = c * QDW_KC;
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT)
                v = *(const ga_u16x8*)(a.dgates + (size_t)r * Q4H + gh * 256 + dg_col0);
            vdg[p] = v;
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            int r = r0 + row;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (r < a.BT) {
                if (ih < 2) {
                    v = *(const ga_u16x8*)(a.x_seq + (size_t)r * QW + ih * 128 + ic_col0);
                } else {
                    int t = r % a.T;
                    if (t > 0)
                        v = *(const ga_u16x8*)(a.h_seq + (size_t)(r - 1) * QW + (ih - 2) * 128 + ic_col0);
                }
            }
            vic[p] = v;
        }
    };
    auto write_chunk = [&](int buf, const ga_u16x8* vdg, const ga_u16x8* vic) {
#pragma unroll
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            *(ga_u16x8*)(QDGT(buf) + (size_t)row * QDW_DGS + dg_col0) = vdg[p];
#pragma unroll
            for (int e = 0; e < 8; ++e) db_reg[e] += ga_bf16_to_f32(vdg[p][e]);
        }
#pragma unroll
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            *(ga_u16x8*)(QICT(buf) + (size_t)row * QDW_ICS + ic_col0) = vic[p];
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
                unsigned short u = QICT(nb)[(size_t)(k0 + e) * QDW_ICS + colc];
                bfr[ct][e] = *(__bf16*)&u;
            }
        }
#pragma unroll
        for (int jt = 0; jt < 4; ++jt) {
            int j = wid * 64 + jt * 16 + (lane & 15);
            ga_bf16x8 afrag;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                unsigned short u = QDGT(nb)[(size_t)(k0 + e) * QDW_DGS + j];
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

    // partial layout: [(gh*4 + ih)][NB][256*128]
    float* out = a.partial +
        ((size_t)(gh * 4 + ih) * a.NB + blockIdx.x) * 256 * 128;
#pragma unroll
    for (int jt = 0; jt < 4; ++jt)
#pragma unroll
        for (int ct = 0; ct < 8; ++ct)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int j = wid * 64 + jt * 16 + (lane >> 4) * 4 + r;
                int col = ct * 16 + (lane & 15);
                out[(size_t)j * 128 + col] = acc[jt][ct][r];
            }
    if (ih == 0) {  // dbias counted once per gate-quarter
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

void lstm_fwd_wide256_launch(torch::Tensor X, c10::optional<torch::Tensor> win,
                             torch::Tensor wcat, torch::Tensor bias, torch::Tensor h_seq,
                             torch::Tensor gates_seq, torch::Tensor c_seq,
                             c10::optional<torch::Tensor> x_seq_out, int64_t B, int64_t T) {
    QFwdArgs a;
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
    size_t smem = 2 * QTB + 4 * HTB;  // 128 KiB
    static bool cfg = [] {
        (void)hipFuncSetAttribute((const void*)lstm_fwd_wide256,
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        return true;
    }();
    (void)cfg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_fwd_wide256, dim3(n_tiles), dim3(512), smem, stream.stream(), a);
}

void lstm_bwd_wide256_launch(torch::Tensor dh_out, torch::Tensor gates_seq,
                             torch::Tensor c_seq, torch::Tensor wcat_t,
                             torch::Tensor dgates_seq, c10::optional<torch::Tensor> dx_seq,
                             int64_t B, int64_t T, bool dh_last_only) {
    QBwdArgs a;
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
    size_t smem = QTB + 4 * HTB;  // 96 KiB
    static bool cfg = [] {
        (void)hipFuncSetAttribute((const void*)lstm_bwd_wide256,
                                  hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        return true;
    }();
    (void)cfg;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_bwd_wide256, dim3(n_tiles), dim3(512), smem, stream.stream(), a);
}

void lstm_dw_wide256_launch(torch::Tensor dgates, torch::Tensor x_seq, torch::Tensor h_seq,
                            torch::Tensor partial, torch::Tensor db_partial,
                            int64_t BT, int64_t T) {
    QDwArgs a;
    a.dgates = (const unsigned short*)dgates.data_ptr();
    a.x_seq = (const unsigned short*)x_seq.data_ptr();
    a.h_seq = (const unsigned short*)h_seq.data_ptr();
    a.partial = partial.data_ptr<float>();
    a.db_partial = db_partial.data_ptr<float>();
    a.BT = (int)BT;
    a.T = (int)T;
    a.NB = (int)partial.size(1);
    TORCH_CHECK(partial.size(0) == 16, "partial must be [16][NB][256*128]");
    TORCH_CHECK(db_partial.size(0) == 4 && db_partial.size(1) == partial.size(1),
                "db_partial must be [4][NB][256]");
    size_t smem = (size_t)2 * QDW_BUF * 2 + 8 * 256 * sizeof(float);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(lstm_dw_wide256, dim3(a.NB, 4, 4), dim3(256), smem,
                       stream.stream(), a);
}
