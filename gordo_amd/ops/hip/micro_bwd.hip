// Ablation microbenchmark for lstm_bwd_cell's occupancy/weight-staging
// trade-off. The production kernel stages the whole 64-KiB WcatT in LDS ->
// 104 KiB/workgroup -> 1 block/CU, so the serial reverse-time chain has no
// second chain to interleave with. Variants:
//   V0: production design — full WcatT LDS image (104 KiB, 1 block/CU)
//   V1: no LDS image — B-fragments streamed from L2 (40 KiB, 2 blocks/CU)
//   V2: hybrid — W_hh half (the recurrence-critical dh_rec GEMM) in LDS,
//       W_ih half (the off-critical-path dx GEMM) streamed from L2
//       (72 KiB, 2 blocks/CU)
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 micro_bwd.hip -o micro_bwd
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
#include "common.h"

#define LSTM_4H 256
#define LSTM_K 128

struct Args {
    const unsigned short* dh_out;
    const unsigned char* gates_seq;
    const unsigned short* c_seq;
    const unsigned short* wcat_t;
    unsigned short* dgates_seq;
    unsigned short* dx_seq;
    int B, T;
    int dh_last_only;
};

extern __shared__ __attribute__((aligned(16))) char sm[];

template <int ABL>
__launch_bounds__(256)
__global__ void bwd(Args a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * GA_MT;
    const int wrow = wid * 16;

    char* dh_tile = sm;
    char* dg_tiles = sm + GA_TILE_BYTES;

    for (int c = tid; c < GA_MT * 8; c += 256) {
        ga_u16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        ga_tile_write8(dh_tile, c >> 3, (c & 7) * 8, z);
    }
    const int em = tid >> 2;
    const int eg = tid & 3;
    const int c0 = eg * 16;
    float dc_reg[16];
#pragma unroll
    for (int e = 0; e < 16; ++e) dc_reg[e] = 0.f;

    // LDS weight image: V0 = both halves [128][256]; V2 = W_hh half [64][256]
    char* wT_lds = sm + 5 * GA_TILE_BYTES;
    if (ABL == 0) {
        for (int g = tid; g < LSTM_K * 32; g += 256) {
            int row = g >> 5, col0 = (g & 31) * 8;
            ga_u16x8 v = *(const ga_u16x8*)(a.wcat_t + (size_t)row * LSTM_4H + col0);
            *(ga_u16x8*)&wT_lds[ga_swW(row, col0 * 2, 512)] = v;
        }
    } else if (ABL == 2) {
        for (int g = tid; g < GA_WMAX * 32; g += 256) {
            int row = g >> 5, col0 = (g & 31) * 8;
            ga_u16x8 v = *(const ga_u16x8*)(a.wcat_t + (size_t)(GA_WMAX + row) * LSTM_4H + col0);
            *(ga_u16x8*)&wT_lds[ga_swW(row, col0 * 2, 512)] = v;
        }
    }

    const int pgi = base + em;
    const bool plive = pgi < a.B;
    uint2 pvi[2], pvf[2], pvg[2], pvo[2];
    ga_u16x8 pvc[2], pvcp[2], pvdh[2];
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
        {
            const size_t bt = (size_t)pgi * a.T + t;
            const bool live = plive;
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                int cc = c0 + hh * 8;
                uint2 vi = pvi[hh], vf = pvf[hh], vg = pvg[hh], vo = pvo[hh];
                ga_u16x8 vc = pvc[hh], vcp = pvcp[hh], vdh = pvdh[hh];
                pvc[hh] = vcp;
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
                    dc_reg[ei] = dc * f_g;
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
        if (t > 0) preload(t - 1);

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
                    ga_bf16x8 bfrag;
                    if (ABL == 1) {
                        bfrag = *(const ga_bf16x8*)(a.wcat_t +
                            (size_t)(GA_WMAX + nt * 16 + (lane & 15)) * LSTM_4H + kk + (lane >> 4) * 8);
                    } else if (ABL == 2) {
                        bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                            nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 512)];
                    } else {
                        bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                            GA_WMAX + nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 512)];
                    }
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            if (need_dx)
#pragma unroll
                for (int nt = 0; nt < 4; ++nt) {
                    ga_bf16x8 bfrag;
                    if (ABL == 0) {
                        bfrag = *(const ga_bf16x8*)&wT_lds[ga_swW(
                            nt * 16 + (lane & 15), (kk + (lane >> 4) * 8) * 2, 512)];
                    } else {
                        bfrag = *(const ga_bf16x8*)(a.wcat_t +
                            (size_t)(nt * 16 + (lane & 15)) * LSTM_4H + kk + (lane >> 4) * 8);
                    }
                    accx[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, accx[nt], 0, 0, 0);
                }
        }
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

int main() {
    const int B = 32768, T = 144;
    const size_t BT = (size_t)B * T;
    unsigned short *dh, *c_seq, *wcat_t, *dgates, *dx;
    unsigned char* gates;
    hipMalloc(&dh, BT * GA_WMAX * 2);
    hipMalloc(&gates, BT * LSTM_4H);
    hipMalloc(&c_seq, BT * GA_WMAX * 2);
    hipMalloc(&wcat_t, (size_t)LSTM_K * LSTM_4H * 2);
    hipMalloc(&dgates, BT * LSTM_4H * 2);
    hipMalloc(&dx, BT * GA_WMAX * 2);
    hipMemset(dh, 0x3c, BT * GA_WMAX * 2);
    hipMemset(gates, 0x55, BT * LSTM_4H);
    hipMemset(c_seq, 0x3c, BT * GA_WMAX * 2);
    hipMemset(wcat_t, 0x2c, (size_t)LSTM_K * LSTM_4H * 2);

    Args a{dh, gates, c_seq, wcat_t, dgates, dx, B, T, 0};
    dim3 grid(B / GA_MT), block(256);
    size_t sm0 = 5 * GA_TILE_BYTES + (size_t)LSTM_K * 512;   // 104 KiB
    size_t sm1 = 5 * GA_TILE_BYTES;                          // 40 KiB
    size_t sm2 = 5 * GA_TILE_BYTES + (size_t)GA_WMAX * 512;  // 72 KiB

    auto run = [&](int abl, size_t smem, const char* name) {
        const void* fn = abl == 0 ? (const void*)bwd<0>
                       : abl == 1 ? (const void*)bwd<1> : (const void*)bwd<2>;
        if (smem > 65536)
            hipFuncSetAttribute(fn, hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);
        hipEvent_t e0, e1;
        hipEventCreate(&e0); hipEventCreate(&e1);
        for (int w = 0; w < 2; ++w) {  // warm
            if (abl == 0) hipLaunchKernelGGL(bwd<0>, grid, block, smem, 0, a);
            if (abl == 1) hipLaunchKernelGGL(bwd<1>, grid, block, smem, 0, a);
            if (abl == 2) hipLaunchKernelGGL(bwd<2>, grid, block, smem, 0, a);
        }
        hipDeviceSynchronize();
        float best = 1e9f;
        for (int it = 0; it < 5; ++it) {
            hipEventRecord(e0);
            if (abl == 0) hipLaunchKernelGGL(bwd<0>, grid, block, smem, 0, a);
            if (abl == 1) hipLaunchKernelGGL(bwd<1>, grid, block, smem, 0, a);
            if (abl == 2) hipLaunchKernelGGL(bwd<2>, grid, block, smem, 0, a);
            hipEventRecord(e1);
            hipEventSynchronize(e1);
            float ms; hipEventElapsedTime(&ms, e0, e1);
            if (ms < best) best = ms;
        }
        printf("%-28s smem=%6zu  %8.3f ms\n", name, smem, best);
    };
    run(0, sm0, "V0 full-LDS (1 blk/CU)");
    run(1, sm1, "V1 full-L2  (2 blk/CU)");
    run(2, sm2, "V2 hh-LDS + ih-L2 (2 blk)");
    // interleaved second round to cancel clock drift
    run(0, sm0, "V0 again");
    run(1, sm1, "V1 again");
    run(2, sm2, "V2 again");
    return 0;
}
