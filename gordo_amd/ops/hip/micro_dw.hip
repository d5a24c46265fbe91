// Ablation microbenchmark for lstm_dw.
// V0 full; V1 no dg stage (no loads/writes); V2 no ic stage; V3 no MFMA;
// V4 stage loads but skip transposed LDS writes; V5 MFMA only (no staging).
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
#include <algorithm>
#include "common.h"

#define LSTM_4H 256
#define LSTM_K 128
#define DW_KC 32
#define DW_STR 40
#define DW_BUF_U16 ((256 + 128) * DW_STR)

struct Args {
    const unsigned short* dgates;
    const unsigned short* x_seq;
    const unsigned short* h_seq;
    float* partial;
    float* db_partial;
    int BT, T, NB;
};

template <int ABL>
__launch_bounds__(256)
__global__ void dw(Args a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned short* const smem_u16 = (unsigned short*)smem;
    float* s_db = (float*)(smem_u16 + 2 * DW_BUF_U16);
#define DGT(buf) (smem_u16 + (buf) * DW_BUF_U16)
#define ICT(buf) (smem_u16 + (buf) * DW_BUF_U16 + 256 * DW_STR)

    ga_f32x4 acc[4][8];
    for (int jt = 0; jt < 4; ++jt) for (int ct = 0; ct < 8; ++ct) acc[jt][ct] = ga_f32x4{0,0,0,0};
    float db_reg[8];
    for (int e = 0; e < 8; ++e) db_reg[e] = 0.f;
    const int dg_row0 = tid >> 5, dg_col0 = (tid & 31) * 8;
    const int ic_row0 = tid >> 4, ic_col0 = (tid & 15) * 8;
    const int n_chunks = (a.BT + DW_KC - 1) / DW_KC;

    auto load_chunk = [&](int c, ga_u16x8* vdg, ga_u16x8* vic) {
        const int r0 = c * DW_KC;
        if (ABL != 1)
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            int r = r0 + row;
            ga_u16x8 v = {0,0,0,0,0,0,0,0};
            if (r < a.BT) v = *(const ga_u16x8*)(a.dgates + (size_t)r * LSTM_4H + dg_col0);
            vdg[p] = v;
        }
        if (ABL != 2)
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            int r = r0 + row;
            ga_u16x8 v = {0,0,0,0,0,0,0,0};
            if (r < a.BT) {
                if (ic_col0 < GA_WMAX) {
                    v = *(const ga_u16x8*)(a.x_seq + (size_t)r * GA_WMAX + ic_col0);
                } else {
                    int t = r % a.T;
                    if (t > 0) v = *(const ga_u16x8*)(a.h_seq + (size_t)(r - 1) * GA_WMAX + ic_col0 - GA_WMAX);
                }
            }
            vic[p] = v;
        }
    };
    auto write_chunk = [&](int buf, const ga_u16x8* vdg, const ga_u16x8* vic) {
        if (ABL == 4) {  // vector junk writes (no transpose) to keep data deps
            for (int p = 0; p < 4; ++p)
                *(ga_u16x8*)(DGT(buf) + (size_t)(p * 256 + tid) * 8 % (256 * DW_STR - 8)) = vdg[p];
            for (int p = 0; p < 2; ++p)
                *(ga_u16x8*)(ICT(buf) + (size_t)(p * 256 + tid) * 8 % (128 * DW_STR - 8)) = vic[p];
            return;
        }
        if (ABL != 1)
        for (int p = 0; p < 4; ++p) {
            int row = p * 8 + dg_row0;
            for (int e = 0; e < 8; ++e) {
                DGT(buf)[(size_t)(dg_col0 + e) * DW_STR + row] = vdg[p][e];
                db_reg[e] += ga_bf16_to_f32(vdg[p][e]);
            }
        }
        if (ABL != 2)
        for (int p = 0; p < 2; ++p) {
            int row = p * 16 + ic_row0;
            for (int e = 0; e < 8; ++e)
                ICT(buf)[(size_t)(ic_col0 + e) * DW_STR + row] = vic[p][e];
        }
    };

    int nb = 0;
    ga_u16x8 vdg[4] = {}, vic[2] = {};
    if (ABL != 5) { load_chunk(blockIdx.x, vdg, vic); write_chunk(0, vdg, vic); }
    __syncthreads();
    for (int c = blockIdx.x; c < n_chunks; c += a.NB) {
        const bool more = (c + a.NB) < n_chunks;
        if (ABL != 5 && more) load_chunk(c + a.NB, vdg, vic);
        if (ABL != 3)
        for (int jt = 0; jt < 4; ++jt) {
            ga_bf16x8 afrag = *(const ga_bf16x8*)(DGT(nb) + (size_t)(wid * 64 + jt * 16 + (lane & 15)) * DW_STR + (lane >> 4) * 8);
            for (int ct = 0; ct < 8; ++ct) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(ICT(nb) + (size_t)(ct * 16 + (lane & 15)) * DW_STR + (lane >> 4) * 8);
                acc[jt][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[jt][ct], 0, 0, 0);
            }
        }
        if (ABL != 5 && more) write_chunk(nb ^ 1, vdg, vic);
        nb ^= 1;
        __syncthreads();
    }
    float* out = a.partial + (size_t)blockIdx.x * LSTM_4H * LSTM_K;
    for (int jt = 0; jt < 4; ++jt)
        for (int ct = 0; ct < 8; ++ct)
            for (int r = 0; r < 4; ++r) {
                int j = wid * 64 + jt * 16 + (lane >> 4) * 4 + r;
                int col = ct * 16 + (lane & 15);
                out[(size_t)j * LSTM_K + col] = acc[jt][ct][r];
            }
    for (int e = 0; e < 8; ++e)
        s_db[(size_t)(tid >> 5) * LSTM_4H + (tid & 31) * 8 + e] = db_reg[e];
    __syncthreads();
    if (tid < LSTM_4H) {
        float ssum = 0.f;
        for (int gr = 0; gr < 8; ++gr) ssum += s_db[(size_t)gr * LSTM_4H + tid];
        a.db_partial[(size_t)blockIdx.x * LSTM_4H + tid] = ssum;
    }
}

int main() {
    int B = 32768, T = 144;
    size_t BT = (size_t)B * T;
    unsigned short *dg, *x, *h;
    float *pp, *dbp;
    (void)hipMalloc(&dg, BT * 256 * 2);
    (void)hipMalloc(&x, BT * 64 * 2);
    (void)hipMalloc(&h, BT * 64 * 2);
    (void)hipMalloc(&pp, 512 * 32768 * 4);
    (void)hipMalloc(&dbp, 512 * 256 * 4);
    (void)hipMemset(dg, 0x3c, BT * 256 * 2);
    (void)hipMemset(x, 0x3c, BT * 64 * 2);
    (void)hipMemset(h, 0x3c, BT * 64 * 2);
    Args a{dg, x, h, pp, dbp, (int)BT, T, 512};
    size_t smem = (size_t)2 * DW_BUF_U16 * 2 + 8 * 256 * 4;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);
    const char* names[6] = {"full", "no_dg", "no_ic", "no_mfma", "vec_writes", "mfma_only"};
    std::vector<std::vector<float>> res(6);
    for (int round = 0; round < 3; ++round)
        for (int v = 0; v < 6; ++v) {
            auto launch = [&]() {
                switch (v) {
                    case 0: hipLaunchKernelGGL(dw<0>, dim3(512), dim3(256), smem, 0, a); break;
                    case 1: hipLaunchKernelGGL(dw<1>, dim3(512), dim3(256), smem, 0, a); break;
                    case 2: hipLaunchKernelGGL(dw<2>, dim3(512), dim3(256), smem, 0, a); break;
                    case 3: hipLaunchKernelGGL(dw<3>, dim3(512), dim3(256), smem, 0, a); break;
                    case 4: hipLaunchKernelGGL(dw<4>, dim3(512), dim3(256), smem, 0, a); break;
                    case 5: hipLaunchKernelGGL(dw<5>, dim3(512), dim3(256), smem, 0, a); break;
                }
            };
            launch(); (void)hipDeviceSynchronize();
            (void)hipEventRecord(e0);
            for (int k = 0; k < 3; ++k) launch();
            (void)hipEventRecord(e1); (void)hipEventSynchronize(e1);
            float ms; (void)hipEventElapsedTime(&ms, e0, e1);
            res[v].push_back(ms / 3);
        }
    for (int v = 0; v < 6; ++v) {
        std::sort(res[v].begin(), res[v].end());
        printf("%-10s median %.3f ms  min %.3f ms\n", names[v], res[v][1], res[v][0]);
    }
    return 0;
}
