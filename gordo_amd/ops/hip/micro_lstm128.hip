// Feasibility micro for a W=128 (wide) persistent LSTM cell forward:
// gate-specialized waves, weights streamed from L2 (256 KB/layer — past the
// register/LDS budget that the W=64 production kernel uses).
// Measures per-timestep cost to decide the ROADMAP item's expected payoff.
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 micro_lstm128.hip -o micro_lstm128
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
#include <algorithm>
#include "common.h"

#define WW 128
#define W4H (4 * WW)   // 512
#define WK (2 * WW)    // 256

// [64][128] bf16 tile, 256-B rows, swizzled 16-B slots
__device__ __forceinline__ unsigned int sw128(int row, int col) {
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 7u) << 4);
    return (unsigned int)row * (WW * 2) + byte;
}

struct Args {
    const unsigned short* X;     // [B][T][128]
    const unsigned short* wcat;  // [512][256]
    const float* bias;           // [512]
    unsigned short* h_seq;       // [B][T][128]
    unsigned short* gates_seq;   // [B][T][512]
    unsigned short* c_seq;       // [B][T][128]
    int B, T;
};

__launch_bounds__(256)
__global__ void lstm_fwd128(Args a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;  // wave = gate
    const int base = blockIdx.x * 64;
    extern __shared__ __attribute__((aligned(16))) char sm[];
    char* x_tile = sm;                         // [64][128]
    char* h_tile = sm + 64 * WW * 2;           // [64][128]
    char* pre_tiles = sm + 2 * 64 * WW * 2;    // 4 x [64][128]

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

    for (int t = 0; t < a.T; ++t) {
        // stage x_t: 64 rows x 16 granules = 1024 granules / 256 threads
        for (int c = tid; c < 64 * 16; c += 256) {
            int m = c >> 4, col0 = (c & 15) * 8;
            int gi = base + m;
            ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
            if (gi < a.B)
                v = *(const ga_u16x8*)(a.X + ((size_t)gi * a.T + t) * WW + col0);
            *(ga_u16x8*)&x_tile[sw128(m, col0)] = v;
        }
        __syncthreads();

        // gate pre-activations: C [64 m][128 j], K = 256, bfrags from L2
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
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)(a.wcat +
                        (size_t)(wid * WW + jt * 16 + (lane & 15)) * WK + kk * 32 + (lane >> 4) * 8);
                    acc[mt][jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mt][jt], 0, 0, 0);
                }
            }
        }
        char* my_pre = pre_tiles + wid * (64 * WW * 2);
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

        // cell update + streaming (rows em, 32 cols each)
        {
            const bool live = egi < a.B;
            const size_t bt = (size_t)egi * a.T + t;
#pragma unroll
            for (int hh = 0; hh < 4; ++hh) {
                int cc = ec0 + hh * 8;
                ga_u16x8 vi = *(ga_u16x8*)&pre_tiles[sw128(em, cc)];
                ga_u16x8 vf = *(ga_u16x8*)&pre_tiles[64 * WW * 2 + sw128(em, cc)];
                ga_u16x8 vg = *(ga_u16x8*)&pre_tiles[2 * 64 * WW * 2 + sw128(em, cc)];
                ga_u16x8 vo = *(ga_u16x8*)&pre_tiles[3 * 64 * WW * 2 + sw128(em, cc)];
                ga_u16x8 oh, oc, oi, of, og, oo;
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
                    oi[e] = ga_f32_to_bf16(i_g);
                    of[e] = ga_f32_to_bf16(f_g);
                    og[e] = ga_f32_to_bf16(g_g);
                    oo[e] = ga_f32_to_bf16(o_g);
                }
                *(ga_u16x8*)&h_tile[sw128(em, cc)] = oh;
                if (live) {
                    *(ga_u16x8*)(a.h_seq + bt * WW + cc) = oh;
                    *(ga_u16x8*)(a.c_seq + bt * WW + cc) = oc;
                    *(ga_u16x8*)(a.gates_seq + bt * W4H + cc) = oi;
                    *(ga_u16x8*)(a.gates_seq + bt * W4H + WW + cc) = of;
                    *(ga_u16x8*)(a.gates_seq + bt * W4H + 2 * WW + cc) = og;
                    *(ga_u16x8*)(a.gates_seq + bt * W4H + 3 * WW + cc) = oo;
                }
            }
        }
        __syncthreads();
    }
}

int main() {
    int B = 16384, T = 144;
    unsigned short *X, *h_seq, *gates, *c_seq, *wcat;
    float* bias;
    (void)hipMalloc(&X, (size_t)B * T * WW * 2);
    (void)hipMalloc(&wcat, (size_t)W4H * WK * 2);
    (void)hipMalloc(&bias, W4H * 4);
    (void)hipMalloc(&h_seq, (size_t)B * T * WW * 2);
    (void)hipMalloc(&c_seq, (size_t)B * T * WW * 2);
    (void)hipMalloc(&gates, (size_t)B * T * W4H * 2);
    (void)hipMemset(X, 0x3c, (size_t)B * T * WW * 2);
    (void)hipMemset(wcat, 0x38, (size_t)W4H * WK * 2);
    (void)hipMemset(bias, 0, W4H * 4);
    Args a{X, wcat, bias, h_seq, gates, c_seq, B, T};
    size_t smem = 6 * 64 * WW * 2;  // x + h + 4 pre = 96 KiB
    (void)hipFuncSetAttribute((const void*)lstm_fwd128,
                              hipFuncAttributeMaxDynamicSharedMemorySize, 131072);
    int grid = B / 64;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);
    std::vector<float> res;
    for (int round = 0; round < 3; ++round) {
        hipLaunchKernelGGL(lstm_fwd128, dim3(grid), dim3(256), smem, 0, a);
        (void)hipDeviceSynchronize();
        (void)hipEventRecord(e0);
        for (int k = 0; k < 3; ++k)
            hipLaunchKernelGGL(lstm_fwd128, dim3(grid), dim3(256), smem, 0, a);
        (void)hipEventRecord(e1);
        (void)hipEventSynchronize(e1);
        float ms; (void)hipEventElapsedTime(&ms, e0, e1);
        res.push_back(ms / 3);
    }
    std::sort(res.begin(), res.end());
    double win_s = B / (res[1] / 1e3);
    printf("lstm_fwd128: %.3f ms/call (B=%d, T=%d) -> fwd-only %.0f windows/s/layer\n",
           res[1], B, T, win_s);
    printf("(production step ~= 2.6x fwd per layer; est. full-step windows/s for 2 layers: %.0f)\n",
           win_s / (2.6 * 2));
    return 0;
}
