// Ablation microbenchmark for lstm_fwd's per-timestep cost structure.
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 micro_lstm.hip -o micro_lstm
// Variants (compile-time): V0 full; V1 no global writer; V2 no x-stage load;
// V3 no MFMA; V4 no transcendentals; V5 barriers+loop only.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
#include <algorithm>
#include "common.h"

#define LSTM_4H 256
#define LSTM_K 128

struct Args {
    const unsigned short* X;
    const int* win;
    const unsigned short* wcat;
    const float* bias;
    unsigned short* h_seq;
    unsigned short* gates_seq;
    unsigned short* c_seq;
    int B, T;
};

template <int ABL>
__launch_bounds__(256, 2)
__global__ void fwd(Args a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * GA_MT;
    const int wrow = wid * 16;
    extern __shared__ __attribute__((aligned(16))) char sm[];
    char* x_tile = sm;
    char* h_tile = sm + GA_TILE_BYTES;
    char* c_tile = sm + 2 * GA_TILE_BYTES;
    char* g_tiles = sm + 3 * GA_TILE_BYTES;
    float* s_bias = (float*)(sm + 7 * GA_TILE_BYTES);
    for (int i = tid; i < LSTM_4H; i += 256) s_bias[i] = a.bias[i];
    for (int c = tid; c < GA_MT * 8; c += 256) {
        ga_u16x8 z = {0,0,0,0,0,0,0,0};
        ga_tile_write8(h_tile, c >> 3, (c & 7) * 8, z);
    }
    float c_reg[4][4];
    for (int jt = 0; jt < 4; ++jt) for (int r = 0; r < 4; ++r) c_reg[jt][r] = 0.f;
    const int wm = tid >> 2, wc0 = (tid & 3) * 16, wgi = base + wm;
    __syncthreads();

    for (int t = 0; t < a.T; ++t) {
        if (ABL < 5) {
            for (int c = tid; c < GA_MT * 8; c += 256) {
                int m = c >> 3, col0 = (c & 7) * 8;
                int gi = base + m;
                ga_u16x8 v = {0,0,0,0,0,0,0,0};
                if (ABL != 2 && gi < a.B) {
                    int row = a.win[gi] + t;
                    v = *(const ga_u16x8*)(a.X + (size_t)row * GA_WMAX + col0);
                }
                ga_tile_write8(x_tile, m, col0, v);
            }
            if (ABL != 1 && t > 0 && wgi < a.B) {
                size_t bt = (size_t)wgi * a.T + (t - 1);
                for (int half = 0; half < 2; ++half) {
                    int c0 = wc0 + half * 8;
                    *(ga_u16x8*)(a.h_seq + bt * GA_WMAX + c0) = *(ga_u16x8*)&h_tile[ga_sw(wm, c0)];
                    *(ga_u16x8*)(a.c_seq + bt * GA_WMAX + c0) = *(ga_u16x8*)&c_tile[ga_sw(wm, c0)];
                    for (int g = 0; g < 4; ++g)
                        *(ga_u16x8*)(a.gates_seq + bt * LSTM_4H + g * GA_WMAX + c0) =
                            *(ga_u16x8*)&g_tiles[g * GA_TILE_BYTES + ga_sw(wm, c0)];
                }
            }
        }
        __syncthreads();
        if (ABL < 5) {
            unsigned short h_stage[4][4];
            for (int jt = 0; jt < 4; ++jt) {
                int col = jt * 16 + (lane & 15);
                ga_f32x4 acc4[4];
                for (int g = 0; g < 4; ++g) acc4[g] = ga_f32x4{0.f,0.f,0.f,0.f};
                if (ABL != 3) {
                    for (int kk = 0; kk < LSTM_K; kk += 32) {
                        const char* src = (kk < 64) ? x_tile : h_tile;
                        int klocal = kk & 63;
                        ga_bf16x8 afrag = ga_tile_read8(src, wrow + (lane & 15), klocal + (lane >> 4) * 8);
                        for (int g = 0; g < 4; ++g) {
                            ga_bf16x8 bfrag = *(const ga_bf16x8*)(a.wcat + (size_t)(g * GA_WMAX + jt * 16 + (lane & 15)) * LSTM_K + kk + (lane >> 4) * 8);
                            acc4[g] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc4[g], 0, 0, 0);
                        }
                    }
                }
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    float i_g, f_g, g_g, o_g;
                    if (ABL != 4) {
                        i_g = ga_sigmoid(acc4[0][r] + s_bias[col]);
                        f_g = ga_sigmoid(acc4[1][r] + s_bias[GA_WMAX + col]);
                        g_g = ga_tanh(acc4[2][r] + s_bias[2 * GA_WMAX + col]);
                        o_g = ga_sigmoid(acc4[3][r] + s_bias[3 * GA_WMAX + col]);
                    } else {
                        i_g = acc4[0][r]; f_g = acc4[1][r]; g_g = acc4[2][r]; o_g = acc4[3][r];
                    }
                    float c_new = f_g * c_reg[jt][r] + i_g * g_g;
                    c_reg[jt][r] = c_new;
                    float h_new = (ABL != 4) ? o_g * ga_tanh(c_new) : o_g * c_new;
                    ga_tile_write1(c_tile, m, col, ga_f32_to_bf16(c_new));
                    ga_tile_write1(g_tiles, m, col, ga_f32_to_bf16(i_g));
                    ga_tile_write1(g_tiles + GA_TILE_BYTES, m, col, ga_f32_to_bf16(f_g));
                    ga_tile_write1(g_tiles + 2 * GA_TILE_BYTES, m, col, ga_f32_to_bf16(g_g));
                    ga_tile_write1(g_tiles + 3 * GA_TILE_BYTES, m, col, ga_f32_to_bf16(o_g));
                    h_stage[jt][r] = ga_f32_to_bf16(h_new);
                }
            }
            __syncthreads();
            for (int jt = 0; jt < 4; ++jt) {
                int col = jt * 16 + (lane & 15);
                for (int r = 0; r < 4; ++r)
                    ga_tile_write1(h_tile, wrow + (lane >> 4) * 4 + r, col, h_stage[jt][r]);
            }
        } else {
            __syncthreads();
        }
        __syncthreads();
    }
    // final writer omitted (timing-only)
}

int main() {
    int B = 32768, T = 144, N = 65536;
    unsigned short *X, *h_seq, *gates_seq, *c_seq, *wcat;
    int* win;
    float* bias;
    (void)hipMalloc(&X, (size_t)N * 64 * 2);
    (void)hipMalloc(&win, (size_t)B * 4);
    (void)hipMalloc(&wcat, 256 * 128 * 2);
    (void)hipMalloc(&bias, 256 * 4);
    (void)hipMalloc(&h_seq, (size_t)B * T * 64 * 2);
    (void)hipMalloc(&c_seq, (size_t)B * T * 64 * 2);
    (void)hipMalloc(&gates_seq, (size_t)B * T * 256 * 2);
    std::vector<int> hwin(B);
    for (int i = 0; i < B; ++i) hwin[i] = (i * 7919) % (N - T);
    (void)hipMemcpy(win, hwin.data(), B * 4, hipMemcpyHostToDevice);
    (void)hipMemset(X, 0x3f, (size_t)N * 64 * 2);
    (void)hipMemset(wcat, 0x3c, 256 * 128 * 2);
    (void)hipMemset(bias, 0, 256 * 4);
    Args a{X, win, wcat, bias, h_seq, gates_seq, c_seq, B, T};
    size_t smem = 7 * GA_TILE_BYTES + 256 * 4;
    int grid = B / 64;
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);
    const char* names[6] = {"full", "no_writer", "no_xload", "no_mfma", "no_transc", "loop_only"};
    std::vector<std::vector<float>> res(6);
    for (int round = 0; round < 3; ++round) {
        for (int v = 0; v < 6; ++v) {
            auto launch = [&]() {
                switch (v) {
                    case 0: hipLaunchKernelGGL(fwd<0>, dim3(grid), dim3(256), smem, 0, a); break;
                    case 1: hipLaunchKernelGGL(fwd<1>, dim3(grid), dim3(256), smem, 0, a); break;
                    case 2: hipLaunchKernelGGL(fwd<2>, dim3(grid), dim3(256), smem, 0, a); break;
                    case 3: hipLaunchKernelGGL(fwd<3>, dim3(grid), dim3(256), smem, 0, a); break;
                    case 4: hipLaunchKernelGGL(fwd<4>, dim3(grid), dim3(256), smem, 0, a); break;
                    case 5: hipLaunchKernelGGL(fwd<5>, dim3(grid), dim3(256), smem, 0, a); break;
                }
            };
            launch();  // warm
            (void)hipDeviceSynchronize();
            (void)hipEventRecord(e0);
            for (int k = 0; k < 3; ++k) launch();
            (void)hipEventRecord(e1);
            (void)hipEventSynchronize(e1);
            float ms;
            (void)hipEventElapsedTime(&ms, e0, e1);
            res[v].push_back(ms / 3);
        }
    }
    for (int v = 0; v < 6; ++v) {
        std::sort(res[v].begin(), res[v].end());
        printf("%-10s median %.3f ms  min %.3f ms\n", names[v], res[v][1], res[v][0]);
    }
    return 0;
}
