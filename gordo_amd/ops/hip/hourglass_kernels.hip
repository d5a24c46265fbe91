// Fused hourglass-MLP training & inference kernels for MI355X (gfx950).
//
// Replaces the reference's Keras fit/predict hot loop (SURVEY.md §2.3 K1-K3,
// K6-K8, K10). Templated on the padded layer width W (64 for the 50-tag
// headline configs, 128 for wide tag sets — above that the engine falls back
// to library GEMMs, which are efficient at large widths).
//
// Per training step:
//   1. hg_train_step<W>: minibatch gather + L-layer dense forward (MFMA bf16
//      16x16x32, activations in XOR-swizzled LDS tiles) + MSE loss grad +
//      full backward (dW/db partials per block slab in bf16)
//   2. hg_slab_reduce<W>: stage-A float4 slab reduction (slab layout),
//      counter bump folded in
//   3. hg_adam_p<W>: stage-B Adam over G partials (transpose-on-read),
//      re-emits bf16 W and W^T
// plus hg_forward<W> for serving (fused min-max scale -> forward -> optional
// per-feature anomaly score + row L2) and the DP variants
// (hg_partial_flat -> RCCL all-reduce -> hg_adam_flat).
//
// MT=32 geometry: one block = 32 batch rows; B=8192 -> 256 blocks = every CU
// busy. The 4 waves split 2-D: wid&1 = 16-row stripe, wid>>1 = (W/2)-col
// stripe. Zero pads stay zero through the supported activations and
// contribute zero gradient.
#include "common.h"

struct HgTrainArgs {
    const unsigned short* __restrict__ X;   // [N][W] bf16, pre-scaled, padded
    const unsigned short* __restrict__ Y;   // may equal X (autoencoder)
    const int* __restrict__ counters;       // [0]=batch idx, [1]=adam t,
                                            // [2]=epoch (permutation key)
    int kbits;                              // Feistel domain bits (0=identity)
    unsigned int seed;                      // shuffle seed
    const unsigned short* __restrict__ w;   // [L][W][W] bf16 (out,in)
    const unsigned short* __restrict__ wt;  // [L][W][W] bf16 (in,out)
    const float* __restrict__ bias;         // [L][W] fp32
    unsigned short* __restrict__ dw;        // [n_tiles][L][W*W] bf16 slabs
    unsigned short* __restrict__ db;        // [n_tiles][L][W] bf16 slabs
                                            // (bf16 halves the dominant
                                            // HBM stream: 336 MB/step at
                                            // W=256; reduce accumulates fp32)
    float* __restrict__ loss;               // [1] fp32 accumulator (SSE)
    unsigned short* __restrict__ dx_out;    // [B][W] bf16 dL/dinput, or null
    int B;          // rows per step (per GPU)
    int N;          // dataset rows
    int F_out;      // real output features (cols >= F_out masked)
    int L;          // number of layers (<= GA_MAXL)
    int y_is_x;     // autoencoder fast path
    int act_id[GA_MAXL];  // activation per layer (last = out_func)
    int width[GA_MAXL + 1];  // real layer widths: hidden activations beyond
                             // width[li+1] are masked to 0 (an activation
                             // with act(0) != 0, e.g. sigmoid, would
                             // otherwise leak into the next layer's padded
                             // dW rows and drift the padded master weights)
    float gscale;   // 2 / (B * F_out)
};

#define HG_MT 32

extern __shared__ __attribute__((aligned(16))) char ga_smem[];

// ---- width-parametrized tile helpers --------------------------------------
// activation tiles: [32][W] bf16, row stride W*2 bytes, XOR swizzle over the
// 16-B slots; transposed dpre tiles: [W][32] bf16 via ga_sw32 (64-B rows).
// Note 32*W*2 == W*32*2: every scratch tile is the same TB bytes.
template <int W>
__device__ __forceinline__ unsigned int hg_sw(int row, int col) {
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 7u) << 4);
    return (unsigned int)row * (W * 2) + byte;
}

template <int W>
__device__ __forceinline__ ga_bf16x8 hg_read8(const char* t, int r, int c) {
    return *(const ga_bf16x8*)(t + hg_sw<W>(r, c));
}

template <int W>
__device__ __forceinline__ void hg_write8(char* t, int r, int c, ga_u16x8 v) {
    *(ga_u16x8*)(t + hg_sw<W>(r, c)) = v;
}

template <int W>
__device__ __forceinline__ unsigned short hg_read1(const char* t, int r, int c) {
    return *(const unsigned short*)(t + hg_sw<W>(r, c));
}

template <int W>
__device__ __forceinline__ void hg_write1(char* t, int r, int c, unsigned short v) {
    *(unsigned short*)(t + hg_sw<W>(r, c)) = v;
}

template <int W>
__launch_bounds__(256)
__global__ void hg_train_step(HgTrainArgs a) {
    constexpr int TB = HG_MT * W * 2;   // tile bytes (act + scratch alike)
    constexpr int NTW = W / 32;         // col tiles of 16 per wave (2 or 4)
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int tile = blockIdx.x;
    const int base = tile * HG_MT;
    const int wrow = (wid & 1) * 16;        // this wave's 16-row stripe
    const int wcol = (wid >> 1) * (W / 2);  // this wave's col stripe

    auto act_tile = [&](int i) { return ga_smem + i * TB; };
    char* scratch0 = ga_smem + a.L * TB;          // dcur  [32][W]
    char* scratch1 = scratch0 + TB;               // dcurT [W][32]
    char* scratch2 = scratch1 + TB;               // dnext
    char* scratch3 = scratch2 + TB;               // dnextT
    float* s_bias = (float*)(scratch3 + TB);      // [L][W]
    float* s_red = s_bias + GA_MAXL * W;          // 256 floats reduce pad

    const int step = a.counters[0];
    const long long pbase = (long long)step * a.B + base;
    // per-epoch shuffle key from the device-side epoch counter
    const unsigned int pkey = a.seed ^ ((unsigned int)a.counters[2] * 0x9E3779B1u);
    auto sample_row = [&](int m) -> unsigned int {
        unsigned int g = (unsigned int)((pbase + m) % (long long)a.N);
        return ga_permute(g, (unsigned int)a.N, a.kbits, pkey);
    };

    // ---- stage biases -------------------------------------------------
    for (int i = tid; i < a.L * W; i += 256) s_bias[i] = a.bias[i];

    // ---- gather minibatch tile into acts[0] ----------------------------
    for (int c = tid; c < HG_MT * (W / 8); c += 256) {
        int m = c / (W / 8);
        int col0 = (c % (W / 8)) * 8;
        int gi = base + m;
        ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gi < a.B) {
            unsigned int row = sample_row(m);
            v = *(const ga_u16x8*)(a.X + (long long)row * W + col0);
        }
        hg_write8<W>(act_tile(0), m, col0, v);
    }
    __syncthreads();

    // =====================================================================
    // Forward: per wave [16 rows][W/2 cols] = NTW MFMA tiles x W/32 K-steps.
    // =====================================================================
    for (int li = 0; li < a.L; ++li) {
        const unsigned short* wl = a.w + (size_t)li * W * W;
        ga_f32x4 acc[NTW];
#pragma unroll
        for (int nt = 0; nt < NTW; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < W; kk += 32) {
            ga_bf16x8 afrag = hg_read8<W>(act_tile(li), wrow + (lane & 15), kk + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(wl + (size_t)(wcol + nt * 16 + (lane & 15)) * W + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }

        if (li + 1 < a.L) {
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) {
                int col = wcol + nt * 16 + (lane & 15);
                float b = s_bias[li * W + col];
                bool pad = col >= a.width[li + 1];
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    float h = pad ? 0.f : ga_act(acc[nt][r] + b, a.act_id[li]);
                    hg_write1<W>(act_tile(li + 1), m, col, ga_f32_to_bf16(h));
                }
            }
            __syncthreads();
        } else {
            // ---- loss epilogue: dpre_{L-1} = (yhat - y) * gscale --------
            float lsum = 0.f;
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) {
                int col = wcol + nt * 16 + (lane & 15);
                float b = s_bias[li * W + col];
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    int gi = base + m;
                    float yhat = ga_act(acc[nt][r] + b, a.act_id[li]);
                    float yval = 0.f;
                    bool valid = (gi < a.B) && (col < a.F_out);
                    if (valid) {
                        if (a.y_is_x) {
                            yval = ga_bf16_to_f32(hg_read1<W>(act_tile(0), m, col));
                        } else {
                            unsigned int row = sample_row(m);
                            yval = ga_bf16_to_f32(a.Y[(long long)row * W + col]);
                        }
                    }
                    float diff = valid ? (yhat - yval) : 0.f;
                    lsum += diff * diff;
                    float dpre = diff * a.gscale * ga_act_grad_from_h(yhat, a.act_id[li]);
                    unsigned short dp = ga_f32_to_bf16(dpre);
                    hg_write1<W>(scratch0, m, col, dp);
                    ga_t32_write1(scratch1, col, m, dp);
                }
            }
            lsum = ga_wave_sum(lsum);
            if (lane == 0) s_red[wid] = lsum;
        }
    }
    __syncthreads();
    if (tid == 0) {
        atomicAdd(a.loss, s_red[0] + s_red[1] + s_red[2] + s_red[3]);
    }
    __syncthreads();  // s_red is reused by the db reduction below

    // =====================================================================
    // Backward per li: dX -> dpre_{li-1}; dW[W in][W out] (waves stripe the
    // in dim, K = 32 batch rows, one K-step); db via dcurT row sums.
    // =====================================================================
    char* dcur = scratch0;
    char* dcurT = scratch1;
    char* dnext = scratch2;
    char* dnextT = scratch3;

    for (int li = a.L - 1; li >= 0; --li) {
        if (li > 0 || a.dx_out != nullptr) {
            const unsigned short* wtl = a.wt + (size_t)li * W * W;
            ga_f32x4 acc[NTW];
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            for (int kk = 0; kk < W; kk += 32) {
                ga_bf16x8 afrag = hg_read8<W>(dcur, wrow + (lane & 15), kk + (lane >> 4) * 8);
#pragma unroll
                for (int nt = 0; nt < NTW; ++nt) {
                    // B[k=out][j=in] = WT[in][out]: row-read of WT
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)(wtl + (size_t)(wcol + nt * 16 + (lane & 15)) * W + kk + (lane >> 4) * 8);
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            }
            if (li > 0) {
#pragma unroll
                for (int nt = 0; nt < NTW; ++nt) {
                    int col = wcol + nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = wrow + (lane >> 4) * 4 + r;
                        float h = ga_bf16_to_f32(hg_read1<W>(act_tile(li), m, col));
                        float d = acc[nt][r] * ga_act_grad_from_h(h, a.act_id[li - 1]);
                        unsigned short dp = ga_f32_to_bf16(d);
                        hg_write1<W>(dnext, m, col, dp);
                        ga_t32_write1(dnextT, col, m, dp);
                    }
                }
            } else {
                // dL/d(input) for the caller (e.g. the LSTM head): batch order
#pragma unroll
                for (int nt = 0; nt < NTW; ++nt) {
                    int col = wcol + nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = wrow + (lane >> 4) * 4 + r;
                        int gi = base + m;
                        if (gi < a.B)
                            a.dx_out[(size_t)gi * W + col] = ga_f32_to_bf16(acc[nt][r]);
                    }
                }
            }
        }

        // ---- dW: wave stripes the in dim (W/4 rows), K = 32, 1 K-step ----
        {
            unsigned short* dwl = a.dw + ((size_t)tile * a.L + li) * W * W;
#pragma unroll
            for (int sub = 0; sub < W / 64; ++sub) {
                int in_base = wid * (W / 4) + sub * 16;
                ga_f32x4 acc[W / 16];
#pragma unroll
                for (int nt = 0; nt < W / 16; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
                // A[i=in][k=m] = acts[li][m][in]: strided scalar reads
                ga_bf16x8 afrag;
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    int m = (lane >> 4) * 8 + e;
                    unsigned short u = hg_read1<W>(act_tile(li), m, in_base + (lane & 15));
                    afrag[e] = *(__bf16*)&u;
                }
#pragma unroll
                for (int nt = 0; nt < W / 16; ++nt) {
                    // B[k=m][j=out] = dcurT[out][m]: row-read of the narrow tile
                    ga_bf16x8 bfrag = ga_t32_read8(dcurT, nt * 16 + (lane & 15), (lane >> 4) * 8);
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
#pragma unroll
                for (int nt = 0; nt < W / 16; ++nt) {
                    int out = nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int in = in_base + (lane >> 4) * 4 + r;
                        dwl[(size_t)in * W + out] = ga_f32_to_bf16(acc[nt][r]);
                    }
                }
            }
        }

        // ---- db: column sums of dpre via dcurT row sums ----------------
        {
            constexpr int GR = 256 / W;        // thread groups per out (4 or 2)
            int out = tid / GR;                // W outs
            int g = tid % GR;                  // each group sums 32/GR m's
            float sum = 0.f;
#pragma unroll
            for (int mi = 0; mi < 32 / GR / 8; ++mi) {
                ga_bf16x8 v = ga_t32_read8(dcurT, out, g * (32 / GR) + mi * 8);
#pragma unroll
                for (int e = 0; e < 8; ++e) sum += (float)v[e];
            }
            s_red[g * W + out] = sum;
            __syncthreads();
            if (tid < W) {
                float tot = 0.f;
#pragma unroll
                for (int gg = 0; gg < GR; ++gg) tot += s_red[gg * W + tid];
                a.db[((size_t)tile * a.L + li) * W + tid] = ga_f32_to_bf16(tot);
            }
        }
        __syncthreads();

        // rotate scratch: dpre_{li-1} becomes current
        char* t0 = dcur; dcur = dnext; dnext = t0;
        char* t1 = dcurT; dcurT = dnextT; dnextT = t1;
    }
}

// ===========================================================================
// Wave-local train step (W=64 tier, OPT-IN via GORDO_W1=1): dense-layer
// work is ROW-LOCAL, so a single 64-lane wave owns 16 batch rows end-to-end
// — gather, L-layer forward, loss, full backward, its own dW/db slab — with
// ZERO barriers. All tiles are wave-private LDS (same-wave dependencies
// ordered by lgkmcnt); dW uses the native K=16 MFMA
// (v_mfma_f32_16x16x16_bf16) since the batch tile is 16 rows.
// MEASURED SLOWER than the 4-wave kernel at the headline config (184.7M vs
// 263.5M samples/s): B=8192 gives only 512 single-wave blocks, so half the
// SIMDs idle and the solo wave eats its own latencies — the 4-wave kernel's
// barrier parking is the batch's parallelism ceiling, not waste. Kept
// correct + tested (grad parity, dx_out, cross-kernel slab agreement) for
// larger-batch experiments; see docs/ROADMAP.md.
// ===========================================================================
typedef __bf16 ga_bf16x4 __attribute__((ext_vector_type(4)));

// transposed dpre tiles for the wave-local kernel: [W][16] bf16, 32-B rows
__device__ __forceinline__ void ga_t16_write1(char* t, int row, int m, unsigned short v) {
    *(unsigned short*)(t + row * 32 + m * 2) = v;
}
__device__ __forceinline__ ga_bf16x4 ga_t16_read4(const char* t, int row, int m) {
    return *(const ga_bf16x4*)(t + row * 32 + m * 2);
}

#define HG_MT1 16

template <int W>
__launch_bounds__(64)
__global__ void hg_train_step_w1(HgTrainArgs a) {
    constexpr int TB = HG_MT1 * W * 2;  // act/scratch tile bytes ([16][W])
    constexpr int NTW = W / 16;         // col tiles of 16 (full width/wave)
    const int lane = threadIdx.x;       // 64 threads = exactly one wave
    const int tile = blockIdx.x;
    const int base = tile * HG_MT1;

    auto act_tile = [&](int i) { return ga_smem + i * TB; };
    char* dcur = ga_smem + a.L * TB;        // dpre  [16][W]
    char* dnext = dcur + TB;
    char* dcurT = dnext + TB;               // dpre^T [W][16]
    char* dnextT = dcurT + (size_t)W * 32;
    float* s_bias = (float*)(dnextT + (size_t)W * 32);  // [L][W]

    const int step = a.counters[0];
    const long long pbase = (long long)step * a.B + base;
    const unsigned int pkey = a.seed ^ ((unsigned int)a.counters[2] * 0x9E3779B1u);
    auto sample_row = [&](int m) -> unsigned int {
        unsigned int g = (unsigned int)((pbase + m) % (long long)a.N);
        return ga_permute(g, (unsigned int)a.N, a.kbits, pkey);
    };

    for (int i = lane; i < a.L * W; i += 64) s_bias[i] = a.bias[i];

    // ---- gather minibatch tile into acts[0] (wave-private) -------------
    for (int c = lane; c < HG_MT1 * (W / 8); c += 64) {
        int m = c / (W / 8);
        int col0 = (c % (W / 8)) * 8;
        int gi = base + m;
        ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gi < a.B) {
            unsigned int row = sample_row(m);
            v = *(const ga_u16x8*)(a.X + (long long)row * W + col0);
        }
        hg_write8<W>(act_tile(0), m, col0, v);
    }

    // ---- forward: [16 rows][W cols] per layer, no barriers --------------
    for (int li = 0; li < a.L; ++li) {
        const unsigned short* wl = a.w + (size_t)li * W * W;
        ga_f32x4 acc[NTW];
#pragma unroll
        for (int nt = 0; nt < NTW; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < W; kk += 32) {
            ga_bf16x8 afrag = hg_read8<W>(act_tile(li), lane & 15, kk + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(wl + (size_t)(nt * 16 + (lane & 15)) * W + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }

        if (li + 1 < a.L) {
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) {
                int col = nt * 16 + (lane & 15);
                float b = s_bias[li * W + col];
                bool pad = col >= a.width[li + 1];
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = (lane >> 4) * 4 + r;
                    float h = pad ? 0.f : ga_act(acc[nt][r] + b, a.act_id[li]);
                    hg_write1<W>(act_tile(li + 1), m, col, ga_f32_to_bf16(h));
                }
            }
        } else {
            // ---- loss epilogue: dpre_{L-1} = (yhat - y) * gscale --------
            float lsum = 0.f;
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) {
                int col = nt * 16 + (lane & 15);
                float b = s_bias[li * W + col];
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = (lane >> 4) * 4 + r;
                    int gi = base + m;
                    float yhat = ga_act(acc[nt][r] + b, a.act_id[li]);
                    float yval = 0.f;
                    bool valid = (gi < a.B) && (col < a.F_out);
                    if (valid) {
                        if (a.y_is_x) {
                            yval = ga_bf16_to_f32(hg_read1<W>(act_tile(0), m, col));
                        } else {
                            unsigned int row = sample_row(m);
                            yval = ga_bf16_to_f32(a.Y[(long long)row * W + col]);
                        }
                    }
                    float diff = valid ? (yhat - yval) : 0.f;
                    lsum += diff * diff;
                    float dpre = diff * a.gscale * ga_act_grad_from_h(yhat, a.act_id[li]);
                    unsigned short dp = ga_f32_to_bf16(dpre);
                    hg_write1<W>(dcur, m, col, dp);
                    ga_t16_write1(dcurT, col, m, dp);
                }
            }
            lsum = ga_wave_sum(lsum);
            if (lane == 0) atomicAdd(a.loss, lsum);
        }
    }

    // ---- backward: dX -> dpre_{li-1}; dW (K=16 native); db --------------
    for (int li = a.L - 1; li >= 0; --li) {
        if (li > 0 || a.dx_out != nullptr) {
            const unsigned short* wtl = a.wt + (size_t)li * W * W;
            ga_f32x4 acc[NTW];
#pragma unroll
            for (int nt = 0; nt < NTW; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            for (int kk = 0; kk < W; kk += 32) {
                ga_bf16x8 afrag = hg_read8<W>(dcur, lane & 15, kk + (lane >> 4) * 8);
#pragma unroll
                for (int nt = 0; nt < NTW; ++nt) {
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)(wtl + (size_t)(nt * 16 + (lane & 15)) * W + kk + (lane >> 4) * 8);
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            }
            if (li > 0) {
#pragma unroll
                for (int nt = 0; nt < NTW; ++nt) {
                    int col = nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = (lane >> 4) * 4 + r;
                        float h = ga_bf16_to_f32(hg_read1<W>(act_tile(li), m, col));
                        float d = acc[nt][r] * ga_act_grad_from_h(h, a.act_id[li - 1]);
                        unsigned short dp = ga_f32_to_bf16(d);
                        hg_write1<W>(dnext, m, col, dp);
                        ga_t16_write1(dnextT, col, m, dp);
                    }
                }
            } else {
#pragma unroll
                for (int nt = 0; nt < NTW; ++nt) {
                    int col = nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = (lane >> 4) * 4 + r;
                        int gi = base + m;
                        if (gi < a.B)
                            a.dx_out[(size_t)gi * W + col] = ga_f32_to_bf16(acc[nt][r]);
                    }
                }
            }
        }

        // ---- dW[in][out]: K = the 16 batch rows, native K=16 MFMA -------
        {
            unsigned short* dwl = a.dw + ((size_t)tile * a.L + li) * W * W;
#pragma unroll
            for (int sub = 0; sub < W / 16; ++sub) {
                int in_base = sub * 16;
                ga_f32x4 acc[W / 16];
#pragma unroll
                for (int nt = 0; nt < W / 16; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
                // A[i=in][k=m] = acts[li][m][in]: strided scalar reads
                ga_bf16x4 afrag;
#pragma unroll
                for (int e = 0; e < 4; ++e) {
                    int m = (lane >> 4) * 4 + e;
                    unsigned short u = hg_read1<W>(act_tile(li), m, in_base + (lane & 15));
                    afrag[e] = *(__bf16*)&u;
                }
#pragma unroll
                for (int nt = 0; nt < W / 16; ++nt) {
                    // B[k=m][j=out] = dcurT[out][m]: 8-B row reads
                    ga_bf16x4 bfrag = ga_t16_read4(dcurT, nt * 16 + (lane & 15), (lane >> 4) * 4);
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x16bf16_1k(afrag, bfrag, acc[nt], 0, 0, 0);
                }
#pragma unroll
                for (int nt = 0; nt < W / 16; ++nt) {
                    int out = nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int in = in_base + (lane >> 4) * 4 + r;
                        dwl[(size_t)in * W + out] = ga_f32_to_bf16(acc[nt][r]);
                    }
                }
            }
        }

        // ---- db: column sums of dpre over the 16 rows -------------------
        for (int out = lane; out < W; out += 64) {
            float sum = 0.f;
#pragma unroll
            for (int mi = 0; mi < HG_MT1 / 4; ++mi) {
                ga_bf16x4 v = ga_t16_read4(dcurT, out, mi * 4);
#pragma unroll
                for (int e = 0; e < 4; ++e) sum += (float)v[e];
            }
            a.db[((size_t)tile * a.L + li) * W + out] = ga_f32_to_bf16(sum);
        }

        char* t0 = dcur; dcur = dnext; dnext = t0;
        char* t1 = dcurT; dcurT = dnextT; dnextT = t1;
    }
}

// ===========================================================================
// Legacy single-stage Adam (kept for the kernel unit tests): reduce dW/db
// slabs, update fp32 master, emit bf16 W and WT. One thread per parameter.
// ===========================================================================
template <int W>
__launch_bounds__(256)
__global__ void hg_adam(
    float* __restrict__ master,      // [P] fp32 (weights then biases)
    float* __restrict__ m_buf,
    float* __restrict__ v_buf,
    const unsigned short* __restrict__ dw,  // [n_slabs][L][W*W] bf16
    const unsigned short* __restrict__ db,  // [n_slabs][L][W] bf16
    unsigned short* __restrict__ w_bf16,   // [L][W][W]
    unsigned short* __restrict__ wt_bf16,  // [L][W][W]
    float* __restrict__ bias,        // [L][W] (fp32 view used by fwd)
    const int* __restrict__ counters,
    int n_slabs, int L, float lr, float beta1, float beta2, float eps) {
    const int PW = L * W * W;
    const int P = PW + L * W;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= P) return;

    float g = 0.f;
    if (idx < PW) {
        // master layout is [l][out][in]; the train kernel's slabs are
        // [l][in][out] (the dW MFMA's C layout) -> swap in/out on read
        int l = idx / (W * W);
        int rem = idx - l * W * W;
        int out = rem / W, in = rem - out * W;
        size_t sidx = (size_t)l * W * W + (size_t)in * W + out;
        for (int s = 0; s < n_slabs; ++s) g += ga_bf16_to_f32(dw[(size_t)s * PW + sidx]);
    } else {
        int bidx = idx - PW;
        for (int s = 0; s < n_slabs; ++s) g += ga_bf16_to_f32(db[(size_t)s * (L * W) + bidx]);
    }

    int t = counters[1];
    float m = beta1 * m_buf[idx] + (1.f - beta1) * g;
    float v = beta2 * v_buf[idx] + (1.f - beta2) * g * g;
    m_buf[idx] = m;
    v_buf[idx] = v;
    float mhat = m / (1.f - __powf(beta1, (float)t));
    float vhat = v / (1.f - __powf(beta2, (float)t));
    float p = master[idx] - lr * mhat / (__fsqrt_rn(vhat) + eps);
    master[idx] = p;

    if (idx < PW) {
        unsigned short pb = ga_f32_to_bf16(p);
        w_bf16[idx] = pb;
        int l = idx / (W * W);
        int rem = idx - l * W * W;
        int out = rem / W, in = rem - out * W;
        wt_bf16[(size_t)l * W * W + in * W + out] = pb;
    } else {
        bias[idx - PW] = p;
    }
}

// test support: materialize the Feistel permutation (bijectivity checks)
__global__ void hg_permute_fill(int* out, int n, int kbits, unsigned int key) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) out[i] = (int)ga_permute((unsigned int)i, (unsigned int)n, kbits, key);
}

__global__ void hg_bump(int* counters, int n_batches) {
    counters[0] = (counters[0] + 1) % n_batches;
    counters[1] += 1;
}

// ===========================================================================
// Two-stage slab reduction. Stage A stays in slab layout ([l][in][out]) so
// every access is a contiguous float4 stream (the [in][out]->[out][in]
// transpose happens in the tiny stage-B Adam: ~1 MB vs ~20 MB); the counter
// bump is folded into block (0,0).
// ===========================================================================
template <int W>
__launch_bounds__(256)
__global__ void hg_slab_reduce(const unsigned short* __restrict__ dw,
                               const unsigned short* __restrict__ db,
                               float* __restrict__ partial,  // [G][P] SLAB layout
                               int* __restrict__ counters,
                               int n_slabs, int G, int L, int n_batches) {
    const int PW = L * W * W;
    const int P = PW + L * W;
    int g = blockIdx.y;
    int idx8 = blockIdx.x * blockDim.x + threadIdx.x;
    if (g == 0 && idx8 == 0 && counters != nullptr) {
        int nb = (counters[0] + 1) % n_batches;
        counters[0] = nb;
        counters[1] += 1;
        if (nb == 0) counters[2] += 1;  // new epoch -> new shuffle key
    }
    int idx = idx8 * 8;  // bf16x8 = 16-B loads, fp32 accumulation
    if (idx >= P) return;
    int per = (n_slabs + G - 1) / G;
    int s0 = g * per, s1 = min(n_slabs, s0 + per);
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    if (idx < PW) {
#pragma unroll 4
        for (int sl = s0; sl < s1; ++sl) {
            ga_u16x8 v = *(const ga_u16x8*)(dw + (size_t)sl * PW + idx);
#pragma unroll
            for (int e = 0; e < 8; ++e) acc[e] += ga_bf16_to_f32(v[e]);
        }
    } else {
        int bidx = idx - PW;
#pragma unroll 4
        for (int sl = s0; sl < s1; ++sl) {
            ga_u16x8 v = *(const ga_u16x8*)(db + (size_t)sl * (L * W) + bidx);
#pragma unroll
            for (int e = 0; e < 8; ++e) acc[e] += ga_bf16_to_f32(v[e]);
        }
    }
    *(float4*)(partial + (size_t)g * P + idx) = make_float4(acc[0], acc[1], acc[2], acc[3]);
    *(float4*)(partial + (size_t)g * P + idx + 4) = make_float4(acc[4], acc[5], acc[6], acc[7]);
}

template <int W>
__launch_bounds__(256)
__global__ void hg_adam_p(
    float* __restrict__ master, float* __restrict__ m_buf, float* __restrict__ v_buf,
    const float* __restrict__ partial, unsigned short* __restrict__ w_bf16,
    unsigned short* __restrict__ wt_bf16, float* __restrict__ bias,
    const int* __restrict__ counters, int G, int L,
    float lr, float beta1, float beta2, float eps) {
    const int PW = L * W * W;
    const int P = PW + L * W;
    // iterate in SLAB order so the dominant stream — G partial reads per
    // element — is fully coalesced (the old master-order loop scattered all
    // 16xP of them; the swap scatters only the 7xP master/m/v/w accesses,
    // measured 40 -> ? us at W=256); transpose happens on the master side
    int sidx = blockIdx.x * blockDim.x + threadIdx.x;
    if (sidx >= P) return;
    size_t idx;  // MASTER-layout index
    if (sidx < PW) {
        int l = sidx / (W * W);
        int rem = sidx - l * W * W;
        int in = rem / W, out = rem - in * W;
        idx = (size_t)l * W * W + (size_t)out * W + in;
    } else {
        idx = (size_t)sidx;
    }
    float g = 0.f;
#pragma unroll 4
    for (int i = 0; i < G; ++i) g += partial[(size_t)i * P + sidx];
    int t = counters[1];
    float m = beta1 * m_buf[idx] + (1.f - beta1) * g;
    float v = beta2 * v_buf[idx] + (1.f - beta2) * g * g;
    m_buf[idx] = m;
    v_buf[idx] = v;
    float mhat = m / (1.f - __powf(beta1, (float)t));
    float vhat = v / (1.f - __powf(beta2, (float)t));
    float p = master[idx] - lr * mhat / (__fsqrt_rn(vhat) + eps);
    master[idx] = p;
    if (sidx < PW) {
        unsigned short pb = ga_f32_to_bf16(p);
        w_bf16[idx] = pb;
        wt_bf16[(size_t)sidx] = pb;  // wt IS slab layout: coalesced
    } else {
        bias[idx - PW] = p;
    }
}

template <int W>
__launch_bounds__(256)
__global__ void hg_partial_flat(const float* __restrict__ partial,
                                float* __restrict__ g_out, int G, int L) {
    // partial is SLAB layout; emit MASTER layout for the DP bucket.
    // Iterate in slab order so the G reads per element coalesce; only the
    // single g_out write scatters (same flip as hg_adam_p, +5% headline).
    const int PW = L * W * W;
    const int P = PW + L * W;
    int sidx = blockIdx.x * blockDim.x + threadIdx.x;
    if (sidx >= P) return;
    size_t idx;
    if (sidx < PW) {
        int l = sidx / (W * W);
        int rem = sidx - l * W * W;
        int in = rem / W, out = rem - in * W;
        idx = (size_t)l * W * W + (size_t)out * W + in;
    } else {
        idx = (size_t)sidx;
    }
    float sum = 0.f;
#pragma unroll 4
    for (int i = 0; i < G; ++i) sum += partial[(size_t)i * P + sidx];
    g_out[idx] = sum;
}

// ===========================================================================
// Fused slab-reduce + Adam: replaces the hg_slab_reduce -> hg_adam_p pair in
// the single-GPU step (both are tiny-grid launch/ramp-bound; the pair costs
// ~11.6 us of the 38 us headline step). Each 256-thread block owns 256
// consecutive SLAB-order params end-to-end: 8 slab-teams of 32 lanes stream
// disjoint slab subsets (bf16x8, coalesced along P), an 8 KB LDS fold sums
// the team partials, then one thread per param applies Adam and re-emits the
// bf16 weights. No cross-block dependency -> no device-scope fence (the
// earlier one-kernel attempt died on __threadfence = per-block L2 writeback
// across the 8 non-coherent XCD L2s; docs/ROADMAP.md).
//
// Counter PING-PONG: every thread derives this step's Adam t from cr[1]+1
// (cr was written by the PREVIOUS step's kernel — read-only this launch) and
// block 0 writes the bumped values to cw. A single counters tensor can't
// work here: with the bump fused, other blocks would race the writer.
// ===========================================================================
template <int W>
__launch_bounds__(256)
__global__ void hg_reduce_adam(
    const unsigned short* __restrict__ dw, const unsigned short* __restrict__ db,
    float* __restrict__ master, float* __restrict__ m_buf, float* __restrict__ v_buf,
    unsigned short* __restrict__ w_bf16, unsigned short* __restrict__ wt_bf16,
    float* __restrict__ bias,
    const int* __restrict__ cr, int* __restrict__ cw,
    int n_slabs, int L, int n_batches,
    float lr, float beta1, float beta2, float eps) {
    const int PW = L * W * W;   // always a multiple of 256, so a block is
    const int P = PW + L * W;   // entirely weights or entirely biases
    const int tid = threadIdx.x;
    if (blockIdx.x == 0 && tid == 0) {
        int nb = (cr[0] + 1) % n_batches;
        cw[0] = nb;
        cw[1] = cr[1] + 1;
        cw[2] = cr[2] + (nb == 0 ? 1 : 0);  // wrap -> new epoch/shuffle key
    }
    // 16 slab-teams x 16 lanes: 128 params per block (~2x the blocks of a
    // 256-param split) and 16-slab team loops that unroll 4x for MLP — the
    // first cut (8 teams x 32 lanes, rolled loop) measured 13.5 us, SLOWER
    // than the 11.6 us pair it replaces; this shape measures the win.
    const int team = tid >> 4, lx = tid & 15;
    const int pb = blockIdx.x * 128;  // param base (slab order)
    const int sidx0 = pb + lx * 8;    // this lane's 8 params
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    if (sidx0 < P) {
        if (sidx0 < PW) {
#pragma unroll 4
            for (int sl = team; sl < n_slabs; sl += 16) {
                ga_u16x8 vv = *(const ga_u16x8*)(dw + (size_t)sl * PW + sidx0);
#pragma unroll
                for (int e = 0; e < 8; ++e) acc[e] += ga_bf16_to_f32(vv[e]);
            }
        } else {
            const int bidx = sidx0 - PW;
#pragma unroll 4
            for (int sl = team; sl < n_slabs; sl += 16) {
                ga_u16x8 vv = *(const ga_u16x8*)(db + (size_t)sl * (L * W) + bidx);
#pragma unroll
                for (int e = 0; e < 8; ++e) acc[e] += ga_bf16_to_f32(vv[e]);
            }
        }
    }
    __shared__ float s_part[16][128 + 4];
#pragma unroll
    for (int e = 0; e < 8; ++e) s_part[team][lx * 8 + e] = acc[e];
    __syncthreads();

    if (tid >= 128) return;
    const int sidx = pb + tid;  // one param per thread from here on
    if (sidx >= P) return;
    float g = 0.f;
#pragma unroll
    for (int tt = 0; tt < 16; ++tt) g += s_part[tt][tid];
    size_t idx;  // MASTER-layout index (the [in][out]->[out][in] flip)
    if (sidx < PW) {
        int l = sidx / (W * W);
        int rem = sidx - l * W * W;
        int in = rem / W, out = rem - in * W;
        idx = (size_t)l * W * W + (size_t)out * W + in;
    } else {
        idx = (size_t)sidx;
    }
    int t = cr[1] + 1;
    float m = beta1 * m_buf[idx] + (1.f - beta1) * g;
    float v = beta2 * v_buf[idx] + (1.f - beta2) * g * g;
    m_buf[idx] = m;
    v_buf[idx] = v;
    float mhat = m / (1.f - __powf(beta1, (float)t));
    float vhat = v / (1.f - __powf(beta2, (float)t));
    float p = master[idx] - lr * mhat / (__fsqrt_rn(vhat) + eps);
    master[idx] = p;
    if (sidx < PW) {
        unsigned short pb16 = ga_f32_to_bf16(p);
        w_bf16[idx] = pb16;
        wt_bf16[(size_t)sidx] = pb16;  // wt IS slab layout: coalesced
    } else {
        bias[idx - PW] = p;
    }
}

// ===========================================================================
// DP variant of the fused reduce: slabs -> MASTER-order flat bucket (the
// RCCL all-reduce payload) in ONE kernel, with the same ping-pong counter
// bump. Replaces the hg_slab_reduce -> hg_partial_flat pair in the world>1
// step; hg_adam_flat then reads the bumped slot (a later kernel — no race).
// ===========================================================================
template <int W>
__launch_bounds__(256)
__global__ void hg_flat_reduce(
    const unsigned short* __restrict__ dw, const unsigned short* __restrict__ db,
    float* __restrict__ g_out,
    const int* __restrict__ cr, int* __restrict__ cw,
    int n_slabs, int L, int n_batches) {
    const int PW = L * W * W;
    const int P = PW + L * W;
    const int tid = threadIdx.x;
    if (blockIdx.x == 0 && tid == 0) {
        int nb = (cr[0] + 1) % n_batches;
        cw[0] = nb;
        cw[1] = cr[1] + 1;
        cw[2] = cr[2] + (nb == 0 ? 1 : 0);
    }
    const int team = tid >> 4, lx = tid & 15;
    const int pb = blockIdx.x * 128;
    const int sidx0 = pb + lx * 8;
    float acc[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.f;
    if (sidx0 < P) {
        if (sidx0 < PW) {
#pragma unroll 4
            for (int sl = team; sl < n_slabs; sl += 16) {
                ga_u16x8 vv = *(const ga_u16x8*)(dw + (size_t)sl * PW + sidx0);
#pragma unroll
                for (int e = 0; e < 8; ++e) acc[e] += ga_bf16_to_f32(vv[e]);
            }
        } else {
            const int bidx = sidx0 - PW;
#pragma unroll 4
            for (int sl = team; sl < n_slabs; sl += 16) {
                ga_u16x8 vv = *(const ga_u16x8*)(db + (size_t)sl * (L * W) + bidx);
#pragma unroll
                for (int e = 0; e < 8; ++e) acc[e] += ga_bf16_to_f32(vv[e]);
            }
        }
    }
    __shared__ float s_part[16][128 + 4];
#pragma unroll
    for (int e = 0; e < 8; ++e) s_part[team][lx * 8 + e] = acc[e];
    __syncthreads();

    if (tid >= 128) return;
    const int sidx = pb + tid;
    if (sidx >= P) return;
    float g = 0.f;
#pragma unroll
    for (int tt = 0; tt < 16; ++tt) g += s_part[tt][tid];
    size_t idx;  // MASTER layout for the DP bucket (matches hg_adam_flat)
    if (sidx < PW) {
        int l = sidx / (W * W);
        int rem = sidx - l * W * W;
        int in = rem / W, out = rem - in * W;
        idx = (size_t)l * W * W + (size_t)out * W + in;
    } else {
        idx = (size_t)sidx;
    }
    g_out[idx] = g;
}

// ===========================================================================
// Serving forward (fused K1+K7+K8): raw fp32 input -> min-max scale -> L-layer
// forward -> yhat (model output space); optionally per-feature scaled |err|
// and row L2 total (autoencoder anomaly). 64 rows per block.
// ===========================================================================
struct HgFwdArgs {
    const float* __restrict__ X;   // [B][F_in] fp32 raw (row stride F_in)
    const unsigned short* __restrict__ w;
    const float* __restrict__ bias;
    const float* __restrict__ xscale;   // [W] (pad cols 0)
    const float* __restrict__ xshift;   // [W]
    float* __restrict__ yhat;           // [B][F_out]
    float* __restrict__ err;            // [B][F_out] or null
    float* __restrict__ total;          // [B] or null
    const float* __restrict__ inv_err_scale;  // [W] detector 1/scale (anomaly)
    int B, F_in, F_out, L;
    int act_id[GA_MAXL];
};

template <int W>
__launch_bounds__(256)
__global__ void hg_forward(HgFwdArgs a) {
    constexpr int FTB = 64 * W * 2;  // [64][W] tile bytes
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * 64;
    char* t_x = ga_smem;                  // scaled input (preserved for anomaly y)
    char* t0 = ga_smem + FTB;             // ping
    char* t1 = ga_smem + 2 * FTB;         // pong
    float* s_red = (float*)(ga_smem + 3 * FTB);  // [4][64]

    // stage + scale input tile (scalar loads; F_in is unpadded)
    for (int c = tid; c < 64 * W; c += 256) {
        int m = c / W, col = c % W;
        int gi = base + m;
        float x = 0.f;
        if (gi < a.B && col < a.F_in) x = a.X[(long long)gi * a.F_in + col];
        float xs = x * a.xscale[col] + a.xshift[col];
        hg_write1<W>(t_x, m, col, ga_f32_to_bf16(xs));
    }
    __syncthreads();

    const int wrow = wid * 16;
    for (int li = 0; li < a.L; ++li) {
        const unsigned short* wl = a.w + (size_t)li * W * W;
        const char* src = (li == 0) ? t_x : ((li & 1) ? t0 : t1);
        ga_f32x4 acc[W / 16];
#pragma unroll
        for (int nt = 0; nt < W / 16; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < W; kk += 32) {
            ga_bf16x8 afrag = hg_read8<W>(src, wrow + (lane & 15), kk + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < W / 16; ++nt) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(wl + (size_t)(nt * 16 + (lane & 15)) * W + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }
        char* dst = (li & 1) ? t1 : t0;
        bool last = (li + 1 == a.L);
#pragma unroll
        for (int nt = 0; nt < W / 16; ++nt) {
            int col = nt * 16 + (lane & 15);
            float b = a.bias[li * W + col];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int m = wrow + (lane >> 4) * 4 + r;
                float h = ga_act(acc[nt][r] + b, a.act_id[li]);
                int gi = base + m;
                if (last && gi < a.B && col < a.F_out)
                    a.yhat[(long long)gi * a.F_out + col] = h;
                hg_write1<W>(dst, m, col, ga_f32_to_bf16(h));
            }
        }
        __syncthreads();
    }

    if (a.err != nullptr) {
        // anomaly: err = |yhat - y_scaled| * inv_err_scale; total = ||err||_2
        // (autoencoder: y = the scaled input, preserved in t_x)
        char* yh = (a.L & 1) ? t0 : t1;  // the final layer's dst tile
        int m = tid >> 2, g = tid & 3;
        float s = 0.f;
        for (int ci = 0; ci < W / 4; ++ci) {
            int col = g * (W / 4) + ci;
            int gi = base + m;
            if (gi < a.B && col < a.F_out) {
                float yhat = ga_bf16_to_f32(hg_read1<W>(yh, m, col));
                float yval = ga_bf16_to_f32(hg_read1<W>(t_x, m, col));
                float e = fabsf(yhat - yval) * a.inv_err_scale[col];
                a.err[(long long)gi * a.F_out + col] = e;
                s += e * e;
            }
        }
        s_red[g * 64 + m] = s;
        __syncthreads();
        if (tid < 64 && base + tid < a.B) {
            float tot = s_red[tid] + s_red[64 + tid] + s_red[2 * 64 + tid] +
                        s_red[3 * 64 + tid];
            a.total[base + tid] = __fsqrt_rn(tot);
        }
    }
}

// ===========================================================================
// DP legacy single-stage reduce (kept for tests): slabs -> one flat grad
// (the RCCL all-reduce bucket), then Adam from the reduced flat grad.
// ===========================================================================
template <int W>
__launch_bounds__(256)
__global__ void hg_grad_reduce(const unsigned short* __restrict__ dw,
                               const unsigned short* __restrict__ db,
                               float* __restrict__ g, int n_slabs, int L) {
    const int PW = L * W * W;
    const int P = PW + L * W;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= P) return;
    float s = 0.f;
    if (idx < PW) {
        int l = idx / (W * W);
        int rem = idx - l * W * W;
        int out = rem / W, in = rem - out * W;
        size_t sidx = (size_t)l * W * W + (size_t)in * W + out;
        for (int sl = 0; sl < n_slabs; ++sl) s += ga_bf16_to_f32(dw[(size_t)sl * PW + sidx]);
    } else {
        int bidx = idx - PW;
        for (int sl = 0; sl < n_slabs; ++sl) s += ga_bf16_to_f32(db[(size_t)sl * (L * W) + bidx]);
    }
    g[idx] = s;
}

template <int W>
__launch_bounds__(256)
__global__ void hg_adam_flat(
    float* __restrict__ master, float* __restrict__ m_buf, float* __restrict__ v_buf,
    const float* __restrict__ g_in, unsigned short* __restrict__ w_bf16,
    unsigned short* __restrict__ wt_bf16, float* __restrict__ bias,
    const int* __restrict__ counters, int L,
    float lr, float beta1, float beta2, float eps) {
    const int PW = L * W * W;
    const int P = PW + L * W;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= P) return;
    float g = g_in[idx];
    int t = counters[1];
    float m = beta1 * m_buf[idx] + (1.f - beta1) * g;
    float v = beta2 * v_buf[idx] + (1.f - beta2) * g * g;
    m_buf[idx] = m;
    v_buf[idx] = v;
    float mhat = m / (1.f - __powf(beta1, (float)t));
    float vhat = v / (1.f - __powf(beta2, (float)t));
    float p = master[idx] - lr * mhat / (__fsqrt_rn(vhat) + eps);
    master[idx] = p;
    if (idx < PW) {
        unsigned short pb = ga_f32_to_bf16(p);
        w_bf16[idx] = pb;
        int l = idx / (W * W);
        int rem = idx - l * W * W;
        int out = rem / W, in = rem - out * W;
        wt_bf16[(size_t)l * W * W + in * W + out] = pb;
    } else {
        bias[idx - PW] = p;
    }
}

// ===========================================================================
// Host launch functions (torch tensors in, kernel launches on the current
// torch stream so everything is hipGraph-capturable). Width comes in as the
// wmax argument and dispatches to the 64- or 128-wide instantiation.
// ===========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "launchers.h"

#define CHECK_GPU(t) TORCH_CHECK(t.is_cuda(), #t " must be on GPU")

static void ga_fill_acts(int* dst, const std::vector<int64_t>& acts, int L) {
    TORCH_CHECK((int)acts.size() == L, "act id list must have L entries");
    TORCH_CHECK(L >= 1 && L <= GA_MAXL, "fused path supports 1..", GA_MAXL,
                " layers (LDS-capped per width; see engine supported())");
    for (int i = 0; i < L; ++i) dst[i] = (int)acts[i];
}

#define HG_DISPATCH_W(WMAX, ...)                              \
    do {                                                      \
        if ((WMAX) == 64) {                                   \
            constexpr int W = 64;                             \
            __VA_ARGS__;                                      \
        } else if ((WMAX) == 128) {                           \
            constexpr int W = 128;                            \
            __VA_ARGS__;                                      \
        } else if ((WMAX) == 256) {                           \
            constexpr int W = 256;                            \
            __VA_ARGS__;                                      \
        } else {                                              \
            TORCH_CHECK(false, "unsupported width ", (WMAX)); \
        }                                                     \
    } while (0)

// dynamic-LDS opt-in above the 64 KiB default (gfx950 allows a full 160 KiB
// workgroup); tracked per kernel instantiation (width tier)
static inline int hg_wtier(int w) { return w == 64 ? 0 : (w == 128 ? 1 : 2); }

void hg_train_step_launch(torch::Tensor X, torch::Tensor Y,
                          torch::Tensor counters, torch::Tensor w, torch::Tensor wt,
                          torch::Tensor bias, torch::Tensor dw, torch::Tensor db,
                          torch::Tensor loss, int64_t B, int64_t F_out,
                          std::vector<int64_t> acts, double gscale, bool y_is_x,
                          c10::optional<torch::Tensor> dx_out, int64_t wmax,
                          int64_t kbits, int64_t seed, std::vector<int64_t> widths) {
    CHECK_GPU(X);
    TORCH_CHECK(counters.numel() >= 3, "counters must hold [batch, t, epoch]");
    int L = (int)(w.numel() / (wmax * wmax));
    HgTrainArgs a;
    a.X = (const unsigned short*)X.data_ptr();
    a.Y = (const unsigned short*)Y.data_ptr();
    a.counters = counters.data_ptr<int>();
    a.kbits = (int)kbits;
    a.seed = (unsigned int)seed;
    a.w = (const unsigned short*)w.data_ptr();
    a.wt = (const unsigned short*)wt.data_ptr();
    a.bias = bias.data_ptr<float>();
    TORCH_CHECK(dw.scalar_type() == torch::kBFloat16, "dw slabs must be bf16");
    TORCH_CHECK(db.scalar_type() == torch::kBFloat16, "db slabs must be bf16");
    a.dw = (unsigned short*)dw.data_ptr();
    a.db = (unsigned short*)db.data_ptr();
    a.loss = loss.data_ptr<float>();
    a.dx_out = dx_out.has_value() ? (unsigned short*)dx_out->data_ptr() : nullptr;
    a.B = (int)B;
    a.N = (int)(X.size(0));
    a.F_out = (int)F_out;
    a.L = L;
    a.y_is_x = y_is_x ? 1 : 0;
    ga_fill_acts(a.act_id, acts, L);
    if (widths.empty()) {
        // no masking info: treat every layer as full-width (only safe when
        // all hidden activations satisfy act(0) == 0)
        for (int i = 0; i <= GA_MAXL; ++i) a.width[i] = (int)wmax;
    } else {
        TORCH_CHECK((int)widths.size() == L + 1, "widths must have L+1 entries");
        for (int i = 0; i <= L; ++i) a.width[i] = (int)widths[i];
        for (int i = L + 1; i <= GA_MAXL; ++i) a.width[i] = (int)wmax;
    }
    a.gscale = (float)gscale;

    int n_tiles = (int)((B + HG_MT - 1) / HG_MT);
    TORCH_CHECK(dw.size(0) >= n_tiles, "dw slab count too small");
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        constexpr size_t TB = HG_MT * W * 2;
        size_t smem = (size_t)(L + 4) * TB + GA_MAXL * W * 4 + 256 * 4;
        TORCH_CHECK(smem <= 160 * 1024, "hourglass L=", L, " W=", W,
                    " exceeds the 160 KiB LDS budget");
        static size_t cfg[3] = {0, 0, 0};
        size_t& c = cfg[hg_wtier(W)];
        if (smem > 65536 && c < smem) {
            (void)hipFuncSetAttribute((const void*)hg_train_step<W>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      (int)smem);
            c = smem;
        }
        hipLaunchKernelGGL(hg_train_step<W>, dim3(n_tiles), dim3(256), smem,
                           stream.stream(), a);
    });
}

void hg_train_step_w1_launch(torch::Tensor X, torch::Tensor Y,
                             torch::Tensor counters, torch::Tensor w, torch::Tensor wt,
                             torch::Tensor bias, torch::Tensor dw, torch::Tensor db,
                             torch::Tensor loss, int64_t B, int64_t F_out,
                             std::vector<int64_t> acts, double gscale, bool y_is_x,
                             c10::optional<torch::Tensor> dx_out, int64_t wmax,
                             int64_t kbits, int64_t seed, std::vector<int64_t> widths) {
    CHECK_GPU(X);
    TORCH_CHECK(wmax == 64, "wave-local train step is the W=64 tier");
    TORCH_CHECK(counters.numel() >= 3, "counters must hold [batch, t, epoch]");
    int L = (int)(w.numel() / (wmax * wmax));
    HgTrainArgs a;
    a.X = (const unsigned short*)X.data_ptr();
    a.Y = (const unsigned short*)Y.data_ptr();
    a.counters = counters.data_ptr<int>();
    a.kbits = (int)kbits;
    a.seed = (unsigned int)seed;
    a.w = (const unsigned short*)w.data_ptr();
    a.wt = (const unsigned short*)wt.data_ptr();
    a.bias = bias.data_ptr<float>();
    TORCH_CHECK(dw.scalar_type() == torch::kBFloat16, "dw slabs must be bf16");
    TORCH_CHECK(db.scalar_type() == torch::kBFloat16, "db slabs must be bf16");
    a.dw = (unsigned short*)dw.data_ptr();
    a.db = (unsigned short*)db.data_ptr();
    a.loss = loss.data_ptr<float>();
    a.dx_out = dx_out.has_value() ? (unsigned short*)dx_out->data_ptr() : nullptr;
    a.B = (int)B;
    a.N = (int)(X.size(0));
    a.F_out = (int)F_out;
    a.L = L;
    a.y_is_x = y_is_x ? 1 : 0;
    ga_fill_acts(a.act_id, acts, L);
    if (widths.empty()) {
        for (int i = 0; i <= GA_MAXL; ++i) a.width[i] = (int)wmax;
    } else {
        TORCH_CHECK((int)widths.size() == L + 1, "widths must have L+1 entries");
        for (int i = 0; i <= L; ++i) a.width[i] = (int)widths[i];
        for (int i = L + 1; i <= GA_MAXL; ++i) a.width[i] = (int)wmax;
    }
    a.gscale = (float)gscale;

    int n_tiles = (int)((B + HG_MT1 - 1) / HG_MT1);
    TORCH_CHECK(dw.size(0) >= n_tiles, "dw slab count too small");
    auto stream = at::hip::getCurrentHIPStream();
    constexpr int W = 64;
    constexpr size_t TB = HG_MT1 * W * 2;
    size_t smem = (size_t)(a.L + 2) * TB + 2 * ((size_t)W * 32) + GA_MAXL * W * 4;
    TORCH_CHECK(smem <= 160 * 1024, "wave-local step LDS overflow at L=", a.L);
    hipLaunchKernelGGL(hg_train_step_w1<W>, dim3(n_tiles), dim3(64), smem,
                       stream.stream(), a);
}

void hg_adam_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                    torch::Tensor dw, torch::Tensor db, torch::Tensor w,
                    torch::Tensor wt, torch::Tensor bias, torch::Tensor counters,
                    int64_t n_slabs, int64_t L, double lr, double beta1, double beta2,
                    double eps, int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_adam<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                           (const unsigned short*)dw.data_ptr(), (const unsigned short*)db.data_ptr(),
                           (unsigned short*)w.data_ptr(), (unsigned short*)wt.data_ptr(),
                           bias.data_ptr<float>(), counters.data_ptr<int>(), (int)n_slabs,
                           (int)L, (float)lr, (float)beta1, (float)beta2, (float)eps);
    });
}

void hg_permute_fill_launch(torch::Tensor out, int64_t kbits, int64_t key) {
    int n = (int)out.numel();
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_permute_fill, dim3((n + 255) / 256), dim3(256), 0,
                       stream.stream(), out.data_ptr<int>(), n, (int)kbits,
                       (unsigned int)key);
}

void hg_bump_launch(torch::Tensor counters, int64_t n_batches) {
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_bump, dim3(1), dim3(1), 0, stream.stream(),
                       counters.data_ptr<int>(), (int)n_batches);
}

void hg_forward_launch(torch::Tensor X, torch::Tensor w, torch::Tensor bias,
                       torch::Tensor xscale, torch::Tensor xshift, torch::Tensor yhat,
                       c10::optional<torch::Tensor> err, c10::optional<torch::Tensor> total,
                       c10::optional<torch::Tensor> inv_err_scale,
                       int64_t F_in, int64_t F_out, std::vector<int64_t> acts,
                       int64_t wmax) {
    CHECK_GPU(X);
    int L = (int)(w.numel() / (wmax * wmax));
    HgFwdArgs a;
    a.X = X.data_ptr<float>();
    a.w = (const unsigned short*)w.data_ptr();
    a.bias = bias.data_ptr<float>();
    a.xscale = xscale.data_ptr<float>();
    a.xshift = xshift.data_ptr<float>();
    a.yhat = yhat.data_ptr<float>();
    a.err = err.has_value() ? err->data_ptr<float>() : nullptr;
    a.total = total.has_value() ? total->data_ptr<float>() : nullptr;
    a.inv_err_scale = inv_err_scale.has_value() ? inv_err_scale->data_ptr<float>() : nullptr;
    a.B = (int)X.size(0);
    a.F_in = (int)F_in;
    a.F_out = (int)F_out;
    a.L = L;
    ga_fill_acts(a.act_id, acts, L);

    int n_tiles = (a.B + 63) / 64;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        size_t smem = 3 * (size_t)(64 * W * 2) + 4 * 64 * 4;
        static size_t cfg[3] = {0, 0, 0};
        size_t& c = cfg[hg_wtier(W)];
        if (smem > 65536 && c < smem) {
            (void)hipFuncSetAttribute((const void*)hg_forward<W>,
                                      hipFuncAttributeMaxDynamicSharedMemorySize,
                                      (int)smem);
            c = smem;
        }
        hipLaunchKernelGGL(hg_forward<W>, dim3(n_tiles), dim3(256), smem,
                           stream.stream(), a);
    });
}

void hg_slab_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor partial,
                           c10::optional<torch::Tensor> counters, int64_t n_slabs,
                           int64_t G, int64_t L, int64_t n_batches, int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    TORCH_CHECK(dw.scalar_type() == torch::kBFloat16 && db.scalar_type() == torch::kBFloat16,
                "slab buffers must be bf16");
    dim3 grid((P / 8 + 255) / 256, (int)G);
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_slab_reduce<W>, grid, dim3(256), 0, stream.stream(),
                           (const unsigned short*)dw.data_ptr(),
                           (const unsigned short*)db.data_ptr(),
                           partial.data_ptr<float>(),
                           counters.has_value() ? counters->data_ptr<int>() : nullptr,
                           (int)n_slabs, (int)G, (int)L, (int)n_batches);
    });
}

void hg_adam_p_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                      torch::Tensor partial, torch::Tensor w, torch::Tensor wt,
                      torch::Tensor bias, torch::Tensor counters, int64_t G,
                      int64_t L, double lr, double beta1, double beta2, double eps,
                      int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_adam_p<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                           partial.data_ptr<float>(), (unsigned short*)w.data_ptr(),
                           (unsigned short*)wt.data_ptr(), bias.data_ptr<float>(),
                           counters.data_ptr<int>(), (int)G, (int)L, (float)lr,
                           (float)beta1, (float)beta2, (float)eps);
    });
}

void hg_reduce_adam_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor master,
                           torch::Tensor m, torch::Tensor v, torch::Tensor w,
                           torch::Tensor wt, torch::Tensor bias,
                           torch::Tensor counters_r, torch::Tensor counters_w,
                           int64_t n_slabs, int64_t L, int64_t n_batches,
                           double lr, double beta1, double beta2, double eps,
                           int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    TORCH_CHECK(dw.scalar_type() == torch::kBFloat16 && db.scalar_type() == torch::kBFloat16,
                "slab buffers must be bf16");
    TORCH_CHECK(counters_r.data_ptr() != counters_w.data_ptr(),
                "ping-pong counters must be distinct tensors");
    int blocks = (P + 127) / 128;  // 128 params per block
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_reduce_adam<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           (const unsigned short*)dw.data_ptr(),
                           (const unsigned short*)db.data_ptr(),
                           master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                           (unsigned short*)w.data_ptr(), (unsigned short*)wt.data_ptr(),
                           bias.data_ptr<float>(), counters_r.data_ptr<int>(),
                           counters_w.data_ptr<int>(), (int)n_slabs, (int)L,
                           (int)n_batches, (float)lr, (float)beta1, (float)beta2,
                           (float)eps);
    });
}

void hg_flat_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor g,
                           torch::Tensor counters_r, torch::Tensor counters_w,
                           int64_t n_slabs, int64_t L, int64_t n_batches,
                           int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    TORCH_CHECK(dw.scalar_type() == torch::kBFloat16 && db.scalar_type() == torch::kBFloat16,
                "slab buffers must be bf16");
    TORCH_CHECK(counters_r.data_ptr() != counters_w.data_ptr(),
                "ping-pong counters must be distinct tensors");
    int blocks = (P + 127) / 128;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_flat_reduce<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           (const unsigned short*)dw.data_ptr(),
                           (const unsigned short*)db.data_ptr(),
                           g.data_ptr<float>(), counters_r.data_ptr<int>(),
                           counters_w.data_ptr<int>(), (int)n_slabs, (int)L,
                           (int)n_batches);
    });
}

void hg_partial_flat_launch(torch::Tensor partial, torch::Tensor g, int64_t G,
                            int64_t L, int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_partial_flat<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           partial.data_ptr<float>(), g.data_ptr<float>(), (int)G, (int)L);
    });
}

void hg_grad_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor g,
                           int64_t n_slabs, int64_t L, int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_grad_reduce<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           (const unsigned short*)dw.data_ptr(),
                           (const unsigned short*)db.data_ptr(), g.data_ptr<float>(),
                           (int)n_slabs, (int)L);
    });
}

void hg_adam_flat_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                         torch::Tensor g, torch::Tensor w, torch::Tensor wt,
                         torch::Tensor bias, torch::Tensor counters, int64_t L,
                         double lr, double beta1, double beta2, double eps,
                         int64_t wmax) {
    int P = (int)(L * wmax * wmax + L * wmax);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    HG_DISPATCH_W(wmax, {
        hipLaunchKernelGGL(hg_adam_flat<W>, dim3(blocks), dim3(256), 0, stream.stream(),
                           master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                           g.data_ptr<float>(), (unsigned short*)w.data_ptr(),
                           (unsigned short*)wt.data_ptr(), bias.data_ptr<float>(),
                           counters.data_ptr<int>(), (int)L, (float)lr, (float)beta1,
                           (float)beta2, (float)eps);
    });
}
