// Fused hourglass-MLP training & inference kernels for MI355X (gfx950).
//
// Replaces the reference's Keras fit/predict hot loop (SURVEY.md §2.3 K1-K3,
// K6-K8, K10) with three kernels per training step:
//   1. hg_train_step: minibatch gather + L-layer dense forward (MFMA bf16
//      16x16x32, activations staged in XOR-swizzled LDS tiles) + MSE loss
//      grad + full backward (dW/db partials per block slab in fp32)
//   2. hg_adam: slab reduction + fp32-master Adam + bf16 weight (and
//      transposed-weight) emission
//   3. hg_bump: advance the batch/t counters (device-side, so the whole step
//      is hipGraph-capturable with no host round trip)
// plus hg_forward for serving (fused min-max scale -> forward -> optional
// per-feature anomaly score + row L2 total).
//
// Geometry: one block = 64 batch rows x 4 waves; all layer widths padded to
// 64 (zero pads stay zero through tanh/linear layers and contribute zero
// gradient). Weights are read straight from L2 (the whole model is KB-scale);
// activations and the dpre scratch run through LDS.
#include "common.h"

#define GA_NT (GA_WMAX / 16)  // 4 column tiles of 16
#define GA_KK (GA_WMAX / 32)  // 2 K-steps of 32

struct HgTrainArgs {
    const unsigned short* __restrict__ X;   // [N][64] bf16, pre-scaled, padded
    const unsigned short* __restrict__ Y;   // may equal X (autoencoder)
    const int* __restrict__ perm;           // [n_batches*B] row indices
    const int* __restrict__ counters;       // [0]=batch idx, [1]=adam t
    const unsigned short* __restrict__ w;   // [L][64][64] bf16 (out,in)
    const unsigned short* __restrict__ wt;  // [L][64][64] bf16 (in,out)
    const float* __restrict__ bias;         // [L][64] fp32
    float* __restrict__ dw;                 // [n_tiles][L][64*64] fp32 slabs
    float* __restrict__ db;                 // [n_tiles][L][64] fp32 slabs
    float* __restrict__ loss;               // [1] fp32 accumulator (SSE)
    unsigned short* __restrict__ dx_out;    // [B][64] bf16 dL/dinput, or null
    int B;          // rows per step (per GPU)
    int N;          // dataset rows
    int F_out;      // real output features (cols >= F_out masked)
    int L;          // number of layers (<= GA_MAXL)
    int y_is_x;     // autoencoder fast path
    int act_id[GA_MAXL];  // activation per layer (last = out_func)
    float gscale;   // 2 / (B * F_out)
};

// MT=32 geometry: one block = 32 batch rows; B=8192 -> 256 blocks = every
// CU busy (the original 64-row tiling left half the chip idle at the
// headline batch). The 4 waves split the work 2-D: wid&1 = 16-row stripe,
// wid>>1 = 32-col stripe. LDS: acts[L] + 4 dpre scratch tiles, all 4 KiB
// ([32][64] via ga_sw, [64][32] transposed via ga_sw32) + bias ~ 38 KiB.
#define HG_MT 32
#define HG_TILE_BYTES (HG_MT * GA_WMAX * 2)  // 4 KiB

extern __shared__ __attribute__((aligned(16))) char ga_smem[];

__device__ __forceinline__ char* act_tile(int i) { return ga_smem + i * HG_TILE_BYTES; }

__launch_bounds__(256)
__global__ void hg_train_step(HgTrainArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int tile = blockIdx.x;
    const int base = tile * HG_MT;
    const int wrow = (wid & 1) * 16;   // this wave's 16-row stripe
    const int wcol = (wid >> 1) * 32;  // this wave's 32-col stripe

    char* scratch0 = ga_smem + a.L * HG_TILE_BYTES;        // dcur  [32][64]
    char* scratch1 = scratch0 + HG_TILE_BYTES;             // dcurT [64][32]
    char* scratch2 = scratch1 + HG_TILE_BYTES;             // dnext
    char* scratch3 = scratch2 + HG_TILE_BYTES;             // dnextT
    float* s_bias = (float*)(scratch3 + HG_TILE_BYTES);    // [L][64]
    float* s_red = s_bias + GA_MAXL * GA_WMAX;             // [4][64] reduce pad

    const int step = a.counters[0];
    const long long pbase = (long long)step * a.B + base;

    // ---- stage biases -------------------------------------------------
    for (int i = tid; i < a.L * GA_WMAX; i += 256) s_bias[i] = a.bias[i];

    // ---- gather minibatch tile into acts[0]: one 16-B granule/thread ---
    {
        int m = tid >> 3;
        int col0 = (tid & 7) * 8;
        int gi = base + m;
        ga_u16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
        if (gi < a.B) {
            int row = a.perm[pbase + m];
            v = *(const ga_u16x8*)(a.X + (long long)row * GA_WMAX + col0);
        }
        ga_tile_write8(act_tile(0), m, col0, v);
    }
    __syncthreads();

    // =====================================================================
    // Forward: per wave [16 rows][32 cols] = 2 MFMA tiles x 2 K-steps.
    // =====================================================================
    for (int li = 0; li < a.L; ++li) {
        const unsigned short* wl = a.w + (size_t)li * GA_WMAX * GA_WMAX;
        ga_f32x4 acc[2];
        acc[0] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        acc[1] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < GA_WMAX; kk += 32) {
            ga_bf16x8 afrag = ga_tile_read8(act_tile(li), wrow + (lane & 15), kk + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(wl + (size_t)(wcol + nt * 16 + (lane & 15)) * GA_WMAX + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }

        if (li + 1 < a.L) {
#pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                int col = wcol + nt * 16 + (lane & 15);
                float b = s_bias[li * GA_WMAX + col];
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    float h = ga_act(acc[nt][r] + b, a.act_id[li]);
                    ga_tile_write1(act_tile(li + 1), m, col, ga_f32_to_bf16(h));
                }
            }
            __syncthreads();
        } else {
            // ---- loss epilogue: dpre_{L-1} = (yhat - y) * gscale --------
            float lsum = 0.f;
#pragma unroll
            for (int nt = 0; nt < 2; ++nt) {
                int col = wcol + nt * 16 + (lane & 15);
                float b = s_bias[li * GA_WMAX + col];
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int m = wrow + (lane >> 4) * 4 + r;
                    int gi = base + m;
                    float yhat = ga_act(acc[nt][r] + b, a.act_id[li]);
                    float yval = 0.f;
                    bool valid = (gi < a.B) && (col < a.F_out);
                    if (valid) {
                        if (a.y_is_x) {
                            yval = ga_bf16_to_f32(ga_tile_read1(act_tile(0), m, col));
                        } else {
                            int row = a.perm[pbase + m];
                            yval = ga_bf16_to_f32(a.Y[(long long)row * GA_WMAX + col]);
                        }
                    }
                    float diff = valid ? (yhat - yval) : 0.f;
                    lsum += diff * diff;
                    float dpre = diff * a.gscale * ga_act_grad_from_h(yhat, a.act_id[li]);
                    unsigned short dp = ga_f32_to_bf16(dpre);
                    ga_tile_write1(scratch0, m, col, dp);
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
    // Backward: per li:  dX -> dpre_{li-1} (waves split [16r][32c]);
    //   dW[64 in][64 out] (wave = 16-in stripe, K = 32 rows, 1 K-step);
    //   db via dcurT row sums.
    // =====================================================================
    char* dcur = scratch0;
    char* dcurT = scratch1;
    char* dnext = scratch2;
    char* dnextT = scratch3;

    for (int li = a.L - 1; li >= 0; --li) {
        if (li > 0 || a.dx_out != nullptr) {
            const unsigned short* wtl = a.wt + (size_t)li * GA_WMAX * GA_WMAX;
            ga_f32x4 acc[2];
            acc[0] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            acc[1] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            for (int kk = 0; kk < GA_WMAX; kk += 32) {
                ga_bf16x8 afrag = ga_tile_read8(dcur, wrow + (lane & 15), kk + (lane >> 4) * 8);
#pragma unroll
                for (int nt = 0; nt < 2; ++nt) {
                    // B[k=out][j=in] = WT[in][out]: row-read of WT
                    ga_bf16x8 bfrag = *(const ga_bf16x8*)(wtl + (size_t)(wcol + nt * 16 + (lane & 15)) * GA_WMAX + kk + (lane >> 4) * 8);
                    acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
                }
            }
            if (li > 0) {
#pragma unroll
                for (int nt = 0; nt < 2; ++nt) {
                    int col = wcol + nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = wrow + (lane >> 4) * 4 + r;
                        float h = ga_bf16_to_f32(ga_tile_read1(act_tile(li), m, col));
                        float d = acc[nt][r] * ga_act_grad_from_h(h, a.act_id[li - 1]);
                        unsigned short dp = ga_f32_to_bf16(d);
                        ga_tile_write1(dnext, m, col, dp);
                        ga_t32_write1(dnextT, col, m, dp);
                    }
                }
            } else {
                // dL/d(input) for the caller (e.g. the LSTM head): batch order
#pragma unroll
                for (int nt = 0; nt < 2; ++nt) {
                    int col = wcol + nt * 16 + (lane & 15);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int m = wrow + (lane >> 4) * 4 + r;
                        int gi = base + m;
                        if (gi < a.B)
                            a.dx_out[(size_t)gi * GA_WMAX + col] = ga_f32_to_bf16(acc[nt][r]);
                    }
                }
            }
        }

        // ---- dW: wave = 16-in stripe x 64 out, K = m (32, one K-step) ----
        {
            ga_f32x4 acc[GA_NT];
            for (int nt = 0; nt < GA_NT; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
            // A[i=in][k=m] = acts[li][m][in]: strided scalar reads
            ga_bf16x8 afrag;
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                int m = (lane >> 4) * 8 + e;
                unsigned short u = ga_tile_read1(act_tile(li), m, wid * 16 + (lane & 15));
                afrag[e] = *(__bf16*)&u;
            }
#pragma unroll
            for (int nt = 0; nt < GA_NT; ++nt) {
                // B[k=m][j=out] = dcurT[out][m]: row-read of the narrow tile
                ga_bf16x8 bfrag = ga_t32_read8(dcurT, nt * 16 + (lane & 15), (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
            float* dwl = a.dw + ((size_t)tile * a.L + li) * GA_WMAX * GA_WMAX;
#pragma unroll
            for (int nt = 0; nt < GA_NT; ++nt) {
                int out = nt * 16 + (lane & 15);
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    int in = wid * 16 + (lane >> 4) * 4 + r;
                    dwl[(size_t)in * GA_WMAX + out] = acc[nt][r];
                }
            }
        }

        // ---- db: column sums of dpre via dcurT row sums ----------------
        {
            int out = tid >> 2;          // 64 outs x 4 groups of 8 m's
            int g = tid & 3;
            ga_bf16x8 v = ga_t32_read8(dcurT, out, g * 8);
            float sum = 0.f;
#pragma unroll
            for (int e = 0; e < 8; ++e) sum += (float)v[e];
            s_red[g * GA_WMAX + out] = sum;
            __syncthreads();
            if (tid < GA_WMAX) {
                float tot = s_red[tid] + s_red[GA_WMAX + tid] + s_red[2 * GA_WMAX + tid] +
                            s_red[3 * GA_WMAX + tid];
                a.db[((size_t)tile * a.L + li) * GA_WMAX + tid] = tot;
            }
        }
        __syncthreads();

        // rotate scratch: dpre_{li-1} becomes current
        char* t0 = dcur; dcur = dnext; dnext = t0;
        char* t1 = dcurT; dcurT = dnextT; dnextT = t1;
    }
}

// ===========================================================================
// Adam: reduce dW/db slabs, update fp32 master, emit bf16 W and WT.
// Param space: [L*64*64 weights][L*64 biases]. One thread per parameter.
// ===========================================================================
__launch_bounds__(256)
__global__ void hg_adam(
    float* __restrict__ master,      // [P] fp32 (weights then biases)
    float* __restrict__ m_buf,
    float* __restrict__ v_buf,
    const float* __restrict__ dw,    // [n_slabs][L][64*64]
    const float* __restrict__ db,    // [n_slabs][L][64]
    unsigned short* __restrict__ w_bf16,   // [L][64][64]
    unsigned short* __restrict__ wt_bf16,  // [L][64][64]
    float* __restrict__ bias,        // [L][64] (fp32 view used by fwd)
    const int* __restrict__ counters,
    int n_slabs, int L, float lr, float beta1, float beta2, float eps) {
    const int PW = L * GA_WMAX * GA_WMAX;
    const int P = PW + L * GA_WMAX;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= P) return;

    float g = 0.f;
    if (idx < PW) {
        // master layout is [l][out][in]; the train kernel's slabs are
        // [l][in][out] (the dW MFMA's C layout) -> swap in/out on read
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        size_t sidx = (size_t)l * GA_WMAX * GA_WMAX + (size_t)in * GA_WMAX + out;
        for (int s = 0; s < n_slabs; ++s) g += dw[(size_t)s * PW + sidx];
    } else {
        int bidx = idx - PW;
        for (int s = 0; s < n_slabs; ++s) g += db[(size_t)s * (L * GA_WMAX) + bidx];
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
        // transposed copy: layer-local [out][in] -> [in][out]
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        wt_bf16[(size_t)l * GA_WMAX * GA_WMAX + in * GA_WMAX + out] = pb;
    } else {
        bias[idx - PW] = p;
    }
}

__global__ void hg_bump(int* counters, int n_batches) {
    counters[0] = (counters[0] + 1) % n_batches;
    counters[1] += 1;
}

// ===========================================================================
// Two-stage slab reduction (the single-stage hg_adam loop over 128 slabs with
// only P/256 blocks was 51% of the hourglass step): stage A fans the slab sum
// over G groups (G x P/256 blocks, ~7x more parallelism), folding the counter
// bump in (block (0,0) — no other block touches counters, and every consumer
// of counters runs later in the stream); stage B is Adam over the G partials.
// Layout conversion slab[l][in][out] -> master[l][out][in] happens in stage A.
// ===========================================================================
__launch_bounds__(256)
__global__ void hg_slab_reduce(const float* __restrict__ dw,
                               const float* __restrict__ db,
                               float* __restrict__ partial,  // [G][P] SLAB layout
                               int* __restrict__ counters,
                               int n_slabs, int G, int L, int n_batches) {
    // stays in slab layout ([l][in][out]) so every access is a contiguous
    // float4 stream; the [in][out]->[out][in] transpose happens in the tiny
    // stage-B Adam instead (1.3 MB vs 21 MB here).
    const int PW = L * GA_WMAX * GA_WMAX;
    const int P = PW + L * GA_WMAX;
    int g = blockIdx.y;
    int idx4 = blockIdx.x * blockDim.x + threadIdx.x;
    if (g == 0 && idx4 == 0 && counters != nullptr) {
        counters[0] = (counters[0] + 1) % n_batches;
        counters[1] += 1;
    }
    int idx = idx4 * 4;
    if (idx >= P) return;
    int per = (n_slabs + G - 1) / G;
    int s0 = g * per, s1 = min(n_slabs, s0 + per);
    float4 acc = make_float4(0.f, 0.f, 0.f, 0.f);
    if (idx < PW) {
        for (int sl = s0; sl < s1; ++sl) {
            float4 v = *(const float4*)(dw + (size_t)sl * PW + idx);
            acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
        }
    } else {
        int bidx = idx - PW;
        for (int sl = s0; sl < s1; ++sl) {
            float4 v = *(const float4*)(db + (size_t)sl * (L * GA_WMAX) + bidx);
            acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
        }
    }
    *(float4*)(partial + (size_t)g * P + idx) = acc;
}

__launch_bounds__(256)
__global__ void hg_adam_p(
    float* __restrict__ master, float* __restrict__ m_buf, float* __restrict__ v_buf,
    const float* __restrict__ partial, unsigned short* __restrict__ w_bf16,
    unsigned short* __restrict__ wt_bf16, float* __restrict__ bias,
    const int* __restrict__ counters, int G, int L,
    float lr, float beta1, float beta2, float eps) {
    const int PW = L * GA_WMAX * GA_WMAX;
    const int P = PW + L * GA_WMAX;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;  // MASTER-layout index
    if (idx >= P) return;
    // partial is in SLAB layout: transpose on read
    size_t sidx;
    if (idx < PW) {
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        sidx = (size_t)l * GA_WMAX * GA_WMAX + (size_t)in * GA_WMAX + out;
    } else {
        sidx = (size_t)idx;
    }
    float g = 0.f;
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
    if (idx < PW) {
        unsigned short pb = ga_f32_to_bf16(p);
        w_bf16[idx] = pb;
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        wt_bf16[(size_t)l * GA_WMAX * GA_WMAX + in * GA_WMAX + out] = pb;
    } else {
        bias[idx - PW] = p;
    }
}

// partial [G][P] -> flat grad bucket [P] (DP all-reduce input)
__launch_bounds__(256)
__global__ void hg_partial_flat(const float* __restrict__ partial,
                                float* __restrict__ g_out, int G, int L) {
    // partial is SLAB layout; emit MASTER layout for the DP bucket
    const int PW = L * GA_WMAX * GA_WMAX;
    const int P = PW + L * GA_WMAX;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= P) return;
    size_t sidx;
    if (idx < PW) {
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        sidx = (size_t)l * GA_WMAX * GA_WMAX + (size_t)in * GA_WMAX + out;
    } else {
        sidx = (size_t)idx;
    }
    float sum = 0.f;
    for (int i = 0; i < G; ++i) sum += partial[(size_t)i * P + sidx];
    g_out[idx] = sum;
}

// ===========================================================================
// Serving forward (fused K1+K7+K8): raw fp32 input -> min-max scale -> L-layer
// forward -> yhat (scaled space); optionally per-feature scaled |err| and row
// L2 total (autoencoder anomaly: y = scaled x). hipGraph-capturable.
// ===========================================================================
struct HgFwdArgs {
    const float* __restrict__ X;   // [B][F_in] fp32 raw (row stride F_in)
    const unsigned short* __restrict__ w;
    const float* __restrict__ bias;
    const float* __restrict__ xscale;   // [64] (pad cols 0)
    const float* __restrict__ xshift;   // [64]
    float* __restrict__ yhat;           // [B][F_out]
    float* __restrict__ err;            // [B][F_out] or null
    float* __restrict__ total;          // [B] or null
    const float* __restrict__ inv_err_scale;  // [64] detector 1/scale (anomaly)
    int B, F_in, F_out, L;
    int act_id[GA_MAXL];
};

__launch_bounds__(256)
__global__ void hg_forward(HgFwdArgs a) {
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int base = blockIdx.x * GA_MT;
    char* t_x = ga_smem;                        // scaled input (preserved for anomaly y)
    char* t0 = ga_smem + GA_TILE_BYTES;         // ping
    char* t1 = ga_smem + 2 * GA_TILE_BYTES;     // pong
    float* s_red = (float*)(ga_smem + 3 * GA_TILE_BYTES);  // [4][64]

    // stage + scale input tile (scalar loads; F_in is unpadded)
    for (int c = tid; c < GA_MT * GA_WMAX; c += 256) {
        int m = c >> 6, col = c & 63;
        int gi = base + m;
        float x = 0.f;
        if (gi < a.B && col < a.F_in) x = a.X[(long long)gi * a.F_in + col];
        float xs = x * a.xscale[col] + a.xshift[col];
        ga_tile_write1(t_x, m, col, ga_f32_to_bf16(xs));
    }
    __syncthreads();

    const int wrow = wid * 16;
    for (int li = 0; li < a.L; ++li) {
        const unsigned short* wl = a.w + (size_t)li * GA_WMAX * GA_WMAX;
        const char* src = (li == 0) ? t_x : ((li & 1) ? t0 : t1);
        ga_f32x4 acc[GA_NT];
        for (int nt = 0; nt < GA_NT; ++nt) acc[nt] = ga_f32x4{0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < GA_WMAX; kk += 32) {
            ga_bf16x8 afrag = ga_tile_read8(src, wrow + (lane & 15), kk + (lane >> 4) * 8);
#pragma unroll
            for (int nt = 0; nt < GA_NT; ++nt) {
                ga_bf16x8 bfrag = *(const ga_bf16x8*)(wl + (size_t)(nt * 16 + (lane & 15)) * GA_WMAX + kk + (lane >> 4) * 8);
                acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[nt], 0, 0, 0);
            }
        }
        char* dst = (li & 1) ? t1 : t0;
        bool last = (li + 1 == a.L);
#pragma unroll
        for (int nt = 0; nt < GA_NT; ++nt) {
            int col = nt * 16 + (lane & 15);
            float b = a.bias[li * GA_WMAX + col];
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int m = wrow + (lane >> 4) * 4 + r;
                float h = ga_act(acc[nt][r] + b, a.act_id[li]);
                int gi = base + m;
                if (last) {
                    if (gi < a.B && col < a.F_out)
                        a.yhat[(long long)gi * a.F_out + col] = h;
                    ga_tile_write1(dst, m, col, ga_f32_to_bf16(h));
                } else {
                    ga_tile_write1(dst, m, col, ga_f32_to_bf16(h));
                }
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
        for (int ci = 0; ci < 16; ++ci) {
            int col = g * 16 + ci;
            int gi = base + m;
            if (gi < a.B && col < a.F_out) {
                float yhat = ga_bf16_to_f32(ga_tile_read1(yh, m, col));
                float yval = ga_bf16_to_f32(ga_tile_read1(t_x, m, col));
                float e = fabsf(yhat - yval) * a.inv_err_scale[col];
                a.err[(long long)gi * a.F_out + col] = e;
                s += e * e;
            }
        }
        s_red[g * GA_WMAX + m] = s;
        __syncthreads();
        if (tid < GA_MT && base + tid < a.B) {
            float tot = s_red[tid] + s_red[GA_WMAX + tid] + s_red[2 * GA_WMAX + tid] +
                        s_red[3 * GA_WMAX + tid];
            a.total[base + tid] = __fsqrt_rn(tot);
        }
    }
}


// ===========================================================================
// DP split: reduce slabs to one flat grad (the RCCL all-reduce bucket), then
// Adam from the reduced flat grad. Single-GPU uses the fused hg_adam instead.
// ===========================================================================
__launch_bounds__(256)
__global__ void hg_grad_reduce(const float* __restrict__ dw,
                               const float* __restrict__ db,
                               float* __restrict__ g, int n_slabs, int L) {
    const int PW = L * GA_WMAX * GA_WMAX;
    const int P = PW + L * GA_WMAX;
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= P) return;
    float s = 0.f;
    if (idx < PW) {
        // emit the flat bucket in MASTER layout [l][out][in] (slabs are
        // [l][in][out]) so hg_adam_flat can consume g[idx] directly
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        size_t sidx = (size_t)l * GA_WMAX * GA_WMAX + (size_t)in * GA_WMAX + out;
        for (int sl = 0; sl < n_slabs; ++sl) s += dw[(size_t)sl * PW + sidx];
    } else {
        int bidx = idx - PW;
        for (int sl = 0; sl < n_slabs; ++sl) s += db[(size_t)sl * (L * GA_WMAX) + bidx];
    }
    g[idx] = s;
}

__launch_bounds__(256)
__global__ void hg_adam_flat(
    float* __restrict__ master, float* __restrict__ m_buf, float* __restrict__ v_buf,
    const float* __restrict__ g_in, unsigned short* __restrict__ w_bf16,
    unsigned short* __restrict__ wt_bf16, float* __restrict__ bias,
    const int* __restrict__ counters, int L,
    float lr, float beta1, float beta2, float eps) {
    const int PW = L * GA_WMAX * GA_WMAX;
    const int P = PW + L * GA_WMAX;
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
        int l = idx / (GA_WMAX * GA_WMAX);
        int rem = idx - l * GA_WMAX * GA_WMAX;
        int out = rem / GA_WMAX, in = rem - out * GA_WMAX;
        wt_bf16[(size_t)l * GA_WMAX * GA_WMAX + in * GA_WMAX + out] = pb;
    } else {
        bias[idx - PW] = p;
    }
}

// ===========================================================================
// Host launch functions (torch tensors in, kernel launches on the current
// torch stream so everything is hipGraph-capturable).
// ===========================================================================
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "launchers.h"

#define CHECK_GPU(t) TORCH_CHECK(t.is_cuda(), #t " must be on GPU")

static void ga_fill_acts(int* dst, const std::vector<int64_t>& acts, int L) {
    TORCH_CHECK((int)acts.size() == L, "act id list must have L entries");
    TORCH_CHECK(L >= 1 && L <= GA_MAXL, "fused path supports 1..5 layers");
    for (int i = 0; i < L; ++i) dst[i] = (int)acts[i];
}

void hg_train_step_launch(torch::Tensor X, torch::Tensor Y, torch::Tensor perm,
                          torch::Tensor counters, torch::Tensor w, torch::Tensor wt,
                          torch::Tensor bias, torch::Tensor dw, torch::Tensor db,
                          torch::Tensor loss, int64_t B, int64_t F_out,
                          std::vector<int64_t> acts, double gscale, bool y_is_x,
                          c10::optional<torch::Tensor> dx_out) {
    CHECK_GPU(X);
    int L = (int)(w.numel() / (GA_WMAX * GA_WMAX));
    HgTrainArgs a;
    a.X = (const unsigned short*)X.data_ptr();
    a.Y = (const unsigned short*)Y.data_ptr();
    a.perm = perm.data_ptr<int>();
    a.counters = counters.data_ptr<int>();
    a.w = (const unsigned short*)w.data_ptr();
    a.wt = (const unsigned short*)wt.data_ptr();
    a.bias = bias.data_ptr<float>();
    a.dw = dw.data_ptr<float>();
    a.db = db.data_ptr<float>();
    a.loss = loss.data_ptr<float>();
    a.dx_out = dx_out.has_value() ? (unsigned short*)dx_out->data_ptr() : nullptr;
    a.B = (int)B;
    a.N = (int)(X.size(0));
    a.F_out = (int)F_out;
    a.L = L;
    a.y_is_x = y_is_x ? 1 : 0;
    ga_fill_acts(a.act_id, acts, L);
    a.gscale = (float)gscale;

    int n_tiles = (int)((B + HG_MT - 1) / HG_MT);
    TORCH_CHECK(dw.size(0) >= n_tiles, "dw slab count too small");
    size_t smem = (size_t)(L + 4) * HG_TILE_BYTES + GA_MAXL * GA_WMAX * 4 + 4 * GA_WMAX * 4;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_train_step, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void hg_adam_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                    torch::Tensor dw, torch::Tensor db, torch::Tensor w,
                    torch::Tensor wt, torch::Tensor bias, torch::Tensor counters,
                    int64_t n_slabs, int64_t L, double lr, double beta1, double beta2,
                    double eps) {
    int P = (int)(L * GA_WMAX * GA_WMAX + L * GA_WMAX);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_adam, dim3(blocks), dim3(256), 0, stream.stream(),
                       master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                       dw.data_ptr<float>(), db.data_ptr<float>(),
                       (unsigned short*)w.data_ptr(), (unsigned short*)wt.data_ptr(),
                       bias.data_ptr<float>(), counters.data_ptr<int>(), (int)n_slabs,
                       (int)L, (float)lr, (float)beta1, (float)beta2, (float)eps);
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
                       int64_t F_in, int64_t F_out, std::vector<int64_t> acts) {
    CHECK_GPU(X);
    int L = (int)(w.numel() / (GA_WMAX * GA_WMAX));
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

    int n_tiles = (a.B + GA_MT - 1) / GA_MT;
    size_t smem = 3 * GA_TILE_BYTES + 4 * GA_WMAX * 4;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_forward, dim3(n_tiles), dim3(256), smem, stream.stream(), a);
}

void hg_slab_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor partial,
                           c10::optional<torch::Tensor> counters, int64_t n_slabs,
                           int64_t G, int64_t L, int64_t n_batches) {
    int P = (int)(L * GA_WMAX * GA_WMAX + L * GA_WMAX);
    dim3 grid((P / 4 + 255) / 256, (int)G);
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_slab_reduce, grid, dim3(256), 0, stream.stream(),
                       dw.data_ptr<float>(), db.data_ptr<float>(),
                       partial.data_ptr<float>(),
                       counters.has_value() ? counters->data_ptr<int>() : nullptr,
                       (int)n_slabs, (int)G, (int)L, (int)n_batches);
}

void hg_adam_p_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                      torch::Tensor partial, torch::Tensor w, torch::Tensor wt,
                      torch::Tensor bias, torch::Tensor counters, int64_t G,
                      int64_t L, double lr, double beta1, double beta2, double eps) {
    int P = (int)(L * GA_WMAX * GA_WMAX + L * GA_WMAX);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_adam_p, dim3(blocks), dim3(256), 0, stream.stream(),
                       master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                       partial.data_ptr<float>(), (unsigned short*)w.data_ptr(),
                       (unsigned short*)wt.data_ptr(), bias.data_ptr<float>(),
                       counters.data_ptr<int>(), (int)G, (int)L, (float)lr,
                       (float)beta1, (float)beta2, (float)eps);
}

void hg_partial_flat_launch(torch::Tensor partial, torch::Tensor g, int64_t G, int64_t L) {
    int P = (int)(L * GA_WMAX * GA_WMAX + L * GA_WMAX);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_partial_flat, dim3(blocks), dim3(256), 0, stream.stream(),
                       partial.data_ptr<float>(), g.data_ptr<float>(), (int)G, (int)L);
}

void hg_grad_reduce_launch(torch::Tensor dw, torch::Tensor db, torch::Tensor g,
                           int64_t n_slabs, int64_t L) {
    int P = (int)(L * GA_WMAX * GA_WMAX + L * GA_WMAX);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_grad_reduce, dim3(blocks), dim3(256), 0, stream.stream(),
                       dw.data_ptr<float>(), db.data_ptr<float>(), g.data_ptr<float>(),
                       (int)n_slabs, (int)L);
}

void hg_adam_flat_launch(torch::Tensor master, torch::Tensor m, torch::Tensor v,
                         torch::Tensor g, torch::Tensor w, torch::Tensor wt,
                         torch::Tensor bias, torch::Tensor counters, int64_t L,
                         double lr, double beta1, double beta2, double eps) {
    int P = (int)(L * GA_WMAX * GA_WMAX + L * GA_WMAX);
    int blocks = (P + 255) / 256;
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(hg_adam_flat, dim3(blocks), dim3(256), 0, stream.stream(),
                       master.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                       g.data_ptr<float>(), (unsigned short*)w.data_ptr(),
                       (unsigned short*)wt.data_ptr(), bias.data_ptr<float>(),
                       counters.data_ptr<int>(), (int)L, (float)lr, (float)beta1,
                       (float)beta2, (float)eps);
}
