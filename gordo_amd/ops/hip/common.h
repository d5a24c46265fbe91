// Common device helpers for gordo_amd CDNA4 (gfx950) kernels.
// Hand-written for MI355X: wave64, MFMA bf16 16x16x32, XOR-swizzled LDS tiles.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

#define GA_WAVE 64
#define GA_WMAX 64   // padded layer width (max supported by fused path)
#define GA_MT 64     // batch rows per block tile
#define GA_MAXL 8    // arg-array cap; the REAL per-width layer limit
                     // is LDS ((L+4) act/scratch tiles must fit 160 KiB
                     // -> L<=8 at W=64/128, L<=5 at W=256)

typedef __bf16 ga_bf16x8 __attribute__((ext_vector_type(8)));
typedef float ga_f32x4 __attribute__((ext_vector_type(4)));
typedef unsigned short ga_u16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float ga_bf16_to_f32(unsigned short u) {
    union { unsigned int i; float f; } c;
    c.i = ((unsigned int)u) << 16;
    return c.f;
}

// single v_cvt_pk_bf16_f32 (HW round-to-nearest-even) — the manual shift/add
// RTNE emulation costs ~6 VALU ops per convert and showed up as the top VALU
// consumer in the LSTM cell kernels (≈100 converts/thread/timestep).
__device__ __forceinline__ unsigned short ga_f32_to_bf16(float f) {
    __bf16 h = (__bf16)f;
    union { __bf16 h; unsigned short u; } c;
    c.h = h;
    return c.u;
}

// ---------------------------------------------------------------------------
// LDS tile addressing: [64][64] bf16 tiles, row stride 128 B, XOR-swizzled so
// that per-lane row-varying ds_read_b128 (MFMA fragment loads) spread over 8
// bank slots instead of hitting one (guide T2: byte ^= (row&7)<<4).
// Tile size: 64*64*2 = 8 KiB.
// ---------------------------------------------------------------------------
#define GA_TILE_BYTES (GA_WMAX * GA_WMAX * 2)

__device__ __forceinline__ unsigned int ga_sw(int row, int col) {
    // byte offset of element [row][col] in a swizzled [64][64] bf16 tile
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 7u) << 4);
    return ((unsigned int)row << 7) + byte;
}

// Load an 8-wide contiguous bf16 group [row][col0..col0+7] (col0 % 8 == 0)
__device__ __forceinline__ ga_bf16x8 ga_tile_read8(const char* tile, int row, int col0) {
    return *(const ga_bf16x8*)(tile + ga_sw(row, col0));
}

__device__ __forceinline__ void ga_tile_write8(char* tile, int row, int col0, ga_u16x8 v) {
    *(ga_u16x8*)(tile + ga_sw(row, col0)) = v;
}

__device__ __forceinline__ unsigned short ga_tile_read1(const char* tile, int row, int col) {
    return *(const unsigned short*)(tile + ga_sw(row, col));
}

__device__ __forceinline__ void ga_tile_write1(char* tile, int row, int col, unsigned short v) {
    *(unsigned short*)(tile + ga_sw(row, col)) = v;
}

// ---------------------------------------------------------------------------
// Narrow tiles for the MT=32 hourglass step: [R][32] bf16, 64-B rows, XOR
// swizzle over the 4 16-B slots per row.
// ---------------------------------------------------------------------------
__device__ __forceinline__ unsigned int ga_sw32(int row, int col) {
    unsigned int byte = (unsigned int)(col * 2) ^ (((unsigned int)row & 3u) << 4);
    return ((unsigned int)row << 6) + byte;
}

__device__ __forceinline__ ga_bf16x8 ga_t32_read8(const char* tile, int row, int col0) {
    return *(const ga_bf16x8*)(tile + ga_sw32(row, col0));
}

__device__ __forceinline__ void ga_t32_write1(char* tile, int row, int col, unsigned short v) {
    *(unsigned short*)(tile + ga_sw32(row, col)) = v;
}

// ---------------------------------------------------------------------------
// Swizzled LDS image for weight matrices read as MFMA B-fragments: 16-lane
// groups read 16 consecutive ROWS at one column offset; a plain row-major
// image puts all 16 on one bank (row stride is a multiple of 256 B). XORing
// bits 4..7 of the in-row byte offset with (row&15) permutes the 16-B slots
// so the 16 rows land on 16 distinct slots. rowbytes must be a multiple of
// 256 and the in-row offset a multiple of 16.
// ---------------------------------------------------------------------------
__device__ __forceinline__ unsigned int ga_swW(int row, int colbyte, int rowbytes) {
    return (unsigned int)row * (unsigned int)rowbytes +
           ((unsigned int)colbyte ^ (((unsigned int)row & 15u) << 4));
}

// ---------------------------------------------------------------------------
// Activations (ids shared with the Python engine)
// ---------------------------------------------------------------------------
#define GA_ACT_LINEAR 0
#define GA_ACT_TANH 1
#define GA_ACT_RELU 2
#define GA_ACT_SIGMOID 3

// v_rcp_f32 (1 ULP) — plain `1.f/x` and even __fdividef lower to the IEEE
// div_scale/div_fmas/div_fixup sequence (~8 instrs, serially dependent),
// which was the top instruction cluster in the LSTM cell epilogue.
__device__ __forceinline__ float ga_rcp(float x) { return __builtin_amdgcn_rcpf(x); }

// fast tanh via the hardware-exp path (libm tanhf is an ocml call, ~10x the
// cost; rel err ~1e-7, far below bf16 resolution). Clamp keeps exp finite.
__device__ __forceinline__ float ga_tanh(float x) {
    float xc = fminf(fmaxf(x, -15.f), 15.f);
    float e = __expf(2.f * xc);
    return (e - 1.f) * ga_rcp(e + 1.f);
}

__device__ __forceinline__ float ga_sigmoid(float x) {
    return ga_rcp(1.f + __expf(-x));
}

__device__ __forceinline__ float ga_act(float x, int id) {
    switch (id) {
        case GA_ACT_TANH: return ga_tanh(x);
        case GA_ACT_RELU: return x > 0.f ? x : 0.f;
        case GA_ACT_SIGMOID: return ga_sigmoid(x);
        default: return x;
    }
}

// derivative expressed in terms of the activation OUTPUT h
__device__ __forceinline__ float ga_act_grad_from_h(float h, int id) {
    switch (id) {
        case GA_ACT_TANH: return 1.f - h * h;
        case GA_ACT_RELU: return h > 0.f ? 1.f : 0.f;
        case GA_ACT_SIGMOID: return h * (1.f - h);
        default: return 1.f;
    }
}

// ---------------------------------------------------------------------------
// On-the-fly minibatch shuffle: a keyed bijection on [0, n) via a 4-round
// Feistel network over the next-pow2 domain with cycle walking. Replaces the
// host-side randperm + perm buffer (rocprim sorts were ~8% of the hourglass
// step and forced host work at epoch boundaries; with the epoch key coming
// from a device counter the whole training run is pure graph replay).
// kbits = even ceil(log2 n); kbits == 0 means identity.
// ---------------------------------------------------------------------------
__device__ __forceinline__ unsigned int ga_mix(unsigned int x, unsigned int key) {
    x ^= key;
    x *= 0x9E3779B9u;
    x ^= x >> 16;
    x *= 0x85EBCA6Bu;
    x ^= x >> 13;
    return x;
}

__device__ __forceinline__ unsigned int ga_permute(unsigned int i, unsigned int n,
                                                   int kbits, unsigned int key) {
    if (kbits == 0) return i;
    const unsigned int half = (unsigned int)kbits >> 1;
    const unsigned int hmask = (1u << half) - 1u;
    unsigned int x = i;
    do {
        unsigned int L = x >> half, R = x & hmask;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            unsigned int nl = R;
            R = (L ^ ga_mix(R, key + (unsigned int)r * 0x6A09E667u)) & hmask;
            L = nl;
        }
        x = (L << half) | R;
    } while (x >= n);
    return x;
}

// wave-wide f32 sum (64 lanes)
__device__ __forceinline__ float ga_wave_sum(float v) {
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    return v;
}

// ---------------------------------------------------------------------------
// u8 fixed-point gate streams (the LSTM BPTT state saves). The activated
// gates are bounded — i,f,o in (0,1), g in (-1,1) — so uniform 8-bit
// quantization (abs err <= 1/510, vs ~0.03 for fp8-e4m3 near 1.0) halves the
// two largest HBM streams of the LSTM step at better precision than fp8.
// Decode lowers to v_cvt_f32_ubyteN + one multiply. g is stored offset:
// (g+1)/2 * 255.
// ---------------------------------------------------------------------------
__device__ __forceinline__ unsigned int ga_q8_pack4(float v0, float v1, float v2, float v3) {
    unsigned int b0 = (unsigned int)__float2uint_rn(v0 * 255.f);
    unsigned int b1 = (unsigned int)__float2uint_rn(v1 * 255.f);
    unsigned int b2 = (unsigned int)__float2uint_rn(v2 * 255.f);
    unsigned int b3 = (unsigned int)__float2uint_rn(v3 * 255.f);
    return b0 | (b1 << 8) | (b2 << 16) | (b3 << 24);
}

__device__ __forceinline__ uint2 ga_q8_pack8(const float* v) {
    return uint2{ga_q8_pack4(v[0], v[1], v[2], v[3]),
                 ga_q8_pack4(v[4], v[5], v[6], v[7])};
}

// element e (0..7) of a packed 8-vector, back to [0,1]
__device__ __forceinline__ float ga_q8_at(uint2 p, int e) {
    unsigned int w = (e < 4) ? p.x : p.y;
    return (float)((w >> ((e & 3) * 8)) & 0xffu) * (1.f / 255.f);
}
