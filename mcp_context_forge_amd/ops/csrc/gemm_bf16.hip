// bf16 MFMA GEMM (C = A @ B^T) with fused bias + activation epilogue.
//
// This runs the gateway's classifier matmuls (content_moderation /
// harmful_content MLP) and the semantic-cache cosine-similarity sweep
// (response_cache_by_prompt) per BASELINE.json — the MFMA consumers of the
// plugin chain. Reference gateway has no GPU analog; the CPU oracle is the
// fp32 torch forward in models/classifier.py.
//
// Structure (per the CDNA4 guide's verified ladder, m97-class):
//   128x128 tile, BK=64, 256 threads = 4 waves (2x2), 64x64 output per wave,
//   v_mfma_f32_16x16x32_bf16 with f32x4 accumulators,
//   global_load_lds dwordx4 staging (16 B/lane, LDS dest linear),
//   XOR-swizzled LDS reads (T2: physical = logical ^ ((row&7)<<4)) with the
//   inverse swizzle applied to the *global source* address (guide rule #21:
//   linear dest + inverse-swizzled source + swizzled read).
//
// Preconditions (host wrapper pads): M%128==0, N%128==0, K%64==0,
// A row-major [M,K] bf16, BT row-major [N,K] bf16, C row-major [M,N].

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define BM 128
#define BN 128
#define BK 64           // bf16 elems; 128 bytes per row
#define ROW_BYTES (BK * 2)

// LDS swizzle: spread 8 consecutive rows across 8 distinct 16B slots so the
// 16 lanes of a quarter-wave reading a column-slice hit different banks.
__device__ __forceinline__ uint32_t swz(uint32_t byte_off) {
    return byte_off ^ (((byte_off >> 7) & 7u) << 4);
}

// ACT: 0 = none, 1 = gelu(tanh), 2 = sigmoid.  OUT_BF16: emit bf16 instead of f32.
template <int ACT, bool OUT_BF16, bool HAS_BIAS>
__global__ __launch_bounds__(256, 2) void gemm_bt_kernel(
    const short* __restrict__ A,   // [M,K] bf16 bits
    const short* __restrict__ BT,  // [N,K] bf16 bits
    const float* __restrict__ bias,  // [N] or null
    void* __restrict__ C,          // [M,N] f32 or bf16
    int M, int N, int K)
{
    __shared__ short As[BM * BK];
    __shared__ short Bs[BN * BK];

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wave = tid >> 6;           // 0..3
    const int wm = wave >> 1;            // 0..1 — wave row
    const int wn = wave & 1;             // 0..1 — wave col
    const int tile_m = blockIdx.x * BM;
    const int tile_n = blockIdx.y * BN;

    f32x4 acc[4][4];
    #pragma unroll
    for (int m = 0; m < 4; ++m)
        #pragma unroll
        for (int n = 0; n < 4; ++n)
            acc[m][n] = (f32x4){0.f, 0.f, 0.f, 0.f};

    const int n_tiles = K / BK;
    for (int kt = 0; kt < n_tiles; ++kt) {
        const int k0 = kt * BK;
        // ---- stage A and B tiles: 4 global_load_lds x 16B per thread per tile ----
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
            // physical LDS byte offset this (thread, issue) writes
            uint32_t P = (uint32_t)(i * 256 + tid) * 16u;
            uint32_t L = swz(P);                         // logical offset whose data lands at P
            uint32_t row = L >> 7, colb = L & 127u;
            const short* gA = A + ((size_t)(tile_m + row) * K + k0) + (colb >> 1);
            const short* gB = BT + ((size_t)(tile_n + row) * K + k0) + (colb >> 1);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)gA,
                (__attribute__((address_space(3))) uint32_t*)((char*)As + (i * 256 + (wave * 64)) * 16),
                16, /*offset*/ 0, /*aux*/ 0);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)gB,
                (__attribute__((address_space(3))) uint32_t*)((char*)Bs + (i * 256 + (wave * 64)) * 16),
                16, 0, 0);
        }
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __syncthreads();

        // ---- compute: 2 K-steps x 4x4 fragments ----
        #pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8 a_frag[4], b_frag[4];
            #pragma unroll
            for (int m = 0; m < 4; ++m) {
                uint32_t row = wm * 64 + m * 16 + (lane & 15);
                uint32_t colb = (kk * 32 + (lane >> 4) * 8) * 2;
                uint32_t off = swz(row * ROW_BYTES + colb);
                a_frag[m] = *(const bf16x8*)((const char*)As + off);
            }
            #pragma unroll
            for (int n = 0; n < 4; ++n) {
                uint32_t row = wn * 64 + n * 16 + (lane & 15);
                uint32_t colb = (kk * 32 + (lane >> 4) * 8) * 2;
                uint32_t off = swz(row * ROW_BYTES + colb);
                b_frag[n] = *(const bf16x8*)((const char*)Bs + off);
            }
            #pragma unroll
            for (int m = 0; m < 4; ++m)
                #pragma unroll
                for (int n = 0; n < 4; ++n)
                    acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[m], b_frag[n], acc[m][n], 0, 0, 0);
        }
        __syncthreads();
    }

    // ---- epilogue: C/D layout col=lane&15, row=(lane>>4)*4+reg (guide §3, m89/m91) ----
    #pragma unroll
    for (int m = 0; m < 4; ++m) {
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
            int col = tile_n + wn * 64 + n * 16 + (lane & 15);
            float b = HAS_BIAS ? bias[col] : 0.0f;
            #pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                int row = tile_m + wm * 64 + m * 16 + (lane >> 4) * 4 + reg;
                float v = acc[m][n][reg] + b;
                if constexpr (ACT == 1) v = gelu_tanh(v);
                if constexpr (ACT == 2) v = sigmoidf(v);
                if constexpr (OUT_BF16)
                    ((short*)C)[(size_t)row * N + col] = f_to_bf16(v);
                else
                    ((float*)C)[(size_t)row * N + col] = v;
            }
        }
    }
}

extern "C" int forge_gemm_bt(
    const void* A, const void* BT, const void* bias, void* C,
    int M, int N, int K, int act, int out_bf16, void* stream)
{
    if ((M % BM) || (N % BN) || (K % BK)) return 9001;
    hipStream_t s = (hipStream_t)stream;
    dim3 grid(M / BM, N / BN);
    dim3 block(256);
    bool hb = bias != nullptr;
    #define DISPATCH(A_, O_, B_)                                                                  \
        hipLaunchKernelGGL((gemm_bt_kernel<A_, O_, B_>), grid, block, 0, s,                       \
                           (const short*)A, (const short*)BT, (const float*)bias, C, M, N, K)
    switch (act * 4 + (out_bf16 ? 2 : 0) + (hb ? 1 : 0)) {
        case 0: DISPATCH(0, false, false); break;
        case 1: DISPATCH(0, false, true); break;
        case 2: DISPATCH(0, true, false); break;
        case 3: DISPATCH(0, true, true); break;
        case 4: DISPATCH(1, false, false); break;
        case 5: DISPATCH(1, false, true); break;
        case 6: DISPATCH(1, true, false); break;
        case 7: DISPATCH(1, true, true); break;
        case 8: DISPATCH(2, false, false); break;
        case 9: DISPATCH(2, false, true); break;
        case 10: DISPATCH(2, true, false); break;
        case 11: DISPATCH(2, true, true); break;
        default: return 9002;
    }
    #undef DISPATCH
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Skinny classifier head: C = act(A[M,K] @ WT[C,K]^T + bias), C <= 32.
// One wave per output row; vectorized bf16x8 loads (guide G13).
// ---------------------------------------------------------------------------

template <int ACT>
__global__ __launch_bounds__(256) void gemv_head_kernel(
    const short* __restrict__ A, const short* __restrict__ WT,
    const float* __restrict__ bias, float* __restrict__ out,
    int M, int C, int K)
{
    int wave_global = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    if (wave_global >= M) return;
    const short* a_row = A + (size_t)wave_global * K;

    for (int c = 0; c < C; ++c) {
        const short* w_row = WT + (size_t)c * K;
        float dot = 0.0f;
        for (int k = lane * 8; k < K; k += WAVE * 8) {
            short8 av = *(const short8*)(a_row + k);
            short8 wv = *(const short8*)(w_row + k);
            #pragma unroll
            for (int j = 0; j < 8; ++j) dot += bf16_to_f(av[j]) * bf16_to_f(wv[j]);
        }
        #pragma unroll
        for (int off = WAVE / 2; off > 0; off >>= 1) dot += __shfl_down(dot, off);
        if (lane == 0) {
            float v = dot + (bias ? bias[c] : 0.0f);
            if constexpr (ACT == 1) v = gelu_tanh(v);
            if constexpr (ACT == 2) v = sigmoidf(v);
            out[(size_t)wave_global * C + c] = v;
        }
    }
}

extern "C" int forge_gemv_head(
    const void* A, const void* WT, const void* bias, void* out,
    int M, int C, int K, int act, void* stream)
{
    if (K % (WAVE * 8)) return 9003;
    hipStream_t s = (hipStream_t)stream;
    int waves_per_block = 256 / WAVE;
    int grid = ceil_div(M, waves_per_block);
    switch (act) {
        case 0: hipLaunchKernelGGL(gemv_head_kernel<0>, dim3(grid), dim3(256), 0, s, (const short*)A, (const short*)WT, (const float*)bias, (float*)out, M, C, K); break;
        case 1: hipLaunchKernelGGL(gemv_head_kernel<1>, dim3(grid), dim3(256), 0, s, (const short*)A, (const short*)WT, (const float*)bias, (float*)out, M, C, K); break;
        case 2: hipLaunchKernelGGL(gemv_head_kernel<2>, dim3(grid), dim3(256), 0, s, (const short*)A, (const short*)WT, (const float*)bias, (float*)out, M, C, K); break;
        default: return 9002;
    }
    return (int)hipGetLastError();
}

// ---------------------------------------------------------------------------
// Row-wise running argmax merge for the chunked semantic-cache sweep:
// for each row, fold scores[row, 0:Nc] (with column validity) into
// (best_val[row], best_idx[row]).  One wave per row.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void rows_argmax_merge_kernel(
    const float* __restrict__ scores, int M, int Nc, int idx_base,
    const uint8_t* __restrict__ valid,  // [Nc] or null
    float* __restrict__ best_val, int32_t* __restrict__ best_idx)
{
    int row = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
    int lane = threadIdx.x & (WAVE - 1);
    if (row >= M) return;
    float bv = -1e30f;
    int bi = -1;
    const float* srow = scores + (size_t)row * Nc;
    if ((Nc & 3) == 0) {
        // vectorized: float4 per lane (16 B coalesced) + packed validity,
        // branch-free select — the scalar form ran ~4x under HBM bandwidth
        int n4 = Nc >> 2;
        for (int q = lane; q < n4; q += WAVE) {
            float4v v = *(const float4v*)(srow + q * 4);
            uint32_t vm = 0x01010101u;
            if (valid) __builtin_memcpy(&vm, valid + (size_t)q * 4, 4);
            #pragma unroll
            for (int e = 0; e < 4; ++e) {
                float x = ((vm >> (8 * e)) & 0xFF) ? v[e] : -1e30f;
                if (x > bv) { bv = x; bi = idx_base + q * 4 + e; }
            }
        }
    } else {
        for (int c = lane; c < Nc; c += WAVE) {
            if (valid && !valid[c]) continue;
            float v = srow[c];
            if (v > bv) { bv = v; bi = idx_base + c; }
        }
    }
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        float ov = __shfl_down(bv, off);
        int oi = __shfl_down(bi, off);
        if (ov > bv || (ov == bv && oi >= 0 && (bi < 0 || oi < bi))) { bv = ov; bi = oi; }
    }
    if (lane == 0) {
        if (bv > best_val[row]) { best_val[row] = bv; best_idx[row] = bi; }
    }
}

extern "C" int forge_rows_argmax_merge(
    const void* scores, int M, int Nc, int idx_base, const void* valid,
    void* best_val, void* best_idx, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    int waves_per_block = 256 / WAVE;
    int grid = ceil_div(M, waves_per_block);
    hipLaunchKernelGGL(rows_argmax_merge_kernel, dim3(grid), dim3(256), 0, s,
                       (const float*)scores, M, Nc, idx_base, (const uint8_t*)valid,
                       (float*)best_val, (int32_t*)best_idx);
    return (int)hipGetLastError();
}
