// bf16 MFMA GEMM v2 — 256×256 tile, 4-slot LDS ring, counted vmcnt.
//
// Applies the CDNA4 guide's verified technique stack (§5.5): T1-less variant
// of the 256² schedule — T3+T4 (phase interleave with counted `s_waitcnt
// vmcnt(N)`, never drained to 0 in the main loop), T5 (`s_setprio(1)`
// around the MFMA cluster), and a T2-style LDS swizzle adapted to 64-byte
// rows (phys = logical ^ ((row&3)<<4): spreads a column-slice read across
// the 4 16-byte units of a row → 8-way bank conflict drops to ~2-4-way).
//
// Geometry: BM=BN=256, BK=32, 512 threads = 8 waves (2×4). Per-wave output
// 128×64 = 8×4 fragments of 16×16; 32 MFMA + 12 ds_read_b128 per K-tile per
// wave. LDS = 4-slot ring × (A 16 KB + B 16 KB) = 128 KB; prefetch distance
// 2 K-tiles keeps 4 loads in flight across the tile boundary (vmcnt(4)).
//
// Preconditions (host wrapper pads): M%256==0, N%256==0, K%32==0.

#include "common.h"

typedef __bf16 bf16x8_v2 __attribute__((ext_vector_type(8)));
typedef float f32x4_v2 __attribute__((ext_vector_type(4)));

#define V2_BM 256
#define V2_BN 256
#define V2_BK 32
#define V2_ROWB 64                      // bytes per LDS row (32 bf16)
#define V2_SLOT_A (V2_BM * V2_ROWB)     // 16 KB per A slot
#define V2_SLOT_B (V2_BN * V2_ROWB)     // 16 KB per B slot

__device__ __forceinline__ uint32_t swz2(uint32_t L) {
    // spread the 4 16B units of each 64B row by row&3 (involution)
    return L ^ (((L >> 6) & 3u) << 4);
}

// swizzle ablation (PMC: VAR2 still shows ~25M LDS bank conflicts/call):
//   VAR 4: the guide's m201 st_16x32 formula (row&8 flips col-byte bit 5)
//   VAR 5: two-bit spread (row&8 -> bit5, row&4 -> bit4)
template <int VAR>
__device__ __forceinline__ uint32_t swz_sel(uint32_t L) {
    if constexpr (VAR == 4) return L ^ (((L >> 9) & 1u) << 5);
    else if constexpr (VAR == 5) return L ^ (((L >> 9) & 1u) << 5) ^ (((L >> 8) & 1u) << 4);
    else return L ^ (((L >> 6) & 3u) << 4);
}

template <int ACT, bool OUT_BF16, bool HAS_BIAS, int VAR = 0>
__global__ __launch_bounds__(512, 1) void gemm_bt_v2_kernel(
    const short* __restrict__ A,   // [M,K] bf16
    const short* __restrict__ BT,  // [N,K] bf16
    const float* __restrict__ bias,
    void* __restrict__ C,
    int M, int N, int K)
{
    // LDS ring: A slots [4][16KB] then B slots [4][16KB]
    __shared__ short lds[4 * (V2_SLOT_A + V2_SLOT_B) / 2];
    char* lds_a = (char*)lds;
    char* lds_b = (char*)lds + 4 * V2_SLOT_A;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wave = tid >> 6;       // 0..7
    const int wm = wave >> 2;        // 0..1  (M half)
    const int wn = wave & 3;         // 0..3  (N quarter)
    int bm = blockIdx.x, bn = blockIdx.y;
    if constexpr (VAR == 3) {
        // T1 XCD-aware bijective swizzle (guide m204): 1D grid; consecutive
        // ids on one XCD share an N-panel (bn) whose 2 MB B slab L2-fits.
        int gx = M / V2_BM;
        int nwg = gx * (N / V2_BN);
        int id = blockIdx.x;
        int q = nwg / 8, r = nwg % 8;
        int xcd = id % 8, pos = id / 8;
        int swz_id = (xcd < r) ? xcd * (q + 1) + pos : r * (q + 1) + (xcd - r) * q + pos;
        bm = swz_id % gx;
        bn = swz_id / gx;
    }
    const int tile_m = bm * V2_BM;
    const int tile_n = bn * V2_BN;
    const int n_tiles = K / V2_BK;

    f32x4_v2 acc[8][4];
    #pragma unroll
    for (int i = 0; i < 8; ++i)
        #pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = (f32x4_v2){0.f, 0.f, 0.f, 0.f};

    // ---- staging: thread stages 2 A-loads + 2 B-loads per tile (16B each) ----
    // A load i (i=0,1): linear LDS byte P = (i*512 + tid)*16 within the slot;
    // the data fetched is the *logical* offset swz2(P) (rule #21: linear dest,
    // inverse-swizzled source, swizzled read).
    auto stage_a = [&](int t, int i) {
        uint32_t P = (uint32_t)(i * 512 + tid) * 16u;
        uint32_t L = swz_sel<VAR>(P);
        uint32_t row = L >> 6, colb = L & 63u;
        const short* g = A + ((size_t)(tile_m + row) * K + t * V2_BK) + (colb >> 1);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)g,
            (__attribute__((address_space(3))) uint32_t*)(lds_a + (size_t)(t & 3) * V2_SLOT_A
                                                          + (i * 512 + (wave * 64)) * 16),
            16, 0, 0);
    };
    auto stage_b = [&](int t, int i) {
        uint32_t P = (uint32_t)(i * 512 + tid) * 16u;
        uint32_t L = swz_sel<VAR>(P);
        uint32_t row = L >> 6, colb = L & 63u;
        const short* g = BT + ((size_t)(tile_n + row) * K + t * V2_BK) + (colb >> 1);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)g,
            (__attribute__((address_space(3))) uint32_t*)(lds_b + (size_t)(t & 3) * V2_SLOT_B
                                                          + (i * 512 + (wave * 64)) * 16),
            16, 0, 0);
    };
    auto stage_quarter = [&](int t, int q) {
        // per-thread load q of tile t: [A0, A1, B0, B1]
        if (q < 2) stage_a(t, q); else stage_b(t, q - 2);
    };

    // ---- prologue: stage tiles 0 and 1, retire tile 0 ----
    #pragma unroll
    for (int q = 0; q < 4; ++q) stage_quarter(0, q);
    if (n_tiles > 1) {
        #pragma unroll
        for (int q = 0; q < 4; ++q) stage_quarter(1, q);
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    const uint32_t a_col = (uint32_t)((lane >> 4) * 16);  // byte col of this lane's k-slice
    const uint32_t a_row_base = (uint32_t)(wm * 128 + (lane & 15));
    const uint32_t b_row_base = (uint32_t)(wn * 64 + (lane & 15));

    for (int kt = 0; kt < n_tiles; ++kt) {
        const char* slot_a = lds_a + (size_t)(kt & 3) * V2_SLOT_A;
        const char* slot_b = lds_b + (size_t)(kt & 3) * V2_SLOT_B;
        bf16x8_v2 b_frag[4];

        #pragma unroll
        for (int q = 0; q < 4; ++q) {
            if (kt + 2 < n_tiles) stage_quarter(kt + 2, q);
            if (q == 0) {
                #pragma unroll
                for (int fc = 0; fc < 4; ++fc) {
                    uint32_t row = b_row_base + fc * 16;
                    b_frag[fc] = *(const bf16x8_v2*)(slot_b + swz_sel<VAR>(row * V2_ROWB + a_col));
                }
            }
            bf16x8_v2 a0, a1;
            {
                uint32_t r0 = a_row_base + (2 * q) * 16;
                uint32_t r1 = a_row_base + (2 * q + 1) * 16;
                a0 = *(const bf16x8_v2*)(slot_a + swz_sel<VAR>(r0 * V2_ROWB + a_col));
                a1 = *(const bf16x8_v2*)(slot_a + swz_sel<VAR>(r1 * V2_ROWB + a_col));
            }
            if constexpr (VAR < 2) __builtin_amdgcn_s_setprio(1);
            #pragma unroll
            for (int fc = 0; fc < 4; ++fc) {
                acc[2 * q][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b_frag[fc], acc[2 * q][fc], 0, 0, 0);
                acc[2 * q + 1][fc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b_frag[fc], acc[2 * q + 1][fc], 0, 0, 0);
            }
            if constexpr (VAR < 2) __builtin_amdgcn_s_setprio(0);
            if constexpr (VAR == 0) __builtin_amdgcn_s_barrier();
        }
        // tile boundary: retire tile kt+1's loads (counted — tile kt+2 stays in flight)
        if (kt + 1 < n_tiles) {
            if (kt + 2 < n_tiles) {
                asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            } else {
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            }
            __builtin_amdgcn_s_barrier();
        }
    }

    // ---- epilogue: C/D layout col=lane&15, row=(lane>>4)*4+reg (guide §3) ----
    #pragma unroll
    for (int fr = 0; fr < 8; ++fr) {
        #pragma unroll
        for (int fc = 0; fc < 4; ++fc) {
            int col = tile_n + wn * 64 + fc * 16 + (lane & 15);
            float b = HAS_BIAS ? bias[col] : 0.0f;
            #pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
                int row = tile_m + wm * 128 + fr * 16 + (lane >> 4) * 4 + reg;
                float v = acc[fr][fc][reg] + b;
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

extern "C" int forge_gemm_bt_v2_var(
    const void* A, const void* BT, void* C, int M, int N, int K, int var, void* stream)
{
    if ((M % V2_BM) || (N % V2_BN) || (K % V2_BK)) return 9001;
    hipStream_t s = (hipStream_t)stream;
    dim3 grid(M / V2_BM, N / V2_BN);
    dim3 block(512);
    switch (var) {
        case 0: hipLaunchKernelGGL((gemm_bt_v2_kernel<0, false, false, 0>), grid, block, 0, s, (const short*)A, (const short*)BT, nullptr, C, M, N, K); break;
        case 1: hipLaunchKernelGGL((gemm_bt_v2_kernel<0, false, false, 1>), grid, block, 0, s, (const short*)A, (const short*)BT, nullptr, C, M, N, K); break;
        case 2: hipLaunchKernelGGL((gemm_bt_v2_kernel<0, false, false, 2>), grid, block, 0, s, (const short*)A, (const short*)BT, nullptr, C, M, N, K); break;
        case 3: hipLaunchKernelGGL((gemm_bt_v2_kernel<0, false, false, 3>), dim3((M / V2_BM) * (N / V2_BN)), block, 0, s, (const short*)A, (const short*)BT, nullptr, C, M, N, K); break;
        case 4: hipLaunchKernelGGL((gemm_bt_v2_kernel<0, false, false, 4>), grid, block, 0, s, (const short*)A, (const short*)BT, nullptr, C, M, N, K); break;
        case 5: hipLaunchKernelGGL((gemm_bt_v2_kernel<0, false, false, 5>), grid, block, 0, s, (const short*)A, (const short*)BT, nullptr, C, M, N, K); break;
        default: return 9002;
    }
    return (int)hipGetLastError();
}

extern "C" int forge_gemm_bt_v2(
    const void* A, const void* BT, const void* bias, void* C,
    int M, int N, int K, int act, int out_bf16, void* stream)
{
    if ((M % V2_BM) || (N % V2_BN) || (K % V2_BK)) return 9001;
    hipStream_t s = (hipStream_t)stream;
    dim3 grid(M / V2_BM, N / V2_BN);
    dim3 block(512);
    bool hb = bias != nullptr;
    // production configuration = VAR 4 (free-running ring + counted vmcnt +
    // m201 st_16x32 swizzle — ablation table in profiles/README.md: 1133 TF
    // vs 1094 for the old row&3 spread; phase barriers/setprio measured
    // NEGATIVE at this occupancy)
    #define DISPATCH2(A_, O_, B_)                                                                \
        hipLaunchKernelGGL((gemm_bt_v2_kernel<A_, O_, B_, 4>), grid, block, 0, s,                \
                           (const short*)A, (const short*)BT, (const float*)bias, C, M, N, K)
    switch (act * 4 + (out_bf16 ? 2 : 0) + (hb ? 1 : 0)) {
        case 0: DISPATCH2(0, false, false); break;
        case 1: DISPATCH2(0, false, true); break;
        case 2: DISPATCH2(0, true, false); break;
        case 3: DISPATCH2(0, true, true); break;
        case 4: DISPATCH2(1, false, false); break;
        case 5: DISPATCH2(1, false, true); break;
        case 6: DISPATCH2(1, true, false); break;
        case 7: DISPATCH2(1, true, true); break;
        case 8: DISPATCH2(2, false, false); break;
        case 9: DISPATCH2(2, false, true); break;
        case 10: DISPATCH2(2, true, false); break;
        case 11: DISPATCH2(2, true, true); break;
        default: return 9002;
    }
    #undef DISPATCH2
    return (int)hipGetLastError();
}
