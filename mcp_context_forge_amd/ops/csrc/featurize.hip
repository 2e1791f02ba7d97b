// Hashed count-vector featurization of a packed request batch.
//
// Exact semantics of ops/featurize.py (the CPU oracle): tokens are maximal
// [A-Za-z0-9_] runs, ASCII-lowercased, FNV-1a 32-bit hashed, bucketed with
// & (dim-1), counted, then L2-normalized. Output row-major [B, dim] bf16
// (classifier/semantic-cache input) and optionally fp32.
//
// Reference-gateway analog: response_cache_by_prompt._vectorize (:55) and
// the moderation-classifier featurization — per-request Python there, one
// LDS-histogram workgroup per request here (dim*4 B histogram in LDS;
// dim<=8192 fits the 160 KiB/CU budget trivially, guide §2).

#include "common.h"

#define FNV_OFFSET 0x811C9DC5u
#define FNV_PRIME 0x01000193u

__device__ __forceinline__ bool is_word(uint8_t b) {
    return (b >= '0' && b <= '9') || (b >= 'A' && b <= 'Z') || (b >= 'a' && b <= 'z') || b == '_';
}
__device__ __forceinline__ uint8_t lower(uint8_t b) { return (b >= 'A' && b <= 'Z') ? b + 0x20 : b; }

__global__ __launch_bounds__(256) void featurize_kernel(
    const uint8_t* __restrict__ data,
    const int32_t* __restrict__ beg,
    const int32_t* __restrict__ end_,
    int dim,
    short* __restrict__ out_bf16,   // [B, dim] or nullptr
    float* __restrict__ out_f32)    // [B, dim] or nullptr
{
    extern __shared__ uint32_t hist[];  // [dim] counts, then 1 extra slot for norm
    int r = blockIdx.x;
    int32_t rbeg = beg[r], rend = end_[r];
    int len = rend - rbeg;

    for (int i = threadIdx.x; i < dim; i += blockDim.x) hist[i] = 0u;
    __syncthreads();

    // Each thread owns tokens that START inside its chunk; it follows them to
    // their end even past the chunk boundary (exact, no split tokens).
    int chunk = (len + blockDim.x - 1) / blockDim.x;
    int c0 = threadIdx.x * chunk;
    int c1 = min(c0 + chunk, len);
    for (int i = c0; i < c1; ++i) {
        uint8_t b = data[rbeg + i];
        if (!is_word(b)) continue;
        bool starts = (i == 0) || !is_word(data[rbeg + i - 1]);
        if (!starts) continue;
        uint32_t h = FNV_OFFSET;
        int j = i;
        while (j < len) {
            uint8_t bj = data[rbeg + j];
            if (!is_word(bj)) break;
            h = (h ^ (uint32_t)lower(bj)) * FNV_PRIME;
            ++j;
        }
        atomicAdd(&hist[h & (uint32_t)(dim - 1)], 1u);
    }
    __syncthreads();

    // L2 norm: block reduction of sum(counts^2).
    float ss = 0.0f;
    for (int i = threadIdx.x; i < dim; i += blockDim.x) {
        float c = (float)hist[i];
        ss += c * c;
    }
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) ss += __shfl_down(ss, off);
    __shared__ float wave_ss[8];
    int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) wave_ss[wid] = ss;
    __syncthreads();
    if (threadIdx.x == 0) {
        float total = 0.0f;
        for (int w = 0; w < (int)(blockDim.x / WAVE); ++w) total += wave_ss[w];
        wave_ss[0] = (total > 0.0f) ? rsqrtf(total) : 0.0f;
    }
    __syncthreads();
    float inv = wave_ss[0];

    for (int i = threadIdx.x; i < dim; i += blockDim.x) {
        float v = (float)hist[i] * inv;
        if (out_f32) out_f32[(size_t)r * dim + i] = v;
        if (out_bf16) out_bf16[(size_t)r * dim + i] = f_to_bf16(v);
    }
}

extern "C" int forge_featurize(
    const void* data, const void* beg, const void* end_, int batch, int dim,
    void* out_bf16, void* out_f32, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    size_t lds = (size_t)dim * 4;
    hipLaunchKernelGGL(featurize_kernel, dim3(batch), dim3(256), lds, s,
                       (const uint8_t*)data, (const int32_t*)beg, (const int32_t*)end_, dim,
                       (short*)out_bf16, (float*)out_f32);
    return (int)hipGetLastError();
}
