// Small runtime utilities exposed through the C API: device info and a
// batched bf16 row-scatter used by the HBM-resident semantic cache insert
// (response_cache_by_prompt analog — keys live in device memory, not Redis).

#include "common.h"

__global__ __launch_bounds__(256) void rows_scatter_bf16_kernel(
    const short* __restrict__ src,   // [R, D]
    const int32_t* __restrict__ slots,  // [R] destination row indices
    short* __restrict__ dst,         // [capacity, D]
    int R, int D)
{
    int r = blockIdx.x;
    if (r >= R) return;
    int32_t slot = slots[r];
    if (slot < 0) return;
    const short* s = src + (size_t)r * D;
    short* d = dst + (size_t)slot * D;
    for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
        *(short8*)(d + i) = *(const short8*)(s + i);
    }
}

extern "C" int forge_rows_scatter_bf16(
    const void* src, const void* slots, void* dst, int R, int D, void* stream)
{
    if (D % 8) return 9004;
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(rows_scatter_bf16_kernel, dim3(R), dim3(256), 0, s,
                       (const short*)src, (const int32_t*)slots, (short*)dst, R, D);
    return (int)hipGetLastError();
}

// Fused semcache insert: gather feature rows by index and scatter into the
// key matrix slots, setting validity — replaces a 64 MB torch gather copy +
// two indexing kernels per batch.
__global__ __launch_bounds__(256) void rows_gather_scatter_bf16_kernel(
    const short* __restrict__ src,       // [B, D]
    const int32_t* __restrict__ src_rows,  // [R]
    const int32_t* __restrict__ dst_slots, // [R]
    short* __restrict__ dst,             // [capacity, D]
    uint8_t* __restrict__ valid,         // [capacity] or null
    int R, int D)
{
    int r = blockIdx.x;
    if (r >= R) return;
    int32_t row = src_rows[r];
    int32_t slot = dst_slots[r];
    if (slot < 0 || row < 0) return;
    const short* s = src + (size_t)row * D;
    short* d = dst + (size_t)slot * D;
    for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
        *(short8*)(d + i) = *(const short8*)(s + i);
    }
    if (threadIdx.x == 0 && valid != nullptr) valid[slot] = 1;
}

extern "C" int forge_rows_gather_scatter_bf16(
    const void* src, const void* src_rows, const void* dst_slots, void* dst,
    void* valid, int R, int D, void* stream)
{
    if (D % 8) return 9004;
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(rows_gather_scatter_bf16_kernel, dim3(R), dim3(256), 0, s,
                       (const short*)src, (const int32_t*)src_rows, (const int32_t*)dst_slots,
                       (short*)dst, (uint8_t*)valid, R, D);
    return (int)hipGetLastError();
}

extern "C" int forge_device_count(int* n) {
    HIP_CHECK(hipGetDeviceCount(n));
    return 0;
}

extern "C" int forge_device_name(int dev, char* buf, int len) {
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, dev));
    snprintf(buf, len, "%s (gcnArch %s, %d CUs)", prop.name, prop.gcnArchName, prop.multiProcessorCount);
    return 0;
}

extern "C" int forge_synchronize(void* stream) {
    HIP_CHECK(hipStreamSynchronize((hipStream_t)stream));
    return 0;
}

// Exact verification for the two-stage semcache search: one wave per row
// computes the full-dim bf16 dot product between the query row and its
// sketch-selected candidate key. 4 rows per block (4 waves), coalesced
// 16-byte loads; wave-level reduction via shfl.
__global__ __launch_bounds__(256) void verify_dot_kernel(
    const short* __restrict__ feats,   // [M, D] bf16
    const short* __restrict__ keys,    // [cap, D] bf16
    const int32_t* __restrict__ idx,   // [M] candidate slot (-1 = none)
    float* __restrict__ out,           // [M] exact dot (or -1e30)
    int M, int D)
{
    int wave = threadIdx.x / 64;
    int lane = threadIdx.x % 64;
    int r = blockIdx.x * 4 + wave;
    if (r >= M) return;
    int32_t slot = idx[r];
    if (slot < 0) { if (lane == 0) out[r] = -1e30f; return; }
    const short* f = feats + (size_t)r * D;
    const short* k = keys + (size_t)slot * D;
    float acc = 0.f;
    for (int i = lane * 8; i < D; i += 64 * 8) {
        short8 fv = *(const short8*)(f + i);
        short8 kv = *(const short8*)(k + i);
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
            short fs = fv[e], ks = kv[e];
            bf16_t fb = *(const bf16_t*)&fs;
            bf16_t kb = *(const bf16_t*)&ks;
            acc += __bfloat162float(fb) * __bfloat162float(kb);
        }
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, 64);
    if (lane == 0) out[r] = acc;
}

extern "C" int forge_verify_dot(
    const void* feats, const void* keys, const void* idx, void* out,
    int M, int D, void* stream)
{
    if (D % 8) return 9004;
    hipStream_t s = (hipStream_t)stream;
    hipLaunchKernelGGL(verify_dot_kernel, dim3((M + 3) / 4), dim3(256), 0, s,
                       (const short*)feats, (const short*)keys,
                       (const int32_t*)idx, (float*)out, M, D);
    return (int)hipGetLastError();
}
