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
