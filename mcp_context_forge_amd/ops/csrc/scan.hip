// Multi-pattern DFA scan over a packed request batch.
//
// Tables come from ops/dfa.py (compile_patterns); semantics are identical to
// dfa.scan_reference / match_mask_reference — the CPU oracle in tests.
// Reference-gateway analog: the per-request regex/deny/PII/harm scans
// (plugins/deny_filter, regex_filter, cpex-pii-filter,
// harmful_content_detector) executed per request in Python `re`; here one
// kernel scans the whole micro-batch.
//
// Parallelization: one thread per request (requests are short JSON payloads,
// ~100 B–4 KB). The transition table is small (tens of KB) and L2/LDS-
// resident. When the table fits in 64 KB of LDS the block cooperatively
// stages it (fast path); otherwise lanes read it through L2.

#include "common.h"

template <bool USE_LDS>
__global__ __launch_bounds__(256) void scan_kernel(
    const uint8_t* __restrict__ data,     // packed bytes
    const int32_t* __restrict__ beg,
    const int32_t* __restrict__ end_,
    int batch,
    const uint16_t* __restrict__ next,    // [S, C]
    const uint8_t* __restrict__ klass,    // [256]
    const uint32_t* __restrict__ accept,  // [S]
    int n_states, int n_classes,
    uint32_t* __restrict__ out_mask,      // [B] pattern-id bitmask
    int32_t* __restrict__ out_first_end)  // [B] first match end (or -1); optional
{
    extern __shared__ uint8_t lds_raw[];
    const uint16_t* tnext = next;
    const uint8_t* tklass = klass;
    const uint32_t* taccept = accept;

    if constexpr (USE_LDS) {
        uint16_t* lnext = (uint16_t*)lds_raw;
        uint8_t* lklass = (uint8_t*)(lds_raw + (size_t)n_states * n_classes * 2);
        uint32_t* laccept = (uint32_t*)(lklass + 256);
        int total16 = n_states * n_classes;
        for (int i = threadIdx.x; i < total16; i += blockDim.x) lnext[i] = next[i];
        for (int i = threadIdx.x; i < 256; i += blockDim.x) lklass[i] = klass[i];
        for (int i = threadIdx.x; i < n_states; i += blockDim.x) laccept[i] = accept[i];
        __syncthreads();
        tnext = lnext; tklass = lklass; taccept = laccept;
    }

    int r = blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= batch) return;
    int32_t rbeg = beg[r], rend = end_[r];
    uint32_t mask = 0;
    int32_t first_end = -1;
    uint32_t state = 0;
    for (int32_t p = rbeg; p < rend; ++p) {
        uint8_t b = data[p];
        state = tnext[state * n_classes + tklass[b]];
        uint32_t a = taccept[state];
        if (a) {
            if (first_end < 0) first_end = p + 1 - rbeg;
            mask |= a;
        }
    }
    out_mask[r] = mask;
    if (out_first_end) out_first_end[r] = first_end;
}

// ---------------------------------------------------------------------------
// Fused multi-bank scan: ALL DFA banks in ONE launch (grid.y = bank).
//
// Measured motivation (profiles/README_r02.md stage budget): at HTTP batch
// sizes (~1-3k rows) the per-bank launches + per-bank D2H copies dominate
// the scan stage — six launches of ~128 waves each also underfill 256 CUs.
// One launch with grid (rows/256, n_banks) fills the chip bank-parallel,
// writes one [n_banks, B] mask matrix, and the host does ONE D2H.
//
// Bank descriptors live in device memory (built once per bank-set); each
// block stages ITS bank's table into LDS when it fits the per-block
// allocation (banks over the limit read through L2, same as scan_kernel).

struct ScanBankDesc {
    const uint16_t* next;
    const uint8_t* klass;
    const uint32_t* accept;
    int n_states;
    int n_classes;
    int use_lds;   // table fits the dynamic LDS allocation
    int _pad[2];
};

__global__ __launch_bounds__(256) void scan_multi_kernel(
    const uint8_t* __restrict__ data,
    const int32_t* __restrict__ beg,
    const int32_t* __restrict__ end_,
    int batch,
    const ScanBankDesc* __restrict__ descs,
    uint32_t* __restrict__ out_mask)      // [n_banks, batch]
{
    extern __shared__ uint8_t lds_raw[];
    const ScanBankDesc d = descs[blockIdx.y];
    const uint16_t* tnext = d.next;
    const uint8_t* tklass = d.klass;
    const uint32_t* taccept = d.accept;
    if (d.use_lds) {
        uint16_t* lnext = (uint16_t*)lds_raw;
        uint8_t* lklass = (uint8_t*)(lds_raw + (size_t)d.n_states * d.n_classes * 2);
        uint32_t* laccept = (uint32_t*)(lklass + 256);
        int total16 = d.n_states * d.n_classes;
        for (int i = threadIdx.x; i < total16; i += blockDim.x) lnext[i] = d.next[i];
        for (int i = threadIdx.x; i < 256; i += blockDim.x) lklass[i] = d.klass[i];
        for (int i = threadIdx.x; i < d.n_states; i += blockDim.x) laccept[i] = d.accept[i];
        __syncthreads();
        tnext = lnext; tklass = lklass; taccept = laccept;
    }
    int r = blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= batch) return;
    int32_t rbeg = beg[r], rend = end_[r];
    uint32_t mask = 0;
    uint32_t state = 0;
    int nc = d.n_classes;
    for (int32_t p = rbeg; p < rend; ++p) {
        state = tnext[state * nc + tklass[data[p]]];
        mask |= taccept[state];
    }
    out_mask[(size_t)blockIdx.y * batch + r] = mask;
}

extern "C" int forge_scan_multi(
    const void* data, const void* beg, const void* end_, int batch,
    const void* descs_dev, int n_banks, int lds_bytes,
    void* out_mask, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    int block = 256;
    dim3 grid((unsigned)ceil_div(batch, block), (unsigned)n_banks);
    hipLaunchKernelGGL(scan_multi_kernel, grid, dim3(block), (size_t)lds_bytes, s,
                       (const uint8_t*)data, (const int32_t*)beg, (const int32_t*)end_, batch,
                       (const ScanBankDesc*)descs_dev, (uint32_t*)out_mask);
    return (int)hipGetLastError();
}

extern "C" int forge_scan(
    const void* data, const void* beg, const void* end_, int batch,
    const void* next, const void* klass, const void* accept,
    int n_states, int n_classes,
    void* out_mask, void* out_first_end, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    int block = 256;
    int grid = ceil_div(batch, block);
    size_t table_bytes = (size_t)n_states * n_classes * 2 + 256 + (size_t)n_states * 4;
    if (table_bytes <= 64 * 1024) {
        hipLaunchKernelGGL((scan_kernel<true>), dim3(grid), dim3(block), table_bytes, s,
                           (const uint8_t*)data, (const int32_t*)beg, (const int32_t*)end_, batch,
                           (const uint16_t*)next, (const uint8_t*)klass, (const uint32_t*)accept,
                           n_states, n_classes, (uint32_t*)out_mask, (int32_t*)out_first_end);
    } else {
        hipLaunchKernelGGL((scan_kernel<false>), dim3(grid), dim3(block), 0, s,
                           (const uint8_t*)data, (const int32_t*)beg, (const int32_t*)end_, batch,
                           (const uint16_t*)next, (const uint8_t*)klass, (const uint32_t*)accept,
                           n_states, n_classes, (uint32_t*)out_mask, (int32_t*)out_first_end);
    }
    return (int)hipGetLastError();
}
