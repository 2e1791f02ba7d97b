// Structural JSON guard over a packed request batch.
//
// Reference-gateway analog: SecurityValidator depth/length checks
// (mcpgateway/common/validators.py) + orjson parse rejection on /rpc
// (main.py:11225). The GPU pipeline runs this before anything else touches
// the payload: UTF-8 validity, bracket/brace balance + max depth, string
// quoting sanity, max string length, and NUL/control-char rejection.
//
// Output status bitmask per request (0 = structurally OK):
//   bit 0: unbalanced braces/brackets  bit 1: depth > max_depth
//   bit 2: unterminated string         bit 3: string longer than max_string
//   bit 4: invalid UTF-8               bit 5: raw control char inside string
// Plus out_depth[r] = max nesting depth observed.

#include "common.h"

__global__ __launch_bounds__(256) void json_guard_kernel(
    const uint8_t* __restrict__ data,
    const int32_t* __restrict__ beg,
    const int32_t* __restrict__ end_,
    int batch, int max_depth, int max_string,
    int32_t* __restrict__ out_status,
    int32_t* __restrict__ out_depth)
{
    int r = blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= batch) return;
    int32_t rbeg = beg[r], rend = end_[r];

    int status = 0;
    int depth = 0, maxd = 0;
    bool in_str = false, esc = false;
    int str_len = 0;
    int cont = 0;  // pending UTF-8 continuation bytes

    for (int32_t p = rbeg; p < rend; ++p) {
        uint8_t b = data[p];
        // UTF-8 validity
        if (cont > 0) {
            if ((b & 0xC0) != 0x80) { status |= 1 << 4; cont = 0; }
            else { --cont; }
        } else if (b >= 0x80) {
            if ((b & 0xE0) == 0xC0) cont = 1;
            else if ((b & 0xF0) == 0xE0) cont = 2;
            else if ((b & 0xF8) == 0xF0) cont = 3;
            else status |= 1 << 4;
        }
        if (in_str) {
            ++str_len;
            if (str_len > max_string) status |= 1 << 3;
            if (esc) { esc = false; continue; }
            if (b == '\\') { esc = true; continue; }
            if (b == '"') { in_str = false; str_len = 0; continue; }
            if (b < 0x20) status |= 1 << 5;
        } else {
            if (b == '"') { in_str = true; str_len = 0; }
            else if (b == '{' || b == '[') { if (++depth > maxd) maxd = depth; if (depth > max_depth) status |= 1 << 1; }
            else if (b == '}' || b == ']') { if (--depth < 0) { status |= 1 << 0; depth = 0; } }
        }
    }
    if (depth != 0) status |= 1 << 0;
    if (in_str) status |= 1 << 2;
    if (cont != 0) status |= 1 << 4;
    out_status[r] = status;
    if (out_depth) out_depth[r] = maxd;
}

extern "C" int forge_json_guard(
    const void* data, const void* beg, const void* end_, int batch, int max_depth, int max_string,
    void* out_status, void* out_depth, void* stream)
{
    hipStream_t s = (hipStream_t)stream;
    int block = 256;
    hipLaunchKernelGGL(json_guard_kernel, dim3(ceil_div(batch, block)), dim3(block), 0, s,
                       (const uint8_t*)data, (const int32_t*)beg, (const int32_t*)end_, batch, max_depth, max_string,
                       (int32_t*)out_status, (int32_t*)out_depth);
    return (int)hipGetLastError();
}
