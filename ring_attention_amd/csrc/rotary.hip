// Fused rotary position embedding for gfx950.
//
// out = x * cos(freqs) + rotate_half(x) * sin(freqs), applied to q and k in
// ONE memory-bound pass (the torch composition is ~8 elementwise kernels per
// tensor).  freqs (n, d) fp32 are host-precomputed (guide: keep trig tables
// off the device hot path).  Backward is the inverse rotation (sin negated),
// so the same kernel serves both directions.
//
// x: bf16 (B, N, H, D) contiguous; freqs: fp32 (N, D).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "attn_common.h"

namespace ring_attn {

template <int D>
__global__ __launch_bounds__(256) void rotary_kernel(RotaryParams p) {
    // one wave per row; lane l handles elements l and l + D/2 (pairs rotate
    // together, so one lane holding both halves needs no cross-lane traffic)
    constexpr int PER_LANE = D / 2 / 64 > 0 ? D / 2 / 64 : 1;  // D<=128: 1
    const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
    if (row >= p.rows) return;
    const int lane = threadIdx.x & 63;
    const long nidx = (row / p.h) % p.n;

    const __bf16* xr = (const __bf16*)p.x + row * D;
    __bf16* orow = (__bf16*)p.out + row * D;
    const float* ct = p.cos_t + nidx * (D / 2);
    const float* st = p.sin_t + nidx * (D / 2);

    #pragma unroll
    for (int e = 0; e < PER_LANE; ++e) {
        int i = lane + e * 64;           // first-half index
        if (i < D / 2) {
            float x1 = (float)xr[i];
            float x2 = (float)xr[i + D / 2];
            float c1 = ct[i];
            float s1 = st[i] * p.sin_sign;
            // freqs duplicate across halves (cat(f, f)): same c/s both halves
            orow[i] = (__bf16)(x1 * c1 - x2 * s1);
            orow[i + D / 2] = (__bf16)(x2 * c1 + x1 * s1);
        }
    }
}

void launch_rotary(const RotaryParams& p, int head_dim, hipStream_t stream) {
    dim3 grid((p.rows + 3) / 4);
    dim3 block(256);
    if (head_dim == 64) {
        hipLaunchKernelGGL(rotary_kernel<64>, grid, block, 0, stream, p);
    } else if (head_dim == 32) {
        hipLaunchKernelGGL(rotary_kernel<32>, grid, block, 0, stream, p);
    } else if (head_dim == 128) {
        hipLaunchKernelGGL(rotary_kernel<128>, grid, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}

}  // namespace ring_attn
