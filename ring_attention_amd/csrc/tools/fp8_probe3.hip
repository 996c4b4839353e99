// Disambiguate: does lane L's scale_a apply to the 32 A-bytes lane L itself
// provides, or to the OTHER k-half's bytes?  A bytes: lane<32 -> 1.0,
// lane>=32 -> 0.5 (for every row); B = 1.0; C(i,j) = 48 at unit scales.
//   scale2 on lane 0 (row 0):  own-bytes semantics -> C(0,:) = 80
//                              cross semantics     -> C(0,:) = 64
//   scale2 on lane 32 (row 0): own-bytes -> 64;  cross -> 80 ... wait:
//   lane32 bytes are 0.5*32=16; doubled -> 32: C = 32+32 = 64 (own)
//   cross: lane32's scale doubles lane0's bytes -> 64+16 = 80.
// Same check for scale_b with asymmetric B halves.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef int i32x8 __attribute__((ext_vector_type(8)));
#define CHECK(x) do { hipError_t e = (x); if (e) { printf("HIP err %d\n", (int)e); return 1; } } while (0)

__global__ void probeA(float* C, int L) {
    int l = threadIdx.x & 63;
    i32x8 af, bf;
    for (int r = 0; r < 8; ++r) {
        af[r] = (l >> 5) ? 0x30303030 : 0x38383838;   // 0.5 : 1.0 e4m3
        bf[r] = 0x38383838;
    }
    int sa = (l == L) ? 0x7F7F7F80 : 0x7F7F7F7F;
    f32x16 c = {};
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(af, bf, c, 0, 0, 0, sa, 0, 0x7F7F7F7F);
    for (int r = 0; r < 16; ++r) {
        int ci = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5), cj = l & 31;
        C[ci * 32 + cj] = c[r];
    }
}
__global__ void probeB(float* C, int L) {
    int l = threadIdx.x & 63;
    i32x8 af, bf;
    for (int r = 0; r < 8; ++r) {
        af[r] = 0x38383838;
        bf[r] = (l >> 5) ? 0x30303030 : 0x38383838;
    }
    int sb = (l == L) ? 0x7F7F7F80 : 0x7F7F7F7F;
    f32x16 c = {};
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(af, bf, c, 0, 0, 0, 0x7F7F7F7F, 0, sb);
    for (int r = 0; r < 16; ++r) {
        int ci = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5), cj = l & 31;
        C[ci * 32 + cj] = c[r];
    }
}
int main() {
    float* dC; CHECK(hipMalloc(&dC, 32 * 32 * 4));
    std::vector<float> C(32 * 32);
    for (int L : {0, 32}) {
        hipLaunchKernelGGL(probeA, dim3(1), dim3(64), 0, 0, dC, L);
        CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
        printf("A-scale2 on lane %2d: C(0,0)=%g C(0,5)=%g C(1,0)=%g\n", L, C[0], C[5], C[32]);
    }
    for (int L : {0, 32}) {
        hipLaunchKernelGGL(probeB, dim3(1), dim3(64), 0, 0, dC, L);
        CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
        printf("B-scale2 on lane %2d: C(0,0)=%g C(5,0)=%g C(0,1)=%g\n", L, C[0], C[5 * 32], C[1]);
    }
    return 0;
}
