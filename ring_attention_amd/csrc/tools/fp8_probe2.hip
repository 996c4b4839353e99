// Scale-operand mapping probe for v_mfma_scale_f32_32x32x64_f8f6f4.
// A = B = 1.0 (e4m3 0x38): C_ij = 64 at unit scales.  Give lane L scale 2.0
// on A only; the C entries that read 96 reveal which (row, kblock) lane L's
// scale governs.  Also sweeps opsel_a byte selection.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef int i32x8 __attribute__((ext_vector_type(8)));
#define CHECK(x) do { hipError_t e = (x); if (e) { printf("HIP err %d\n", (int)e); return 1; } } while (0)

__global__ void probe(float* C, int L, int sa_word_L, int opsel_a) {
    int l = threadIdx.x & 63;
    i32x8 af, bf;
    for (int r = 0; r < 8; ++r) { af[r] = 0x38383838; bf[r] = 0x38383838; }
    int sa = (l == L) ? sa_word_L : 0x7F7F7F7F;
    f32x16 c = {};
    switch (opsel_a) {   // opsel must be an immediate
    case 0: c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(af, bf, c, 0, 0, 0, sa, 0, 0x7F7F7F7F); break;
    case 1: c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(af, bf, c, 0, 0, 1, sa, 0, 0x7F7F7F7F); break;
    case 2: c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(af, bf, c, 0, 0, 2, sa, 0, 0x7F7F7F7F); break;
    case 3: c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(af, bf, c, 0, 0, 3, sa, 0, 0x7F7F7F7F); break;
    }
    for (int r = 0; r < 16; ++r) {
        int ci = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5), cj = l & 31;
        C[ci * 32 + cj] = c[r];
    }
}

int main() {
    float* dC; CHECK(hipMalloc(&dC, 32 * 32 * 4));
    std::vector<float> C(32 * 32);
    // part 1: single lane L gets byte0=128 (2.0), opsel 0
    for (int L : {0, 1, 5, 31, 32, 40, 63}) {
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dC, L, 0x7F7F7F80, 0);
        CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
        printf("L=%2d opsel0 changed:", L);
        int n = 0;
        for (int i = 0; i < 32 && n < 40; ++i)
            for (int j = 0; j < 32 && n < 40; ++j)
                if (C[i * 32 + j] != 64.f) { printf(" (%d,%d)=%g", i, j, C[i * 32 + j]); ++n; }
        printf("%s\n", n ? "" : " none");
    }
    // part 2: lane 0, scale word with distinct bytes: b0=127,b1=128,b2=129,b3=130, opsel sweep
    for (int op = 0; op < 4; ++op) {
        hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dC, 0, (130<<24)|(129<<16)|(128<<8)|127, op);
        CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
        printf("L=0 opsel%d: C00=%g C(8,0)=%g\n", op, C[0], C[8 * 32 + 0]);
    }
    return 0;
}
