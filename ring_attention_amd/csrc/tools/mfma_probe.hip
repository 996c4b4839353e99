// MFMA fragment-layout probe for gfx950 (CDNA4).
//
// Verifies the assumed lane->element maps for the bf16 MFMA shapes this
// framework's flash-attention kernels use, with ASYMMETRIC operand fills so a
// transposed/swapped map cannot pass by accident:
//
//   v_mfma_f32_32x32x16_bf16 (D[32][32] += A[32][16] B[16][32]):
//     A: lane l, reg r -> (i = l&31,                k = 8*(l>>5) + r)
//     B: lane l, reg r -> (k = 8*(l>>5) + r,        j = l&31)
//     C: lane l, reg r -> (i = (r&3)+8*(r>>2)+4*(l>>5), j = l&31)
//   v_mfma_f32_16x16x32_bf16 (D[16][16] += A[16][32] B[32][16]):
//     A: lane l, reg r -> (i = l&15,                k = 8*(l>>4) + r)
//     B: lane l, reg r -> (k = 8*(l>>4) + r,        j = l&15)
//     C: lane l, reg r -> (i = (l>>4)*4 + r,        j = l&15)
//
// Also prints v_permlane32_swap semantics.
//
// Build: hipcc --offload-arch=gfx950 -O2 mfma_probe.hip -o mfma_probe

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstring>
#include <cmath>
#include <vector>

typedef __bf16 bf16_t;
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef unsigned int u32x2 __attribute__((ext_vector_type(2)));

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

__global__ void probe32(const float* A, const float* B, float* C) {
    int l = threadIdx.x & 63;
    bf16x8 af, bf;
    for (int r = 0; r < 8; ++r) {
        int ai = l & 31, ak = 8 * (l >> 5) + r;
        int bk = 8 * (l >> 5) + r, bj = l & 31;
        af[r] = (bf16_t)A[ai * 16 + ak];
        bf[r] = (bf16_t)B[bk * 32 + bj];
    }
    f32x16 acc = {};
    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
    for (int r = 0; r < 16; ++r) C[l * 16 + r] = acc[r];
}

__global__ void probe16(const float* A, const float* B, float* C) {
    int l = threadIdx.x & 63;
    bf16x8 af, bf;
    for (int r = 0; r < 8; ++r) {
        int ai = l & 15, ak = 8 * (l >> 4) + r;
        int bk = 8 * (l >> 4) + r, bj = l & 15;
        af[r] = (bf16_t)A[ai * 32 + ak];
        bf[r] = (bf16_t)B[bk * 16 + bj];
    }
    f32x4 acc = {};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
    for (int r = 0; r < 4; ++r) C[l * 4 + r] = acc[r];
}

__global__ void probe_permlane(float* out) {
    int l = threadIdx.x & 63;
    unsigned x = l, y = 1000 + l;
    u32x2 r = __builtin_amdgcn_permlane32_swap(x, y, false, false);
    out[l] = (float)r[0];
    out[64 + l] = (float)r[1];
}

static float tobf(float x) {  // round-trip through bf16 like the device does
    union { float f; unsigned u; } c;
    c.f = x;
    c.u = (c.u + 0x8000u) & 0xffff0000u;
    return c.f;
}

int main() {
    // asymmetric, exactly-representable-in-bf16 fills
    std::vector<float> A32(32 * 16), B32(16 * 32), A16(16 * 32), B16(32 * 16);
    for (int i = 0; i < 32; ++i) for (int k = 0; k < 16; ++k) A32[i * 16 + k] = tobf((i * 2 + k) % 13 - 6);
    for (int k = 0; k < 16; ++k) for (int j = 0; j < 32; ++j) B32[k * 32 + j] = tobf((k * 5 + j * 3) % 11 - 5);
    for (int i = 0; i < 16; ++i) for (int k = 0; k < 32; ++k) A16[i * 32 + k] = tobf((i * 3 + k) % 13 - 6);
    for (int k = 0; k < 32; ++k) for (int j = 0; j < 16; ++j) B16[k * 16 + j] = tobf((k * 7 + j * 2) % 11 - 5);

    float *dA, *dB, *dC;
    CHECK(hipMalloc(&dA, 4096 * 4)); CHECK(hipMalloc(&dB, 4096 * 4)); CHECK(hipMalloc(&dC, 4096 * 4));

    // --- 32x32x16 ---
    CHECK(hipMemcpy(dA, A32.data(), A32.size() * 4, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dB, B32.data(), B32.size() * 4, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(probe32, dim3(1), dim3(64), 0, 0, dA, dB, dC);
    CHECK(hipDeviceSynchronize());
    std::vector<float> C(64 * 16);
    CHECK(hipMemcpy(C.data(), dC, C.size() * 4, hipMemcpyDeviceToHost));
    int bad = 0;
    for (int l = 0; l < 64; ++l) for (int r = 0; r < 16; ++r) {
        int i = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5), j = l & 31;
        float ref = 0;
        for (int k = 0; k < 16; ++k) ref += A32[i * 16 + k] * B32[k * 32 + j];
        if (fabsf(C[l * 16 + r] - ref) > 1e-3f * (1 + fabsf(ref))) {
            if (bad < 8) printf("32x32 MISMATCH l=%d r=%d got %f want %f\n", l, r, C[l * 16 + r], ref);
            ++bad;
        }
    }
    printf("mfma_f32_32x32x16_bf16 layout: %s (%d mismatches)\n", bad ? "WRONG" : "OK", bad);
    if (bad) {  // dump for manual decode
        printf("dump l r val:\n");
        for (int l = 0; l < 64; ++l) for (int r = 0; r < 16; ++r)
            printf("%d %d %.1f\n", l, r, C[l * 16 + r]);
    }

    // --- 16x16x32 ---
    CHECK(hipMemcpy(dA, A16.data(), A16.size() * 4, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dB, B16.data(), B16.size() * 4, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(probe16, dim3(1), dim3(64), 0, 0, dA, dB, dC);
    CHECK(hipDeviceSynchronize());
    CHECK(hipMemcpy(C.data(), dC, 64 * 4 * 4, hipMemcpyDeviceToHost));
    bad = 0;
    for (int l = 0; l < 64; ++l) for (int r = 0; r < 4; ++r) {
        int i = (l >> 4) * 4 + r, j = l & 15;
        float ref = 0;
        for (int k = 0; k < 32; ++k) ref += A16[i * 32 + k] * B16[k * 16 + j];
        if (fabsf(C[l * 4 + r] - ref) > 1e-3f * (1 + fabsf(ref))) {
            if (bad < 8) printf("16x16 MISMATCH l=%d r=%d got %f want %f\n", l, r, C[l * 4 + r], ref);
            ++bad;
        }
    }
    printf("mfma_f32_16x16x32_bf16 layout: %s (%d mismatches)\n", bad ? "WRONG" : "OK", bad);

    // --- permlane32_swap ---
    hipLaunchKernelGGL(probe_permlane, dim3(1), dim3(64), 0, 0, dC);
    CHECK(hipDeviceSynchronize());
    CHECK(hipMemcpy(C.data(), dC, 128 * 4, hipMemcpyDeviceToHost));
    bool ok = true;
    for (int l = 0; l < 64; ++l) {
        float r0 = C[l], r1 = C[64 + l];
        float e0 = l < 32 ? l : 1000 + (l - 32);       // vdst: hi lanes take src's lo
        float e1 = l < 32 ? 32 + l : 1000 + l;          // src:  lo lanes take vdst's hi
        if (r0 != e0 || r1 != e1) { ok = false; printf("permlane l=%d r0=%.0f r1=%.0f\n", l, r0, r1); }
    }
    printf("permlane32_swap semantics: %s\n", ok ? "OK (r0: hi<-src.lo, r1: lo<-vdst.hi)" : "see dump above");
    return 0;
}
