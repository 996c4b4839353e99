// MX-FP8 mfma_scale fragment-layout probe for gfx950 (CDNA4).
//
// Verifies the hypothesized lane->element maps for
// v_mfma_scale_f32_32x32x64_f8f6f4 (D[32][32] += A[32][64] B[64][32],
// A/B in OCP e4m3, per-32-element-block e8m0 scales):
//
//   A: lane l, byte r (r=0..31) -> (i = l&31, k = 32*(l>>5) + r)
//   B: lane l, byte r          -> (k = 32*(l>>5) + r, j = l&31)
//   C: lane l, reg r (r=0..15) -> (i = (r&3)+8*(r>>2)+4*(l>>5), j = l&31)
//     (C/D layout is dtype-independent on gfx950 — same as the bf16 shape)
//
// and the scale-operand semantics: each lane contributes ONE e8m0 scale for
// its (row, k-block) — byte `opsel` of the 32-bit scale_a/scale_b operand;
// value 2^(x-127).
//
// Build: hipcc --offload-arch=gfx950 -O2 fp8_probe.hip -o fp8_probe

#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include <cstdio>
#include <cmath>
#include <vector>

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef int i32x8 __attribute__((ext_vector_type(8)));

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

// one wave; A,B given as fp8 BYTES laid out row-major [32][64] / [64][32]
__global__ void probe_fp8(const unsigned char* A, const unsigned char* B,
                          float* C, unsigned sa_byte, unsigned sb_byte) {
    int l = threadIdx.x & 63;
    union { i32x8 v; unsigned char b[32]; } af, bf;
    for (int r = 0; r < 32; ++r) {
        int ai = l & 31, ak = 32 * (l >> 5) + r;
        int bk = 32 * (l >> 5) + r, bj = l & 31;
        af.b[r] = A[ai * 64 + ak];
        bf.b[r] = B[bk * 32 + bj];
    }
    f32x16 c = {};
    // opsel 0: scale in byte 0.  sa/sb passed per-lane identical here; the
    // per-(row,kblock) granularity test varies them by lane below.
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            af.v, bf.v, c, 0, 0, 0, (int)sa_byte, 0, (int)sb_byte);
    for (int r = 0; r < 16; ++r) {
        int ci = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5), cj = l & 31;
        C[ci * 32 + cj] = c[r];
    }
}

// scale granularity: lanes in k-block 1 (l>=32) get scale 2.0 on A
__global__ void probe_scale(const unsigned char* A, const unsigned char* B,
                            float* C) {
    int l = threadIdx.x & 63;
    union { i32x8 v; unsigned char b[32]; } af, bf;
    for (int r = 0; r < 32; ++r) {
        int ai = l & 31, ak = 32 * (l >> 5) + r;
        int bk = 32 * (l >> 5) + r, bj = l & 31;
        af.b[r] = A[ai * 64 + ak];
        bf.b[r] = B[bk * 32 + bj];
    }
    f32x16 c = {};
    int sa = (l >> 5) ? 128 : 127;     // k-block 1 lanes: 2^1
    c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
            af.v, bf.v, c, 0, 0, 0, sa, 0, 127);
    for (int r = 0; r < 16; ++r) {
        int ci = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5), cj = l & 31;
        C[ci * 32 + cj] = c[r];
    }
}

int main() {
    std::vector<unsigned char> A(32 * 64), B(64 * 32);
    std::vector<float> Af(32 * 64), Bf(64 * 32);
    unsigned seed = 12345;
    auto rnd = [&]() { seed = seed * 1664525u + 1013904223u; return ((seed >> 16) & 0xff) / 64.f - 2.f; };
    for (int i = 0; i < 32 * 64; ++i) {
        __hip_fp8_e4m3 q(rnd());
        A[i] = q.__x; Af[i] = (float)q;
    }
    for (int i = 0; i < 64 * 32; ++i) {
        __hip_fp8_e4m3 q(rnd());
        B[i] = q.__x; Bf[i] = (float)q;
    }
    std::vector<float> ref(32 * 32, 0.f), ref2(32 * 32, 0.f);
    for (int i = 0; i < 32; ++i)
        for (int j = 0; j < 32; ++j) {
            double s = 0, s2 = 0;
            for (int k = 0; k < 64; ++k) {
                double t = (double)Af[i * 64 + k] * Bf[k * 32 + j];
                s += t;
                s2 += (k >= 32 ? 2.0 : 1.0) * t;
            }
            ref[i * 32 + j] = (float)s;
            ref2[i * 32 + j] = (float)s2;
        }

    unsigned char *dA, *dB; float* dC;
    CHECK(hipMalloc(&dA, 32 * 64)); CHECK(hipMalloc(&dB, 64 * 32));
    CHECK(hipMalloc(&dC, 32 * 32 * 4));
    CHECK(hipMemcpy(dA, A.data(), 32 * 64, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dB, B.data(), 64 * 32, hipMemcpyHostToDevice));

    std::vector<float> C(32 * 32);
    hipLaunchKernelGGL(probe_fp8, dim3(1), dim3(64), 0, 0, dA, dB, dC, 127u, 127u);
    CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
    float err = 0; for (int i = 0; i < 32 * 32; ++i) err = fmaxf(err, fabsf(C[i] - ref[i]));
    printf("fp8 32x32x64 unit-scale max err: %g  (c[0]=%g ref=%g)\n", err, C[0], ref[0]);

    hipLaunchKernelGGL(probe_scale, dim3(1), dim3(64), 0, 0, dA, dB, dC);
    CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
    float err2 = 0; for (int i = 0; i < 32 * 32; ++i) err2 = fmaxf(err2, fabsf(C[i] - ref2[i]));
    printf("fp8 scale-granularity (A kblock1 x2) max err: %g\n", err2);

    // e8m0 sanity: scale 2^-1 on B
    hipLaunchKernelGGL(probe_fp8, dim3(1), dim3(64), 0, 0, dA, dB, dC, 127u, 126u);
    CHECK(hipMemcpy(C.data(), dC, 32 * 32 * 4, hipMemcpyDeviceToHost));
    float err3 = 0; for (int i = 0; i < 32 * 32; ++i) err3 = fmaxf(err3, fabsf(C[i] - 0.5f * ref[i]));
    printf("fp8 B-scale 2^-1 max err: %g\n", err3);
    return 0;
}
