// cvt_pk_fp8_f32 byte-order probe
#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include <cstdio>
__global__ void k(unsigned* o) {
    int u = __builtin_amdgcn_cvt_pk_fp8_f32(1.0f, 2.0f, 0, false);
    u = __builtin_amdgcn_cvt_pk_fp8_f32(3.0f, 4.0f, u, true);
    o[0] = (unsigned)u;
}
int main() {
    unsigned* d; hipMalloc(&d, 4);
    hipLaunchKernelGGL(k, dim3(1), dim3(1), 0, 0, d);
    unsigned u; hipMemcpy(&u, d, 4, hipMemcpyDeviceToHost);
    printf("packed = %08x\n", u);
    for (int i = 0; i < 4; ++i) {
        __hip_fp8_e4m3 f; f.__x = (u >> (8 * i)) & 0xff;
        printf("byte %d = %g\n", i, (float)f);
    }
    return 0;
}
