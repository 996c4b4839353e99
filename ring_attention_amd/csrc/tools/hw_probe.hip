// gfx950 hardware-semantics probe for the round-2 forward-kernel rebuild.
//
// Verifies, on real MI355X hardware (no emulation; run via gpurun):
//   1. ds_read_b64_tr_b16 lane->element mapping (no builtin exists; the
//      kernel will use inline asm, so the exact transpose semantics must be
//      measured, not assumed).  Guide formula under test: lane l, elem j
//      reads lds_bf16[(l&15) + j*16 + (l>>4)*64] per 16-lane group.
//   2. __builtin_amdgcn_global_load_lds size=16: destination is
//      wave-uniform-base + lane*16 (lane-linear), source address per-lane.
//   3. inline-asm `buffer_load_dwordx4 ... offen lds` with M0 = LDS base —
//      the LDS-DMA form the pwg4x64 structure uses (SRD in SGPRs).
//
// Build: hipcc --offload-arch=gfx950 -O2 hw_probe.hip -o hw_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); return 1; } } while (0)

typedef unsigned int u32;
typedef unsigned long long u64;

// ---------------------------------------------------------------------------
// 1. ds_read_b64_tr_b16: fill LDS bf16[i] = i, issue tr read at per-lane
//    address patterns, dump the 4 u16 each lane receives.
// ---------------------------------------------------------------------------
__global__ void probe_tr16(unsigned short* out, int pattern) {
    __shared__ unsigned short lds[2048];
    int t = threadIdx.x;
    for (int i = t; i < 2048; i += 64) lds[i] = (unsigned short)i;
    __syncthreads();
    // ds_* operands are raw LDS byte offsets: extract the array's base via an
    // explicit addrspacecast (generic -> AS3), not a flat-address cast
    unsigned base = (unsigned)(uintptr_t)(__attribute__((address_space(3))) unsigned short*)lds;
    unsigned addr;
    switch (pattern) {
        case 0: addr = t * 8; break;              // contiguous 8B per lane
        case 1: addr = (t & 15) * 8; break;       // repeat per 16-lane group
        case 2: addr = 0; break;                  // uniform
        case 3: addr = t * 8 + 256; break;        // offset tile
        default: addr = t * 8; break;
    }
    addr += base;
    u64 r;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(r) : "v"(addr) : "memory");
    __builtin_amdgcn_sched_barrier(0);
    unsigned short* o = out + t * 4;
    o[0] = (unsigned short)(r & 0xffff);
    o[1] = (unsigned short)((r >> 16) & 0xffff);
    o[2] = (unsigned short)((r >> 32) & 0xffff);
    o[3] = (unsigned short)((r >> 48) & 0xffff);
}

// ---------------------------------------------------------------------------
// 2. global_load_lds width 16: g[i] = i (u32); one wave issues a single DMA
//    with per-lane source g + lane*4 u32s; read LDS back.
// ---------------------------------------------------------------------------
__global__ void probe_glds(const u32* g, u32* out) {
    __shared__ u32 lds[512];
    int t = threadIdx.x;
    for (int i = t; i < 512; i += 64) lds[i] = 0xdeadbeef;
    __syncthreads();
    if (t < 64) {
        const u32* src = g + t * 4;  // 16B per lane
        __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) u32*)src,
                                         (__attribute__((address_space(3))) u32*)lds, 16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    for (int i = t; i < 512; i += 64) out[i] = lds[i];
}

// ---------------------------------------------------------------------------
// 3. buffer_load_dwordx4 ... lds with M0: same expectation as (2).
//    SRD built from kernarg pointer (wave-uniform); voffset per-lane.
// ---------------------------------------------------------------------------
__global__ void probe_bufl_lds(const u32* g, u32* out) {
    __shared__ u32 lds[512];
    int t = threadIdx.x;
    for (int i = t; i < 512; i += 64) lds[i] = 0xdeadbeef;
    __syncthreads();
    if (t < 64) {
        auto rsrc = __builtin_amdgcn_make_buffer_rsrc((void*)g, 0, 64 * 16, 0x00020000);
        u32 voff = t * 16;
        u32 lds_base = 1024;           // byte offset into LDS: land in lds[256..]
        unsigned keep;
        asm volatile(
            "s_mov_b32 %0, m0\n\t"
            "s_mov_b32 m0, %3\n\t"
            "s_nop 0\n\t"
            "buffer_load_dwordx4 %1, %2, 0 offen lds\n\t"
            "s_mov_b32 m0, %0"
            : "=&s"(keep)
            : "v"(voff), "s"(rsrc), "s"(lds_base)
            : "memory");
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    for (int i = t; i < 512; i += 64) out[i] = lds[i];
}

int main() {
    unsigned short* dus;
    u32 *dg, *dout;
    CHECK(hipMalloc(&dus, 64 * 4 * 2));
    CHECK(hipMalloc(&dg, 4096));
    CHECK(hipMalloc(&dout, 4096));
    std::vector<u32> g(1024);
    for (int i = 0; i < 1024; ++i) g[i] = i;
    CHECK(hipMemcpy(dg, g.data(), 4096, hipMemcpyHostToDevice));

    std::vector<unsigned short> us(256);
    for (int pat = 0; pat < 4; ++pat) {
        hipLaunchKernelGGL(probe_tr16, dim3(1), dim3(64), 0, 0, dus, pat);
        CHECK(hipDeviceSynchronize());
        CHECK(hipMemcpy(us.data(), dus, 512, hipMemcpyDeviceToHost));
        printf("tr16 pattern %d:\n", pat);
        for (int l = 0; l < 64; ++l)
            printf("  l=%2d: %4d %4d %4d %4d\n", l, us[l*4], us[l*4+1], us[l*4+2], us[l*4+3]);
        // check the guide formula for pattern 0: elem j = (l&15) + j*16 + (l>>4)*64
        if (pat == 0) {
            int bad = 0;
            for (int l = 0; l < 64; ++l)
                for (int j = 0; j < 4; ++j)
                    if (us[l*4+j] != (l & 15) + j * 16 + (l >> 4) * 64) ++bad;
            printf("tr16 guide-formula (pattern 0): %s (%d mismatches)\n", bad ? "WRONG" : "OK", bad);
        }
    }

    std::vector<u32> o(512);
    hipLaunchKernelGGL(probe_glds, dim3(1), dim3(64), 0, 0, dg, dout);
    CHECK(hipDeviceSynchronize());
    CHECK(hipMemcpy(o.data(), dout, 2048, hipMemcpyDeviceToHost));
    int bad = 0;
    for (int i = 0; i < 256; ++i) if (o[i] != (u32)i) ++bad;
    printf("global_load_lds x16 lane-linear dest: %s (%d mismatches; lds[0..3]=%u %u %u %u)\n",
           bad ? "WRONG" : "OK", bad, o[0], o[1], o[2], o[3]);

    hipLaunchKernelGGL(probe_bufl_lds, dim3(1), dim3(64), 0, 0, dg, dout);
    CHECK(hipDeviceSynchronize());
    CHECK(hipMemcpy(o.data(), dout, 2048, hipMemcpyDeviceToHost));
    bad = 0;
    for (int i = 0; i < 256; ++i) if (o[256 + i] != (u32)i) ++bad;
    printf("buffer_load_dwordx4 lds (M0=1024) dest: %s (%d mismatches; lds[256..259]=%u %u %u %u)\n",
           bad ? "WRONG" : "OK", bad, o[256], o[257], o[258], o[259]);
    return 0;
}
