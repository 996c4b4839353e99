// CDNA4 (gfx950) flash-attention BACKWARD kernels (recompute-based).
//
// Capability counterpart of the reference's Triton _bwd_kernel
// (/root/reference/ring_attention_pytorch/triton_flash_attn.py:509-1128),
// re-designed for wave64/MFMA; no code ported.  TWO kernels, each with the
// contraction laid out so every accumulator lives in registers and no
// cross-workgroup atomics exist at all:
//
//  * attn_bwd_dq_kernel — ROW-parallel (mirrors the forward kernel): each
//    wave owns 32 q rows, Q^T and dO^T fragments live in registers, K/V/K^T
//    tiles stream through LDS; ds^T is built in-register (lane = q) and
//    dq^T accumulates in registers; epilogue does ONE plain fp32 += per
//    element (each q row is owned by exactly one workgroup).
//  * attn_bwd_dkv_kernel — COLUMN-parallel: each wave owns 32 kv rows with
//    K/V fragments in registers across the whole q loop; Q/dO stream through
//    LDS (row-major + pair-staged transpose); p/ds fragments are built
//    in-register (lane = kv); dk/dv accumulate in registers and are written
//    once.  GQA: the group's query heads are iterated with K/V resident, so
//    dk/dv need no cross-head reduction.
//
// The reference needed tl.debug_barrier() workarounds and an atomic-vs-
// serialized dq autotune choice (triton_flash_attn.py:648-776); this design
// removes the hazards structurally (profiled on MI355X: the single-kernel
// atomic variant was LDS/atomic-bound at 35% LDS-array cycles).
//
// causality/striping/lookback use the same (diag, win) reduction as forward;
// softclamp backward applies the dtanh factor exactly as the oracle.
//
// Output layouts:
//   dq: fp32 (B, Nq, H, D)   — plain accumulate (+=) across hops
//   dk: fp32 (B, HK, Nk, D)  — plain writes (one WG owns each row)
//   dv: fp32 (B, HK, D, Nk)  — transposed scratch, coalesced from dv^T regs

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#include "attn_common.h"

namespace ring_attn {

template <int CH>
__device__ __forceinline__ int bswz(int row, int chunk) {
    return chunk ^ (row & (CH < 8 ? CH - 1 : 7));   // see swz in attn_fwd.hip
}

__device__ __forceinline__ float bfast_tanhf(float x) {
    float e = __builtin_amdgcn_exp2f(x * 2.885390081777927f);
    return 1.f - 2.f * __builtin_amdgcn_rcpf(e + 1.f);
}

// ---------------------------------------------------------------------------
// shared staging helpers (512-thread workgroups)
// ---------------------------------------------------------------------------

// [rows][D] bf16 row-major, 16B chunks XOR-swizzled within the row
template <int D, int ROWS>
__device__ __forceinline__ void stage_rowmajor(
    const __bf16* gbase, long row0, long rowmax, long row_stride,
    __bf16* lds_rm, int tid) {
    constexpr int CH = D * 2 / 16;
    for (int c = tid; c < ROWS * CH; c += 512) {
        int row = c / CH, ch = c % CH;
        long gr = row0 + row;
        uint4 val = (gr <= rowmax) ? *(const uint4*)(gbase + gr * row_stride + ch * 8)
                                   : uint4{0, 0, 0, 0};
        *(uint4*)(lds_rm + row * D + bswz<D / 8>(row, ch) * 8) = val;
    }
}

// [D][ROWS] bf16 transposed image via row-pair loads -> b32 writes.
// Swizzle: 16B chunk of the d-row XORed with (d & TM).
template <int D, int ROWS>
__device__ __forceinline__ void stage_transposed(
    const __bf16* gbase, long row0, long rowmax, long row_stride,
    __bf16* lds_t, int tid) {
    constexpr int TM = (ROWS / 8 - 1) < 7 ? (ROWS / 8 - 1) : 7;
    constexpr int PAIRS = (ROWS / 2) * (D / 8);
    for (int c = tid; c < PAIRS; c += 512) {
        int jp = c % (ROWS / 2);
        int d0 = (c / (ROWS / 2)) * 8;
        long ja = row0 + jp * 2, jb = ja + 1;
        bf16x8 va = (ja <= rowmax) ? *(const bf16x8*)(gbase + ja * row_stride + d0) : bf16x8{};
        bf16x8 vb = (jb <= rowmax) ? *(const bf16x8*)(gbase + jb * row_stride + d0) : bf16x8{};
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
            int d = d0 + e;
            int byte_off = d * ROWS * 2 + ((jp * 4) ^ ((d & TM) << 4));
            __bf16 pair[2] = {va[e], vb[e]};
            *(uint32_t*)((char*)lds_t + byte_off) = *(uint32_t*)pair;
        }
    }
}

// ---------------------------------------------------------------------------
// dq kernel: row-parallel, forward-like (same pipeline as attn_fwd_kernel:
// double-buffered LDS, one barrier per kv tile, running-pointer staging)
// ---------------------------------------------------------------------------
static constexpr int DQ_WAVES = 8;
static constexpr int DQ_QROWS_WG = DQ_WAVES * 32;     // 256
// The kt image is gone (K^T fragments come from the row-major k image via
// ds_read_b64_tr_b16 — the v2 forward's V recipe, hardware-verified in
// tools/hw_probe.hip), which frees a third of the LDS and the transpose
// staging.  d128 stays at KVB=64: the KVB=128 variant its freed LDS allows
// spills 280-336 B/lane into the hot loop and measured SLOWER (174 vs 204
// TF headline) — the round-1 "fwd KVBLK=128 at d128" trap again.
template <class F>
__device__ __attribute__((noinline)) void dkv_noinline_call(F&& f) { f(); }

template <int D> constexpr int dq_kvblk() { return D == 64 ? 128 : 64; }

template <int D>
struct DqLds {
    static constexpr int KVB = dq_kvblk<D>();
    __align__(16) __bf16 k[2][KVB * D];    // [kv][d] swizzled
    __align__(16) __bf16 v[2][KVB * D];    // [kv][d] swizzled
    unsigned char kmask[2][KVB];
};

template <int D, bool SOFTCLAMP, bool PAIRED, bool BIAS = false>
__global__ __launch_bounds__(512, 2)   // 8-wave WGs need exactly 2 waves/SIMD:
void attn_bwd_dq_kernel(BwdParams p) { // cap 256 VGPR, stop the 220 B/thread
                                       // scratch spills seen at the 128 cap
    constexpr int DBLK = D / 32;
    constexpr int KSTEPS = D / 16;
    constexpr int DQ_KVBLK = dq_kvblk<D>();
    constexpr int DQ_NBLK = DQ_KVBLK / 32;

    __shared__ DqLds<D> lds;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l31 = lane & 31;
    const int lhi = lane >> 5;

    const int bh = blockIdx.y;
    const int b = bh / p.h;
    const int h = bh % p.h;
    const int hk = h % p.hk;   // reference tile GQA pairing

    // causal pairing: WG x runs q-tiles (x, T-1-x) — uniform per-WG work
    const int n_pit = PAIRED
        ? (p.paired - 1 - (int)blockIdx.x == (int)blockIdx.x ? 1 : 2) : 1;
    for (int pit = 0; pit < n_pit; ++pit) {
    const int qtile = PAIRED
        ? (pit == 0 ? (int)blockIdx.x : p.paired - 1 - (int)blockIdx.x)
        : (p.desc ? p.desc[(long)blockIdx.x * 3] : (int)blockIdx.x);
    if (PAIRED && pit == 1) {
        __syncthreads();                       // LDS handoff between tiles
    }

    auto dq_body = [&]() {
    const BwdParams P = p;   // register-local copy (see dkv_noinline_call)
    const long i = (long)qtile * DQ_QROWS_WG + wid * 32 + l31;
    const bool row_valid = i < P.nq;
    const long ic = row_valid ? i : 0;

    // Q^T and dO^T fragments in registers (B-operand layout, lane = q)
    const __bf16* qbase = (const __bf16*)P.q + ((long)b * P.nq + ic) * P.h * D + (long)h * D;
    const __bf16* dobase = (const __bf16*)P.dout + ((long)b * P.nq + ic) * P.h * D + (long)h * D;
    bf16x8 qf[KSTEPS], dof[KSTEPS];
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
        qf[ks] = *(const bf16x8*)(qbase + ks * 16 + lhi * 8);
        dof[ks] = *(const bf16x8*)(dobase + ks * 16 + lhi * 8);
    }
    const float lse_i = P.lse[((long)b * P.h + h) * P.nq + ic] * 1.4426950408889634f;
    const float delta_i = P.delta[((long)b * P.h + h) * P.nq + ic];

    f32x16 dq_acc[DBLK];
    #pragma unroll
    for (int db = 0; db < DBLK; ++db) dq_acc[db] = f32x16{};

    const long wg_i_min = (long)qtile * DQ_QROWS_WG;
    const long wg_i_max = min((long)(qtile + 1) * DQ_QROWS_WG, P.nq) - 1;
    const long wg_q_min = wg_i_min * P.q_stride + P.diag;
    const long wg_q_max = wg_i_max * P.q_stride + P.diag;
    const long qpos_i = i * P.q_stride + P.diag;
    const int num_kv_tiles = (int)((P.nk + DQ_KVBLK - 1) / DQ_KVBLK);

    int t_lo = 0, t_hi = num_kv_tiles;
    if (P.causal)
        t_hi = wg_q_max < 0 ? 0 : min((long)num_kv_tiles, wg_q_max / DQ_KVBLK + 1);
    if (P.has_win) {
        long x = wg_q_min - P.win - DQ_KVBLK + 1;
        t_lo = x <= 0 ? 0 : (int)((x + DQ_KVBLK - 1) / DQ_KVBLK);
        if (t_lo > t_hi) t_lo = t_hi;
    }
    if (P.split > 1) {
        // fractional split of this WG's own valid range (see attn_fwd)
        int valid = t_hi - t_lo;
        int per = (valid + P.split - 1) / P.split;
        int base = t_lo;
        t_lo = base + min(valid, (int)(blockIdx.z * per));
        t_hi = base + min(valid, (int)((blockIdx.z + 1) * per));
    }
    if (!PAIRED && P.desc) {     // descriptor mode: exact unit bounds
        t_lo = P.desc[(long)blockIdx.x * 3 + 1];
        t_hi = P.desc[(long)blockIdx.x * 3 + 2];
    }

    // staging: K + V row chunks, K^T pairs, running pointers
    constexpr int CH = D * 2 / 16;
    constexpr int KCHUNKS = DQ_KVBLK * CH;
    constexpr int KREGS = (KCHUNKS + 511) / 512;
    const __bf16* kbase = (const __bf16*)P.k + ((long)b * P.nk) * P.hk * D + (long)hk * D;
    const __bf16* vbase = (const __bf16*)P.v + ((long)b * P.nk) * P.hk * D + (long)hk * D;
    const unsigned char* mbase = P.kmask ? (const unsigned char*)P.kmask + (long)b * P.nk : nullptr;

    const long kv_row_stride = (long)P.hk * D;
    const long tile_stride = DQ_KVBLK * kv_row_stride;
    const __bf16* kptr = kbase + (long)t_lo * tile_stride
        + (tid / CH) * kv_row_stride + (tid % CH) * 8;
    const __bf16* vptr = vbase + (long)t_lo * tile_stride
        + (tid / CH) * kv_row_stride + (tid % CH) * 8;
    long j0_next = (long)t_lo * DQ_KVBLK;

    uint4 kst[KREGS], vst[KREGS];
    unsigned char mst = 1;

    auto load_tile = [&]() {
        const long j0 = j0_next;
        const long jmax = min(j0 + DQ_KVBLK, P.nk) - 1;
        const bool full = jmax - j0 == DQ_KVBLK - 1;
        #pragma unroll
        for (int r = 0; r < KREGS; ++r) {
            int c = tid + r * 512;
            if (c < KCHUNKS) {
                long off = (long)(r * (512 / CH)) * kv_row_stride;
                bool okr = full || (j0 + c / CH) <= jmax;
                kst[r] = okr ? *(const uint4*)(kptr + off) : uint4{0, 0, 0, 0};
                vst[r] = okr ? *(const uint4*)(vptr + off) : uint4{0, 0, 0, 0};
            }
        }
        if (mbase && tid < DQ_KVBLK)
            mst = (j0 + tid <= jmax) ? mbase[j0 + tid] : 0;
        kptr += tile_stride; vptr += tile_stride;
        j0_next += DQ_KVBLK;
    };

    auto write_tile = [&](int par) {
        #pragma unroll
        for (int r = 0; r < KREGS; ++r) {
            int c = tid + r * 512;
            if (c < KCHUNKS) {
                int row = c / CH, ch = c % CH;
                *(uint4*)(lds.k[par] + row * D + bswz<D / 8>(row, ch) * 8) = kst[r];
                *(uint4*)(lds.v[par] + row * D + bswz<D / 8>(row, ch) * 8) = vst[r];
            }
        }
        if (mbase && tid < DQ_KVBLK) lds.kmask[par][tid] = mst;
    };

    const float scale2 = P.scale * 1.4426950408889634f;   // exp2 domain
    if (t_lo < t_hi) {
        load_tile();
        write_tile(t_lo & 1);
        if (t_lo + 1 < t_hi) load_tile();
    }

    for (int t = t_lo; t < t_hi; ++t) {
        const int par = t & 1;
        const long j0 = (long)t * DQ_KVBLK;
        const long jmax = min(j0 + DQ_KVBLK, P.nk) - 1;
        const bool full_tile = !BIAS &&
            (jmax - j0 == DQ_KVBLK - 1) &&
            (!P.causal || jmax <= wg_q_min) &&
            (!P.has_win || (wg_q_max - j0) <= P.win) &&
            !P.kmask;

        __syncthreads();

        // s^T and dp^T computed PER 32-row BLOCK and packed immediately —
        // keeping all NBLK blocks' accumulators live (128 VGPRs) forced
        // scratch spills at the 256-register budget
        uint32_t pk[DQ_NBLK * 8];
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kb = 0; kb < DQ_NBLK; ++kb) {
            f32x16 s = f32x16{}, dp = f32x16{};
            int krow = kb * 32 + l31;
            #pragma unroll
            for (int ks = 0; ks < KSTEPS; ++ks) {
                int chunk = ks * 2 + lhi;
                bf16x8 kfr = *(const bf16x8*)(lds.k[par] + krow * D + bswz<D / 8>(krow, chunk) * 8);
                bf16x8 vfr = *(const bf16x8*)(lds.v[par] + krow * D + bswz<D / 8>(krow, chunk) * 8);
                s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfr, qf[ks], s, 0, 0, 0);
                dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfr, dof[ks], dp, 0, 0, 0);
            }
            if (full_tile && row_valid) {
                #pragma unroll
                for (int x2 = 0; x2 < 8; ++x2) {
                    float dse[2];
                    #pragma unroll
                    for (int e = 0; e < 2; ++e) {
                        int r = 2 * x2 + e;
                        float x, dtanh = 1.f;
                        if constexpr (SOFTCLAMP) {
                            float inv_v = __builtin_amdgcn_rcpf(P.softclamp_value);
                            float th = bfast_tanhf(s[r] * P.scale * inv_v);
                            x = P.softclamp_value * th * 1.4426950408889634f;
                            dtanh = 1.f - th * th;
                        } else {
                            x = __builtin_fmaf(s[r], scale2, -lse_i);  // fold
                        }
                        float pv = __builtin_amdgcn_exp2f(SOFTCLAMP ? x - lse_i : x);
                        dse[e] = pv * (dp[r] - delta_i) * dtanh * P.scale;
                    }
                    union { __hip_bfloat162 h2; uint32_t u; } cvt;
                    cvt.h2 = __float22bfloat162_rn(float2{dse[0], dse[1]});
                    pk[kb * 8 + x2] = cvt.u;
                }
            } else {
                #pragma unroll
                for (int x2 = 0; x2 < 8; ++x2) {
                    float dse[2];
                    #pragma unroll
                    for (int e = 0; e < 2; ++e) {
                        int r = 2 * x2 + e;
                        long jj = j0 + kb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                        float x, dtanh = 1.f;
                        if constexpr (SOFTCLAMP) {
                            float inv_v = __builtin_amdgcn_rcpf(P.softclamp_value);
                            float th = bfast_tanhf(s[r] * P.scale * inv_v);
                            x = P.softclamp_value * th * 1.4426950408889634f;
                            dtanh = 1.f - th * th;
                        } else {
                            x = s[r] * scale2;
                        }
                        bool ok = row_valid && jj <= jmax;
                        if constexpr (BIAS) {
                            if (ok) {
                                const long bi = P.bias_mat
                                    ? (((long)b * P.h + h) * P.nq + ic) * P.nk + jj
                                    : ((long)b * P.h + h) * P.nk + jj;
                                x += P.bias[bi] * 1.4426950408889634f;
                            }
                        }
                        if (P.causal) ok = ok && (jj <= qpos_i);
                        if (P.has_win) ok = ok && (qpos_i - jj <= P.win);
                        if (P.kmask) ok = ok && lds.kmask[par][jj - j0];
                        float pv = ok ? __builtin_amdgcn_exp2f(
                            SOFTCLAMP ? x - lse_i : x - lse_i) : 0.f;
                        dse[e] = pv * (dp[r] - delta_i) * dtanh * P.scale;
                    }
                    union { __hip_bfloat162 h2; uint32_t u; } cvt;
                    cvt.h2 = __float22bfloat162_rn(float2{dse[0], dse[1]});
                    pk[kb * 8 + x2] = cvt.u;
                }
            }
        }
        __builtin_amdgcn_s_setprio(0);

        if (t + 1 < t_hi) write_tile(par ^ 1);
        if (t + 2 < t_hi) load_tile();

        // build B-operand fragments (lane = q, k = kv contiguous): pairs +2
        uint32_t frag[DQ_NBLK * 2][4];
        #pragma unroll
        for (int kb = 0; kb < DQ_NBLK; ++kb)
            #pragma unroll
            for (int half = 0; half < 2; ++half)
                #pragma unroll
                for (int c = 0; c < 2; ++c) {
                    u32x2 r = __builtin_amdgcn_permlane32_swap(
                        pk[kb * 8 + half * 4 + c], pk[kb * 8 + half * 4 + c + 2], false, false);
                    frag[kb * 2 + half][c] = r[0];
                    frag[kb * 2 + half][c + 2] = r[1];
                }

        // dq^T[d][q] += K^T[d][kv] x ds^T[kv][q].  K^T A-fragments come
        // straight from the row-major swizzled k image via
        // ds_read_b64_tr_b16: lane f=l&15 of each 16-lane group fetches
        // K[kvq + (f>>2)][dg + 4*(f&3) ..+4] and the hardware
        // redistribution delivers out[l][j] = K[kvq + j][dg + (l&15)] —
        // i.e. lane l31 holds d column db*32 + l31, 4 kv per read.
        {
            const int f15 = lane & 15;
            const int g16 = (lane >> 4) & 1;
            const unsigned kb_lds = (unsigned)(uintptr_t)
                (__attribute__((address_space(3))) __bf16*)lds.k[par];
            __builtin_amdgcn_s_setprio(1);
            // issue ALL K^T tr-reads up front (asm loads cannot be software-
            // pipelined by the scheduler, so one batched wait beats a wait
            // per d-block), then run the whole MFMA stream
            unsigned long long ktr[DBLK][DQ_NBLK * 2][2];
            #pragma unroll
            for (int db = 0; db < DBLK; ++db) {
                const int dg = db * 32 + g16 * 16 + 4 * (f15 & 3);
                #pragma unroll
                for (int ks = 0; ks < DQ_NBLK * 2; ++ks) {
                    #pragma unroll
                    for (int half = 0; half < 2; ++half) {
                        const int kv = ks * 16 + 8 * lhi + 4 * half + (f15 >> 2);
                        // swizzled element address of (kv, dg): 16B chunk
                        // c = dg/8 XORed with the row's swizzle key
                        const int c = (dg / 8) ^ (kv & (CH < 8 ? CH - 1 : 7));
                        const unsigned a = kb_lds
                            + (unsigned)(kv * (D * 2) + c * 16 + (dg % 8) * 2);
                        asm volatile("ds_read_b64_tr_b16 %0, %1"
                                     : "=&v"(ktr[db][ks][half]) : "v"(a) : "memory");
                    }
                }
            }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            __builtin_amdgcn_sched_barrier(0);
            #pragma unroll
            for (int db = 0; db < DBLK; ++db)
                #pragma unroll
                for (int ks = 0; ks < DQ_NBLK * 2; ++ks) {
                    bf16x8 ktf;
                    *(unsigned long long*)&ktf = ktr[db][ks][0];
                    *((unsigned long long*)&ktf + 1) = ktr[db][ks][1];
                    dq_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        ktf, *(const bf16x8*)frag[ks], dq_acc[db], 0, 0, 0);
                }
            __builtin_amdgcn_s_setprio(0);
        }
    }

    if (!row_valid) return;
    // epilogue: dq (B, Nq, H, D) fp32; unique writer per row unless the kv
    // walk is split across grid.z (then fp32 atomics, contention = split)
    float* dqp = P.dq + ((long)b * P.nq + i) * P.h * D + (long)h * D;
    #pragma unroll
    for (int db = 0; db < DBLK; ++db)
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
            int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
            if (P.split > 1 || P.desc) atomicAdd(dqp + d, dq_acc[db][r]);
            else if (P.accumulate) dqp[d] += dq_acc[db][r];
            else dqp[d] = dq_acc[db][r];
        }
    };
    // measured: the noinline frame REGRESSES d128-PAIRED dq (105 vs 123 TF)
    // unlike fwd/dkv, but the lambda restructure itself (register-local P
    // copy) lifts d128-plain (200 -> 207) — keep the body direct-called.
    dq_body();
    }  // pair loop
}

// ---------------------------------------------------------------------------
// dk/dv kernel: column-parallel, K/V resident in registers
// ---------------------------------------------------------------------------
static constexpr int BWD_WAVES = 8;
static constexpr int KVROWS_WAVE = 32;
static constexpr int KVROWS_WG = BWD_WAVES * KVROWS_WAVE;   // 256

template <int D, int QT>
struct DkvLds {
    // double-buffered q-tile images (one barrier per q tile; staging writes
    // and next-next tile's loads overlap the MFMAs, as in the forward kernel)
    __align__(16) __bf16 q[2][QT * D];      // [q][d]   swizzled rows
    __align__(16) __bf16 qt[2][D * QT];     // [d][q]   pair-staged transpose
    __align__(16) __bf16 do_[2][QT * D];    // [q][d]
    __align__(16) __bf16 dot[2][D * QT];    // [d][q]
    __align__(16) float lse[2][QT];
    __align__(16) float delta[2][QT];
};


template <int D, int QT, bool SOFTCLAMP, bool PAIRED, bool BIAS = false>
__global__ __launch_bounds__(512, 2)   // see dq kernel note (552 B spills)
void attn_bwd_dkv_kernel(BwdParams p) {
    static_assert(D % 32 == 0 && QT % 32 == 0);
    constexpr int DBLK = D / 32;
    constexpr int KSTEPS = D / 16;
    constexpr int QBLKS = QT / 32;
    constexpr int TM = (QT / 8 - 1) < 7 ? (QT / 8 - 1) : 7;

    __shared__ DkvLds<D, QT> lds;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l31 = lane & 31;
    const int lhi = lane >> 5;

    const int bhk = blockIdx.y;
    const int b = bhk / p.hk;
    const int hkh = bhk % p.hk;

    // causal pairing (mirrored): kv-tile x attends T-x q-tiles, so WG x
    // runs kv-tiles (x, T-1-x) for uniform per-WG work
    const int n_pit = PAIRED
        ? (p.paired - 1 - (int)blockIdx.x == (int)blockIdx.x ? 1 : 2) : 1;
    for (int pit = 0; pit < n_pit; ++pit) {
    const int kvtile = PAIRED
        ? (pit == 0 ? (int)blockIdx.x : p.paired - 1 - (int)blockIdx.x)
        : (p.desc ? p.desc[(long)blockIdx.x * 3] : (int)blockIdx.x);
    if (PAIRED && pit == 1) {
        __syncthreads();                       // LDS handoff between tiles
    }

    auto dkv_body = [&]() {
    const BwdParams P = p;   // register-local copy: the noinline frame would
                             // otherwise re-read fields through scratch
    const long j0_wg = (long)kvtile * KVROWS_WG;
    const long jmax = min(j0_wg + KVROWS_WG, P.nk) - 1;
    const long j = j0_wg + wid * KVROWS_WAVE + l31;
    const bool col_valid = j <= jmax;
    const long jc = col_valid ? j : j0_wg;

    const __bf16* kb = (const __bf16*)P.k + ((long)b * P.nk + jc) * P.hk * D + (long)hkh * D;
    const __bf16* vb = (const __bf16*)P.v + ((long)b * P.nk + jc) * P.hk * D + (long)hkh * D;
    bf16x8 kf[KSTEPS], vf[KSTEPS];
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
        kf[ks] = *(const bf16x8*)(kb + ks * 16 + lhi * 8);
        vf[ks] = *(const bf16x8*)(vb + ks * 16 + lhi * 8);
    }
    unsigned char kmask_own = 1;
    if (P.kmask) kmask_own = col_valid ? ((const unsigned char*)P.kmask)[(long)b * P.nk + j] : 0;

    f32x16 dv_acc[DBLK], dk_acc[DBLK];
    #pragma unroll
    for (int db = 0; db < DBLK; ++db) { dv_acc[db] = f32x16{}; dk_acc[db] = f32x16{}; }

    const int num_q_tiles = (int)((P.nq + QT - 1) / QT);

    for (int g = 0; g < P.group; ++g) {
        // group boundary barrier: the previous group's last tile is still
        // being read from LDS by slower waves when this group's prologue
        // write_qtile targets the same buffer (GQA-only race; widest at
        // split>1 where a chunk is a single tile)
        if (g > 0) __syncthreads();
        const int h = hkh + g * P.hk;   // reference tile GQA: group g's q head
        const float* lse_row = P.lse + ((long)b * P.h + h) * P.nq;
        const float* delta_row = P.delta + ((long)b * P.h + h) * P.nq;
        const __bf16* qg = (const __bf16*)P.q + ((long)b * P.nq) * P.h * D + (long)h * D;
        const __bf16* dog = (const __bf16*)P.dout + ((long)b * P.nq) * P.h * D + (long)h * D;

        int t0 = 0, t1 = num_q_tiles;
        if (P.causal) {
            // need qpos(i) >= j0_wg  =>  i >= (j0_wg - diag) / q_stride
            long i_min_needed = (j0_wg - P.diag + P.q_stride - 1) / P.q_stride;
            if (i_min_needed > 0) t0 = (int)(i_min_needed / QT);
        }
        if (P.has_win) {
            // need qpos(i) <= jmax + win  =>  i <= (jmax + win - diag) / q_stride
            long num = jmax + P.win - P.diag;
            long i_max_needed = num < 0 ? -1 : num / P.q_stride;
            if (i_max_needed < (long)num_q_tiles * QT)
                t1 = (int)min((long)num_q_tiles,
                              i_max_needed < 0 ? 0 : i_max_needed / QT + 1);
        }
        if (P.split > 1) {
            // grid.z takes a fractional share of this WG's own valid q walk
            // (a global-range slice is skewed against the causal trapezoid)
            int valid = t1 > t0 ? t1 - t0 : 0;
            int per = (valid + P.split - 1) / P.split;
            int base = t0;
            t0 = base + min(valid, (int)(blockIdx.z * per));
            t1 = base + min(valid, (int)((blockIdx.z + 1) * per));
        }
        if (!PAIRED && P.desc) {   // descriptor mode: exact unit bounds
            t0 = P.desc[(long)blockIdx.x * 3 + 1];
            t1 = P.desc[(long)blockIdx.x * 3 + 2];
        }

        // ---- T14 pipeline: per-thread staging registers
        constexpr int CH = D * 2 / 16;
        constexpr int QCHUNKS = QT * CH;
        constexpr int QREGS = (QCHUNKS + 511) / 512;
        constexpr int TPAIRS = (QT / 2) * (D / 8);
        constexpr int TREGS = (TPAIRS + 511) / 512;
        uint4 qst[QREGS], dost[QREGS];
        bf16x8 qtv_a[TREGS], qtv_b[TREGS], dov_a[TREGS], dov_b[TREGS];
        float lse_st = 0.f, delta_st = 0.f;
        long t_next = t0;

        auto load_qtile = [&]() {
            const long i0 = t_next * QT;
            const long imax_ = min(i0 + QT, P.nq) - 1;
            const bool full = imax_ - i0 == QT - 1;
            #pragma unroll
            for (int r = 0; r < QREGS; ++r) {
                int c = tid + r * 512;
                if (c < QCHUNKS) {
                    long gr = i0 + c / CH;
                    int ch = c % CH;
                    const __bf16* src = qg + gr * P.h * D + ch * 8;
                    const __bf16* srd = dog + gr * P.h * D + ch * 8;
                    bool okr = full || gr <= imax_;
                    qst[r] = okr ? *(const uint4*)src : uint4{0, 0, 0, 0};
                    dost[r] = okr ? *(const uint4*)srd : uint4{0, 0, 0, 0};
                }
            }
            #pragma unroll
            for (int r = 0; r < TREGS; ++r) {
                int c = tid + r * 512;
                if (c < TPAIRS) {
                    int jp = c % (QT / 2);
                    int d0 = (c / (QT / 2)) * 8;
                    long ra = i0 + jp * 2;
                    bool oka = full || ra <= imax_;
                    bool okb = full || ra + 1 <= imax_;
                    const __bf16* qa_ = qg + ra * P.h * D + d0;
                    const __bf16* da_ = dog + ra * P.h * D + d0;
                    qtv_a[r] = oka ? *(const bf16x8*)qa_ : bf16x8{};
                    qtv_b[r] = okb ? *(const bf16x8*)(qa_ + P.h * D) : bf16x8{};
                    dov_a[r] = oka ? *(const bf16x8*)da_ : bf16x8{};
                    dov_b[r] = okb ? *(const bf16x8*)(da_ + P.h * D) : bf16x8{};
                }
            }
            if (tid < QT) {
                long gi = i0 + tid;
                bool okl = gi <= imax_;
                lse_st = okl ? lse_row[gi] : 0.f;
                delta_st = okl ? delta_row[gi] : 0.f;
            }
            ++t_next;
        };

        auto write_qtile = [&](int par) {
            #pragma unroll
            for (int r = 0; r < QREGS; ++r) {
                int c = tid + r * 512;
                if (c < QCHUNKS) {
                    int row = c / CH, ch = c % CH;
                    *(uint4*)(lds.q[par] + row * D + bswz<D / 8>(row, ch) * 8) = qst[r];
                    *(uint4*)(lds.do_[par] + row * D + bswz<D / 8>(row, ch) * 8) = dost[r];
                }
            }
            #pragma unroll
            for (int r = 0; r < TREGS; ++r) {
                int c = tid + r * 512;
                if (c < TPAIRS) {
                    int jp = c % (QT / 2);
                    int d0 = (c / (QT / 2)) * 8;
                    #pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        int dd = d0 + e;
                        int byte_off = dd * QT * 2 + ((jp * 4) ^ ((dd & TM) << 4));
                        __bf16 pq[2] = {qtv_a[r][e], qtv_b[r][e]};
                        __bf16 pd[2] = {dov_a[r][e], dov_b[r][e]};
                        *(uint32_t*)((char*)lds.qt[par] + byte_off) = *(uint32_t*)pq;
                        *(uint32_t*)((char*)lds.dot[par] + byte_off) = *(uint32_t*)pd;
                    }
                }
            }
            if (tid < QT) {
                lds.lse[par][tid] = lse_st;
                lds.delta[par][tid] = delta_st;
            }
        };

        if (t0 < t1) {
            load_qtile();
            write_qtile(t0 & 1);
            if (t0 + 1 < t1) load_qtile();
        }

        for (int t = t0; t < t1; ++t) {
            const int par = t & 1;
            const long i0 = (long)t * QT;
            const long imax = min(i0 + QT, P.nq) - 1;
            const long q_lo = i0 * P.q_stride + P.diag;
            const long q_hi = imax * P.q_stride + P.diag;
            const bool full_tile = !BIAS &&
                (imax - i0 == QT - 1) &&
                (!P.causal || q_lo >= jmax) &&
                (!P.has_win || ((q_hi - j0_wg) <= P.win)) &&
                !P.kmask;

            __syncthreads();

            if (t + 1 < t1) write_qtile(par ^ 1);
            if (t + 2 < t1) load_qtile();

            #pragma unroll
            for (int qb = 0; qb < QBLKS; ++qb) {
                // S2[q][kv], dP[q][kv]: lane = kv, q rows in regs
                f32x16 s2 = f32x16{}, dp = f32x16{};
                #pragma unroll
                for (int ks = 0; ks < KSTEPS; ++ks) {
                    int qrow = qb * 32 + l31;
                    int chunk = ks * 2 + lhi;
                    bf16x8 qa = *(const bf16x8*)(lds.q[par] + qrow * D + bswz<D / 8>(qrow, chunk) * 8);
                    bf16x8 da = *(const bf16x8*)(lds.do_[par] + qrow * D + bswz<D / 8>(qrow, chunk) * 8);
                    s2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kf[ks], s2, 0, 0, 0);
                    dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, vf[ks], dp, 0, 0, 0);
                }

                uint32_t p_pk[8], ds_pk[8];
                if (full_tile && col_valid) {
                    #pragma unroll
                    for (int x2 = 0; x2 < 8; ++x2) {
                        float pe[2], dse[2];
                        // per-group wide loads (8 live regs, not 32): regs
                        // 4g..4g+3 are consecutive qloc -> one b128 each
                        f32x4 lse4, delta4;
                        {
                            int base = qb * 32 + 8 * (x2 >> 1) + 4 * lhi;
                            lse4 = *(const f32x4*)(lds.lse[par] + base);
                            delta4 = *(const f32x4*)(lds.delta[par] + base);
                        }
                        #pragma unroll
                        for (int e = 0; e < 2; ++e) {
                            int r = 2 * x2 + e;
                            float x, dtanh = 1.f;
                            if constexpr (SOFTCLAMP) {
                                float inv_v = __builtin_amdgcn_rcpf(P.softclamp_value);
                                float th = bfast_tanhf(s2[r] * P.scale * inv_v);
                                x = P.softclamp_value * th * 1.4426950408889634f;
                                dtanh = 1.f - th * th;
                            } else {
                                x = __builtin_fmaf(s2[r], P.scale * 1.4426950408889634f,
                                                   -lse4[r & 3] * 1.4426950408889634f);  // fold
                            }
                            float pv = __builtin_amdgcn_exp2f(SOFTCLAMP ? x - lse4[r & 3] * 1.4426950408889634f : x);
                            pe[e] = pv;
                            dse[e] = pv * (dp[r] - delta4[r & 3]) * dtanh * P.scale;
                        }
                        union { __hip_bfloat162 h2; uint32_t u; } c1, c2;
                        c1.h2 = __float22bfloat162_rn(float2{pe[0], pe[1]});
                        c2.h2 = __float22bfloat162_rn(float2{dse[0], dse[1]});
                        p_pk[x2] = c1.u;
                        ds_pk[x2] = c2.u;
                    }
                } else {
                    #pragma unroll
                    for (int x2 = 0; x2 < 8; ++x2) {
                        float pe[2], dse[2];
                        // per-group wide loads (8 live regs, not 32): regs
                        // 4g..4g+3 are consecutive qloc -> one b128 each
                        f32x4 lse4, delta4;
                        {
                            int base = qb * 32 + 8 * (x2 >> 1) + 4 * lhi;
                            lse4 = *(const f32x4*)(lds.lse[par] + base);
                            delta4 = *(const f32x4*)(lds.delta[par] + base);
                        }
                        #pragma unroll
                        for (int e = 0; e < 2; ++e) {
                            int r = 2 * x2 + e;
                            int qloc = qb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                            long i = i0 + qloc;
                            float x, dtanh = 1.f;
                            if constexpr (SOFTCLAMP) {
                                float inv_v = __builtin_amdgcn_rcpf(P.softclamp_value);
                                float th = bfast_tanhf(s2[r] * P.scale * inv_v);
                                x = P.softclamp_value * th * 1.4426950408889634f;
                                dtanh = 1.f - th * th;
                            } else {
                                x = s2[r] * (P.scale * 1.4426950408889634f);
                            }
                            bool ok = col_valid && i <= imax;
                            if constexpr (BIAS) {
                                if (ok) {
                                    const long bi = P.bias_mat
                                        ? (((long)b * P.h + h) * P.nq + i) * P.nk + j
                                        : ((long)b * P.h + h) * P.nk + j;
                                    x += P.bias[bi] * 1.4426950408889634f;
                                }
                            }
                            long qpos = i * P.q_stride + P.diag;
                            if (P.causal) ok = ok && (j <= qpos);
                            if (P.has_win) ok = ok && (qpos - j <= P.win);
                            if (P.kmask) ok = ok && kmask_own;
                            float pv = ok ? __builtin_amdgcn_exp2f(x - lse4[r & 3] * 1.4426950408889634f) : 0.f;
                            pe[e] = pv;
                            dse[e] = pv * (dp[r] - delta4[r & 3]) * dtanh * P.scale;
                        }
                        union { __hip_bfloat162 h2; uint32_t u; } c1, c2;
                        c1.h2 = __float22bfloat162_rn(float2{pe[0], pe[1]});
                        c2.h2 = __float22bfloat162_rn(float2{dse[0], dse[1]});
                        p_pk[x2] = c1.u;
                        ds_pk[x2] = c2.u;
                    }
                }

                // fragments (lane = kv, k = q contiguous)
                uint32_t p_frag[2][4], ds_frag[2][4];
                #pragma unroll
                for (int half = 0; half < 2; ++half)
                    #pragma unroll
                    for (int c = 0; c < 2; ++c) {
                        u32x2 r1 = __builtin_amdgcn_permlane32_swap(
                            p_pk[half * 4 + c], p_pk[half * 4 + c + 2], false, false);
                        p_frag[half][c] = r1[0];
                        p_frag[half][c + 2] = r1[1];
                        u32x2 r2 = __builtin_amdgcn_permlane32_swap(
                            ds_pk[half * 4 + c], ds_pk[half * 4 + c + 2], false, false);
                        ds_frag[half][c] = r2[0];
                        ds_frag[half][c + 2] = r2[1];
                    }

                // dv^T[d][kv] += dO^T x p ; dk[kv][d] += ds^T x Q^T-read
                #pragma unroll
                for (int db = 0; db < DBLK; ++db)
                    #pragma unroll
                    for (int half = 0; half < 2; ++half) {
                        int qk = qb * 2 + half;
                        int drow = db * 32 + l31;
                        int ch = qk * 2 + lhi;
                        bf16x8 doa = *(const bf16x8*)(lds.dot[par] + drow * QT +
                                                      ((ch ^ (drow & TM))) * 8);
                        dv_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            doa, *(const bf16x8*)p_frag[half], dv_acc[db], 0, 0, 0);
                        bf16x8 qta = *(const bf16x8*)(lds.qt[par] + drow * QT +
                                                      ((ch ^ (drow & TM))) * 8);
                        dk_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            *(const bf16x8*)ds_frag[half], qta, dk_acc[db], 0, 0, 0);
                    }
            }
        }
    }

    // write dk (B,HK,Nk,D) and dv^T (B,HK,D,Nk)
    if (col_valid) {
        float* dkb = P.dk + (((long)b * P.hk + hkh) * P.nk) * D;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                long kvrow = j0_wg + wid * KVROWS_WAVE + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                int d = db * 32 + l31;
                if (kvrow <= jmax) {
                    float* dst = dkb + kvrow * D + d;
                    if (P.split > 1 || P.desc) atomicAdd(dst, dk_acc[db][r]);
                    else if (P.accumulate) *dst += dk_acc[db][r];
                    else *dst = dk_acc[db][r];
                }
            }
        float* dvb = P.dv + (((long)b * P.hk + hkh) * D) * P.nk;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                float* dst = dvb + (long)d * P.nk + j;
                if (P.split > 1 || P.desc) atomicAdd(dst, dv_acc[db][r]);
                else if (P.accumulate) *dst += dv_acc[db][r];
                else *dst = dv_acc[db][r];
            }
    }
    };
    // the noinline frame confines register allocation to the body — at d64
    // it cuts the hot-loop spills roughly in half for BOTH instantiations
    // (measured headline +4%, causal +17-24%); d128 is better DIRECT for
    // both forms (paired: 125 -> 138 TF causal 8k — the same value-copy
    // preference as the d128 fwd/dq bodies), so only d64 takes the call
    if constexpr (D == 64) dkv_noinline_call(dkv_body);
    else dkv_body();
    }  // pair loop
}

void launch_attn_bwd_dq(const BwdParams& p, int head_dim, hipStream_t stream) {
    dim3 block(512);
    int z = p.split > 1 ? p.split : 1;
    long qt_ = (p.nq + DQ_QROWS_WG - 1) / DQ_QROWS_WG;
    if (p.desc) qt_ = p.n_units;
    else if (p.paired) qt_ = (qt_ + 1) / 2;
    dim3 grid_dq(qt_, p.b * p.h, z);
    const bool pr = p.paired > 0;
    if (p.bias) {   // bias runs unpaired (bindings force paired=0)
        if (head_dim == 64) {
            if (p.softclamp) hipLaunchKernelGGL((attn_bwd_dq_kernel<64, true, false, true>), grid_dq, block, 0, stream, p);
            else hipLaunchKernelGGL((attn_bwd_dq_kernel<64, false, false, true>), grid_dq, block, 0, stream, p);
        } else if (head_dim == 128) {
            if (p.softclamp) hipLaunchKernelGGL((attn_bwd_dq_kernel<128, true, false, true>), grid_dq, block, 0, stream, p);
            else hipLaunchKernelGGL((attn_bwd_dq_kernel<128, false, false, true>), grid_dq, block, 0, stream, p);
        } else {
            if (p.softclamp) hipLaunchKernelGGL((attn_bwd_dq_kernel<32, true, false, true>), grid_dq, block, 0, stream, p);
            else hipLaunchKernelGGL((attn_bwd_dq_kernel<32, false, false, true>), grid_dq, block, 0, stream, p);
        }
        return;
    }
    if (head_dim == 64) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_bwd_dq_kernel<64, true, true>), grid_dq, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dq_kernel<64, true, false>), grid_dq, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_bwd_dq_kernel<64, false, true>), grid_dq, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dq_kernel<64, false, false>), grid_dq, block, 0, stream, p);
    } else if (head_dim == 128) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_bwd_dq_kernel<128, true, true>), grid_dq, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dq_kernel<128, true, false>), grid_dq, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_bwd_dq_kernel<128, false, true>), grid_dq, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dq_kernel<128, false, false>), grid_dq, block, 0, stream, p);
    } else if (head_dim == 32) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_bwd_dq_kernel<32, true, true>), grid_dq, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dq_kernel<32, true, false>), grid_dq, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_bwd_dq_kernel<32, false, true>), grid_dq, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dq_kernel<32, false, false>), grid_dq, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}

void launch_attn_bwd_dkv(const BwdParams& p, int head_dim, hipStream_t stream) {
    dim3 block(512);
    int z = p.split > 1 ? p.split : 1;
    long kt_ = (p.nk + KVROWS_WG - 1) / KVROWS_WG;
    if (p.desc) kt_ = p.n_units;
    else if (p.paired) kt_ = (kt_ + 1) / 2;
    dim3 grid_dkv(kt_, p.b * p.hk, z);
    const bool pr = p.paired > 0;
    if (p.bias) {   // bias runs unpaired (bindings force paired=0)
        if (head_dim == 64) {
            if (p.softclamp) hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, 64, true, false, true>), grid_dkv, block, 0, stream, p);
            else hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, 64, false, false, true>), grid_dkv, block, 0, stream, p);
        } else if (head_dim == 128) {
            if (p.softclamp) hipLaunchKernelGGL((attn_bwd_dkv_kernel<128, 64, true, false, true>), grid_dkv, block, 0, stream, p);
            else hipLaunchKernelGGL((attn_bwd_dkv_kernel<128, 64, false, false, true>), grid_dkv, block, 0, stream, p);
        } else {
            if (p.softclamp) hipLaunchKernelGGL((attn_bwd_dkv_kernel<32, 64, true, false, true>), grid_dkv, block, 0, stream, p);
            else hipLaunchKernelGGL((attn_bwd_dkv_kernel<32, 64, false, false, true>), grid_dkv, block, 0, stream, p);
        }
        return;
    }
    if (head_dim == 64) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, 64, true, true>), grid_dkv, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, 64, true, false>), grid_dkv, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, 64, false, true>), grid_dkv, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, 64, false, false>), grid_dkv, block, 0, stream, p);
    } else if (head_dim == 128) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_bwd_dkv_kernel<128, 64, true, true>), grid_dkv, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dkv_kernel<128, 64, true, false>), grid_dkv, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_bwd_dkv_kernel<128, 64, false, true>), grid_dkv, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dkv_kernel<128, 64, false, false>), grid_dkv, block, 0, stream, p);
    } else if (head_dim == 32) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_bwd_dkv_kernel<32, 64, true, true>), grid_dkv, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dkv_kernel<32, 64, true, false>), grid_dkv, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_bwd_dkv_kernel<32, 64, false, true>), grid_dkv, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_bwd_dkv_kernel<32, 64, false, false>), grid_dkv, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}

// ---------------------------------------------------------------------------
// delta = rowsum(dO * O) preprocess, fused over the bf16 inputs
// (replaces two f32 casts + mul + reduce torch kernels per backward)
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void attn_delta_kernel(DeltaParams p) {
    constexpr int LPR = D / 8;                 // lanes per row (16B each)
    constexpr int RPW = 64 / LPR;              // rows per wave
    const int lane = threadIdx.x & 63;
    const int wave = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
    const long row = (long)wave * RPW + lane / LPR;
    if (row >= p.rows) return;
    const int c = lane % LPR;

    const __bf16* dop = (const __bf16*)p.dout + row * D + c * 8;
    const __bf16* op = (const __bf16*)p.out + row * D + c * 8;
    bf16x8 a = *(const bf16x8*)dop;
    bf16x8 b = *(const bf16x8*)op;
    float s = 0.f;
    #pragma unroll
    for (int e = 0; e < 8; ++e) s += (float)a[e] * (float)b[e];
    #pragma unroll
    for (int off = LPR / 2; off > 0; off >>= 1) s += __shfl_xor(s, off);
    if (c == 0) {
        // row = (bb * n + i) * h + hh  ->  delta[(bb * h + hh) * n + i]
        const long hh = row % p.h;
        const long bi = row / p.h;
        const long i = bi % p.n, bb = bi / p.n;
        p.delta[(bb * p.h + hh) * p.n + i] = s;
    }
}

void launch_attn_delta(const DeltaParams& p, int head_dim, hipStream_t stream) {
    dim3 block(256);
    if (head_dim == 64) {
        long waves = (p.rows + 7) / 8;
        hipLaunchKernelGGL(attn_delta_kernel<64>, dim3((waves + 3) / 4), block, 0, stream, p);
    } else if (head_dim == 128) {
        long waves = (p.rows + 3) / 4;
        hipLaunchKernelGGL(attn_delta_kernel<128>, dim3((waves + 3) / 4), block, 0, stream, p);
    } else if (head_dim == 32) {
        long waves = (p.rows + 15) / 16;
        hipLaunchKernelGGL(attn_delta_kernel<32>, dim3((waves + 3) / 4), block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}

}  // namespace ring_attn
