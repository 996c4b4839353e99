// CDNA4 (gfx950) flash-attention FORWARD v2 — one-wave-per-SIMD segmented
// pipeline (the "pwg4x64" structure of the CDNA guide, realized in plain HIP
// with inline-asm DMA waits and tr_b16 V reads).
//
// Brand-new MI355X design (capability counterpart of the reference's Triton
// _fwd_kernel, /root/reference/ring_attention_pytorch/triton_flash_attn.py:
// 52-430; same resume contract as attn_fwd.hip v1, which remains the
// fallback for shapes v2 does not cover).
//
// Structure (differences from v1, all measured round-1 bottlenecks):
//   * 4 waves / 256 threads per workgroup, ONE wave per SIMD
//     (__launch_bounds__(256, 1) => up to 512 unified VGPR+AGPR per lane).
//     v1's 8-wave lockstep left ~50% of cycles parked on intra-SIMD
//     arbitration (profiles/README.md): with one wave per SIMD the whole
//     issue stream belongs to this wave and softmax VALU interleaves with
//     the MFMA pipe instead of convoying behind it.
//   * each wave owns 64 q rows (two 32-row blocks); KV tile = KVB rows.
//   * K and V arrive by LDS-DMA (__builtin_amdgcn_global_load_lds, 16B/lane)
//     into 2-deep LDS rings — the wave never stages through registers, so
//     staging costs issue slots only.  ONE counted s_waitcnt vmcnt per tile
//     (leaving the next tiles' DMA in flight) + two RAW s_barriers; no
//     __syncthreads (its implicit vmcnt(0) would drain the DMA pipeline).
//   * K LDS image is XOR-16B-chunk swizzled via the DMA *source* address
//     (both-sides rule: glds writes lane-linear, so the swizzle must ride on
//     the global address); V image is row-major and read with
//     ds_read_b64_tr_b16, whose lane redistribution out[l][j] =
//     fetch(4j + ((l&15)>>2))[(l&15)&3] turns per-lane row fetches into the
//     column-major PV A-fragments directly (hardware-verified in
//     tools/hw_probe.hip).
//   * software pipeline with a one-tile skew:
//       phase A(t): QK^T(t) MFMAs ∥ softmax-finish(t-1) (P pack -> frags,
//                   rowsum finish, m/l update, O rescale) ∥ V(t-1) tr-reads
//       phase B(t): PV(t-1) MFMAs ∥ softmax-start(t) (mask, row-max,
//                   exp2, pack, partial row sums) ∥ DMA K(t+2), V(t+1)
//     so the 145us of non-overlapped softmax VALU measured in v1 runs under
//     the opposite phase's MFMA pipe.  Exact numerics (no defer-max): the
//     max decision for tile t completes inside phase B(t) before its exps.
//   * same masking algebra as v1: qpos(i) = i*q_stride + diag; causal,
//     sliding window, key-pad mask, softclamp, kv_split, resume (o_acc/m/l).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>
#include <cstdlib>
#include <type_traits>

#include "attn_common.h"

namespace ring_attn {

namespace v2 {

static constexpr int WAVES = 4;
static constexpr int NTHREADS = 256;
static constexpr int QROWS_WAVE = 64;
static constexpr int QROWS_WG = WAVES * QROWS_WAVE;   // 256

static constexpr float LOG2E = 1.4426950408889634f;
static constexpr float LN2 = 0.6931471805599453f;

typedef unsigned long long u64_t;
typedef __attribute__((address_space(3))) unsigned int lds_u32;
typedef __attribute__((address_space(1))) unsigned int glb_u32;

__device__ __forceinline__ float fast_tanhf(float x) {
    float e = __builtin_amdgcn_exp2f(x * 2.885390081777927f);
    return 1.f - 2.f * __builtin_amdgcn_rcpf(e + 1.f);
}

__device__ __forceinline__ float cross_half(float x) {
    union { float f; unsigned u; } c; c.f = x;
    u32x2 r = __builtin_amdgcn_permlane32_swap(c.u, c.u, false, false);
    union { unsigned u; float f; } lo, hi; lo.u = r[0]; hi.u = r[1];
    return (threadIdx.x & 32) ? lo.f : hi.f;
}

template <int KVB, int D>
struct V2Lds {
    __bf16 k[2][KVB * D];          // [kv][d], 16B chunks XOR-swizzled by (kv&7)
    __bf16 v[2][KVB * D];          // [kv][d], linear (tr_b16-read)
    unsigned char km[2][KVB];
};

// one LDS-DMA "piece": 64 lanes x 16 B = 1 KiB, 1024/(2*D) kv rows
template <int D>
__device__ __forceinline__ void glds16(const __bf16* src, __bf16* dst_lds) {
    __builtin_amdgcn_global_load_lds((const glb_u32*)src, (lds_u32*)dst_lds, 16, 0, 0);
}

template <int D, int KVB, bool SOFTCLAMP>
__global__ __launch_bounds__(NTHREADS, 1) void attn_fwd_v2_kernel(FwdParams p) {
    static_assert(D % 32 == 0 && KVB % 32 == 0);
    constexpr int DBLK = D / 32;        // 32-d output blocks
    constexpr int KSTEPS = D / 16;      // QK^T k-steps
    constexpr int NKV32 = KVB / 32;     // kv 32-blocks
    constexpr int PVKS = KVB / 16;      // PV k-steps
    constexpr int CHROW = D / 8;        // 16B chunks per kv row
    constexpr int ROWS_PER_PIECE = 1024 / (2 * D);
    constexpr int PIECES = KVB * D * 2 / 1024;       // per K (or V) tile
    constexpr int PW = PIECES / WAVES;               // pieces per wave
    static_assert(PW * WAVES == PIECES);

    __shared__ __align__(16) V2Lds<KVB, D> lds;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l31 = lane & 31;
    const int lhi = lane >> 5;

    // work item: (qtile, bh); bh fastest so blockIdx%8 pins (b,h) per XCD
    const int nbh = p.b * p.h;
    const int bh = blockIdx.x % nbh;
    const int qtile = blockIdx.x / nbh;
    const int b = bh / p.h;
    const int h = bh % p.h;
    const int hk = h % p.hk;            // reference tile GQA pairing

    const long irow0 = (long)qtile * QROWS_WG + wid * QROWS_WAVE;
    // this lane's two q rows (one per 32-row block)
    long iq[2];
    bool rowv[2];
    #pragma unroll
    for (int qb = 0; qb < 2; ++qb) {
        iq[qb] = irow0 + qb * 32 + l31;
        rowv[qb] = iq[qb] < p.nq;
        if (!rowv[qb]) iq[qb] = 0;
    }

    // ---- Q fragments: qf[qb][ks] = q[b, iq, h, ks*16 + lhi*8 ..+8]
    bf16x8 qf[2][KSTEPS];
    #pragma unroll
    for (int qb = 0; qb < 2; ++qb) {
        const __bf16* qbase = (const __bf16*)p.q
            + ((long)b * p.nq + iq[qb]) * p.h * D + (long)h * D;
        #pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks)
            qf[qb][ks] = *(const bf16x8*)(qbase + ks * 16 + lhi * 8);
    }

    // ---- accumulators
    float m_run[2] = {MASK_VALUE_F, MASK_VALUE_F};
    float l_run[2] = {0.f, 0.f};
    f32x16 o_acc[2][DBLK];
    #pragma unroll
    for (int qb = 0; qb < 2; ++qb)
        #pragma unroll
        for (int db = 0; db < DBLK; ++db) o_acc[qb][db] = f32x16{};

    const bool split_mode = p.kv_split > 1;
    const int zsplit = blockIdx.z;
    if (!p.is_first && !split_mode) {
        const float* mrow = p.m + ((long)b * p.h + h) * p.nq;
        const float* lrow = p.l + ((long)b * p.h + h) * p.nq;
        const float* oa = p.o_acc + (((long)b * p.h + h) * D) * p.nq;
        #pragma unroll
        for (int qb = 0; qb < 2; ++qb) {
            m_run[qb] = mrow[iq[qb]] * LOG2E;
            l_run[qb] = lrow[iq[qb]];
            #pragma unroll
            for (int db = 0; db < DBLK; ++db)
                #pragma unroll
                for (int r = 0; r < 16; ++r) {
                    int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                    o_acc[qb][db][r] = oa[(long)d * p.nq + iq[qb]];
                }
        }
    }

    // anchor the compiler's vmcnt wait for the Q/o_acc loads HERE (before
    // any LDS-DMA is in flight): an asm input is a "use", so hipcc emits its
    // wait at this point instead of inside the tile loop where it would
    // drain the DMA pipeline every iteration
    #pragma unroll
    for (int qb = 0; qb < 2; ++qb) {
        #pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks)
            asm volatile("" :: "v"(qf[qb][ks]));
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            asm volatile("" :: "a"(o_acc[qb][db]));
    }

    // ---- tile bounds (same algebra as v1)
    const long wg_i_min = (long)qtile * QROWS_WG;
    const long wg_i_max = min((long)(qtile + 1) * QROWS_WG, p.nq) - 1;
    const long wg_q_min = wg_i_min * p.q_stride + p.diag;
    const long wg_q_max = wg_i_max * p.q_stride + p.diag;
    long qpos[2];
    #pragma unroll
    for (int qb = 0; qb < 2; ++qb) qpos[qb] = iq[qb] * p.q_stride + p.diag;
    const int num_kv_tiles = (int)((p.nk + KVB - 1) / KVB);

    int t_lo = 0, t_hi = num_kv_tiles;
    if (p.causal)
        t_hi = wg_q_max < 0 ? 0 : min((long)num_kv_tiles, wg_q_max / KVB + 1);
    if (p.has_win) {
        long x = wg_q_min - p.win - KVB + 1;
        t_lo = x <= 0 ? 0 : (int)((x + KVB - 1) / KVB);
        if (t_lo > t_hi) t_lo = t_hi;
    }
    if (split_mode) {
        int valid = t_hi - t_lo;
        int per_split = (valid + p.kv_split - 1) / p.kv_split;
        int base = t_lo;
        t_lo = base + min(valid, zsplit * per_split);
        t_hi = base + min(valid, (zsplit + 1) * per_split);
    }

    // ---- DMA source bases
    const long kv_row_stride = (long)p.hk * D;
    const __bf16* kbase = (const __bf16*)p.k + ((long)b * p.nk) * kv_row_stride + (long)hk * D;
    const __bf16* vbase = (const __bf16*)p.v + ((long)b * p.nk) * kv_row_stride + (long)hk * D;
    const unsigned char* mbase = p.kmask ? (const unsigned char*)p.kmask + (long)b * p.nk : nullptr;

    // per-lane RUNNING source pointers (advance by a constant per tile: a
    // per-tile 64-bit address rebuild costs ~2x the whole MFMA issue).  K and
    // V each keep their own pointer/row, self-advanced per call, so the call
    // SEQUENCE walks consecutive tiles; the t argument only picks the slot.
    const int src_row_in_piece = lane / CHROW;
    const int src_ch = lane % CHROW;
    const long tile_stride = (long)KVB * kv_row_stride;
    const int row0 = wid * PW * ROWS_PER_PIECE + src_row_in_piece;
    // NOTE: the K source swizzle depends on the PIECE's row ((row0 +
    // pc*ROWS_PER_PIECE) & 7), which differs per piece when ROWS_PER_PIECE
    // < 8 (D=128) — so kdma points at the ROW (chunk added per piece)
    const __bf16* kdma = kbase + (long)t_lo * tile_stride
        + (long)row0 * kv_row_stride;
    const __bf16* vdma = vbase + (long)t_lo * tile_stride
        + (long)row0 * kv_row_stride + (long)(src_ch * 8);
    long krow_dma = (long)t_lo * KVB + row0;   // lane's kv row of its piece 0
    long vrow_dma = krow_dma;
    const __bf16* const ksrc_last = kbase + (p.nk - 1) * kv_row_stride;
    const __bf16* const vsrc_last = vbase + (p.nk - 1) * kv_row_stride;

    auto dma_k = [&](int t) {
        const int par = t & 1;
        #pragma unroll
        for (int pc = 0; pc < PW; ++pc) {
            const int piece = wid * PW + pc;
            const int chunk = (src_ch ^ ((row0 + pc * ROWS_PER_PIECE) & 7)) * 8;
            const __bf16* src = kdma + (long)(pc * ROWS_PER_PIECE) * kv_row_stride + chunk;
            if (krow_dma + pc * ROWS_PER_PIECE >= p.nk)      // tail clamp (rare)
                src = ksrc_last + chunk;
            glds16<D>(src, lds.k[par] + piece * 512);
        }
        kdma += tile_stride; krow_dma += KVB;
    };
    auto dma_v = [&](int t) {
        const int par = t & 1;
        #pragma unroll
        for (int pc = 0; pc < PW; ++pc) {
            const int piece = wid * PW + pc;
            const __bf16* src = vdma + (long)(pc * ROWS_PER_PIECE) * kv_row_stride;
            if (vrow_dma + pc * ROWS_PER_PIECE >= p.nk)
                src = vsrc_last + (long)(src_ch * 8);
            glds16<D>(src, lds.v[par] + piece * 512);
        }
        vdma += tile_stride; vrow_dma += KVB;
    };
    // key-pad mask tile staging (ordinary load + ds_write; only when kmask
    // given — the load's compiler wait costs a DMA drain per tile, accepted
    // on this rare path)
    auto stage_km = [&](int t) {
        if (!mbase) return;
        const int par = t & 1;
        const long j0 = (long)t * KVB;
        if (wid == 3 && lane < KVB / 16) {
            uint4 mv{0, 0, 0, 0};
            const long jb = j0 + lane * 16;
            if (jb + 15 < p.nk) mv = *(const uint4*)(mbase + jb);
            else {
                unsigned char tmp[16];
                #pragma unroll
                for (int e = 0; e < 16; ++e)
                    tmp[e] = (jb + e < p.nk) ? mbase[jb + e] : 0;
                mv = *(const uint4*)tmp;
            }
            *(uint4*)(lds.km[par] + lane * 16) = mv;
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        }
    };

    const float scale2 = p.scale * LOG2E;

    // ---- per-tile state carried across phases (t-1 -> t)
    f32x16 s[2][NKV32];                 // S^T of tile t (phase B consumes)
    uint32_t pfrag[2][PVKS][4];         // PV B-fragments of tile t-1
    float psum[2][2];                   // partial row sums of tile t-1
    bool growth[2];                     // did the max grow at tile t-1
    float alpha[2];                     // exp2(m_old - m_new) of tile t-1

    // ---- prologue: fill both ring slots, then enter the pipeline
    if (t_lo < t_hi) {
        dma_k(t_lo); dma_v(t_lo);
        if (t_lo + 1 < t_hi) dma_k(t_lo + 1);
        stage_km(t_lo);
        // K(t_lo), V(t_lo) landed; leave K(t_lo+1) in flight
        if (t_lo + 1 < t_hi)
            asm volatile("s_waitcnt vmcnt(%0)" :: "i"(PW) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");

    const bool stamp0 = p.ticks && blockIdx.x == 0 && blockIdx.z == 0 && tid == 0;
    // pipeline: iteration t in [t_lo, t_hi]; t == t_hi is the drain step
    for (int t = t_lo; t <= t_hi; ++t) {
        if (t_lo >= t_hi) break;
        const int par = t & 1;
        const bool have_cur = t < t_hi;          // QK^T(t)/softmax(t) exist
        const bool have_prev = t > t_lo;         // PV(t-1)/finish(t-1) exist
        const bool stamp = stamp0 && t < 64;
        if (stamp) p.ticks[t * 7 + 0] = __builtin_amdgcn_s_memtime();
        // V slot t&1's previous occupant V(t-2) was read in phase B(t-1),
        // the other side of the entry barrier — the only placement where no
        // wave can still be reading the slot when the DMA lands.  V(t) is
        // consumed by PV(t) in phase B(t+1), ~1.5 phases of flight.
        if (t > t_lo && have_cur) { dma_v(t); stage_km(t); }

        // ================= PHASE A =================
        // QK^T(t) ∥ finish-softmax(t-1) ∥ V(t-1) tr-reads
        if (have_cur) {
            #pragma unroll
            for (int qb = 0; qb < 2; ++qb)
                #pragma unroll
                for (int kb = 0; kb < NKV32; ++kb) s[qb][kb] = f32x16{};
            #pragma unroll
            for (int kb = 0; kb < NKV32; ++kb) {
                const int krow = kb * 32 + l31;
                #pragma unroll
                for (int ks = 0; ks < KSTEPS; ++ks) {
                    const int chunk = (ks * 2 + lhi) ^ (krow & 7);
                    bf16x8 kf = *(const bf16x8*)(lds.k[par] + krow * D + chunk * 8);
                    #pragma unroll
                    for (int qb = 0; qb < 2; ++qb)
                        s[qb][kb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            kf, qf[qb][ks], s[qb][kb], 0, 0, 0);
                }
            }
        }
        if (stamp) p.ticks[t * 7 + 1] = __builtin_amdgcn_s_memtime();
        if (have_prev) {
            // finish-softmax(t-1): row sum, l/m bookkeeping, O rescale
            #pragma unroll
            for (int qb = 0; qb < 2; ++qb) {
                float rowsum = psum[qb][0] + psum[qb][1];
                rowsum += cross_half(rowsum);
                if (__builtin_amdgcn_readfirstlane((int)growth[qb])) {
                    asm volatile("");   // keep this a scalar branch
                    l_run[qb] = l_run[qb] * alpha[qb] + rowsum;
                    #pragma unroll
                    for (int db = 0; db < DBLK; ++db)
                        #pragma unroll
                        for (int r = 0; r < 16; ++r)
                            o_acc[qb][db][r] *= alpha[qb];
                } else {
                    l_run[qb] += rowsum;
                }
            }
        }

        // barrier: all phase-A LDS reads issued; DMA into the slots they
        // read is only issued after this point (phase B), and lands >=
        // ~200 cycles later, far behind the in-flight reads.  The empty-asm
        // fences pin compiler memory ops to their phase (the raw barrier
        // builtin is not a compiler memory fence).
        if (stamp) p.ticks[t * 7 + 2] = __builtin_amdgcn_s_memtime();
        asm volatile("" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("" ::: "memory");
        if (stamp) p.ticks[t * 7 + 3] = __builtin_amdgcn_s_memtime();

        // ================= PHASE B =================
        // PV(t-1) ∥ softmax-start(t) ∥ DMA K(t+2), V(t+1)
        if (t + 2 < t_hi) dma_k(t + 2);   // kdma sits at tile t+2 (K leads V)

        if (have_prev) {
            // V(t-1) tr-reads, per-dblock just-in-time (16 VGPRs live):
            // vf covers kv quad (pks*16 + 8*lhi + 4*half),
            // d = db*32 + 16*((l>>4)&1) + (l&15)
            const int f = lane & 15;
            const int g16 = (lane >> 4) & 1;
            const unsigned vb_lds = (unsigned)(uintptr_t)
                (__attribute__((address_space(3))) __bf16*)lds.v[(t - 1) & 1];
            const unsigned dpart = (unsigned)((g16 * 16 + 4 * (f & 3)) * 2);
            u64_t vtr[DBLK][PVKS][2];
            #pragma unroll
            for (int db = 0; db < DBLK; ++db) {
                #pragma unroll
                for (int pks = 0; pks < PVKS; ++pks) {
                    const unsigned kvq = pks * 16 + 8 * lhi + (f >> 2);
                    unsigned a0 = vb_lds + kvq * (D * 2) + db * 64 + dpart;
                    unsigned a1 = a0 + 4 * (D * 2);
                    asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
                                 "ds_read_b64_tr_b16 %1, %3"
                                 : "=&v"(vtr[db][pks][0]), "=&v"(vtr[db][pks][1])
                                 : "v"(a0), "v"(a1) : "memory");
                }
                // wait for this dblock's reads before its MFMAs; the next
                // dblock's reads are issued behind them
                asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
                __builtin_amdgcn_sched_barrier(0);
                #pragma unroll
                for (int pks = 0; pks < PVKS; ++pks) {
                    bf16x8 vf;
                    *(u64_t*)&vf = vtr[db][pks][0];
                    *((u64_t*)&vf + 1) = vtr[db][pks][1];
                    #pragma unroll
                    for (int qb = 0; qb < 2; ++qb)
                        o_acc[qb][db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            vf, *(const bf16x8*)pfrag[qb][pks], o_acc[qb][db], 0, 0, 0);
                }
            }
        }
        if (stamp) p.ticks[t * 7 + 4] = __builtin_amdgcn_s_memtime();

        if (have_cur) {
            const long j0 = (long)t * KVB;
            const long jmax = min(j0 + KVB, p.nk) - 1;
            const bool full_tile =
                (jmax - j0 == KVB - 1) &&
                (!p.causal || jmax <= wg_q_min) &&
                (!p.has_win || (wg_q_max - j0) <= p.win) &&
                !p.kmask;

            // softmax-start for both q blocks, SPECIALIZED on full/masked at
            // statement level: a shared exp/pack tail would make the
            // compiler phi-merge per-element s values from both arms into
            // cndmask chains that execute on EVERY tile (measured: the
            // masked arm's i64 compares ran on full tiles, 4k cycles/tile)
            auto softmax_start = [&](auto fullc) {
                constexpr bool FULL = decltype(fullc)::value;
                #pragma unroll
                for (int qb = 0; qb < 2; ++qb) {
                    float smax = MASK_VALUE_F;
                    float sv[NKV32 * 16];
                    if constexpr (FULL && !SOFTCLAMP) {
                        #pragma unroll
                        for (int kb = 0; kb < NKV32; ++kb)
                            #pragma unroll
                            for (int r = 0; r < 16; ++r) {
                                sv[kb * 16 + r] = s[qb][kb][r];
                                smax = fmaxf(smax, sv[kb * 16 + r]);
                            }
                        smax *= scale2;
                    } else {
                        #pragma unroll
                        for (int kb = 0; kb < NKV32; ++kb)
                            #pragma unroll
                            for (int r = 0; r < 16; ++r) {
                                float x;
                                if constexpr (SOFTCLAMP) {
                                    float xs = s[qb][kb][r]
                                        * (p.scale * __builtin_amdgcn_rcpf(p.softclamp_value));
                                    x = p.softclamp_value * fast_tanhf(xs) * LOG2E;
                                } else {
                                    x = s[qb][kb][r] * scale2;
                                }
                                if constexpr (!FULL) {
                                    long j = j0 + kb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                                    bool ok = j <= jmax;
                                    if (p.causal) ok = ok && (j <= qpos[qb]);
                                    if (p.has_win) ok = ok && (qpos[qb] - j <= p.win);
                                    if (p.kmask) ok = ok && lds.km[par][j - j0];
                                    if (!ok) x = MASK_VALUE_F;
                                }
                                sv[kb * 16 + r] = x;
                                smax = fmaxf(smax, x);
                            }
                    }
                    smax = fmaxf(smax, cross_half(smax));

                    float m_new0 = fmaxf(m_run[qb], smax);
                    growth[qb] = !__all(smax <= m_run[qb]);
                    alpha[qb] = growth[qb]
                        ? __builtin_amdgcn_exp2f(m_run[qb] - m_new0) : 1.f;
                    m_run[qb] = m_new0;
                    // all-masked rows: clamp the exp-domain max so
                    // exp2(MASK - MASK) cannot become 1 (see attn_fwd.hip)
                    const float m_new = fmaxf(m_new0, -1.7e38f);
                    constexpr bool RAW = FULL && !SOFTCLAMP;
                    float part[4] = {0.f, 0.f, 0.f, 0.f};
                    uint32_t pk[NKV32 * 8];
                    #pragma unroll
                    for (int x2 = 0; x2 < NKV32 * 8; ++x2) {
                        float e0, e1;
                        if constexpr (RAW) {
                            e0 = __builtin_amdgcn_exp2f(
                                __builtin_fmaf(sv[2 * x2], scale2, -m_new));
                            e1 = __builtin_amdgcn_exp2f(
                                __builtin_fmaf(sv[2 * x2 + 1], scale2, -m_new));
                        } else {
                            e0 = __builtin_amdgcn_exp2f(sv[2 * x2] - m_new);
                            e1 = __builtin_amdgcn_exp2f(sv[2 * x2 + 1] - m_new);
                        }
                        part[x2 & 3] += e0 + e1;
                        union { __hip_bfloat162 h2; uint32_t u; } cvt;
                        cvt.h2 = __float22bfloat162_rn(float2{e0, e1});
                        pk[x2] = cvt.u;
                    }
                    psum[qb][0] = part[0] + part[1];
                    psum[qb][1] = part[2] + part[3];
                    // build the PV B-fragments now (pk dies inside this
                    // phase; only pfrag crosses the barrier)
                    #pragma unroll
                    for (int kb = 0; kb < NKV32; ++kb) {
                        #pragma unroll
                        for (int half = 0; half < 2; ++half) {
                            #pragma unroll
                            for (int c = 0; c < 2; ++c) {
                                u32x2 r = __builtin_amdgcn_permlane32_swap(
                                    pk[kb * 8 + half * 4 + c],
                                    pk[kb * 8 + half * 4 + c + 2], false, false);
                                pfrag[qb][kb * 2 + half][c] = r[0];
                                pfrag[qb][kb * 2 + half][c + 2] = r[1];
                            }
                        }
                    }
                }
            };
            if (full_tile) {
                asm volatile("");
                softmax_start(std::integral_constant<bool, true>{});
            } else {
                asm volatile("");
                softmax_start(std::integral_constant<bool, false>{});
            }
        }

        if (stamp) p.ticks[t * 7 + 5] = __builtin_amdgcn_s_memtime();
        // one counted wait per tile: K(t+1) (read by the next phase A) and
        // V(t) (read by the next phase B) must have landed; leave only the
        // just-issued K(t+2) in flight
        if (t + 2 < t_hi)
            asm volatile("s_waitcnt vmcnt(%0)" :: "i"(PW) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        if (stamp) p.ticks[t * 7 + 6] = __builtin_amdgcn_s_memtime();
        asm volatile("" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("" ::: "memory");
    }

    // ---- epilogue (same contracts as v1)
    #pragma unroll
    for (int qb = 0; qb < 2; ++qb) {
        if (!rowv[qb]) continue;
        const long i = iq[qb];
        if (split_mode) {
            const long partz = (long)zsplit * p.b * p.h;
            float* mrow = p.m + (partz + (long)b * p.h + h) * p.nq;
            float* lrow = p.l + (partz + (long)b * p.h + h) * p.nq;
            if (lhi == 0) { mrow[i] = m_run[qb] * LN2; lrow[i] = l_run[qb]; }
            float* oa = p.o_acc + (partz + (long)b * p.h + h) * D * p.nq;
            #pragma unroll
            for (int db = 0; db < DBLK; ++db)
                #pragma unroll
                for (int r = 0; r < 16; ++r) {
                    int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                    oa[(long)d * p.nq + i] = o_acc[qb][db][r];
                }
            continue;
        }
        if (p.is_last) {
            float l_safe = fmaxf(l_run[qb], 1e-38f);
            float inv_l = 1.f / l_safe;
            __bf16* ob = (__bf16*)p.out + ((long)b * p.nq + i) * p.h * D + (long)h * D;
            #pragma unroll
            for (int db = 0; db < DBLK; ++db)
                #pragma unroll
                for (int g = 0; g < 4; ++g) {
                    __bf16 four[4];
                    #pragma unroll
                    for (int e = 0; e < 4; ++e)
                        four[e] = (__bf16)(o_acc[qb][db][g * 4 + e] * inv_l);
                    int d = db * 32 + 8 * g + 4 * lhi;
                    *(uint2*)(ob + d) = *(uint2*)four;
                }
            if (lhi == 0) {
                float* lsep = p.lse + ((long)b * p.h + h) * p.nq;
                lsep[i] = __logf(l_safe) + m_run[qb] * LN2;
            }
        } else {
            float* mrow = p.m + ((long)b * p.h + h) * p.nq;
            float* lrow = p.l + ((long)b * p.h + h) * p.nq;
            if (lhi == 0) { mrow[i] = m_run[qb] * LN2; lrow[i] = l_run[qb]; }
            float* oa = p.o_acc + (((long)b * p.h + h) * D) * p.nq;
            #pragma unroll
            for (int db = 0; db < DBLK; ++db)
                #pragma unroll
                for (int r = 0; r < 16; ++r) {
                    int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                    oa[(long)d * p.nq + i] = o_acc[qb][db][r];
                }
        }
    }
}

}  // namespace v2

// returns true if v2 handled this launch
bool launch_attn_fwd_v2(const FwdParams& p, int head_dim, hipStream_t stream) {
    if (p.ablate || p.bias) return false;   // ticks ARE supported (7 stamps/tile)
    long qtiles = (p.nq + v2::QROWS_WG - 1) / v2::QROWS_WG;
    dim3 grid(qtiles * p.b * p.h, 1, p.kv_split > 1 ? p.kv_split : 1);
    dim3 block(v2::NTHREADS);
    if (head_dim == 64) {
        static const char* kvbe = std::getenv("RING_ATTN_V2_KVB");
        const bool kvb64 = kvbe && kvbe[0] == '6';
        if (kvb64) {
            if (p.softclamp)
                hipLaunchKernelGGL((v2::attn_fwd_v2_kernel<64, 64, true>), grid, block, 0, stream, p);
            else
                hipLaunchKernelGGL((v2::attn_fwd_v2_kernel<64, 64, false>), grid, block, 0, stream, p);
        } else if (p.softclamp)
            hipLaunchKernelGGL((v2::attn_fwd_v2_kernel<64, 128, true>), grid, block, 0, stream, p);
        else
            hipLaunchKernelGGL((v2::attn_fwd_v2_kernel<64, 128, false>), grid, block, 0, stream, p);
        return true;
    }
    if (head_dim == 128) {
        if (p.softclamp)
            hipLaunchKernelGGL((v2::attn_fwd_v2_kernel<128, 64, true>), grid, block, 0, stream, p);
        else
            hipLaunchKernelGGL((v2::attn_fwd_v2_kernel<128, 64, false>), grid, block, 0, stream, p);
        return true;
    }
    return false;
}

}  // namespace ring_attn
