// CDNA4 (gfx950) flash-attention BACKWARD kernel (recompute-based).
//
// Capability counterpart of the reference's Triton _bwd_kernel
// (/root/reference/ring_attention_pytorch/triton_flash_attn.py:509-1128),
// re-designed for wave64/MFMA; no code ported.  Differences by design:
//   * column-parallel over KV tiles: one workgroup owns a 256-row KV tile
//     (8 waves x 32 kv rows); K and V fragments live in REGISTERS across the
//     whole Q loop.  dk/dv accumulate in registers, written once at the end
//     (the reference re-stored per column block and needed debug barriers).
//   * dq is accumulated with fp32 global atomics (CDNA global_atomic_add_f32),
//     coalesced along the head dim — no serialized load-modify-store variant
//     needed, and no cross-workgroup ordering hazards by construction.
//   * GQA: the workgroup iterates the group's query heads with K/V resident,
//     so dk/dv need no cross-head reduction at all.
//   * causality/striping/lookback use the same (diag, win) integer reduction
//     as the forward kernel.
//   * softclamp backward applies the dtanh factor exactly as the oracle
//     (ops/ring_flash.py backward).
//
// Per (q 32-row block):  S2[q][kv] = mfma(A=Q_lds, B=K_regs)      (lane = kv)
//                        dP[q][kv] = mfma(A=dO_lds, B=V_regs)
//   p = exp(s*scale - lse); ds = p*(dP - delta)*scale (*dtanh)
//   pack p, ds -> bf16 fragments (cvt_pk + permlane32_swap, lane = kv)
//   dv^T += mfma(A=dOT_lds, B=p_frag)                              (lane = kv)
//   dk   += mfma(A=ds_frag, B=QT_lds)                              (lane = d)
//   ds -> LDS [q][kv]; dq += mfma(A=ds_lds, B=KT_lds) -> atomicAdd (lane = d)
//
// Output layouts (host transposes once at the end of the ring):
//   dq: fp32 (B, Nq, H, D)   — atomically accumulated across tiles AND hops
//   dk: fp32 (B, HK, Nk, D)  — plain writes/adds (one WG owns each row)
//   dv: fp32 (B, HK, D, Nk)  — transposed scratch, coalesced from dv^T regs

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

#include "attn_common.h"

namespace ring_attn {

static constexpr int BWD_WAVES = 8;
static constexpr int KVROWS_WAVE = 32;
static constexpr int KVROWS_WG = BWD_WAVES * KVROWS_WAVE;   // 256

__device__ __forceinline__ int bswz(int row, int chunk) { return chunk ^ (row & 7); }

template <int D, int QT>
struct BwdLds {
    __align__(16) __bf16 q[QT * D];        // [q][d]   swizzled rows
    __align__(16) __bf16 qt[D * QT];       // [d][q]   swizzled rows
    __align__(16) __bf16 do_[QT * D];      // [q][d]
    __align__(16) __bf16 dot[D * QT];      // [d][q]
    __align__(16) __bf16 kt[D * KVROWS_WG]; // [d][kv]  whole-WG K^T, swizzled
    __align__(16) __bf16 ds[BWD_WAVES][QT * KVROWS_WAVE];  // per-wave [q][kv32]
    float lse[QT];
    float delta[QT];
};

// XOR swizzle valid for rows of CH chunks (CH a power of two, <= 8 kept)
template <int CH>
__device__ __forceinline__ int tmask() { return (CH - 1) < 7 ? (CH - 1) : 7; }

// stage a [rows][D] bf16 tile row-major (swizzled) AND transposed (swizzled)
template <int D, int QT>
__device__ void stage_rowmajor_and_t(
    const __bf16* gbase, long row0, long rowmax, long row_stride,
    __bf16* lds_rm, __bf16* lds_t, int tid) {
    constexpr int CH_PER_ROW = D * 2 / 16;
    constexpr int TM = (QT / 8 - 1) < 7 ? (QT / 8 - 1) : 7;   // transposed-row chunk mask
    for (int c = tid; c < QT * CH_PER_ROW; c += 512) {
        int row = c / CH_PER_ROW, ch = c % CH_PER_ROW;
        long gr = row0 + row;
        uint4 val = (gr <= rowmax) ? *(const uint4*)(gbase + gr * row_stride + ch * 8)
                                   : uint4{0, 0, 0, 0};
        *(uint4*)(lds_rm + row * D + bswz(row, ch) * 8) = val;
        // transposed image: 8 bf16 of one row -> 8 columns of lds_t
        const __bf16* vals = (const __bf16*)&val;
        #pragma unroll
        for (int e = 0; e < 8; ++e) {
            int d = ch * 8 + e;
            int byte_off = d * QT * 2 + ((row * 2) ^ ((d & TM) << 4));
            *((__bf16*)((char*)lds_t + byte_off)) = vals[e];
        }
    }
}

template <int D, int QT>
__global__ __launch_bounds__(512, 1) void attn_bwd_kernel(BwdParams p) {
    static_assert(D % 32 == 0 && QT % 32 == 0);
    constexpr int DBLK = D / 32;
    constexpr int KSTEPS = D / 16;
    constexpr int QBLKS = QT / 32;

    __shared__ BwdLds<D, QT> lds;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l31 = lane & 31;
    const int lhi = lane >> 5;

    const int kvtile = blockIdx.x;
    const int bhk = blockIdx.y;
    const int b = bhk / p.hk;
    const int hkh = bhk % p.hk;

    const long j0_wg = (long)kvtile * KVROWS_WG;
    const long jmax = min(j0_wg + KVROWS_WG, p.nk) - 1;
    const long j = j0_wg + wid * KVROWS_WAVE + l31;     // this lane's kv row
    const bool col_valid = j <= jmax;
    const long j_clamped = col_valid ? j : j0_wg;

    // ---- K, V fragments in registers: [kv = l31][d = ks*16 + lhi*8 ..+8]
    const __bf16* kb = (const __bf16*)p.k + ((long)b * p.nk + j_clamped) * p.hk * D + (long)hkh * D;
    const __bf16* vb = (const __bf16*)p.v + ((long)b * p.nk + j_clamped) * p.hk * D + (long)hkh * D;
    bf16x8 kf[KSTEPS], vf[KSTEPS];
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
        kf[ks] = *(const bf16x8*)(kb + ks * 16 + lhi * 8);
        vf[ks] = *(const bf16x8*)(vb + ks * 16 + lhi * 8);
    }
    unsigned char kmask_own = 1;
    if (p.kmask) kmask_own = col_valid ? ((const unsigned char*)p.kmask)[(long)b * p.nk + j] : 0;

    // ---- stage WG-wide K^T image [D][256]
    {
        const __bf16* kwg = (const __bf16*)p.k + ((long)b * p.nk) * p.hk * D + (long)hkh * D;
        constexpr int CH_PER_ROW = D * 2 / 16;
        for (int c = tid; c < KVROWS_WG * CH_PER_ROW; c += 512) {
            int row = c / CH_PER_ROW, ch = c % CH_PER_ROW;
            long gr = j0_wg + row;
            uint4 val = (gr <= jmax) ? *(const uint4*)(kwg + gr * p.hk * D + ch * 8)
                                     : uint4{0, 0, 0, 0};
            const __bf16* vals = (const __bf16*)&val;
            #pragma unroll
            for (int e = 0; e < 8; ++e) {
                int d = ch * 8 + e;
                int byte_off = d * KVROWS_WG * 2 + ((row * 2) ^ ((d & 7) << 4));
                *((__bf16*)((char*)lds.kt + byte_off)) = vals[e];
            }
        }
    }

    // dv^T accum [dblk][16] (j = lane = kv), dk accum [dblk][16] (j = lane = d)
    f32x16 dv_acc[DBLK], dk_acc[DBLK];
    #pragma unroll
    for (int db = 0; db < DBLK; ++db) { dv_acc[db] = f32x16{}; dk_acc[db] = f32x16{}; }

    const int num_q_tiles = (int)((p.nq + QT - 1) / QT);

    for (int g = 0; g < p.group; ++g) {
        const int h = hkh * p.group + g;
        const float* lse_row = p.lse + ((long)b * p.h + h) * p.nq;
        const float* delta_row = p.delta + ((long)b * p.h + h) * p.nq;
        const __bf16* qg = (const __bf16*)p.q + ((long)b * p.nq) * p.h * D + (long)h * D;
        const __bf16* dog = (const __bf16*)p.dout + ((long)b * p.nq) * p.h * D + (long)h * D;
        float* dqg = p.dq + ((long)b * p.nq) * p.h * D + (long)h * D;

        // q-tile range limited by causality / window for this kv tile
        int t0 = 0, t1 = num_q_tiles;
        if (p.causal) {
            long i_min_needed = j0_wg - p.diag;              // need i >= j - diag
            t0 = (int)max(0L, i_min_needed / QT);
        }
        if (p.has_win) {
            long i_max_needed = jmax + p.win;                // need i <= j + win
            t1 = (int)min((long)num_q_tiles, i_max_needed / QT + 1);
        }

        for (int t = t0; t < t1; ++t) {
            const long i0 = (long)t * QT;
            const long imax = min(i0 + QT, p.nq) - 1;
            const bool full_tile =
                (imax - i0 == QT - 1) &&
                (!p.causal || (i0 - (jmax - p.diag)) >= 0) &&     // every i >= j - diag
                (!p.has_win || ((imax - j0_wg) <= p.win)) &&
                !p.kmask;

            __syncthreads();   // protect LDS from previous iteration's readers
            stage_rowmajor_and_t<D, QT>(qg, i0, imax, (long)p.h * D, lds.q, lds.qt, tid);
            stage_rowmajor_and_t<D, QT>(dog, i0, imax, (long)p.h * D, lds.do_, lds.dot, tid);
            for (int c = tid; c < QT; c += 512) {
                long gi = i0 + c;
                lds.lse[c] = (gi <= imax) ? lse_row[gi] : 0.f;
                lds.delta[c] = (gi <= imax) ? delta_row[gi] : 0.f;
            }
            __syncthreads();

            #pragma unroll
            for (int qb = 0; qb < QBLKS; ++qb) {
                // ---- S2[q][kv], dP[q][kv]: lane = kv, q rows in regs
                f32x16 s2 = f32x16{}, dp = f32x16{};
                #pragma unroll
                for (int ks = 0; ks < KSTEPS; ++ks) {
                    int qrow = qb * 32 + l31;
                    int chunk = ks * 2 + lhi;
                    // A operand rows = q (lane = l31 selects q row)
                    bf16x8 qa = *(const bf16x8*)(lds.q + qrow * D + bswz(qrow, chunk) * 8);
                    bf16x8 da = *(const bf16x8*)(lds.do_ + qrow * D + bswz(qrow, chunk) * 8);
                    s2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kf[ks], s2, 0, 0, 0);
                    dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, vf[ks], dp, 0, 0, 0);
                }

                // ---- p, ds per reg (q row = i0 + qb*32 + pattern, kv = j)
                uint32_t p_pk[8], ds_pk[8];
                #pragma unroll
                for (int x2 = 0; x2 < 8; ++x2) {
                    float pe[2], dse[2];
                    #pragma unroll
                    for (int e = 0; e < 2; ++e) {
                        int r = 2 * x2 + e;
                        int qloc = qb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                        long i = i0 + qloc;
                        float x = s2[r] * p.scale;
                        float dtanh = 1.f;
                        if (p.softclamp) {
                            x = p.softclamp_value * tanhf(x / p.softclamp_value);
                            dtanh = 1.f - (x / p.softclamp_value) * (x / p.softclamp_value);
                        }
                        bool ok = col_valid && i <= imax;
                        if (!full_tile) {
                            if (p.causal) ok = ok && (j <= i + p.diag);
                            if (p.has_win) ok = ok && (i - j <= p.win);
                            if (p.kmask) ok = ok && kmask_own;
                        }
                        if (!ok) x = MASK_VALUE_F;
                        float pv = __expf(x - lds.lse[qloc]);
                        if (!ok) pv = 0.f;   // exact zero even for garbage lse rows
                        pe[e] = pv;
                        dse[e] = pv * (dp[r] - lds.delta[qloc]) * dtanh * p.scale;
                    }
                    union { __hip_bfloat162 h2; uint32_t u; } c1, c2;
                    c1.h2 = __float22bfloat162_rn(float2{pe[0], pe[1]});
                    c2.h2 = __float22bfloat162_rn(float2{dse[0], dse[1]});
                    p_pk[x2] = c1.u;
                    ds_pk[x2] = c2.u;
                }

                // ---- fragments (lane = kv, k = q contiguous): swap pairs +2
                uint32_t p_frag[2][4], ds_frag[2][4];
                #pragma unroll
                for (int half = 0; half < 2; ++half) {
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
                }

                // ---- dv^T[d][kv] += dO^T x p ; dk[kv][d] += ds^T x Q^T-read
                #pragma unroll
                for (int db = 0; db < DBLK; ++db) {
                    #pragma unroll
                    for (int half = 0; half < 2; ++half) {
                        int qk = qb * 2 + half;                  // 16-q k-step index
                        int drow = db * 32 + l31;
                        // dO^T row = d, contiguous q at qk*16 + lhi*8
                        constexpr int TM = (QT / 8 - 1) < 7 ? (QT / 8 - 1) : 7;
                        int ch_dot = qk * 2 + lhi;
                        bf16x8 doa = *(const bf16x8*)(lds.dot + drow * QT +
                                                      ((ch_dot ^ (drow & TM))) * 8);
                        dv_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            doa, *(const bf16x8*)p_frag[half], dv_acc[db], 0, 0, 0);
                        // Q^T as B: lane j = d column; read Q^T row (d) contiguous q
                        bf16x8 qta = *(const bf16x8*)(lds.qt + drow * QT +
                                                      ((ch_dot ^ (drow & TM))) * 8);
                        dk_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            *(const bf16x8*)ds_frag[half], qta, dk_acc[db], 0, 0, 0);
                    }
                }

                // ---- ds -> LDS [q][kv32] (lane = kv writes its column pairs)
                __bf16* dsw = lds.ds[wid];
                #pragma unroll
                for (int x2 = 0; x2 < 8; ++x2) {
                    int r = 2 * x2;
                    int qloc = qb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                    // rows qloc and qloc+1 hold ds_pk[x2] lo/hi for kv = l31
                    union { uint32_t u; __bf16 h[2]; } val; val.u = ds_pk[x2];
                    dsw[qloc * KVROWS_WAVE + l31] = val.h[0];
                    dsw[(qloc + 1) * KVROWS_WAVE + l31] = val.h[1];
                }
            }
            __syncthreads();   // ds images complete

            // ---- dq[q][d] += ds x K (contract over this wave's 32 kv), atomic
            #pragma unroll
            for (int qb = 0; qb < QBLKS; ++qb) {
                f32x16 dq_acc[DBLK];
                #pragma unroll
                for (int db = 0; db < DBLK; ++db) dq_acc[db] = f32x16{};
                #pragma unroll
                for (int ks = 0; ks < 2; ++ks) {                  // 2 x 16 kv of this wave
                    int qrow = qb * 32 + l31;
                    bf16x8 dsa = *(const bf16x8*)(lds.ds[wid] + qrow * KVROWS_WAVE +
                                                  ks * 16 + lhi * 8);
                    #pragma unroll
                    for (int db = 0; db < DBLK; ++db) {
                        // K^T image: B[k=kv][j=d]: lane j = d col; row = d, contig
                        // kv; mirror the (d&7)<<4 byte-XOR used on the write side
                        int drow = db * 32 + l31;
                        int kvoff = wid * KVROWS_WAVE + ks * 16 + lhi * 8;
                        bf16x8 kta = *(const bf16x8*)((char*)lds.kt +
                                                      drow * KVROWS_WG * 2 +
                                                      ((kvoff * 2) ^ ((drow & 7) << 4)));
                        dq_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                            dsa, kta, dq_acc[db], 0, 0, 0);
                    }
                }
                #pragma unroll
                for (int db = 0; db < DBLK; ++db)
                    #pragma unroll
                    for (int r = 0; r < 16; ++r) {
                        int qloc = qb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                        long gi = i0 + qloc;
                        if (gi <= imax) {
                            int d = db * 32 + l31;
                            atomicAdd(dqg + gi * p.h * D + d, dq_acc[db][r]);
                        }
                    }
            }
        }
    }

    // ---- write dk (B,HK,Nk,D) and dv^T (B,HK,D,Nk)
    if (col_valid) {
        float* dkb = p.dk + (((long)b * p.hk + hkh) * p.nk) * D;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                // dk: i = kv pattern rows, j = lane = d
                long kvrow = j0_wg + wid * KVROWS_WAVE + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                int d = db * 32 + l31;
                if (kvrow <= jmax) {
                    float* dst = dkb + kvrow * D + d;
                    if (p.accumulate) *dst += dk_acc[db][r]; else *dst = dk_acc[db][r];
                }
            }
        float* dvb = p.dv + (((long)b * p.hk + hkh) * D) * p.nk;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                // dv^T: i = d pattern rows, j = lane = kv
                int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                float* dst = dvb + (long)d * p.nk + j;
                if (p.accumulate) *dst += dv_acc[db][r]; else *dst = dv_acc[db][r];
            }
    }
}

void launch_attn_bwd(const BwdParams& p, int head_dim, hipStream_t stream) {
    dim3 grid((p.nk + KVROWS_WG - 1) / KVROWS_WG, p.b * p.hk);
    dim3 block(512);
    if (head_dim == 64) {
        hipLaunchKernelGGL((attn_bwd_kernel<64, 64>), grid, block, 0, stream, p);
    } else if (head_dim == 128) {
        hipLaunchKernelGGL((attn_bwd_kernel<128, 32>), grid, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}

}  // namespace ring_attn
