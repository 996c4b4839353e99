// CDNA4 (gfx950) flash-attention FORWARD kernel with online-softmax
// resume across ring passes.
//
// Brand-new MI355X design (capability counterpart of the reference's Triton
// _fwd_kernel, /root/reference/ring_attention_pytorch/triton_flash_attn.py:52-430,
// re-thought for wave64 + MFMA; no code ported):
//
//   * 8 waves / 512 threads per workgroup; each wave owns 32 Q rows
//     (256-row Q tile per workgroup), KV tile = KVBLK (128 at d64).
//   * swapped QK^T: S^T[kv][q] = mfma(A=K, B=Q^T) so each lane holds a full
//     slice of ONE q row's scores -> softmax is almost entirely in-register
//     (31 VALU max/sum + one cross-half __shfl_xor), no LDS round trip.
//   * P -> bf16 via v_cvt_pk_bf16_f32 pairs + v_permlane32_swap to build the
//     PV B-operand fragments in-register (T12 pattern).
//   * K and V^T staged in LDS with an XOR-16B swizzle (conflict-free
//     ds_read_b128 column slices); 3-deep pipeline over DOUBLE-buffered LDS
//     (one barrier per tile; staging overlaps the MFMAs).
//   * causality, striping and lookback reduce to integers: with
//     qpos(i) = i*q_stride + diag, attend(i,j) <=> j <= qpos(i) AND
//     qpos(i) - j <= win (host folds ring-rank offsets / layout strides;
//     q_stride > 1 expresses a striped q shard against gathered global KV).
//   * resume contract: fp32 o_acc (B,H,D,Nq transposed scratch), m, l
//     (B,H,Nq) persist between ring hops; IS_FIRST initializes instead of
//     loading, IS_LAST normalizes and emits bf16 out (B,Nq,H,D) + lse.
//     Single-hop launches (IS_FIRST && IS_LAST) never touch o_acc at all.
//
// Verified fragment maps (csrc/tools/mfma_probe.hip, run on MI355X):
//   mfma_f32_32x32x16_bf16: A(i=l&31, k=8*(l>>5)+r)  B(k=8*(l>>5)+r, j=l&31)
//                           C(i=(r&3)+8*(r>>2)+4*(l>>5), j=l&31)
//   permlane32_swap: r0 = {lo: vdst.lo, hi: src.lo}, r1 = {lo: vdst.hi, hi: src.hi}

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>
#include <cstdlib>

#include "attn_common.h"

namespace ring_attn {

// ---------------------------------------------------------------------------
// geometry
// ---------------------------------------------------------------------------
// QROWS_WG = 256 (8 waves x 32 rows), KVBLK = 64
static constexpr int WAVES = 8;          // 8-wave WGs: the 2-waves/SIMD paired
static constexpr int QROWS_WAVE = 32;    // regime (4-wave variant measured slower)
static constexpr int QROWS_WG = WAVES * QROWS_WAVE;
template <int D> constexpr int fwd_kvblk() { return D == 64 ? 128 : 64; }  // LDS budget
static constexpr int NTHREADS = WAVES * 64;

// XOR swizzle of a 16-byte chunk index within a row (row stride D*2 bytes):
// chunk' = chunk ^ (row & 7).  Applied identically on the staging write and
// the fragment read, so the LDS image is consistent (both-sides rule).
template <int CH>
__device__ __forceinline__ int swz(int row, int chunk) {
    // swizzle mask must stay inside the row: CH chunks per row (CH-1 when
    // CH <= 8; wider rows keep the measured 8-slot spread)
    return chunk ^ (row & (CH < 8 ? CH - 1 : 7));
}

// fast tanh from builtins: tanh(x) = 1 - 2/(exp2(2x*log2e) + 1); avoids the
// libm tanhf whose inlined body injects v_div_* sequences into the hot loop
__device__ __forceinline__ float fast_tanhf(float x) {
    float e = __builtin_amdgcn_exp2f(x * 2.885390081777927f);   // 2*log2(e)
    return 1.f - 2.f * __builtin_amdgcn_rcpf(e + 1.f);
}

// cross-half (lane <-> lane^32) exchange via v_permlane32_swap — pure VALU,
// no LDS round trip (a __shfl_xor(x, 32) lowers to ds_bpermute + addressing)
__device__ __forceinline__ float cross_half(float x) {
    union { float f; unsigned u; } c; c.f = x;
    u32x2 r = __builtin_amdgcn_permlane32_swap(c.u, c.u, false, false);
    // r0 = {lo: own lo, hi: own lo}; r1 = {lo: own hi, hi: own hi}
    union { unsigned u; float f; } lo, hi; lo.u = r[0]; hi.u = r[1];
    // lanes < 32 want the partner's value = own hi-half's value = r1.lo;
    // lanes >= 32 want r0.hi = own lo-half's value — both are "the other
    // half's x" exactly when we select r1 on lo lanes and r0 on hi lanes
    return (threadIdx.x & 32) ? lo.f : hi.f;
}

template <int D>
struct FwdLds {
    static constexpr int KVB = fwd_kvblk<D>();
    // double-buffered: K tile [kv][D] + V^T tile [d][kv], 16B-chunk swizzled
    __align__(16) __bf16 k[2][KVB * D];
    __align__(16) __bf16 vt[2][D * KVB];
    unsigned char kmask[2][KVB];
};

static constexpr float LOG2E = 1.4426950408889634f;
static constexpr float LN2 = 0.6931471805599453f;

template <class F>
__device__ __attribute__((noinline)) void fwd_noinline_call(F&& f) { f(); }

// ---------------------------------------------------------------------------
// forward kernel
// ---------------------------------------------------------------------------
template <int D, bool SOFTCLAMP, bool PAIRED>
__global__ __launch_bounds__(NTHREADS, 1) void attn_fwd_kernel(FwdParams p) {
    static_assert(D % 32 == 0);
    constexpr int DBLK = D / 32;     // 32-d output blocks
    constexpr int KSTEPS = D / 16;   // QK^T k-steps
    constexpr int KVBLK = fwd_kvblk<D>();
    constexpr int NBLK = KVBLK / 32;

    __shared__ FwdLds<D> lds;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int l31 = lane & 31;
    const int lhi = lane >> 5;        // 0 or 1

    const int bh = blockIdx.y;
    const int b = bh / p.h;
    const int h = bh % p.h;
    const int hk = h % p.hk;          // kv head (reference tile GQA: qh pairs qh % hk)

    // causal pairing: WG x runs q-tiles (x, T-1-x) — per-WG work is the
    // uniform T+1 tiles instead of the 2:1 triangle imbalance
    const int n_pit = PAIRED
        ? (p.paired - 1 - (int)blockIdx.x == (int)blockIdx.x ? 1 : 2) : 1;
    for (int pit = 0; pit < n_pit; ++pit) {
    const int qtile = PAIRED
        ? (pit == 0 ? (int)blockIdx.x : p.paired - 1 - (int)blockIdx.x)
        : (int)blockIdx.x;
    if (PAIRED && pit == 1) {
        __syncthreads();                       // LDS handoff between tiles
    }

    auto fwd_body = [&]() {
    const FwdParams P = p;   // register-local copy (see dkv_noinline_call):
                             // the noinline frame would otherwise re-read
                             // fields through scratch
    const long irow0 = (long)qtile * QROWS_WG + wid * QROWS_WAVE;  // this wave's first q row
    const long i = irow0 + l31;                                     // this lane's q row
    const bool row_valid = i < P.nq;
    const long i_clamped = row_valid ? i : 0;

    // ---- load Q fragments (bf16x8 per k-step): q[b, i, h, ks*16 + lhi*8 .. +8]
    const __bf16* qbase = (const __bf16*)P.q + ((long)b * P.nq + i_clamped) * P.h * D + (long)h * D;
    bf16x8 qf[KSTEPS];
    #pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
        qf[ks] = *(const bf16x8*)(qbase + ks * 16 + lhi * 8);

    // ---- accumulators
    float m_run = MASK_VALUE_F;   // running row max of scaled scores (lane's q row)
    float l_run = 0.f;            // running row sum
    f32x16 o_acc[DBLK];           // O^T[d][q]: j = l31 = q, rows = d pattern
    #pragma unroll
    for (int db = 0; db < DBLK; ++db) o_acc[db] = f32x16{};

    const bool split_mode = P.kv_split > 1;
    const int zsplit = blockIdx.z;
    if (!P.is_first && !split_mode) {   // resume from a previous ring hop
        const float* mrow = P.m + ((long)b * P.h + h) * P.nq;
        const float* lrow = P.l + ((long)b * P.h + h) * P.nq;
        m_run = mrow[i_clamped] * LOG2E;   // external contract is natural log
        l_run = lrow[i_clamped];
        const float* oa = P.o_acc + (((long)b * P.h + h) * D) * P.nq;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                o_acc[db][r] = oa[(long)d * P.nq + i_clamped];
            }
    }

    // ---- causal / lookback tile bounds for this workgroup: the masked
    // region is contiguous, so the tile range is computed once (keeps the
    // staging pipeline branch-free)
    // q positions in kv-local coordinates: qpos(i) = i * q_stride + diag
    const long wg_i_min = (long)qtile * QROWS_WG;
    const long wg_i_max = min((long)(qtile + 1) * QROWS_WG, P.nq) - 1;
    const long wg_q_min = wg_i_min * P.q_stride + P.diag;
    const long wg_q_max = wg_i_max * P.q_stride + P.diag;
    const long qpos_i = i * P.q_stride + P.diag;      // this lane's q position
    const int num_kv_tiles = (int)((P.nk + KVBLK - 1) / KVBLK);

    int t_lo = 0, t_hi = num_kv_tiles;
    if (P.causal) {
        t_hi = wg_q_max < 0 ? 0 : min((long)num_kv_tiles, wg_q_max / KVBLK + 1);
    }
    if (P.has_win) {
        // tile t attends iff j0 + KVBLK - 1 >= wg_q_min - win
        long x = wg_q_min - P.win - KVBLK + 1;
        t_lo = x <= 0 ? 0 : (int)((x + KVBLK - 1) / KVBLK);
        if (t_lo > t_hi) t_lo = t_hi;
    }
    if (split_mode) {
        // this z's share of THIS WG's valid tile range (fractional split:
        // a global-range split is skewed against the causal trapezoid)
        int valid = t_hi - t_lo;
        int per_split = (valid + P.kv_split - 1) / P.kv_split;
        int base = t_lo;
        t_lo = base + min(valid, zsplit * per_split);
        t_hi = base + min(valid, (zsplit + 1) * per_split);
    }

    // ---- T14 async-stage split: per-thread staging registers
    constexpr int CH_PER_ROW = D * 2 / 16;
    constexpr int KCHUNKS = KVBLK * CH_PER_ROW;          // K-tile 16B chunks
    constexpr int KREGS = (KCHUNKS + NTHREADS - 1) / NTHREADS;
    constexpr int VPAIRS = (KVBLK / 2) * (D / 8);
    constexpr int VREGS = (VPAIRS + NTHREADS - 1) / NTHREADS;
    const __bf16* kbase = (const __bf16*)P.k + ((long)b * P.nk) * P.hk * D + (long)hk * D;
    const __bf16* vbase = (const __bf16*)P.v + ((long)b * P.nk) * P.hk * D + (long)hk * D;
    const unsigned char* mbase = P.kmask ? (const unsigned char*)P.kmask + (long)b * P.nk : nullptr;

    uint4 kst[KREGS];
    bf16x8 vsta[VREGS], vstb[VREGS];
    unsigned char mst = 1;

    // running per-thread source pointers: load_tile is always called for
    // consecutive tiles, so addresses advance by a constant — no per-tile
    // 64-bit multiplies in the hot loop
    const long kv_row_stride = (long)P.hk * D;
    const long tile_stride = KVBLK * kv_row_stride;
    const __bf16* kptr = kbase + (long)t_lo * tile_stride
        + (tid / CH_PER_ROW) * kv_row_stride + (tid % CH_PER_ROW) * 8;
    const __bf16* vptra = vbase + (long)t_lo * tile_stride
        + ((tid % (KVBLK / 2)) * 2) * kv_row_stride + (tid / (KVBLK / 2)) * 8;
    const __bf16* vptrb = vptra + kv_row_stride;
    long j0_next = (long)t_lo * KVBLK;

    auto load_tile = [&]() {
        const long j0 = j0_next;
        const long jmax = min(j0 + KVBLK, P.nk) - 1;
        const bool full = jmax - j0 == KVBLK - 1;
        #pragma unroll
        for (int r = 0; r < KREGS; ++r) {
            int c = tid + r * NTHREADS;
            if (c < KCHUNKS) {
                const __bf16* src = kptr + (long)(r * (NTHREADS / CH_PER_ROW)) * kv_row_stride;
                kst[r] = (full || (j0 + c / CH_PER_ROW) <= jmax)
                         ? *(const uint4*)src : uint4{0, 0, 0, 0};
            }
        }
        #pragma unroll
        for (int r = 0; r < VREGS; ++r) {
            int c = tid + r * NTHREADS;
            if (c < VPAIRS) {
                // extra r steps advance along d (same kv pair)
                const __bf16* sa = vptra + r * (NTHREADS / (KVBLK / 2)) * 8;
                const __bf16* sb = vptrb + r * (NTHREADS / (KVBLK / 2)) * 8;
                long ja = j0 + (c % (KVBLK / 2)) * 2;
                vsta[r] = (full || ja <= jmax) ? *(const bf16x8*)sa : bf16x8{};
                vstb[r] = (full || ja + 1 <= jmax) ? *(const bf16x8*)sb : bf16x8{};
            }
        }
        if (mbase && tid < KVBLK)
            mst = (j0 + tid <= jmax) ? mbase[j0 + tid] : 0;
        kptr += tile_stride;
        vptra += tile_stride;
        vptrb += tile_stride;
        j0_next += KVBLK;
    };

    auto write_tile = [&](int par) {
        #pragma unroll
        for (int r = 0; r < KREGS; ++r) {
            int c = tid + r * NTHREADS;
            if (c < KCHUNKS) {
                int row = c / CH_PER_ROW, ch = c % CH_PER_ROW;
                *(uint4*)(lds.k[par] + row * D + swz<D / 8>(row, ch) * 8) = kst[r];
            }
        }
        #pragma unroll
        for (int r = 0; r < VREGS; ++r) {
            int c = tid + r * NTHREADS;
            if (c < VPAIRS) {
                int jp = c % (KVBLK / 2);
                int d0 = (c / (KVBLK / 2)) * 8;
                #pragma unroll
                for (int e = 0; e < 8; ++e) {
                    int d = d0 + e;
                    int byte_off = d * KVBLK * 2 + ((jp * 4) ^ ((d & 7) << 4));
                    __bf16 pair[2] = {vsta[r][e], vstb[r][e]};
                    *(uint32_t*)((char*)lds.vt[par] + byte_off) = *(uint32_t*)pair;
                }
            }
        }
        if (mbase && tid < KVBLK) lds.kmask[par][tid] = mst;
    };

    // 3-deep pipeline over DOUBLE-buffered LDS: one barrier per tile; LDS
    // writes and the next-next tile's HBM loads fully overlap the MFMAs
    const float scale2 = P.scale * LOG2E;    // softmax runs in the exp2 domain
    if (t_lo < t_hi) {
        load_tile();
        write_tile(t_lo & 1);
        if (P.ablate == 1) write_tile((t_lo & 1) ^ 1);   // both buffers valid
        if (t_lo + 1 < t_hi && P.ablate != 1) load_tile();
    }

    for (int t = t_lo; t < t_hi; ++t) {
        const int par = t & 1;
        const long j0 = (long)t * KVBLK;
        const long jmax = min(j0 + KVBLK, P.nk) - 1;
        const bool full_tile =
            (jmax - j0 == KVBLK - 1) &&
            (!P.causal || jmax <= wg_q_min) &&
            (!P.has_win || (wg_q_max - j0) <= P.win) &&
            !P.kmask && !P.bias;

        __syncthreads();
        const bool stamp = P.ticks && blockIdx.x == 0 && bh == 0 && tid == 0
                           && blockIdx.z == 0;
        if (stamp) P.ticks[t * 6 + 0] = __builtin_amdgcn_s_memtime();

        // ---- QK^T: S^T[kv][q] for the NBLK 32-row kv blocks
        f32x16 s[NBLK];
        #pragma unroll
        for (int kb = 0; kb < NBLK; ++kb) s[kb] = f32x16{};
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int kb = 0; kb < NBLK; ++kb) {
            int krow = kb * 32 + l31;
            #pragma unroll
            for (int ks = 0; ks < KSTEPS; ++ks) {
                int chunk = ks * 2 + lhi;
                bf16x8 kf = *(const bf16x8*)(lds.k[par] + krow * D + swz<D / 8>(krow, chunk) * 8);
                s[kb] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[ks], s[kb], 0, 0, 0);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        if (stamp) P.ticks[t * 6 + 1] = __builtin_amdgcn_s_memtime();

        // stage tile t+1 into the other buffer while the MFMAs above retire
        if (P.ablate != 1) {
            if (t + 1 < t_hi) write_tile(par ^ 1);
            if (t + 2 < t_hi) load_tile();
        }
        if (stamp) P.ticks[t * 6 + 2] = __builtin_amdgcn_s_memtime();

        if (P.ablate == 2) {
            // diagnostics: skip softmax VALU, feed PV garbage fragments kept
            // alive via asm (rule 17: a skipped phase must not DCE upstream)
            uint32_t gfrag[4];
            #pragma unroll
            for (int c = 0; c < 4; ++c) {
                union { float f; uint32_t u; } cv; cv.f = s[0][c];
                gfrag[c] = cv.u;
                asm volatile("" :: "v"(gfrag[c]));
            }
            #pragma unroll
            for (int db = 0; db < DBLK; ++db) {
                int drow = db * 32 + l31;
                #pragma unroll
                for (int ks = 0; ks < NBLK * 2; ++ks) {
                    int chunk = ks * 2 + lhi;
                    bf16x8 vf = *(const bf16x8*)(lds.vt[par] + drow * KVBLK + swz<KVBLK / 8>(drow, chunk) * 8);
                    o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        vf, *(const bf16x8*)gfrag, o_acc[db], 0, 0, 0);
                }
            }
            continue;
        }

        // ---- scale (exp2 domain), clamp, mask in place.  The full-tile
        // variant must contain NO per-element conditions: a condition inside
        // the unrolled loop is PREDICATED (cndmask per element, executed on
        // every tile) rather than branched — hoist to one scalar branch.
        float smax = MASK_VALUE_F;
        if (full_tile) {
            if constexpr (!SOFTCLAMP) {
                // exp2+fma fold: fmax runs on RAW scores (scale2 > 0 is
                // monotone); the scale is folded into the exp argument as a
                // single v_fma below — saves one VALU per element
                #pragma unroll
                for (int kb = 0; kb < NBLK; ++kb)
                    #pragma unroll
                    for (int r = 0; r < 16; ++r)
                        smax = fmaxf(smax, s[kb][r]);
                smax *= scale2;
            } else {
                #pragma unroll
                for (int kb = 0; kb < NBLK; ++kb)
                    #pragma unroll
                    for (int r = 0; r < 16; ++r) {
                        float xs = s[kb][r] * (P.scale * __builtin_amdgcn_rcpf(P.softclamp_value));
                        float x = P.softclamp_value * fast_tanhf(xs) * LOG2E;
                        s[kb][r] = x;
                        smax = fmaxf(smax, x);
                    }
            }
        } else {
            #pragma unroll
            for (int kb = 0; kb < NBLK; ++kb)
                #pragma unroll
                for (int r = 0; r < 16; ++r) {
                    float x;
                    if constexpr (SOFTCLAMP) {
                        float xs = s[kb][r] * (P.scale * __builtin_amdgcn_rcpf(P.softclamp_value));
                        x = P.softclamp_value * fast_tanhf(xs) * LOG2E;
                    } else {
                        x = s[kb][r] * scale2;
                    }
                    long j = j0 + kb * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                    bool ok = j <= jmax;
                    if (P.bias && ok) {
                        // reference semantics: sim = qk*scale + bias
                        // (natural log), applied after softclamp; our
                        // softmax runs in the exp2 domain
                        const long bi = P.bias_mat
                            ? (((long)b * P.h + h) * P.nq + i_clamped) * P.nk + j
                            : ((long)b * P.h + h) * P.nk + j;
                        x += P.bias[bi] * LOG2E;
                    }
                    if (P.causal) ok = ok && (j <= qpos_i);
                    if (P.has_win) ok = ok && (qpos_i - j <= P.win);
                    if (P.kmask) ok = ok && lds.kmask[par][j - j0];
                    if (!ok) x = MASK_VALUE_F;
                    s[kb][r] = x;
                    smax = fmaxf(smax, x);
                }
        }
        smax = fmaxf(smax, cross_half(smax));
        if (stamp) P.ticks[t * 6 + 3] = __builtin_amdgcn_s_memtime();

        // ---- online softmax update (defer-max THR=0: exact — skip the O
        // rescale whenever the running max did not grow on any lane)
        float m_new = fmaxf(m_run, smax);
        const bool any_growth = !__all(smax <= m_run);
        // all-masked rows: m_new == MASK -> exp2(x - m_new) would be 1 for
        // every masked element (out = mean(V) instead of 0).  Clamping the
        // exp-domain max keeps those exps at 0 while leaving real rows
        // (|scores| << 1e37) untouched.
        const float m_exp = fmaxf(m_new, -1.7e38f);
        // full tiles (no softclamp) kept RAW scores: exp2(fma(s, scale2, -m))
        const float escale = (full_tile && !SOFTCLAMP) ? scale2 : 1.f;
        uint32_t pk[NBLK * 8];                                      // packed bf16 pairs
        float partial[NBLK * 8];
        #pragma unroll
        for (int x2 = 0; x2 < NBLK * 8; ++x2) {
            float e0 = __builtin_amdgcn_exp2f(
                __builtin_fmaf(s[x2 >> 3][(2 * x2) & 15], escale, -m_exp));
            float e1 = __builtin_amdgcn_exp2f(
                __builtin_fmaf(s[x2 >> 3][(2 * x2 + 1) & 15], escale, -m_exp));
            partial[x2] = e0 + e1;
            union { __hip_bfloat162 h2; uint32_t u; } cvt;
            cvt.h2 = __float22bfloat162_rn(float2{e0, e1});
            pk[x2] = cvt.u;
        }
        // pairwise tree instead of a 32-deep serial dependency chain
        #pragma unroll
        for (int w = NBLK * 4; w >= 1; w >>= 1)
            #pragma unroll
            for (int x2 = 0; x2 < w; ++x2) partial[x2] += partial[x2 + w];
        float rowsum = partial[0];
        rowsum += cross_half(rowsum);
        if (any_growth) {
            float alpha = __builtin_amdgcn_exp2f(m_run - m_new);
            l_run = l_run * alpha + rowsum;
            #pragma unroll
            for (int db = 0; db < DBLK; ++db)
                #pragma unroll
                for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;
            m_run = m_new;
        } else {
            l_run += rowsum;
        }

        // ---- build PV B-operand fragments via permlane32_swap
        // pk[pb + x] holds the exp'd pair for kv rows (pattern):
        //   x=0:(0,1)+4lhi  x=1:(2,3)+4lhi  x=2:(8,9)+4lhi   x=3:(10,11)+4lhi
        //   x=4:(16,17)+4lhi x=5:(18,19)+4lhi x=6:(24,25)+4lhi x=7:(26,27)+4lhi
        // B fragment for k-step needs u32 slot c = kv (8*lhi + 2c, +1), so:
        //   swap(pk[pb+h*4+c], pk[pb+h*4+c+2]) -> r0 = slot c, r1 = slot c+2
        uint32_t frag[NBLK * 2][4];
        #pragma unroll
        for (int kb = 0; kb < NBLK; ++kb) {
            #pragma unroll
            for (int half = 0; half < 2; ++half) {
                #pragma unroll
                for (int c = 0; c < 2; ++c) {
                    u32x2 r = __builtin_amdgcn_permlane32_swap(
                        pk[kb * 8 + half * 4 + c], pk[kb * 8 + half * 4 + c + 2], false, false);
                    frag[kb * 2 + half][c] = r[0];
                    frag[kb * 2 + half][c + 2] = r[1];
                }
            }
        }

        if (stamp) P.ticks[t * 6 + 4] = __builtin_amdgcn_s_memtime();
        // ---- PV: O^T[d][q] += V^T[d][kv] P^T[kv][q]
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int db = 0; db < DBLK; ++db) {
            int drow = db * 32 + l31;
            #pragma unroll
            for (int ks = 0; ks < NBLK * 2; ++ks) {
                int chunk = ks * 2 + lhi;
                bf16x8 vf = *(const bf16x8*)(lds.vt[par] + drow * KVBLK + swz<KVBLK / 8>(drow, chunk) * 8);
                o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    vf, *(const bf16x8*)frag[ks], o_acc[db], 0, 0, 0);
            }
        }
        __builtin_amdgcn_s_setprio(0);
        if (stamp) P.ticks[t * 6 + 5] = __builtin_amdgcn_s_memtime();
    }

    // ---- epilogue
    if (!row_valid) return;

    if (split_mode) {
        // write this split's unnormalized partial (merged by attn_fwd_merge)
        const long part = (long)zsplit * P.b * P.h;
        float* mrow = P.m + (part + (long)b * P.h + h) * P.nq;
        float* lrow = P.l + (part + (long)b * P.h + h) * P.nq;
        if (lhi == 0) { mrow[i] = m_run * LN2; lrow[i] = l_run; }
        float* oa = P.o_acc + (part + (long)b * P.h + h) * D * P.nq;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                oa[(long)d * P.nq + i] = o_acc[db][r];
            }
        return;
    }

    if (P.is_last) {
        float l_safe = fmaxf(l_run, 1e-38f);
        float inv_l = 1.f / l_safe;
        __bf16* ob = (__bf16*)P.out + ((long)b * P.nq + i) * P.h * D + (long)h * D;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int g = 0; g < 4; ++g) {                            // groups of 4 contiguous d
                __bf16 four[4];
                #pragma unroll
                for (int e = 0; e < 4; ++e)
                    four[e] = (__bf16)(o_acc[db][g * 4 + e] * inv_l);
                int d = db * 32 + 8 * g + 4 * lhi;
                *(uint2*)(ob + d) = *(uint2*)four;
            }
        if (lhi == 0) {
            float* lsep = P.lse + ((long)b * P.h + h) * P.nq;
            lsep[i] = __logf(l_safe) + m_run * LN2;
        }
    } else {
        float* mrow = P.m + ((long)b * P.h + h) * P.nq;
        float* lrow = P.l + ((long)b * P.h + h) * P.nq;
        if (lhi == 0) { mrow[i] = m_run * LN2; lrow[i] = l_run; }
        float* oa = P.o_acc + (((long)b * P.h + h) * D) * P.nq;
        #pragma unroll
        for (int db = 0; db < DBLK; ++db)
            #pragma unroll
            for (int r = 0; r < 16; ++r) {
                int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * lhi;
                oa[(long)d * P.nq + i] = o_acc[db][r];
            }
    }
    };
    // noinline frame confines register allocation to the body — the pair
    // loop's liveness made the PAIRED instantiations spill (d64 532 B,
    // d128 656 B/lane) while the plain forms are clean; same lever as
    // dkv_noinline_call (attn_bwd.hip)
    if constexpr (PAIRED) fwd_noinline_call(fwd_body);
    else fwd_body();
    }  // pair loop
}

// ---------------------------------------------------------------------------
// split-merge: combine kv_split partials (and optionally the running ring
// accumulator) with the standard online-softmax merge; memory-bound.
// One thread per q row; d-loop vectorized along n across the wave.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void attn_fwd_merge_kernel(FwdMergeParams p) {
    const long row = (long)blockIdx.x * 256 + threadIdx.x;   // global (b*h*n) row
    const long total = (long)p.b * p.h * p.nq;
    if (row >= total) return;
    const long bh = row / p.nq;
    const long n = row % p.nq;
    const long b = bh / p.h, h = bh % p.h;

    const long bhn = bh * p.nq + n;
    const long stride_bh = (long)p.b * p.h;

    float m_tot = MASK_VALUE_F;
    if (!p.is_first) m_tot = p.m[bhn];
    for (int s = 0; s < p.splits; ++s)
        m_tot = fmaxf(m_tot, p.m_part[s * stride_bh * p.nq + bhn]);

    float l_tot = 0.f;
    float alpha_prev = 0.f;
    if (!p.is_first) {
        alpha_prev = __expf(p.m[bhn] - m_tot);
        l_tot = p.l[bhn] * alpha_prev;
    }
    float alpha_s[16];                      // splits <= 16
    for (int s = 0; s < p.splits; ++s) {
        alpha_s[s] = __expf(p.m_part[s * stride_bh * p.nq + bhn] - m_tot);
        l_tot += p.l_part[s * stride_bh * p.nq + bhn] * alpha_s[s];
    }

    const long od_base = (bh * D) * p.nq + n;     // (b,h,d,n) index at d=0
    if (p.is_last) {
        float l_safe = fmaxf(l_tot, 1e-38f);
        float inv = 1.f / l_safe;
        __bf16* ob = (__bf16*)p.out + ((long)b * p.nq + n) * p.h * D + h * D;
        for (int d = 0; d < D; d += 4) {
            float acc[4];
            #pragma unroll
            for (int e = 0; e < 4; ++e) {
                float v = p.is_first ? 0.f : p.o_acc[od_base + (long)(d + e) * p.nq] * alpha_prev;
                for (int s = 0; s < p.splits; ++s)
                    v += p.o_part[s * stride_bh * D * p.nq + od_base + (long)(d + e) * p.nq]
                         * alpha_s[s];
                acc[e] = v * inv;
            }
            __bf16 four[4] = {(__bf16)acc[0], (__bf16)acc[1], (__bf16)acc[2], (__bf16)acc[3]};
            *(uint2*)(ob + d) = *(uint2*)four;
        }
        p.lse[bhn] = __logf(l_safe) + m_tot;
    } else {
        for (int d = 0; d < D; ++d) {
            float v = p.is_first ? 0.f : p.o_acc[od_base + (long)d * p.nq] * alpha_prev;
            for (int s = 0; s < p.splits; ++s)
                v += p.o_part[s * stride_bh * D * p.nq + od_base + (long)d * p.nq] * alpha_s[s];
            p.o_acc[od_base + (long)d * p.nq] = v;
        }
        p.m[bhn] = m_tot;
        p.l[bhn] = l_tot;
    }
}

void launch_attn_fwd_merge(const FwdMergeParams& p, int head_dim, hipStream_t stream) {
    long total = (long)p.b * p.h * p.nq;
    dim3 grid((total + 255) / 256);
    dim3 block(256);
    if (head_dim == 64) {
        hipLaunchKernelGGL(attn_fwd_merge_kernel<64>, grid, block, 0, stream, p);
    } else if (head_dim == 32) {
        hipLaunchKernelGGL(attn_fwd_merge_kernel<32>, grid, block, 0, stream, p);
    } else {
        hipLaunchKernelGGL(attn_fwd_merge_kernel<128>, grid, block, 0, stream, p);
    }
}

void launch_attn_fwd(const FwdParams& p, int head_dim, hipStream_t stream) {
    // v2 (one-wave-per-SIMD pipeline) opt-in via RING_ATTN_FWD_V2=1;
    // RING_ATTN_FWD_V2=0 forces v1.  Default currently v1 (flip after the
    // promotion criterion: v2 beats v1 with full-flag oracle parity).
    static const char* v2env = std::getenv("RING_ATTN_FWD_V2");
    if (v2env && v2env[0] == '1') {
        if (launch_attn_fwd_v2(p, head_dim, stream)) return;
    }
    long qtiles = (p.nq + QROWS_WG - 1) / QROWS_WG;
    dim3 grid(p.paired ? (qtiles + 1) / 2 : qtiles, p.b * p.h,
              p.kv_split > 1 ? p.kv_split : 1);
    dim3 block(NTHREADS);
    const bool pr = p.paired > 0;
    if (head_dim == 64) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_fwd_kernel<64, true, true>), grid, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_fwd_kernel<64, true, false>), grid, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_fwd_kernel<64, false, true>), grid, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_fwd_kernel<64, false, false>), grid, block, 0, stream, p);
    } else if (head_dim == 128) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_fwd_kernel<128, true, true>), grid, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_fwd_kernel<128, true, false>), grid, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_fwd_kernel<128, false, true>), grid, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_fwd_kernel<128, false, false>), grid, block, 0, stream, p);
    } else if (head_dim == 32) {
        if (p.softclamp) if (pr) hipLaunchKernelGGL((attn_fwd_kernel<32, true, true>), grid, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_fwd_kernel<32, true, false>), grid, block, 0, stream, p);
        else if (pr) hipLaunchKernelGGL((attn_fwd_kernel<32, false, true>), grid, block, 0, stream, p);
        else hipLaunchKernelGGL((attn_fwd_kernel<32, false, false>), grid, block, 0, stream, p);
    } else {
        // unsupported head dim is a host-side error (checked in bindings)
        __builtin_trap();
    }
}

}  // namespace ring_attn
