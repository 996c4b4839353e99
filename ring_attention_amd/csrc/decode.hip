// Single-query (decode) attention partial for tree-attention decoding.
//
// Capability counterpart of the reference's tree-decode local partial
// (/root/reference/ring_attention_pytorch/tree_attn_decoding.py:54-79).
// Memory-bound: one wave per (b, h); KV streamed with 16-byte loads; softmax
// via in-wave shuffle reduction; fp32 out + lse emitted for the cross-rank
// RCCL merge (done in Python with a packed all-reduce).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "attn_common.h"

namespace ring_attn {

typedef float f32x2 __attribute__((ext_vector_type(2)));

template <int D>
__global__ __launch_bounds__(256) void decode_partial_kernel(DecodeParams p) {
    // one wave per (b, q-head, query token); 4 waves per block; blockIdx.y
    // splits the KV range into `chunks` partials (merged on the host) so
    // long sequences use the whole chip.  Multi-query (speculative / tree
    // heads) and GQA share the kv stream through L2/L3: the extra waves of
    // one kv head re-read lines the first wave just fetched, so >= 2
    // queries/step cost far less than 2x one query.
    const int rows = p.b * p.h * p.nq;
    const int wave_global = (blockIdx.x * 4) + (threadIdx.x >> 6);
    if (wave_global >= rows) return;
    // group-major wave->row mapping: consecutive waves (the 4 of a block,
    // which land on ONE CU and one XCD L2) cover the query tokens and then
    // the GQA group-mates of the SAME kv head, so their shared kv stream is
    // fetched once per L2.  The natural (b,h,iq) order instead put 4
    // DIFFERENT kv heads in each block and scattered a group's readers
    // across the 8 XCDs' private L2s — each re-fetched the stream from HBM
    // (measured: 32q/4kv nq=1 decode at 0.27 TB/s effective).  Output
    // layout is unchanged; only the wave->row assignment differs.
    const int G = p.h / p.hk;
    const int iq = wave_global % p.nq;
    int r = wave_global / p.nq;
    const int g = r % G;  r /= G;
    const int hk = r % p.hk;           // reference tile GQA pairing
    const int b = r / p.hk;
    const int h = hk + g * p.hk;       // group-mate g of kv head hk
    const int bh = b * p.h + h;
    const int lane = threadIdx.x & 63;
    const int chunk = blockIdx.y;
    const long per = (p.n + gridDim.y - 1) / gridDim.y;
    const long j_lo = chunk * per;
    const long j_hi = min(p.n, j_lo + per);
    const long part_off = (long)chunk * rows;

    const __bf16* qp = (const __bf16*)p.q + (((long)b * p.h + h) * p.nq + iq) * D;
    const __bf16* kp = (const __bf16*)p.k + ((long)b * p.hk + hk) * p.n * D;
    const __bf16* vp = (const __bf16*)p.v + ((long)b * p.hk + hk) * p.n * D;

    // q in registers (fp32), replicated per lane as needed
    float qreg[D];
    #pragma unroll
    for (int d = 0; d < D; ++d) qreg[d] = (float)qp[d];

    // pass 1+2 fused online: each lane handles keys lane, lane+64, ...
    float m = MASK_VALUE_F, l = 0.f;
    float acc[D];
    #pragma unroll
    for (int d = 0; d < D; ++d) acc[d] = 0.f;

    for (long j = j_lo + lane; j < j_hi; j += 64) {
        const __bf16* krow = kp + j * D;
        float s = 0.f;
        #pragma unroll
        for (int d = 0; d < D; d += 8) {
            bf16x8 kv8 = *(const bf16x8*)(krow + d);
            #pragma unroll
            for (int e = 0; e < 8; ++e) s += qreg[d + e] * (float)kv8[e];
        }
        s *= p.scale;
        float m_new = fmaxf(m, s);
        float alpha = __expf(m - m_new);
        float w = __expf(s - m_new);
        l = l * alpha + w;
        const __bf16* vrow = vp + j * D;
        #pragma unroll
        for (int d = 0; d < D; d += 8) {
            bf16x8 vv8 = *(const bf16x8*)(vrow + d);
            #pragma unroll
            for (int e = 0; e < 8; ++e) acc[d + e] = acc[d + e] * alpha + w * (float)vv8[e];
        }
        m = m_new;
    }

    // cross-lane merge (all 64 lanes -> lane 0's running stats)
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float m2 = __shfl_xor(m, off);
        float l2 = __shfl_xor(l, off);
        float m_new = fmaxf(m, m2);
        float a1 = __expf(m - m_new), a2 = __expf(m2 - m_new);
        l = l * a1 + l2 * a2;
        #pragma unroll
        for (int d = 0; d < D; ++d) {
            float o2 = __shfl_xor(acc[d], off);
            acc[d] = acc[d] * a1 + o2 * a2;
        }
        m = m_new;
    }

    float l_safe = fmaxf(l, 1e-38f);
    if (lane == 0) {
        float inv = 1.f / l_safe;
        float* op = p.out + (part_off + (long)bh * p.nq + iq) * D;
        #pragma unroll
        for (int d = 0; d < D; ++d) op[d] = acc[d] * inv;
        p.lse[part_off + (long)bh * p.nq + iq] = __logf(l_safe) + m;
    }
}

void launch_decode_partial(const DecodeParams& p, int head_dim, hipStream_t stream) {
    int waves = p.b * p.h * p.nq;
    // kv-split fallback when the caller did not set p.chunks (the binding
    // always does): mirror its GPU-swept heuristic — ~512 keys per wave,
    // total waves capped near 8192, S capped at 256 (the chunk merge grows
    // with S).  See bindings.cpp decode_partial.
    long chunks = p.chunks > 0 ? p.chunks
                 : max(1L, min(min((long)(p.n / 512 + 1), 256L),
                               8192L / max((long)waves, 1L)));
    dim3 grid((waves + 3) / 4, (unsigned)chunks);
    dim3 block(256);
    if (head_dim == 64) {
        hipLaunchKernelGGL(decode_partial_kernel<64>, grid, block, 0, stream, p);
    } else if (head_dim == 128) {
        hipLaunchKernelGGL(decode_partial_kernel<128>, grid, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}


// ---------------------------------------------------------------------------
// FP8 KV-cache decode: same structure, 8-bit KV stream (half the HBM bytes —
// decode is bandwidth-bound, measured 1.5-1.6 TB/s on the bf16 stream).
// k8/v8 are e4m3 rows with ONE e8m0 scale per (kv row): the k scale folds
// into the score (s_true = 2^ek * dot(q, k8)) and the v scale folds into the
// softmax weight (acc += (w * 2^ev) * v8) — the inner loop does the same
// fp32 math as the bf16 kernel, only the loads halve.  q stays bf16.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void decode_partial_fp8_kernel(DecodeParams p) {
    const int rows = p.b * p.h * p.nq;
    const int wave_global = (blockIdx.x * 4) + (threadIdx.x >> 6);
    if (wave_global >= rows) return;
    const int G = p.h / p.hk;
    const int iq = wave_global % p.nq;
    int r = wave_global / p.nq;
    const int g = r % G;  r /= G;
    const int hk = r % p.hk;
    const int b = r / p.hk;
    const int h = hk + g * p.hk;
    const int bh = b * p.h + h;
    const int lane = threadIdx.x & 63;
    const int chunk = blockIdx.y;
    const long per = (p.n + gridDim.y - 1) / gridDim.y;
    const long j_lo = chunk * per;
    const long j_hi = min(p.n, j_lo + per);
    const long part_off = (long)chunk * rows;

    const __bf16* qp = (const __bf16*)p.q + (((long)b * p.h + h) * p.nq + iq) * D;
    const unsigned char* kp = (const unsigned char*)p.k + ((long)b * p.hk + hk) * p.n * D;
    const unsigned char* vp = (const unsigned char*)p.v + ((long)b * p.hk + hk) * p.n * D;
    const unsigned char* ksp = (const unsigned char*)p.kscale + ((long)b * p.hk + hk) * p.n;
    const unsigned char* vsp = (const unsigned char*)p.vscale + ((long)b * p.hk + hk) * p.n;

    float qreg[D];
    #pragma unroll
    for (int d = 0; d < D; ++d) qreg[d] = (float)qp[d];

    float m = MASK_VALUE_F, l = 0.f;
    float acc[D];
    #pragma unroll
    for (int d = 0; d < D; ++d) acc[d] = 0.f;

    for (long j = j_lo + lane; j < j_hi; j += 64) {
        const unsigned char* krow = kp + j * D;
        float s = 0.f;
        #pragma unroll
        for (int d = 0; d < D; d += 16) {
            uint4 k16 = *(const uint4*)(krow + d);
            const unsigned* kw = (const unsigned*)&k16;
            #pragma unroll
            for (int w4 = 0; w4 < 4; ++w4) {
                f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(kw[w4], false);
                f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(kw[w4], true);
                s += qreg[d + 4 * w4 + 0] * lo[0] + qreg[d + 4 * w4 + 1] * lo[1]
                   + qreg[d + 4 * w4 + 2] * hi[0] + qreg[d + 4 * w4 + 3] * hi[1];
            }
        }
        s *= p.scale * exp2f((float)ksp[j] - 127.f);
        float m_new = fmaxf(m, s);
        float alpha = __expf(m - m_new);
        float w = __expf(s - m_new) * exp2f((float)vsp[j] - 127.f);
        l = l * alpha + __expf(s - m_new);
        const unsigned char* vrow = vp + j * D;
        #pragma unroll
        for (int d = 0; d < D; d += 16) {
            uint4 v16 = *(const uint4*)(vrow + d);
            const unsigned* vw = (const unsigned*)&v16;
            #pragma unroll
            for (int w4 = 0; w4 < 4; ++w4) {
                f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8(vw[w4], false);
                f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8(vw[w4], true);
                int base = d + 4 * w4;
                acc[base + 0] = acc[base + 0] * alpha + w * lo[0];
                acc[base + 1] = acc[base + 1] * alpha + w * lo[1];
                acc[base + 2] = acc[base + 2] * alpha + w * hi[0];
                acc[base + 3] = acc[base + 3] * alpha + w * hi[1];
            }
        }
        m = m_new;
    }

    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float m2 = __shfl_xor(m, off);
        float l2 = __shfl_xor(l, off);
        float m_new = fmaxf(m, m2);
        float a1 = __expf(m - m_new), a2 = __expf(m2 - m_new);
        l = l * a1 + l2 * a2;
        #pragma unroll
        for (int d = 0; d < D; ++d) {
            float o2 = __shfl_xor(acc[d], off);
            acc[d] = acc[d] * a1 + o2 * a2;
        }
        m = m_new;
    }

    float l_safe = fmaxf(l, 1e-38f);
    if (lane == 0) {
        float inv = 1.f / l_safe;
        float* op = p.out + (part_off + (long)bh * p.nq + iq) * D;
        #pragma unroll
        for (int d = 0; d < D; ++d) op[d] = acc[d] * inv;
        p.lse[part_off + (long)bh * p.nq + iq] = __logf(l_safe) + m;
    }
}

void launch_decode_partial_fp8(const DecodeParams& p, int head_dim, hipStream_t stream) {
    int waves = p.b * p.h * p.nq;
    // same swept kv-split fallback as launch_decode_partial
    long chunks = p.chunks > 0 ? p.chunks
                 : max(1L, min(min((long)(p.n / 512 + 1), 256L),
                               8192L / max((long)waves, 1L)));
    dim3 grid((waves + 3) / 4, (unsigned)chunks);
    dim3 block(256);
    if (head_dim == 64) {
        hipLaunchKernelGGL(decode_partial_fp8_kernel<64>, grid, block, 0, stream, p);
    } else if (head_dim == 128) {
        hipLaunchKernelGGL(decode_partial_fp8_kernel<128>, grid, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}


// ---------------------------------------------------------------------------
// Fused kv-chunk merge: combine the S per-chunk partials (out, lse) into one
// (out, lse) with the logsumexp weights.  The torch expression for this
// (max + exp + weighted sums over the S axis) costs ~50 us of reduce/
// elementwise dispatches per decode step at S = 256 — more than the decode
// kernel itself (30 us).  One wave per (b, h, nq) row; lanes parallel over d.
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void decode_merge_kernel(DecodeMergeParams p) {
    // one BLOCK per row: the 4 waves split the S chunks (a single wave
    // serially streaming S*D floats was latency-bound at decode's tiny
    // row counts), lanes parallel over d; LDS combines the 4 partials
    constexpr int DR = D / 64;
    const long row = blockIdx.x;
    const int wid = threadIdx.x >> 6;
    const int lane = threadIdx.x & 63;

    __shared__ float lm[4];
    __shared__ float lden[4];
    __shared__ float lacc[4][D];

    // global max over all S (each thread strides the whole range)
    float m = MASK_VALUE_F;
    for (int sC = (int)threadIdx.x; sC < p.s; sC += 256)
        m = fmaxf(m, p.lses[(long)sC * p.rows + row]);
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_xor(m, off));
    if (lane == 0) lm[wid] = m;
    __syncthreads();
    m = fmaxf(fmaxf(lm[0], lm[1]), fmaxf(lm[2], lm[3]));

    // wave w accumulates chunks s = w, w+4, ...
    float den = 0.f;
    float acc[DR];
    #pragma unroll
    for (int r = 0; r < DR; ++r) acc[r] = 0.f;
    for (int sC = wid; sC < p.s; sC += 4) {
        float w = __expf(p.lses[(long)sC * p.rows + row] - m);
        den += w;
        const float* op = p.outs + ((long)sC * p.rows + row) * D;
        #pragma unroll
        for (int r = 0; r < DR; ++r)
            acc[r] += w * op[r * 64 + lane];
    }
    if (lane == 0) lden[wid] = den;
    #pragma unroll
    for (int r = 0; r < DR; ++r) lacc[wid][r * 64 + lane] = acc[r];
    __syncthreads();

    if (wid == 0) {
        float den_t = lden[0] + lden[1] + lden[2] + lden[3];
        float den_safe = fmaxf(den_t, 1e-38f);
        float inv = 1.f / den_safe;
        float* dst = p.out + row * D;
        #pragma unroll
        for (int r = 0; r < DR; ++r) {
            int d = r * 64 + lane;
            dst[d] = (lacc[0][d] + lacc[1][d] + lacc[2][d] + lacc[3][d]) * inv;
        }
        if (lane == 0) p.lse[row] = __logf(den_safe) + m;
    }
}

void launch_decode_merge(const DecodeMergeParams& p, int head_dim, hipStream_t stream) {
    dim3 grid((unsigned)p.rows);
    dim3 block(256);
    if (head_dim == 64) {
        hipLaunchKernelGGL(decode_merge_kernel<64>, grid, block, 0, stream, p);
    } else if (head_dim == 128) {
        hipLaunchKernelGGL(decode_merge_kernel<128>, grid, block, 0, stream, p);
    } else {
        __builtin_trap();
    }
}

}  // namespace ring_attn
