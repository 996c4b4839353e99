// Shared types for the ring_attention_amd CDNA4 kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#ifdef __HIPCC__
#include <hip/hip_bf16.h>
#endif

namespace ring_attn {

#ifdef __HIPCC__
typedef __bf16 bf16_t;
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef unsigned int u32x2 __attribute__((ext_vector_type(2)));
#endif

// Must match ring_attention_amd.ops.reference.MASK_VALUE (torch.finfo(f32).min)
static constexpr float MASK_VALUE_F = -3.4028234663852886e38f;

struct FwdParams {
    const void* q;          // bf16 (B, Nq, H, D)
    const void* k;          // bf16 (B, Nk, HK, D)
    const void* v;          // bf16 (B, Nk, HK, D)
    const void* kmask;      // uint8 (B, Nk) or nullptr; 1 = attend
    float* o_acc;           // fp32 (B, H, D, Nq) scratch (nullptr for single-pass)
    float* m;               // fp32 (B, H, Nq)
    float* l;               // fp32 (B, H, Nq)
    void* out;              // bf16 (B, Nq, H, D) (written when is_last)
    float* lse;             // fp32 (B, H, Nq)    (written when is_last)
    int b, h, hk, group;    // group = H / HK
    long nq, nk;
    float scale;
    float softclamp_value;
    int softclamp;          // bool
    int causal;             // bool
    long diag;              // q position base: qpos(i) = i*q_stride + diag
    long q_stride;          // attend iff j <= qpos(i) (causal)
    long win;               // attend iff qpos(i) - j <= win (when has_win)
    int has_win;            // lookback window enabled
    int is_first;           // initialize m/l/o instead of loading
    int is_last;            // normalize + write out/lse instead of o_acc/m/l
    int kv_split;           // >1: grid.z splits the kv range; o_acc/m/l hold
                            // kv_split partials (merged by attn_fwd_merge)
    int ablate;             // diagnostics: 1 = stage first tile only
    int paired;             // causal load balance: >0 = total q-tiles T;
                            // grid.x = ceil(T/2), each WG runs tiles
                            // (x, T-1-x) so per-WG causal work is uniform
    unsigned long long* ticks;  // diagnostics: per-tile segment s_memtime
                                // stamps from block(0,0,0) wave 0 (or null)
    const float* bias;      // additive attention bias (natural-log domain)
                            // or null; (B,H,Nk) vector / (B,H,Nq,Nk) matrix
                            // (reference triton_flash_attn.py:1047-1063)
    int bias_mat;           // 1 = matrix (per-row) bias
};

struct FwdMergeParams {
    const float* o_part;    // fp32 (S, B, H, D, Nq) unnormalized partials
    const float* m_part;    // fp32 (S, B, H, Nq)
    const float* l_part;    // fp32 (S, B, H, Nq)
    float* o_acc;           // fp32 (B, H, D, Nq) running accumulator (ring)
    float* m;               // fp32 (B, H, Nq)
    float* l;               // fp32 (B, H, Nq)
    void* out;              // bf16 (B, Nq, H, D) written when is_last
    float* lse;             // fp32 (B, H, Nq)
    int splits;
    int b, h;
    long nq;
    int is_first;           // no running state to fold in
    int is_last;            // normalize + emit out/lse
};

void launch_attn_fwd_merge(const FwdMergeParams& p, int head_dim, hipStream_t stream);

void launch_attn_fwd(const FwdParams& p, int head_dim, hipStream_t stream);

// v2: one-wave-per-SIMD pipelined forward (attn_fwd_v2.hip); returns false
// when the config is not covered (caller falls back to v1)
bool launch_attn_fwd_v2(const FwdParams& p, int head_dim, hipStream_t stream);

struct BwdParams {
    const void* q;          // bf16 (B, Nq, H, D)
    const void* k;          // bf16 (B, Nk, HK, D)
    const void* v;          // bf16 (B, Nk, HK, D)
    const void* dout;       // bf16 (B, Nq, H, D)
    const void* kmask;      // uint8 (B, Nk) or nullptr
    const float* lse;       // fp32 (B, H, Nq)
    const float* delta;     // fp32 (B, H, Nq)  rowsum(do*o)
    float* dq;              // fp32 (B, Nq, H, D)  accumulated via atomics
    float* dk;              // fp32 (B, Nk, HK, D) accumulated (plain adds per WG)
    float* dv;              // fp32 (B, Nk, HK, D)
    int b, h, hk, group;
    long nq, nk;
    float scale;
    float softclamp_value;
    int softclamp;
    int causal;
    long diag;
    long q_stride;
    long win;
    int has_win;
    int accumulate;         // dk/dv + dq: 0 = overwrite, 1 = add to existing
    int split;              // >1: grid.z splits the contraction range; dq/dk/dv
                            // accumulated with fp32 atomics instead of plain ops
    int paired;             // causal balance: >0 = total walk-parallel tiles T;
                            // grid.x = ceil(T/2), WG x runs tiles (x, T-1-x)
    const float* bias;      // additive bias (see FwdParams) or null
    int bias_mat;
    const int* desc;        // descriptor scheduling (or null): grid.x units,
                            // desc[3u] = tile, desc[3u+1] = t_lo,
                            // desc[3u+2] = t_hi — constant work per unit;
                            // outputs accumulate with fp32 atomics
    long n_units;           // rows in desc (grid.x when desc mode)
};

void launch_attn_bwd_dq(const BwdParams& p, int head_dim, hipStream_t stream);
void launch_attn_bwd_dkv(const BwdParams& p, int head_dim, hipStream_t stream);

struct DeltaParams {
    const void* dout;       // bf16 (B, N, H, D)
    const void* out;        // bf16 (B, N, H, D)
    float* delta;           // fp32 (B, H, N)
    long rows;              // B*N*H
    long n;
    int h;
};

void launch_attn_delta(const DeltaParams& p, int head_dim, hipStream_t stream);

struct DecodeParams {
    const void* q;          // bf16 (B, HQ, NQ, D): NQ query tokens per head
                            // (speculative / tree-decode heads)
    const void* k;          // bf16 (B, HK, N, D)
    const void* v;          // bf16 (B, HK, N, D)
    float* out;             // fp32 (S, B, HQ, NQ, D) per-chunk partials
    float* lse;             // fp32 (S, B, HQ, NQ, 1)
    int b, h;               // h = HQ (query heads)
    int hk;                 // kv heads; q head qh reads kv head qh % hk
    int nq;                 // query tokens per head (>= 1)
    long n;
    float scale;
    long chunks;            // kv-split S (0 = auto)
    const void* kscale;     // fp8 mode: e8m0 per kv row (B, HK, N)
    const void* vscale;     // fp8 mode: e8m0 per kv row (B, HK, N)
};

void launch_decode_partial(const DecodeParams& p, int head_dim, hipStream_t stream);
void launch_decode_partial_fp8(const DecodeParams& p, int head_dim, hipStream_t stream);

struct DecodeMergeParams {
    const float* outs;   // (S, rows, D)
    const float* lses;   // (S, rows)
    float* out;          // (rows, D)
    float* lse;          // (rows)
    long rows;           // B * H * NQ
    int s;               // chunk count S
};

void launch_decode_merge(const DecodeMergeParams& p, int head_dim, hipStream_t stream);

struct Fp8FwdParams {
    const void* q;      // e4m3 bytes (B, Nq, H, D)
    const void* k;      // e4m3 bytes (B, Nk, H, D)
    const void* vt;     // e4m3 bytes (B, H, D, Nk)  — host-pre-transposed V
    const void* qs;     // e8m0 bytes (B, Nq, H, D/64): per (row, 64-d chunk)
    const void* ks;     // e8m0 bytes (B, Nk, HK, D/64)
    const void* vs;     // e8m0 bytes (B, HK, D, Nvs): per (d row, 64-kv chunk)
    void* out;          // bf16 (B, Nq, H, D)
    float* lse;         // fp32 (B, H, Nq)
    int b, h, hk;       // GQA: kv heads (qh pairs qh % hk)
    long nq, nk;        // PADDED sizes (nq % 256 == 0, nk % 128 == 0) — the
                        // wrapper pads the quantized buffers; strides use these
    long nk_true;       // true kv length: keys j >= nk_true are masked out
    int nvs;            // Nk / 64
    float scale;
    int causal;         // standard causal (qpos(i) = i), paired-tile grid
    int paired;         // total q tiles T (causal launcher)
};

void launch_attn_fwd_fp8(const Fp8FwdParams& p, int head_dim, hipStream_t stream);

struct RotaryParams {
    const void* x;      // bf16 (B, N, H, D)
    const float* cos_t; // fp32 (N, D/2) host-precomputed table
    const float* sin_t; // fp32 (N, D/2)
    void* out;          // bf16 (B, N, H, D)
    long rows;          // B * N * H
    int n, h, d;
    float sin_sign;     // +1 forward, -1 inverse (backward)
};

void launch_rotary(const RotaryParams& p, int head_dim, hipStream_t stream);

}  // namespace ring_attn
