// PyTorch bindings for the ring_attention_amd CDNA4 kernels.
//
// Thin layer: shape/dtype/contiguity checks + parameter marshalling.  All
// allocation happens in Python; kernels are launched on the current HIP
// stream so they compose with RCCL comm streams and hipGraph capture.

#include <torch/extension.h>
#include <cstdlib>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#include "attn_common.h"

namespace ring_attn {

#define CHECK_BF16_CONTIG(t) \
    TORCH_CHECK(t.is_contiguous() && t.scalar_type() == at::kBFloat16, #t " must be contiguous bf16")
#define CHECK_F32_CONTIG(t) \
    TORCH_CHECK(t.is_contiguous() && t.scalar_type() == at::kFloat, #t " must be contiguous fp32")

void attn_fwd(
    at::Tensor q, at::Tensor k, at::Tensor v,
    std::optional<at::Tensor> kmask,
    std::optional<at::Tensor> o_acc,
    std::optional<at::Tensor> m,
    std::optional<at::Tensor> l,
    std::optional<at::Tensor> out,
    std::optional<at::Tensor> lse,
    double scale, bool causal, int64_t diag, int64_t q_stride,
    int64_t win, bool has_win,
    bool softclamp, double softclamp_value,
    bool is_first, bool is_last, int64_t kv_split, int64_t ablate,
    std::optional<at::Tensor> ticks,
    std::optional<at::Tensor> bias, bool bias_mat) {
    CHECK_BF16_CONTIG(q); CHECK_BF16_CONTIG(k); CHECK_BF16_CONTIG(v);
    TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4, "q/k/v must be (B,N,H,D)");
    const int64_t B = q.size(0), Nq = q.size(1), H = q.size(2), D = q.size(3);
    const int64_t Nk = k.size(1), HK = k.size(2);
    TORCH_CHECK(D == 32 || D == 64 || D == 128,
                "head dim must be 32/64/128 (got ", D, "); arbitrary d <= 128 is "
                "padded at the python layer");
    TORCH_CHECK(H % HK == 0, "q heads must be a multiple of kv heads");
    TORCH_CHECK(k.size(0) == B && v.size(0) == B && v.size(1) == Nk && v.size(2) == HK && v.size(3) == D);

    FwdParams p{};
    p.q = q.data_ptr(); p.k = k.data_ptr(); p.v = v.data_ptr();
    p.kmask = nullptr;
    if (kmask.has_value()) {
        TORCH_CHECK(kmask->is_contiguous() && kmask->scalar_type() == at::kByte, "kmask must be contiguous uint8");
        TORCH_CHECK(kmask->size(0) == B && kmask->size(1) == Nk);
        p.kmask = kmask->data_ptr();
    }
    if (kv_split > 1) {
        TORCH_CHECK(o_acc && m && l, "split launches need partial o/m/l buffers");
        CHECK_F32_CONTIG((*o_acc)); CHECK_F32_CONTIG((*m)); CHECK_F32_CONTIG((*l));
        TORCH_CHECK(kv_split <= 16, "kv_split capped at 16");
        TORCH_CHECK(o_acc->numel() == kv_split * B * H * D * Nq
                    && m->numel() == kv_split * B * H * Nq);
        p.o_acc = o_acc->data_ptr<float>();
        p.m = m->data_ptr<float>();
        p.l = l->data_ptr<float>();
    } else if (!(is_first && is_last)) {
        TORCH_CHECK(o_acc && m && l, "multi-pass launches need o_acc/m/l scratch");
        CHECK_F32_CONTIG((*o_acc)); CHECK_F32_CONTIG((*m)); CHECK_F32_CONTIG((*l));
        TORCH_CHECK(o_acc->numel() == B * H * D * Nq && m->numel() == B * H * Nq);
        p.o_acc = o_acc->data_ptr<float>();
        p.m = m->data_ptr<float>();
        p.l = l->data_ptr<float>();
    }
    if (is_last && kv_split <= 1) {
        TORCH_CHECK(out && lse, "last pass needs out/lse");
        CHECK_BF16_CONTIG((*out)); CHECK_F32_CONTIG((*lse));
        TORCH_CHECK(out->sizes() == q.sizes() && lse->numel() == B * H * Nq);
        p.out = out->data_ptr();
        p.lse = lse->data_ptr<float>();
    }
    p.b = (int)B; p.h = (int)H; p.hk = (int)HK; p.group = (int)(H / HK);
    p.nq = Nq; p.nk = Nk;
    p.scale = (float)scale;
    p.softclamp = softclamp; p.softclamp_value = (float)softclamp_value;
    p.causal = causal; p.diag = diag; p.q_stride = q_stride;
    p.win = win; p.has_win = has_win;
    p.is_first = is_first; p.is_last = is_last;
    p.kv_split = (int)kv_split;
    p.ablate = (int)ablate;
    p.ticks = nullptr;
    if (ticks.has_value()) p.ticks = (unsigned long long*)ticks->data_ptr();
    p.bias = nullptr; p.bias_mat = bias_mat ? 1 : 0;
    if (bias.has_value()) {
        CHECK_F32_CONTIG((*bias));
        TORCH_CHECK(bias->numel() == (bias_mat ? B * H * Nq * Nk : B * H * Nk),
                    "bias must be (B,H,Nk) or (B,H,Nq,Nk) fp32");
        p.bias = bias->data_ptr<float>();
    }
    // causal pairing (uniform per-WG work) whenever the diagonal cuts this
    // kv range and nothing incompatible is on (splits balance differently;
    // windows are already uniform; ticks index per tile)
    p.paired = 0;
    {
        const long T = (Nq + 255) / 256;      // QROWS_WG
        // pairing halves grid.x — only engage when the paired grid (times
        // any grid.z split) still fills the 256 CUs (measured: engaging
        // below that idles half the chip and loses)
        const long z = kv_split > 1 ? kv_split : 1;
        if (causal && !has_win && diag < Nk && !bias.has_value()
            && ((T + 1) / 2) * B * H * z >= 256 && !ticks.has_value()
            && ablate == 0 && !std::getenv("RING_ATTN_NO_PAIR"))
            p.paired = (int)T;
    }

    launch_attn_fwd(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "attn_fwd launch failed");
}

void attn_fwd_merge(
    at::Tensor o_part, at::Tensor m_part, at::Tensor l_part,
    std::optional<at::Tensor> o_acc, std::optional<at::Tensor> m,
    std::optional<at::Tensor> l,
    std::optional<at::Tensor> out, std::optional<at::Tensor> lse,
    int64_t splits, int64_t B, int64_t H, int64_t D, int64_t Nq,
    bool is_first, bool is_last) {
    CHECK_F32_CONTIG(o_part); CHECK_F32_CONTIG(m_part); CHECK_F32_CONTIG(l_part);
    TORCH_CHECK(splits >= 1 && splits <= 16);
    TORCH_CHECK(o_part.numel() == splits * B * H * D * Nq);
    FwdMergeParams p{};
    p.o_part = o_part.data_ptr<float>();
    p.m_part = m_part.data_ptr<float>();
    p.l_part = l_part.data_ptr<float>();
    if (!(is_first && is_last)) {
        TORCH_CHECK(o_acc && m && l, "ring merge needs running o_acc/m/l");
        p.o_acc = o_acc->data_ptr<float>();
        p.m = m->data_ptr<float>();
        p.l = l->data_ptr<float>();
    }
    if (is_last) {
        TORCH_CHECK(out && lse);
        CHECK_BF16_CONTIG((*out)); CHECK_F32_CONTIG((*lse));
        p.out = out->data_ptr();
        p.lse = lse->data_ptr<float>();
    }
    p.splits = (int)splits; p.b = (int)B; p.h = (int)H; p.nq = Nq;
    p.is_first = is_first; p.is_last = is_last;
    launch_attn_fwd_merge(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "attn_fwd_merge launch failed");
}

void attn_bwd(
    at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor dout,
    std::optional<at::Tensor> kmask,
    at::Tensor lse, at::Tensor delta,
    at::Tensor dq, at::Tensor dk, at::Tensor dv,
    double scale, bool causal, int64_t diag, int64_t q_stride,
    int64_t win, bool has_win,
    bool softclamp, double softclamp_value, bool accumulate, int64_t split,
    int64_t which,     // 0 = both, 1 = dq only, 2 = dk/dv only
    std::optional<at::Tensor> desc_dq,    // int32 (U,3): tile, t_lo, t_hi
    std::optional<at::Tensor> desc_dkv,
    std::optional<at::Tensor> bias, bool bias_mat) {
    CHECK_BF16_CONTIG(q); CHECK_BF16_CONTIG(k); CHECK_BF16_CONTIG(v); CHECK_BF16_CONTIG(dout);
    CHECK_F32_CONTIG(lse); CHECK_F32_CONTIG(delta);
    CHECK_F32_CONTIG(dq); CHECK_F32_CONTIG(dk); CHECK_F32_CONTIG(dv);
    const int64_t B = q.size(0), Nq = q.size(1), H = q.size(2), D = q.size(3);
    const int64_t Nk = k.size(1), HK = k.size(2);
    TORCH_CHECK(D == 32 || D == 64 || D == 128, "head dim must be 32/64/128");
    TORCH_CHECK(dq.numel() == q.numel() && dk.numel() == k.numel() && dv.numel() == v.numel());
    TORCH_CHECK(lse.numel() == B * H * Nq && delta.numel() == B * H * Nq);

    BwdParams p{};
    p.q = q.data_ptr(); p.k = k.data_ptr(); p.v = v.data_ptr(); p.dout = dout.data_ptr();
    p.kmask = nullptr;
    if (kmask.has_value()) {
        TORCH_CHECK(kmask->is_contiguous() && kmask->scalar_type() == at::kByte);
        p.kmask = kmask->data_ptr();
    }
    p.lse = lse.data_ptr<float>(); p.delta = delta.data_ptr<float>();
    p.dq = dq.data_ptr<float>(); p.dk = dk.data_ptr<float>(); p.dv = dv.data_ptr<float>();
    p.b = (int)B; p.h = (int)H; p.hk = (int)HK; p.group = (int)(H / HK);
    p.nq = Nq; p.nk = Nk;
    p.scale = (float)scale;
    p.softclamp = softclamp; p.softclamp_value = (float)softclamp_value;
    p.causal = causal; p.diag = diag; p.q_stride = q_stride;
    p.win = win; p.has_win = has_win;
    p.accumulate = accumulate;
    p.split = (int)split;
    p.bias = nullptr; p.bias_mat = bias_mat ? 1 : 0;
    if (bias.has_value()) {
        CHECK_F32_CONTIG((*bias));
        TORCH_CHECK(bias->numel() == (bias_mat ? B * H * Nq * Nk : B * H * Nk));
        p.bias = bias->data_ptr<float>();
    }

    // causal pairing: uniform per-WG work when the diagonal cuts this range.
    // dq pairs over Q tiles, dkv over KV tiles (different counts when the
    // kv range is gathered), so set per launch.
    const bool pair_ok = causal && !has_win && diag < Nk
                         && !bias.has_value()
                         && !std::getenv("RING_ATTN_NO_PAIR");
    const long z = split > 1 ? split : 1;
    auto set_desc = [&](const std::optional<at::Tensor>& dsc) {
        p.desc = nullptr; p.n_units = 0;
        if (dsc.has_value()) {
            TORCH_CHECK(dsc->is_contiguous() && dsc->scalar_type() == at::kInt
                        && dsc->dim() == 2 && dsc->size(1) == 3,
                        "desc must be contiguous int32 (U,3)");
            p.desc = dsc->data_ptr<int>();
            p.n_units = dsc->size(0);
        }
    };
    if (which == 0 || which == 1) {
        const long Tq = (Nq + 255) / 256;
        set_desc(desc_dq);
        p.paired = (!p.desc && pair_ok && ((Tq + 1) / 2) * B * H * z >= 256)
                       ? (int)Tq : 0;
        launch_attn_bwd_dq(p, (int)D, at::hip::getCurrentHIPStream());
    }
    if (which == 0 || which == 2) {
        const long Tk = (Nk + 255) / 256;
        set_desc(desc_dkv);
        p.paired = (!p.desc && pair_ok && ((Tk + 1) / 2) * B * HK * z >= 256)
                       ? (int)Tk : 0;
        launch_attn_bwd_dkv(p, (int)D, at::hip::getCurrentHIPStream());
    }
    TORCH_CHECK(hipGetLastError() == hipSuccess, "attn_bwd launch failed");
}

at::Tensor rotary_apply(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t, double sin_sign) {
    // x bf16 (B, N, H, D); cos/sin fp32 (N, D/2)
    CHECK_BF16_CONTIG(x);
    CHECK_F32_CONTIG(cos_t); CHECK_F32_CONTIG(sin_t);
    const int64_t B = x.size(0), N = x.size(1), H = x.size(2), D = x.size(3);
    TORCH_CHECK(D == 32 || D == 64 || D == 128, "head dim must be 32/64/128");
    TORCH_CHECK(cos_t.size(0) == N && cos_t.size(1) == D / 2);
    auto out = at::empty_like(x);
    RotaryParams p{};
    p.x = x.data_ptr();
    p.cos_t = cos_t.data_ptr<float>();
    p.sin_t = sin_t.data_ptr<float>();
    p.out = out.data_ptr();
    p.rows = B * N * H;
    p.n = (int)N; p.h = (int)H; p.d = (int)D;
    p.sin_sign = (float)sin_sign;
    launch_rotary(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "rotary launch failed");
    return out;
}

std::vector<at::Tensor> decode_partial(at::Tensor q, at::Tensor k, at::Tensor v,
                                       double sm_scale, int64_t chunks_in) {
    // q (B,HQ,NQ,D); k,v (B,HK,N,D) bf16 — NQ query tokens (speculative /
    // tree heads), HQ % HK == 0 (GQA, tile pairing qh % hk)
    // -> (out fp32 (S,B,HQ,NQ,D), lse fp32 (S,B,HQ,NQ,1)): S kv-chunk
    //    partials, merged by the caller with the logsumexp combine
    CHECK_BF16_CONTIG(q); CHECK_BF16_CONTIG(k); CHECK_BF16_CONTIG(v);
    const int64_t B = q.size(0), H = q.size(1), NQ = q.size(2), D = q.size(3);
    const int64_t HK = k.size(1), N = k.size(2);
    TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128");
    TORCH_CHECK(H % HK == 0, "q heads must be a multiple of kv heads");
    int64_t waves = B * H * NQ;
    // kv-split heuristic (GPU-swept, tools/*decode_sweep*.py): ~512 keys
    // per wave with total waves capped near 8192 and S capped at 256 (the
    // host-side partial merge grows with S).  128k/1M bf16: 155/983 ->
    // 112/607 us; fp8: 109/250 -> 60/247 us; GQA 32q/4kv: 229 -> 177 us.
    int64_t chunks = chunks_in > 0 ? chunks_in : std::max<int64_t>(
        1, std::min<int64_t>(std::min<int64_t>(N / 512 + 1, 256),
                             8192 / std::max<int64_t>(waves, 1)));
    auto out = at::empty({chunks, B, H, NQ, D}, q.options().dtype(at::kFloat));
    auto lse = at::empty({chunks, B, H, NQ, 1}, q.options().dtype(at::kFloat));
    DecodeParams p{};
    p.q = q.data_ptr(); p.k = k.data_ptr(); p.v = v.data_ptr();
    p.out = out.data_ptr<float>(); p.lse = lse.data_ptr<float>();
    p.b = (int)B; p.h = (int)H; p.hk = (int)HK; p.nq = (int)NQ; p.n = N;
    p.scale = sm_scale > 0 ? (float)sm_scale : (float)(1.0 / std::sqrt((double)D));
    p.chunks = chunks;
    launch_decode_partial(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "decode launch failed");
    return {out, lse};
}

std::vector<at::Tensor> attn_fwd_fp8(at::Tensor q8, at::Tensor k8, at::Tensor v8t,
                                      at::Tensor qs, at::Tensor ks, at::Tensor vs,
                                      double sm_scale, bool causal, int64_t nk_true) {
    // MX-FP8 serving forward (see attn_fwd_fp8.hip header for scope):
    //   q8 (B,Nq,H,D) u8 e4m3; k8 (B,Nk,H,D) u8; v8t (B,H,D,Nk) u8
    //   qs (B,Nq,H) u8 e8m0(+127); ks (B,Nk,H) u8; vs (B,H,D,Nk/64) u8
    // -> out bf16 (B,Nq,H,D), lse fp32 (B,H,Nq)
    for (auto* t : {&q8, &k8, &v8t, &qs, &ks, &vs}) {
        TORCH_CHECK(t->scalar_type() == at::kByte && t->is_contiguous(),
                    "fp8 operands must be contiguous uint8 views");
    }
    const int64_t B = q8.size(0), NQ = q8.size(1), H = q8.size(2), D = q8.size(3);
    const int64_t NK = k8.size(1), HK = k8.size(2);
    TORCH_CHECK(D == 64 || D == 128, "fp8 path: head dim 64 or 128");
    TORCH_CHECK(H % HK == 0, "fp8 path: q heads must be a multiple of kv heads");
    TORCH_CHECK(NQ % 256 == 0, "fp8 path: nq must be a multiple of 256 (v0)");
    TORCH_CHECK(NK % 128 == 0, "fp8 path: nk must be a multiple of 128 (v0)");
    if (nk_true <= 0) nk_true = NK;
    TORCH_CHECK(v8t.size(1) == HK && v8t.size(3) == NK && v8t.size(2) == D
                && vs.size(1) == HK && vs.size(3) == NK / 64);
    TORCH_CHECK(qs.numel() == B * NQ * H * (D / 64)
                && ks.numel() == B * NK * HK * (D / 64),
                "q/k scales must be per (row, 64-d chunk)");
    auto out = at::empty({B, NQ, H, D}, q8.options().dtype(at::kBFloat16));
    auto lse = at::empty({B, H, NQ}, q8.options().dtype(at::kFloat));
    Fp8FwdParams p{};
    p.q = q8.data_ptr(); p.k = k8.data_ptr(); p.vt = v8t.data_ptr();
    p.qs = qs.data_ptr(); p.ks = ks.data_ptr(); p.vs = vs.data_ptr();
    p.out = out.data_ptr(); p.lse = lse.data_ptr<float>();
    p.b = (int)B; p.h = (int)H; p.hk = (int)HK; p.nq = NQ; p.nk = NK;
    p.nk_true = nk_true;
    p.nvs = (int)(NK / 64);
    p.causal = causal ? 1 : 0;
    p.scale = sm_scale > 0 ? (float)sm_scale : (float)(1.0 / std::sqrt((double)D));
    launch_attn_fwd_fp8(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "attn_fwd_fp8 launch failed");
    return {out, lse};
}

std::vector<at::Tensor> decode_partial_fp8(at::Tensor q, at::Tensor k8, at::Tensor v8,
                                           at::Tensor ks, at::Tensor vs,
                                           double sm_scale, int64_t chunks_in) {
    // FP8 KV-cache decode partial: q bf16 (B,HQ,NQ,D); k8/v8 e4m3 u8
    // (B,HK,N,D) with per-row e8m0 scales ks/vs (B,HK,N).  Same output
    // contract as decode_partial: (out fp32 (S,B,HQ,NQ,D), lse (S,B,HQ,NQ,1)).
    CHECK_BF16_CONTIG(q);
    for (auto* t : {&k8, &v8, &ks, &vs})
        TORCH_CHECK(t->scalar_type() == at::kByte && t->is_contiguous(),
                    "fp8 cache operands must be contiguous uint8");
    const int64_t B = q.size(0), H = q.size(1), NQ = q.size(2), D = q.size(3);
    const int64_t HK = k8.size(1), N = k8.size(2);
    TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128");
    TORCH_CHECK(H % HK == 0, "q heads must be a multiple of kv heads");
    TORCH_CHECK(ks.numel() == B * HK * N && vs.numel() == B * HK * N);
    int64_t waves = B * H * NQ;
    // kv-split heuristic (GPU-swept, tools/*decode_sweep*.py): ~512 keys
    // per wave with total waves capped near 8192 and S capped at 256 (the
    // host-side partial merge grows with S).  128k/1M bf16: 155/983 ->
    // 112/607 us; fp8: 109/250 -> 60/247 us; GQA 32q/4kv: 229 -> 177 us.
    int64_t chunks = chunks_in > 0 ? chunks_in : std::max<int64_t>(
        1, std::min<int64_t>(std::min<int64_t>(N / 512 + 1, 256),
                             8192 / std::max<int64_t>(waves, 1)));
    auto out = at::empty({chunks, B, H, NQ, D}, q.options().dtype(at::kFloat));
    auto lse = at::empty({chunks, B, H, NQ, 1}, q.options().dtype(at::kFloat));
    DecodeParams p{};
    p.q = q.data_ptr(); p.k = k8.data_ptr(); p.v = v8.data_ptr();
    p.kscale = ks.data_ptr(); p.vscale = vs.data_ptr();
    p.out = out.data_ptr<float>(); p.lse = lse.data_ptr<float>();
    p.b = (int)B; p.h = (int)H; p.hk = (int)HK; p.nq = (int)NQ; p.n = N;
    p.scale = sm_scale > 0 ? (float)sm_scale : (float)(1.0 / std::sqrt((double)D));
    p.chunks = chunks;
    launch_decode_partial_fp8(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "decode fp8 launch failed");
    return {out, lse};
}

std::vector<at::Tensor> decode_merge(at::Tensor outs, at::Tensor lses) {
    // fuse the S kv-chunk partial merge: outs (S,B,H,NQ,D) fp32,
    // lses (S,B,H,NQ,1) fp32 -> (out (B,H,NQ,D) fp32, lse (B,H,NQ,1) fp32)
    CHECK_F32_CONTIG(outs); CHECK_F32_CONTIG(lses);
    const int64_t S = outs.size(0), B = outs.size(1), H = outs.size(2),
                  NQ = outs.size(3), D = outs.size(4);
    TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128");
    auto out = at::empty({B, H, NQ, D}, outs.options());
    auto lse = at::empty({B, H, NQ, 1}, outs.options());
    DecodeMergeParams p{};
    p.outs = outs.data_ptr<float>(); p.lses = lses.data_ptr<float>();
    p.out = out.data_ptr<float>(); p.lse = lse.data_ptr<float>();
    p.rows = B * H * NQ; p.s = (int)S;
    launch_decode_merge(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "decode merge launch failed");
    return {out, lse};
}

at::Tensor attn_delta(at::Tensor dout, at::Tensor out) {
    // dout, out (B,N,H,D) bf16 -> delta fp32 (B,H,N) = rowsum(dout*out)
    CHECK_BF16_CONTIG(dout); CHECK_BF16_CONTIG(out);
    const int64_t B = dout.size(0), N = dout.size(1), H = dout.size(2), D = dout.size(3);
    TORCH_CHECK(D == 32 || D == 64 || D == 128, "head dim must be 32/64/128");
    auto delta = at::empty({B, H, N}, dout.options().dtype(at::kFloat));
    DeltaParams p{};
    p.dout = dout.data_ptr(); p.out = out.data_ptr();
    p.delta = delta.data_ptr<float>();
    p.rows = B * N * H; p.n = N; p.h = (int)H;
    launch_attn_delta(p, (int)D, at::hip::getCurrentHIPStream());
    TORCH_CHECK(hipGetLastError() == hipSuccess, "delta launch failed");
    return delta;
}

}  // namespace ring_attn

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
    mod.def("attn_fwd", &ring_attn::attn_fwd, "CDNA4 flash attention forward (resumable)",
            py::arg("q"), py::arg("k"), py::arg("v"), py::arg("kmask"),
            py::arg("o_acc"), py::arg("m"), py::arg("l"), py::arg("out"),
            py::arg("lse"), py::arg("scale"), py::arg("causal"), py::arg("diag"),
            py::arg("q_stride"), py::arg("win"), py::arg("has_win"),
            py::arg("softclamp"), py::arg("softclamp_value"), py::arg("is_first"),
            py::arg("is_last"), py::arg("kv_split"), py::arg("ablate"),
            py::arg("ticks"), py::arg("bias") = std::nullopt,
            py::arg("bias_mat") = false);
    mod.def("attn_fwd_merge", &ring_attn::attn_fwd_merge, "merge kv-split partials");
    mod.def("attn_bwd", &ring_attn::attn_bwd, "CDNA4 flash attention backward",
            py::arg("q"), py::arg("k"), py::arg("v"), py::arg("dout"),
            py::arg("kmask"), py::arg("lse"), py::arg("delta"), py::arg("dq"),
            py::arg("dk"), py::arg("dv"), py::arg("scale"), py::arg("causal"),
            py::arg("diag"), py::arg("q_stride"), py::arg("win"),
            py::arg("has_win"), py::arg("softclamp"), py::arg("softclamp_value"),
            py::arg("accumulate"), py::arg("split"), py::arg("which"),
            py::arg("desc_dq"), py::arg("desc_dkv"),
            py::arg("bias") = std::nullopt, py::arg("bias_mat") = false);
    mod.def("decode_partial", &ring_attn::decode_partial, "CDNA4 single-query decode partial",
            py::arg("q"), py::arg("k"), py::arg("v"), py::arg("sm_scale") = -1.0,
            py::arg("chunks") = 0);
    mod.def("attn_fwd_fp8", &ring_attn::attn_fwd_fp8,
            "CDNA4 MX-FP8 serving forward (e4m3 + e8m0 row scales)",
            py::arg("q8"), py::arg("k8"), py::arg("v8t"), py::arg("qs"),
            py::arg("ks"), py::arg("vs"), py::arg("sm_scale") = -1.0,
            py::arg("causal") = false, py::arg("nk_true") = 0);
    mod.def("decode_partial_fp8", &ring_attn::decode_partial_fp8,
            "CDNA4 FP8 KV-cache decode partial",
            py::arg("q"), py::arg("k8"), py::arg("v8"), py::arg("ks"),
            py::arg("vs"), py::arg("sm_scale") = -1.0, py::arg("chunks") = 0);
    mod.def("decode_merge", &ring_attn::decode_merge,
            "fused kv-chunk partial merge for decode");
    mod.def("attn_delta", &ring_attn::attn_delta, "fused delta = rowsum(dO*O) preprocess");
    mod.def("rotary_apply", &ring_attn::rotary_apply, "fused rotary embedding (table-driven)");
}
