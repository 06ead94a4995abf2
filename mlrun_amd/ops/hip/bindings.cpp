// Copyright 2026 mlrun_amd authors
//
// Licensed under the Apache License, Version 2.0 (the "License");
// you may not use this file except in compliance with the License.
//
// Torch bindings for the CDNA4 kernel library (mlrun_amd._hip_ops).
// Thin validation + launch layer; kernels live in kernels.hip.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "kernels.h"

#define CHECK_DEV(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " not contiguous")
#define CHECK_BF16(x) \
  TORCH_CHECK((x).scalar_type() == torch::kBFloat16, #x " must be bf16")
#define CHECK_F32(x) \
  TORCH_CHECK((x).scalar_type() == torch::kFloat32, #x " must be f32")

#define CHECK_U8(x) \
  TORCH_CHECK((x).scalar_type() == torch::kUInt8, #x " must be uint8")

#define CHECK_I32(x) \
  TORCH_CHECK((x).scalar_type() == torch::kInt32, #x " must be int32")

// native ROCm stream accessor (c10::hip — no hipify pass needed; the
// kernels in kernels.hip are pure CDNA4 HIP already)
static void* stream() {
  return (void*)c10::hip::getCurrentHIPStream().stream();
}

// out <- rmsnorm(x [+ residual]); residual updated in-place when given
void fused_add_rmsnorm(torch::Tensor out, torch::Tensor x,
                       torch::Tensor weight,
                       c10::optional<torch::Tensor> residual, double eps,
                       c10::optional<torch::Tensor> out8,
                       c10::optional<torch::Tensor> out_scale) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(x); CHECK_CONTIG(x); CHECK_BF16(x);
  CHECK_DEV(weight); CHECK_CONTIG(weight); CHECK_BF16(weight);
  int64_t hidden = x.size(-1);
  int64_t rows = x.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  TORCH_CHECK(weight.numel() == hidden, "weight size mismatch");
  void* res_ptr = nullptr;
  if (residual.has_value()) {
    CHECK_DEV(*residual); CHECK_CONTIG(*residual); CHECK_BF16(*residual);
    TORCH_CHECK(residual->numel() == x.numel(), "residual size mismatch");
    res_ptr = residual->data_ptr();
  }
  void* out8_ptr = nullptr;
  void* scale_ptr = nullptr;
  if (out8.has_value()) {
    TORCH_CHECK(out_scale.has_value(), "out8 needs out_scale");
    TORCH_CHECK(out8->numel() == x.numel() &&
                out_scale->numel() >= rows, "fp8 sidecar sizes");
    TORCH_CHECK(hidden % 64 == 0, "fp8 sidecar needs hidden % 64 == 0");
    out8_ptr = out8->data_ptr();
    scale_ptr = out_scale->data_ptr();
  }
  launch_fused_add_rmsnorm(out.data_ptr(), res_ptr, x.data_ptr(),
                           weight.data_ptr(), (int)rows, (int)hidden,
                           (float)eps, out8_ptr, scale_ptr, stream());
}

// q may be a strided row-view into a fused qkv buffer:
// required layout [T, heads, dim] with strides (row_stride, dim, 1)
static long long check_head_view(const torch::Tensor& q) {
  TORCH_CHECK(q.dim() == 3, "expected [T, heads, dim]");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "inner dims must be dense (strides (*, dim, 1))");
  return (long long)q.stride(0);
}

void rope(torch::Tensor q, torch::Tensor positions, torch::Tensor cos_sin) {
  CHECK_DEV(q); CHECK_BF16(q);
  CHECK_DEV(positions); CHECK_CONTIG(positions); CHECK_I32(positions);
  CHECK_DEV(cos_sin); CHECK_CONTIG(cos_sin); CHECK_F32(cos_sin);
  long long row_stride = check_head_view(q);
  int T = q.size(0), heads = q.size(1), dim = q.size(2);
  TORCH_CHECK(dim % 2 == 0 && dim / 2 <= 1024, "bad head dim");
  TORCH_CHECK(positions.numel() == T, "positions size mismatch");
  launch_rope(q.data_ptr(), positions.data_ptr(), cos_sin.data_ptr(), T,
              heads, dim, row_stride, stream());
}

void swiglu_fused(torch::Tensor out, torch::Tensor gu) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(gu); CHECK_CONTIG(gu); CHECK_BF16(gu);
  TORCH_CHECK(gu.dim() == 2 && out.dim() == 2, "need 2-D tensors");
  int rows = gu.size(0), inter = gu.size(1) / 2;
  TORCH_CHECK(out.size(0) == rows && out.size(1) == inter, "shape mismatch");
  TORCH_CHECK(inter % 8 == 0, "inter must be a multiple of 8");
  launch_swiglu_fused(out.data_ptr(), gu.data_ptr(), rows, inter, stream());
}

void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(gate); CHECK_CONTIG(gate); CHECK_BF16(gate);
  CHECK_DEV(up); CHECK_CONTIG(up); CHECK_BF16(up);
  TORCH_CHECK(out.numel() == gate.numel() && gate.numel() == up.numel(),
              "size mismatch");
  TORCH_CHECK(out.numel() % 8 == 0, "numel must be a multiple of 8");
  launch_silu_mul(out.data_ptr(), gate.data_ptr(), up.data_ptr(),
                  out.numel(), stream());
}

// out_bf16 [M,N] <- A [M,K] @ W [N,K]^T.  part_f32 is the split-K
// scratch ([ksplit*M*N] f32, unused when ksplit==1 — pass any tensor).
void skinny_gemm(torch::Tensor out_bf16, torch::Tensor part_f32,
                 torch::Tensor a, torch::Tensor w, int64_t ksplit,
                 int64_t variant) {
  CHECK_DEV(out_bf16); CHECK_CONTIG(out_bf16); CHECK_BF16(out_bf16);
  CHECK_DEV(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_DEV(w); CHECK_CONTIG(w); CHECK_BF16(w);
  int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(M <= 64, "skinny_gemm supports M <= 64");
  TORCH_CHECK(K % 32 == 0, "K must be a multiple of 32");
  TORCH_CHECK(out_bf16.numel() == (int64_t)M * N, "out shape mismatch");
  if (ksplit > 1) {
    CHECK_DEV(part_f32); CHECK_CONTIG(part_f32); CHECK_F32(part_f32);
    TORCH_CHECK(part_f32.numel() >= ksplit * (int64_t)M * N,
                "part_f32 scratch too small");
  }
  launch_skinny_gemm(out_bf16.data_ptr(),
                     ksplit > 1 ? part_f32.data_ptr() : nullptr,
                     a.data_ptr(), w.data_ptr(), M, N, K, (int)ksplit,
                     (int)variant, stream());
}

void skinny_gemm_slabs(torch::Tensor part_f32, torch::Tensor a,
                       torch::Tensor w, int64_t ksplit, int64_t variant) {
  CHECK_DEV(part_f32); CHECK_CONTIG(part_f32); CHECK_F32(part_f32);
  CHECK_DEV(a); CHECK_CONTIG(a); CHECK_BF16(a);
  CHECK_DEV(w); CHECK_CONTIG(w); CHECK_BF16(w);
  int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && M <= 64 && K % 32 == 0, "bad shapes");
  TORCH_CHECK(part_f32.numel() >= ksplit * (int64_t)M * N,
              "part_f32 scratch too small");
  launch_skinny_gemm_slabs(part_f32.data_ptr(), a.data_ptr(), w.data_ptr(),
                           M, N, K, (int)ksplit, (int)variant, stream());
}

void fused_add_rmsnorm_slab(torch::Tensor out, torch::Tensor residual,
                            torch::Tensor slabs, torch::Tensor weight,
                            int64_t ksplit, double eps) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(residual); CHECK_CONTIG(residual); CHECK_BF16(residual);
  CHECK_DEV(slabs); CHECK_CONTIG(slabs); CHECK_F32(slabs);
  CHECK_DEV(weight); CHECK_CONTIG(weight); CHECK_BF16(weight);
  int64_t hidden = out.size(-1);
  int64_t rows = out.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  TORCH_CHECK(slabs.numel() >= ksplit * rows * hidden, "slabs too small");
  launch_fused_add_rmsnorm_slab(out.data_ptr(), residual.data_ptr(),
                                slabs.data_ptr(), weight.data_ptr(),
                                (int)rows, (int)hidden, (float)eps,
                                (int)ksplit, stream());
}

void swiglu_slab(torch::Tensor out, torch::Tensor slabs, int64_t ksplit) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(slabs); CHECK_CONTIG(slabs); CHECK_F32(slabs);
  int rows = out.size(0), inter = out.size(1);
  TORCH_CHECK(inter % 4 == 0, "inter must be a multiple of 4");
  TORCH_CHECK(slabs.numel() >= (int64_t)ksplit * rows * 2 * inter,
              "slabs too small");
  launch_swiglu_slab(out.data_ptr(), slabs.data_ptr(), rows, inter,
                     (int)ksplit, stream());
}

void rope_kv_slab(torch::Tensor qkv, torch::Tensor kc, torch::Tensor vc,
                  torch::Tensor slabs, torch::Tensor positions,
                  torch::Tensor cos_sin, int64_t hq, int64_t ksplit) {
  CHECK_DEV(qkv); CHECK_CONTIG(qkv); CHECK_BF16(qkv);
  CHECK_DEV(kc); CHECK_CONTIG(kc); CHECK_BF16(kc);
  CHECK_DEV(slabs); CHECK_CONTIG(slabs); CHECK_F32(slabs);
  CHECK_DEV(positions); CHECK_CONTIG(positions); CHECK_I32(positions);
  CHECK_DEV(cos_sin); CHECK_CONTIG(cos_sin); CHECK_F32(cos_sin);
  int B = kc.size(0), Hkv = kc.size(1), Smax = kc.size(2), D = kc.size(3);
  int qkv_row = qkv.size(1);
  TORCH_CHECK(qkv.size(0) == B && qkv_row >= (hq + 2 * Hkv) * D,
              "qkv buffer too small");
  TORCH_CHECK(slabs.numel() >= (int64_t)ksplit * B * qkv_row,
              "slabs too small");
  launch_rope_kv_slab(qkv.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                      slabs.data_ptr(), positions.data_ptr(),
                      cos_sin.data_ptr(), B, (int)hq, Hkv, Smax, D, qkv_row,
                      (int)ksplit, stream());
}

void rope_kv_fused(torch::Tensor qkv, torch::Tensor kc, torch::Tensor vc,
                   torch::Tensor positions, torch::Tensor cos_sin,
                   int64_t hq,
                   c10::optional<torch::Tensor> k_scale = c10::nullopt,
                   c10::optional<torch::Tensor> v_scale = c10::nullopt) {
  CHECK_DEV(qkv); CHECK_CONTIG(qkv); CHECK_BF16(qkv);
  CHECK_DEV(kc); CHECK_CONTIG(kc);
  CHECK_DEV(vc); CHECK_CONTIG(vc);
  CHECK_DEV(positions); CHECK_CONTIG(positions); CHECK_I32(positions);
  CHECK_DEV(cos_sin); CHECK_CONTIG(cos_sin); CHECK_F32(cos_sin);
  int B = kc.size(0), Hkv = kc.size(1), Smax = kc.size(2), D = kc.size(3);
  TORCH_CHECK(qkv.dim() == 2 && qkv.size(0) == B, "qkv must be [B, rows]");
  TORCH_CHECK(qkv.size(1) >= (hq + 2 * Hkv) * D, "qkv row too small");
  if (k_scale.has_value()) {
    // fp8 KV mode: caches are uint8 e4m3 + per-row f32 scales
    TORCH_CHECK(v_scale.has_value(), "fp8 KV needs both scales");
    CHECK_U8(kc); CHECK_U8(vc);
    CHECK_DEV(*k_scale); CHECK_CONTIG(*k_scale); CHECK_F32(*k_scale);
    CHECK_DEV(*v_scale); CHECK_CONTIG(*v_scale); CHECK_F32(*v_scale);
    TORCH_CHECK(k_scale->numel() >= (int64_t)B * Hkv * Smax &&
                v_scale->numel() >= (int64_t)B * Hkv * Smax,
                "scale tensors too small");
    TORCH_CHECK(D == 128, "fp8 KV requires head_dim 128");
    launch_rope_kv_fused_q8(qkv.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                            k_scale->data_ptr(), v_scale->data_ptr(),
                            positions.data_ptr(), cos_sin.data_ptr(), B,
                            (int)hq, Hkv, Smax, D, qkv.size(1), stream());
    return;
  }
  CHECK_BF16(kc); CHECK_BF16(vc);
  launch_rope_kv_fused(qkv.data_ptr(), kc.data_ptr(), vc.data_ptr(),
                       positions.data_ptr(), cos_sin.data_ptr(), B, (int)hq,
                       Hkv, Smax, D, qkv.size(1), stream());
}

void skinny_gemm_fp8(torch::Tensor out_bf16, torch::Tensor part_f32,
                     torch::Tensor a8, torch::Tensor a_scale,
                     torch::Tensor w8, torch::Tensor w_scale,
                     int64_t ksplit) {
  CHECK_DEV(out_bf16); CHECK_CONTIG(out_bf16); CHECK_BF16(out_bf16);
  CHECK_DEV(a8); CHECK_CONTIG(a8); CHECK_U8(a8);
  CHECK_DEV(w8); CHECK_CONTIG(w8); CHECK_U8(w8);
  CHECK_DEV(a_scale); CHECK_F32(a_scale);
  CHECK_DEV(w_scale); CHECK_F32(w_scale);
  int M = a8.size(0), K = a8.size(1), N = w8.size(0);
  TORCH_CHECK(w8.size(1) == K && M <= 64 && K % 64 == 0,
              "bad shapes (fp8 path needs K % 64 == 0)");
  TORCH_CHECK(a_scale.numel() >= M && w_scale.numel() >= N,
              "scale sizes");
  if (ksplit > 1) {
    CHECK_DEV(part_f32); CHECK_CONTIG(part_f32); CHECK_F32(part_f32);
    TORCH_CHECK(part_f32.numel() >= ksplit * (int64_t)M * N,
                "part_f32 scratch too small");
  }
  launch_skinny_gemm_fp8(out_bf16.data_ptr(),
                         ksplit > 1 ? part_f32.data_ptr() : nullptr,
                         a8.data_ptr(), a_scale.data_ptr(), w8.data_ptr(),
                         w_scale.data_ptr(), M, N, K, (int)ksplit,
                         stream());
}

void quant_fp8_rows(torch::Tensor a8, torch::Tensor a_scale,
                    torch::Tensor a) {
  CHECK_DEV(a8); CHECK_CONTIG(a8); CHECK_U8(a8);
  CHECK_DEV(a_scale); CHECK_F32(a_scale);
  CHECK_DEV(a); CHECK_CONTIG(a); CHECK_BF16(a);
  int64_t cols = a.size(-1);
  int64_t rows = a.numel() / cols;
  TORCH_CHECK(cols % 8 == 0, "cols must be a multiple of 8");
  TORCH_CHECK(a8.numel() == a.numel() && a_scale.numel() >= rows,
              "size mismatch");
  launch_quant_fp8_rows(a8.data_ptr(), a_scale.data_ptr(), a.data_ptr(),
                        (int)rows, (int)cols, stream());
}

void cast_f32_bf16(torch::Tensor out, torch::Tensor in) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(in); CHECK_CONTIG(in); CHECK_F32(in);
  TORCH_CHECK(out.numel() == in.numel(), "size mismatch");
  launch_cast_f32_bf16(out.data_ptr(), in.data_ptr(), in.numel(), stream());
}

void attn_decode(torch::Tensor o, torch::Tensor q, torch::Tensor kc,
                 torch::Tensor vc, torch::Tensor seq_lens, double scale,
                 c10::optional<torch::Tensor> partial_ws, int64_t nsplit,
                 c10::optional<torch::Tensor> k_scale = c10::nullopt,
                 c10::optional<torch::Tensor> v_scale = c10::nullopt) {
  CHECK_DEV(o); CHECK_CONTIG(o); CHECK_BF16(o);
  CHECK_DEV(q); CHECK_BF16(q);
  CHECK_DEV(kc); CHECK_CONTIG(kc);
  CHECK_DEV(vc); CHECK_CONTIG(vc);
  const bool kvq = k_scale.has_value();
  if (kvq) {
    TORCH_CHECK(v_scale.has_value(), "fp8 KV needs both scales");
    CHECK_U8(kc); CHECK_U8(vc);
    CHECK_DEV(*k_scale); CHECK_CONTIG(*k_scale); CHECK_F32(*k_scale);
    CHECK_DEV(*v_scale); CHECK_CONTIG(*v_scale); CHECK_F32(*v_scale);
  } else {
    CHECK_BF16(kc); CHECK_BF16(vc);
  }
  CHECK_DEV(seq_lens); CHECK_CONTIG(seq_lens); CHECK_I32(seq_lens);
  TORCH_CHECK(q.dim() == 3 && kc.dim() == 4, "q [B,Hq,D], kc [B,Hkv,S,D]");
  long long q_row_stride = check_head_view(q);
  int B = q.size(0), Hq = q.size(1), D = q.size(2);
  int Hkv = kc.size(1), Smax = kc.size(2);
  TORCH_CHECK(D == 128, "attn_decode requires head_dim 128");
  TORCH_CHECK(Hq % Hkv == 0 && Hq / Hkv <= 8,
              "grouped heads per kv head must divide and be <= 8");
  TORCH_CHECK(kc.size(0) == B && kc.size(3) == D, "kc shape mismatch");
  float* ws = nullptr;
  if (nsplit > 1) {
    TORCH_CHECK(partial_ws.has_value(), "nsplit>1 needs partial_ws");
    CHECK_DEV(*partial_ws); CHECK_CONTIG(*partial_ws);
    CHECK_F32(*partial_ws);
    TORCH_CHECK(partial_ws->numel() >=
                (int64_t)B * Hq * nsplit * (D + 2),
                "partial_ws too small");
    ws = (float*)partial_ws->data_ptr();
  }
  if (kvq)
    TORCH_CHECK(k_scale->numel() >= (int64_t)B * Hkv * Smax &&
                v_scale->numel() >= (int64_t)B * Hkv * Smax,
                "scale tensors too small");
  launch_attn_decode(o.data_ptr(), q.data_ptr(), kc.data_ptr(),
                     vc.data_ptr(),
                     kvq ? k_scale->data_ptr() : nullptr,
                     kvq ? v_scale->data_ptr() : nullptr,
                     seq_lens.data_ptr(), B, Hq, Hkv, Smax,
                     (float)scale, q_row_stride, ws, (int)nsplit, stream());
}

void kv_append(torch::Tensor kc, torch::Tensor vc, torch::Tensor knew,
               torch::Tensor vnew, torch::Tensor positions) {
  CHECK_DEV(kc); CHECK_CONTIG(kc); CHECK_BF16(kc);
  CHECK_DEV(knew); CHECK_BF16(knew);
  CHECK_DEV(positions); CHECK_CONTIG(positions); CHECK_I32(positions);
  long long src_stride = check_head_view(knew);
  TORCH_CHECK(check_head_view(vnew) == src_stride,
              "knew/vnew strides must match");
  int B = kc.size(0), Hkv = kc.size(1), Smax = kc.size(2), D = kc.size(3);
  TORCH_CHECK(D % 8 == 0, "head_dim must be a multiple of 8");
  launch_kv_append(kc.data_ptr(), vc.data_ptr(), knew.data_ptr(),
                   vnew.data_ptr(), positions.data_ptr(), B, Hkv, Smax, D,
                   src_stride, stream());
}

void softmax(torch::Tensor out, torch::Tensor in) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_BF16(out);
  CHECK_DEV(in); CHECK_CONTIG(in); CHECK_BF16(in);
  int64_t cols = in.size(-1);
  int64_t rows = in.numel() / cols;
  launch_softmax(out.data_ptr(), in.data_ptr(), (int)rows, (int)cols,
                 stream());
}

void tree_ensemble(torch::Tensor out, torch::Tensor features,
                   torch::Tensor feature_idx, torch::Tensor threshold,
                   torch::Tensor left, torch::Tensor right,
                   torch::Tensor leaf_value, torch::Tensor tree_offsets,
                   double base_score) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_F32(out);
  CHECK_DEV(features); CHECK_CONTIG(features); CHECK_F32(features);
  CHECK_I32(feature_idx); CHECK_I32(left); CHECK_I32(right);
  CHECK_I32(tree_offsets); CHECK_F32(threshold); CHECK_F32(leaf_value);
  int n_samples = features.size(0), n_features = features.size(1);
  int n_trees = tree_offsets.numel() - 1;
  launch_tree_ensemble(out.data_ptr(), features.data_ptr(),
                       feature_idx.data_ptr(), threshold.data_ptr(),
                       left.data_ptr(), right.data_ptr(),
                       leaf_value.data_ptr(), tree_offsets.data_ptr(),
                       n_trees, n_samples, n_features, (float)base_score,
                       stream());
}

void window_ingest(torch::Tensor ring, torch::Tensor keys,
                   torch::Tensor values, torch::Tensor period_idx) {
  CHECK_DEV(ring); CHECK_CONTIG(ring);
  CHECK_I32(keys); CHECK_F32(values); CHECK_I32(period_idx);
  TORCH_CHECK(ring.dim() == 3 && ring.size(2) == 4,
              "ring must be [keys, periods, 4]");
  if (ring.scalar_type() == torch::kFloat64) {
    // f64 ring: sum-of-SQUARES accumulation (stdvar precision)
    launch_window_ingest64(ring.data_ptr(), keys.data_ptr(),
                           values.data_ptr(), period_idx.data_ptr(),
                           keys.numel(), (int)ring.size(1), stream());
    return;
  }
  CHECK_F32(ring);
  launch_window_ingest(ring.data_ptr(), keys.data_ptr(), values.data_ptr(),
                       period_idx.data_ptr(), keys.numel(),
                       (int)ring.size(1), stream());
}

void window_reduce(torch::Tensor out, torch::Tensor ring,
                   int64_t window_periods, int64_t current_period) {
  CHECK_DEV(out); CHECK_CONTIG(out);
  CHECK_DEV(ring); CHECK_CONTIG(ring);
  if (ring.scalar_type() == torch::kFloat64) {
    TORCH_CHECK(out.scalar_type() == torch::kFloat64,
                "out must be f64 for an f64 ring");
    launch_window_reduce64(out.data_ptr(), ring.data_ptr(),
                           (int)ring.size(0), (int)ring.size(1),
                           (int)window_periods, (int)current_period,
                           stream());
    return;
  }
  CHECK_F32(out); CHECK_F32(ring);
  launch_window_reduce(out.data_ptr(), ring.data_ptr(), (int)ring.size(0),
                       (int)ring.size(1), (int)window_periods,
                       (int)current_period, stream());
}

void window_ingest_mm(torch::Tensor ring_mm, torch::Tensor keys,
                      torch::Tensor values, torch::Tensor period_idx) {
  CHECK_DEV(ring_mm); CHECK_CONTIG(ring_mm);
  TORCH_CHECK(ring_mm.scalar_type() == torch::kInt32,
              "ring_mm must be int32 (ordered-f32 bits)");
  CHECK_I32(keys); CHECK_F32(values); CHECK_I32(period_idx);
  TORCH_CHECK(ring_mm.dim() == 3 && ring_mm.size(2) == 2,
              "ring_mm must be [keys, periods, 2]");
  launch_window_ingest_mm(ring_mm.data_ptr(), keys.data_ptr(),
                          values.data_ptr(), period_idx.data_ptr(),
                          keys.numel(), (int)ring_mm.size(1), stream());
}

void window_ingest_fl(torch::Tensor ring_fl, torch::Tensor keys,
                      torch::Tensor values, torch::Tensor timestamps,
                      torch::Tensor period_idx) {
  CHECK_DEV(ring_fl); CHECK_CONTIG(ring_fl);
  TORCH_CHECK(ring_fl.scalar_type() == torch::kInt64,
              "ring_fl must be int64 (packed ts|ordered-f32)");
  CHECK_I32(keys); CHECK_F32(values); CHECK_I32(timestamps);
  CHECK_I32(period_idx);
  TORCH_CHECK(ring_fl.dim() == 3 && ring_fl.size(2) == 2,
              "ring_fl must be [keys, periods, 2]");
  launch_window_ingest_fl(ring_fl.data_ptr(), keys.data_ptr(),
                          values.data_ptr(), timestamps.data_ptr(),
                          period_idx.data_ptr(), keys.numel(),
                          (int)ring_fl.size(1), stream());
}

void window_reduce_mmfl(torch::Tensor out, torch::Tensor ring_mm,
                        torch::Tensor ring_fl, int64_t window_periods,
                        int64_t current_period) {
  CHECK_DEV(out); CHECK_CONTIG(out); CHECK_F32(out);
  CHECK_DEV(ring_mm); CHECK_CONTIG(ring_mm);
  CHECK_DEV(ring_fl); CHECK_CONTIG(ring_fl);
  TORCH_CHECK(out.dim() == 2 && out.size(1) == 4,
              "out must be [keys, 4]");
  launch_window_reduce_mmfl(out.data_ptr(), ring_mm.data_ptr(),
                            ring_fl.data_ptr(), (int)ring_mm.size(0),
                            (int)ring_mm.size(1), (int)window_periods,
                            (int)current_period, stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "fused residual add + RMSNorm (bf16)");
  m.def("rope", &rope, "in-place rotary embedding (bf16)");
  m.def("silu_mul", &silu_mul, "silu(gate) * up (bf16)");
  m.def("swiglu_fused", &swiglu_fused,
        "silu(gu[:, :I]) * gu[:, I:] over a fused gate|up buffer");
  m.def("skinny_gemm", &skinny_gemm,
        "decode GEMM C=A@W^T on MFMA (bf16 in/out, split-K f32 slabs)");
  m.def("rope_kv_fused", &rope_kv_fused, py::arg("qkv"), py::arg("kc"),
        py::arg("vc"), py::arg("positions"), py::arg("cos_sin"),
        py::arg("hq"), py::arg("k_scale") = py::none(),
        py::arg("v_scale") = py::none(),
        "fused decode rope(q,k) + KV-cache append");
  m.def("skinny_gemm_slabs", &skinny_gemm_slabs,
        "split-K GEMM emitting f32 slabs only");
  m.def("fused_add_rmsnorm_slab", &fused_add_rmsnorm_slab,
        "slab-sum + residual add + RMSNorm");
  m.def("swiglu_slab", &swiglu_slab, "slab-sum + SwiGLU");
  m.def("rope_kv_slab", &rope_kv_slab,
        "slab-sum + rope(q,k) + KV append");
  m.def("cast_f32_bf16", &cast_f32_bf16, "f32 -> bf16 cast");
  m.def("skinny_gemm_fp8", &skinny_gemm_fp8,
        "fp8-weight decode GEMM (e4m3, per-row scales)");
  m.def("quant_fp8_rows", &quant_fp8_rows,
        "dynamic per-row bf16 -> fp8 e4m3 quantization");
  m.def("attn_decode", &attn_decode, py::arg("o"), py::arg("q"),
        py::arg("kc"), py::arg("vc"), py::arg("seq_lens"),
        py::arg("scale"), py::arg("partial_ws") = py::none(),
        py::arg("nsplit") = 1, py::arg("k_scale") = py::none(),
        py::arg("v_scale") = py::none(),
        "GQA decode attention (bf16 or fp8 KV cache)");
  m.def("kv_append", &kv_append, "append token K/V into cache");
  m.def("softmax", &softmax, "row softmax (bf16)");
  m.def("tree_ensemble", &tree_ensemble, "GBDT ensemble inference (f32)");
  m.def("window_ingest", &window_ingest,
        "feature-store window ring ingest (f32)");
  m.def("window_reduce", &window_reduce,
        "feature-store window ring reduce (f32)");
  m.def("window_ingest_mm", &window_ingest_mm,
        "per-period min/max ingest (ordered-f32 atomics)");
  m.def("window_ingest_fl", &window_ingest_fl,
        "per-period first/last ingest (packed ts|value u64 atomics)");
  m.def("window_reduce_mmfl", &window_reduce_mmfl,
        "window min/max/first/last reduce over period cells");
}
