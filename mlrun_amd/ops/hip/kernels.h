// Copyright 2026 mlrun_amd authors
//
// Licensed under the Apache License, Version 2.0 (the "License");
// you may not use this file except in compliance with the License.
//
// Launch-wrapper declarations for the CDNA4 kernel library.
// All pointers are device pointers; stream is a hipStream_t.
#pragma once

extern "C++" {

void launch_fused_add_rmsnorm(void* out, void* residual, const void* x,
                              const void* weight, int rows, int hidden,
                              float eps, void* out8, void* out_scale,
                              void* stream);

void launch_rope(void* q, const void* positions, const void* cos_sin, int T,
                 int heads, int dim, long long row_stride, void* stream);

void launch_swiglu_fused(void* out, const void* gu, int rows, int inter,
                         void* stream);

void launch_silu_mul(void* out, const void* gate, const void* up,
                     long long n, void* stream);

void launch_skinny_gemm_slabs(void* part_f32, const void* A, const void* W,
                              int M, int N, int K, int ksplit, int variant,
                              void* stream);

void launch_fused_add_rmsnorm_slab(void* out, void* residual,
                                   const void* slabs, const void* weight,
                                   int rows, int hidden, float eps,
                                   int ksplit, void* stream);

void launch_swiglu_slab(void* out, const void* slabs, int rows, int inter,
                        int ksplit, void* stream);

void launch_rope_kv_slab(void* qkv, void* Kc, void* Vc, const void* slabs,
                         const void* positions, const void* cos_sin, int B,
                         int Hq, int Hkv, int Smax, int D, int qkv_row,
                         int ksplit, void* stream);

void launch_skinny_gemm(void* out_bf16, void* part_f32, const void* A,
                        const void* W, int M, int N, int K, int ksplit,
                        int variant, void* stream);

void launch_rope_kv_fused(void* qkv, void* Kc, void* Vc,
                          const void* positions, const void* cos_sin, int B,
                          int Hq, int Hkv, int Smax, int D,
                          long long row_stride, void* stream);

void launch_rope_kv_fused_q8(void* qkv, void* Kc8, void* Vc8,
                             void* kscale, void* vscale,
                             const void* positions, const void* cos_sin,
                             int B, int Hq, int Hkv, int Smax, int D,
                             long long row_stride, void* stream);

void launch_skinny_gemm_fp8(void* out_bf16, void* part_f32, const void* A8,
                            const void* a_scale, const void* W8,
                            const void* w_scale, int M, int N, int K,
                            int ksplit, void* stream);

void launch_quant_fp8_rows(void* a8, void* a_scale, const void* a, int rows,
                           int cols, void* stream);

void launch_cast_f32_bf16(void* out, const void* in, long long n,
                          void* stream);

void launch_zero_f32(void* p, long long n, void* stream);

void launch_attn_decode(void* O, const void* Q, const void* Kc,
                        const void* Vc, const void* kscale,
                        const void* vscale, const void* seq_lens, int B,
                        int Hq, int Hkv, int Smax, float scale,
                        long long q_row_stride, float* partial_ws,
                        int nsplit, void* stream);

void launch_kv_append(void* Kc, void* Vc, const void* knew, const void* vnew,
                      const void* positions, int B, int Hkv, int Smax, int D,
                      long long src_row_stride, void* stream);

void launch_softmax(void* out, const void* in, int rows, int cols,
                    void* stream);

void launch_tree_ensemble(void* out, const void* features,
                          const void* feature_idx, const void* threshold,
                          const void* left, const void* right,
                          const void* leaf_value, const void* tree_offsets,
                          int n_trees, int n_samples, int n_features,
                          float base_score, void* stream);

void launch_window_ingest(void* ring, const void* keys, const void* values,
                          const void* period_idx, long long n_events,
                          int n_periods, void* stream);

void launch_window_ingest64(void* ring, const void* keys,
                            const void* values, const void* period_idx,
                            long long n_events, int n_periods,
                            void* stream);
void launch_window_reduce64(void* out, const void* ring, int n_keys,
                            int n_periods, int window_periods,
                            int current_period, void* stream);
void launch_window_ingest_mm(void* ring_mm, const void* keys,
                             const void* values, const void* period_idx,
                             long long n_events, int n_periods,
                             void* stream);
void launch_window_ingest_fl(void* ring_fl, const void* keys,
                             const void* values, const void* timestamps,
                             const void* period_idx, long long n_events,
                             int n_periods, void* stream);
void launch_window_reduce_mmfl(void* out, const void* ring_mm,
                               const void* ring_fl, int n_keys,
                               int n_periods, int window_periods,
                               int current_period, void* stream);
void launch_window_reduce(void* out, const void* ring, int n_keys,
                          int n_periods, int window_periods,
                          int current_period, void* stream);

}  // extern "C++"
