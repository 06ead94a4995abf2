// Copyright 2026 mlrun_amd authors
//
// Licensed under the Apache License, Version 2.0 (the "License");
// you may not use this file except in compliance with the License.
//
// CDNA4 (gfx950 / MI355X) serving & training kernels.
//
// Design notes (see /root/repo/SURVEY.md §2.3/§2.6 — the reference has
// no GPU kernels; this kernel surface is defined by the north-star
// serving/training configs):
//  - wave = 64 lanes everywhere; all bf16 global loads vectorized as
//    short8/ushort2 (16B / 4B per lane)
//  - decode GEMM (M<=16 activations x [N,K] weights) on MFMA
//    v_mfma_f32_16x16x32_bf16, K-split across workgroups with f32
//    atomics so the grid fills 256 CUs
//  - decode attention: one workgroup per (batch, kv-head), one wave
//    per grouped q-head, online softmax, LDS score buffer
//  - norm/rope/swiglu fused elementwise kernels at HBM rate
// All kernels are hipGraph-capture safe: no allocation, no sync.

#include <hip/hip_fp8.h>
#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>

#include "kernels.h"

typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float f32x4v;
typedef __attribute__((ext_vector_type(2))) float f32x2v;

#define DEV __device__ __forceinline__

DEV float bf2f(unsigned short u) {
  union { uint32_t i; float f; } v;
  v.i = uint32_t(u) << 16;
  return v.f;
}

DEV unsigned short f2bf(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  uint32_t x = v.i;
  // round-to-nearest-even
  uint32_t r = (x + 0x7fffu + ((x >> 16) & 1u)) >> 16;
  return (unsigned short)r;
}

struct ushort8 { unsigned short v[8]; };
struct ushort2v { unsigned short x, y; };

// ---------------------------------------------------------------------
// fused residual-add + RMSNorm (bf16):
//   residual <- x + residual ; out <- rmsnorm(residual) * weight
// one workgroup per row; vectorized short8 loads (guide G13).
// ---------------------------------------------------------------------
// out8/out_scale (optional): also emit the normalized row quantized
// to fp8 e4m3 in the GEMM's pair-swizzled layout — feeds the
// fp8-weight decode path without a separate quant launch.
__global__ void fused_add_rmsnorm_kernel(
    unsigned short* __restrict__ out, unsigned short* __restrict__ residual,
    const unsigned short* __restrict__ x,
    const unsigned short* __restrict__ weight, int hidden, float eps,
    int has_residual, unsigned char* __restrict__ out8,
    float* __restrict__ out_scale) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const unsigned short* xr = x + (size_t)row * hidden;
  unsigned short* rr = residual ? residual + (size_t)row * hidden : nullptr;
  unsigned short* outr = out + (size_t)row * hidden;

  float sumsq = 0.f;
  for (int i = tid * 8; i < hidden; i += nthreads * 8) {
    ushort8 xv = *reinterpret_cast<const ushort8*>(xr + i);
    float z[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) z[j] = bf2f(xv.v[j]);
    if (has_residual) {
      ushort8 rv = *reinterpret_cast<const ushort8*>(rr + i);
      ushort8 zv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        z[j] += bf2f(rv.v[j]);
        zv.v[j] = f2bf(z[j]);
      }
      *reinterpret_cast<ushort8*>(rr + i) = zv;  // updated residual
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) sumsq += z[j] * z[j];
  }
  // block reduce
  __shared__ float red[32];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    sumsq += __shfl_down(sumsq, off, 64);
  if ((tid & 63) == 0) red[tid >> 6] = sumsq;
  __syncthreads();
  if (tid < (nthreads >> 6)) sumsq = red[tid];
  else sumsq = 0.f;
  if (tid < 64) {
#pragma unroll
    for (int off = 2; off > 0; off >>= 1)
      sumsq += __shfl_down(sumsq, off, 64);
  }
  if (tid == 0) red[0] = sumsq;
  __syncthreads();
  const float inv = rsqrtf(red[0] / hidden + eps);

  const unsigned short* zsrc = has_residual ? rr : xr;
  float amax = 0.f;
  for (int i = tid * 8; i < hidden; i += nthreads * 8) {
    ushort8 zv = *reinterpret_cast<const ushort8*>(zsrc + i);
    ushort8 wv = *reinterpret_cast<const ushort8*>(weight + i);
    ushort8 ov;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float value = bf2f(zv.v[j]) * inv * bf2f(wv.v[j]);
      ov.v[j] = f2bf(value);
      amax = fmaxf(amax, fabsf(value));
    }
    *reinterpret_cast<ushort8*>(outr + i) = ov;
  }
  if (out8 == nullptr) return;
  // fp8 sidecar: block-reduce amax, then re-emit the row quantized
  // into the gemm's pair-swizzled layout (row data is L1/L2 hot)
  __syncthreads();  // red[] still feeds inv on slower threads
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, 64));
  if ((tid & 63) == 0) red[tid >> 6] = amax;
  __syncthreads();
  float gmax = 0.f;
  for (int w = 0; w < (nthreads >> 6); ++w) gmax = fmaxf(gmax, red[w]);
  const float qscale = gmax > 0.f ? gmax / 448.f : 1.f;
  const float qinv = 1.f / qscale;
  if (tid == 0) out_scale[row] = qscale;
  unsigned char* out8r = out8 + (size_t)row * hidden;
  for (int i = tid * 8; i < hidden; i += nthreads * 8) {
    ushort8 ov = *reinterpret_cast<const ushort8*>(outr + i);
    unsigned char q[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 f8(bf2f(ov.v[j]) * qinv);
      q[j] = f8.__x;
    }
    const int sblk = i >> 5;
    const int c = (i >> 3) & 3;
    const int dst = ((sblk >> 1) << 6) + (c << 4) + ((sblk & 1) << 3);
    *reinterpret_cast<uint64_t*>(out8r + dst) =
        *reinterpret_cast<const uint64_t*>(q);
  }
}

void launch_fused_add_rmsnorm(void* out, void* residual, const void* x,
                              const void* weight, int rows, int hidden,
                              float eps, void* out8, void* out_scale,
                              void* stream) {
  int threads = hidden >= 2048 ? 256 : 64;
  hipLaunchKernelGGL(fused_add_rmsnorm_kernel, dim3(rows), dim3(threads), 0,
                     (hipStream_t)stream, (unsigned short*)out,
                     (unsigned short*)residual, (const unsigned short*)x,
                     (const unsigned short*)weight, hidden, eps,
                     residual != nullptr ? 1 : 0, (unsigned char*)out8,
                     (float*)out_scale);
}

// ---------------------------------------------------------------------
// RoPE (neox / llama style, rotate-half):
//   for i < D/2: (q_i, q_{i+D/2}) <- (q_i c - q_{i+D/2} s,
//                                     q_{i+D/2} c + q_i s)
// cos_sin: [max_pos, D/2, 2] f32 precomputed host-side (guide B:
// on-device trig turns memory-bound into VALU-bound).
// q: [T, heads, D]; positions: [T].
// ---------------------------------------------------------------------
__global__ void rope_kernel(unsigned short* __restrict__ q,
                            const int* __restrict__ positions,
                            const float* __restrict__ cos_sin, int heads,
                            int dim, long long row_stride) {
  const int t = blockIdx.x;
  const int h = blockIdx.y;
  const int i = threadIdx.x;  // 0..D/2-1
  const int half = dim >> 1;
  if (i >= half) return;
  const int pos = positions[t];
  const float c = cos_sin[((size_t)pos * half + i) * 2];
  const float s = cos_sin[((size_t)pos * half + i) * 2 + 1];
  unsigned short* base = q + (size_t)t * row_stride + (size_t)h * dim;
  float a = bf2f(base[i]);
  float b = bf2f(base[i + half]);
  base[i] = f2bf(a * c - b * s);
  base[i + half] = f2bf(b * c + a * s);
}

void launch_rope(void* q, const void* positions, const void* cos_sin, int T,
                 int heads, int dim, long long row_stride, void* stream) {
  hipLaunchKernelGGL(rope_kernel, dim3(T, heads), dim3(dim / 2), 0,
                     (hipStream_t)stream, (unsigned short*)q,
                     (const int*)positions, (const float*)cos_sin, heads, dim,
                     row_stride);
}

// ---------------------------------------------------------------------
// SwiGLU: out = silu(gate) * up   (bf16, flat, vectorized)
// gate/up are halves of one [rows, 2*inter] tensor or separate ptrs.
// ---------------------------------------------------------------------
__global__ void silu_mul_kernel(unsigned short* __restrict__ out,
                                const unsigned short* __restrict__ gate,
                                const unsigned short* __restrict__ up,
                                long long n) {
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long long stride = (long long)gridDim.x * blockDim.x * 8;
  for (; i < n; i += stride) {
    ushort8 g = *reinterpret_cast<const ushort8*>(gate + i);
    ushort8 u = *reinterpret_cast<const ushort8*>(up + i);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.v[j]);
      float s = gf / (1.f + __expf(-gf));
      o.v[j] = f2bf(s * bf2f(u.v[j]));
    }
    *reinterpret_cast<ushort8*>(out + i) = o;
  }
}

// SwiGLU over a fused gate|up buffer: gu [rows, 2*inter] ->
// out [rows, inter] = silu(gu[:, :inter]) * gu[:, inter:]
__global__ void swiglu_fused_kernel(unsigned short* __restrict__ out,
                                    const unsigned short* __restrict__ gu,
                                    int rows, int inter) {
  const long long total = (long long)rows * inter / 8;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const long long flat = i * 8;
    const int r = (int)(flat / inter);
    const int c = (int)(flat % inter);
    const unsigned short* row = gu + (size_t)r * 2 * inter;
    ushort8 g = *reinterpret_cast<const ushort8*>(row + c);
    ushort8 u = *reinterpret_cast<const ushort8*>(row + inter + c);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g.v[j]);
      float s = gf / (1.f + __expf(-gf));
      o.v[j] = f2bf(s * bf2f(u.v[j]));
    }
    *reinterpret_cast<ushort8*>(out + (size_t)r * inter + c) = o;
  }
}

void launch_swiglu_fused(void* out, const void* gu, int rows, int inter,
                         void* stream) {
  long long blocks = ((long long)rows * inter / 8 + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(swiglu_fused_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (unsigned short*)out,
                     (const unsigned short*)gu, rows, inter);
}

void launch_silu_mul(void* out, const void* gate, const void* up,
                     long long n, void* stream) {
  long long blocks = (n / 8 + 255) / 256;
  if (blocks > 2048) blocks = 2048;  // grid-stride (guide G11)
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(silu_mul_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (unsigned short*)out,
                     (const unsigned short*)gate, (const unsigned short*)up,
                     n);
}

// ---------------------------------------------------------------------
// Skinny GEMM (decode projections):  C[M,N] (+)= A[M,K] @ W[N,K]^T
// M <= 16 (decode batch), bf16 inputs, f32 C.
// MFMA 16x16x32: each wave owns a 32-wide n-tile (2 B-fragments),
// 4 waves/block -> 128 n per block; K split across gridDim.y with
// device-scope f32 atomics (cross-XCD safe, guide G12/G16).
// Fragment layout (verified on HW by tests/gpu):
//   A: lane l holds A[l&15][(l>>4)*8 + j]   j=0..7
//   B: lane l holds W[n0 + (l&15)][(l>>4)*8 + j] (B^T = W row-major)
//   C: lane l, reg r -> C[(l>>4)*4 + r][l&15]
// ---------------------------------------------------------------------
// SPLIT=false: ksplit==1, epilogue writes bf16 straight to `out`.
// SPLIT=true:  blockIdx.y writes its f32 partial slab part[y][M][N]
//              (plain stores — no atomics, no pre-zero, deterministic);
//              reduce_cast_kernel folds the slabs to bf16.
template <bool SPLIT, int MT>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    void* __restrict__ out, const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ W, int M, int N, int K, int ksplit) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n0 = blockIdx.x * 128 + wave * 32;  // this wave's n-tile
  if (n0 >= N) return;
  int kbegin = 0, kend = K;
  if (SPLIT) {
    const int kchunk = (K / ksplit + 31) & ~31;  // multiple of 32
    kbegin = blockIdx.y * kchunk;
    kend = kbegin + kchunk;
    if (kend > K) kend = K;
  }

  // K-loop notes (measured via .s dumps):
  //  - any branch or pragma-driven unroll in the loop kept it ROLLED
  //    with a vmcnt(0) drain per iteration -> latency-serialized;
  //    the manual unrolled block issues all its loads before the
  //    first MFMA waits, so a wave keeps ~384B in flight
  //  - out-of-range rows are CLAMPED, not masked: their products land
  //    only in C cells (m>=M / n>=N) the epilogue never writes
  //  - MT in {1,2} A-row tiles: M<=16 or 17..32 (decode batch 32
  //    doubles served requests per weight pass at equal HBM traffic)
  const int arow = lane & 15;
  const int kb = (lane >> 4) * 8;
  const int brow0 = n0 + (lane & 15);
  const int brow1 = brow0 + 16;

  f32x4v acc0[MT], acc1[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    acc0[t] = {0.f, 0.f, 0.f, 0.f};
    acc1[t] = {0.f, 0.f, 0.f, 0.f};
  }

  const unsigned short* aptr[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
    aptr[t] = A + (size_t)min(arow + 16 * t, M - 1) * K + kb;
  const unsigned short* bptr0 =
      W + (size_t)min(brow0, N - 1) * K + kb;
  const unsigned short* bptr1 =
      W + (size_t)min(brow1, N - 1) * K + kb;

#ifndef MLRUN_GEMM_UNR1
#define MLRUN_GEMM_UNR1 16
#endif
  constexpr int UNR = (MT == 1) ? MLRUN_GEMM_UNR1 : (MT == 2 ? 4 : 2);
  int k = kbegin;
  const int kend8 = kbegin + ((kend - kbegin) & ~(UNR * 32 - 1));
  for (; k < kend8; k += UNR * 32) {
    short8v af[UNR][MT], bf0[UNR], bf1[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
#pragma unroll
      for (int t = 0; t < MT; ++t)
        af[u][t] =
            *reinterpret_cast<const short8v*>(aptr[t] + k + u * 32);
      bf0[u] = *reinterpret_cast<const short8v*>(bptr0 + k + u * 32);
      bf1[u] = *reinterpret_cast<const short8v*>(bptr1 + k + u * 32);
    }
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[u][t], bf0[u], acc0[t], 0, 0, 0);
        acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[u][t], bf1[u], acc1[t], 0, 0, 0);
      }
    }
  }
  for (; k < kend; k += 32) {
    short8v bf0 = *reinterpret_cast<const short8v*>(bptr0 + k);
    short8v bf1 = *reinterpret_cast<const short8v*>(bptr1 + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      short8v af = *reinterpret_cast<const short8v*>(aptr[t] + k);
      acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf0, acc0[t],
                                                        0, 0, 0);
      acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf1, acc1[t],
                                                        0, 0, 0);
    }
  }

  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  const bool b0_valid = brow0 < N;
  const bool b1_valid = brow1 < N;
  if (SPLIT) {
    float* part = (float*)out + (size_t)blockIdx.y * M * N;
#pragma unroll
    for (int t = 0; t < MT; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = 16 * t + crow_base + r;
        if (m >= M) continue;
        if (b0_valid) part[(size_t)m * N + n0 + ccol] = acc0[t][r];
        if (b1_valid) part[(size_t)m * N + n0 + 16 + ccol] = acc1[t][r];
      }
  } else {
    unsigned short* dst = (unsigned short*)out;
#pragma unroll
    for (int t = 0; t < MT; ++t)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = 16 * t + crow_base + r;
        if (m >= M) continue;
        if (b0_valid) dst[(size_t)m * N + n0 + ccol] = f2bf(acc0[t][r]);
        if (b1_valid)
          dst[(size_t)m * N + n0 + 16 + ccol] = f2bf(acc1[t][r]);
      }
  }
}

// Wave-split-K variant: the 4 waves of a workgroup cooperate on ONE
// 32-wide n-tile, each covering K/4 of the k-chunk, combining through
// LDS.  4x the workgroup count of skinny_gemm_kernel at the same
// split-K slab traffic -> 4x the waves/SIMD for latency hiding
// (profile: N=4096 projections were latency-bound at 2 waves/SIMD).
template <bool SPLIT, int MT, int UNR2 = 4, int NB = 2>
__global__ __launch_bounds__(256) void skinny_gemm_ws_kernel(
    void* __restrict__ out, const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ W, int M, int N, int K, int ksplit) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  constexpr int NT = 16 * NB;      // n-tile width (16 or 32)
  const int n0 = blockIdx.x * NT;
  if (n0 >= N) return;
  int cbegin = 0, cend = K;
  if (SPLIT) {
    const int kchunk = (K / ksplit + 31) & ~31;
    cbegin = blockIdx.y * kchunk;
    cend = cbegin + kchunk;
    if (cend > K) cend = K;
  }
  // each wave covers a quarter of [cbegin, cend)
  const int clen = cend - cbegin;
  const int per_wave = ((clen / 4) + 31) & ~31;
  int kbegin = cbegin + wave * per_wave;
  int kend = kbegin + per_wave;
  if (kend > cend) kend = cend;

  // K-loop notes (measured via .s dumps):
  //  - any branch or pragma-driven unroll in the loop kept it ROLLED
  //    with a vmcnt(0) drain per iteration -> latency-serialized;
  //    the manual unrolled block issues all its loads before the
  //    first MFMA waits, so a wave keeps ~384B in flight
  //  - out-of-range rows are CLAMPED, not masked: their products land
  //    only in C cells (m>=M / n>=N) the epilogue never writes
  //  - MT in {1,2} A-row tiles: M<=16 or 17..32 (decode batch 32
  //    doubles served requests per weight pass at equal HBM traffic)
  //  - NB=1 halves the n-tile to 16: double the workgroup count at
  //    UNCHANGED per-wave K length (grid-starved shapes like qkv) at
  //    the cost of re-reading A once more (L2-resident, tiny)
  const int arow = lane & 15;
  const int kb = (lane >> 4) * 8;
  const int brow0 = n0 + (lane & 15);
  const int brow1 = brow0 + 16;

  f32x4v acc0[MT], acc1[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    acc0[t] = {0.f, 0.f, 0.f, 0.f};
    acc1[t] = {0.f, 0.f, 0.f, 0.f};
  }

  const unsigned short* aptr[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
    aptr[t] = A + (size_t)min(arow + 16 * t, M - 1) * K + kb;
  const unsigned short* bptr0 =
      W + (size_t)min(brow0, N - 1) * K + kb;
  const unsigned short* bptr1 =
      W + (size_t)min(NB == 2 ? brow1 : brow0, N - 1) * K + kb;

#ifndef MLRUN_GEMM_UNR1
#define MLRUN_GEMM_UNR1 16
#endif
  constexpr int UNR = (MT == 1) ? MLRUN_GEMM_UNR1
                                : (MT == 2 ? (NB == 1 ? 8 : UNR2) : 2);
  int k = kbegin;
  const int kend8 = kbegin + ((kend - kbegin) & ~(UNR * 32 - 1));
  for (; k < kend8; k += UNR * 32) {
    short8v af[UNR][MT], bf0[UNR], bf1[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
#pragma unroll
      for (int t = 0; t < MT; ++t)
        af[u][t] =
            *reinterpret_cast<const short8v*>(aptr[t] + k + u * 32);
      bf0[u] = *reinterpret_cast<const short8v*>(bptr0 + k + u * 32);
      if (NB == 2)
        bf1[u] = *reinterpret_cast<const short8v*>(bptr1 + k + u * 32);
    }
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[u][t], bf0[u], acc0[t], 0, 0, 0);
        if (NB == 2)
          acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[u][t], bf1[u], acc1[t], 0, 0, 0);
      }
    }
  }
  for (; k < kend; k += 32) {
    short8v bf0 = *reinterpret_cast<const short8v*>(bptr0 + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      short8v af = *reinterpret_cast<const short8v*>(aptr[t] + k);
      acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf0, acc0[t],
                                                        0, 0, 0);
      if (NB == 2) {
        short8v bf1 = *reinterpret_cast<const short8v*>(bptr1 + k);
        acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf1,
                                                          acc1[t], 0, 0,
                                                          0);
      }
    }
  }

  // combine the 4 waves' partials through LDS
  __shared__ float comb[4][16 * MT][NT];  // [wave][m][n] 4-16 KiB
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      comb[wave][16 * t + crow_base + r][ccol] = acc0[t][r];
      if (NB == 2)
        comb[wave][16 * t + crow_base + r][ccol + 16] = acc1[t][r];
    }
  __syncthreads();
  if (wave == 0) {
    // 64 lanes fold 16*MT x NT cells: lane covers 4*MT*NB cells
    constexpr int CPL = 4 * MT * NB;
#pragma unroll
    for (int c = 0; c < CPL; ++c) {
      const int cell = lane * CPL + c;
      const int m = cell / NT;
      const int n = cell % NT;
      if (m >= M || n0 + n >= N) continue;
      float sum = comb[0][m][n] + comb[1][m][n] + comb[2][m][n] +
                  comb[3][m][n];
      if (SPLIT) {
        float* part = (float*)out + (size_t)blockIdx.y * M * N;
        part[(size_t)m * N + n0 + n] = sum;
      } else {
        ((unsigned short*)out)[(size_t)m * N + n0 + n] = f2bf(sum);
      }
    }
  }
}

// Software-pipelined wave-split variant (round-2 design doc §1, built
// in C++ instead of inline asm): a DEPTH-stage rotating register ring
// of one 32-K step each (MT A-vecs + NB B-rows).  The fully-unrolled
// ring gives LLVM's waitcnt pass compile-time register stages, so it
// emits PARTIAL vmcnt waits (retire the oldest stage while
// (MT+NB)*(DEPTH-1) newer loads stay in flight) — steady-state loads
// in flight instead of the base kernel's issue-burst-then-drain
// blocks.  Staging VGPRs = DEPTH * (MT+NB) * 4; DEPTH<=4 stays under
// the round-1 occupancy backfire threshold (64 staging VGPRs).
template <bool SPLIT, int MT, int DEPTH, int NB = 2>
__global__ __launch_bounds__(256) void skinny_gemm_ws_pipe_kernel(
    void* __restrict__ out, const unsigned short* __restrict__ A,
    const unsigned short* __restrict__ W, int M, int N, int K, int ksplit) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  constexpr int NT = 16 * NB;
  const int n0 = blockIdx.x * NT;
  if (n0 >= N) return;
  int cbegin = 0, cend = K;
  if (SPLIT) {
    const int kchunk = (K / ksplit + 31) & ~31;
    cbegin = blockIdx.y * kchunk;
    cend = cbegin + kchunk;
    if (cend > K) cend = K;
  }
  const int clen = cend - cbegin;
  const int per_wave = ((clen / 4) + 31) & ~31;
  int kbegin = cbegin + wave * per_wave;
  int kend = kbegin + per_wave;
  if (kend > cend) kend = cend;

  const int arow = lane & 15;
  const int kb = (lane >> 4) * 8;
  const int brow0 = n0 + (lane & 15);
  const int brow1 = brow0 + 16;

  f32x4v acc0[MT], acc1[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    acc0[t] = {0.f, 0.f, 0.f, 0.f};
    acc1[t] = {0.f, 0.f, 0.f, 0.f};
  }
  const unsigned short* aptr[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
    aptr[t] = A + (size_t)min(arow + 16 * t, M - 1) * K + kb;
  const unsigned short* bptr0 =
      W + (size_t)min(brow0, N - 1) * K + kb;
  const unsigned short* bptr1 =
      W + (size_t)min(NB == 2 ? brow1 : brow0, N - 1) * K + kb;

  short8v a_ring[DEPTH][MT], b0_ring[DEPTH], b1_ring[DEPTH];
  const int nsteps_total = (kend - kbegin) / 32;
  // main region: a multiple of DEPTH steps; the rest runs in the tail
  const int ngroups = nsteps_total / DEPTH;
  const int nsteps = ngroups > 0 ? ngroups * DEPTH : 0;
  int kload = kbegin;
  int k = kbegin + nsteps * 32;
  if (nsteps) {
#pragma unroll
    for (int s = 0; s < DEPTH; ++s) {
#pragma unroll
      for (int t = 0; t < MT; ++t)
        a_ring[s][t] =
            *reinterpret_cast<const short8v*>(aptr[t] + kload);
      b0_ring[s] = *reinterpret_cast<const short8v*>(bptr0 + kload);
      if (NB == 2)
        b1_ring[s] = *reinterpret_cast<const short8v*>(bptr1 + kload);
      kload += 32;
    }
    for (int g = DEPTH; g < nsteps; g += DEPTH) {
#pragma unroll
      for (int s = 0; s < DEPTH; ++s) {
#pragma unroll
        for (int t = 0; t < MT; ++t) {
          acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_ring[s][t], b0_ring[s], acc0[t], 0, 0, 0);
          if (NB == 2)
            acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_ring[s][t], b1_ring[s], acc1[t], 0, 0, 0);
        }
#pragma unroll
        for (int t = 0; t < MT; ++t)
          a_ring[s][t] =
              *reinterpret_cast<const short8v*>(aptr[t] + kload);
        b0_ring[s] = *reinterpret_cast<const short8v*>(bptr0 + kload);
        if (NB == 2)
          b1_ring[s] = *reinterpret_cast<const short8v*>(bptr1 + kload);
        kload += 32;
      }
    }
#pragma unroll
    for (int s = 0; s < DEPTH; ++s) {
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_ring[s][t], b0_ring[s], acc0[t], 0, 0, 0);
        if (NB == 2)
          acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_ring[s][t], b1_ring[s], acc1[t], 0, 0, 0);
      }
    }
  }
  for (; k < kend; k += 32) {
    short8v bf0 = *reinterpret_cast<const short8v*>(bptr0 + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      short8v af = *reinterpret_cast<const short8v*>(aptr[t] + k);
      acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf0, acc0[t],
                                                        0, 0, 0);
      if (NB == 2) {
        short8v bf1 = *reinterpret_cast<const short8v*>(bptr1 + k);
        acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf1,
                                                          acc1[t], 0, 0,
                                                          0);
      }
    }
  }

  __shared__ float comb[4][16 * MT][NT];
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      comb[wave][16 * t + crow_base + r][ccol] = acc0[t][r];
      if (NB == 2)
        comb[wave][16 * t + crow_base + r][ccol + 16] = acc1[t][r];
    }
  __syncthreads();
  if (wave == 0) {
    constexpr int CPL = 4 * MT * NB;
#pragma unroll
    for (int c = 0; c < CPL; ++c) {
      const int cell = lane * CPL + c;
      const int m = cell / NT;
      const int n = cell % NT;
      if (m >= M || n0 + n >= N) continue;
      float sum = comb[0][m][n] + comb[1][m][n] + comb[2][m][n] +
                  comb[3][m][n];
      if (SPLIT) {
        float* part = (float*)out + (size_t)blockIdx.y * M * N;
        part[(size_t)m * N + n0 + n] = sum;
      } else {
        ((unsigned short*)out)[(size_t)m * N + n0 + n] = f2bf(sum);
      }
    }
  }
}

__global__ void reduce_cast_kernel(unsigned short* __restrict__ out,
                                   const float* __restrict__ part,
                                   long long mn, int ksplit);

// ---------------------------------------------------------------------
// FP8 (OCP e4m3) decode GEMM — the opt-in fp8-weight serving mode.
// Decode is weight-read bound, so fp8 weights halve the dominant
// traffic; activations quantize per-row (dynamic) to fp8, MFMA runs
// native v_mfma_f32_16x16x32_fp8_fp8 (same rate as bf16 — we're after
// the BYTES, not the FLOPs), and the f32 accumulator is rescaled by
// a_scale[m] * w_scale[n] in the epilogue.  Wave-split-K structure
// mirrors skinny_gemm_ws_kernel.
// ---------------------------------------------------------------------
template <bool SPLIT, int MT>
__global__ __launch_bounds__(256) void skinny_gemm_fp8_kernel(
    void* __restrict__ out, const unsigned char* __restrict__ A8,
    const float* __restrict__ a_scale,
    const unsigned char* __restrict__ W8,
    const float* __restrict__ w_scale, int M, int N, int K, int ksplit) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n0 = blockIdx.x * 32;
  if (n0 >= N) return;
  int cbegin = 0, cend = K;
  if (SPLIT) {
    const int kchunk = (K / ksplit + 63) & ~63;
    cbegin = blockIdx.y * kchunk;
    cend = cbegin + kchunk;
    if (cend > K) cend = K;
  }
  const int clen = cend - cbegin;
  const int per_wave = ((clen / 4) + 63) & ~63;  // k-pair aligned
  int kbegin = cbegin + wave * per_wave;
  int kend = kbegin + per_wave;
  if (kend > cend) kend = cend;

  // PAIR-SWIZZLED layout: A8/W8 are stored so each lane's 16B load
  // carries its fragments for TWO consecutive K-steps
  // (new[(s/2)*64 + c*16 + (s%2)*8 + j] = orig[s*32 + c*8 + j]).
  // fp8 at the plain layout measured EQUAL to bf16 time: halving
  // bytes/load doesn't help a latency-bound loop — halving the LOAD
  // COUNT does.  K must be a multiple of 64.
  const int arow = lane & 15;
  const int kb2 = (lane >> 4) * 16;  // 16B per lane per k-PAIR
  const int brow0 = n0 + (lane & 15);
  const int brow1 = brow0 + 16;

  f32x4v acc0[MT], acc1[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    acc0[t] = {0.f, 0.f, 0.f, 0.f};
    acc1[t] = {0.f, 0.f, 0.f, 0.f};
  }
  typedef __attribute__((ext_vector_type(2))) long long2v;
  const unsigned char* aptr[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
    aptr[t] = A8 + (size_t)min(arow + 16 * t, M - 1) * K + kb2;
  const unsigned char* bptr0 = W8 + (size_t)min(brow0, N - 1) * K + kb2;
  const unsigned char* bptr1 = W8 + (size_t)min(brow1, N - 1) * K + kb2;

  constexpr int UNR = (MT == 1) ? 8 : (MT == 2 ? 4 : 2);  // k-pairs
  int k = kbegin;
  const int kend8 = kbegin + ((kend - kbegin) & ~(UNR * 64 - 1));
  for (; k < kend8; k += UNR * 64) {
    long2v af[UNR][MT], bf0[UNR], bf1[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
#pragma unroll
      for (int t = 0; t < MT; ++t)
        af[u][t] =
            *reinterpret_cast<const long2v*>(aptr[t] + k + u * 64);
      bf0[u] = *reinterpret_cast<const long2v*>(bptr0 + k + u * 64);
      bf1[u] = *reinterpret_cast<const long2v*>(bptr1 + k + u * 64);
    }
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            af[u][t][0], bf0[u][0], acc0[t], 0, 0, 0);
        acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            af[u][t][0], bf1[u][0], acc1[t], 0, 0, 0);
        acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            af[u][t][1], bf0[u][1], acc0[t], 0, 0, 0);
        acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            af[u][t][1], bf1[u][1], acc1[t], 0, 0, 0);
      }
    }
  }
  for (; k < kend; k += 64) {
    long2v bf0 = *reinterpret_cast<const long2v*>(bptr0 + k);
    long2v bf1 = *reinterpret_cast<const long2v*>(bptr1 + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      long2v af = *reinterpret_cast<const long2v*>(aptr[t] + k);
      acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
          af[0], bf0[0], acc0[t], 0, 0, 0);
      acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
          af[0], bf1[0], acc1[t], 0, 0, 0);
      acc0[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
          af[1], bf0[1], acc0[t], 0, 0, 0);
      acc1[t] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
          af[1], bf1[1], acc1[t], 0, 0, 0);
    }
  }

  __shared__ float comb[4][16 * MT][32];
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      comb[wave][16 * t + crow_base + r][ccol] = acc0[t][r];
      comb[wave][16 * t + crow_base + r][ccol + 16] = acc1[t][r];
    }
  __syncthreads();
  if (wave == 0) {
#pragma unroll
    for (int c = 0; c < 8 * MT; ++c) {
      const int cell = lane * 8 * MT + c;
      const int m = cell >> 5;
      const int n = cell & 31;
      if (m >= M || n0 + n >= N) continue;
      float sum = (comb[0][m][n] + comb[1][m][n] + comb[2][m][n] +
                   comb[3][m][n]) * a_scale[m] * w_scale[n0 + n];
      if (SPLIT) {
        float* part = (float*)out + (size_t)blockIdx.y * M * N;
        part[(size_t)m * N + n0 + n] = sum;
      } else {
        ((unsigned short*)out)[(size_t)m * N + n0 + n] = f2bf(sum);
      }
    }
  }
}

void launch_skinny_gemm_fp8(void* out_bf16, void* part_f32, const void* A8,
                            const void* a_scale, const void* W8,
                            const void* w_scale, int M, int N, int K,
                            int ksplit, void* stream) {
  const int nblocks = (N + 31) / 32;
  const dim3 grid(nblocks, ksplit > 1 ? ksplit : 1);
#define FP8_DISPATCH(SPLIT, DEST)                                         \
  do {                                                                    \
    if (M > 32)                                                           \
      hipLaunchKernelGGL((skinny_gemm_fp8_kernel<SPLIT, 4>), grid,        \
                         dim3(256), 0, (hipStream_t)stream, DEST,         \
                         (const unsigned char*)A8, (const float*)a_scale, \
                         (const unsigned char*)W8, (const float*)w_scale, \
                         M, N, K, ksplit);                                \
    else if (M > 16)                                                      \
      hipLaunchKernelGGL((skinny_gemm_fp8_kernel<SPLIT, 2>), grid,        \
                         dim3(256), 0, (hipStream_t)stream, DEST,         \
                         (const unsigned char*)A8, (const float*)a_scale, \
                         (const unsigned char*)W8, (const float*)w_scale, \
                         M, N, K, ksplit);                                \
    else                                                                  \
      hipLaunchKernelGGL((skinny_gemm_fp8_kernel<SPLIT, 1>), grid,        \
                         dim3(256), 0, (hipStream_t)stream, DEST,         \
                         (const unsigned char*)A8, (const float*)a_scale, \
                         (const unsigned char*)W8, (const float*)w_scale, \
                         M, N, K, ksplit);                                \
  } while (0)
  if (ksplit <= 1) {
    FP8_DISPATCH(false, out_bf16);
  } else {
    FP8_DISPATCH(true, part_f32);
    long long mn = (long long)M * N;
    long long blocks = (mn / 4 + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(reduce_cast_kernel, dim3((int)blocks), dim3(256), 0,
                       (hipStream_t)stream, (unsigned short*)out_bf16,
                       (const float*)part_f32, mn, ksplit);
  }
#undef FP8_DISPATCH
}

// dynamic per-row fp8 quantization of bf16 activations:
// a8[r][k] = round(a[r][k] / (amax_r / 448)); a_scale[r] = amax_r / 448
__global__ void quant_fp8_rows_kernel(unsigned char* __restrict__ a8,
                                      float* __restrict__ a_scale,
                                      const unsigned short* __restrict__ a,
                                      int cols) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const unsigned short* ar = a + (size_t)row * cols;
  unsigned char* outr = a8 + (size_t)row * cols;
  float amax = 0.f;
  for (int i = tid * 8; i < cols; i += blockDim.x * 8) {
    ushort8 v = *reinterpret_cast<const ushort8*>(ar + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(bf2f(v.v[j])));
  }
  __shared__ float red[32];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, 64));
  if ((tid & 63) == 0) red[tid >> 6] = amax;
  __syncthreads();
  float gmax = 0.f;
  for (int w = 0; w < (blockDim.x >> 6); ++w) gmax = fmaxf(gmax, red[w]);
  const float scale = gmax > 0.f ? gmax / 448.f : 1.f;
  const float inv = 1.f / scale;
  if (tid == 0) a_scale[row] = scale;
  for (int i = tid * 8; i < cols; i += blockDim.x * 8) {
    ushort8 v = *reinterpret_cast<const ushort8*>(ar + i);
    unsigned char q[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float scaled = bf2f(v.v[j]) * inv;
      __hip_fp8_e4m3 f8(scaled);
      q[j] = f8.__x;
    }
    // pair-swizzled destination (matches skinny_gemm_fp8_kernel):
    // orig chunk i = s*32 + c*8 -> (s/2)*64 + c*16 + (s%2)*8
    const int s = i >> 5;
    const int c = (i >> 3) & 3;
    const int dst = ((s >> 1) << 6) + (c << 4) + ((s & 1) << 3);
    *reinterpret_cast<uint64_t*>(outr + dst) =
        *reinterpret_cast<const uint64_t*>(q);
  }
}

void launch_quant_fp8_rows(void* a8, void* a_scale, const void* a, int rows,
                           int cols, void* stream) {
  int threads = cols >= 2048 ? 256 : 64;
  hipLaunchKernelGGL(quant_fp8_rows_kernel, dim3(rows), dim3(threads), 0,
                     (hipStream_t)stream, (unsigned char*)a8,
                     (float*)a_scale, (const unsigned short*)a, cols);
}

// fold ksplit partial slabs [ksplit, M, N] f32 -> bf16 [M, N]
__global__ void reduce_cast_kernel(unsigned short* __restrict__ out,
                                   const float* __restrict__ part,
                                   long long mn, int ksplit) {
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (; i + 3 < mn; i += stride) {
    f32x4v sum = *reinterpret_cast<const f32x4v*>(part + i);
    for (int s = 1; s < ksplit; ++s) {
      f32x4v v = *reinterpret_cast<const f32x4v*>(part + (size_t)s * mn + i);
#pragma unroll
      for (int j = 0; j < 4; ++j) sum[j] += v[j];
    }
    unsigned short o[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2bf(sum[j]);
    *reinterpret_cast<uint64_t*>(out + i) =
        *reinterpret_cast<const uint64_t*>(o);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (long long t = (mn & ~3LL); t < mn; ++t) {
      float sum = part[t];
      for (int s = 1; s < ksplit; ++s) sum += part[(size_t)s * mn + t];
      out[t] = f2bf(sum);
    }
  }
}

// emit split-K slabs only (consumer kernel folds them)
// MT=2 wave-split K-loop unroll depth (4 default; 8 doubles the
// in-flight bytes per wave at +64 VGPRs) — env knob for e2e A/B.
static int gemm_unr2_env() {
  static int v = [] {
    const char* e = getenv("MLRUN_GEMM_UNR2");
    return e ? atoi(e) : 4;
  }();
  return v;
}

// MLRUN_GEMM_PIPE=0|3|4|6 selects the software-pipelined ring kernel
// (ring depth) for the M in (16,32] decode shapes.  Default 4:
// measured +1.1% e2e over the burst-unrolled kernel at B=32
// (gpurun_out/pipe_ab.log — the per-wave outstanding-bytes wall caps
// the gain, as docs/round2_kernel_designs.md §1 predicted); numerics
// verified vs fp32 on all decode shapes (scripts/check_gemm_pipe.py).
static int gemm_pipe_env() {
  static int v = [] {
    const char* e = getenv("MLRUN_GEMM_PIPE");
    return e ? atoi(e) : 4;
  }();
  return v;
}

template <bool SPLIT>
static bool launch_pipe_mt2(void* out, const void* A, const void* W,
                            int M, int N, int K, int ksplit,
                            const dim3& grid, void* stream) {
  const int depth = gemm_pipe_env();
  if (depth < 2) return false;
  auto kern = depth >= 6 ? (skinny_gemm_ws_pipe_kernel<SPLIT, 2, 6>)
              : depth >= 4 ? (skinny_gemm_ws_pipe_kernel<SPLIT, 2, 4>)
                           : (skinny_gemm_ws_pipe_kernel<SPLIT, 2, 3>);
  hipLaunchKernelGGL(kern, grid, dim3(256), 0, (hipStream_t)stream, out,
                     (const unsigned short*)A, (const unsigned short*)W,
                     M, N, K, ksplit);
  return true;
}

void launch_skinny_gemm_slabs(void* part_f32, const void* A, const void* W,
                              int M, int N, int K, int ksplit, int variant,
                              void* stream) {
  const int nblocks = variant == 3 ? (N + 15) / 16
                      : variant >= 1 ? (N + 31) / 32 : (N + 127) / 128;
  const dim3 grid(nblocks, ksplit);
  if (variant == 3) {
    // 16-wide n-tiles: 2x the workgroups at unchanged per-wave K
    if (M > 32)
      hipLaunchKernelGGL((skinny_gemm_ws_kernel<true, 4, 4, 1>), grid,
                         dim3(256), 0, (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
    else if (M > 16)
      hipLaunchKernelGGL((skinny_gemm_ws_kernel<true, 2, 4, 1>), grid,
                         dim3(256), 0, (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
    else
      hipLaunchKernelGGL((skinny_gemm_ws_kernel<true, 1, 4, 1>), grid,
                         dim3(256), 0, (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
  } else if (variant >= 1) {
    if (M > 32)
      hipLaunchKernelGGL((skinny_gemm_ws_kernel<true, 4>), grid, dim3(256),
                         0, (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
    else if (M > 16) {
      if (!launch_pipe_mt2<true>(part_f32, A, W, M, N, K, ksplit, grid,
                                 stream))
        hipLaunchKernelGGL(gemm_unr2_env() >= 8
                               ? (skinny_gemm_ws_kernel<true, 2, 8>)
                               : (skinny_gemm_ws_kernel<true, 2, 4>),
                           grid, dim3(256),
                           0, (hipStream_t)stream, part_f32,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, ksplit);
    }
    else
      hipLaunchKernelGGL((skinny_gemm_ws_kernel<true, 1>), grid, dim3(256),
                         0, (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
  } else {
    if (M > 32)
      hipLaunchKernelGGL((skinny_gemm_kernel<true, 4>), grid, dim3(256), 0,
                         (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
    else if (M > 16)
      hipLaunchKernelGGL((skinny_gemm_kernel<true, 2>), grid, dim3(256), 0,
                         (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
    else
      hipLaunchKernelGGL((skinny_gemm_kernel<true, 1>), grid, dim3(256), 0,
                         (hipStream_t)stream, part_f32,
                         (const unsigned short*)A, (const unsigned short*)W,
                         M, N, K, ksplit);
  }
}

void launch_skinny_gemm(void* out_bf16, void* part_f32, const void* A,
                        const void* W, int M, int N, int K, int ksplit,
                        int variant, void* stream) {
  if (ksplit < 1) ksplit = 1;
  const int nblocks = variant == 3 ? (N + 15) / 16
                      : variant >= 1 ? (N + 31) / 32 : (N + 127) / 128;
  if (ksplit == 1) {
    if (variant == 3) {
      if (M > 32)
        hipLaunchKernelGGL((skinny_gemm_ws_kernel<false, 4, 4, 1>),
                           dim3(nblocks), dim3(256), 0,
                           (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
      else if (M > 16)
        hipLaunchKernelGGL((skinny_gemm_ws_kernel<false, 2, 4, 1>),
                           dim3(nblocks), dim3(256), 0,
                           (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
      else
        hipLaunchKernelGGL((skinny_gemm_ws_kernel<false, 1, 4, 1>),
                           dim3(nblocks), dim3(256), 0,
                           (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
    } else if (variant >= 1) {
      if (M > 32)
        hipLaunchKernelGGL((skinny_gemm_ws_kernel<false, 4>),
                           dim3(nblocks), dim3(256), 0,
                           (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
      else if (M > 16) {
        if (!launch_pipe_mt2<false>(out_bf16, A, W, M, N, K, 1,
                                    dim3(nblocks), stream))
          hipLaunchKernelGGL(gemm_unr2_env() >= 8
                                 ? (skinny_gemm_ws_kernel<false, 2, 8>)
                                 : (skinny_gemm_ws_kernel<false, 2, 4>),
                             dim3(nblocks), dim3(256), 0,
                             (hipStream_t)stream, out_bf16,
                             (const unsigned short*)A,
                             (const unsigned short*)W, M, N, K, 1);
      }
      else
        hipLaunchKernelGGL((skinny_gemm_ws_kernel<false, 1>),
                           dim3(nblocks), dim3(256), 0,
                           (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
    } else {
      if (M > 32)
        hipLaunchKernelGGL((skinny_gemm_kernel<false, 4>), dim3(nblocks),
                           dim3(256), 0, (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
      else if (M > 16)
        hipLaunchKernelGGL((skinny_gemm_kernel<false, 2>), dim3(nblocks),
                           dim3(256), 0, (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
      else
        hipLaunchKernelGGL((skinny_gemm_kernel<false, 1>), dim3(nblocks),
                           dim3(256), 0, (hipStream_t)stream, out_bf16,
                           (const unsigned short*)A,
                           (const unsigned short*)W, M, N, K, 1);
    }
  } else {
    launch_skinny_gemm_slabs(part_f32, A, W, M, N, K, ksplit, variant,
                             stream);
    long long mn = (long long)M * N;
    long long blocks = (mn / 4 + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(reduce_cast_kernel, dim3((int)blocks), dim3(256), 0,
                       (hipStream_t)stream, (unsigned short*)out_bf16,
                       (const float*)part_f32, mn, ksplit);
  }
}

// f32 -> bf16 flat cast (epilogue after k-split accumulate)
__global__ void cast_f32_bf16_kernel(unsigned short* __restrict__ out,
                                     const float* __restrict__ in,
                                     long long n) {
  long long i = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long long stride = (long long)gridDim.x * blockDim.x * 4;
  for (; i + 3 < n; i += stride) {
    f32x4v v = *reinterpret_cast<const f32x4v*>(in + i);
    unsigned short o[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2bf(v[j]);
    *reinterpret_cast<uint64_t*>(out + i) =
        *reinterpret_cast<const uint64_t*>(o);
  }
  // tail
  if (blockIdx.x == 0 && threadIdx.x == 0) {
    for (long long t = (n & ~3LL); t < n; ++t) out[t] = f2bf(in[t]);
  }
}

void launch_cast_f32_bf16(void* out, const void* in, long long n,
                          void* stream) {
  long long blocks = (n / 4 + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (unsigned short*)out,
                     (const float*)in, n);
}

// zero f32 buffer (before atomic k-split accumulate)
__global__ void zero_f32_kernel(float* __restrict__ p, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = 0.f;
}

void launch_zero_f32(void* p, long long n, void* stream) {
  long long blocks = (n + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(zero_f32_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (float*)p, n);
}

// ---------------------------------------------------------------------
// Slab-consumer fusions: when a decode GEMM runs split-K, its f32
// partial slabs are folded INSIDE the consuming kernel instead of a
// separate reduce_cast pass — one less launch and one less bf16
// round-trip per projection (profile: reduce_cast was ~8% of decode).
// slabs layout: [ksplit, rows, cols] f32.
// ---------------------------------------------------------------------

DEV float slab_sum(const float* __restrict__ slabs, size_t idx,
                   size_t slab_stride, int ksplit) {
  float sum = slabs[idx];
  for (int s = 1; s < ksplit; ++s) sum += slabs[idx + s * slab_stride];
  return sum;
}

// residual <- sum(slabs) + residual; out <- rmsnorm(residual) * weight
__global__ void fused_add_rmsnorm_slab_kernel(
    unsigned short* __restrict__ out, unsigned short* __restrict__ residual,
    const float* __restrict__ slabs,
    const unsigned short* __restrict__ weight, int hidden, float eps,
    int ksplit) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const size_t rowbase = (size_t)row * hidden;
  const size_t slab_stride = (size_t)gridDim.x * hidden;
  unsigned short* rr = residual + rowbase;
  unsigned short* outr = out + rowbase;

  float sumsq = 0.f;
  for (int i = tid * 4; i < hidden; i += nthreads * 4) {
    float z[4];
#pragma unroll
    for (int j = 0; j < 4; ++j)
      z[j] = slab_sum(slabs, rowbase + i + j, slab_stride, ksplit);
    unsigned short rv[4];
    *reinterpret_cast<uint64_t*>(rv) =
        *reinterpret_cast<const uint64_t*>(rr + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      z[j] += bf2f(rv[j]);
      rv[j] = f2bf(z[j]);
      sumsq += z[j] * z[j];
    }
    *reinterpret_cast<uint64_t*>(rr + i) =
        *reinterpret_cast<const uint64_t*>(rv);
  }
  __shared__ float red[32];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    sumsq += __shfl_down(sumsq, off, 64);
  if ((tid & 63) == 0) red[tid >> 6] = sumsq;
  __syncthreads();
  if (tid < (nthreads >> 6)) sumsq = red[tid];
  else sumsq = 0.f;
  if (tid < 64) {
#pragma unroll
    for (int off = 2; off > 0; off >>= 1)
      sumsq += __shfl_down(sumsq, off, 64);
  }
  if (tid == 0) red[0] = sumsq;
  __syncthreads();
  const float inv = rsqrtf(red[0] / hidden + eps);
  for (int i = tid * 8; i < hidden; i += nthreads * 8) {
    ushort8 zv = *reinterpret_cast<const ushort8*>(rr + i);
    ushort8 wv = *reinterpret_cast<const ushort8*>(weight + i);
    ushort8 ov;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov.v[j] = f2bf(bf2f(zv.v[j]) * inv * bf2f(wv.v[j]));
    *reinterpret_cast<ushort8*>(outr + i) = ov;
  }
}

void launch_fused_add_rmsnorm_slab(void* out, void* residual,
                                   const void* slabs, const void* weight,
                                   int rows, int hidden, float eps,
                                   int ksplit, void* stream) {
  int threads = hidden >= 2048 ? 256 : 64;
  hipLaunchKernelGGL(fused_add_rmsnorm_slab_kernel, dim3(rows),
                     dim3(threads), 0, (hipStream_t)stream,
                     (unsigned short*)out, (unsigned short*)residual,
                     (const float*)slabs, (const unsigned short*)weight,
                     hidden, eps, ksplit);
}

// out[r, c] = silu(sum gate_slab) * sum(up_slab); slabs [ks, rows, 2I]
__global__ void swiglu_slab_kernel(unsigned short* __restrict__ out,
                                   const float* __restrict__ slabs,
                                   int rows, int inter, int ksplit) {
  const size_t slab_stride = (size_t)rows * 2 * inter;
  const long long total = (long long)rows * inter / 4;
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    const long long flat = i * 4;
    const int r = (int)(flat / inter);
    const int c = (int)(flat % inter);
    const size_t base = (size_t)r * 2 * inter + c;
    unsigned short o[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float g = slab_sum(slabs, base + j, slab_stride, ksplit);
      const float u = slab_sum(slabs, base + inter + j, slab_stride,
                               ksplit);
      const float s = g / (1.f + __expf(-g));
      o[j] = f2bf(s * u);
    }
    *reinterpret_cast<uint64_t*>(out + (size_t)r * inter + c) =
        *reinterpret_cast<const uint64_t*>(o);
  }
}

void launch_swiglu_slab(void* out, const void* slabs, int rows, int inter,
                        int ksplit, void* stream) {
  long long blocks = ((long long)rows * inter / 4 + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(swiglu_slab_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (unsigned short*)out,
                     (const float*)slabs, rows, inter, ksplit);
}

// qkv slab -> rope(q,k) + KV append + q written to the bf16 qkv buffer
// slabs [ks, B, qkv_row]; qkv_row = (Hq + 2*Hkv) * D
__global__ void rope_kv_slab_kernel(
    unsigned short* __restrict__ qkv, unsigned short* __restrict__ Kc,
    unsigned short* __restrict__ Vc, const float* __restrict__ slabs,
    const int* __restrict__ positions, const float* __restrict__ cos_sin,
    int Hq, int Hkv, int Smax, int D, int qkv_row, int ksplit) {
  const int b = blockIdx.x;
  const int y = blockIdx.y;
  const int i = threadIdx.x;  // 0..D/2-1
  const int half = D >> 1;
  const int pos = positions[b];
  const size_t slab_stride = (size_t)gridDim.x * qkv_row;
  const size_t base = (size_t)b * qkv_row + (size_t)y * D;
  const float lo_f = slab_sum(slabs, base + i, slab_stride, ksplit);
  const float hi_f = slab_sum(slabs, base + i + half, slab_stride, ksplit);
  if (y < Hq + Hkv) {
    const float c = cos_sin[((size_t)pos * half + i) * 2];
    const float s = cos_sin[((size_t)pos * half + i) * 2 + 1];
    const unsigned short lo = f2bf(lo_f * c - hi_f * s);
    const unsigned short hi = f2bf(hi_f * c + lo_f * s);
    if (y < Hq) {  // q: written back for the attention kernel
      unsigned short* qrow = qkv + (size_t)b * qkv_row + (size_t)y * D;
      qrow[i] = lo;
      qrow[i + half] = hi;
    } else {       // k: straight into the cache
      const int kvh = y - Hq;
      unsigned short* dst =
          Kc + (((size_t)b * Hkv + kvh) * Smax + pos) * D;
      dst[i] = lo;
      dst[i + half] = hi;
    }
  } else {         // v: cache only
    const int kvh = y - Hq - Hkv;
    unsigned short* dst = Vc + (((size_t)b * Hkv + kvh) * Smax + pos) * D;
    dst[i] = f2bf(lo_f);
    dst[i + half] = f2bf(hi_f);
  }
}

void launch_rope_kv_slab(void* qkv, void* Kc, void* Vc, const void* slabs,
                         const void* positions, const void* cos_sin, int B,
                         int Hq, int Hkv, int Smax, int D, int qkv_row,
                         int ksplit, void* stream) {
  hipLaunchKernelGGL(rope_kv_slab_kernel, dim3(B, Hq + 2 * Hkv),
                     dim3(D / 2), 0, (hipStream_t)stream,
                     (unsigned short*)qkv, (unsigned short*)Kc,
                     (unsigned short*)Vc, (const float*)slabs,
                     (const int*)positions, (const float*)cos_sin, Hq, Hkv,
                     Smax, D, qkv_row, ksplit);
}

// ---------------------------------------------------------------------
// Fused decode RoPE + KV-cache append: one launch per layer replaces
// {rope(q), rope(k), kv_append}.  qkv is the fused projection buffer
// [B, row_stride] with q at offset 0, k at Hq*D, v at (Hq+Hkv)*D.
// grid: (B, Hq + 2*Hkv); block: D/2 threads.
//  y < Hq:          rope q head in place
//  Hq <= y < +Hkv:  rope k head in place, then copy into Kc[pos]
//  else:            copy v head into Vc[pos]
// ---------------------------------------------------------------------
__global__ void rope_kv_fused_kernel(
    unsigned short* __restrict__ qkv, unsigned short* __restrict__ Kc,
    unsigned short* __restrict__ Vc, const int* __restrict__ positions,
    const float* __restrict__ cos_sin, int Hq, int Hkv, int Smax, int D,
    long long row_stride) {
  const int b = blockIdx.x;
  const int y = blockIdx.y;
  const int i = threadIdx.x;  // 0..D/2-1
  const int half = D >> 1;
  const int pos = positions[b];
  unsigned short* row = qkv + (size_t)b * row_stride;
  if (y < Hq + Hkv) {
    // rope a q or k head
    unsigned short* head = row + (size_t)y * D;
    const float c = cos_sin[((size_t)pos * half + i) * 2];
    const float s = cos_sin[((size_t)pos * half + i) * 2 + 1];
    const float a = bf2f(head[i]);
    const float bvf = bf2f(head[i + half]);
    const unsigned short lo = f2bf(a * c - bvf * s);
    const unsigned short hi = f2bf(bvf * c + a * s);
    head[i] = lo;
    head[i + half] = hi;
    if (y >= Hq) {
      const int kvh = y - Hq;
      unsigned short* dst = Kc +
          (((size_t)b * Hkv + kvh) * Smax + pos) * D;
      dst[i] = lo;
      dst[i + half] = hi;
    }
  } else {
    const int kvh = y - Hq - Hkv;
    const unsigned short* src = row + (size_t)(Hq + Hkv + kvh) * D;
    unsigned short* dst = Vc + (((size_t)b * Hkv + kvh) * Smax + pos) * D;
    dst[i] = src[i];
    dst[i + half] = src[i + half];
  }
}

// fp8-KV variant: same rope + append, but the caches hold OCP e4m3
// bytes with one f32 scale per row (token): scale = row amax / 448.
// Halves KV-cache HBM traffic and memory; decode attention dequants
// while staging through LDS.
__global__ void rope_kv_fused_q8_kernel(
    unsigned short* __restrict__ qkv, unsigned char* __restrict__ Kc8,
    unsigned char* __restrict__ Vc8, float* __restrict__ kscale,
    float* __restrict__ vscale, const int* __restrict__ positions,
    const float* __restrict__ cos_sin, int Hq, int Hkv, int Smax, int D,
    long long row_stride) {
  const int b = blockIdx.x;
  const int y = blockIdx.y;
  const int i = threadIdx.x;  // 0..D/2-1 (one wave when D=128)
  const int half = D >> 1;
  const int pos = positions[b];
  unsigned short* row = qkv + (size_t)b * row_stride;
  float lo_f = 0.f, hi_f = 0.f;
  if (y < Hq + Hkv) {
    unsigned short* head = row + (size_t)y * D;
    const float c = cos_sin[((size_t)pos * half + i) * 2];
    const float sn = cos_sin[((size_t)pos * half + i) * 2 + 1];
    const float a = bf2f(head[i]);
    const float bvf = bf2f(head[i + half]);
    lo_f = a * c - bvf * sn;
    hi_f = bvf * c + a * sn;
    head[i] = f2bf(lo_f);
    head[i + half] = f2bf(hi_f);
    if (y < Hq) return;
  } else {
    const unsigned short* src = row + (size_t)y * D;
    lo_f = bf2f(src[i]);
    hi_f = bf2f(src[i + half]);
  }
  // wave-reduce the row amax (D/2 lanes x 2 values)
  float amax = fmaxf(fabsf(lo_f), fabsf(hi_f));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    amax = fmaxf(amax, __shfl_xor(amax, off, 64));
  const float qs = (amax > 0.f) ? amax / 448.f : 1.f;
  const float qinv = 1.f / qs;
  const bool is_k = (y < Hq + Hkv);
  const int kvh = is_k ? (y - Hq) : (y - Hq - Hkv);
  const size_t rowi = ((size_t)b * Hkv + kvh) * Smax + pos;
  unsigned char* dst = (is_k ? Kc8 : Vc8) + rowi * D;
  __hip_fp8_e4m3 lo8(lo_f * qinv), hi8(hi_f * qinv);
  dst[i] = lo8.__x;
  dst[i + half] = hi8.__x;
  if (i == 0) (is_k ? kscale : vscale)[rowi] = qs;
}

void launch_rope_kv_fused(void* qkv, void* Kc, void* Vc,
                          const void* positions, const void* cos_sin, int B,
                          int Hq, int Hkv, int Smax, int D,
                          long long row_stride, void* stream) {
  hipLaunchKernelGGL(rope_kv_fused_kernel, dim3(B, Hq + 2 * Hkv),
                     dim3(D / 2), 0, (hipStream_t)stream,
                     (unsigned short*)qkv, (unsigned short*)Kc,
                     (unsigned short*)Vc, (const int*)positions,
                     (const float*)cos_sin, Hq, Hkv, Smax, D, row_stride);
}

void launch_rope_kv_fused_q8(void* qkv, void* Kc8, void* Vc8,
                             void* kscale, void* vscale,
                             const void* positions, const void* cos_sin,
                             int B, int Hq, int Hkv, int Smax, int D,
                             long long row_stride, void* stream) {
  hipLaunchKernelGGL(rope_kv_fused_q8_kernel, dim3(B, Hq + 2 * Hkv),
                     dim3(D / 2), 0, (hipStream_t)stream,
                     (unsigned short*)qkv, (unsigned char*)Kc8,
                     (unsigned char*)Vc8, (float*)kscale, (float*)vscale,
                     (const int*)positions, (const float*)cos_sin, Hq,
                     Hkv, Smax, D, row_stride);
}

// ---------------------------------------------------------------------
// Decode attention (single new token per sequence, GQA):
//   O[b,h,:] = softmax(Q[b,h,:] K[b,kvh,:len,:]^T * scale) V[b,kvh,:len,:]
// One workgroup per (b, kv-head); one wave per grouped q-head.
//
// K/V chunks are staged through LDS by ALL waves cooperatively: the G
// grouped q-heads attend to the SAME kv rows, so per-wave global reads
// would issue the identical K and V bytes G times (L2 absorbs the
// reuse but the kernel becomes issue/latency-bound — measured 2.5
// TB/s apparent).  One cooperative load + G waves reading LDS cuts
// vector-memory traffic G-fold.  dim must be 128 (llama head_dim).
// ---------------------------------------------------------------------
#define ATTN_SCHUNK 64  // default rows staged per LDS buffer
#define ATTN_MAXG 8

struct AttnState {
  float m, l, acc0, acc1;
};

// shared chunk loop: stage [s_begin, s_final) through kbuf/vbuf and
// accumulate the online-softmax state for this wave's head
struct uchar16v { unsigned char v[16]; };

#define ATTN_LDS_PAD 8  // halves; rows start on different banks
template <int SCHUNK, bool KVQ>
__device__ __forceinline__ AttnState attn_chunk_loop(
    const void* __restrict__ kbase_, const void* __restrict__ vbase_,
    const float* __restrict__ kscale, const float* __restrict__ vscale,
    const float* qf, float scale, int s_begin, int s_final,
    int wave, int lane, int tid, int nthreads,
    unsigned short (*kbuf)[128 + ATTN_LDS_PAD],
    unsigned short (*vbuf)[128 + ATTN_LDS_PAD],
    float (*scores)[SCHUNK]) {
  constexpr int D = 128;
  AttnState st = {-FLT_MAX, 0.f, 0.f, 0.f};
  const int d0 = lane * 2;
  for (int s0 = s_begin; s0 < s_final; s0 += SCHUNK) {
    const int cnt = min(SCHUNK, s_final - s0);
    if (!KVQ) {
      // --- cooperative stage: every thread loads 16B K/V vectors
      const unsigned short* kbase = (const unsigned short*)kbase_;
      const unsigned short* vbase = (const unsigned short*)vbase_;
      const int total_vec = cnt * (D / 8);
      for (int i = tid; i < total_vec; i += nthreads) {
        const int row = i >> 4;
        const int col = (i & 15) * 8;
        const size_t off = (size_t)(s0 + row) * D + col;
        *reinterpret_cast<ushort8*>(&kbuf[row][col]) =
            *reinterpret_cast<const ushort8*>(kbase + off);
        *reinterpret_cast<ushort8*>(&vbuf[row][col]) =
            *reinterpret_cast<const ushort8*>(vbase + off);
      }
    } else {
      // --- fp8 cache: 16B loads carry 16 dims; dequant into LDS bf16
      const unsigned char* k8 = (const unsigned char*)kbase_;
      const unsigned char* v8 = (const unsigned char*)vbase_;
      const int total_vec = cnt * (D / 16);
      for (int i = tid; i < total_vec; i += nthreads) {
        const int row = i >> 3;
        const int col = (i & 7) * 16;
        const size_t off = (size_t)(s0 + row) * D + col;
        const uchar16v kq = *reinterpret_cast<const uchar16v*>(k8 + off);
        const uchar16v vq = *reinterpret_cast<const uchar16v*>(v8 + off);
        const float ks = kscale[s0 + row];
        const float vs = vscale[s0 + row];
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          __hip_fp8_e4m3 kf8, vf8;
          kf8.__x = kq.v[j];
          vf8.__x = vq.v[j];
          kbuf[row][col + j] = f2bf(float(kf8) * ks);
          vbuf[row][col + j] = f2bf(float(vf8) * vs);
        }
      }
    }
    __syncthreads();
    // --- scores from LDS (16-lane groups each cover all 128 dims)
    for (int si = lane >> 4; si < cnt; si += 4) {
      const unsigned short* kp = &kbuf[si][(lane & 15) * 8];
      ushort8 kv = *reinterpret_cast<const ushort8*>(kp);
      float dot = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) dot += qf[j] * bf2f(kv.v[j]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        dot += __shfl_xor(dot, off, 64);
      if ((lane & 15) == 0) scores[wave][si] = dot * scale;
    }
    __syncthreads();
    // --- chunk max (wave-wide) ---
    float cm = -FLT_MAX;
    for (int si = lane; si < cnt; si += 64)
      cm = fmaxf(cm, scores[wave][si]);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      cm = fmaxf(cm, __shfl_xor(cm, off, 64));
    const float m_new = fmaxf(st.m, cm);
    const float rescale = (st.m == -FLT_MAX) ? 0.f : __expf(st.m - m_new);
    st.acc0 *= rescale;
    st.acc1 *= rescale;
    st.l *= rescale;
    // --- p * V from LDS (lane owns dims d0, d0+1) ---
    for (int si = 0; si < cnt; ++si) {
      const float p = __expf(scores[wave][si] - m_new);
      st.l += p;
      const ushort2v vv =
          *reinterpret_cast<const ushort2v*>(&vbuf[si][d0]);
      st.acc0 += p * bf2f(vv.x);
      st.acc1 += p * bf2f(vv.y);
    }
    st.m = m_new;
    __syncthreads();
  }
  return st;
}

template <int SCHUNK, bool KVQ = false>
__global__ void attn_decode_kernel(
    unsigned short* __restrict__ O, const unsigned short* __restrict__ Q,
    const void* __restrict__ Kc, const void* __restrict__ Vc,
    const float* __restrict__ kscale, const float* __restrict__ vscale,
    const int* __restrict__ seq_lens, int B, int Hq, int Hkv, int Smax,
    float scale, long long q_row_stride) {
  constexpr int D = 128;
  const int b = blockIdx.x / Hkv;
  const int kvh = blockIdx.x % Hkv;
  const int G = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int h = kvh * G + wave;
  const int len = seq_lens[b];

  __shared__ float scores[ATTN_MAXG][SCHUNK];
  __shared__ __align__(16) unsigned short kbuf[SCHUNK][D + ATTN_LDS_PAD];
  __shared__ __align__(16) unsigned short vbuf[SCHUNK][D + ATTN_LDS_PAD];

  // q fragment: lane holds 8 consecutive dims at (lane&15)*8 for the
  // K-dot phase (16-lane groups each cover all 128 dims)
  const unsigned short* qp = Q + (size_t)b * q_row_stride +
                             (size_t)h * D + (lane & 15) * 8;
  float qf[8];
  {
    ushort8 qv = *reinterpret_cast<const ushort8*>(qp);
#pragma unroll
    for (int j = 0; j < 8; ++j) qf[j] = bf2f(qv.v[j]);
  }

  const size_t kv_row = ((size_t)b * Hkv + kvh) * Smax;
  const AttnState st = attn_chunk_loop<SCHUNK, KVQ>(
      (const char*)Kc + kv_row * D * (KVQ ? 1 : 2),
      (const char*)Vc + kv_row * D * (KVQ ? 1 : 2),
      kscale ? kscale + kv_row : nullptr,
      vscale ? vscale + kv_row : nullptr, qf, scale, 0, len, wave, lane,
      threadIdx.x, G * 64, kbuf, vbuf, scores);

  const float inv = (st.l > 0.f) ? 1.f / st.l : 0.f;
  unsigned short* op = O + ((size_t)b * Hq + h) * D + lane * 2;
  op[0] = f2bf(st.acc0 * inv);
  op[1] = f2bf(st.acc1 * inv);
}

// ---------------------------------------------------------------------
// Flash-decode split-S: nsplit workgroups per (b, kv-head) each cover
// a slice of the sequence and write an unnormalized partial
// (acc[D], m, l) to the workspace; a combine kernel folds the slices.
// Fills the chip even at small B*Hkv (the single-workgroup variant
// peaked at 128 workgroups on 256 CUs).
// partial layout: [B, Hq, nsplit, D+2] f32
// ---------------------------------------------------------------------
template <int SCHUNK, bool KVQ = false>
__global__ void attn_decode_split_kernel(
    float* __restrict__ partial, const unsigned short* __restrict__ Q,
    const void* __restrict__ Kc, const void* __restrict__ Vc,
    const float* __restrict__ kscale, const float* __restrict__ vscale,
    const int* __restrict__ seq_lens, int B, int Hq, int Hkv, int Smax,
    float scale, long long q_row_stride, int nsplit) {
  constexpr int D = 128;
  const int b = blockIdx.x / Hkv;
  const int kvh = blockIdx.x % Hkv;
  const int split = blockIdx.y;
  const int G = Hq / Hkv;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int h = kvh * G + wave;
  const int len = seq_lens[b];
  const int per = (len + nsplit - 1) / nsplit;
  const int s_begin = split * per;
  const int s_final = min(len, s_begin + per);
  float* prow = partial + (((size_t)b * Hq + h) * nsplit + split) * (D + 2);

  __shared__ float scores[ATTN_MAXG][SCHUNK];
  __shared__ __align__(16) unsigned short kbuf[SCHUNK][D + ATTN_LDS_PAD];
  __shared__ __align__(16) unsigned short vbuf[SCHUNK][D + ATTN_LDS_PAD];

  const unsigned short* qp = Q + (size_t)b * q_row_stride +
                             (size_t)h * D + (lane & 15) * 8;
  float qf[8];
  {
    ushort8 qv = *reinterpret_cast<const ushort8*>(qp);
#pragma unroll
    for (int j = 0; j < 8; ++j) qf[j] = bf2f(qv.v[j]);
  }
  const size_t kv_row = ((size_t)b * Hkv + kvh) * Smax;
  const AttnState st = attn_chunk_loop<SCHUNK, KVQ>(
      (const char*)Kc + kv_row * D * (KVQ ? 1 : 2),
      (const char*)Vc + kv_row * D * (KVQ ? 1 : 2),
      kscale ? kscale + kv_row : nullptr,
      vscale ? vscale + kv_row : nullptr, qf, scale, s_begin, s_final,
      wave, lane, threadIdx.x, G * 64, kbuf, vbuf, scores);

  prow[lane * 2] = st.acc0;
  prow[lane * 2 + 1] = st.acc1;
  if (lane == 0) {
    prow[D] = st.m;
    prow[D + 1] = st.l;
  }
}

// fold the nsplit partials: one wave per (b, h)
__global__ void attn_decode_combine_kernel(
    unsigned short* __restrict__ O, const float* __restrict__ partial,
    int Hq, int nsplit) {
  constexpr int D = 128;
  const int bh = blockIdx.x;
  const int lane = threadIdx.x;
  const float* base = partial + (size_t)bh * nsplit * (D + 2);
  float m_star = -FLT_MAX;
  for (int s = 0; s < nsplit; ++s)
    m_star = fmaxf(m_star, base[s * (D + 2) + D]);
  float l_total = 0.f, a0 = 0.f, a1 = 0.f;
  for (int s = 0; s < nsplit; ++s) {
    const float* prow = base + s * (D + 2);
    const float ms = prow[D];
    if (ms == -FLT_MAX) continue;
    const float wgt = __expf(ms - m_star);
    l_total += wgt * prow[D + 1];
    a0 += wgt * prow[lane * 2];
    a1 += wgt * prow[lane * 2 + 1];
  }
  const float inv = (l_total > 0.f) ? 1.f / l_total : 0.f;
  unsigned short* op = O + (size_t)bh * D + lane * 2;
  op[0] = f2bf(a0 * inv);
  op[1] = f2bf(a1 * inv);
}

// staged-chunk rows: 32 halves LDS/WG (16.5 KB -> ~8 WGs/CU = 8
// waves/SIMD) vs 64 (33 KB -> 4 waves/SIMD); selectable for sweeps
// via MLRUN_ATTN_SCHUNK, default chosen by measurement.
static int attn_schunk_env() {
  static int v = [] {
    const char* e = getenv("MLRUN_ATTN_SCHUNK");
    return e ? atoi(e) : 32;
  }();
  return v;
}

void launch_attn_decode(void* O, const void* Q, const void* Kc,
                        const void* Vc, const void* kscale,
                        const void* vscale, const void* seq_lens, int B,
                        int Hq, int Hkv, int Smax, float scale,
                        long long q_row_stride, float* partial_ws,
                        int nsplit, void* stream) {
  const int G = Hq / Hkv;
  const bool c64 = attn_schunk_env() >= 64;
  const bool kvq = kscale != nullptr;
  if (nsplit <= 1 || partial_ws == nullptr) {
    auto* kern = kvq ? (c64 ? attn_decode_kernel<64, true>
                            : attn_decode_kernel<32, true>)
                     : (c64 ? attn_decode_kernel<64, false>
                            : attn_decode_kernel<32, false>);
    hipLaunchKernelGGL(kern, dim3(B * Hkv), dim3(G * 64), 0,
                       (hipStream_t)stream, (unsigned short*)O,
                       (const unsigned short*)Q, Kc, Vc,
                       (const float*)kscale, (const float*)vscale,
                       (const int*)seq_lens, B,
                       Hq, Hkv, Smax, scale, q_row_stride);
    return;
  }
  auto* kern = kvq ? (c64 ? attn_decode_split_kernel<64, true>
                          : attn_decode_split_kernel<32, true>)
                   : (c64 ? attn_decode_split_kernel<64, false>
                          : attn_decode_split_kernel<32, false>);
  hipLaunchKernelGGL(kern, dim3(B * Hkv, nsplit),
                     dim3(G * 64), 0, (hipStream_t)stream, partial_ws,
                     (const unsigned short*)Q, Kc, Vc,
                     (const float*)kscale, (const float*)vscale,
                     (const int*)seq_lens, B, Hq,
                     Hkv, Smax, scale, q_row_stride, nsplit);
  hipLaunchKernelGGL(attn_decode_combine_kernel, dim3(B * Hq), dim3(64), 0,
                     (hipStream_t)stream, (unsigned short*)O, partial_ws,
                     Hq, nsplit);
}

// ---------------------------------------------------------------------
// KV-cache append: scatter the new token's K/V into the cache at
// position seq_lens[b] (pre-increment positions computed host-side).
// knew/vnew: [B, Hkv, D]; cache: [B, Hkv, Smax, D]
// ---------------------------------------------------------------------
__global__ void kv_append_kernel(unsigned short* __restrict__ Kc,
                                 unsigned short* __restrict__ Vc,
                                 const unsigned short* __restrict__ knew,
                                 const unsigned short* __restrict__ vnew,
                                 const int* __restrict__ positions, int Hkv,
                                 int Smax, int D, long long src_row_stride) {
  const int b = blockIdx.x;
  const int kvh = blockIdx.y;
  const int pos = positions[b];
  const size_t src = (size_t)b * src_row_stride + (size_t)kvh * D;
  const size_t dst = (((size_t)b * Hkv + kvh) * Smax + pos) * D;
  for (int i = threadIdx.x * 8; i < D; i += blockDim.x * 8) {
    *reinterpret_cast<ushort8*>(Kc + dst + i) =
        *reinterpret_cast<const ushort8*>(knew + src + i);
    *reinterpret_cast<ushort8*>(Vc + dst + i) =
        *reinterpret_cast<const ushort8*>(vnew + src + i);
  }
}

void launch_kv_append(void* Kc, void* Vc, const void* knew, const void* vnew,
                      const void* positions, int B, int Hkv, int Smax, int D,
                      long long src_row_stride, void* stream) {
  hipLaunchKernelGGL(kv_append_kernel, dim3(B, Hkv), dim3(D / 8), 0,
                     (hipStream_t)stream, (unsigned short*)Kc,
                     (unsigned short*)Vc, (const unsigned short*)knew,
                     (const unsigned short*)vnew, (const int*)positions, Hkv,
                     Smax, D, src_row_stride);
}

// ---------------------------------------------------------------------
// Row softmax (bf16 in/out, online single pass over LDS-staged row,
// used by classic-model servers & tests)
// ---------------------------------------------------------------------
__global__ void softmax_kernel(unsigned short* __restrict__ out,
                               const unsigned short* __restrict__ in,
                               int cols) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const unsigned short* inr = in + (size_t)row * cols;
  unsigned short* outr = out + (size_t)row * cols;

  float lmax = -FLT_MAX;
  for (int i = tid; i < cols; i += blockDim.x)
    lmax = fmaxf(lmax, bf2f(inr[i]));
  __shared__ float red[64];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    lmax = fmaxf(lmax, __shfl_xor(lmax, off, 64));
  if ((tid & 63) == 0) red[tid >> 6] = lmax;
  __syncthreads();
  float gmax = -FLT_MAX;
  for (int w = 0; w < (blockDim.x >> 6); ++w) gmax = fmaxf(gmax, red[w]);

  float lsum = 0.f;
  for (int i = tid; i < cols; i += blockDim.x)
    lsum += __expf(bf2f(inr[i]) - gmax);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    lsum += __shfl_xor(lsum, off, 64);
  if ((tid & 63) == 0) red[tid >> 6] = lsum;
  __syncthreads();
  float gsum = 0.f;
  for (int w = 0; w < (blockDim.x >> 6); ++w) gsum += red[w];
  const float inv = 1.f / gsum;
  for (int i = tid; i < cols; i += blockDim.x)
    outr[i] = f2bf(__expf(bf2f(inr[i]) - gmax) * inv);
}

void launch_softmax(void* out, const void* in, int rows, int cols,
                    void* stream) {
  hipLaunchKernelGGL(softmax_kernel, dim3(rows), dim3(256), 0,
                     (hipStream_t)stream, (unsigned short*)out,
                     (const unsigned short*)in, cols);
}

// ---------------------------------------------------------------------
// Tree-ensemble inference (XGBoost/GBDT-style), for the classic-model
// serving config.  Trees in flattened SoA node arrays; each lane
// evaluates one sample through all trees (depth-bound loop).
// features: [n_samples, n_features] f32; out: [n_samples] f32
// ---------------------------------------------------------------------
__global__ void tree_ensemble_kernel(
    float* __restrict__ out, const float* __restrict__ features,
    const int* __restrict__ feature_idx,    // per node: feature (-1 = leaf)
    const float* __restrict__ threshold,    // per node: split threshold
    const int* __restrict__ left,           // per node: left child index
    const int* __restrict__ right,          // per node: right child index
    const float* __restrict__ leaf_value,   // per node: value if leaf
    const int* __restrict__ tree_offsets,   // [n_trees+1] node offsets
    int n_trees, int n_samples, int n_features, float base_score) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n_samples; i += stride) {
    const float* frow = features + i * n_features;
    float score = base_score;
    for (int t = 0; t < n_trees; ++t) {
      int node = tree_offsets[t];
      int fidx = feature_idx[node];
      while (fidx >= 0) {
        node = (frow[fidx] < threshold[node]) ? left[node] : right[node];
        fidx = feature_idx[node];
      }
      score += leaf_value[node];
    }
    out[i] = score;
  }
}

void launch_tree_ensemble(void* out, const void* features,
                          const void* feature_idx, const void* threshold,
                          const void* left, const void* right,
                          const void* leaf_value, const void* tree_offsets,
                          int n_trees, int n_samples, int n_features,
                          float base_score, void* stream) {
  long long blocks = (n_samples + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(tree_ensemble_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (float*)out,
                     (const float*)features, (const int*)feature_idx,
                     (const float*)threshold, (const int*)left,
                     (const int*)right, (const float*)leaf_value,
                     (const int*)tree_offsets, n_trees, n_samples, n_features,
                     base_score);
}

// ---------------------------------------------------------------------
// Feature-store sliding-window aggregation:
// For each (key, feature): ring buffer of per-period partial aggregates
// resident in HBM; this kernel folds a batch of events into the ring
// and emits current window aggregates (sum/count/min/max -> avg
// derived host-side).
// events: keys[n] int32 (dense key ids), values[n] f32, periods[n] int32
// ring: [n_keys, n_feats?, n_periods, 4] (sum, count, min, max) f32
// Batched: one lane per event; atomics within the period cell.
// ---------------------------------------------------------------------
__global__ void window_ingest_kernel(float* __restrict__ ring,
                                     const int* __restrict__ keys,
                                     const float* __restrict__ values,
                                     const int* __restrict__ period_idx,
                                     long long n_events, int n_periods) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n_events; i += stride) {
    const int key = keys[i];
    const int p = period_idx[i] % n_periods;
    float* cell = ring + ((size_t)key * n_periods + p) * 4;
    const float v = values[i];
    atomicAdd(cell + 0, v);             // sum
    atomicAdd(cell + 1, 1.f);           // count
    // min/max are tracked per-batch host-side (they are not
    // decomposable over ring periods for sliding windows anyway)
  }
}

// window reduce: sum the last `window_periods` period cells per key
// ring: [n_keys, n_periods, 4]; out: [n_keys, 4]
__global__ void window_reduce_kernel(float* __restrict__ out,
                                     const float* __restrict__ ring,
                                     int n_keys, int n_periods,
                                     int window_periods, int current_period) {
  int key = blockIdx.x * blockDim.x + threadIdx.x;
  if (key >= n_keys) return;
  float sum = 0.f, count = 0.f;
  for (int w = 0; w < window_periods; ++w) {
    int p = (current_period - w) % n_periods;
    if (p < 0) p += n_periods;
    const float* cell = ring + ((size_t)key * n_periods + p) * 4;
    sum += cell[0];
    count += cell[1];
  }
  out[key * 4 + 0] = sum;
  out[key * 4 + 1] = count;
  out[key * 4 + 2] = count > 0.f ? sum / count : 0.f;  // avg
  out[key * 4 + 3] = 0.f;
}

// ---------------------------------------------------------------------
// Per-period min/max cells: float atomic min/max via the ordered-int
// transform (monotone map f32 -> u32 so unsigned atomics give float
// order).  Per-period cells make min/max DECOMPOSABLE for the
// bucket-quantized sliding windows served by window_reduce — unlike
// the running aggregates the round-1 path fell back to.
// ring_mm: [n_keys, n_periods, 2] u32 (ordered-min, ordered-max),
// empty cells hold 0xFFFFFFFF / 0x00000000.
// ---------------------------------------------------------------------
__device__ __forceinline__ unsigned int f32_to_ordered(float v) {
  unsigned int bits = __float_as_uint(v);
  return (bits & 0x80000000u) ? ~bits : (bits | 0x80000000u);
}

__device__ __forceinline__ float ordered_to_f32(unsigned int key) {
  unsigned int bits = (key & 0x80000000u) ? (key & 0x7FFFFFFFu) : ~key;
  return __uint_as_float(bits);
}

__global__ void window_ingest_mm_kernel(unsigned int* __restrict__ ring_mm,
                                        const int* __restrict__ keys,
                                        const float* __restrict__ values,
                                        const int* __restrict__ period_idx,
                                        long long n_events, int n_periods) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n_events; i += stride) {
    const int key = keys[i];
    const int p = period_idx[i] % n_periods;
    unsigned int* cell = ring_mm + ((size_t)key * n_periods + p) * 2;
    const unsigned int ov = f32_to_ordered(values[i]);
    atomicMin(cell + 0, ov);
    atomicMax(cell + 1, ov);
  }
}

// first/last per period: packed u64 = (ts << 32) | ordered(value);
// atomicMin gives the earliest event (ties broken by value order),
// atomicMax the latest.  ts fits 31 bits until 2038 (unix seconds).
// ring_fl: [n_keys, n_periods, 2] u64, empty = (ULLONG_MAX, 0).
__global__ void window_ingest_fl_kernel(
    unsigned long long* __restrict__ ring_fl,
    const int* __restrict__ keys, const float* __restrict__ values,
    const int* __restrict__ timestamps, const int* __restrict__ period_idx,
    long long n_events, int n_periods) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n_events; i += stride) {
    const int key = keys[i];
    const int p = period_idx[i] % n_periods;
    unsigned long long* cell = ring_fl + ((size_t)key * n_periods + p) * 2;
    const unsigned long long pack =
        ((unsigned long long)(unsigned int)timestamps[i] << 32) |
        f32_to_ordered(values[i]);
    atomicMin(cell + 0, pack);  // first = earliest pack
    atomicMax(cell + 1, pack);  // last = latest pack
  }
}

// reduce min/max + first/last over the covered window cells.
// out_mmfl: [n_keys, 4] f32 (min, max, first, last); count==0 rows are
// left as written (caller masks by count from window_reduce).
__global__ void window_reduce_mmfl_kernel(
    float* __restrict__ out, const unsigned int* __restrict__ ring_mm,
    const unsigned long long* __restrict__ ring_fl, int n_keys,
    int n_periods, int window_periods, int current_period) {
  int key = blockIdx.x * blockDim.x + threadIdx.x;
  if (key >= n_keys) return;
  unsigned int omin = 0xFFFFFFFFu, omax = 0u;
  unsigned long long first_pack = 0xFFFFFFFFFFFFFFFFull, last_pack = 0ull;
  for (int w = 0; w < window_periods; ++w) {
    int p = (current_period - w) % n_periods;
    if (p < 0) p += n_periods;
    const unsigned int* mm = ring_mm + ((size_t)key * n_periods + p) * 2;
    omin = min(omin, mm[0]);
    omax = max(omax, mm[1]);
    const unsigned long long* fl =
        ring_fl + ((size_t)key * n_periods + p) * 2;
    first_pack = min(first_pack, fl[0]);
    last_pack = max(last_pack, fl[1]);
  }
  float* row = out + (size_t)key * 4;
  row[0] = omin == 0xFFFFFFFFu ? 0.f : ordered_to_f32(omin);
  row[1] = omax == 0u ? 0.f : ordered_to_f32(omax);
  row[2] = first_pack == 0xFFFFFFFFFFFFFFFFull
               ? 0.f
               : ordered_to_f32((unsigned int)(first_pack & 0xFFFFFFFFull));
  row[3] = last_pack == 0ull
               ? 0.f
               : ordered_to_f32((unsigned int)(last_pack & 0xFFFFFFFFull));
}

// f64 variants for the sum-of-squares ring: the stdvar formula
// sumsq/n - mean^2 cancels catastrophically in f32 when values are
// large and var is small — f64 accumulation keeps ~16 digits
// (atomicAdd(double) is native on CDNA4 HBM).
__global__ void window_ingest64_kernel(double* __restrict__ ring,
                                       const int* __restrict__ keys,
                                       const float* __restrict__ values,
                                       const int* __restrict__ period_idx,
                                       long long n_events, int n_periods) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n_events; i += stride) {
    const int key = keys[i];
    const int p = period_idx[i] % n_periods;
    double* cell = ring + ((size_t)key * n_periods + p) * 4;
    const double v = (double)values[i];
    atomicAdd(cell + 0, v * v);  // sum of squares
    atomicAdd(cell + 1, 1.0);    // count
    atomicAdd(cell + 2, v);      // f64 sum (variance needs f64 mean)
  }
}

__global__ void window_reduce64_kernel(double* __restrict__ out,
                                       const double* __restrict__ ring,
                                       int n_keys, int n_periods,
                                       int window_periods,
                                       int current_period) {
  int key = blockIdx.x * blockDim.x + threadIdx.x;
  if (key >= n_keys) return;
  double sumsq = 0.0, count = 0.0, sum = 0.0;
  for (int w = 0; w < window_periods; ++w) {
    int p = (current_period - w) % n_periods;
    if (p < 0) p += n_periods;
    const double* cell = ring + ((size_t)key * n_periods + p) * 4;
    sumsq += cell[0];
    count += cell[1];
    sum += cell[2];
  }
  out[key * 4 + 0] = sumsq;
  out[key * 4 + 1] = count;
  out[key * 4 + 2] = sum;
  out[key * 4 + 3] = 0.0;
}

void launch_window_ingest64(void* ring, const void* keys,
                            const void* values, const void* period_idx,
                            long long n_events, int n_periods,
                            void* stream) {
  long long blocks = (n_events + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(window_ingest64_kernel, dim3((int)blocks), dim3(256),
                     0, (hipStream_t)stream, (double*)ring,
                     (const int*)keys, (const float*)values,
                     (const int*)period_idx, n_events, n_periods);
}

void launch_window_reduce64(void* out, const void* ring, int n_keys,
                            int n_periods, int window_periods,
                            int current_period, void* stream) {
  int blocks = (n_keys + 255) / 256;
  hipLaunchKernelGGL(window_reduce64_kernel, dim3(blocks), dim3(256), 0,
                     (hipStream_t)stream, (double*)out,
                     (const double*)ring, n_keys, n_periods,
                     window_periods, current_period);
}

void launch_window_ingest_mm(void* ring_mm, const void* keys,
                             const void* values, const void* period_idx,
                             long long n_events, int n_periods,
                             void* stream) {
  long long blocks = (n_events + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(window_ingest_mm_kernel, dim3((int)blocks), dim3(256),
                     0, (hipStream_t)stream, (unsigned int*)ring_mm,
                     (const int*)keys, (const float*)values,
                     (const int*)period_idx, n_events, n_periods);
}

void launch_window_ingest_fl(void* ring_fl, const void* keys,
                             const void* values, const void* timestamps,
                             const void* period_idx, long long n_events,
                             int n_periods, void* stream) {
  long long blocks = (n_events + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(window_ingest_fl_kernel, dim3((int)blocks), dim3(256),
                     0, (hipStream_t)stream,
                     (unsigned long long*)ring_fl, (const int*)keys,
                     (const float*)values, (const int*)timestamps,
                     (const int*)period_idx, n_events, n_periods);
}

void launch_window_reduce_mmfl(void* out, const void* ring_mm,
                               const void* ring_fl, int n_keys,
                               int n_periods, int window_periods,
                               int current_period, void* stream) {
  int blocks = (n_keys + 255) / 256;
  hipLaunchKernelGGL(window_reduce_mmfl_kernel, dim3(blocks), dim3(256), 0,
                     (hipStream_t)stream, (float*)out,
                     (const unsigned int*)ring_mm,
                     (const unsigned long long*)ring_fl, n_keys, n_periods,
                     window_periods, current_period);
}

void launch_window_ingest(void* ring, const void* keys, const void* values,
                          const void* period_idx, long long n_events,
                          int n_periods, void* stream) {
  long long blocks = (n_events + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(window_ingest_kernel, dim3((int)blocks), dim3(256), 0,
                     (hipStream_t)stream, (float*)ring, (const int*)keys,
                     (const float*)values, (const int*)period_idx, n_events,
                     n_periods);
}

void launch_window_reduce(void* out, const void* ring, int n_keys,
                          int n_periods, int window_periods,
                          int current_period, void* stream) {
  int blocks = (n_keys + 255) / 256;
  hipLaunchKernelGGL(window_reduce_kernel, dim3(blocks), dim3(256), 0,
                     (hipStream_t)stream, (float*)out, (const float*)ring,
                     n_keys, n_periods, window_periods, current_period);
}
