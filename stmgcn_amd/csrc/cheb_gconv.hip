// Chebyshev graph-convolution support recurrence on CSR — CDNA4 (gfx950).
//
// Replaces the reference's dense einsum('ij,bjp->bip') per materialized
// support (GCN.py:34-36): the (K_s, N, N) T_k stacks are never built; only
// the sparse generator G is stored and the first-kind recurrence
// T_k x = 2 G (T_{k-1} x) - T_{k-2} x runs on-device.
//
// spmm_step computes one recurrence step as a fused SpMM + axpby:
//     out = alpha * (G @ xin) + beta * p1 + gamma * p2
// over (B, N, C) slices embedded in a (B, N, K, C) stack (per-tensor row
// strides). Thread mapping: 256-thread blocks; each thread owns one (row,
// channel) pair, so a row's C channels sit in consecutive lanes and every
// gather xin[col, c] is a coalesced C-wide segment. Host-side sequencing of
// the K steps lives in bindings.cpp (cheb_apply / cheb_combine via Clenshaw).

#include <type_traits>

#include "common.h"

namespace {

template <typename T>
__global__ void spmm_step_kernel(
    const int* __restrict__ rowptr, const int* __restrict__ colidx,
    const float* __restrict__ vals,
    const T* __restrict__ xin, const T* __restrict__ p1, const T* __restrict__ p2,
    T* __restrict__ out,
    int N, int C, int rows_per_block,
    long sx, long s1, long s2, long so,          // element stride between rows
    long bx, long b1, long b2, long bo,          // element stride between batches
    float alpha, float beta, float gamma) {
  const int b = blockIdx.y;
  const int local = (C <= 256) ? (threadIdx.x / C) : 0;
  const int row = blockIdx.x * rows_per_block + local;
  if (row >= N || (C <= 256 && threadIdx.x >= rows_per_block * C)) return;

  const int c0 = (C <= 256) ? (threadIdx.x % C) : threadIdx.x;
  const int cstep = (C <= 256) ? C : 256;

  const T* xb = xin ? xin + (long)b * bx : nullptr;
  for (int c = c0; c < C; c += cstep) {
    float acc = 0.0f;
    if (alpha != 0.0f && xb) {
      const int s = rowptr[row], e = rowptr[row + 1];
      // 4-wide unroll: the col/val loads and the x gathers of a chunk issue
      // together instead of forming one serial load-latency chain per nnz
      int j = s;
      for (; j + 4 <= e; j += 4) {
        int cc[4];
        float vv[4];
        #pragma unroll
        for (int k = 0; k < 4; ++k) { cc[k] = colidx[j + k]; vv[k] = vals[j + k]; }
        float xv[4];
        #pragma unroll
        for (int k = 0; k < 4; ++k) xv[k] = toF<T>(xb[(long)cc[k] * sx + c]);
        #pragma unroll
        for (int k = 0; k < 4; ++k) acc = fmaf(vv[k], xv[k], acc);
      }
      for (; j < e; ++j)
        acc += vals[j] * toF<T>(xb[(long)colidx[j] * sx + c]);
      acc *= alpha;
    }
    const long ro = (long)b * bo + (long)row * so + c;
    if (p1) acc += beta * toF<T>(p1[(long)b * b1 + (long)row * s1 + c]);
    if (p2) acc += gamma * toF<T>(p2[(long)b * b2 + (long)row * s2 + c]);
    out[ro] = fromF<T>(acc);
  }
}

template <typename T>
void launch(hipStream_t stream, const int* rowptr, const int* colidx,
            const float* vals, const void* xin, const void* p1, const void* p2,
            void* out, int B, int N, int C,
            long sx, long s1, long s2, long so,
            long bx, long b1, long b2, long bo,
            float alpha, float beta, float gamma) {
  int rows_per_block = (C <= 256) ? (256 / C > 0 ? 256 / C : 1) : 1;
  dim3 grid((N + rows_per_block - 1) / rows_per_block, B);
  hipLaunchKernelGGL((spmm_step_kernel<T>), grid, dim3(256), 0, stream,
                     rowptr, colidx, vals,
                     (const T*)xin, (const T*)p1, (const T*)p2, (T*)out,
                     N, C, rows_per_block, sx, s1, s2, so, bx, b1, b2, bo,
                     alpha, beta, gamma);
}

}  // namespace

extern "C" void stmgcn_spmm_step(
    void* stream_v, int dtype, const int* rowptr, const int* colidx,
    const float* vals, const void* xin, const void* p1, const void* p2,
    void* out, int B, int N, int C,
    long sx, long s1, long s2, long so,
    long bx, long b1, long b2, long bo,
    float alpha, float beta, float gamma) {
  hipStream_t stream = (hipStream_t)stream_v;
  switch (dtype) {
    case STM_F32:
      launch<float>(stream, rowptr, colidx, vals, xin, p1, p2, out, B, N, C,
                    sx, s1, s2, so, bx, b1, b2, bo, alpha, beta, gamma);
      break;
    case STM_BF16:
      launch<__hip_bfloat16>(stream, rowptr, colidx, vals, xin, p1, p2, out, B, N, C,
                             sx, s1, s2, so, bx, b1, b2, bo, alpha, beta, gamma);
      break;
    case STM_F16:
      launch<__half>(stream, rowptr, colidx, vals, xin, p1, p2, out, B, N, C,
                     sx, s1, s2, so, bx, b1, b2, bo, alpha, beta, gamma);
      break;
  }
}

// ===========================================================================
// Fully fused ChebConv (SURVEY K1+K2): support recurrence + mix GEMM + bias
// + activation in-kernel. The reference materializes a dense (K,N,N) support
// stack offline and a (B,N,K_s,C) feature concat per forward
// (GCN.py:34-42); here NEITHER exists: each recurrence step k computes its
// (B,N,C) state p_k = alpha*(G @ p_gather) + beta*p_prev, stages the 32-row
// tile in LDS (lds_swz layout), and folds p_k @ W_k straight into an fp32
// accumulator of y with v_mfma_f32_16x16x32 matrix cores. The final step
// adds the bias and applies the activation. A scalar epilogue serves fp32
// parity and channel counts that are not MFMA-tileable. The backward
// (Clenshaw) variant fuses U_j = dz @ W_j^T into each combine step the same
// way, so the dgrad U stack is never materialized either.

#define CG_ST 32  // graph-node rows per workgroup tile

typedef __attribute__((ext_vector_type(8))) __bf16 cg_bf16x8;
typedef __attribute__((ext_vector_type(8))) _Float16 cg_f16x8;
typedef __attribute__((ext_vector_type(4))) float cg_f32x4;

template <typename T> struct CFrag8 { using type = float; using elem = float; };
template <> struct CFrag8<__hip_bfloat16> { using type = cg_bf16x8; using elem = __bf16; };
template <> struct CFrag8<__half> { using type = cg_f16x8; using elem = _Float16; };

__device__ __forceinline__ cg_f32x4 cg_mfma(cg_bf16x8 a, cg_bf16x8 b, cg_f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ cg_f32x4 cg_mfma(cg_f16x8 a, cg_f16x8 b, cg_f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
}

namespace {

// One fused forward step over (B, N, C) state tensors:
//   tile = alpha * (G @ xin)[rows] + beta * p1[rows]     (either term optional)
//   pout[rows] = tile                                    (optional)
//   mix (W != null): yacc[rows] (+)= tile @ W[kofs:kofs+C, :]
//                    (+ x0 != null: also += x0[rows] @ W[0:C, :] — the T_0
//                     term folded into the T_1 launch, one launch fewer)
//   final (yout != null): yout = act(yacc_total + bias)
// p1 is read only at the thread's own (row, c) element, so p1 == pout
// aliasing is safe (the gather operand xin must be a distinct buffer).
template <typename T, bool MF>
__global__ void __launch_bounds__(256)
cheb_fused_fwd_kernel(const int* __restrict__ rowptr,
                      const int* __restrict__ colidx,
                      const float* __restrict__ vals,
                      const T* __restrict__ xin, const T* __restrict__ p1,
                      const T* __restrict__ x0,
                      const T* __restrict__ W, const T* __restrict__ bias,
                      T* __restrict__ pout, float* __restrict__ yacc,
                      T* __restrict__ yout, int N, int C, int Cout, int kofs,
                      float alpha, float beta, int first, int act) {
  const int b = blockIdx.y;
  const int n0 = blockIdx.x * CG_ST;
  extern __shared__ char lds[];
  float* fls = (float*)lds;
  char* l0 = lds + (MF ? CG_ST * 128 : CG_ST * 64 * 4);  // x0 tile slot
  float* fl0 = (float*)l0;
  const T* xb = xin ? xin + (long)b * N * C : nullptr;
  const T* pb = p1 ? p1 + (long)b * N * C : nullptr;
  const T* x0b = x0 ? x0 + (long)b * N * C : nullptr;

  // ---- phase 1: recurrence (SpMM + axpby), tile staged in LDS ----
  for (int i = threadIdx.x; i < CG_ST * C; i += 256) {
    const int c = i % C, rl = i / C;
    const int row = n0 + rl;
    float acc = 0.f;
    if (x0b != nullptr) {
      const float v0 = (row < N) ? toF<T>(x0b[(long)row * C + c]) : 0.f;
      if (MF) *(T*)&l0[lds_swz(rl, c * 2)] = fromF<T>(v0);
      else fl0[rl * 64 + c] = v0;
    }
    if (row < N) {
      if (xb != nullptr && alpha != 0.f) {
        const int s = rowptr[row], e = rowptr[row + 1];
        int j = s;
        for (; j + 4 <= e; j += 4) {
          int cc[4];
          float vv[4];
          #pragma unroll
          for (int k = 0; k < 4; ++k) { cc[k] = colidx[j + k]; vv[k] = vals[j + k]; }
          float xv[4];
          #pragma unroll
          for (int k = 0; k < 4; ++k) xv[k] = toF<T>(xb[(long)cc[k] * C + c]);
          #pragma unroll
          for (int k = 0; k < 4; ++k) acc = fmaf(vv[k], xv[k], acc);
        }
        for (; j < e; ++j) acc += vals[j] * toF<T>(xb[(long)colidx[j] * C + c]);
        acc *= alpha;
      }
      if (pb != nullptr) acc = fmaf(beta, toF<T>(pb[(long)row * C + c]), acc);
      if (pout != nullptr) pout[((long)b * N + row) * C + c] = fromF<T>(acc);
    }
    if (W != nullptr) {
      if (MF) *(T*)&lds[lds_swz(rl, c * 2)] = fromF<T>(acc);
      else fls[rl * 64 + c] = acc;
    }
  }
  if (W == nullptr) return;
  __syncthreads();

  // ---- phase 2: mix epilogue y += tile @ W_k ----
  if constexpr (MF) {
    using frag = typename CFrag8<T>::type;
    using elem = typename CFrag8<T>::elem;
    const int wv = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int l16 = lane & 15, lgrp = lane >> 4;
    if (wv < (Cout >> 4)) {            // wave wv owns output cols [16wv,16wv+16)
      const int kchunks = C >> 5;
      cg_f32x4 acc[2];
      acc[0] = cg_f32x4{0.f, 0.f, 0.f, 0.f};
      acc[1] = cg_f32x4{0.f, 0.f, 0.f, 0.f};
      for (int kk = 0; kk < kchunks; ++kk) {
        frag bfr;
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          ((elem*)&bfr)[j] =
              *(const elem*)&W[(kofs + kk * 32 + lgrp * 8 + j) * Cout + 16 * wv + l16];
        frag a0 = *(const frag*)&lds[lds_swz(l16, (kk * 32 + lgrp * 8) * 2)];
        frag a1 = *(const frag*)&lds[lds_swz(16 + l16, (kk * 32 + lgrp * 8) * 2)];
        acc[0] = cg_mfma(a0, bfr, acc[0]);
        acc[1] = cg_mfma(a1, bfr, acc[1]);
      }
      if (x0 != nullptr) {               // + x0 @ W_0 (weight rows [0, C))
        for (int kk = 0; kk < kchunks; ++kk) {
          frag bfr;
          #pragma unroll
          for (int j = 0; j < 8; ++j)
            ((elem*)&bfr)[j] =
                *(const elem*)&W[(kk * 32 + lgrp * 8 + j) * Cout + 16 * wv + l16];
          frag a0 = *(const frag*)&l0[lds_swz(l16, (kk * 32 + lgrp * 8) * 2)];
          frag a1 = *(const frag*)&l0[lds_swz(16 + l16, (kk * 32 + lgrp * 8) * 2)];
          acc[0] = cg_mfma(a0, bfr, acc[0]);
          acc[1] = cg_mfma(a1, bfr, acc[1]);
        }
      }
      #pragma unroll
      for (int m = 0; m < 2; ++m)
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = n0 + 16 * m + 4 * lgrp + r;
          if (row < N) {
            const long idx = ((long)b * N + row) * Cout + 16 * wv + l16;
            float v = acc[m][r];
            if (!first) v += yacc[idx];
            if (yout != nullptr) {
              if (bias != nullptr) v += toF<T>(bias[16 * wv + l16]);
              if (act == 1 && v < 0.f) v = 0.f;
              yout[idx] = fromF<T>(v);
            } else {
              yacc[idx] = v;
            }
          }
        }
    }
  } else {
    for (int i = threadIdx.x; i < CG_ST * Cout; i += 256) {
      const int co = i % Cout, rl = i / Cout;
      const int row = n0 + rl;
      if (row >= N) continue;
      float v = 0.f;
      for (int c = 0; c < C; ++c)
        v = fmaf(fls[rl * 64 + c], toF<T>(W[(kofs + c) * Cout + co]), v);
      if (x0 != nullptr)
        for (int c = 0; c < C; ++c)
          v = fmaf(fl0[rl * 64 + c], toF<T>(W[c * Cout + co]), v);
      const long idx = ((long)b * N + row) * Cout + co;
      if (!first) v += yacc[idx];
      if (yout != nullptr) {
        if (bias != nullptr) v += toF<T>(bias[co]);
        if (act == 1 && v < 0.f) v = 0.f;
        yout[idx] = fromF<T>(v);
      } else {
        yacc[idx] = v;
      }
    }
  }
}

// One fused backward (Clenshaw) step over the G^T CSR:
//   out[rows] = alpha * (G^T @ xin)[rows] + beta * p1[rows] + (dz @ W_j^T)[rows]
// The U_j = dz W_j^T term is computed in-kernel from an LDS-staged dz tile —
// the dgrad U stack (B,N,K_s,C) of the unfused design never exists.
template <typename T, bool MF>
__global__ void __launch_bounds__(256)
cheb_fused_bwd_kernel(const int* __restrict__ rowptr,
                      const int* __restrict__ colidx,
                      const float* __restrict__ vals,
                      const T* __restrict__ xin, const T* __restrict__ p1,
                      const T* __restrict__ dz, const T* __restrict__ W,
                      T* __restrict__ out, int N, int C, int Cout, int kofs,
                      float alpha, float beta) {
  const int b = blockIdx.y;
  const int n0 = blockIdx.x * CG_ST;
  extern __shared__ char lds[];
  char* dzt = lds;
  float* fdz = (float*)lds;
  float* ut = (float*)(lds + (MF ? CG_ST * 128 : CG_ST * 64 * 4));

  // ---- phase A: stage the dz tile ----
  const T* dzb = dz + (long)b * N * Cout;
  for (int i = threadIdx.x; i < CG_ST * Cout; i += 256) {
    const int co = i % Cout, rl = i / Cout;
    const int row = n0 + rl;
    const float v = (row < N) ? toF<T>(dzb[(long)row * Cout + co]) : 0.f;
    if (MF) *(T*)&dzt[lds_swz(rl, co * 2)] = fromF<T>(v);
    else fdz[rl * 64 + co] = v;
  }
  __syncthreads();

  // ---- phase B: U = dz @ W_j^T  (U[row][c] = sum_co dz[row][co] W[kofs+c][co])
  if constexpr (MF) {
    using frag = typename CFrag8<T>::type;
    const int wv = threadIdx.x >> 6, lane = threadIdx.x & 63;
    const int l16 = lane & 15, lgrp = lane >> 4;
    if (wv < (C >> 4)) {               // wave wv owns U cols [16wv, 16wv+16)
      const int kchunks = Cout >> 5;
      cg_f32x4 acc[2];
      acc[0] = cg_f32x4{0.f, 0.f, 0.f, 0.f};
      acc[1] = cg_f32x4{0.f, 0.f, 0.f, 0.f};
      for (int kk = 0; kk < kchunks; ++kk) {
        // W_j^T[co][c] = W[kofs + c][co] -> 8 consecutive co per lane: one load
        frag bfr = *(const frag*)&W[(kofs + 16 * wv + l16) * Cout + kk * 32 + lgrp * 8];
        frag a0 = *(const frag*)&dzt[lds_swz(l16, (kk * 32 + lgrp * 8) * 2)];
        frag a1 = *(const frag*)&dzt[lds_swz(16 + l16, (kk * 32 + lgrp * 8) * 2)];
        acc[0] = cg_mfma(a0, bfr, acc[0]);
        acc[1] = cg_mfma(a1, bfr, acc[1]);
      }
      #pragma unroll
      for (int m = 0; m < 2; ++m)
        #pragma unroll
        for (int r = 0; r < 4; ++r)
          ut[(16 * m + 4 * lgrp + r) * 64 + 16 * wv + l16] = acc[m][r];
    }
  } else {
    for (int i = threadIdx.x; i < CG_ST * C; i += 256) {
      const int c = i % C, rl = i / C;
      float v = 0.f;
      for (int co = 0; co < Cout; ++co)
        v = fmaf(fdz[rl * 64 + co], toF<T>(W[(kofs + c) * Cout + co]), v);
      ut[rl * 64 + c] = v;
    }
  }
  __syncthreads();

  // ---- phase C: out = alpha*(G^T @ xin) + beta*p1 + U ----
  const T* xb = xin ? xin + (long)b * N * C : nullptr;
  const T* pb = p1 ? p1 + (long)b * N * C : nullptr;
  for (int i = threadIdx.x; i < CG_ST * C; i += 256) {
    const int c = i % C, rl = i / C;
    const int row = n0 + rl;
    if (row >= N) continue;
    float acc = ut[rl * 64 + c];
    if (xb != nullptr && alpha != 0.f) {
      float sp = 0.f;
      const int s = rowptr[row], e = rowptr[row + 1];
      int j = s;
      for (; j + 4 <= e; j += 4) {
        int cc[4];
        float vv[4];
        #pragma unroll
        for (int k = 0; k < 4; ++k) { cc[k] = colidx[j + k]; vv[k] = vals[j + k]; }
        float xv[4];
        #pragma unroll
        for (int k = 0; k < 4; ++k) xv[k] = toF<T>(xb[(long)cc[k] * C + c]);
        #pragma unroll
        for (int k = 0; k < 4; ++k) sp = fmaf(vv[k], xv[k], sp);
      }
      for (; j < e; ++j) sp += vals[j] * toF<T>(xb[(long)colidx[j] * C + c]);
      acc = fmaf(alpha, sp, acc);
    }
    if (pb != nullptr) acc = fmaf(beta, toF<T>(pb[(long)row * C + c]), acc);
    out[((long)b * N + row) * C + c] = fromF<T>(acc);
  }
}

template <typename T>
void launch_fused_fwd(hipStream_t st, const int* rp, const int* ci,
                      const float* v, const void* xin, const void* p1,
                      const void* x0, const void* W, const void* bias,
                      void* pout, float* yacc, void* yout, int B, int N,
                      int C, int Cout, int kofs, float alpha, float beta,
                      int first, int act) {
  dim3 grid((N + CG_ST - 1) / CG_ST, B);
  const size_t nslot = x0 ? 2 : 1;
  if constexpr (!std::is_same<T, float>::value) {
    if (W != nullptr && (C % 32 == 0) && (Cout % 16 == 0)) {
      hipLaunchKernelGGL((cheb_fused_fwd_kernel<T, true>), grid, dim3(256),
                         nslot * CG_ST * 128, st, rp, ci, v, (const T*)xin,
                         (const T*)p1, (const T*)x0, (const T*)W,
                         (const T*)bias, (T*)pout, yacc, (T*)yout, N, C, Cout,
                         kofs, alpha, beta, first, act);
      return;
    }
  }
  hipLaunchKernelGGL((cheb_fused_fwd_kernel<T, false>), grid, dim3(256),
                     nslot * CG_ST * 64 * 4, st, rp, ci, v, (const T*)xin,
                     (const T*)p1, (const T*)x0, (const T*)W, (const T*)bias,
                     (T*)pout, yacc, (T*)yout, N, C, Cout, kofs, alpha, beta,
                     first, act);
}

template <typename T>
void launch_fused_bwd(hipStream_t st, const int* rp, const int* ci,
                      const float* v, const void* xin, const void* p1,
                      const void* dz, const void* W, void* out, int B, int N,
                      int C, int Cout, int kofs, float alpha, float beta) {
  dim3 grid((N + CG_ST - 1) / CG_ST, B);
  if constexpr (!std::is_same<T, float>::value) {
    if ((Cout % 32 == 0) && (C % 16 == 0) && (C <= 64)) {
      hipLaunchKernelGGL((cheb_fused_bwd_kernel<T, true>), grid, dim3(256),
                         CG_ST * 128 + CG_ST * 64 * 4, st, rp, ci, v,
                         (const T*)xin, (const T*)p1, (const T*)dz,
                         (const T*)W, (T*)out, N, C, Cout, kofs, alpha, beta);
      return;
    }
  }
  hipLaunchKernelGGL((cheb_fused_bwd_kernel<T, false>), grid, dim3(256),
                     2 * CG_ST * 64 * 4, st, rp, ci, v, (const T*)xin,
                     (const T*)p1, (const T*)dz, (const T*)W, (T*)out, N, C,
                     Cout, kofs, alpha, beta);
}

}  // namespace

extern "C" void stmgcn_cheb_fused_fwd_step(
    void* stream_v, int dtype, const int* rowptr, const int* colidx,
    const float* vals, const void* xin, const void* p1, const void* x0,
    const void* W, const void* bias, void* pout, float* yacc, void* yout,
    int B, int N, int C, int Cout, int kofs, float alpha, float beta,
    int first, int act) {
  hipStream_t st = (hipStream_t)stream_v;
  switch (dtype) {
    case STM_F32:
      launch_fused_fwd<float>(st, rowptr, colidx, vals, xin, p1, x0, W, bias,
                              pout, yacc, yout, B, N, C, Cout, kofs, alpha,
                              beta, first, act);
      break;
    case STM_BF16:
      launch_fused_fwd<__hip_bfloat16>(st, rowptr, colidx, vals, xin, p1, x0,
                                       W, bias, pout, yacc, yout, B, N, C,
                                       Cout, kofs, alpha, beta, first, act);
      break;
    case STM_F16:
      launch_fused_fwd<__half>(st, rowptr, colidx, vals, xin, p1, x0, W, bias,
                               pout, yacc, yout, B, N, C, Cout, kofs, alpha,
                               beta, first, act);
      break;
  }
}

extern "C" void stmgcn_cheb_fused_bwd_step(
    void* stream_v, int dtype, const int* rowptr, const int* colidx,
    const float* vals, const void* xin, const void* p1, const void* dz,
    const void* W, void* out, int B, int N, int C, int Cout, int kofs,
    float alpha, float beta) {
  hipStream_t st = (hipStream_t)stream_v;
  switch (dtype) {
    case STM_F32:
      launch_fused_bwd<float>(st, rowptr, colidx, vals, xin, p1, dz, W, out,
                              B, N, C, Cout, kofs, alpha, beta);
      break;
    case STM_BF16:
      launch_fused_bwd<__hip_bfloat16>(st, rowptr, colidx, vals, xin, p1, dz,
                                       W, out, B, N, C, Cout, kofs, alpha, beta);
      break;
    case STM_F16:
      launch_fused_bwd<__half>(st, rowptr, colidx, vals, xin, p1, dz, W, out,
                               B, N, C, Cout, kofs, alpha, beta);
      break;
  }
}
