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
