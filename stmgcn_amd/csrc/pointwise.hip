// Fused pointwise/reduction kernels — CDNA4 (gfx950).
//
//  K3  seqsum_permute: (B,T,N,C) -> (B,N,T) feature-sum + transpose
//      (reference STMGCN.py:36,39)
//  K4  contextual gate: z = mean_n(gconv + x_seq); s = sigmoid(FC(relu(FC(z))))
//      with ONE weight-tied FC applied twice (STMGCN.py:43, quirk 2);
//      out = obs * s  (eqs. 6-9). Forward = one per-batch reduce+gate kernel
//      + one broadcast-multiply; backward mirrors it.
//  K7  branch-fuse + FC head: y = (sum_m feats_m) @ W^T + b (STMGCN.py:116-118)
//  K8  fused MSE loss + grad  (Model_Trainer.py:38)
//  K9  multi-tensor Adam over a flat fp32 master arena with bf16/f16 working
//      params (torch.optim.Adam semantics incl. L2-style weight decay)

#include "common.h"

#define GATE_MAX_T 16

namespace {

// ---- K3 ------------------------------------------------------------------
template <typename T>
__global__ void seqsum_permute_kernel(const T* __restrict__ obs,  // (B,T,N,C)
                                      T* __restrict__ out,        // (B,N,T)
                                      int B, int Tst, int N, int C) {
  // iterate in (b,t,n) source order: consecutive threads read consecutive
  // C-runs (fully coalesced); the (B,N,T) write is a small-stride scatter
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;  // over B*T*N
  if (i >= (long)B * Tst * N) return;
  const int n = i % N;
  const int t = (i / N) % Tst;
  const long b = i / ((long)N * Tst);
  float acc = 0.f;
  const T* src = obs + i * (long)C;
  int c = 0;
  for (; c + 4 <= C; c += 4) {
    acc += toF<T>(src[c]) + toF<T>(src[c + 1]) + toF<T>(src[c + 2]) +
           toF<T>(src[c + 3]);
  }
  for (; c < C; ++c) acc += toF<T>(src[c]);
  out[(b * N + n) * (long)Tst + t] = fromF<T>(acc);
}

template <typename T>
__global__ void seqsum_permute_bwd_kernel(const T* __restrict__ dxs,  // (B,N,T)
                                          T* __restrict__ dobs,       // (B,T,N,C)
                                          int B, int Tst, int N, int C) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;  // over B*T*N*C
  if (i >= (long)B * Tst * N * C) return;
  const int n = (i / C) % N;
  const int t = (i / ((long)C * N)) % Tst;
  const long b = i / ((long)C * N * Tst);
  // every element written exactly once -> caller may pass uninitialized dobs
  dobs[i] = dxs[(b * N + n) * (long)Tst + t];
}

// ---- K4 forward part A1: multi-block node reduce -> zsum (B,T) fp32 -------
// grid (nchunks, B): scales to any N (the old one-block-per-batch form was
// 3.5 ms/call at N=4096 — 16 blocks on 256 CUs).
template <typename T>
__global__ void gate_fwd_reduce_kernel(const T* __restrict__ g,     // (B,N,T)
                                       const T* __restrict__ xs,    // (B,N,T)
                                       float* __restrict__ zsum,    // (B,T) zeroed
                                       int N, int Tst) {
  __shared__ float red[256];
  const long b = blockIdx.y;
  const int nchunks = gridDim.x;
  const int chunk = (N + nchunks - 1) / nchunks;
  const int n0 = blockIdx.x * chunk;
  const int n1 = min(n0 + chunk, N);
  float zacc[GATE_MAX_T];
  for (int t = 0; t < Tst; ++t) zacc[t] = 0.f;
  const T* gb = g + b * (long)N * Tst;
  const T* xb = xs + b * (long)N * Tst;
  for (int n = n0 + threadIdx.x; n < n1; n += blockDim.x)
    for (int t = 0; t < Tst; ++t)
      zacc[t] += toF<T>(gb[(long)n * Tst + t]) + toF<T>(xb[(long)n * Tst + t]);
  for (int t = 0; t < Tst; ++t) {
    float v = zacc[t];
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if ((threadIdx.x & 63) == 0) red[(threadIdx.x >> 6) * GATE_MAX_T + t] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      float tot = 0.f;
      for (int w = 0; w < blockDim.x / 64; ++w) tot += red[w * GATE_MAX_T + t];
      unsafeAtomicAdd(&zsum[b * Tst + t], tot);
    }
    __syncthreads();
  }
}

// ---- K4 forward part A2: tiny per-batch tied double-FC --------------------
template <typename T>
__global__ void gate_fwd_fc_kernel(const float* __restrict__ zsum, // (B,T)
                                   const T* __restrict__ fcw,      // (T,T)
                                   const T* __restrict__ fcb,      // (T,)
                                   float* __restrict__ z_out, float* __restrict__ u_out,
                                   float* __restrict__ s_out, int N, int Tst, int B) {
  const long b = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  float z[GATE_MAX_T], u[GATE_MAX_T];
  for (int t = 0; t < Tst; ++t) z[t] = zsum[b * Tst + t] / (float)N;
  for (int t = 0; t < Tst; ++t) {                  // u = relu(W z + b)
    float a = toF<T>(fcb[t]);
    for (int j = 0; j < Tst; ++j) a += toF<T>(fcw[t * Tst + j]) * z[j];
    u[t] = a > 0.f ? a : 0.f;
  }
  for (int t = 0; t < Tst; ++t) {                  // s = sigmoid(W u + b)
    float a = toF<T>(fcb[t]);
    for (int j = 0; j < Tst; ++j) a += toF<T>(fcw[t * Tst + j]) * u[j];
    z_out[b * Tst + t] = z[t];
    u_out[b * Tst + t] = u[t];
    s_out[b * Tst + t] = stm_sigmoid(a);
  }
}

// ---- K4 forward part B: out = obs * s[b,t] --------------------------------
template <typename T>
__global__ void gate_scale_kernel(const T* __restrict__ obs,   // (B,T,N,C)
                                  const float* __restrict__ s, // (B,T)
                                  T* __restrict__ out, int Tst, long NC, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  const long bt = i / NC;
  out[i] = fromF<T>(toF<T>(obs[i]) * s[bt]);
}

// ---- K4 backward part A1: ds = sum_{n,c} dout*obs (multi-block + atomics) -
template <typename T>
__global__ void gate_bwd_reduce_kernel(const T* __restrict__ dout, // (B,T,N,C)
                                       const T* __restrict__ obs,  // (B,T,N,C)
                                       float* __restrict__ ds_out, // (B,T) zeroed
                                       int N, int C, int Tst) {
  __shared__ float red[256];
  const long b = blockIdx.y;
  const long NC = (long)N * C;
  const int nchunks = gridDim.x;
  const long chunk = (NC + nchunks - 1) / nchunks;
  const long i0 = blockIdx.x * chunk;
  const long i1 = min(i0 + chunk, NC);
  float dsacc[GATE_MAX_T];
  for (int t = 0; t < Tst; ++t) dsacc[t] = 0.f;
  const T* db_ = dout + b * Tst * NC;
  const T* ob = obs + b * Tst * NC;
  for (int t = 0; t < Tst; ++t)
    for (long i = i0 + threadIdx.x; i < i1; i += blockDim.x)
      dsacc[t] += toF<T>(db_[t * NC + i]) * toF<T>(ob[t * NC + i]);
  for (int t = 0; t < Tst; ++t) {
    float v = dsacc[t];
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if ((threadIdx.x & 63) == 0) red[(threadIdx.x >> 6) * GATE_MAX_T + t] = v;
    __syncthreads();
    if (threadIdx.x == 0) {
      float tot = 0.f;
      for (int w = 0; w < blockDim.x / 64; ++w) tot += red[w * GATE_MAX_T + t];
      unsafeAtomicAdd(&ds_out[b * Tst + t], tot);
    }
    __syncthreads();
  }
}

// ---- K4 backward part A2: tiny per-batch tied-FC backward -----------------
template <typename T>
__global__ void gate_bwd_fc_kernel(const float* __restrict__ ds,  // (B,T)
                                   const T* __restrict__ fcw,     // (T,T)
                                   const float* __restrict__ z,
                                   const float* __restrict__ u,
                                   const float* __restrict__ s,
                                   float* __restrict__ dz_out,    // (B,T)
                                   float* __restrict__ dw_part,   // (B,T,T)
                                   float* __restrict__ db_part,   // (B,T)
                                   int Tst, int B) {
  const long b = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  // tied double-FC backward: s = sig(W u + b), u = relu(W z + b)
  float dv2[GATE_MAX_T], du[GATE_MAX_T], dv1[GATE_MAX_T];
  for (int t = 0; t < Tst; ++t) {
    const float sv = s[b * Tst + t];
    dv2[t] = ds[b * Tst + t] * sv * (1.f - sv);
  }
  for (int j = 0; j < Tst; ++j) {                   // du = W^T dv2
    float a = 0.f;
    for (int t = 0; t < Tst; ++t) a += toF<T>(fcw[t * Tst + j]) * dv2[t];
    du[j] = a;
  }
  for (int t = 0; t < Tst; ++t) dv1[t] = u[b * Tst + t] > 0.f ? du[t] : 0.f;
  for (int j = 0; j < Tst; ++j) {                   // dz = W^T dv1
    float a = 0.f;
    for (int t = 0; t < Tst; ++t) a += toF<T>(fcw[t * Tst + j]) * dv1[t];
    dz_out[b * Tst + j] = a;
  }
  // dW partial = dv2 (x) u + dv1 (x) z ; db partial = dv2 + dv1
  for (int t = 0; t < Tst; ++t) {
    for (int j = 0; j < Tst; ++j)
      dw_part[(b * Tst + t) * Tst + j] =
          dv2[t] * u[b * Tst + j] + dv1[t] * z[b * Tst + j];
    db_part[b * Tst + t] = dv2[t] + dv1[t];
  }
}

// ---- K4 backward part B: dobs = dout*s + dz/N ; dg = dz/N -----------------
template <typename T>
__global__ void gate_bwd_scatter_kernel(const T* __restrict__ dout, // (B,T,N,C)
                                        const float* __restrict__ s,
                                        const float* __restrict__ dz, // (B,T)
                                        T* __restrict__ dobs,         // (B,T,N,C)
                                        T* __restrict__ dg,           // (B,N,T)
                                        int Tst, int N, int C, float invN,
                                        long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  const long NC = (long)N * C;
  const long bt = i / NC;
  const long b = bt / Tst;
  const int t = bt % Tst;
  const float dzv = dz[bt] * invN;
  dobs[i] = fromF<T>(toF<T>(dout[i]) * s[bt] + dzv);
  if ((i % NC) % C == 0) {
    const int n = (i % NC) / C;
    dg[(b * N + n) * (long)Tst + t] = fromF<T>(dzv);
  }
}

// ---- K7: head y[b,n] = sum_m feats_m[b,n,:] . w + bias --------------------
template <typename T>
__global__ void head_fwd_kernel(const T* __restrict__ f0, const T* __restrict__ f1,
                                const T* __restrict__ f2, const T* __restrict__ w,
                                const T* __restrict__ bias_p, T* __restrict__ y,
                                T* __restrict__ fsum,  // saved (B*N, G)
                                int G, long BN) {
  // one wave per row; lane g covers channel g (G <= 64)
  const long row = ((long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  const int lane = threadIdx.x & 63;
  if (row >= BN) return;
  float v = 0.f;
  if (lane < G) {
    v = toF<T>(f0[row * G + lane]);
    if (f1) v += toF<T>(f1[row * G + lane]);
    if (f2) v += toF<T>(f2[row * G + lane]);
    fsum[row * G + lane] = fromF<T>(v);
    v *= toF<T>(w[lane]);
  }
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  if (lane == 0) y[row] = fromF<T>(v + toF<T>(bias_p[0]));
}

// dfeat_m[b,n,g] = dy[b,n] * w[g]  (same for every branch)
template <typename T>
__global__ void head_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ w,
                                T* __restrict__ dfeat, int G, long total) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= total) return;
  dfeat[i] = fromF<T>(toF<T>(dy[i / G]) * toF<T>(w[i % G]));
}

// head wgrad: dw[g] = sum_r dy[r] * fsum[r,g]; db = sum_r dy[r] — a pure
// bandwidth reduction (replaces the last library GEMM on the step: a
// hipBLASLt MT64x16x512 tall-skinny at ~85 us for 1x64 output).
template <typename T>
__global__ void head_wgrad_kernel(const T* __restrict__ dy,
                                  const T* __restrict__ fsum,
                                  float* __restrict__ dw,   // (G,) zeroed
                                  float* __restrict__ db,   // (1,) zeroed
                                  int G, long BN) {
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const long chunk = (BN + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * chunk;
  const long r1 = (r0 + chunk < BN) ? r0 + chunk : BN;
  // two independent accumulator chains keep 2+ row loads in flight
  float acc = 0.f, acc2 = 0.f, accb = 0.f, accb2 = 0.f;
  long r = r0 + wv;
  for (; r + 4 < r1; r += 8) {
    const float d = toF<T>(dy[r]);        // wave-uniform scalar loads
    const float d2 = toF<T>(dy[r + 4]);
    if (lane < G) {
      acc += d * toF<T>(fsum[r * G + lane]);
      acc2 += d2 * toF<T>(fsum[(r + 4) * G + lane]);
    }
    accb += d;
    accb2 += d2;
  }
  for (; r < r1; r += 4) {
    const float d = toF<T>(dy[r]);
    if (lane < G) acc += d * toF<T>(fsum[r * G + lane]);
    accb += d;
  }
  acc += acc2;
  accb += accb2;
  __shared__ float red[4][64];
  __shared__ float redb[4];
  red[wv][lane] = acc;
  if (lane == 0) redb[wv] = accb;
  __syncthreads();
  if (wv == 0) {
    const float v = red[0][lane] + red[1][lane] + red[2][lane] + red[3][lane];
    if (lane < G) unsafeAtomicAdd(&dw[lane], v);
    if (lane == 0)
      unsafeAtomicAdd(db, redb[0] + redb[1] + redb[2] + redb[3]);
  }
}

// ---- K8: fused MSE loss + grad -------------------------------------------
template <typename T>
__global__ void mse_fwd_kernel(const T* __restrict__ pred, const T* __restrict__ tgt,
                               float* __restrict__ loss_out,  // (1,) pre-zeroed
                               T* __restrict__ diff, long n) {
  __shared__ float red[4];
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  float v = 0.f;
  if (i < n) {
    const float d = toF<T>(pred[i]) - toF<T>(tgt[i]);
    diff[i] = fromF<T>(d);
    v = d * d;
  }
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = v;
  __syncthreads();
  if (threadIdx.x == 0)
    atomicAdd(loss_out, (red[0] + red[1] + red[2] + red[3]) / (float)n);
}

template <typename T>
__global__ void mse_bwd_kernel(const T* __restrict__ diff,
                               const float* __restrict__ gscale,  // d(loss) scalar
                               T* __restrict__ dpred, float two_over_n, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dpred[i] = fromF<T>(toF<T>(diff[i]) * two_over_n * gscale[0]);
}

// ---- K9: multi-tensor Adam over a flat arena ------------------------------
// torch.optim.Adam semantics: g' = g + wd*p ; m,v updates; p -= lr * mhat /
// (sqrt(vhat) + eps). fp32 master params; T-typed working copy rewritten.
template <typename T>
__global__ void adam_kernel(float* __restrict__ master, T* __restrict__ param,
                            const T* __restrict__ grad, float* __restrict__ m,
                            float* __restrict__ v, const float* __restrict__ hyper,
                            // hyper: lr, beta1, beta2, eps, wd, bc1, bc2
                            long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const float lr = hyper[0], b1 = hyper[1], b2 = hyper[2], eps = hyper[3],
              wd = hyper[4], bc1 = hyper[5], bc2 = hyper[6];
  const float p = master[i];
  float g = toF<T>(grad[i]) + wd * p;
  const float mn = b1 * m[i] + (1.f - b1) * g;
  const float vn = b2 * v[i] + (1.f - b2) * g * g;
  m[i] = mn; v[i] = vn;
  const float pn = p - lr * (mn / bc1) / (sqrtf(vn / bc2) + eps);
  master[i] = pn;
  param[i] = fromF<T>(pn);
}

// device-side Adam hyper-state advance: keeps the whole optimizer step
// inside a hipGraph (no host-computed bias correction per step).
// hyper: [lr, b1, b2, eps, wd, bc1, bc2, b1pow, b2pow]
__global__ void adam_prep_kernel(float* hyper) {
  if (threadIdx.x == 0) {
    hyper[7] *= hyper[1];
    hyper[8] *= hyper[2];
    hyper[5] = 1.f - hyper[7];
    hyper[6] = 1.f - hyper[8];
  }
}

// chunks so the node reduction fills the chip regardless of B
static int gate_chunks(int B, long work_per_b) {
  long per_chunk = 2048;
  long nch = (work_per_b + per_chunk - 1) / per_chunk;
  long cap = (B > 0) ? (2048 / B > 0 ? 2048 / B : 1) : 1;
  if (nch > cap) nch = cap;
  if (nch < 1) nch = 1;
  return (int)nch;
}

template <typename T>
static void launch_gate_fwd(hipStream_t st, const T* g, const T* xs,
                            const T* fcw, const T* fcb, const T* obs,
                            float* z, float* u, float* s, T* out,
                            int B, int Tst, int N, int C) {
  // z arrives zeroed and doubles as the zsum atomic target (A2 rescales)
  const dim3 gridA(gate_chunks(B, N), B);
  hipLaunchKernelGGL(gate_fwd_reduce_kernel<T>, gridA, dim3(256), 0, st,
                     g, xs, z, N, Tst);
  hipLaunchKernelGGL(gate_fwd_fc_kernel<T>, dim3((B + 63) / 64), dim3(64), 0,
                     st, z, fcw, fcb, z, u, s, N, Tst, B);
  hipLaunchKernelGGL(gate_scale_kernel<T>, dim3(((long)B * Tst * N * C + 255) / 256),
                     dim3(256), 0, st, obs, s, out, Tst, (long)N * C,
                     (long)B * Tst * N * C);
}

template <typename T>
static void launch_gate_bwd(hipStream_t st, const T* dout, const T* obs,
                            const T* fcw, const float* z, const float* u,
                            const float* s, float* dz, float* dw_part,
                            float* db_part, T* dobs, T* dg,
                            int B, int Tst, int N, int C) {
  const long total = (long)B * Tst * N * C;
  const float invN = 1.f / (float)N;
  // dz arrives zeroed and doubles as the ds atomic target (fc pass rewrites)
  const dim3 gridA(gate_chunks(B, (long)N * C), B);
  hipLaunchKernelGGL(gate_bwd_reduce_kernel<T>, gridA, dim3(256), 0, st,
                     dout, obs, dz, N, C, Tst);
  hipLaunchKernelGGL(gate_bwd_fc_kernel<T>, dim3((B + 63) / 64), dim3(64), 0,
                     st, dz, fcw, z, u, s, dz, dw_part, db_part, Tst, B);
  hipLaunchKernelGGL(gate_bwd_scatter_kernel<T>, dim3((total + 255) / 256),
                     dim3(256), 0, st, dout, s, dz, dobs, dg, Tst, N, C, invN,
                     total);
}

}  // namespace

#define DISPATCH(fn, ...)                                        \
  switch (dtype) {                                               \
    case STM_F32: fn<float>(__VA_ARGS__); break;                 \
    case STM_BF16: fn<__hip_bfloat16>(__VA_ARGS__); break;       \
    case STM_F16: fn<__half>(__VA_ARGS__); break;                \
  }

extern "C" {

void stmgcn_seqsum_permute(void* stream, int dtype, const void* obs, void* out,
                           int B, int Tst, int N, int C) {
  const long total = (long)B * N * Tst;
  dim3 grid((total + 255) / 256);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(seqsum_permute_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)obs, (float*)out, B, Tst, N, C); break;
    case STM_BF16: hipLaunchKernelGGL(seqsum_permute_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)obs, (__hip_bfloat16*)out, B, Tst, N, C); break;
    case STM_F16: hipLaunchKernelGGL(seqsum_permute_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)obs, (__half*)out, B, Tst, N, C); break;
  }
}

void stmgcn_seqsum_permute_bwd(void* stream, int dtype, const void* dxs,
                               void* dobs, int B, int Tst, int N, int C) {
  const long total = (long)B * Tst * N * C;
  dim3 grid((total + 255) / 256);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(seqsum_permute_bwd_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)dxs, (float*)dobs, B, Tst, N, C); break;
    case STM_BF16: hipLaunchKernelGGL(seqsum_permute_bwd_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)dxs, (__hip_bfloat16*)dobs, B, Tst, N, C); break;
    case STM_F16: hipLaunchKernelGGL(seqsum_permute_bwd_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)dxs, (__half*)dobs, B, Tst, N, C); break;
  }
}


void stmgcn_gate_fwd(void* stream, int dtype, const void* g, const void* xs,
                     const void* fcw, const void* fcb, const void* obs,
                     float* z, float* u, float* s, void* out,
                     int B, int Tst, int N, int C) {
  hipStream_t st = (hipStream_t)stream;
  switch (dtype) {
    case STM_F32:
      launch_gate_fwd<float>(st, (const float*)g, (const float*)xs, (const float*)fcw, (const float*)fcb, (const float*)obs, z, u, s, (float*)out, B, Tst, N, C);
      break;
    case STM_BF16:
      launch_gate_fwd<__hip_bfloat16>(st, (const __hip_bfloat16*)g, (const __hip_bfloat16*)xs, (const __hip_bfloat16*)fcw, (const __hip_bfloat16*)fcb, (const __hip_bfloat16*)obs, z, u, s, (__hip_bfloat16*)out, B, Tst, N, C);
      break;
    case STM_F16:
      launch_gate_fwd<__half>(st, (const __half*)g, (const __half*)xs, (const __half*)fcw, (const __half*)fcb, (const __half*)obs, z, u, s, (__half*)out, B, Tst, N, C);
      break;
  }
}


void stmgcn_gate_bwd(void* stream, int dtype, const void* dout, const void* obs,
                     const void* fcw, const float* z, const float* u,
                     const float* s, float* dz, float* dw_part, float* db_part,
                     void* dobs, void* dg, int B, int Tst, int N, int C) {
  hipStream_t st = (hipStream_t)stream;
  switch (dtype) {
    case STM_F32:
      launch_gate_bwd<float>(st, (const float*)dout, (const float*)obs, (const float*)fcw, z, u, s, dz, dw_part, db_part, (float*)dobs, (float*)dg, B, Tst, N, C);
      break;
    case STM_BF16:
      launch_gate_bwd<__hip_bfloat16>(st, (const __hip_bfloat16*)dout, (const __hip_bfloat16*)obs, (const __hip_bfloat16*)fcw, z, u, s, dz, dw_part, db_part, (__hip_bfloat16*)dobs, (__hip_bfloat16*)dg, B, Tst, N, C);
      break;
    case STM_F16:
      launch_gate_bwd<__half>(st, (const __half*)dout, (const __half*)obs, (const __half*)fcw, z, u, s, dz, dw_part, db_part, (__half*)dobs, (__half*)dg, B, Tst, N, C);
      break;
  }
}

void stmgcn_head_fwd(void* stream, int dtype, const void* f0, const void* f1,
                     const void* f2, const void* w, const void* bias, void* y,
                     void* fsum, int G, long BN) {
  dim3 grid((BN * 64 + 255) / 256);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(head_fwd_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)f0, (const float*)f1, (const float*)f2, (const float*)w, (const float*)bias, (float*)y, (float*)fsum, G, BN); break;
    case STM_BF16: hipLaunchKernelGGL(head_fwd_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)f0, (const __hip_bfloat16*)f1, (const __hip_bfloat16*)f2, (const __hip_bfloat16*)w, (const __hip_bfloat16*)bias, (__hip_bfloat16*)y, (__hip_bfloat16*)fsum, G, BN); break;
    case STM_F16: hipLaunchKernelGGL(head_fwd_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)f0, (const __half*)f1, (const __half*)f2, (const __half*)w, (const __half*)bias, (__half*)y, (__half*)fsum, G, BN); break;
  }
}

void stmgcn_head_bwd(void* stream, int dtype, const void* dy, const void* w,
                     void* dfeat, int G, long BN) {
  const long total = BN * G;
  dim3 grid((total + 255) / 256);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(head_bwd_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)dy, (const float*)w, (float*)dfeat, G, total); break;
    case STM_BF16: hipLaunchKernelGGL(head_bwd_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)dy, (const __hip_bfloat16*)w, (__hip_bfloat16*)dfeat, G, total); break;
    case STM_F16: hipLaunchKernelGGL(head_bwd_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)dy, (const __half*)w, (__half*)dfeat, G, total); break;
  }
}

void stmgcn_head_wgrad(void* stream, int dtype, const void* dy,
                       const void* fsum, float* dw, float* db, int G,
                       long BN) {
  long nblk = (BN + 255) / 256;   // fill the chip: 128+ blocks at bench size
  if (nblk > 512) nblk = 512;
  if (nblk < 1) nblk = 1;
  dim3 grid((unsigned)nblk);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(head_wgrad_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)dy, (const float*)fsum, dw, db, G, BN); break;
    case STM_BF16: hipLaunchKernelGGL(head_wgrad_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)dy, (const __hip_bfloat16*)fsum, dw, db, G, BN); break;
    case STM_F16: hipLaunchKernelGGL(head_wgrad_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)dy, (const __half*)fsum, dw, db, G, BN); break;
  }
}

void stmgcn_mse_fwd(void* stream, int dtype, const void* pred, const void* tgt,
                    float* loss, void* diff, long n) {
  dim3 grid((n + 255) / 256);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(mse_fwd_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)pred, (const float*)tgt, loss, (float*)diff, n); break;
    case STM_BF16: hipLaunchKernelGGL(mse_fwd_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)pred, (const __hip_bfloat16*)tgt, loss, (__hip_bfloat16*)diff, n); break;
    case STM_F16: hipLaunchKernelGGL(mse_fwd_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)pred, (const __half*)tgt, loss, (__half*)diff, n); break;
  }
}

void stmgcn_mse_bwd(void* stream, int dtype, const void* diff, const float* gscale,
                    void* dpred, long n) {
  dim3 grid((n + 255) / 256);
  const float ton = 2.f / (float)n;
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(mse_bwd_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, (const float*)diff, gscale, (float*)dpred, ton, n); break;
    case STM_BF16: hipLaunchKernelGGL(mse_bwd_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, (const __hip_bfloat16*)diff, gscale, (__hip_bfloat16*)dpred, ton, n); break;
    case STM_F16: hipLaunchKernelGGL(mse_bwd_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, (const __half*)diff, gscale, (__half*)dpred, ton, n); break;
  }
}

void stmgcn_adam(void* stream, int dtype, float* master, void* param,
                 const void* grad, float* m, float* v, float* hyper,
                 long n) {
  hipLaunchKernelGGL(adam_prep_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream, hyper);
  dim3 grid((n + 255) / 256);
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(adam_kernel<float>, grid, dim3(256), 0, (hipStream_t)stream, master, (float*)param, (const float*)grad, m, v, hyper, n); break;
    case STM_BF16: hipLaunchKernelGGL(adam_kernel<__hip_bfloat16>, grid, dim3(256), 0, (hipStream_t)stream, master, (__hip_bfloat16*)param, (const __hip_bfloat16*)grad, m, v, hyper, n); break;
    case STM_F16: hipLaunchKernelGGL(adam_kernel<__half>, grid, dim3(256), 0, (hipStream_t)stream, master, (__half*)param, (const __half*)grad, m, v, hyper, n); break;
  }
}

}  // extern "C"

// ---- K9b: multi-segment gradient gather into the flat Adam arena ----------
// FusedAdam runs autograd with .grad = None (no per-param accumulate-add
// kernels); the fresh grads are packed into the contiguous arena by ONE
// launch per <=64 segments. Null src -> zero-fill (param got no grad).
namespace {
struct GatherSeg {
  const void* src[64];
  long ofs[64];
  int len[64];
};

template <typename T>
__global__ void gather_grads_kernel(GatherSeg a, T* __restrict__ arena,
                                    int nseg) {
  const int seg = blockIdx.x;
  if (seg >= nseg) return;
  T* dst = arena + a.ofs[seg];
  const T* s = (const T*)a.src[seg];
  const int len = a.len[seg];
  const int step = blockDim.x * gridDim.y;
  for (int i = blockIdx.y * blockDim.x + threadIdx.x; i < len; i += step)
    dst[i] = s ? s[i] : fromF<T>(0.f);
}
}  // namespace

extern "C" void stmgcn_gather_grads(void* stream, int dtype, const void** srcs,
                                    const long* ofs, const int* lens, int nseg,
                                    void* arena) {
  GatherSeg a;
  int maxlen = 1;
  for (int i = 0; i < nseg && i < 64; ++i) {
    a.src[i] = srcs[i]; a.ofs[i] = ofs[i]; a.len[i] = lens[i];
    if (lens[i] > maxlen) maxlen = lens[i];
  }
  const int chunks = min(8, (maxlen + 2047) / 2048);
  const dim3 grid(nseg, chunks), blk(256);
  hipStream_t st = (hipStream_t)stream;
  switch (dtype) {
    case STM_F32: hipLaunchKernelGGL(gather_grads_kernel<float>, grid, blk, 0, st, a, (float*)arena, nseg); break;
    case STM_BF16: hipLaunchKernelGGL(gather_grads_kernel<__hip_bfloat16>, grid, blk, 0, st, a, (__hip_bfloat16*)arena, nseg); break;
    case STM_F16: hipLaunchKernelGGL(gather_grads_kernel<__half>, grid, blk, 0, st, a, (__half*)arena, nseg); break;
  }
}
