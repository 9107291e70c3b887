// Weight-gradient (reduction-GEMM) kernels for CDNA4 (gfx950) — SURVEY K10.
//
// The wgrad shapes here are tall-skinny reductions: C[M<=256, N<=64] =
// A[rows, M]^T @ B[rows, N] with rows up to B*N*T (262k+). hipBLASLt serves
// these poorly (measured 437 us for (256,64|262k) bf16 on MI355X — see
// profiles/r01_bench1024_kernel_stats.md, 47.8% of step time); these kernels
// replace 15 library GEMM + 9 cat + 9 reduce launches per step with one
// launch per RNN branch + one per graph-conv.
//
// Scheme per workgroup (256 threads, 4 waves): march a private chunk of the
// reduction dim in 32-row K-tiles; stage A^T (and B^T) tiles in LDS with an
// XOR swizzle that makes the 16B MFMA fragment reads bank-conflict-free;
// accumulate the full output in registers (v_mfma_f32_16x16x32, fp32 acc);
// one unsafeAtomicAdd(f32) per output element per workgroup at the end.
// Bias grads (column sums) ride along in the staging pass for free.
//
// LSTM variant reads the natural-layout dA stream (L, Tst*S_pad, 4H) written
// by lstm_bwd_kernel and does ALL layers in one launch (grid.y = L). The
// h_{t-1} operand for dw_hh is hseq offset by -S_pad rows (zero for t == 0):
// the host-side torch.cat shift this replaces was 1.9% of step time.
// grid.z splits the 256 gate rows into 256/GT slices: GT=128 halves the
// per-wave accumulator state (full-GT compiled to 126 VGPR + 140 AGPR ->
// 1 wave/SIMD; measured flat ~400 us across chunk counts = latency-bound).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

template <typename T> struct WFrag8;
template <> struct WFrag8<__hip_bfloat16> { using type = bf16x8; using elem = __bf16; };
template <> struct WFrag8<__half> { using type = f16x8; using elem = _Float16; };

__device__ __forceinline__ f32x4 wmfma(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ f32x4 wmfma(f16x8 a, f16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
}

// Transposed-tile LDS addressing: column g (output dim), row-pair byte offset
// off in [0,64). 64B per column; XOR displaces the 16B fragment window so the
// MFMA reads (16 consecutive g, 16B each) hit 64 distinct banks.
__device__ __forceinline__ int tswz(int g, int off) {
  return g * 64 + (off ^ ((((unsigned)g >> 2) & 3) << 4));
}

// Stage one 32-row tile of src (rows-major, ld elems) transposed into LDS
// tile [ncols][32] (64B/col). Unit u -> (pair pr = u & 15, col-block cb =
// u >> 4, 8 cols each); zero-fills guarded rows/cols. Each thread covers
// units {tid, tid+256, ...} so a caller tracking column sums per unit slot
// keeps a fixed col-block per slot.
template <typename T, int NUNITS>
__device__ __forceinline__ void stage_tileT(
    char* lds_tile, const T* __restrict__ src, long ld, long r0, long rmax,
    int ncols, float* colsum /* per-thread [NUNITS/256][8] or null */) {
  using frag = typename WFrag8<T>::type;
  for (int u = threadIdx.x, slot = 0; u < NUNITS; u += 256, ++slot) {
    const int pr = u & 15, cb = u >> 4;
    const long r = r0 + pr * 2;
    frag v0 = {}, v1 = {};
    if (cb * 8 < ncols) {
      if (r < rmax) v0 = *(const frag*)&src[r * ld + cb * 8];
      if (r + 1 < rmax) v1 = *(const frag*)&src[(r + 1) * ld + cb * 8];
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int g = cb * 8 + j;
      union { T t2[2]; int i; } pk;
      pk.t2[0] = ((const T*)&v0)[j];
      pk.t2[1] = ((const T*)&v1)[j];
      *(int*)&lds_tile[tswz(g, pr * 4)] = pk.i;
      if (colsum) colsum[slot * 8 + j] += toF<T>(pk.t2[0]) + toF<T>(pk.t2[1]);
    }
  }
}

// A-fragment (or B-fragment) read from a transposed LDS tile: output index
// g = g0 + l16, reduction k = lgrp*8..+8 -> 16B at tswz(g, lgrp*16).
template <typename T>
__device__ __forceinline__ typename WFrag8<T>::type frag_from(
    const char* lds_tile, int g, int lgrp) {
  return *(const typename WFrag8<T>::type*)&lds_tile[tswz(g, lgrp * 16)];
}

}  // namespace

// ===========================================================================
// Fused multi-layer LSTM weight gradients: one launch, grid (nchunks, L).
//   dwih[l] += dA_l^T @ (l == 0 ? x : hseq[l-1])      (4H, cin_l<=64)
//   dwhh[l] += dA_l^T @ hseq[l] shifted one step back (4H, H)
//   db[l]   += colsum(dA_l)                           (4H,)
// dA: (L, R=Tst*S_pad, 4H) natural; hseq: (L, R, H); x: (S, Tst, Cin).
// Row r of the flat reduction dim maps to (t = r / S_pad, s = r % S_pad).
template <typename T, bool CIN1, int GT>
__global__ void __launch_bounds__(256, 1)
lstm_wgrad_kernel(const T* __restrict__ dA, const T* __restrict__ hseq,
                  const T* __restrict__ x, float* __restrict__ dwih,
                  float* __restrict__ dwhh, float* __restrict__ db,
                  long R, long S_pad, int S, int Tst, int L) {
  using frag = typename WFrag8<T>::type;
  constexpr int MTG = GT / 64;      // gate m-tiles per wave (GT=128 -> 2)
  constexpr int GU = (GT / 8) * 16; // dA staging units per K-tile
  extern __shared__ char lds[];
  char* dAT = lds;                     // [GT][32] T
  char* hpT = lds + GT * 64;           // [64][32] T -> 4 KiB
  char* hxT = lds + GT * 64 + 4096;    // [64][32] T -> 4 KiB
  float* red = (float*)lds;            // db reduction scratch (reuses dAT)

  const int layer = blockIdx.y;
  const int g0 = blockIdx.z * GT;      // gate-row slice of this WG
  const int wv = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int l16 = lane & 15, lgrp = lane >> 4;
  const long chunk = (R + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * chunk;
  const long r1 = (r0 + chunk < R) ? r0 + chunk : R;

  const T* dA_l = dA + (long)layer * R * 256;
  const T* hp_l = hseq + (long)layer * R * 64;       // index r - S_pad
  const T* hx_l = (layer > 0) ? hseq + (long)(layer - 1) * R * 64 : nullptr;
  const bool l0 = (layer == 0);

  f32x4 acc_hh[MTG][4], acc_ih[MTG][4];
  #pragma unroll
  for (int mt = 0; mt < MTG; ++mt)
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      acc_hh[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
      acc_ih[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
    }
  float dbp[GU / 256][8];
  #pragma unroll
  for (int s_ = 0; s_ < GU / 256; ++s_)
    #pragma unroll
    for (int j = 0; j < 8; ++j) dbp[s_][j] = 0.f;

  // Software-pipelined main loop: global loads for tile k+1 issue right
  // after the barrier, overlapping tile k's MFMA phase (at 2-3 WGs/CU the
  // measured kernel was load-latency-bound: flat across chunk counts).
  static_assert(GU == 256, "one dA unit per thread assumed by the pipeline");
  const int pr = threadIdx.x & 15, cb = threadIdx.x >> 4;   // dA unit
  const int hcb = cb & 7;                                   // hp/hx unit (tid<128)
  const bool hthread = threadIdx.x < 128;

  frag va0, va1, vh0, vh1, vx0, vx1;
  auto load_tiles = [&](long kt, frag& a0, frag& a1, frag& h0, frag& h1,
                        frag& x0f, frag& x1f) {
    const frag fz = {};
    const long ra = kt + pr * 2, rb = ra + 1;
    a0 = fz; a1 = fz;
    if (ra < r1) a0 = *(const frag*)&dA_l[ra * 256 + g0 + cb * 8];
    if (rb < r1) a1 = *(const frag*)&dA_l[rb * 256 + g0 + cb * 8];
    h0 = fz; h1 = fz; x0f = fz; x1f = fz;
    if (hthread) {
      // h_{t-1}: hseq[l] offset -S_pad rows, zero for t==0 (the guard must
      // gate the LOAD — for layer 0, r - S_pad points before the allocation)
      if (ra < r1 && ra >= S_pad)
        h0 = *(const frag*)&hp_l[(ra - S_pad) * 64 + hcb * 8];
      if (rb < r1 && rb >= S_pad)
        h1 = *(const frag*)&hp_l[(rb - S_pad) * 64 + hcb * 8];
      if (!l0) {
        if (ra < r1) x0f = *(const frag*)&hx_l[ra * 64 + hcb * 8];
        if (rb < r1) x1f = *(const frag*)&hx_l[rb * 64 + hcb * 8];
      } else if (CIN1) {
        if (hcb == 0) {
          const long sa = ra % S_pad, ta = ra / S_pad;
          const long sb = rb % S_pad, tb = rb / S_pad;
          if (ra < r1 && sa < S) ((T*)&x0f)[0] = x[sa * Tst + ta];
          if (rb < r1 && sb < S) ((T*)&x1f)[0] = x[sb * Tst + tb];
        }
      } else {
        if (ra < r1) {
          const long s_ = ra % S_pad, t_ = ra / S_pad;
          if (s_ < S) x0f = *(const frag*)&x[(s_ * Tst + t_) * 64 + hcb * 8];
        }
        if (rb < r1) {
          const long s_ = rb % S_pad, t_ = rb / S_pad;
          if (s_ < S) x1f = *(const frag*)&x[(s_ * Tst + t_) * 64 + hcb * 8];
        }
      }
    }
  };
  auto commit_tiles = [&]() {
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      union { T t2[2]; int i; } pk;
      pk.t2[0] = ((const T*)&va0)[j];
      pk.t2[1] = ((const T*)&va1)[j];
      *(int*)&dAT[tswz(cb * 8 + j, pr * 4)] = pk.i;
      dbp[0][j] += toF<T>(pk.t2[0]) + toF<T>(pk.t2[1]);
    }
    if (hthread) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { T t2[2]; int i; } pk;
        pk.t2[0] = ((const T*)&vh0)[j];
        pk.t2[1] = ((const T*)&vh1)[j];
        *(int*)&hpT[tswz(hcb * 8 + j, pr * 4)] = pk.i;
        pk.t2[0] = ((const T*)&vx0)[j];
        pk.t2[1] = ((const T*)&vx1)[j];
        *(int*)&hxT[tswz(hcb * 8 + j, pr * 4)] = pk.i;
      }
    }
  };

  load_tiles(r0, va0, va1, vh0, vh1, vx0, vx1);
  for (long kt = r0; kt < r1; kt += 32) {
    commit_tiles();
    __syncthreads();
    frag na0, na1, nh0, nh1, nx0, nx1;
    const bool more = kt + 32 < r1;
    if (more) load_tiles(kt + 32, na0, na1, nh0, nh1, nx0, nx1);

    // ---- MFMA: wave wv owns gate rows g0 + [wv*16*MTG, ...) --------------
    #pragma unroll
    for (int mt = 0; mt < MTG; ++mt) {
      const frag a = frag_from<T>(dAT, wv * 16 * MTG + mt * 16 + l16, lgrp);
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        acc_hh[mt][nt] = wmfma(a, frag_from<T>(hpT, nt * 16 + l16, lgrp),
                               acc_hh[mt][nt]);
      if (l0 && CIN1) {
        acc_ih[mt][0] = wmfma(a, frag_from<T>(hxT, l16, lgrp), acc_ih[mt][0]);
      } else {
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt)
          acc_ih[mt][nt] = wmfma(a, frag_from<T>(hxT, nt * 16 + l16, lgrp),
                                 acc_ih[mt][nt]);
      }
    }
    __syncthreads();  // WAR: next commit overwrites the tiles
    if (more) {
      va0 = na0; va1 = na1; vh0 = nh0; vh1 = nh1; vx0 = nx0; vx1 = nx1;
    }
  }

  // ---- writeback: one f32 atomic per output element per workgroup --------
  float* whh = dwhh + (long)layer * 256 * 64;
  float* wih = dwih + (long)layer * 256 * 64;
  #pragma unroll
  for (int mt = 0; mt < MTG; ++mt)
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt)
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int m = g0 + wv * 16 * MTG + mt * 16 + lgrp * 4 + rr;
        const int n = nt * 16 + l16;
        unsafeAtomicAdd(&whh[m * 64 + n], acc_hh[mt][nt][rr]);
        if (!(l0 && CIN1) || nt == 0)
          unsafeAtomicAdd(&wih[m * 64 + n], acc_ih[mt][nt][rr]);
      }

  // ---- db: LDS cross-thread reduction, then atomics ----------------------
  __syncthreads();      // all MFMA reads of dAT done before red reuses it
  #pragma unroll
  for (int s_ = 0; s_ < GU / 256; ++s_)
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      red[(s_ * 256 + threadIdx.x) * 8 + j] = dbp[s_][j];
  __syncthreads();
  if (threadIdx.x < GT) {
    const int g = threadIdx.x;             // local gate within the slice
    const int gb = g >> 3, j = g & 7;      // staging col-block of gate g
    // contributors: threads whose unit had cb == gb (16 pr values)
    const int set = gb >> 4, base = (gb & 15) * 16;
    float v = 0.f;
    #pragma unroll
    for (int k = 0; k < 16; ++k) v += red[(set * 256 + base + k) * 8 + j];
    unsafeAtomicAdd(&db[layer * 256 + g0 + g], v);
  }
}

extern "C" void stmgcn_lstm_wgrad(void* stream_v, int dtype, const void* dA,
                                  const void* hseq, const void* x, float* dwih,
                                  float* dwhh, float* db, long R, long S_pad,
                                  int S, int Tst, int L, int cin) {
  static long env_chunks = -1;   // STMGCN_WGRAD_CHUNKS: perf experiments
  if (env_chunks < 0) {
    const char* e = getenv("STMGCN_WGRAD_CHUNKS");
    env_chunks = e ? atol(e) : 0;
  }
  constexpr int GT = 128;           // gate-slice width (grid.z = 256/GT)
  const long target = env_chunks > 0 ? env_chunks : 512 / (L > 0 ? L : 1);
  long nchunks = (R + 1023) / 1024;
  if (nchunks > target) nchunks = target;
  if (nchunks < 1) nchunks = 1;
  const dim3 grid((unsigned)nchunks, L, 256 / GT), blk(256);
  const size_t lds = GT * 64 + 4096 + 4096;
  hipStream_t stream = (hipStream_t)stream_v;
  if (dtype == STM_BF16) {
    if (cin == 1)
      hipLaunchKernelGGL((lstm_wgrad_kernel<__hip_bfloat16, true, GT>), grid, blk,
                         lds, stream, (const __hip_bfloat16*)dA,
                         (const __hip_bfloat16*)hseq, (const __hip_bfloat16*)x,
                         dwih, dwhh, db, R, S_pad, S, Tst, L);
    else
      hipLaunchKernelGGL((lstm_wgrad_kernel<__hip_bfloat16, false, GT>), grid, blk,
                         lds, stream, (const __hip_bfloat16*)dA,
                         (const __hip_bfloat16*)hseq, (const __hip_bfloat16*)x,
                         dwih, dwhh, db, R, S_pad, S, Tst, L);
  } else if (dtype == STM_F16) {
    if (cin == 1)
      hipLaunchKernelGGL((lstm_wgrad_kernel<__half, true, GT>), grid, blk, lds,
                         stream, (const __half*)dA, (const __half*)hseq,
                         (const __half*)x, dwih, dwhh, db, R, S_pad, S, Tst, L);
    else
      hipLaunchKernelGGL((lstm_wgrad_kernel<__half, false, GT>), grid, blk, lds,
                         stream, (const __half*)dA, (const __half*)hseq,
                         (const __half*)x, dwih, dwhh, db, R, S_pad, S, Tst, L);
  }
}

// ===========================================================================
// Generic reduction GEMM: C[M,N] += A[rows,M]^T @ B[rows,N], optional
// db[N] += colsum(B). M <= 256, N <= 64 (zero-padded to 16-multiples in LDS).
// Serves the graph-conv dW = feat^T @ dZ and db = colsum(dZ) (SURVEY K2
// backward) — measured 465 us each as hipBLASLt calls.
template <typename T>
__global__ void __launch_bounds__(256, 1)
atb_wgrad_kernel(const T* __restrict__ A, const T* __restrict__ B,
                 float* __restrict__ C, float* __restrict__ db,
                 long rows, int M, int N) {
  using frag = typename WFrag8<T>::type;
  extern __shared__ char lds[];
  char* AT = lds;                  // [256][32] -> 16 KiB
  char* BT = lds + 16384;         // [64][32]  ->  4 KiB
  float* red = (float*)lds;
  const int wv = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int l16 = lane & 15, lgrp = lane >> 4;
  const long chunk = (rows + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * chunk;
  const long r1 = (r0 + chunk < rows) ? r0 + chunk : rows;
  const int mtiles = (M + 15) / 16;       // <= 16
  const int ntiles = (N + 15) / 16;       // <= 4
  // wave wv handles m-tiles wv, wv+4, wv+8, wv+12
  const int mt_mine = (mtiles - wv + 3) / 4;

  f32x4 acc[4][4];
  #pragma unroll
  for (int i = 0; i < 4; ++i)
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc[i][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
  float dbp[8];
  #pragma unroll
  for (int j = 0; j < 8; ++j) dbp[j] = 0.f;

  for (long kt = r0; kt < r1; kt += 32) {
    stage_tileT<T, 512>(AT, A, M, kt, r1, M, nullptr);
    stage_tileT<T, 128>(BT, B, N, kt, r1, N, db ? dbp : nullptr);
    __syncthreads();
    for (int i = 0; i < mt_mine; ++i) {
      const int mt = wv + i * 4;
      const frag a = frag_from<T>(AT, mt * 16 + l16, lgrp);
      for (int nt = 0; nt < ntiles; ++nt)
        acc[i][nt] = wmfma(a, frag_from<T>(BT, nt * 16 + l16, lgrp), acc[i][nt]);
    }
    __syncthreads();
  }

  for (int i = 0; i < mt_mine; ++i) {
    const int mt = wv + i * 4;
    for (int nt = 0; nt < ntiles; ++nt)
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int m = mt * 16 + lgrp * 4 + rr, n = nt * 16 + l16;
        if (m < M && n < N) unsafeAtomicAdd(&C[(long)m * N + n], acc[i][nt][rr]);
      }
  }

  if (db) {
    // B-staging thread tid < 128 held col-block tid >> 4
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      red[threadIdx.x * 8 + j] = (threadIdx.x < 128) ? dbp[j] : 0.f;
    __syncthreads();
    if (threadIdx.x < 64 && threadIdx.x < N) {
      const int n = threadIdx.x, cb = n >> 3, j = n & 7;
      float v = 0.f;
      #pragma unroll
      for (int k = 0; k < 16; ++k) v += red[(cb * 16 + k) * 8 + j];
      unsafeAtomicAdd(&db[n], v);
    }
  }
}

extern "C" void stmgcn_atb_wgrad(void* stream_v, int dtype, const void* A,
                                 const void* B, float* C, float* db, long rows,
                                 int M, int N) {
  // fill the 256-CU chip: ~128 rows per workgroup minimum (4 K-tiles)
  long nchunks = (rows + 127) / 128;
  if (nchunks > 512) nchunks = 512;
  if (nchunks < 1) nchunks = 1;
  const dim3 grid((unsigned)nchunks), blk(256);
  const size_t lds = 16384 + 4096;
  hipStream_t stream = (hipStream_t)stream_v;
  if (dtype == STM_BF16)
    hipLaunchKernelGGL((atb_wgrad_kernel<__hip_bfloat16>), grid, blk, lds,
                       stream, (const __hip_bfloat16*)A,
                       (const __hip_bfloat16*)B, C, db, rows, M, N);
  else if (dtype == STM_F16)
    hipLaunchKernelGGL((atb_wgrad_kernel<__half>), grid, blk, lds, stream,
                       (const __half*)A, (const __half*)B, C, db, rows, M, N);
  else
    printf("stmgcn_atb_wgrad: unsupported dtype %d (bf16/f16 only)\n", dtype);
}
