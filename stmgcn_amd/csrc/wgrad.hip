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

// Transposed-tile LDS addressing: column g (output dim), row byte offset off.
// 32-row tiles (64 B/col): XOR displaces the 16 B fragment window so the MFMA
// reads (16 consecutive g, 16 B each) hit 64 distinct banks; 64-row tiles
// (128 B/col) use row parity for the 32-bank half + ((g>>1)&7) window perm.
__device__ __forceinline__ int tswz(int g, int off) {
  return g * 64 + (off ^ ((((unsigned)g >> 2) & 3) << 4));
}
__device__ __forceinline__ int tswz64(int g, int off) {
  return g * 128 + (off ^ ((((unsigned)g >> 1) & 7) << 4));
}
template <typename T>
__device__ __forceinline__ typename WFrag8<T>::type frag_from64(
    const char* lds_tile, int g, int koff_bytes) {
  return *(const typename WFrag8<T>::type*)&lds_tile[tswz64(g, koff_bytes)];
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
// L0CIN1 instantiation covers ONLY layer 0 of the scalar-input (C_in == 1)
// model: its dw_ih is a single column -> acc_ih collapses to one n-tile,
// which keeps the unified register count at 2 waves/SIMD (the combined
// variant compiled to 268 regs -> occupancy 1). layer = blockIdx.y + lbase.
template <typename T, bool L0CIN1, int GT>
__global__ void __launch_bounds__(256, 1)
lstm_wgrad_kernel(const T* __restrict__ dA, const T* __restrict__ hseq,
                  const T* __restrict__ x, float* __restrict__ dwih,
                  float* __restrict__ dwhh, float* __restrict__ db,
                  long R, long S_pad, int S, int Tst, int L, int lbase) {
  using frag = typename WFrag8<T>::type;
  constexpr int KT = 64;            // K-tile rows (PMC: 32-row tiles left the
                                    // waves 65% parked on load latency —
                                    // 2 MFMAs per frag per barrier now)
  constexpr int MTG = GT / 64;      // gate m-tiles per wave (GT=128 -> 2)
  constexpr int GU = (GT / 8) * (KT / 2);  // dA staging units per K-tile
  extern __shared__ char lds[];
  char* dAT = lds;                        // [GT][KT] T
  char* hpT = lds + GT * 2 * KT;          // [64][KT] T
  char* hxT = hpT + 64 * 2 * KT;          // [64][KT] T
  float* red = (float*)lds;               // db reduction scratch (reuses dAT)

  const int layer = blockIdx.y + lbase;
  const int g0 = blockIdx.z * GT;      // gate-row slice of this WG
  const int wv = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int l16 = lane & 15, lgrp = lane >> 4;
  const long chunk = (R + gridDim.x - 1) / gridDim.x;
  const long r0 = (long)blockIdx.x * chunk;
  const long r1 = (r0 + chunk < R) ? r0 + chunk : R;

  const T* dA_l = dA + (long)layer * R * 256;
  const T* hp_l = hseq + (long)layer * R * 64;       // index r - S_pad
  const T* hx_l = (layer > 0) ? hseq + (long)(layer - 1) * R * 64 : nullptr;
  const bool l0 = L0CIN1 || (layer == 0);
  constexpr int IHNT = L0CIN1 ? 1 : 4;   // dw_ih n-tiles

  f32x4 acc_hh[MTG][4], acc_ih[MTG][IHNT];
  #pragma unroll
  for (int mt = 0; mt < MTG; ++mt) {
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc_hh[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int nt = 0; nt < IHNT; ++nt) acc_ih[mt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
  }
  float dbp[GU / 256][8];
  #pragma unroll
  for (int s_ = 0; s_ < GU / 256; ++s_)
    #pragma unroll
    for (int j = 0; j < 8; ++j) dbp[s_][j] = 0.f;

  // Software-pipelined main loop: global loads for tile k+1 issue right
  // after the barrier, overlapping tile k's MFMA phase.
  static_assert(GU == 512, "two dA units per thread assumed by the pipeline");
  const int pr = threadIdx.x & 31, cb = threadIdx.x >> 5;   // dA unit 0 (of 2)
  const int pr2 = pr, cb2 = cb + 8;                         // dA unit 1
  const int hcb = cb;                                       // hp/hx unit (8 blocks)
  const bool hthread = true;  // all 256 threads carry one hp+hx unit

  frag va[2][2], vh[2], vx[2];     // [unit][row-pair half] / [row-pair half]
  auto load_tiles = [&](long kt, frag a[2][2], frag h[2], frag xf[2]) {
    const frag fz = {};
    const long ra = kt + pr * 2, rb = ra + 1;
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      const int cbu = (u == 0) ? cb : cb2;
      a[u][0] = fz; a[u][1] = fz;
      if (ra < r1) a[u][0] = *(const frag*)&dA_l[ra * 256 + g0 + cbu * 8];
      if (rb < r1) a[u][1] = *(const frag*)&dA_l[rb * 256 + g0 + cbu * 8];
    }
    h[0] = fz; h[1] = fz; xf[0] = fz; xf[1] = fz;
    // h_{t-1}: hseq[l] offset -S_pad rows, zero for t==0 (the guard must
    // gate the LOAD — for layer 0, r - S_pad points before the allocation)
    if (ra < r1 && ra >= S_pad)
      h[0] = *(const frag*)&hp_l[(ra - S_pad) * 64 + hcb * 8];
    if (rb < r1 && rb >= S_pad)
      h[1] = *(const frag*)&hp_l[(rb - S_pad) * 64 + hcb * 8];
    if (!l0) {
      if (ra < r1) xf[0] = *(const frag*)&hx_l[ra * 64 + hcb * 8];
      if (rb < r1) xf[1] = *(const frag*)&hx_l[rb * 64 + hcb * 8];
    } else if (L0CIN1) {
      if (hcb == 0) {
        const long sa = ra % S_pad, ta = ra / S_pad;
        const long sb = rb % S_pad, tb = rb / S_pad;
        if (ra < r1 && sa < S) ((T*)&xf[0])[0] = x[sa * Tst + ta];
        if (rb < r1 && sb < S) ((T*)&xf[1])[0] = x[sb * Tst + tb];
      }
    } else {
      if (ra < r1) {
        const long s_ = ra % S_pad, t_ = ra / S_pad;
        if (s_ < S) xf[0] = *(const frag*)&x[(s_ * Tst + t_) * 64 + hcb * 8];
      }
      if (rb < r1) {
        const long s_ = rb % S_pad, t_ = rb / S_pad;
        if (s_ < S) xf[1] = *(const frag*)&x[(s_ * Tst + t_) * 64 + hcb * 8];
      }
    }
  };
  auto commit_tiles = [&]() {
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      const int cbu = (u == 0) ? cb : cb2;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { T t2[2]; int i; } pk;
        pk.t2[0] = ((const T*)&va[u][0])[j];
        pk.t2[1] = ((const T*)&va[u][1])[j];
        *(int*)&dAT[tswz64(cbu * 8 + j, pr * 4)] = pk.i;
        dbp[u][j] += toF<T>(pk.t2[0]) + toF<T>(pk.t2[1]);
      }
    }
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      union { T t2[2]; int i; } pk;
      pk.t2[0] = ((const T*)&vh[0])[j];
      pk.t2[1] = ((const T*)&vh[1])[j];
      *(int*)&hpT[tswz64(hcb * 8 + j, pr * 4)] = pk.i;
      pk.t2[0] = ((const T*)&vx[0])[j];
      pk.t2[1] = ((const T*)&vx[1])[j];
      *(int*)&hxT[tswz64(hcb * 8 + j, pr * 4)] = pk.i;
    }
  };

  load_tiles(r0, va, vh, vx);
  for (long kt = r0; kt < r1; kt += KT) {
    commit_tiles();
    __syncthreads();
    frag na[2][2], nh[2], nx[2];
    const bool more = kt + KT < r1;
    if (more) load_tiles(kt + KT, na, nh, nx);

    // ---- MFMA: wave wv owns gate rows g0 + [wv*16*MTG, ...); two K-halves
    #pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      const int koff = kh * KT + lgrp * 16;   // byte offset of the K half
      #pragma unroll
      for (int mt = 0; mt < MTG; ++mt) {
        const frag a = frag_from64<T>(dAT, wv * 16 * MTG + mt * 16 + l16, koff);
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt)
          acc_hh[mt][nt] = wmfma(a, frag_from64<T>(hpT, nt * 16 + l16, koff),
                                 acc_hh[mt][nt]);
        #pragma unroll
        for (int nt = 0; nt < IHNT; ++nt)
          acc_ih[mt][nt] = wmfma(a, frag_from64<T>(hxT, nt * 16 + l16, koff),
                                 acc_ih[mt][nt]);
      }
    }
    __syncthreads();  // WAR: next commit overwrites the tiles
    if (more) {
      #pragma unroll
      for (int u = 0; u < 2; ++u) {
        va[u][0] = na[u][0]; va[u][1] = na[u][1];
      }
      vh[0] = nh[0]; vh[1] = nh[1]; vx[0] = nx[0]; vx[1] = nx[1];
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
        if (nt < IHNT)
          unsafeAtomicAdd(&wih[m * 64 + n], acc_ih[mt][nt < IHNT ? nt : 0][rr]);
      }

  // ---- db: LDS cross-thread reduction, then atomics ----------------------
  // unit u of thread tid covered gate block (tid>>5) + 8u, rows pr = tid&31
  __syncthreads();      // all MFMA reads of dAT done before red reuses it
  #pragma unroll
  for (int u = 0; u < 2; ++u)
    #pragma unroll
    for (int j = 0; j < 8; ++j)
      red[(u * 256 + threadIdx.x) * 8 + j] = dbp[u][j];
  __syncthreads();
  if (threadIdx.x < GT) {
    const int g = threadIdx.x;             // local gate within the slice
    const int gb = g >> 3, j = g & 7;      // staging col-block of gate g
    // contributors: 32 threads tid = (gb & 7) * 32 + pr, unit set gb >> 3
    const int set = gb >> 3, base = (gb & 7) * 32;
    float v = 0.f;
    #pragma unroll
    for (int k = 0; k < 32; ++k) v += red[(set * 256 + base + k) * 8 + j];
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
  const size_t lds = (size_t)GT * 128 + 8192 + 8192;  // 64-row tiles
  hipStream_t stream = (hipStream_t)stream_v;
  const dim3 blk(256);
  auto launch = [&](auto tptr) {
    using TT = std::remove_pointer_t<decltype(tptr)>;
    if (cin == 1) {
      // layer 0 (single ih column) and layers >= 1 as separate instantiations
      hipLaunchKernelGGL((lstm_wgrad_kernel<TT, true, GT>),
                         dim3((unsigned)nchunks, 1, 256 / GT), blk, lds, stream,
                         (const TT*)dA, (const TT*)hseq, (const TT*)x,
                         dwih, dwhh, db, R, S_pad, S, Tst, L, 0);
      if (L > 1)
        hipLaunchKernelGGL((lstm_wgrad_kernel<TT, false, GT>),
                           dim3((unsigned)nchunks, L - 1, 256 / GT), blk, lds,
                           stream, (const TT*)dA, (const TT*)hseq, (const TT*)x,
                           dwih, dwhh, db, R, S_pad, S, Tst, L, 1);
    } else {
      hipLaunchKernelGGL((lstm_wgrad_kernel<TT, false, GT>),
                         dim3((unsigned)nchunks, L, 256 / GT), blk, lds, stream,
                         (const TT*)dA, (const TT*)hseq, (const TT*)x,
                         dwih, dwhh, db, R, S_pad, S, Tst, L, 0);
    }
  };
  if (dtype == STM_BF16) launch((__hip_bfloat16*)nullptr);
  else if (dtype == STM_F16) launch((__half*)nullptr);
}

// ===========================================================================
// Generic reduction GEMM: C[M,N] += A[rows,M]^T @ B[rows,N], optional
// db[N] += colsum(B). M <= 256, N <= 64 (zero-padded to 16-multiples in LDS).
// Serves the graph-conv dW = feat^T @ dZ and db = colsum(dZ) (SURVEY K2
// backward) — measured 465 us each as hipBLASLt calls.
//
// Multi-source form: A is up to 4 separate (rows, CA) tensors laid out as
// the column blocks of a virtual (rows, M = K*CA) matrix — the fused
// ChebConv wgrad passes its recurrence states [x, p_1, .., p_{K_s-1}] so
// ALL supports' dW_k land in one launch with dZ streamed ONCE (the
// (B,N,K_s,C) concat stack never exists). CA must be a multiple of 8 so
// every 8-col fragment stays within one source.
struct AtbSrc { const void* a[4]; };

template <typename T>
__global__ void __launch_bounds__(256, 1)
atb_wgrad_kernel(AtbSrc As, int CA, const T* __restrict__ B,
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

  // register-prefetch pipeline (same rationale as lstm_wgrad_kernel): A has
  // up to 2 units/thread (unit u: pair pr = u & 15, col-block cb = u >> 4),
  // B one unit for tid < 128
  const int pr = threadIdx.x & 15;
  const int cba = threadIdx.x >> 4, cba2 = cba + 16;  // A col-blocks
  const int cbb = cba & 7;                            // B col-block (tid<128)
  // resolve this thread's two col-blocks to (source tensor, column offset)
  const T* Aa = nullptr;
  const T* Ab = nullptr;
  int oa = 0, ob = 0;
  if (cba * 8 < M) { Aa = (const T*)As.a[(cba * 8) / CA]; oa = (cba * 8) % CA; }
  if (cba2 * 8 < M) { Ab = (const T*)As.a[(cba2 * 8) / CA]; ob = (cba2 * 8) % CA; }
  const frag fz = {};
  frag va[2][2], vb[2];
  auto load_tiles = [&](long kt, frag a[2][2], frag b[2]) {
    const long ra = kt + pr * 2, rb = ra + 1;
    a[0][0] = fz; a[0][1] = fz; a[1][0] = fz; a[1][1] = fz;
    if (Aa != nullptr) {
      if (ra < r1) a[0][0] = *(const frag*)&Aa[ra * CA + oa];
      if (rb < r1) a[0][1] = *(const frag*)&Aa[rb * CA + oa];
    }
    if (Ab != nullptr) {
      if (ra < r1) a[1][0] = *(const frag*)&Ab[ra * CA + ob];
      if (rb < r1) a[1][1] = *(const frag*)&Ab[rb * CA + ob];
    }
    b[0] = fz; b[1] = fz;
    if (threadIdx.x < 128 && cbb * 8 < N) {
      if (ra < r1) b[0] = *(const frag*)&B[ra * N + cbb * 8];
      if (rb < r1) b[1] = *(const frag*)&B[rb * N + cbb * 8];
    }
  };
  auto commit_tiles = [&]() {
    #pragma unroll
    for (int u = 0; u < 2; ++u) {
      const int cb = (u == 0) ? cba : cba2;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { T t2[2]; int i; } pk;
        pk.t2[0] = ((const T*)&va[u][0])[j];
        pk.t2[1] = ((const T*)&va[u][1])[j];
        *(int*)&AT[tswz(cb * 8 + j, pr * 4)] = pk.i;
      }
    }
    if (threadIdx.x < 128) {
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { T t2[2]; int i; } pk;
        pk.t2[0] = ((const T*)&vb[0])[j];
        pk.t2[1] = ((const T*)&vb[1])[j];
        *(int*)&BT[tswz(cbb * 8 + j, pr * 4)] = pk.i;
        if (db) dbp[j] += toF<T>(pk.t2[0]) + toF<T>(pk.t2[1]);
      }
    }
  };

  load_tiles(r0, va, vb);
  for (long kt = r0; kt < r1; kt += 32) {
    commit_tiles();
    __syncthreads();
    frag na[2][2], nb[2];
    const bool more = kt + 32 < r1;
    if (more) load_tiles(kt + 32, na, nb);
    for (int i = 0; i < mt_mine; ++i) {
      const int mt = wv + i * 4;
      const frag a = frag_from<T>(AT, mt * 16 + l16, lgrp);
      for (int nt = 0; nt < ntiles; ++nt)
        acc[i][nt] = wmfma(a, frag_from<T>(BT, nt * 16 + l16, lgrp), acc[i][nt]);
    }
    __syncthreads();
    if (more) {
      va[0][0] = na[0][0]; va[0][1] = na[0][1];
      va[1][0] = na[1][0]; va[1][1] = na[1][1];
      vb[0] = nb[0]; vb[1] = nb[1];
    }
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

extern "C" void stmgcn_atb_wgrad_multi(void* stream_v, int dtype,
                                       const void** As, int nsrc, int CA,
                                       const void* B, float* C, float* db,
                                       long rows, int N) {
  // fill the 256-CU chip: ~128 rows per workgroup minimum (4 K-tiles)
  long nchunks = (rows + 127) / 128;
  if (nchunks > 512) nchunks = 512;
  if (nchunks < 1) nchunks = 1;
  const dim3 grid((unsigned)nchunks), blk(256);
  const size_t lds = 16384 + 4096;
  const int M = nsrc * CA;
  AtbSrc src{};
  for (int i = 0; i < nsrc && i < 4; ++i) src.a[i] = As[i];
  hipStream_t stream = (hipStream_t)stream_v;
  if (dtype == STM_BF16)
    hipLaunchKernelGGL((atb_wgrad_kernel<__hip_bfloat16>), grid, blk, lds,
                       stream, src, CA, (const __hip_bfloat16*)B, C, db, rows,
                       M, N);
  else if (dtype == STM_F16)
    hipLaunchKernelGGL((atb_wgrad_kernel<__half>), grid, blk, lds, stream,
                       src, CA, (const __half*)B, C, db, rows, M, N);
  else
    printf("stmgcn_atb_wgrad: unsupported dtype %d (bf16/f16 only)\n", dtype);
}

extern "C" void stmgcn_atb_wgrad(void* stream_v, int dtype, const void* A,
                                 const void* B, float* C, float* db, long rows,
                                 int M, int N) {
  stmgcn_atb_wgrad_multi(stream_v, dtype, &A, 1, M, B, C, db, rows, N);
}
