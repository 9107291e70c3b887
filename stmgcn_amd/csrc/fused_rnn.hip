// Persistent fused multi-layer LSTM/GRU for CDNA4 (gfx950) — SURVEY K5/K6.
//
// Replaces the reference's nn.LSTM over the flattened (B*N, T, C) sequence
// batch (STMGCN.py:21-22,47-50). One kernel runs ALL layers and ALL
// timesteps: each 256-thread workgroup owns a 32-sequence tile, keeps the
// LIVE hidden state (t-1/t ping/pong) in XOR-swizzled LDS for conflict-free
// ds_read_b128 MFMA fragment reads, computes the recurrent + input
// projections with v_mfma_f32_16x16x32_bf16 matrix cores (fp32 accumulate),
// and applies the gate nonlinearities in the MFMA accumulator fragment
// layout. The cross-LAYER sequence hand-off rides hseq_g in global memory
// (mandatory for the wgrad kernel anyway), with the input fragments
// prefetched one stage ahead.
//
// Geometry per workgroup (H = 64 fixed):
//   4 waves; wave w owns gate-channel slice [16w, 16w+16) of each of the
//   4 (LSTM) / 3 (GRU) gate types -> i/f/g/o for one (seq,channel) pair land
//   in the SAME lane's accumulators, making the cell update lane-local.
//   M = 64 sequences = 4 MFMA row-tiles; K = H (+C for layers > 0) chunked
//   by 32. Weights are loaded once per layer into VGPR B-fragments
//   (contiguous 16B per lane from the (4H, H|C) row-major weight).
//
// Saves for BPTT (training): per (layer, t): post-activation gates + cell
// state in fragment-native order (coalesced 16B/lane), and the hidden
// sequence in natural (S_pad, H) order staged through LDS.
//
// fp32/fp64 paths are not served here: bf16/f16 in, fp32 accumulate.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(8))) _Float16 f16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// 32-row sequence tiles: MT=2 m-tiles/wave halves per-wave register state
// vs the original 64-row tiles (fwd was 230 VGPR + 68 AGPR -> occupancy 1
// wave/SIMD; 32-row tiles fit 2 workgroups/CU, hiding LDS/MFMA latency).
#define SEQ_TILE 32
#define RNN_H 64
#define MAX_LAYERS 8
#define MAX_T 16

struct RnnPtrs {
  const void* w_ih[MAX_LAYERS];   // (G*H, C_l) row-major, model dtype
  const void* w_hh[MAX_LAYERS];   // (G*H, H)
  const void* b_ih[MAX_LAYERS];   // (G*H,) model dtype
  const void* b_hh[MAX_LAYERS];   // (G*H,) model dtype
};

// (LDS swizzle lds_swz for 128 B tile rows lives in common.h — shared with
// the fused ChebConv kernels.)

template <typename T> struct Frag8;
template <> struct Frag8<__hip_bfloat16> { using type = bf16x8; using elem = __bf16; };
template <> struct Frag8<__half> { using type = f16x8; using elem = _Float16; };

__device__ __forceinline__ float elemF(__bf16 v) { return (float)v; }
__device__ __forceinline__ float elemF(_Float16 v) { return (float)v; }

__device__ __forceinline__ f32x4 mfma16x16x32(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ f32x4 mfma16x16x32(f16x8 a, f16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
}

// ---------------------------------------------------------------------------
// MFMA layout probe (validated on hardware by tests/test_gpu_kernels.py):
// D[16,16] = A[16,32] @ B[32,16] with the assumed fragment maps:
//   A: lane l holds A[l%16][(l/16)*8 + j], j = 0..7
//   B: lane l holds B[(l/16)*8 + j][l%16]
//   C/D: lane l holds D[(l/16)*4 + r][l%16], r = 0..3
extern "C" __global__ void mfma_probe_kernel(const __hip_bfloat16* A,
                                             const __hip_bfloat16* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  #pragma unroll
      for (int j = 0; j < 8; ++j) {
    a[j] = *(const __bf16*)&A[(l % 16) * 32 + (l / 16) * 8 + j];
    b[j] = *(const __bf16*)&B[((l / 16) * 8 + j) * 16 + (l % 16)];
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = mfma16x16x32(a, b, c);
  #pragma unroll
      for (int r = 0; r < 4; ++r) D[((l / 16) * 4 + r) * 16 + (l % 16)] = c[r];
}

// ---------------------------------------------------------------------------
// Fused LSTM/GRU forward.
//
// Template: T = element type (bf16/f16), CIN1 = first-layer input_dim == 1
// (base ST-MGCN; otherwise C_in == H, the stacked deep-variant blocks),
// GRU = cell type (the deep variant, BASELINE configs[3]).
//
// GRU rides the 4-gate MFMA structure via host-side weight packing: the
// 4 q-slots carry [r | z | n_input | n_hidden] with
//   W_ih_packed = [Wr; Wz; Wn; 0],  W_hh_packed = [Ur; Uz; 0; Un],
//   bias_packed = [bir+bhr; biz+bhz; bin; bhn]
// so acc[2] = Wn x + bin and acc[3] = Un h + bhn arrive separately and the
// pointwise phase computes n = tanh(acc2 + r*acc3), h' = (1-z) n + z h_prev.
// The c_state register carry doubles as the GRU h_prev; the cseq training
// save holds h_{t-1} (GRU) instead of c_t (LSTM). Everything else — LDS
// staging, MFMA tiling, dA streaming, the batched wgrad kernel — is shared.
//
// LDS holds only the LIVE state: two ping/pong h slots (t-1 and t of the
// CURRENT layer, parity-indexed). The cross-layer sequence hand-off goes
// through hseq_g in GLOBAL memory — in training those writes were already
// mandatory (the wgrad kernel consumes hseq_g), so the hand-off is free and
// LDS drops from (2*Tst+eps) slots (64 KB at T=8 -> 2 workgroups/CU) to
// 2 slots (~8.5 KB -> occupancy is register-bound at 4 waves/SIMD instead).
// CIN1 additionally stages the scalar input sequence ([Tst][ST] T).
template <typename T, bool CIN1, bool GRU, int ST>
__global__ void __launch_bounds__(256, 2)
lstm_fwd_kernel(const T* __restrict__ x,    // (S, Tst, C_in)
                T* __restrict__ out,        // (S, H) or (S, Tst, H)
                T* __restrict__ hseq_g,     // (L, Tst, S_pad, H) — ALWAYS
                                            // (layer hand-off + wgrad save)
                T* __restrict__ cseq_g,     // (L, Tst, S_pad*H) frag-native
                                            // (LSTM c_t / GRU h_{t-1})
                T* __restrict__ gates_g,    // (L, Tst, S_pad*4H) frag-native
                RnnPtrs ptrs,
                int S, int Tst, int L, int ret_seq) {
  using frag = typename Frag8<T>::type;
  constexpr int MT = ST / 16;        // MFMA row-tiles per wave
  constexpr int SLOT = ST * 128;     // one timestep LDS slot, bytes
  extern __shared__ char lds[];
  const int s0 = blockIdx.x * ST;
  const int wv = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lgrp = lane >> 4;
  const long S_pad = (long)gridDim.x * ST;

  // ping/pong h slots by t parity (explicit ternary: an LDS-pointer array
  // initializer is rejected by the AMDGPU backend)
  auto hslot = [&](int par) -> char* { return lds + (par & 1) * SLOT; };
  char* xbuf = lds + 2 * SLOT;                 // CIN1 scalar input stage

  // ---- stage scalar input x into LDS (CIN1 only) -------------------------
  if (CIN1) {
    for (int i = threadIdx.x; i < ST * Tst; i += 256) {
      const int s = i % ST, t = i / ST;
      T v = fromF<T>(0.f);
      if (s0 + s < S) v = x[(long)(s0 + s) * Tst + t];
      ((T*)xbuf)[t * ST + s] = v;
    }
    __syncthreads();
  }

  const int hch = 16 * wv + l16;                      // this lane's h channel

  for (int layer = 0; layer < L; ++layer) {
    const T* Whh = (const T*)ptrs.w_hh[layer];
    const T* Wih = (const T*)ptrs.w_ih[layer];
    const T* bih = (const T*)ptrs.b_ih[layer];
    const T* bhh = (const T*)ptrs.b_hh[layer];

    // B-fragments for the recurrent GEMM: b_hh[q][kk]; W row g = q*64 + hch,
    // cols kk*32 + lgrp*8 .. +8 -> contiguous 16B in the (4H, H) weight.
    frag bhfrag[4][2];
    #pragma unroll
    for (int q = 0; q < 4; ++q)
      #pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bhfrag[q][kk] = *(const frag*)&Whh[(q * 64 + hch) * RNN_H + kk * 32 + lgrp * 8];
    // input-projection B-fragments (dense layers only)
    frag bxfrag[4][2];
    if (!CIN1 || layer > 0)
      #pragma unroll
      for (int q = 0; q < 4; ++q)
        #pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bxfrag[q][kk] = *(const frag*)&Wih[(q * 64 + hch) * RNN_H + kk * 32 + lgrp * 8];
    // layer-0 scalar input weights (CIN1): W_ih[g][0]
    float wih0[4];
    float bias[4];
    #pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int g = q * 64 + hch;
      bias[q] = toF<T>(bih[g]) + toF<T>(bhh[g]);
      if (CIN1 && layer == 0) wih0[q] = toF<T>(Wih[g]);
    }

    // previous layer's output sequence (global hand-off)
    const T* hin = (layer > 0)
        ? hseq_g + ((long)(layer - 1) * Tst) * (S_pad * RNN_H)
        : nullptr;

    float c_state[MT][4];  // [m][reg]
    #pragma unroll
    for (int m = 0; m < MT; ++m)
      #pragma unroll
      for (int r = 0; r < 4; ++r) c_state[m][r] = 0.f;

    // input-term fragments, prefetched one stage ahead (single-buffered:
    // reloaded right after their MFMAs consume them, so the global loads
    // get a full stage of slack instead of stalling inside the MFMA loop)
    frag xfr[MT][2];
    auto load_input = [&](int t) {
      #pragma unroll
      for (int m = 0; m < MT; ++m) {
        const int row = 16 * m + l16;
        #pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          if (layer == 0) {            // dense x (S, Tst, 64): guard the tail
            xfr[m][kk] = frag{};
            if (s0 + row < S)
              xfr[m][kk] = *(const frag*)&x[((long)(s0 + row) * Tst + t) * RNN_H + kk * 32 + lgrp * 8];
          } else {
            xfr[m][kk] = *(const frag*)&hin[((long)t * S_pad + s0 + row) * RNN_H + kk * 32 + lgrp * 8];
          }
        }
      }
    };
    if (!CIN1 || layer > 0) load_input(0);

    for (int t = 0; t < Tst; ++t) {
      f32x4 acc[MT][4];  // [m][q]
      #pragma unroll
      for (int m = 0; m < MT; ++m)
        #pragma unroll
        for (int q = 0; q < 4; ++q) acc[m][q] = f32x4{0.f, 0.f, 0.f, 0.f};

      // recurrent term: h_{t-1} from the previous parity slot (zero at t==0)
      if (t > 0) {
        char* slot = hslot(t - 1);
        #pragma unroll
        for (int m = 0; m < MT; ++m) {
          const int row = 16 * m + l16;
          #pragma unroll
          for (int kk = 0; kk < 2; ++kk) {
            frag a = *(const frag*)&slot[lds_swz(row, (kk * 32 + lgrp * 8) * 2)];
            #pragma unroll
            for (int q = 0; q < 4; ++q)
              acc[m][q] = mfma16x16x32(a, bhfrag[q][kk], acc[m][q]);
          }
        }
      }
      // input term: previous layer's h from GLOBAL (L2-hot: this block wrote
      // the same tile last layer); dense layer 0 reads x directly. The
      // fragments were prefetched last stage; reload for t+1 right away.
      if (!CIN1 || layer > 0) {
        #pragma unroll
        for (int m = 0; m < MT; ++m)
          #pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            #pragma unroll
            for (int q = 0; q < 4; ++q)
              acc[m][q] = mfma16x16x32(xfr[m][kk], bxfrag[q][kk], acc[m][q]);
        if (t + 1 < Tst) load_input(t + 1);
      }

      // pointwise cell update in fragment layout (+ scalar-input term)
      T hval[MT][4];     // [m][reg] this lane's h outputs (ch = hch)
      T gsave[MT][16];   // [m][i f g o | r z n Bn][reg] post-activation gates
      float csave[MT][4];  // [m][reg] training save (LSTM c_t / GRU h_{t-1})
      #pragma unroll
      for (int m = 0; m < MT; ++m) {
        float xv[4];
        if (CIN1 && layer == 0)
          #pragma unroll
          for (int r = 0; r < 4; ++r)
            xv[r] = toF<T>(((const T*)xbuf)[t * ST + 16 * m + 4 * lgrp + r]);
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          float gi = acc[m][0][r] + bias[0];
          float gf = acc[m][1][r] + bias[1];
          float gg = acc[m][2][r] + bias[2];
          float go = acc[m][3][r] + bias[3];
          if (CIN1 && layer == 0) {
            gi += xv[r] * wih0[0]; gf += xv[r] * wih0[1];
            gg += xv[r] * wih0[2]; go += xv[r] * wih0[3];
          }
          if (GRU) {
            // q-slots: gi = r-pre, gf = z-pre, gg = n_input, go = n_hidden
            const float r_ = stm_sigmoid(gi), z_ = stm_sigmoid(gf);
            const float n_ = stm_tanh(gg + r_ * go);
            const float hprev = c_state[m][r];
            csave[m][r] = hprev;                        // bwd needs h_{t-1}
            const float h_ = (1.f - z_) * n_ + z_ * hprev;
            c_state[m][r] = h_;
            hval[m][r] = fromF<T>(h_);
            gsave[m][0 * 4 + r] = fromF<T>(r_);
            gsave[m][1 * 4 + r] = fromF<T>(z_);
            gsave[m][2 * 4 + r] = fromF<T>(n_);
            gsave[m][3 * 4 + r] = fromF<T>(go);          // Bn = Un h + bhn
          } else {
            const float i_ = stm_sigmoid(gi), f_ = stm_sigmoid(gf);
            const float g_ = stm_tanh(gg), o_ = stm_sigmoid(go);
            const float c_ = f_ * c_state[m][r] + i_ * g_;
            c_state[m][r] = c_;
            csave[m][r] = c_;
            const float h_ = o_ * stm_tanh(c_);
            hval[m][r] = fromF<T>(h_);
            gsave[m][0 * 4 + r] = fromF<T>(i_);
            gsave[m][1 * 4 + r] = fromF<T>(f_);
            gsave[m][2 * 4 + r] = fromF<T>(g_);
            gsave[m][3 * 4 + r] = fromF<T>(o_);
          }
        }
      }

      // write h_t into the parity slot: stage t-1's readers of this slot
      // (= slot (t-2)&1 == t&1) finished before the previous stage barrier
      {
        char* slot = hslot(t);
        #pragma unroll
        for (int m = 0; m < MT; ++m)
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = 16 * m + 4 * lgrp + r;
            *(T*)&slot[lds_swz(row, hch * 2)] = hval[m][r];
          }
      }
      __syncthreads();

      // hand the slot off to global (next layer's input + wgrad/out source;
      // the last layer's sequence is dead in eval — only `out` is written)
      {
        T* hp = hseq_g + ((long)layer * Tst + t) * (S_pad * RNN_H)
                + (long)s0 * RNN_H;
        const bool keep_h = (layer < L - 1) || (gates_g != nullptr);
        char* slot = hslot(t);
        for (int i = threadIdx.x; i < ST * 8; i += 256) {
          const int c8 = i & 7, sr = i >> 3;
          frag v = *(frag*)&slot[lds_swz(sr, c8 * 16)];
          if (keep_h) *(frag*)&hp[sr * RNN_H + c8 * 8] = v;
          if (layer == L - 1 && s0 + sr < S) {
            if (ret_seq)
              *(frag*)&out[((long)(s0 + sr) * Tst + t) * RNN_H + c8 * 8] = v;
            else if (t == Tst - 1)
              *(frag*)&out[(long)(s0 + sr) * RNN_H + c8 * 8] = v;
          }
        }
      }

      // training saves: gates + cell, fragment-native (16B contiguous/lane)
      if (gates_g) {
        const long base = ((long)layer * Tst + t);
        // gates: (L,Tst, S_pad*4H) as [wave][m][lane][16] T
        T* gp = gates_g + base * (S_pad * 4 * RNN_H)
                + (long)blockIdx.x * (ST * 4 * RNN_H)
                + ((wv * MT) * 64) * 16;
        #pragma unroll
        for (int m = 0; m < MT; ++m)
          *(((frag*)(gp + (m * 64 + lane) * 16)) + 0) = *(frag*)&gsave[m][0],
          *(((frag*)(gp + (m * 64 + lane) * 16)) + 1) = *(frag*)&gsave[m][8];
        // cell: (L,Tst, S_pad*H) model-dtype as [wave][m][lane][4]
        // (LSTM: c_t; GRU: h_{t-1}) — T-typed save halves HBM traffic vs
        // fp32; bwd tolerances cover the rounding (tests at 8% rel)
        T* cp = cseq_g + base * (S_pad * RNN_H)
                + (long)blockIdx.x * (ST * RNN_H) + (wv * MT) * 64 * 4;
        #pragma unroll
        for (int m = 0; m < MT; ++m) {
          T c4[4];
          #pragma unroll
          for (int r = 0; r < 4; ++r) c4[r] = fromF<T>(csave[m][r]);
          *(ulong1*)(cp + (m * 64 + lane) * 4) = *(ulong1*)c4;
        }
      }
    }
    // layer boundary: the hseq_g stores must be visible to this block's
    // next-layer input loads (workgroup-scope fence + barrier)
    __threadfence_block();
    __syncthreads();
  }
}

template <typename T>
void launch_fwd(hipStream_t stream, const void* x, void* out, void* hseq_g,
                void* cseq_g, void* gates_g, const RnnPtrs& ptrs, int S,
                int Tst, int L, int cin, int ret_seq, int gru) {
  constexpr int ST = SEQ_TILE;
  const int nblk = (S + ST - 1) / ST;
  const bool cin1 = (cin == 1);
  const size_t slot = (size_t)ST * 128;
  const size_t lds_bytes = 2 * slot + (cin1 ? Tst * ST * sizeof(T) : 0);
  auto go = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(nblk), dim3(256), lds_bytes, stream,
                       (const T*)x, (T*)out, (T*)hseq_g, (T*)cseq_g,
                       (T*)gates_g, ptrs, S, Tst, L, ret_seq);
  };
  if (cin1 && !gru) go(lstm_fwd_kernel<T, true, false, ST>);
  else if (cin1 && gru) go(lstm_fwd_kernel<T, true, true, ST>);
  else if (!gru) go(lstm_fwd_kernel<T, false, false, ST>);
  else go(lstm_fwd_kernel<T, false, true, ST>);
}

extern "C" void stmgcn_lstm_fwd(void* stream_v, int dtype, const void* x,
                                void* out, void* hseq_g, void* cseq_g,
                                void* gates_g, const void** w_ih,
                                const void** w_hh, const void** b_ih,
                                const void** b_hh, int S, int Tst, int L,
                                int cin, int ret_seq, int gru) {
  RnnPtrs p;
  for (int l = 0; l < L && l < MAX_LAYERS; ++l) {
    p.w_ih[l] = w_ih[l]; p.w_hh[l] = w_hh[l];
    p.b_ih[l] = b_ih[l]; p.b_hh[l] = b_hh[l];
  }
  hipStream_t stream = (hipStream_t)stream_v;
  if (dtype == STM_BF16)
    launch_fwd<__hip_bfloat16>(stream, x, out, hseq_g, cseq_g, gates_g, p, S,
                               Tst, L, cin, ret_seq, gru);
  else if (dtype == STM_F16)
    launch_fwd<__half>(stream, x, out, hseq_g, cseq_g, gates_g, p, S, Tst, L,
                       cin, ret_seq, gru);
}

extern "C" void stmgcn_mfma_probe(void* stream_v, const void* A, const void* B,
                                  void* D) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream_v, (const __hip_bfloat16*)A,
                     (const __hip_bfloat16*)B, (float*)D);
}

// ===========================================================================
// Fused LSTM backward, dgrad part (BPTT in-kernel) — SURVEY K10.
//
// Same workgroup geometry as forward. Layers walk top-down, timesteps in
// reverse. Per step the pointwise phase turns (dh, dc, saved gates, saved
// cell) into gate-preactivation grads dA — lane-local in the MFMA fragment
// layout — then two MFMA GEMMs produce dh_{t-1} (recurrent carry, kept in
// registers) and dx_t (written into the dh hand-off LDS slot for the layer
// below, or to the dx output at layer 0). dA is also streamed to global in
// natural (L,Tst,S_pad,4H) order; the weight gradients are then two plain
// library GEMMs per layer on the host side (dW = dA^T @ [h_prev | x]),
// which keeps this kernel lean (no long-lived dW accumulators).

__device__ __forceinline__ int swzA(int s, int cbyte) {      // dA rows: 512 B
  return s * 512 + (cbyte ^ ((s & 15) << 4));
}

// GRU backward (same packed-slot scheme as forward — see lstm_fwd_kernel):
// saved gates = [r, z, n, Bn], cseq = h_{t-1}; emits packed
// dA = [dr_pre, dz_pre, dn_pre, dBn] so the dgrad GEMMs against the packed
// transposed weights and the batched wgrad kernel run unchanged. The direct
// dh_{t-1} += dh_t * z term rides the dc[][] register carry.
template <typename T, bool CIN1, bool GRU, int ST>
__global__ void __launch_bounds__(256, 2)
lstm_bwd_kernel(const T* __restrict__ dout,     // (S,H) or (S,Tst,H)
                const T* __restrict__ x,        // (S,Tst,Cin)
                const T* __restrict__ cseq_g,   // model dtype (see fwd)
                const T* __restrict__ gates_g,
                RnnPtrs w,                      // w_ih/w_hh = TRANSPOSED (C|H, 4H)
                T* __restrict__ dx,             // (S,Tst,Cin)
                T* __restrict__ dA_g,           // (L,Tst,S_pad,4H)
                T* __restrict__ dh_g,           // (Tst,S_pad,H) layer hand-off
                                                // scratch (null when L == 1)
                int S, int Tst, int L, int ret_seq) {
  using frag = typename Frag8<T>::type;
  using elem = typename Frag8<T>::elem;
  constexpr int MT = ST / 16;
  extern __shared__ char lds[];
  const int s0 = blockIdx.x * ST;
  const int wv = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lgrp = lane >> 4;
  const long S_pad = (long)gridDim.x * ST;
  const int hch = 16 * wv + l16;

  // LDS holds only the dA tiles (ping/pong by timestep parity: the pointwise
  // writes of step t never alias the buffer step t+1's GEMMs/stream read, so
  // the loop-top WAR barrier drops) + the CIN1 dx reduction scratch. The
  // cross-layer dh hand-off lives in GLOBAL scratch dh_g — dropping the
  // [Tst] dh slots cuts LDS from ~64.5 KB (2 WGs/CU) to ~32.5 KB (4 WGs/CU),
  // doubling the occupancy that hides the serial BPTT latency.
  char* dA_buf0 = lds;                        // [ST][512B] swizzled
  char* dA_buf1 = dA_buf0 + ST * 512;
  float* red = (float*)(dA_buf1 + ST * 512);  // [4][ST] cross-wave scratch
  // lane-PRIVATE gates staging (each lane writes exactly the 2*MT fragments
  // it alone reads next stage -> no cross-wave visibility, no barrier, and
  // the prefetched tile costs LDS instead of 32 live VGPRs)
  char* gstage = (char*)(red + 4 * ST);       // [256][MT*32B]

  for (int layer = L - 1; layer >= 0; --layer) {
    const bool l0cin1 = CIN1 && layer == 0;
    const T* WhhT = (const T*)w.w_hh[layer];  // (H, 4H)
    const T* WihT = (const T*)w.w_ih[layer];  // (Cin_l, 4H)
    float dh_rec[MT][4], dc[MT][4];
    #pragma unroll
    for (int m = 0; m < MT; ++m)
      #pragma unroll
      for (int r = 0; r < 4; ++r) { dh_rec[m][r] = 0.f; dc[m][r] = 0.f; }

    // ---- training-save prefetch pipeline: gates/cseq for step t are
    // loaded during step t+1's GEMM phase (the kernel is latency-bound at
    // 2 WGs/CU; these are the only global loads in the hot loop) ----------
    const long lay_base = (long)layer * Tst;
    auto g_at = [&](int t) {
      return gates_g + (lay_base + t) * (S_pad * 4 * RNN_H)
             + (long)blockIdx.x * (ST * 4 * RNN_H) + (wv * MT) * 64 * 16;
    };
    auto c_at = [&](int t) {
      return cseq_g + (lay_base + t) * (S_pad * RNN_H)
             + (long)blockIdx.x * (ST * RNN_H) + (wv * MT) * 64 * 4;
    };
    // dh hand-off is FRAG-NATIVE: the GEMM2 writer and the pointwise reader
    // use the identical lane mapping (row 16m+4lgrp+r, col hch), so each
    // lane stores/loads its 4 values as one packed 8-byte word.
    auto dh_at = [&](int t) {
      return dh_g + (long)t * (S_pad * RNN_H)
             + (long)blockIdx.x * (ST * RNN_H) + (wv * MT) * 64 * 4;
    };
    char* gmine = gstage + threadIdx.x * (MT * 32);
    #pragma unroll
    for (int m = 0; m < MT; ++m) {
      *(frag*)&gmine[m * 32 + 0] =
          *(((const frag*)(g_at(Tst - 1) + (m * 64 + lane) * 16)) + 0);
      *(frag*)&gmine[m * 32 + 16] =
          *(((const frag*)(g_at(Tst - 1) + (m * 64 + lane) * 16)) + 1);
    }
    // upstream dh (layer above) for step t, prefetched one stage ahead so
    // the global loads never sit at the top of the pointwise critical path
    ulong1 dhup[MT];
    if (layer < L - 1) {
      #pragma unroll
      for (int m = 0; m < MT; ++m)
        dhup[m] = *(const ulong1*)(dh_at(Tst - 1) + (m * 64 + lane) * 4);
    }
    // c rotation: stage t consumes c_t and c_{t-1}; the next stage reuses
    // c_{t-1} as ITS c_t, so only ONE new (prefetched) c load per stage.
    ulong1 cc_t[MT], cc_p[MT];
    #pragma unroll
    for (int m = 0; m < MT; ++m) {
      cc_t[m] = *(const ulong1*)(c_at(Tst - 1) + (m * 64 + lane) * 4);
      cc_p[m] = ulong1{0};
      if (!GRU && Tst >= 2)
        cc_p[m] = *(const ulong1*)(c_at(Tst - 2) + (m * 64 + lane) * 4);
    }
    // layer boundary: dA-buffer parity reuse needs the readers done, and
    // this block's dh_g stores must be visible to its next-layer loads
    __threadfence_block();
    __syncthreads();

    for (int t = Tst - 1; t >= 0; --t) {
      char* dA_lds = (t & 1) ? dA_buf1 : dA_buf0;
      const long base = lay_base + t;

      float wih0[4];
      if (l0cin1) {
        #pragma unroll
        for (int q = 0; q < 4; ++q) wih0[q] = toF<T>(WihT[q * 64 + hch]);
      }
      float dxpart[MT][4];

      #pragma unroll
      for (int m = 0; m < MT; ++m) {
        frag gf0 = *(const frag*)&gmine[m * 32 + 0];                 // i|f
        frag gf1 = *(const frag*)&gmine[m * 32 + 16];                // g|o
        f32x4 ct, cpv;
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          ct[r] = elemF(((elem*)&cc_t[m])[r]);
          cpv[r] = (t > 0) ? elemF(((elem*)&cc_p[m])[r]) : 0.f;
        }
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = 16 * m + 4 * lgrp + r;
          float dh = dh_rec[m][r];
          if (layer < L - 1) {
            dh += elemF(((elem*)&dhup[m])[r]);
          } else if (ret_seq) {
            if (s0 + row < S) dh += toF<T>(dout[((long)(s0 + row) * Tst + t) * RNN_H + hch]);
          } else if (t == Tst - 1) {
            if (s0 + row < S) dh += toF<T>(dout[(long)(s0 + row) * RNN_H + hch]);
          }
          float dAi, dAf, dAg, dAo;
          if (GRU) {
            // gf0 = [r, z], gf1 = [n, Bn]; ct = h_{t-1} (fwd csave)
            const float r_ = elemF(gf0[r]);
            const float z_ = elemF(gf0[4 + r]);
            const float n_ = elemF(gf1[r]);
            const float Bn = elemF(gf1[4 + r]);
            const float hprev = ct[r];
            dh += dc[m][r];                       // direct dh_t * z carry
            const float dz_pre = dh * (hprev - n_) * z_ * (1.f - z_);
            const float dn_pre = dh * (1.f - z_) * (1.f - n_ * n_);
            const float dBn = dn_pre * r_;
            const float dr_pre = dn_pre * Bn * r_ * (1.f - r_);
            dc[m][r] = dh * z_;
            dAi = dr_pre; dAf = dz_pre; dAg = dn_pre; dAo = dBn;
          } else {
            const float i_ = elemF(gf0[r]);
            const float f_ = elemF(gf0[4 + r]);
            const float g_ = elemF(gf1[r]);
            const float o_ = elemF(gf1[4 + r]);
            const float tc = stm_tanh(ct[r]);
            float dcv = dc[m][r] + dh * o_ * (1.f - tc * tc);
            dAo = dh * tc * o_ * (1.f - o_);
            dAi = dcv * g_ * i_ * (1.f - i_);
            dAf = dcv * cpv[r] * f_ * (1.f - f_);
            dAg = dcv * i_ * (1.f - g_ * g_);
            dc[m][r] = dcv * f_;
          }
          *(T*)&dA_lds[swzA(row, (0 * 64 + hch) * 2)] = fromF<T>(dAi);
          *(T*)&dA_lds[swzA(row, (1 * 64 + hch) * 2)] = fromF<T>(dAf);
          *(T*)&dA_lds[swzA(row, (2 * 64 + hch) * 2)] = fromF<T>(dAg);
          *(T*)&dA_lds[swzA(row, (3 * 64 + hch) * 2)] = fromF<T>(dAo);
          if (l0cin1)
            dxpart[m][r] = dAi * wih0[0] + dAf * wih0[1] + dAg * wih0[2] + dAo * wih0[3];
        }
      }
      __syncthreads();   // dA visible to all waves

      // prefetch next stage's saves (overlap the GEMMs below). The gate
      // loads issue NOW but their LDS stage-in happens at the END of the
      // stage: a ds_write forces a vector-memory wait on its operand, and
      // doing that here would drain the dA stream stores of the previous
      // stage too (full HBM store latency exposed every stage — the 70%
      // wave-park of round 1). c_{t-2} and dh go to packed registers.
      ulong1 dhnext[MT], cnext[MT];
      frag gld[MT][2];
      if (t > 0) {
        #pragma unroll
        for (int m = 0; m < MT; ++m) {
          gld[m][0] = *(((const frag*)(g_at(t - 1) + (m * 64 + lane) * 16)) + 0);
          gld[m][1] = *(((const frag*)(g_at(t - 1) + (m * 64 + lane) * 16)) + 1);
        }
        // LSTM rotates c_{t-1} into c_t, so only c_{t-2} is new; GRU's
        // save is h_{t-1} (no reuse), so its next stage needs c_at(t-1)
        if (GRU || t >= 2) {
          #pragma unroll
          for (int m = 0; m < MT; ++m)
            cnext[m] = *(const ulong1*)(c_at(GRU ? t - 1 : t - 2)
                                        + (m * 64 + lane) * 4);
        }
        if (layer < L - 1) {
          #pragma unroll
          for (int m = 0; m < MT; ++m)
            dhnext[m] = *(const ulong1*)(dh_at(t - 1) + (m * 64 + lane) * 4);
        }
      }

      // ---- GEMM1: dh_prev = dA @ W_hh ------------------------------------
      // (the dA global stream moved below the GEMMs: dh_prev is the serial
      // BPTT critical path, the stream stores fill the gaps after it)
      if (t > 0) {
        f32x4 acc[MT];
        #pragma unroll
        for (int m = 0; m < MT; ++m) acc[m] = f32x4{0.f, 0.f, 0.f, 0.f};
        #pragma unroll 4
        for (int kk = 0; kk < 8; ++kk) {
          frag b = *(const frag*)&WhhT[hch * (4 * RNN_H) + kk * 32 + lgrp * 8];
          #pragma unroll
          for (int m = 0; m < MT; ++m) {
            frag a = *(const frag*)&dA_lds[swzA(16 * m + l16, (kk * 32 + lgrp * 8) * 2)];
            acc[m] = mfma16x16x32(a, b, acc[m]);
          }
        }
        #pragma unroll
        for (int m = 0; m < MT; ++m)
          #pragma unroll
          for (int r = 0; r < 4; ++r) dh_rec[m][r] = acc[m][r];
      }

      // ---- GEMM2: dx = dA @ W_ih -> dh_lds slot t / dx output ------------
      if (!l0cin1) {
        f32x4 acc[MT];
        #pragma unroll
        for (int m = 0; m < MT; ++m) acc[m] = f32x4{0.f, 0.f, 0.f, 0.f};
        #pragma unroll 4
        for (int kk = 0; kk < 8; ++kk) {
          frag b = *(const frag*)&WihT[hch * (4 * RNN_H) + kk * 32 + lgrp * 8];
          #pragma unroll
          for (int m = 0; m < MT; ++m) {
            frag a = *(const frag*)&dA_lds[swzA(16 * m + l16, (kk * 32 + lgrp * 8) * 2)];
            acc[m] = mfma16x16x32(a, b, acc[m]);
          }
        }
        if (layer > 0) {
          #pragma unroll
          for (int m = 0; m < MT; ++m) {
            T v4[4];
            #pragma unroll
            for (int r = 0; r < 4; ++r) v4[r] = fromF<T>(acc[m][r]);
            *(ulong1*)(dh_at(t) + (m * 64 + lane) * 4) = *(ulong1*)v4;
          }
        } else {
          #pragma unroll
          for (int m = 0; m < MT; ++m)
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int row = 16 * m + 4 * lgrp + r;
              if (s0 + row < S)
                dx[((long)(s0 + row) * Tst + t) * RNN_H + hch] = fromF<T>(acc[m][r]);
            }
        }
      } else {
        // CIN1 l0: dx[s,t] = sum_g dA[s,g] W_ih[g,0]
        #pragma unroll
        for (int m = 0; m < MT; ++m)
          #pragma unroll
          for (int r = 0; r < 4; ++r) {
            float v = dxpart[m][r];
            #pragma unroll
            for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, 64);
            if (l16 == 0) red[wv * ST + 16 * m + 4 * lgrp + r] = v;
          }
        __syncthreads();
        if (wv == 0) {
          for (int sA = lane; sA < ST; sA += 64) {
            const float v = red[sA] + red[ST + sA] + red[2 * ST + sA] + red[3 * ST + sA];
            if (s0 + sA < S) dx[(long)(s0 + sA) * Tst + t] = fromF<T>(v);
          }
        }
      }

      // ---- stream dA to global (natural layout) for the wgrad kernel -----
      {
        T* out_dA = dA_g + base * (S_pad * 4 * RNN_H) + (long)s0 * 4 * RNN_H;
        for (int i = threadIdx.x; i < ST * 32; i += 256) {
          const int c8 = i & 31, sA = i >> 5;   // 32 x 16B pieces per row
          *(frag*)&out_dA[(long)sA * 4 * RNN_H + c8 * 8] =
              *(const frag*)&dA_lds[swzA(sA, c8 * 16)];
        }
      }

      if (t > 0) {
        #pragma unroll
        for (int m = 0; m < MT; ++m) {
          *(frag*)&gmine[m * 32 + 0] = gld[m][0];   // loads landed long ago
          *(frag*)&gmine[m * 32 + 16] = gld[m][1];
          cc_t[m] = GRU ? cnext[m] : cc_p[m];
          if (!GRU) cc_p[m] = (t >= 2) ? cnext[m] : ulong1{0};
          if (layer < L - 1) dhup[m] = dhnext[m];
        }
      }
    }  // t loop
  }  // layer loop
}


// ===========================================================================
// WAVE-PRIVATE BPTT backward (round-2 experiment, STMGCN_BWD_WAVE=1).
//
// The classic lstm_bwd_kernel couples its 4 waves every stage: the
// pointwise phase writes dA columns that OTHER waves' GEMMs consume, so a
// __syncthreads sits on the serial BPTT critical path 24 times per block —
// measured 71% wave-park at occupancy 2 (profiles/r02_pmc_sq_waits.md).
//
// This variant gives each WAVE 16 sequences end-to-end: pointwise (in the
// MFMA D-fragment layout, which is ALSO the forward save layout, so gates/
// cell bundles arrive as coalesced 16 B fragments), a wave-private LDS
// transpose of dA into A-fragment layout, and both GEMMs over the full
// k = 4H and n = H ranges per wave. Nothing crosses waves: there are ZERO
// barriers in the kernel, the cross-layer dh hand-off is packed so the
// writing lane is the reading lane (plain vmcnt ordering suffices), and a
// 64-row block doubles the rows in flight per CU at the same occupancy.
//
// Geometry: 256 threads = 4 waves; wave w owns rows [64*blockIdx + 16w ..
// +16). D-layout per lane: rows 16w + 4*lgrp + r (r 0..3), channels
// 16*nt + l16 (nt 0..3). Forward saved its bundles from 32-row blocks with
// channel-sliced waves; the (block, m, wave, lane) coordinates of this
// wave's rows are bxf = row0/32, mf = (row0/16)&1, wv_f = nt, lane_f = lane
// — the SAME lane index, so every gates/cseq access is a direct fragment
// load of the forward's save.
template <typename T, bool CIN1, bool GRU>
__global__ void __launch_bounds__(256, 2)
lstm_bwd_wave_kernel(const T* __restrict__ dout,    // (S,H) or (S,Tst,H)
                     const T* __restrict__ cseq_g,  // fwd save (see fwd)
                     const T* __restrict__ gates_g, // fwd save
                     RnnPtrs w,                     // w_ih/w_hh TRANSPOSED (C|H, 4H)
                     T* __restrict__ dx,            // (S,Tst,Cin)
                     T* __restrict__ dA_g,          // (L,Tst,S_pad,4H) natural
                     T* __restrict__ dh_g,          // (Tst*S_pad*H) packed scratch
                     int S, int S_pad, int Tst, int L, int ret_seq) {
  using frag = typename Frag8<T>::type;
  using elem = typename Frag8<T>::elem;
  extern __shared__ char lds[];
  const int wv = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int l16 = lane & 15;
  const int lgrp = lane >> 4;
  const int row0 = blockIdx.x * 64 + 16 * wv;     // wave's first row
  if (row0 >= S_pad) return;                       // whole-wave tail guard

  char* dAw = lds + wv * 8192;                     // wave-private dA (16x512B)
  char* gmine = lds + 32768 + threadIdx.x * 128;   // lane-private gates stage

  const int bxf = row0 >> 5;                       // fwd 32-row block
  const int mf = (row0 >> 4) & 1;                  // fwd m-tile

  for (int layer = L - 1; layer >= 0; --layer) {
    const bool l0cin1 = CIN1 && layer == 0;
    const T* WhhT = (const T*)w.w_hh[layer];       // (H, 4H)
    const T* WihT = (const T*)w.w_ih[layer];       // (Cin_l, 4H)
    const long lay_base = (long)layer * Tst;
    // fwd-save bundle addresses for this lane (16 gate T / 4 cell T per nt)
    auto g_at = [&](int t, int nt) {
      return gates_g + (lay_base + t) * ((long)S_pad * 4 * RNN_H)
             + (long)bxf * (32 * 4 * RNN_H) + nt * (2 * 64 * 16)
             + (mf * 64 + lane) * 16;
    };
    auto c_at = [&](int t, int nt) {
      return cseq_g + (lay_base + t) * ((long)S_pad * RNN_H)
             + (long)bxf * (32 * RNN_H) + nt * (2 * 64 * 4)
             + (mf * 64 + lane) * 4;
    };
    // packed lane-private dh hand-off (writer lane == reader lane)
    auto dh_at = [&](int t) {
      return dh_g + (long)t * ((long)S_pad * RNN_H)
             + (long)blockIdx.x * (64 * RNN_H) + wv * (16 * RNN_H) + lane * 16;
    };

    float dh_rec[4][4], dc[4][4];                  // [nt][r], D-layout
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt)
      #pragma unroll
      for (int r = 0; r < 4; ++r) { dh_rec[nt][r] = 0.f; dc[nt][r] = 0.f; }

    float wihv[4][4];                              // [q][nt] this lane's W_ih column
    if (l0cin1)
      #pragma unroll
      for (int q = 0; q < 4; ++q)
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt)
          wihv[q][nt] = toF<T>(WihT[q * 64 + 16 * nt + l16]);

    // ---- stage-in for t = Tst-1: gates -> lane LDS, c/dh -> registers ----
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      *(frag*)&gmine[nt * 32 + 0] = *(((const frag*)g_at(Tst - 1, nt)) + 0);
      *(frag*)&gmine[nt * 32 + 16] = *(((const frag*)g_at(Tst - 1, nt)) + 1);
    }
    ulong1 cc_t[4], cc_p[4], dhup[4];
    #pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      cc_t[nt] = *(const ulong1*)c_at(Tst - 1, nt);
      cc_p[nt] = ulong1{0};
      if (!GRU && Tst >= 2) cc_p[nt] = *(const ulong1*)c_at(Tst - 2, nt);
    }
    if (layer < L - 1) {
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt)
        dhup[nt] = *(((const ulong1*)dh_at(Tst - 1)) + nt);
    }

    for (int t = Tst - 1; t >= 0; --t) {
      // ---- pointwise in D-layout; dA transposed through wave LDS --------
      float dxacc[4];                              // [r] (CIN1 l0 only)
      if (l0cin1) { dxacc[0] = dxacc[1] = dxacc[2] = dxacc[3] = 0.f; }
      #pragma unroll 1
      for (int nt = 0; nt < 4; ++nt) {             // serialized: live-range cap
        frag gf0 = *(const frag*)&gmine[nt * 32 + 0];   // [i r0..3 | f r0..3]
        frag gf1 = *(const frag*)&gmine[nt * 32 + 16];  // [g r0..3 | o r0..3]
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = row0 + 4 * lgrp + r;
          float dh = dh_rec[nt][r];
          if (layer < L - 1) {
            dh += elemF(((elem*)&dhup[nt])[r]);
          } else if (ret_seq) {
            if (row < S)
              dh += toF<T>(dout[((long)row * Tst + t) * RNN_H + 16 * nt + l16]);
          } else if (t == Tst - 1) {
            if (row < S)
              dh += toF<T>(dout[(long)row * RNN_H + 16 * nt + l16]);
          }
          const float ctv = elemF(((elem*)&cc_t[nt])[r]);
          float dAi, dAf, dAg, dAo;
          if (GRU) {
            const float r_ = elemF(gf0[r]);
            const float z_ = elemF(gf0[4 + r]);
            const float n_ = elemF(gf1[r]);
            const float Bn = elemF(gf1[4 + r]);
            const float hprev = ctv;               // GRU save = h_{t-1}
            dh += dc[nt][r];
            const float dz_pre = dh * (hprev - n_) * z_ * (1.f - z_);
            const float dn_pre = dh * (1.f - z_) * (1.f - n_ * n_);
            dAo = dn_pre * r_;                     // dBn
            dAi = dn_pre * Bn * r_ * (1.f - r_);   // dr_pre
            dAf = dz_pre;
            dAg = dn_pre;
            dc[nt][r] = dh * z_;
          } else {
            const float i_ = elemF(gf0[r]);
            const float f_ = elemF(gf0[4 + r]);
            const float g_ = elemF(gf1[r]);
            const float o_ = elemF(gf1[4 + r]);
            const float cpv = (t > 0) ? elemF(((elem*)&cc_p[nt])[r]) : 0.f;
            const float tc = stm_tanh(ctv);
            float dcv = dc[nt][r] + dh * o_ * (1.f - tc * tc);
            dAo = dh * tc * o_ * (1.f - o_);
            dAi = dcv * g_ * i_ * (1.f - i_);
            dAf = dcv * cpv * f_ * (1.f - f_);
            dAg = dcv * i_ * (1.f - g_ * g_);
            dc[nt][r] = dcv * f_;
          }
          const int rl = 4 * lgrp + r;             // wave-local row
          const int ch = 16 * nt + l16;
          *(T*)&dAw[swzA(rl, (0 * 64 + ch) * 2)] = fromF<T>(dAi);
          *(T*)&dAw[swzA(rl, (1 * 64 + ch) * 2)] = fromF<T>(dAf);
          *(T*)&dAw[swzA(rl, (2 * 64 + ch) * 2)] = fromF<T>(dAg);
          *(T*)&dAw[swzA(rl, (3 * 64 + ch) * 2)] = fromF<T>(dAo);
          if (l0cin1)
            dxacc[r] += dAi * wihv[0][nt] + dAf * wihv[1][nt]
                      + dAg * wihv[2][nt] + dAo * wihv[3][nt];
        }
      }
      // wave-private transpose: in-order LDS, no cross-wave traffic — the
      // A-fragment reads below only need the implicit lgkmcnt waits (kept
      // in LDS, re-read per GEMM: 8 extra ds_reads beat 32 live VGPRs)
      auto a_at = [&](int kk) {
        return *(const frag*)&dAw[swzA(l16, (kk * 32 + lgrp * 8) * 2)];
      };

      // ---- prefetch next stage's saves (land during the GEMMs) ----------
      frag gld[4][2];
      ulong1 cnext[4], dhnext[4];
      if (t > 0) {
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          gld[nt][0] = *(((const frag*)g_at(t - 1, nt)) + 0);
          gld[nt][1] = *(((const frag*)g_at(t - 1, nt)) + 1);
        }
        if (GRU || t >= 2) {
          #pragma unroll
          for (int nt = 0; nt < 4; ++nt)
            cnext[nt] = *(const ulong1*)c_at(GRU ? t - 1 : t - 2, nt);
        }
        if (layer < L - 1) {
          #pragma unroll
          for (int nt = 0; nt < 4; ++nt)
            dhnext[nt] = *(((const ulong1*)dh_at(t - 1)) + nt);
        }
      }

      // ---- GEMM1: dh_prev = dA @ W_hh (full k=4H, n=H per wave) ---------
      if (t > 0) {
        #pragma unroll 2
        for (int nt = 0; nt < 4; ++nt) {
          f32x4 acc = f32x4{0.f, 0.f, 0.f, 0.f};
          #pragma unroll 4
          for (int kk = 0; kk < 8; ++kk) {
            frag b = *(const frag*)&WhhT[(16 * nt + l16) * (4 * RNN_H)
                                         + kk * 32 + lgrp * 8];
            acc = mfma16x16x32(a_at(kk), b, acc);
          }
          #pragma unroll
          for (int r = 0; r < 4; ++r) dh_rec[nt][r] = acc[r];
        }
      }

      // ---- GEMM2: dx / dh hand-off ---------------------------------------
      if (!l0cin1) {
        #pragma unroll 2
        for (int nt = 0; nt < 4; ++nt) {
          f32x4 acc = f32x4{0.f, 0.f, 0.f, 0.f};
          #pragma unroll 4
          for (int kk = 0; kk < 8; ++kk) {
            frag b = *(const frag*)&WihT[(16 * nt + l16) * (4 * RNN_H)
                                         + kk * 32 + lgrp * 8];
            acc = mfma16x16x32(a_at(kk), b, acc);
          }
          if (layer > 0) {                         // packed lane-private
            T v4[4];
            #pragma unroll
            for (int r = 0; r < 4; ++r) v4[r] = fromF<T>(acc[r]);
            *(((ulong1*)dh_at(t)) + nt) = *(ulong1*)v4;
          } else {                                 // dense dx (S,Tst,64)
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
              const int row = row0 + 4 * lgrp + r;
              if (row < S)
                dx[((long)row * Tst + t) * RNN_H + 16 * nt + l16] =
                    fromF<T>(acc[r]);
            }
          }
        }
      } else {
        // CIN1 l0: dx[row,t] = sum_ch dxacc; reduce over the 16 l16 lanes
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = dxacc[r];
          #pragma unroll
          for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, 64);
          const int row = row0 + 4 * lgrp + r;
          if (l16 == 0 && row < S) dx[(long)row * Tst + t] = fromF<T>(v);
        }
      }

      // ---- dA stream to global (natural layout, from the A-fragments) ---
      {
        T* out_dA = dA_g + (lay_base + t) * ((long)S_pad * 4 * RNN_H)
                    + (long)(row0 + l16) * 4 * RNN_H;
        #pragma unroll
        for (int kk = 0; kk < 8; ++kk)
          *(frag*)&out_dA[kk * 32 + lgrp * 8] = a_at(kk);
      }

      // ---- stage-end: stage the prefetched gates into lane LDS, rotate --
      if (t > 0) {
        #pragma unroll
        for (int nt = 0; nt < 4; ++nt) {
          *(frag*)&gmine[nt * 32 + 0] = gld[nt][0];
          *(frag*)&gmine[nt * 32 + 16] = gld[nt][1];
          cc_t[nt] = GRU ? cnext[nt] : cc_p[nt];
          if (!GRU) cc_p[nt] = (t >= 2) ? cnext[nt] : ulong1{0};
          if (layer < L - 1) dhup[nt] = dhnext[nt];
        }
      }
    }  // t loop
  }  // layer loop
}

static int bwd_wave_mode() {
  static int mode = -1;
  if (mode < 0) {
    const char* e = getenv("STMGCN_BWD_WAVE");
    mode = e ? atoi(e) : 0;
  }
  return mode;
}

template <typename T>
void launch_bwd(hipStream_t stream, const void* dout, const void* x,
                const void* cseq_g, const void* gates_g, const RnnPtrs& w,
                void* dx, void* dA_g, void* dh_g, int S, int Tst, int L,
                int cin, int ret_seq, int gru) {
  constexpr int ST = SEQ_TILE;
  const int nblk = (S + ST - 1) / ST;
  const int S_pad = nblk * ST;
  if (bwd_wave_mode()) {
    const int nblk64 = (S_pad + 63) / 64;
    const size_t ldsw = 32768 + 256 * 128;     // wave dA slots + lane gates
    auto gow = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(nblk64), dim3(256), ldsw, stream,
                         (const T*)dout, (const T*)cseq_g, (const T*)gates_g,
                         w, (T*)dx, (T*)dA_g, (T*)dh_g, S, S_pad, Tst, L,
                         ret_seq);
    };
    if (cin == 1 && !gru) gow(lstm_bwd_wave_kernel<T, true, false>);
    else if (cin == 1 && gru) gow(lstm_bwd_wave_kernel<T, true, true>);
    else if (!gru) gow(lstm_bwd_wave_kernel<T, false, false>);
    else gow(lstm_bwd_wave_kernel<T, false, true>);
    return;
  }
  const size_t lds_bytes = 2 * ST * 512 + 4 * ST * sizeof(float)
                           + 256 * (ST / 16) * 32;   // + lane-private gstage
  auto go = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(nblk), dim3(256), lds_bytes, stream,
                       (const T*)dout, (const T*)x, (const T*)cseq_g,
                       (const T*)gates_g, w, (T*)dx, (T*)dA_g, (T*)dh_g,
                       S, Tst, L, ret_seq);
  };
  if (cin == 1 && !gru) go(lstm_bwd_kernel<T, true, false, ST>);
  else if (cin == 1 && gru) go(lstm_bwd_kernel<T, true, true, ST>);
  else if (!gru) go(lstm_bwd_kernel<T, false, false, ST>);
  else go(lstm_bwd_kernel<T, false, true, ST>);
}

extern "C" void stmgcn_lstm_bwd(void* stream_v, int dtype, const void* dout,
                                const void* x, const void* cseq_g,
                                const void* gates_g, const void** w_ihT,
                                const void** w_hhT, void* dx, void* dA_g,
                                void* dh_g, int S, int Tst, int L, int cin,
                                int ret_seq, int gru) {
  RnnPtrs p;
  for (int l = 0; l < L && l < MAX_LAYERS; ++l) {
    p.w_ih[l] = w_ihT[l]; p.w_hh[l] = w_hhT[l];
    p.b_ih[l] = nullptr; p.b_hh[l] = nullptr;
  }
  hipStream_t stream = (hipStream_t)stream_v;
  if (dtype == STM_BF16)
    launch_bwd<__hip_bfloat16>(stream, dout, x, cseq_g, gates_g, p, dx, dA_g,
                               dh_g, S, Tst, L, cin, ret_seq, gru);
  else if (dtype == STM_F16)
    launch_bwd<__half>(stream, dout, x, cseq_g, gates_g, p, dx, dA_g, dh_g,
                       S, Tst, L, cin, ret_seq, gru);
}
