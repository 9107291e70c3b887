// Common device helpers for the stmgcn_amd CDNA4 (gfx950) kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define STM_WAVE 64  // CDNA wavefront width (never 32)

// dtype codes shared with the python side
enum StmDtype : int { STM_F32 = 0, STM_BF16 = 1, STM_F16 = 2 };

template <typename T> __device__ __forceinline__ float toF(T v);
template <> __device__ __forceinline__ float toF<float>(float v) { return v; }
template <> __device__ __forceinline__ float toF<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <> __device__ __forceinline__ float toF<__half>(__half v) { return __half2float(v); }

template <typename T> __device__ __forceinline__ T fromF(float v);
template <> __device__ __forceinline__ float fromF<float>(float v) { return v; }
template <> __device__ __forceinline__ __hip_bfloat16 fromF<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ __half fromF<__half>(float v) { return __float2half(v); }

// Fast device transcendentals: v_exp_f32 + v_rcp_f32 (~1 ulp each) instead
// of libm tanhf (~30 VALU instr) / IEEE divide — the RNN pointwise phase is
// VALU-bound on these (up to 40 per thread per (layer,t) stage) and the
// results are rounded to bf16/f16 anyway.
__device__ __forceinline__ float stm_sigmoid(float x) {
  return __builtin_amdgcn_rcpf(1.0f + __expf(-x));
}
__device__ __forceinline__ float stm_tanh(float x) {
  // tanh(x) = 2*sigmoid(2x) - 1; clamp keeps exp finite for large |x|
  const float t = (x > 15.f) ? 15.f : (x < -15.f ? -15.f : x);
  return fmaf(2.0f, stm_sigmoid(2.0f * t), -1.0f);
}

// LDS swizzle for 128-byte tile rows (64 bf16/f16 channels): element (row s,
// channel byte cbyte) of one tile slot. XOR'd so the 16 lanes of a
// ds_read_b128 fragment group (consecutive rows, same channel window) hit 64
// distinct banks: row parity gives the 32-bank half, ((s>>1)&7) permutes the
// eight 16 B windows within it.
__device__ __forceinline__ int lds_swz(int s, int cbyte) {
  return s * 128 + (cbyte ^ ((((unsigned)s >> 1) & 7) << 4));
}

#define STM_CHECK_HIP(expr)                                                     \
  do {                                                                          \
    hipError_t _e = (expr);                                                     \
    if (_e != hipSuccess) {                                                     \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, __LINE__); \
    }                                                                           \
  } while (0)
