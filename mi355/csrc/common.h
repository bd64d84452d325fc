// Shared helpers for the mi355 CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA4; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

// ---------------------------------------------------------------------------
// dtype traits: load/store conversion to fp32 compute
// ---------------------------------------------------------------------------
template <typename T> struct DT;

template <> struct DT<float> {
  using vec4 = float4;  // 16 B
  static DEVINL float to_f32(float x) { return x; }
  static DEVINL float from_f32(float x) { return x; }
};

template <> struct DT<__hip_bfloat16> {
  static DEVINL float to_f32(__hip_bfloat16 x) { return __bfloat162float(x); }
  static DEVINL __hip_bfloat16 from_f32(float x) { return __float2bfloat16(x); }
};

template <> struct DT<__half> {
  static DEVINL float to_f32(__half x) { return __half2float(x); }
  static DEVINL __half from_f32(float x) { return __float2half(x); }
};

// Vectorized 8-element access for 2-byte dtypes (16 B = dwordx4), 4-element
// for f32. hipcc does not auto-vectorize bf16 loads (guide G13) — always go
// through these.
template <typename T, int N> struct alignas(sizeof(T) * N) VecT {
  T data[N];
};

template <typename T, int N>
DEVINL VecT<T, N> vload(const T* p) {
  return *reinterpret_cast<const VecT<T, N>*>(p);
}
template <typename T, int N>
DEVINL void vstore(T* p, const VecT<T, N>& v) {
  *reinterpret_cast<VecT<T, N>*>(p) = v;
}

// ---------------------------------------------------------------------------
// wave + block reductions
// ---------------------------------------------------------------------------
DEVINL float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

DEVINL float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Block-level reduce for block sizes up to 1024 (<=16 waves). smem must hold
// 16 floats. Result valid on every thread.
DEVINL float block_reduce_sum(float x, float* smem) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) smem[wid] = x;
  __syncthreads();
  float r = (lane < nw) ? smem[lane] : 0.0f;
  r = wave_reduce_sum(r);
  __syncthreads();
  return r;
}

DEVINL float block_reduce_max(float x, float* smem) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  x = wave_reduce_max(x);
  if (lane == 0) smem[wid] = x;
  __syncthreads();
  float r = (lane < nw) ? smem[lane] : -INFINITY;
  r = wave_reduce_max(r);
  __syncthreads();
  return r;
}

// deterministic per-element uniform in [0,1) for dropout masks: same
// (seed, idx) in forward and backward reproduces the same mask without
// storing it (splitmix64 finalizer)
DEVINL float hash_u01(long long seed, long long idx) {
  unsigned long long x = (unsigned long long)seed * 0x9E3779B97F4A7C15ull ^
                         (unsigned long long)idx * 0xBF58476D1CE4E5B9ull;
  x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 27; x *= 0x94D049BB133111EBull;
  x ^= x >> 31;
  return (float)(x >> 40) * (1.0f / 16777216.0f);
}

#define HIP_CHECK_KERNEL()                                           \
  do {                                                               \
    hipError_t e_ = hipGetLastError();                               \
    if (e_ != hipSuccess) {                                          \
      TORCH_CHECK(false, "HIP kernel launch failed: ",               \
                  hipGetErrorString(e_));                            \
    }                                                                \
  } while (0)
