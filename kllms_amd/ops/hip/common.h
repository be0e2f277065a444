// Shared helpers for the kllms_amd CDNA4 (gfx950) kernels.
// Wavefront size is 64 on CDNA; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE 64

typedef __hip_bfloat16 bf16_t;

// 8 bf16 = 16 bytes: one dwordx4 load per lane (G13: always vectorize bf16).
typedef short bf16x8_vec __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short bf16x4_vec __attribute__((ext_vector_type(4)));

__device__ __forceinline__ float bf16_to_f32(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f32_to_bf16(float f) {
  // round-to-nearest-even, matching PyTorch's float->bfloat16 cast
  union { float f; uint32_t i; } c;
  c.f = f;
  uint32_t x = c.i;
  uint32_t lsb = (x >> 16) & 1;
  x += 0x7fff + lsb;
  return (short)(x >> 16);
}

// wave-wide reductions (64 lanes)
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block reductions through LDS (block = N waves, N <= 16)
template <int MAX_WAVES>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  const int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  // every wave reads the per-wave partials so ALL threads get the total
  v = (lane < nwaves) ? lds_scratch[lane] : 0.0f;
  v = wave_reduce_sum(v);
  __syncthreads();
  return v;
}

template <int MAX_WAVES>
__device__ __forceinline__ float block_reduce_max(float v, float* lds_scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  const int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  v = (lane < nwaves) ? lds_scratch[lane] : -INFINITY;
  v = wave_reduce_max(v);
  __syncthreads();
  return v;
}

// fp8 (OCP e4m3) conversions for the optional fp8 KV cache (gfx950 native
// OCP format, NOT the MI300X fnuz variant)
typedef unsigned char u8x16_vec __attribute__((ext_vector_type(16)));
typedef unsigned char u8x8_vec __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float fp8_to_f32(unsigned char u) {
  __hip_fp8_e4m3 v;
  v.__x = u;
  return (float)v;
}

__device__ __forceinline__ unsigned char f32_to_fp8(float f) {
  __hip_fp8_e4m3 v(f);
  return v.__x;
}

// splitmix64: per-(seed, step, token) counter-based RNG for the sampler
__device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

__device__ __forceinline__ float u64_to_uniform(uint64_t h) {
  // (0, 1]: use top 24 bits
  return ((float)((h >> 40) + 1)) * (1.0f / 16777217.0f);
}

#define HIP_CHECK_LAUNCH()                                                     \
  do {                                                                         \
    hipError_t e_ = hipGetLastError();                                         \
    if (e_ != hipSuccess) {                                                    \
      printf("kernel launch failed: %s\n", hipGetErrorString(e_));             \
    }                                                                          \
  } while (0)

// decode-attention split-K geometry (shared by attn_decode.hip and bindings)
#define ATTN_DECODE_CHUNK 256   // keys per workgroup
#define ATTN_DECODE_TKV 256     // keys per LDS tile (one tile per chunk)
