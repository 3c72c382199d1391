// Shared helpers for the gfx950 (MI355X / CDNA4) kernels.
//
// Conventions (per the CDNA4 programming guide):
//  - wavefront = 64 lanes; block sizes are multiples of 64
//  - bf16 memory traffic is vectorized as ushort4/ushort8 (8/16 B per lane)
//  - f32 accumulation everywhere; bf16 only at the memory boundary
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

// vector types for wide loads
typedef ushort4 bf16x4;  // 8 B
struct bf16x8 { ushort4 lo, hi; };  // 16 B

DEVINL float bf2f(unsigned short u) {
  union { float f; unsigned int i; } w;
  w.i = ((unsigned int)u) << 16;
  return w.f;
}

DEVINL unsigned short f2bf(float f) {
  union { float f; unsigned int i; } w;
  w.f = f;
  // round-to-nearest-even
  unsigned int lsb = (w.i >> 16) & 1;
  w.i += 0x7fffu + lsb;
  return (unsigned short)(w.i >> 16);
}

// wave-wide f32 sum/max over all 64 lanes
DEVINL float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEVINL float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// reduction over a 16-lane group (lanes l, l^1, l^2, ..., same l>>4)
DEVINL float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEVINL float group16_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block-wide f32 sum using one LDS slot per wave (call with all threads)
template <int BLOCK>
DEVINL float block_sum(float v, float* lds_scratch /* BLOCK/WAVE floats */) {
  const int wid = threadIdx.x / WAVE;
  const int nw = BLOCK / WAVE;
  v = wave_sum(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) lds_scratch[wid] = v;
  __syncthreads();
  float out = 0.f;
#pragma unroll
  for (int i = 0; i < nw; ++i) out += lds_scratch[i];
  return out;
}

#define HIP_CHECK_LAST()                                             \
  do {                                                               \
    hipError_t e = hipGetLastError();                                \
    if (e != hipSuccess) {                                           \
      TORCH_CHECK(false, "HIP kernel launch failed: ",               \
                  hipGetErrorString(e));                             \
    }                                                                \
  } while (0)
