// RMSNorm + LayerNorm forward/backward for gfx950.
//
// Memory-bound ops: one 256-thread workgroup per row (grid-strided over
// rows), bf16 traffic vectorized 8-wide (ushort4 pairs = 16 B/lane), f32
// accumulation, rstd/mean saved for backward.  dweight/dbias are accumulated
// in LDS per block, then atomically added to fp32 global buffers (one atomic
// per column per block).
//
// Replaces the reference's nn.LayerNorm hot path
// (/root/reference/torchdistpackage/parallel/tensor_parallel/transformer.py:15-44)
// and the layernorm math spec (explore/understand_ops/layernorm.py:3-14).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int MAX_D = 16384;

// ---------------------------------------------------------------- RMSNorm

template <typename T>
DEVINL float load_as_f32(const T* p, int i);
template <> DEVINL float load_as_f32<bf16_t>(const bf16_t* p, int i) {
  return bf2f(((const unsigned short*)p)[i]);
}
template <> DEVINL float load_as_f32<float>(const float* p, int i) {
  return p[i];
}
template <typename T>
DEVINL void store_from_f32(T* p, int i, float v);
template <> DEVINL void store_from_f32<bf16_t>(bf16_t* p, int i, float v) {
  ((unsigned short*)p)[i] = f2bf(v);
}
template <> DEVINL void store_from_f32<float>(float* p, int i, float v) {
  p[i] = v;
}

template <typename T>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x,
                                   const T* __restrict__ w,
                                   T* __restrict__ y,
                                   float* __restrict__ rstd,
                                   int rows, int D, float eps) {
  __shared__ float lds[BLOCK / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * D;
    T* yr = y + (long)row * D;
    float ss = 0.f;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float v = load_as_f32(xr, i);
      ss += v * v;
    }
    ss = block_sum<BLOCK>(ss, lds);
    float r = rsqrtf(ss / D + eps);
    if (threadIdx.x == 0) rstd[row] = r;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float v = load_as_f32(xr, i);
      float wv = load_as_f32(w, i);
      store_from_f32(yr, i, v * r * wv);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const T* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   T* __restrict__ dx,
                                   float* __restrict__ dw_partial,
                                   int rows, int D) {
  // dynamic LDS: [D] floats for the dw accumulator + BLOCK/WAVE scratch
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_lds = (float*)smem;
  float* scratch = dw_lds + D;
  for (int i = threadIdx.x; i < D; i += BLOCK) dw_lds[i] = 0.f;
  __syncthreads();

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + (long)row * D;
    const T* xr = x + (long)row * D;
    T* dxr = dx + (long)row * D;
    float r = rstd[row];
    float dot = 0.f;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float xhat = load_as_f32(xr, i) * r;
      float wdy = load_as_f32(w, i) * load_as_f32(dyr, i);
      dot += wdy * xhat;
    }
    dot = block_sum<BLOCK>(dot, scratch) / D;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float xhat = load_as_f32(xr, i) * r;
      float dyv = load_as_f32(dyr, i);
      float wdy = load_as_f32(w, i) * dyv;
      store_from_f32(dxr, i, r * (wdy - xhat * dot));
      dw_lds[i] += dyv * xhat;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += BLOCK)
    atomicAdd(&dw_partial[i], dw_lds[i]);
}

// -------------------------------------------------------------- LayerNorm

template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const T* __restrict__ w,
                                     const T* __restrict__ b,
                                     T* __restrict__ y,
                                     float* __restrict__ mean,
                                     float* __restrict__ rstd,
                                     int rows, int D, float eps) {
  __shared__ float lds[BLOCK / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * D;
    T* yr = y + (long)row * D;
    float s = 0.f, ss = 0.f;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float v = load_as_f32(xr, i);
      s += v;
      ss += v * v;
    }
    s = block_sum<BLOCK>(s, lds);
    __syncthreads();
    ss = block_sum<BLOCK>(ss, lds);
    float mu = s / D;
    float var = ss / D - mu * mu;
    float r = rsqrtf(var + eps);
    if (threadIdx.x == 0) { mean[row] = mu; rstd[row] = r; }
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float v = load_as_f32(xr, i);
      float wv = load_as_f32(w, i);
      float bv = load_as_f32(b, i);
      store_from_f32(yr, i, (v - mu) * r * wv + bv);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void layernorm_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const T* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ dx,
                                     float* __restrict__ dw_partial,
                                     float* __restrict__ db_partial,
                                     int rows, int D) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_lds = (float*)smem;
  float* db_lds = dw_lds + D;
  float* scratch = db_lds + D;
  for (int i = threadIdx.x; i < D; i += BLOCK) { dw_lds[i] = 0.f; db_lds[i] = 0.f; }
  __syncthreads();

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + (long)row * D;
    const T* xr = x + (long)row * D;
    T* dxr = dx + (long)row * D;
    float mu = mean[row], r = rstd[row];
    float m1 = 0.f, m2 = 0.f;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float xhat = (load_as_f32(xr, i) - mu) * r;
      float wdy = load_as_f32(w, i) * load_as_f32(dyr, i);
      m1 += wdy;
      m2 += wdy * xhat;
    }
    m1 = block_sum<BLOCK>(m1, scratch);
    __syncthreads();
    m2 = block_sum<BLOCK>(m2, scratch);
    m1 /= D;
    m2 /= D;
    for (int i = threadIdx.x; i < D; i += BLOCK) {
      float xhat = (load_as_f32(xr, i) - mu) * r;
      float dyv = load_as_f32(dyr, i);
      float wdy = load_as_f32(w, i) * dyv;
      store_from_f32(dxr, i, r * (wdy - m1 - xhat * m2));
      dw_lds[i] += dyv * xhat;
      db_lds[i] += dyv;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += BLOCK) {
    atomicAdd(&dw_partial[i], dw_lds[i]);
    atomicAdd(&db_partial[i], db_lds[i]);
  }
}

int pick_grid(long rows) {
  // memory-bound: cap at 2048 blocks, grid-stride the rest (guideline 11)
  long g = rows < 2048 ? rows : 2048;
  return (int)(g > 0 ? g : 1);
}

// ------------------- vectorized bf16 variants (D % 8 == 0) ----------------
// 16 B/lane loads (guideline 13: scalar bf16 is 2-2.5x slower).

typedef ushort4 u4;
struct f8 { float v[8]; };

DEVINL f8 load8f(const unsigned short* p) {
  u4 a = *(const u4*)p;
  u4 b = *(const u4*)(p + 4);
  f8 r;
  r.v[0] = bf2f(a.x); r.v[1] = bf2f(a.y); r.v[2] = bf2f(a.z); r.v[3] = bf2f(a.w);
  r.v[4] = bf2f(b.x); r.v[5] = bf2f(b.y); r.v[6] = bf2f(b.z); r.v[7] = bf2f(b.w);
  return r;
}

DEVINL void store8f(unsigned short* p, const f8& r) {
  u4 a, b;
  a.x = f2bf(r.v[0]); a.y = f2bf(r.v[1]); a.z = f2bf(r.v[2]); a.w = f2bf(r.v[3]);
  b.x = f2bf(r.v[4]); b.y = f2bf(r.v[5]); b.z = f2bf(r.v[6]); b.w = f2bf(r.v[7]);
  *(u4*)p = a;
  *(u4*)(p + 4) = b;
}

__global__ void rmsnorm_fwd_bf16v8(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   unsigned short* __restrict__ y,
                                   float* __restrict__ rstd,
                                   int rows, int D, float eps) {
  __shared__ float lds[BLOCK / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * D;
    unsigned short* yr = y + (long)row * D;
    float ss = 0.f;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 v = load8f(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) ss += v.v[j] * v.v[j];
    }
    ss = block_sum<BLOCK>(ss, lds);
    float r = rsqrtf(ss / D + eps);
    if (threadIdx.x == 0) rstd[row] = r;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 v = load8f(xr + i);
      f8 wv = load8f(w + i);
      f8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j) out.v[j] = v.v[j] * r * wv.v[j];
      store8f(yr + i, out);
    }
    __syncthreads();
  }
}

__global__ void rmsnorm_bwd_bf16v8(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ w,
                                   const float* __restrict__ rstd,
                                   unsigned short* __restrict__ dx,
                                   float* __restrict__ dw_partial,
                                   int rows, int D) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_lds = (float*)smem;
  float* scratch = dw_lds + D;
  for (int i = threadIdx.x; i < D; i += BLOCK) dw_lds[i] = 0.f;
  __syncthreads();
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* dyr = dy + (long)row * D;
    const unsigned short* xr = x + (long)row * D;
    unsigned short* dxr = dx + (long)row * D;
    float r = rstd[row];
    float dot = 0.f;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 xv = load8f(xr + i), dyv = load8f(dyr + i), wv = load8f(w + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += wv.v[j] * dyv.v[j] * xv.v[j] * r;
    }
    dot = block_sum<BLOCK>(dot, scratch) / D;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 xv = load8f(xr + i), dyv = load8f(dyr + i), wv = load8f(w + i);
      f8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = xv.v[j] * r;
        out.v[j] = r * (wv.v[j] * dyv.v[j] - xhat * dot);
        dw_lds[i + j] += dyv.v[j] * xhat;
      }
      store8f(dxr + i, out);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += BLOCK)
    atomicAdd(&dw_partial[i], dw_lds[i]);
}

__global__ void layernorm_fwd_bf16v8(const unsigned short* __restrict__ x,
                                     const unsigned short* __restrict__ w,
                                     const unsigned short* __restrict__ b,
                                     unsigned short* __restrict__ y,
                                     float* __restrict__ mean,
                                     float* __restrict__ rstd,
                                     int rows, int D, float eps) {
  __shared__ float lds[2 * BLOCK / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* xr = x + (long)row * D;
    unsigned short* yr = y + (long)row * D;
    float s = 0.f, ss = 0.f;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 v = load8f(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) { s += v.v[j]; ss += v.v[j] * v.v[j]; }
    }
    // two block sums sharing one pass: use separate scratch halves
    {
      const int wid = threadIdx.x / WAVE;
      float sw = wave_sum(s), ssw = wave_sum(ss);
      if ((threadIdx.x & (WAVE - 1)) == 0) {
        lds[wid] = sw;
        lds[BLOCK / WAVE + wid] = ssw;
      }
      __syncthreads();
      s = 0.f; ss = 0.f;
#pragma unroll
      for (int i = 0; i < BLOCK / WAVE; ++i) {
        s += lds[i];
        ss += lds[BLOCK / WAVE + i];
      }
    }
    float mu = s / D;
    float var = ss / D - mu * mu;
    float r = rsqrtf(var + eps);
    if (threadIdx.x == 0) { mean[row] = mu; rstd[row] = r; }
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 v = load8f(xr + i), wv = load8f(w + i), bv = load8f(b + i);
      f8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        out.v[j] = (v.v[j] - mu) * r * wv.v[j] + bv.v[j];
      store8f(yr + i, out);
    }
    __syncthreads();
  }
}

__global__ void layernorm_bwd_bf16v8(const unsigned short* __restrict__ dy,
                                     const unsigned short* __restrict__ x,
                                     const unsigned short* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     unsigned short* __restrict__ dx,
                                     float* __restrict__ dw_partial,
                                     float* __restrict__ db_partial,
                                     int rows, int D) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* dw_lds = (float*)smem;
  float* db_lds = dw_lds + D;
  float* scratch = db_lds + D;   // 2*BLOCK/WAVE floats
  for (int i = threadIdx.x; i < D; i += BLOCK) { dw_lds[i] = 0.f; db_lds[i] = 0.f; }
  __syncthreads();
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* dyr = dy + (long)row * D;
    const unsigned short* xr = x + (long)row * D;
    unsigned short* dxr = dx + (long)row * D;
    float mu = mean[row], r = rstd[row];
    float m1 = 0.f, m2 = 0.f;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 xv = load8f(xr + i), dyv = load8f(dyr + i), wv = load8f(w + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = (xv.v[j] - mu) * r;
        float wdy = wv.v[j] * dyv.v[j];
        m1 += wdy;
        m2 += wdy * xhat;
      }
    }
    {
      const int wid = threadIdx.x / WAVE;
      float a = wave_sum(m1), c = wave_sum(m2);
      if ((threadIdx.x & (WAVE - 1)) == 0) {
        scratch[wid] = a;
        scratch[BLOCK / WAVE + wid] = c;
      }
      __syncthreads();
      m1 = 0.f; m2 = 0.f;
#pragma unroll
      for (int i = 0; i < BLOCK / WAVE; ++i) {
        m1 += scratch[i];
        m2 += scratch[BLOCK / WAVE + i];
      }
    }
    m1 /= D;
    m2 /= D;
    for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
      f8 xv = load8f(xr + i), dyv = load8f(dyr + i), wv = load8f(w + i);
      f8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xhat = (xv.v[j] - mu) * r;
        out.v[j] = r * (wv.v[j] * dyv.v[j] - m1 - xhat * m2);
        dw_lds[i + j] += dyv.v[j] * xhat;
        db_lds[i + j] += dyv.v[j];
      }
      store8f(dxr + i, out);
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < D; i += BLOCK) {
    atomicAdd(&dw_partial[i], dw_lds[i]);
    atomicAdd(&db_partial[i], db_lds[i]);
  }
}

template <int NC>
__global__ void rmsnorm_bwd_bf16_reg(const unsigned short* __restrict__ dy,
                                     const unsigned short* __restrict__ x,
                                     const unsigned short* __restrict__ w,
                                     const float* __restrict__ rstd,
                                     unsigned short* __restrict__ dx,
                                     float* __restrict__ dw_partial,
                                     int rows, int D) {
  __shared__ float scratch[BLOCK / WAVE];
  float acc_dw[NC][8];
#pragma unroll
  for (int c = 0; c < NC; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc_dw[c][j] = 0.f;

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* dyr = dy + (long)row * D;
    const unsigned short* xr = x + (long)row * D;
    unsigned short* dxr = dx + (long)row * D;
    float r = rstd[row];
    float dot = 0.f;
    f8 xv[NC], dyv[NC], wv[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      int i = threadIdx.x * 8 + c * BLOCK * 8;
      if (i < D) {
        xv[c] = load8f(xr + i);
        dyv[c] = load8f(dyr + i);
        wv[c] = load8f(w + i);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          dot += wv[c].v[j] * dyv[c].v[j] * xv[c].v[j] * r;
      }
    }
    dot = block_sum<BLOCK>(dot, scratch) / D;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      int i = threadIdx.x * 8 + c * BLOCK * 8;
      if (i < D) {
        f8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xhat = xv[c].v[j] * r;
          out.v[j] = r * (wv[c].v[j] * dyv[c].v[j] - xhat * dot);
          acc_dw[c][j] += dyv[c].v[j] * xhat;
        }
        store8f(dxr + i, out);
      }
    }
    __syncthreads();
  }
  // per-block partial row (coalesced plain stores; summed host-side) —
  // atomicAdd here serializes gridDim-deep per element and was 2x slower
  float* dwp = dw_partial + (long)blockIdx.x * D;
#pragma unroll
  for (int c = 0; c < NC; ++c) {
    int i = threadIdx.x * 8 + c * BLOCK * 8;
    if (i < D)
#pragma unroll
      for (int j = 0; j < 8; ++j) dwp[i + j] = acc_dw[c][j];
  }
}

template <int NC>
__global__ void layernorm_bwd_bf16_reg(const unsigned short* __restrict__ dy,
                                       const unsigned short* __restrict__ x,
                                       const unsigned short* __restrict__ w,
                                       const float* __restrict__ mean,
                                       const float* __restrict__ rstd,
                                       unsigned short* __restrict__ dx,
                                       float* __restrict__ dw_partial,
                                       float* __restrict__ db_partial,
                                       int rows, int D) {
  __shared__ float scratch[2 * BLOCK / WAVE];
  float acc_dw[NC][8], acc_db[NC][8];
#pragma unroll
  for (int c = 0; c < NC; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) { acc_dw[c][j] = 0.f; acc_db[c][j] = 0.f; }

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const unsigned short* dyr = dy + (long)row * D;
    const unsigned short* xr = x + (long)row * D;
    unsigned short* dxr = dx + (long)row * D;
    float mu = mean[row], r = rstd[row];
    float m1 = 0.f, m2 = 0.f;
    f8 xv[NC], dyv[NC], wv[NC];
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      int i = threadIdx.x * 8 + c * BLOCK * 8;
      if (i < D) {
        xv[c] = load8f(xr + i);
        dyv[c] = load8f(dyr + i);
        wv[c] = load8f(w + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xhat = (xv[c].v[j] - mu) * r;
          float wdy = wv[c].v[j] * dyv[c].v[j];
          m1 += wdy;
          m2 += wdy * xhat;
        }
      }
    }
    {
      const int wid = threadIdx.x / WAVE;
      float a = wave_sum(m1), cgr = wave_sum(m2);
      if ((threadIdx.x & (WAVE - 1)) == 0) {
        scratch[wid] = a;
        scratch[BLOCK / WAVE + wid] = cgr;
      }
      __syncthreads();
      m1 = 0.f; m2 = 0.f;
#pragma unroll
      for (int i = 0; i < BLOCK / WAVE; ++i) {
        m1 += scratch[i];
        m2 += scratch[BLOCK / WAVE + i];
      }
    }
    m1 /= D;
    m2 /= D;
#pragma unroll
    for (int c = 0; c < NC; ++c) {
      int i = threadIdx.x * 8 + c * BLOCK * 8;
      if (i < D) {
        f8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float xhat = (xv[c].v[j] - mu) * r;
          out.v[j] = r * (wv[c].v[j] * dyv[c].v[j] - m1 - xhat * m2);
          acc_dw[c][j] += dyv[c].v[j] * xhat;
          acc_db[c][j] += dyv[c].v[j];
        }
        store8f(dxr + i, out);
      }
    }
    __syncthreads();
  }
  float* dwp = dw_partial + (long)blockIdx.x * D;
  float* dbp = db_partial + (long)blockIdx.x * D;
#pragma unroll
  for (int c = 0; c < NC; ++c) {
    int i = threadIdx.x * 8 + c * BLOCK * 8;
    if (i < D)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        dwp[i + j] = acc_dw[c][j];
        dbp[i + j] = acc_db[c][j];
      }
  }
}

}  // namespace

// ------------------------------------------------------------------ C++ API

std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int D = x.size(-1);
  TORCH_CHECK(D <= MAX_D, "D too large");
  const long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(pick_grid(rows)), block(BLOCK);
  if (x.scalar_type() == torch::kBFloat16 && D % 8 == 0) {
    hipLaunchKernelGGL(rmsnorm_fwd_bf16v8, grid, block, 0, stream,
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       (unsigned short*)y.data_ptr(), rstd.data_ptr<float>(),
                       (int)rows, D, (float)eps);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<bf16_t>, grid, block, 0, stream,
                       (const bf16_t*)x.data_ptr(), (const bf16_t*)w.data_ptr(),
                       (bf16_t*)y.data_ptr(), rstd.data_ptr<float>(),
                       (int)rows, D, (float)eps);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat, "bf16/f32 only");
    hipLaunchKernelGGL(rmsnorm_fwd_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(),
                       y.data_ptr<float>(), rstd.data_ptr<float>(),
                       (int)rows, D, (float)eps);
  }
  HIP_CHECK_LAST();
  return {y, rstd};
}

std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd) {
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(pick_grid(rows)), block(BLOCK);
  size_t lds = (D + BLOCK / WAVE) * sizeof(float);
  const bool reg_path = x.scalar_type() == torch::kBFloat16 &&
      D % 8 == 0 && D <= 4 * BLOCK * 8;
  auto dw_partial = reg_path
      ? torch::empty({(long)grid.x, (long)D}, x.options().dtype(torch::kFloat))
      : torch::zeros({D}, x.options().dtype(torch::kFloat));
  if (reg_path) {
    int nc = (D + BLOCK * 8 - 1) / (BLOCK * 8);
#define RMS_REG(NC)                                                          \
    hipLaunchKernelGGL(rmsnorm_bwd_bf16_reg<NC>, grid, block, 0, stream,     \
                       (const unsigned short*)dy.data_ptr(),                 \
                       (const unsigned short*)x.data_ptr(),                  \
                       (const unsigned short*)w.data_ptr(),                  \
                       rstd.data_ptr<float>(),                               \
                       (unsigned short*)dx.data_ptr(),                       \
                       dw_partial.data_ptr<float>(), (int)rows, D)
    if (nc == 1) RMS_REG(1); else if (nc == 2) RMS_REG(2);
    else if (nc == 3) RMS_REG(3); else RMS_REG(4);
#undef RMS_REG
    HIP_CHECK_LAST();
    return {dx, dw_partial.sum(0).to(w.scalar_type())};
  } else if (x.scalar_type() == torch::kBFloat16 && D % 8 == 0) {
    hipLaunchKernelGGL(rmsnorm_bwd_bf16v8, grid, block, lds, stream,
                       (const unsigned short*)dy.data_ptr(),
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       rstd.data_ptr<float>(), (unsigned short*)dx.data_ptr(),
                       dw_partial.data_ptr<float>(), (int)rows, D);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(rmsnorm_bwd_kernel<bf16_t>, grid, block, lds, stream,
                       (const bf16_t*)dy.data_ptr(), (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)w.data_ptr(), rstd.data_ptr<float>(),
                       (bf16_t*)dx.data_ptr(), dw_partial.data_ptr<float>(),
                       (int)rows, D);
  } else {
    hipLaunchKernelGGL(rmsnorm_bwd_kernel<float>, grid, block, lds, stream,
                       dy.data_ptr<float>(), x.data_ptr<float>(),
                       w.data_ptr<float>(), rstd.data_ptr<float>(),
                       dx.data_ptr<float>(), dw_partial.data_ptr<float>(),
                       (int)rows, D);
  }
  HIP_CHECK_LAST();
  return {dx, dw_partial.to(w.scalar_type())};
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int D = x.size(-1);
  TORCH_CHECK(D <= MAX_D, "D too large");
  const long rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto mean = torch::empty({rows}, x.options().dtype(torch::kFloat));
  auto rstd = torch::empty({rows}, x.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(pick_grid(rows)), block(BLOCK);
  if (x.scalar_type() == torch::kBFloat16 && D % 8 == 0) {
    hipLaunchKernelGGL(layernorm_fwd_bf16v8, grid, block, 0, stream,
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       (const unsigned short*)b.data_ptr(),
                       (unsigned short*)y.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (int)rows, D, (float)eps);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(layernorm_fwd_kernel<bf16_t>, grid, block, 0, stream,
                       (const bf16_t*)x.data_ptr(), (const bf16_t*)w.data_ptr(),
                       (const bf16_t*)b.data_ptr(), (bf16_t*)y.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (int)rows, D, (float)eps);
  } else {
    TORCH_CHECK(x.scalar_type() == torch::kFloat, "bf16/f32 only");
    hipLaunchKernelGGL(layernorm_fwd_kernel<float>, grid, block, 0, stream,
                       x.data_ptr<float>(), w.data_ptr<float>(),
                       b.data_ptr<float>(), y.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (int)rows, D, (float)eps);
  }
  HIP_CHECK_LAST();
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(pick_grid(rows)), block(BLOCK);
  size_t lds = (2 * D + 2 * (BLOCK / WAVE)) * sizeof(float);
  const bool reg_path = x.scalar_type() == torch::kBFloat16 &&
      D % 8 == 0 && D <= 4 * BLOCK * 8;
  auto dw_partial = reg_path
      ? torch::empty({(long)grid.x, (long)D}, x.options().dtype(torch::kFloat))
      : torch::zeros({D}, x.options().dtype(torch::kFloat));
  auto db_partial = reg_path
      ? torch::empty({(long)grid.x, (long)D}, x.options().dtype(torch::kFloat))
      : torch::zeros({D}, x.options().dtype(torch::kFloat));
  if (reg_path) {
    int nc = (D + BLOCK * 8 - 1) / (BLOCK * 8);
#define LN_REG(NC)                                                           \
    hipLaunchKernelGGL(layernorm_bwd_bf16_reg<NC>, grid, block, 0, stream,   \
                       (const unsigned short*)dy.data_ptr(),                 \
                       (const unsigned short*)x.data_ptr(),                  \
                       (const unsigned short*)w.data_ptr(),                  \
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),       \
                       (unsigned short*)dx.data_ptr(),                       \
                       dw_partial.data_ptr<float>(),                         \
                       db_partial.data_ptr<float>(), (int)rows, D)
    if (nc == 1) LN_REG(1); else if (nc == 2) LN_REG(2);
    else if (nc == 3) LN_REG(3); else LN_REG(4);
#undef LN_REG
    HIP_CHECK_LAST();
    return {dx, dw_partial.sum(0).to(w.scalar_type()),
            db_partial.sum(0).to(w.scalar_type())};
  } else if (x.scalar_type() == torch::kBFloat16 && D % 8 == 0) {
    hipLaunchKernelGGL(layernorm_bwd_bf16v8, grid, block, lds, stream,
                       (const unsigned short*)dy.data_ptr(),
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)w.data_ptr(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (unsigned short*)dx.data_ptr(),
                       dw_partial.data_ptr<float>(),
                       db_partial.data_ptr<float>(), (int)rows, D);
  } else if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(layernorm_bwd_kernel<bf16_t>, grid, block, lds, stream,
                       (const bf16_t*)dy.data_ptr(), (const bf16_t*)x.data_ptr(),
                       (const bf16_t*)w.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), (bf16_t*)dx.data_ptr(),
                       dw_partial.data_ptr<float>(), db_partial.data_ptr<float>(),
                       (int)rows, D);
  } else {
    hipLaunchKernelGGL(layernorm_bwd_kernel<float>, grid, block, lds, stream,
                       dy.data_ptr<float>(), x.data_ptr<float>(),
                       w.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), dx.data_ptr<float>(),
                       dw_partial.data_ptr<float>(), db_partial.data_ptr<float>(),
                       (int)rows, D);
  }
  HIP_CHECK_LAST();
  return {dx, dw_partial.to(w.scalar_type()), db_partial.to(w.scalar_type())};
}
