// Fused elementwise / reduction kernels for gfx950:
//   - bias_gelu fwd/bwd (tanh GELU, GPT-2 convention) — fuses the reference's
//     nn.GELU after fc1 (/root/reference/.../tensor_parallel/mlp.py:30-32)
//   - fused AdamW step on flat fp32 shards (replaces torch.optim.AdamW.step
//     per-tensor loop, reference zero_optim.py:265)
//   - EMA update (reference sharded_ema.py:29 mul_/add_ pair -> one pass)
//   - l2norm_sq + inplace scale (grad clip, reference clip_grad_parallel.py)
//
// All memory-bound: grid-stride loops, 256-thread blocks, bf16 vectorized as
// ushort4/bf16x8 (8-16 B per lane) where applicable, f32 math.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr float GELU_C = 0.7978845608028654f;   // sqrt(2/pi)
constexpr float GELU_A = 0.044715f;

// tanh via the hardware exp2 unit: tanh(x) = 1 - 2/(exp2(x*2*log2e)+1).
// Overflow saturates the right way (exp2->inf -> rcp->0 -> +1; exp2->0 -> -1)
// so no range branches are needed; ulp-exact at bf16 output precision and
// ~5x fewer instructions than libm tanhf (which rate-limited the fused
// bias+GELU kernels below ~3.3 TB/s; HBM roofline is ~6).
DEVINL float tanh_fast(float x) {
  float e = __builtin_amdgcn_exp2f(x * 2.885390082f);  // 2*log2(e)
  return 1.f - 2.f / (e + 1.f);
}

DEVINL float gelu_f(float u) {
  float t = tanh_fast(GELU_C * (u + GELU_A * u * u * u));
  return 0.5f * u * (1.f + t);
}

DEVINL float gelu_df(float u) {
  float u2 = u * u;
  float t = tanh_fast(GELU_C * (u + GELU_A * u * u2));
  return 0.5f * (1.f + t) +
         0.5f * u * (1.f - t * t) * GELU_C * (1.f + 3.f * GELU_A * u2);
}

// -------- bias_gelu: x (rows, D) bf16, bias (D) bf16 (or empty) ----------

__global__ void bias_gelu_fwd_bf16(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ bias,
                                   unsigned short* __restrict__ y,
                                   long n, int D, int has_bias) {
  long i0 = ((long)blockIdx.x * BLOCK + threadIdx.x) * 8;
  long stride = (long)gridDim.x * BLOCK * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n) {
      bf16x8 xv = *(const bf16x8*)(x + i);
      // D % 8 == 0 guaranteed by caller for the vector path
      bf16x8 bv;
      if (has_bias) bv = *(const bf16x8*)(bias + (int)(i % D));
      const unsigned short* xs = (const unsigned short*)&xv;
      const unsigned short* bs = (const unsigned short*)&bv;
      bf16x8 out;
      unsigned short* os = (unsigned short*)&out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float b = has_bias ? bf2f(bs[j]) : 0.f;
        os[j] = f2bf(gelu_f(bf2f(xs[j]) + b));
      }
      *(bf16x8*)(y + i) = out;
    } else {
      for (long j = i; j < n; ++j) {
        float b = has_bias ? bf2f(bias[j % D]) : 0.f;
        y[j] = f2bf(gelu_f(bf2f(x[j]) + b));
      }
    }
  }
}

__global__ void bias_gelu_bwd_bf16(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ bias,
                                   unsigned short* __restrict__ dx,
                                   long n, int D, int has_bias) {
  long i0 = ((long)blockIdx.x * BLOCK + threadIdx.x) * 8;
  long stride = (long)gridDim.x * BLOCK * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n) {
      bf16x8 xv = *(const bf16x8*)(x + i);
      bf16x8 gv = *(const bf16x8*)(dy + i);
      bf16x8 bv;
      if (has_bias) bv = *(const bf16x8*)(bias + (int)(i % D));
      const unsigned short* xs = (const unsigned short*)&xv;
      const unsigned short* gs = (const unsigned short*)&gv;
      const unsigned short* bs = (const unsigned short*)&bv;
      bf16x8 out;
      unsigned short* os = (unsigned short*)&out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float b = has_bias ? bf2f(bs[j]) : 0.f;
        os[j] = f2bf(bf2f(gs[j]) * gelu_df(bf2f(xs[j]) + b));
      }
      *(bf16x8*)(dx + i) = out;
    } else {
      for (long j = i; j < n; ++j) {
        float b = has_bias ? bf2f(bias[j % D]) : 0.f;
        dx[j] = f2bf(bf2f(dy[j]) * gelu_df(bf2f(x[j]) + b));
      }
    }
  }
}

__global__ void bias_gelu_fwd_f32(const float* __restrict__ x,
                                  const float* __restrict__ bias,
                                  float* __restrict__ y,
                                  long n, int D, int has_bias) {
  long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * BLOCK;
  for (long i = i0; i < n; i += stride) {
    float b = has_bias ? bias[i % D] : 0.f;
    y[i] = gelu_f(x[i] + b);
  }
}

__global__ void bias_gelu_bwd_f32(const float* __restrict__ dy,
                                  const float* __restrict__ x,
                                  const float* __restrict__ bias,
                                  float* __restrict__ dx,
                                  long n, int D, int has_bias) {
  long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * BLOCK;
  for (long i = i0; i < n; i += stride) {
    float b = has_bias ? bias[i % D] : 0.f;
    dx[i] = dy[i] * gelu_df(x[i] + b);
  }
}

// -------- fused AdamW on flat fp32 tensors -------------------------------

__global__ void adamw_kernel(float* __restrict__ p, const float* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v,
                             long n, float lr, float beta1, float beta2,
                             float eps, float wd, float bc1, float bc2) {
  long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * BLOCK;
  for (long i = i0; i < n; i += stride) {
    float gi = g[i];
    float pi = p[i] * (1.f - lr * wd);
    float mi = m[i] * beta1 + gi * (1.f - beta1);
    float vi = v[i] * beta2 + gi * gi * (1.f - beta2);
    m[i] = mi;
    v[i] = vi;
    p[i] = pi - lr / bc1 * mi / (sqrtf(vi / bc2) + eps);
  }
}

// -------- EMA: ema = d*ema + (1-d)*p (f32 ema; f32 OR bf16 p) ------------

// vectorized 4-wide (the v10 adamw lesson: scalar dword accesses cap the
// HBM-bound kernels well below the roofline); scalar tail for n % 4
__global__ void ema_kernel(float* __restrict__ ema, const float* __restrict__ p,
                           long n, float decay, int vec4) {
  const long n4 = vec4 ? (n >> 2) : 0;
  long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * BLOCK;
  for (long i = i0; i < n4; i += stride) {
    float4 e = *(const float4*)(ema + i * 4);
    float4 v = *(const float4*)(p + i * 4);
    e.x = decay * e.x + (1.f - decay) * v.x;
    e.y = decay * e.y + (1.f - decay) * v.y;
    e.z = decay * e.z + (1.f - decay) * v.z;
    e.w = decay * e.w + (1.f - decay) * v.w;
    *(float4*)(ema + i * 4) = e;
  }
  for (long i = n4 * 4 + i0; i < n; i += stride)
    ema[i] = decay * ema[i] + (1.f - decay) * p[i];
}

// -------- l2 norm squared (any dtype -> f32 scalar) ----------------------

template <typename T>
__global__ void l2norm_kernel(const T* __restrict__ x, float* __restrict__ out,
                              long n) {
  __shared__ float lds[BLOCK / WAVE];
  long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * BLOCK;
  float acc = 0.f;
  for (long i = i0; i < n; i += stride) {
    float v;
    if constexpr (sizeof(T) == 2) v = bf2f(((const unsigned short*)x)[i]);
    else v = ((const float*)x)[i];
    acc += v * v;
  }
  acc = block_sum<BLOCK>(acc, lds);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

template <typename T>
__global__ void scale_kernel(T* __restrict__ x, long n, float s) {
  long i0 = (long)blockIdx.x * BLOCK + threadIdx.x;
  long stride = (long)gridDim.x * BLOCK;
  for (long i = i0; i < n; i += stride) {
    if constexpr (sizeof(T) == 2) {
      unsigned short* p = (unsigned short*)x;
      p[i] = f2bf(bf2f(p[i]) * s);
    } else {
      ((float*)x)[i] *= s;
    }
  }
}

int ew_grid(long n, int per_thread = 1) {
  long blocks = (n + (long)BLOCK * per_thread - 1) / ((long)BLOCK * per_thread);
  return (int)(blocks > 2048 ? 2048 : (blocks > 0 ? blocks : 1));
}

}  // namespace

// ------------------------------------------------------------------ C++ API

torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  auto y = torch::empty_like(x);
  long n = x.numel();
  int D = x.size(-1);
  int has_bias = bias.numel() > 0 ? 1 : 0;
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    TORCH_CHECK(!has_bias || (D % 8 == 0), "D must be multiple of 8");
    hipLaunchKernelGGL(bias_gelu_fwd_bf16, dim3(ew_grid(n, 8)), dim3(BLOCK), 0,
                       stream, (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)bias.data_ptr(),
                       (unsigned short*)y.data_ptr(), n, D, has_bias);
  } else {
    hipLaunchKernelGGL(bias_gelu_fwd_f32, dim3(ew_grid(n)), dim3(BLOCK), 0,
                       stream, x.data_ptr<float>(), bias.data_ptr<float>(),
                       y.data_ptr<float>(), n, D, has_bias);
  }
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                            torch::Tensor bias) {
  auto dx = torch::empty_like(x);
  long n = x.numel();
  int D = x.size(-1);
  int has_bias = bias.numel() > 0 ? 1 : 0;
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    TORCH_CHECK(!has_bias || (D % 8 == 0), "D must be multiple of 8");
    hipLaunchKernelGGL(bias_gelu_bwd_bf16, dim3(ew_grid(n, 8)), dim3(BLOCK), 0,
                       stream, (const unsigned short*)dy.data_ptr(),
                       (const unsigned short*)x.data_ptr(),
                       (const unsigned short*)bias.data_ptr(),
                       (unsigned short*)dx.data_ptr(), n, D, has_bias);
  } else {
    hipLaunchKernelGGL(bias_gelu_bwd_f32, dim3(ew_grid(n)), dim3(BLOCK), 0,
                       stream, dy.data_ptr<float>(), x.data_ptr<float>(),
                       bias.data_ptr<float>(), dx.data_ptr<float>(),
                       n, D, has_bias);
  }
  HIP_CHECK_LAST();
  return dx;
}

void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, long step, double lr, double beta1,
                double beta2, double eps, double wd) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat);
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous());
  long n = p.numel();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(adamw_kernel, dim3(ew_grid(n)), dim3(BLOCK), 0, stream,
                     p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n,
                     (float)lr, (float)beta1, (float)beta2, (float)eps,
                     (float)wd, bc1, bc2);
  HIP_CHECK_LAST();
}

namespace {
__global__ void ema_bf16_kernel(float* __restrict__ ema,
                                const unsigned short* __restrict__ p, long n,
                                float decay, int vec4) {
  // GRID-STRIDED: ew_grid caps launches at 2048 blocks — a plain
  // one-element-per-thread body here updated only the first 524k elements
  // of an 8B-param shard (caught by the large-n parity test); 4-wide
  // vectorized like the fp32 variant (vec4=0: unaligned narrow() views
  // fall back to the scalar loop)
  const long n4 = vec4 ? (n >> 2) : 0;
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n4; i += stride) {
    float4 e = *(const float4*)(ema + i * 4);
    ushort4 v = *(const ushort4*)(p + i * 4);
    e.x = decay * e.x + (1.f - decay) * bf2f(v.x);
    e.y = decay * e.y + (1.f - decay) * bf2f(v.y);
    e.z = decay * e.z + (1.f - decay) * bf2f(v.z);
    e.w = decay * e.w + (1.f - decay) * bf2f(v.w);
    *(float4*)(ema + i * 4) = e;
  }
  for (long i = n4 * 4 + i0; i < n; i += stride)
    ema[i] = decay * ema[i] + (1.f - decay) * bf2f(p[i]);
}
}  // namespace

void ema_update(torch::Tensor ema, torch::Tensor p, double decay) {
  TORCH_CHECK(ema.is_cuda() && ema.scalar_type() == torch::kFloat);
  long n = ema.numel();
  auto stream = at::cuda::getCurrentHIPStream();
  // vector path needs 16B-aligned ema and 16B/8B-aligned p: EMA views are
  // narrow() slices at arbitrary element offsets
  const int vec4 = (((unsigned long long)ema.data_ptr() & 15) == 0 &&
                    ((unsigned long long)p.data_ptr() &
                     (p.scalar_type() == torch::kBFloat16 ? 7 : 15)) == 0)
                   ? 1 : 0;
  if (p.scalar_type() == torch::kBFloat16) {
    // bf16 model params feed the fp32 EMA shard directly — the Python
    // lerp_(p.float()) fallback this replaces materialized an fp32 copy
    // of every param each step (r02 Llama profile: 14% eager elementwise)
    hipLaunchKernelGGL(ema_bf16_kernel, dim3(ew_grid(n, 4)), dim3(BLOCK), 0,
                       stream, ema.data_ptr<float>(),
                       (const unsigned short*)p.data_ptr(), n, (float)decay,
                       vec4);
  } else {
    hipLaunchKernelGGL(ema_kernel, dim3(ew_grid(n, 4)), dim3(BLOCK), 0,
                       stream, ema.data_ptr<float>(), p.data_ptr<float>(), n,
                       (float)decay, vec4);
  }
  HIP_CHECK_LAST();
}

torch::Tensor l2norm_sq(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  auto out = torch::zeros({}, x.options().dtype(torch::kFloat));
  long n = x.numel();
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(l2norm_kernel<bf16_t>, dim3(ew_grid(n)), dim3(BLOCK), 0,
                       stream, (const bf16_t*)x.data_ptr(),
                       out.data_ptr<float>(), n);
  } else {
    hipLaunchKernelGGL(l2norm_kernel<float>, dim3(ew_grid(n)), dim3(BLOCK), 0,
                       stream, x.data_ptr<float>(), out.data_ptr<float>(), n);
  }
  HIP_CHECK_LAST();
  return out;
}

void scale_inplace(torch::Tensor x, double s) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  long n = x.numel();
  auto stream = at::cuda::getCurrentHIPStream();
  if (x.scalar_type() == torch::kBFloat16) {
    hipLaunchKernelGGL(scale_kernel<bf16_t>, dim3(ew_grid(n)), dim3(BLOCK), 0,
                       stream, (bf16_t*)x.data_ptr(), n, (float)s);
  } else {
    hipLaunchKernelGGL(scale_kernel<float>, dim3(ew_grid(n)), dim3(BLOCK), 0,
                       stream, x.data_ptr<float>(), n, (float)s);
  }
  HIP_CHECK_LAST();
}

// -------- multi-tensor AdamW: one launch for ALL params ---------------------
// Chunk table built once host-side; per chunk: param id + element offset.
// Reads bf16 grads and writes bf16 params IN the same pass (the v1 profile
// showed the foreach grad-cast + param-copyback pairs costing ~3.6% of step
// on top of the f32 flat update).

namespace {

constexpr long MT_CHUNK = 1 << 16;

__global__ void multi_adamw_kernel(const int* __restrict__ cpid,
                                   const long* __restrict__ coff,
                                   const long* __restrict__ pptrs,
                                   const long* __restrict__ gptrs,
                                   const long* __restrict__ mptrs,
                                   const long* __restrict__ moffs,
                                   const long* __restrict__ numels,
                                   int nchunks,
                                   float* __restrict__ m,
                                   float* __restrict__ v,
                                   float lr, float beta1, float beta2,
                                   float eps, float wd, float bc1, float bc2,
                                   int param_bf16, int grad_bf16) {
  const int c = blockIdx.x;
  if (c >= nchunks) return;
  const int pid = cpid[c];
  const long off = coff[c];
  const long n = numels[pid];
  const long end = off + MT_CHUNK < n ? off + MT_CHUNK : n;
  const long mvbase = moffs[pid];     // m/v flat offset
  float* master = (float*)mptrs[pid];  // per-param master (MAY alias param)
  void* pp = (void*)pptrs[pid];
  const void* gp = (const void*)gptrs[pid];
  // grad-None params are SKIPPED entirely (no decay, no moment update) —
  // stock torch.optim semantics; substituting grad=0 silently decayed
  // frozen params toward zero.
  if (gp == nullptr) return;
  // HBM-bound: ~26 B/element of read+write traffic, so dwordx4 accesses are
  // what decides the rate.  All pointers/offsets are 16B-aligned in practice
  // (torch allocations + 64Ki chunking); fall back to scalars otherwise.
  const bool vec4 = ((off | mvbase) & 3) == 0 &&
      (((long)master | (long)pp | (long)gp) & 15) == 0;
  if (vec4) {
    const long nv = (end - off) >> 2;   // full float4 groups in this chunk
    for (long g = threadIdx.x; g < nv; g += blockDim.x) {
      const long i = off + g * 4;
      const long k = mvbase + i;
      float4 gi4;
      if (gp == nullptr) gi4 = make_float4(0.f, 0.f, 0.f, 0.f);
      else if (grad_bf16) {
        ushort4 gu = *(const ushort4*)((const unsigned short*)gp + i);
        gi4 = make_float4(bf2f(gu.x), bf2f(gu.y), bf2f(gu.z), bf2f(gu.w));
      } else {
        gi4 = *(const float4*)((const float*)gp + i);
      }
      float4 p4 = *(const float4*)(master + i);
      float4 m4 = *(const float4*)(m + k);
      float4 v4 = *(const float4*)(v + k);
      float* pi = (float*)&p4;
      float* mi = (float*)&m4;
      float* vi = (float*)&v4;
      const float* gi = (const float*)&gi4;
      ushort4 pb;
      unsigned short* pbs = (unsigned short*)&pb;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float pp_ = pi[j] * (1.f - lr * wd);
        float mm_ = mi[j] * beta1 + gi[j] * (1.f - beta1);
        float vv_ = vi[j] * beta2 + gi[j] * gi[j] * (1.f - beta2);
        mi[j] = mm_;
        vi[j] = vv_;
        pp_ = pp_ - lr / bc1 * mm_ / (sqrtf(vv_ / bc2) + eps);
        pi[j] = pp_;
        pbs[j] = f2bf(pp_);
      }
      *(float4*)(m + k) = m4;
      *(float4*)(v + k) = v4;
      *(float4*)(master + i) = p4;
      if (param_bf16) *(ushort4*)((unsigned short*)pp + i) = pb;
    }
    // scalar tail: at most 3 elements at the chunk end
    for (long i = off + nv * 4 + threadIdx.x; i < end; i += blockDim.x) {
      float gi;
      if (gp == nullptr) gi = 0.f;
      else if (grad_bf16) gi = bf2f(((const unsigned short*)gp)[i]);
      else gi = ((const float*)gp)[i];
      const long k = mvbase + i;
      float pi = master[i] * (1.f - lr * wd);
      float mi = m[k] * beta1 + gi * (1.f - beta1);
      float vi = v[k] * beta2 + gi * gi * (1.f - beta2);
      m[k] = mi;
      v[k] = vi;
      pi = pi - lr / bc1 * mi / (sqrtf(vi / bc2) + eps);
      master[i] = pi;
      if (param_bf16) ((unsigned short*)pp)[i] = f2bf(pi);
    }
    return;
  }
  for (long i = off + threadIdx.x; i < end; i += blockDim.x) {
    float gi;
    if (gp == nullptr) gi = 0.f;
    else if (grad_bf16) gi = bf2f(((const unsigned short*)gp)[i]);
    else gi = ((const float*)gp)[i];
    const long k = mvbase + i;
    float pi = master[i] * (1.f - lr * wd);
    float mi = m[k] * beta1 + gi * (1.f - beta1);
    float vi = v[k] * beta2 + gi * gi * (1.f - beta2);
    m[k] = mi;
    v[k] = vi;
    pi = pi - lr / bc1 * mi / (sqrtf(vi / bc2) + eps);
    master[i] = pi;
    if (param_bf16) ((unsigned short*)pp)[i] = f2bf(pi);
  }
}

}  // namespace

void multi_adamw_step(torch::Tensor cpid, torch::Tensor coff,
                      torch::Tensor pptrs, torch::Tensor gptrs,
                      torch::Tensor mptrs, torch::Tensor moffs,
                      torch::Tensor numels,
                      torch::Tensor m, torch::Tensor v,
                      long step, double lr, double beta1, double beta2,
                      double eps, double wd, bool param_bf16,
                      bool grad_bf16) {
  TORCH_CHECK(m.is_cuda() && m.scalar_type() == torch::kFloat);
  int nchunks = cpid.numel();
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(multi_adamw_kernel, dim3(nchunks), dim3(BLOCK), 0,
                     stream, cpid.data_ptr<int>(), coff.data_ptr<long>(),
                     pptrs.data_ptr<long>(), gptrs.data_ptr<long>(),
                     mptrs.data_ptr<long>(), moffs.data_ptr<long>(),
                     numels.data_ptr<long>(),
                     nchunks, m.data_ptr<float>(),
                     v.data_ptr<float>(), (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)wd, bc1, bc2,
                     param_bf16 ? 1 : 0, grad_bf16 ? 1 : 0);
  HIP_CHECK_LAST();
}
