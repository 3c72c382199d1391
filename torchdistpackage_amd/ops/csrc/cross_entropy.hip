// Fused cross-entropy over a large vocab (bf16 logits, fp32 math).
//
// Replaces the eager chain logits.float() -> log_softmax -> nll (+ the
// backward softmax) which materializes fp32 logits (1.6 GB at GPT-2 shapes)
// and re-reads them several times.  Here:
//   fwd: ONE online pass per row (running max + rescaled expsum, the same
//        scheme as flash attention) -> per-row LSE + loss
//   bwd: ONE pass: dlogits = (softmax - onehot) * gscale, written bf16
// Loss reduction (mean) happens on the tiny per-row vector in torch.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int BLOCK = 256;

typedef ushort4 u4;

__global__ void ce_fwd_kernel(const unsigned short* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ lse,
                              float* __restrict__ loss,
                              long N, long V) {
  __shared__ float lds_m[BLOCK / WAVE];
  __shared__ float lds_z[BLOCK / WAVE];
  const long row = blockIdx.x;
  if (row >= N) return;
  const unsigned short* x = logits + row * V;
  float m = -1e30f, z = 0.f;
  for (long i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
    if (i + 8 <= V) {
      u4 a = *(const u4*)(x + i);
      u4 b = *(const u4*)(x + i + 4);
      float vals[8] = {bf2f(a.x), bf2f(a.y), bf2f(a.z), bf2f(a.w),
                       bf2f(b.x), bf2f(b.y), bf2f(b.z), bf2f(b.w)};
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = vals[j];
        if (v > m) { z *= __expf(m - v); m = v; }
        z += __expf(v - m);
      }
    } else {
      for (long j = i; j < V; ++j) {
        float v = bf2f(x[j]);
        if (v > m) { z *= __expf(m - v); m = v; }
        z += __expf(v - m);
      }
    }
  }
  // combine (m, z) across the wave then across waves
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float om = __shfl_xor(m, off, 64);
    float oz = __shfl_xor(z, off, 64);
    float nm = fmaxf(m, om);
    z = z * __expf(m - nm) + oz * __expf(om - nm);
    m = nm;
  }
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & 63) == 0) { lds_m[wid] = m; lds_z[wid] = z; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float fm = lds_m[0], fz = lds_z[0];
#pragma unroll
    for (int i = 1; i < BLOCK / WAVE; ++i) {
      float nm = fmaxf(fm, lds_m[i]);
      fz = fz * __expf(fm - nm) + lds_z[i] * __expf(lds_m[i] - nm);
      fm = nm;
    }
    float l = fm + __logf(fz);
    lse[row] = l;
    long t = targets[row];
    loss[row] = (t >= 0 && t < V) ? l - bf2f(x[t]) : 0.f;
  }
}

// vocab-parallel variant: same online pass, but emits the LOCAL lse and
// the LOCAL target logit (0 when the target lives on another TP rank's
// vocab shard) — combined across ranks by one lse all-gather + one tgt
// all-reduce (parallel/tensor/vocab.py).  ce_bwd_kernel is reused as-is
// with the GLOBAL lse (an out-of-range target simply never matches the
// one-hot test).
__global__ void ce_partial_fwd_kernel(
    const unsigned short* __restrict__ logits,
    const long* __restrict__ targets,   // local (t - vocab_start), or -1
    float* __restrict__ lse, float* __restrict__ tgt, long N, long V) {
  __shared__ float lds_m[BLOCK / WAVE];
  __shared__ float lds_z[BLOCK / WAVE];
  const long row = blockIdx.x;
  if (row >= N) return;
  const unsigned short* x = logits + row * V;
  float m = -1e30f, z = 0.f;
  for (long i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
    if (i + 8 <= V) {
      u4 a = *(const u4*)(x + i);
      u4 b = *(const u4*)(x + i + 4);
      float vals[8] = {bf2f(a.x), bf2f(a.y), bf2f(a.z), bf2f(a.w),
                       bf2f(b.x), bf2f(b.y), bf2f(b.z), bf2f(b.w)};
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = vals[j];
        if (v > m) { z *= __expf(m - v); m = v; }
        z += __expf(v - m);
      }
    } else {
      for (long j = i; j < V; ++j) {
        float v = bf2f(x[j]);
        if (v > m) { z *= __expf(m - v); m = v; }
        z += __expf(v - m);
      }
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float om = __shfl_xor(m, off, 64);
    float oz = __shfl_xor(z, off, 64);
    float nm = fmaxf(m, om);
    z = z * __expf(m - nm) + oz * __expf(om - nm);
    m = nm;
  }
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & 63) == 0) { lds_m[wid] = m; lds_z[wid] = z; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float fm = lds_m[0], fz = lds_z[0];
#pragma unroll
    for (int i = 1; i < BLOCK / WAVE; ++i) {
      float nm = fmaxf(fm, lds_m[i]);
      fz = fz * __expf(fm - nm) + lds_z[i] * __expf(lds_m[i] - nm);
      fm = nm;
    }
    lse[row] = fm + __logf(fz);
    long t = targets[row];
    tgt[row] = (t >= 0 && t < V) ? bf2f(x[t]) : 0.f;
  }
}

__global__ void ce_bwd_kernel(const unsigned short* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              unsigned short* __restrict__ dlogits,
                              const float* __restrict__ gin, long N, long V) {
  const long row = blockIdx.x;
  if (row >= N) return;
  const float gscale = gin[0] / (float)N;  // upstream grad of the mean
  const unsigned short* x = logits + row * V;
  unsigned short* dx = dlogits + row * V;
  const float l = lse[row];
  const long t = targets[row];
  for (long i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
    if (i + 8 <= V) {
      u4 a = *(const u4*)(x + i);
      u4 b = *(const u4*)(x + i + 4);
      float vals[8] = {bf2f(a.x), bf2f(a.y), bf2f(a.z), bf2f(a.w),
                       bf2f(b.x), bf2f(b.y), bf2f(b.z), bf2f(b.w)};
      unsigned short outs[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(vals[j] - l);
        if (i + j == t) p -= 1.f;
        outs[j] = f2bf(p * gscale);
      }
      *(u4*)(dx + i) = *(u4*)outs;
      *(u4*)(dx + i + 4) = *(u4*)(outs + 4);
    } else {
      for (long j = i; j < V; ++j) {
        float p = __expf(bf2f(x[j]) - l);
        if (j == t) p -= 1.f;
        dx[j] = f2bf(p * gscale);
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor targets) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == torch::kLong);
  const long N = logits.size(0), V = logits.size(1);
  auto lse = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto loss = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_fwd_kernel, dim3((unsigned)N), dim3(BLOCK), 0, stream,
                     (const unsigned short*)logits.data_ptr(),
                     targets.contiguous().data_ptr<long>(),
                     lse.data_ptr<float>(), loss.data_ptr<float>(), N, V);
  HIP_CHECK_LAST();
  return {loss, lse};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor targets,
                     torch::Tensor lse, torch::Tensor grad_out) {
  const long N = logits.size(0), V = logits.size(1);
  auto dlogits = torch::empty_like(logits);
  auto stream = at::cuda::getCurrentHIPStream();
  TORCH_CHECK(grad_out.is_cuda() && grad_out.scalar_type() == torch::kFloat);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3((unsigned)N), dim3(BLOCK), 0, stream,
                     (const unsigned short*)logits.data_ptr(),
                     targets.contiguous().data_ptr<long>(),
                     lse.data_ptr<float>(),
                     (unsigned short*)dlogits.data_ptr(),
                     grad_out.data_ptr<float>(), N, V);
  HIP_CHECK_LAST();
  return dlogits;
}

std::vector<torch::Tensor> ce_partial_fwd(torch::Tensor logits,
                                          torch::Tensor targets) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == torch::kLong);
  const long N = logits.size(0), V = logits.size(1);
  auto lse = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto tgt = torch::empty({N}, logits.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(ce_partial_fwd_kernel, dim3((unsigned)N), dim3(BLOCK),
                     0, stream, (const unsigned short*)logits.data_ptr(),
                     targets.data_ptr<long>(), lse.data_ptr<float>(),
                     tgt.data_ptr<float>(), N, V);
  HIP_CHECK_LAST();
  return {lse, tgt};
}
