// Fused RoPE (rotate-half) and SwiGLU kernels for the Llama hot path.
//
// Replaces the eager chains the round-1 model used
// (models/llama.py: rotate-half = 2 cats + 4 muls per call per layer;
// silu(w1x)*w3x unfused) — both are HBM-bound elementwise ops, so the win
// is one read+write pass instead of 4-6 (plus the launch count).
// BASELINE.json's north star names RoPE as a hand-written CDNA4 kernel.
//
// Conventions: bf16 tensors, fp32 cos/sin tables (S, D/2), fp32 math,
// ushort8 (16 B/lane) accesses — the bandwidth rules from the guide that
// took bias_gelu to the roofline in round 1.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

constexpr int BLOCK = 256;

DEVINL void load8(const unsigned short* p, float* f) {
  ushort4 a = *(const ushort4*)p;
  ushort4 b = *(const ushort4*)(p + 4);
  f[0] = bf2f(a.x); f[1] = bf2f(a.y); f[2] = bf2f(a.z); f[3] = bf2f(a.w);
  f[4] = bf2f(b.x); f[5] = bf2f(b.y); f[6] = bf2f(b.z); f[7] = bf2f(b.w);
}

DEVINL void store8(unsigned short* p, const float* f) {
  ushort4 a, b;
  a.x = f2bf(f[0]); a.y = f2bf(f[1]); a.z = f2bf(f[2]); a.w = f2bf(f[3]);
  b.x = f2bf(f[4]); b.y = f2bf(f[5]); b.z = f2bf(f[6]); b.w = f2bf(f[7]);
  *(ushort4*)p = a;
  *(ushort4*)(p + 4) = b;
}

// rows = B*H*S (seq position = row % S), rotate-half pairs (i, i+D/2).
// FWD=false computes the transpose rotation (gradient).  The INPUT may be
// any-strided over (B,H,S) with d contiguous (e.g. a permuted view of the
// (S,B,H*hd) projection output — reading it in place removes the
// permute-contiguous copy per q/k per layer); the output is written
// contiguous (B,H,S,D).
template <bool FWD>
__global__ void rope_kernel(const unsigned short* __restrict__ x,
                            unsigned short* __restrict__ y,
                            const float* __restrict__ cs,   // (S, D/2) cos
                            const float* __restrict__ sn,   // (S, D/2) sin
                            long rows, int S, int D, int pos0,
                            long H, long xsb, long xsh, long xss,
                            const long* __restrict__ pos0p) {
  const int half = D >> 1;
  const int chunks = half >> 3;                 // 8 elems per thread-chunk
  const long total = rows * chunks;
  // pos0p: optional DEVICE position (hipGraph-capturable decode — the
  // graph replays with the position tensor advanced in place)
  const int p0 = pos0p ? (int)*pos0p : pos0;
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < total;
       g += (long)gridDim.x * blockDim.x) {
    const long row = g / chunks;
    const int c = (int)(g - row * chunks) * 8;
    const int s = (int)(row % S) + p0;
    const long bh = row / S;
    const long xbase = (bh / H) * xsb + (bh - (bh / H) * H) * xsh +
                       (row % S) * xss + c;
    const long base = row * D + c;
    float x1[8], x2[8], co[8], si[8];
    load8(x + xbase, x1);
    load8(x + xbase + half, x2);
#pragma unroll
    for (int j = 0; j < 8; j += 4) {
      *(float4*)(co + j) = *(const float4*)(cs + (long)s * half + c + j);
      *(float4*)(si + j) = *(const float4*)(sn + (long)s * half + c + j);
    }
    float y1[8], y2[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (FWD) {
        y1[j] = x1[j] * co[j] - x2[j] * si[j];
        y2[j] = x2[j] * co[j] + x1[j] * si[j];
      } else {   // transpose (inverse rotation): grads
        y1[j] = x1[j] * co[j] + x2[j] * si[j];
        y2[j] = x2[j] * co[j] - x1[j] * si[j];
      }
    }
    store8(y + base, y1);
    store8(y + base + half, y2);
  }
}

// silu(a) * b and its backward (recomputes sigmoid from a: saves storing
// the activation).  exp via the hardware exp2 unit (log2e-scaled).
DEVINL float sigmoidf_fast(float x) {
  return 1.0f / (1.0f + __builtin_exp2f(-1.442695041f * x));
}

__global__ void swiglu_fwd_kernel(const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b,
                                  unsigned short* __restrict__ out,
                                  long n) {
  const long g = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (g >= n) return;
  float fa[8], fb[8], fo[8];
  load8(a + g, fa);
  load8(b + g, fb);
#pragma unroll
  for (int j = 0; j < 8; ++j)
    fo[j] = fa[j] * sigmoidf_fast(fa[j]) * fb[j];
  store8(out + g, fo);
}

__global__ void swiglu_bwd_kernel(const unsigned short* __restrict__ dy,
                                  const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b,
                                  unsigned short* __restrict__ da,
                                  unsigned short* __restrict__ db,
                                  long n) {
  const long g = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (g >= n) return;
  float fdy[8], fa[8], fb[8], fda[8], fdb[8];
  load8(dy + g, fdy);
  load8(a + g, fa);
  load8(b + g, fb);
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const float sig = sigmoidf_fast(fa[j]);
    const float silu = fa[j] * sig;
    fda[j] = fdy[j] * fb[j] * (sig + silu * (1.0f - sig));
    fdb[j] = fdy[j] * silu;
  }
  store8(da + g, fda);
  store8(db + g, fdb);
}

// per-expert bias + tanh-GELU over (E, N, H): bias index = (e, i % H).
// Split out because hipBLASLt's baddbmm faulted on a stride-0 broadcast
// batch bias at MoE bench scale (profiles/r02_notes.md) — the batched
// expert path uses plain bmm + this fused epilogue instead.
DEVINL float gelu_tanh_f(float u) {
  float e = __builtin_exp2f((u + 0.044715f * u * u * u) *
                            0.7978845608f * 2.885390082f);
  float t = 1.f - 2.f / (e + 1.f);
  return 0.5f * u * (1.f + t);
}

DEVINL float gelu_tanh_df(float u) {
  float u2 = u * u;
  float e = __builtin_exp2f((u + 0.044715f * u * u2) *
                            0.7978845608f * 2.885390082f);
  float t = 1.f - 2.f / (e + 1.f);
  return 0.5f * (1.f + t) +
         0.5f * u * (1.f - t * t) * 0.7978845608f *
             (1.f + 3.f * 0.044715f * u2);
}

__global__ void bgelu_b_fwd_kernel(const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ bias,
                                   unsigned short* __restrict__ y,
                                   long n, long NH, int H) {
  const long g = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (g >= n) return;
  const long e = g / NH;
  const int j = (int)(g % H);
  float xv[8], bv[8], ov[8];
  load8(x + g, xv);
  load8(bias + e * H + j, bv);
#pragma unroll
  for (int k = 0; k < 8; ++k) ov[k] = gelu_tanh_f(xv[k] + bv[k]);
  store8(y + g, ov);
}

__global__ void bgelu_b_bwd_kernel(const unsigned short* __restrict__ dy,
                                   const unsigned short* __restrict__ x,
                                   const unsigned short* __restrict__ bias,
                                   unsigned short* __restrict__ dx,
                                   long n, long NH, int H) {
  const long g = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (g >= n) return;
  const long e = g / NH;
  const int j = (int)(g % H);
  float dv[8], xv[8], bv[8], ov[8];
  load8(dy + g, dv);
  load8(x + g, xv);
  load8(bias + e * H + j, bv);
#pragma unroll
  for (int k = 0; k < 8; ++k) ov[k] = dv[k] * gelu_tanh_df(xv[k] + bv[k]);
  store8(dx + g, ov);
}

void check_bf16(const torch::Tensor& t, const char* n) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
              t.is_contiguous(), n, " must be contiguous bf16 CUDA");
}

}  // namespace

// x (B, H, S, D) bf16 contiguous; cos/sin (>=S+pos0, D/2) fp32.
torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cs, torch::Tensor sn,
                         long pos0, bool fwd) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.dim() == 4 && x.stride(3) == 1,
              "rope: x must be 4-D bf16 CUDA with contiguous head dim");
  TORCH_CHECK(cs.scalar_type() == torch::kFloat &&
              cs.is_contiguous() && sn.is_contiguous());
  const int D = x.size(3), S = x.size(2);
  TORCH_CHECK(D % 16 == 0 && cs.size(1) == D / 2);
  const long rows = (long)x.size(0) * x.size(1) * S;
  auto y = torch::empty({x.size(0), x.size(1), x.size(2), x.size(3)},
                        x.options());
  const long total = rows * (D / 16);
  const long nb = (total + BLOCK - 1) / BLOCK;
  auto stream = at::cuda::getCurrentHIPStream();
  auto kern = fwd ? rope_kernel<true> : rope_kernel<false>;
  hipLaunchKernelGGL(kern, dim3((unsigned)std::min(nb, (long)65535 * 8)),
                     dim3(BLOCK), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (unsigned short*)y.data_ptr(), cs.data_ptr<float>(),
                     sn.data_ptr<float>(), rows, S, D, (int)pos0,
                     x.size(1), x.stride(0), x.stride(1), x.stride(2),
                     (const long*)nullptr);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor rope_apply_pos(torch::Tensor x, torch::Tensor cs,
                             torch::Tensor sn, torch::Tensor pos,
                             bool fwd) {
  // device-position variant for hipGraph-captured decode
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.dim() == 4 && x.stride(3) == 1,
              "rope: x must be 4-D bf16 CUDA with contiguous head dim");
  TORCH_CHECK(cs.scalar_type() == torch::kFloat &&
              cs.is_contiguous() && sn.is_contiguous());
  TORCH_CHECK(pos.is_cuda() && pos.scalar_type() == torch::kLong &&
              pos.numel() == 1);
  const int D = x.size(3), S = x.size(2);
  TORCH_CHECK(D % 16 == 0 && cs.size(1) == D / 2);
  const long rows = (long)x.size(0) * x.size(1) * S;
  auto y = torch::empty({x.size(0), x.size(1), x.size(2), x.size(3)},
                        x.options());
  const long total = rows * (D / 16);
  const long nb = (total + BLOCK - 1) / BLOCK;
  auto stream = at::cuda::getCurrentHIPStream();
  auto kern = fwd ? rope_kernel<true> : rope_kernel<false>;
  hipLaunchKernelGGL(kern, dim3((unsigned)std::min(nb, (long)65535 * 8)),
                     dim3(BLOCK), 0, stream,
                     (const unsigned short*)x.data_ptr(),
                     (unsigned short*)y.data_ptr(), cs.data_ptr<float>(),
                     sn.data_ptr<float>(), rows, S, D, 0,
                     x.size(1), x.stride(0), x.stride(1), x.stride(2),
                     (const long*)pos.data_ptr());
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor swiglu_fwd(torch::Tensor a, torch::Tensor b) {
  check_bf16(a, "a"); check_bf16(b, "b");
  TORCH_CHECK(a.numel() == b.numel() && a.numel() % 8 == 0);
  auto out = torch::empty_like(a);
  const long n = a.numel();
  const long nb = (n / 8 + BLOCK - 1) / BLOCK;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3((unsigned)nb), dim3(BLOCK), 0,
                     stream, (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)b.data_ptr(),
                     (unsigned short*)out.data_ptr(), n);
  HIP_CHECK_LAST();
  return out;
}

std::vector<torch::Tensor> swiglu_bwd(torch::Tensor dy, torch::Tensor a,
                                      torch::Tensor b) {
  check_bf16(dy, "dy"); check_bf16(a, "a"); check_bf16(b, "b");
  auto da = torch::empty_like(a);
  auto db = torch::empty_like(b);
  const long n = a.numel();
  const long nb = (n / 8 + BLOCK - 1) / BLOCK;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3((unsigned)nb), dim3(BLOCK), 0,
                     stream, (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)a.data_ptr(),
                     (const unsigned short*)b.data_ptr(),
                     (unsigned short*)da.data_ptr(),
                     (unsigned short*)db.data_ptr(), n);
  HIP_CHECK_LAST();
  return {da, db};
}

// x (E, N, H) bf16 contiguous, bias (E, H) bf16; H % 8 == 0.
torch::Tensor bgelu_b_fwd(torch::Tensor x, torch::Tensor bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 &&
              x.dim() == 3 && x.is_contiguous() && bias.is_contiguous());
  const int H = x.size(2);
  TORCH_CHECK(H % 8 == 0 && bias.size(0) == x.size(0) &&
              bias.size(1) == H);
  auto y = torch::empty_like(x);
  const long n = x.numel();
  const long NH = (long)x.size(1) * H;
  const long nb = (n / 8 + 255) / 256;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(bgelu_b_fwd_kernel, dim3((unsigned)nb), dim3(256), 0,
                     stream, (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)bias.data_ptr(),
                     (unsigned short*)y.data_ptr(), n, NH, H);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor bgelu_b_bwd(torch::Tensor dy, torch::Tensor x,
                          torch::Tensor bias) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const int H = x.size(2);
  auto dx = torch::empty_like(x);
  const long n = x.numel();
  const long NH = (long)x.size(1) * H;
  const long nb = (n / 8 + 255) / 256;
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(bgelu_b_bwd_kernel, dim3((unsigned)nb), dim3(256), 0,
                     stream, (const unsigned short*)dy.data_ptr(),
                     (const unsigned short*)x.data_ptr(),
                     (const unsigned short*)bias.data_ptr(),
                     (unsigned short*)dx.data_ptr(), n, NH, H);
  HIP_CHECK_LAST();
  return dx;
}
