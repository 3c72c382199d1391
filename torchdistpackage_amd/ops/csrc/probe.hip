// MFMA fragment-layout probe: computes D = A(16x32) @ B(32x16) with
// v_mfma_f32_16x16x32_bf16 under the layout assumption used by attention.hip
// (A: lane l holds row l&15, k = (l>>4)*8 + j ; B: col l&15, same k ;
//  C/D: col = l&15, row = (l>>4)*4 + reg).
// A GPU test compares this against torch.matmul — if the assumption is wrong
// the test localizes exactly which mapping to fix.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void mfma_probe_kernel(const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b,
                                  float* __restrict__ d) {
  const int lane = threadIdx.x & 63;
  const int l15 = lane & 15;
  const int lg = lane >> 4;
  unsigned short af[8], bf[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[l15 * 32 + lg * 8 + j];        // A[row][k], row-major 16x32
    bf[j] = b[(lg * 8 + j) * 16 + l15];      // B[k][col], row-major 32x16
  }
  f32x4 acc = (f32x4)(0.f);
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      *(const bf16x8_v*)af, *(const bf16x8_v*)bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(lg * 4 + r) * 16 + l15] = acc[r];
}

}  // namespace

torch::Tensor mfma_probe_16x16x32(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({32, 16}));
  auto d = torch::empty({16, 16}, a.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)a.contiguous().data_ptr(),
                     (const unsigned short*)b.contiguous().data_ptr(),
                     d.data_ptr<float>());
  HIP_CHECK_LAST();
  return d;
}

namespace {

typedef __attribute__((ext_vector_type(16))) float f32x16;

// 32x32x16 layout probe: A[32][16] @ B[16][32] with the assumed maps
//   A: lane l holds A[row=l&31][k=(l>>5)*8+j]  (8 bf16)
//   B: lane l holds B[k=(l>>5)*8+j][col=l&31]
//   C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)   (16 f32)
__global__ void mfma_probe32_kernel(const unsigned short* __restrict__ a,
                                    const unsigned short* __restrict__ b,
                                    float* __restrict__ d) {
  const int lane = threadIdx.x & 63;
  const int l31 = lane & 31;
  const int hi = lane >> 5;
  unsigned short af[8], bf[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[l31 * 16 + hi * 8 + j];
    bf[j] = b[(hi * 8 + j) * 32 + l31];
  }
  f32x16 acc = (f32x16)(0.f);
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
      *(const bf16x8_v*)af, *(const bf16x8_v*)bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    d[row * 32 + l31] = acc[r];
  }
}

}  // namespace

torch::Tensor mfma_probe_32x32x16(torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.is_cuda() && a.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({32, 16}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({16, 32}));
  auto d = torch::empty({32, 32}, a.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, stream,
                     (const unsigned short*)a.contiguous().data_ptr(),
                     (const unsigned short*)b.contiguous().data_ptr(),
                     d.data_ptr<float>());
  HIP_CHECK_LAST();
  return d;
}

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 layout probe.  Hardware-verified semantics (this
// probe, marker fill + four address patterns — uniform, quad-aliased,
// per-lane distinct, cross-group distinct):
//
//   lane l, element j = lds[ floor8B(addr(lane g0 + ((l>>2)&3) + 4j))/2
//                            + (l&3) ]        (g0 = 16*(l/16))
//
// i.e. within each 16-lane group, sub-quad q=(l>>2)&3 reads the 4x4 bf16
// tile whose four ROW addresses come from lanes {q, q+4, q+8, q+12},
// transposed (lane column = l&3); four independent tiles per group, 16
// distinct columns; address bits below 8-byte alignment are floored.
// tests/test_ops_gpu.py asserts this model against all four patterns.
// ---------------------------------------------------------------------------

__global__ void tr16_probe_kernel(unsigned short* __restrict__ out,
                                  int addr_mode) {
  __shared__ __attribute__((aligned(16))) unsigned short lds[256];
  int tid = threadIdx.x;
  for (int i = tid; i < 256; i += blockDim.x)
    lds[i] = (unsigned short)i;          // marker = element index
  __syncthreads();
  if (tid < 64) {
    // per-lane byte address: the ISA form takes a VGPR address; probe both
    // a uniform base and a per-lane 8B-aligned base to see which operand
    // model holds (guide: per-lane addr, no internal lane offset beyond
    // the group gather)
    // the DS address operand is an LDS-segment byte offset: convert the
    // generic (flat) shared pointer through address_space(3)
    unsigned base = (unsigned)(unsigned long long)
        (__attribute__((address_space(3))) unsigned short*)lds;
    unsigned off = 0;
    if (addr_mode == 1) off = (unsigned)((tid & 15) * 2);
    else if (addr_mode == 2) off = (unsigned)((tid & 15) * 8);
    else if (addr_mode == 3) off = (unsigned)((tid & 63) * 8);
    unsigned addr = base + off;
    unsigned long long got;
    asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                 : "=v"(got) : "v"(addr));
    unsigned short* g = (unsigned short*)&got;
#pragma unroll
    for (int j = 0; j < 4; ++j) out[tid * 4 + j] = g[j];
  }
}

torch::Tensor tr16_probe(long addr_mode) {
  auto out = torch::zeros({64, 4},
                          torch::dtype(torch::kInt16).device(torch::kCUDA));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (unsigned short*)out.data_ptr(), (int)addr_mode);
  HIP_CHECK_LAST();
  return out;
}
