// Hand-written CDNA4 bf16 GEMM for the transformer hot shapes (gfx950).
//
// Replaces hipBLASLt on the model GEMMs the reference delegates to cuBLAS
// (/root/reference/torchdistpackage/parallel/tensor_parallel/tp_utils.py:171
// torch.matmul): QKV / out-proj / MLP fprop, dgrad, and the K=16384 wgrad
// family that hipBLASLt runs at ~1.03 PF/s (profiles/r01_notes.md item 4).
//
// Structure = the guide's 256x256 8-phase template (cdna_hip_programming.md
// §5): BM=BN=256, 8 waves (2Mx4N, 128x64 per wave), K staged in 32-deep
// slots through a 4-slot LDS ring per operand (128 KiB total, ONE __shared__
// object), global->LDS via global_load_lds dwordx4 (2 per thread per phase),
// prefetch 3 slots ahead with counted s_waitcnt vmcnt(8) at slot boundaries
// (never vmcnt(0) in the main loop), raw s_barrier phase boundaries, and
// s_setprio(1) around each 16-MFMA cluster (v_mfma_f32_16x16x32_bf16).
//
// Operand layouts (element (row, k)):
//   LAY=0 "k-inner": row-major (rows, K)  — fprop x and nn.Linear weight.
//     LDS slot image [256 rows][32 k] (64B rows), 16B-chunk XOR swizzle
//     c^((row>>2)&1)<<1 applied on the glds SOURCE address (rule 21) so
//     ds_read_b128 A/B-fragment reads are bank-conflict-free.
//   LAY=1 "k-outer": row-major (K, rows) — wgrad operands (dY, X viewed
//     along tokens) and dgrad's weight.  LDS slot image [32 k][256 rows]
//     (512B rows, fully-coalesced glds); fragments gathered with
//     ds_read_b64_tr_b16 pairs using the probe-verified lane model
//     (probe.hip tr16_probe; same recipe as attention_v2.hip dq-lite),
//     optional 16B-chunk swizzle c^((k&3)<<2) against tr-read conflicts.
//
// Split-K (wgrad M=N=2048-class shapes: too few 256^2 tiles to fill 256
// CUs): blockIdx carries tile*splitk+slice after the bijective XCD remap
// (T1) so a tile's slices share an XCD; slices write fp32 slabs, a separate
// reduce kernel folds them to bf16 (slab is MBytes-scale, far past the
// guide's in-launch-combine regime).
//
// MFMA fragment maps (probe-verified on hardware, see probe.hip):
//   A: lane l holds A[row=l&15][k=(l>>4)*8+j], j=0..7
//   B: lane l holds B[col=l&15][k=(l>>4)*8+j]   (B stored (N,K): same map)
//   C/D: col=l&15, row=(l>>4)*4+reg

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int SLOT_HW = 256 * 32;     // one staged slot: 8192 hw = 16 KiB
constexpr int BREG_HW = 4 * SLOT_HW;  // B operand region (halfwords)

#define GLDS16(gp, lp)                                              \
  __builtin_amdgcn_global_load_lds(                                 \
      (const __attribute__((address_space(1))) unsigned int*)(gp),  \
      (__attribute__((address_space(3))) unsigned int*)(lp), 16, 0, 0)

DEVINL unsigned lds_addr_of(const ushort* p) {
  return (unsigned)(unsigned long long)(
      __attribute__((address_space(3))) const ushort*)p;
}

// bijective XCD-aware remap (T1): each XCD gets a contiguous chunk of the
// logical grid so neighbouring tiles (and a tile's split-K slices) share L2
DEVINL int xcd_remap(int wg, int nwg) {
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = wg & 7, idx = wg >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

template <int LAYA, int LAYB, bool SPLIT, bool KSWZ>
__launch_bounds__(512)
__global__ void gemm256_kernel(const ushort* __restrict__ A,
                               const ushort* __restrict__ B,
                               const ushort* __restrict__ bias,
                               ushort* __restrict__ C,
                               float* __restrict__ slab,
                               int M, int N, int K,
                               long lda, long ldb, long ldc,
                               int splitk, int kper, int tiles_n,
                               int gsup) {
  __shared__ __attribute__((aligned(16))) ushort lds[8 * SLOT_HW];  // 128 KiB

  const int nwg = gridDim.x;
  const int wgid = xcd_remap(blockIdx.x, nwg);
  const int tile = wgid / splitk;
  const int slice = wgid - tile * splitk;
  // L2 supertile: an XCD's contiguous wgid chunk visits tiles in
  // gsup-tall tm bands so BOTH operand panels get L2 reuse (a row-major
  // chunk reuses only the A panel)
  const int band = tile / (gsup * tiles_n);
  const int rem = tile - band * (gsup * tiles_n);
  const int tm = band * gsup + (rem - (rem / gsup) * gsup);
  const int tn = rem / gsup;

  const int tid = threadIdx.x;
  const int w = tid >> 6;            // wave id (wave-uniform)
  const int lane = tid & 63;
  const int wm = w >> 2, wn = w & 3; // wave tile (wm*128, wn*64)
  const int l15 = lane & 15, lg = lane >> 4;

  const ushort* Abase = (LAYA == 0) ? A + (long)(tm * 256) * lda
                                    : A + tm * 256;
  const ushort* Bbase = (LAYB == 0) ? B + (long)(tn * 256) * ldb
                                    : B + tn * 256;
  const int k0 = slice * kper;
  const int nslot = kper / 32;

  // ---- staging: 2 glds per thread per call (one 16 KiB slot).
  // ring = destination ring slot; ss = SOURCE slot (clamped at the tail:
  // the target ring slot is no longer read, the data is garbage-by-design)
  auto stage = [&](const ushort* gb, long ld, int ring, int ss, int region,
                   int lay) {
    const int kk = k0 + ss * 32;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const int dhw = region + ring * SLOT_HW + w * 512 + i * 4096;
      const ushort* g;
      if (lay == 0) {
        const int o = w * 1024 + i * 8192 + lane * 16;  // byte within slot
        const int row = o >> 6;                          // 64 B rows
        const int cs = (lane & 3) ^ (((row >> 2) & 1) << 1);
        g = gb + (long)row * ld + kk + cs * 8;
      } else {
        const int o = w * 1024 + i * 8192 + lane * 16;
        const int krow = o >> 9;                         // 512 B rows
        const int c = lane & 31;
        const int cs = KSWZ ? (c ^ ((krow & 3) << 2)) : c;
        g = gb + (long)(kk + krow) * ld + cs * 8;
      }
      GLDS16(g, lds + dhw);
    }
  };
  auto stageA = [&](int s) {
    stage(Abase, lda, s & 3, s < nslot ? s : nslot - 1, 0, LAYA);
  };
  auto stageB = [&](int s) {
    stage(Bbase, ldb, s & 3, s < nslot ? s : nslot - 1, BREG_HW, LAYB);
  };

  // ---- fragment reads
  // k-inner: ds_read_b128 at swizzled per-lane base + frag*1024B immediates
  const int csr = lg ^ (((l15 >> 2) & 1) << 1);
  const int rdA = (wm * 128 + l15) * 32 + csr * 8;             // hw offset
  const int rdB = BREG_HW + (wn * 64 + l15) * 32 + csr * 8;
  // k-outer: tr16 per-lane byte base (probe model); frag adds rowbase*2
  // (XOR'd with j<<6 when swizzled)
  const unsigned trlane = (unsigned)((8 * lg + ((lane >> 2) & 3)) * 512 +
                                     ((lane & 3) >> 1) * 16 + (lane & 1) * 8);
  const unsigned trJ = (unsigned)(((lane >> 2) & 3) << 6);
  const unsigned lds0 = lds_addr_of(lds);

  // ALL of an operand's tr16 gathers issue inside ONE asm block whose
  // final instruction is the lgkm wait: the reads stay in flight together
  // (a per-pair wait serialized 12 LDS round trips per slot), and no
  // compiler-inserted register move can touch an output while its read is
  // still outstanding (free-floating no-wait asm reads raced exactly that
  // way — schedule-dependent corruption on the LAY=1 paths).
  auto tr16x8 = [&](const unsigned* ad, bf16x8_v* fr, int n) {
    unsigned long long r[16];
    if (n == 8) {
      asm volatile(
          "ds_read_b64_tr_b16 %0, %16 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %16 offset:2048\n\t"
          "ds_read_b64_tr_b16 %2, %17 offset:0\n\t"
          "ds_read_b64_tr_b16 %3, %17 offset:2048\n\t"
          "ds_read_b64_tr_b16 %4, %18 offset:0\n\t"
          "ds_read_b64_tr_b16 %5, %18 offset:2048\n\t"
          "ds_read_b64_tr_b16 %6, %19 offset:0\n\t"
          "ds_read_b64_tr_b16 %7, %19 offset:2048\n\t"
          "ds_read_b64_tr_b16 %8, %20 offset:0\n\t"
          "ds_read_b64_tr_b16 %9, %20 offset:2048\n\t"
          "ds_read_b64_tr_b16 %10, %21 offset:0\n\t"
          "ds_read_b64_tr_b16 %11, %21 offset:2048\n\t"
          "ds_read_b64_tr_b16 %12, %22 offset:0\n\t"
          "ds_read_b64_tr_b16 %13, %22 offset:2048\n\t"
          "ds_read_b64_tr_b16 %14, %23 offset:0\n\t"
          "ds_read_b64_tr_b16 %15, %23 offset:2048\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(r[0]), "=&v"(r[1]), "=&v"(r[2]), "=&v"(r[3]),
            "=&v"(r[4]), "=&v"(r[5]), "=&v"(r[6]), "=&v"(r[7]),
            "=&v"(r[8]), "=&v"(r[9]), "=&v"(r[10]), "=&v"(r[11]),
            "=&v"(r[12]), "=&v"(r[13]), "=&v"(r[14]), "=&v"(r[15])
          : "v"(ad[0]), "v"(ad[1]), "v"(ad[2]), "v"(ad[3]),
            "v"(ad[4]), "v"(ad[5]), "v"(ad[6]), "v"(ad[7]));
    } else {
      asm volatile(
          "ds_read_b64_tr_b16 %0, %8 offset:0\n\t"
          "ds_read_b64_tr_b16 %1, %8 offset:2048\n\t"
          "ds_read_b64_tr_b16 %2, %9 offset:0\n\t"
          "ds_read_b64_tr_b16 %3, %9 offset:2048\n\t"
          "ds_read_b64_tr_b16 %4, %10 offset:0\n\t"
          "ds_read_b64_tr_b16 %5, %10 offset:2048\n\t"
          "ds_read_b64_tr_b16 %6, %11 offset:0\n\t"
          "ds_read_b64_tr_b16 %7, %11 offset:2048\n\t"
          "s_waitcnt lgkmcnt(0)"
          : "=&v"(r[0]), "=&v"(r[1]), "=&v"(r[2]), "=&v"(r[3]),
            "=&v"(r[4]), "=&v"(r[5]), "=&v"(r[6]), "=&v"(r[7])
          : "v"(ad[0]), "v"(ad[1]), "v"(ad[2]), "v"(ad[3]));
    }
#pragma unroll
    for (int i = 0; i < n; ++i) {
      ((unsigned long long*)&fr[i])[0] = r[2 * i];
      ((unsigned long long*)&fr[i])[1] = r[2 * i + 1];
    }
  };
  auto trA_addr = [&](int ring, int mf) -> unsigned {
    const unsigned r16 = (unsigned)((wm * 128 + mf * 16) * 2);
    return lds0 + (unsigned)(ring * SLOT_HW * 2) + trlane +
           (KSWZ ? (r16 ^ trJ) : r16);
  };
  auto trB_addr = [&](int ring, int nf) -> unsigned {
    const unsigned r16 = (unsigned)((wn * 64 + nf * 16) * 2);
    return lds0 + (unsigned)(BREG_HW * 2) +
           (unsigned)(ring * SLOT_HW * 2) + trlane +
           (KSWZ ? (r16 ^ trJ) : r16);
  };
  auto readA0 = [&](int ring, int mf) -> bf16x8_v {
    return *(const bf16x8_v*)&lds[ring * SLOT_HW + rdA + mf * 512];
  };
  auto readB0 = [&](int ring, int nf) -> bf16x8_v {
    return *(const bf16x8_v*)&lds[ring * SLOT_HW + rdB + nf * 512];
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int mf = 0; mf < 8; ++mf)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) acc[mf][nf] = (f32x4)(0.f);

  // one 32-deep slot: 2 phases.  All 12 fragment reads issue at the top
  // (counted lgkm waits let phase b's operands arrive under phase a's
  // MFMAs); staging for slot s+3 is split across the two phases.
  // wait_mode: 0 = no vmcnt at this slot boundary (covered by the
  // previous even-slot wait), 1 = vmcnt(4) (even slots: guarantees the
  // NEXT TWO slots' staging landed), 2 = vmcnt(8) every slot (tail-safe)
  auto do_slot = [&](int s, int ring, bool stage_ok, int wait_mode) {
    stage(Abase, lda, (ring + 3) & 3, stage_ok ? s + 3 : nslot - 1, 0, LAYA);
    bf16x8_v af[8], bfr[4];
    if (LAYA == 0) {
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) af[mf] = readA0(ring, mf);
    } else {
      unsigned ad[8];
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) ad[mf] = trA_addr(ring, mf);
      tr16x8(ad, af, 8);
    }
    if (LAYB == 0) {
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) bfr[nf] = readB0(ring, nf);
    } else {
      unsigned ad[4];
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) ad[nf] = trB_addr(ring, nf);
      tr16x8(ad, bfr, 4);
    }
    stage(Bbase, ldb, (ring + 3) & 3, stage_ok ? s + 3 : nslot - 1,
          BREG_HW, LAYB);
    asm volatile("s_barrier" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
      acc[mf][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          af[mf], bfr[0], acc[mf][0], 0, 0, 0);
      acc[mf][1] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          af[mf], bfr[1], acc[mf][1], 0, 0, 0);
    }
#pragma unroll
    for (int mf = 0; mf < 8; ++mf) {
      acc[mf][2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          af[mf], bfr[2], acc[mf][2], 0, 0, 0);
      acc[mf][3] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          af[mf], bfr[3], acc[mf][3], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    // slot boundary (counted waits, never 0 in the loop)
    if (wait_mode == 1)
      asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
    else if (wait_mode == 2)
      asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
    else
      asm volatile("s_barrier" ::: "memory");
  };

  // ---- prologue: slots 0..2 staged; wait until slot 0 landed
  stageA(0); stageB(0);
  stageA(1); stageB(1);
  stageA(2); stageB(2);
  asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");

  // ---- main loop: 4-slot groups (compile-time ring ids, no tail clamp)
  const int ns_main = nslot > 3 ? (nslot - 3) & ~3 : 0;
  int s = 0;
  for (; s < ns_main; s += 4) {
    do_slot(s, 0, true, 1);
    do_slot(s + 1, 1, true, 0);
    do_slot(s + 2, 2, true, 1);
    do_slot(s + 3, 3, true, 1);
  }
  for (; s < nslot; ++s)   // tail: staging clamped to the last slot
    do_slot(s, s & 3, s + 3 < nslot, 2);

  // ---- epilogue
  const long crow0 = (long)tm * 256 + wm * 128;
  const int ccol0 = tn * 256 + wn * 64;
  if (SPLIT) {
    float* sl = slab + (long)slice * M * N;
#pragma unroll
    for (int mf = 0; mf < 8; ++mf)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const long row = crow0 + mf * 16 + lg * 4;
        const int col = ccol0 + nf * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          sl[(row + r) * (long)N + col] = acc[mf][nf][r];
      }
  } else {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int col = ccol0 + nf * 16 + l15;
      const float bv = bias ? bf2f(bias[col]) : 0.f;
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) {
        const long row = crow0 + mf * 16 + lg * 4;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          C[(row + r) * ldc + col] = f2bf(acc[mf][nf][r] + bv);
      }
    }
  }
}

// fold SPLITK fp32 slabs into bf16 (grid-strided, float4-vectorized)
__global__ void splitk_reduce_kernel(const float* __restrict__ slab,
                                     ushort* __restrict__ C, long total,
                                     int splitk) {
  const long i4 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  if (i4 >= total) return;
  float4 acc = *(const float4*)(slab + i4);
  for (int s = 1; s < splitk; ++s) {
    const float4 v = *(const float4*)(slab + (long)s * total + i4);
    acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
  }
  ushort4 o;
  o.x = f2bf(acc.x); o.y = f2bf(acc.y); o.z = f2bf(acc.z); o.w = f2bf(acc.w);
  *(ushort4*)(C + i4) = o;
}

void check_bf16_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kBFloat16 &&
              t.dim() == 2 && t.is_contiguous(), name,
              " must be contiguous 2-D bf16 CUDA");
}

template <int LAYA, int LAYB>
torch::Tensor launch_gemm(const torch::Tensor& A, const torch::Tensor& B,
                          const c10::optional<torch::Tensor>& bias,
                          int M, int N, int K, long lda, long ldb,
                          int splitk, bool kswz) {
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0,
              "tdpa_gemm: M,N must be multiples of 256 (got ", M, "x", N,
              ") — dispatch layer should fall back");
  TORCH_CHECK(splitk >= 1 && K % (32 * splitk) == 0,
              "tdpa_gemm: K (", K, ") must be a multiple of 32*splitk");
  auto C = torch::empty({M, N}, A.options());
  const int tiles_m = M / 256, tiles_n = N / 256;
  const int nwg = tiles_m * tiles_n * splitk;
  const int gsup = (tiles_m % 4 == 0) ? 4 : ((tiles_m % 2 == 0) ? 2 : 1);
  const ushort* bptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->numel() == N && bias->scalar_type() == torch::kBFloat16);
    bptr = (const ushort*)bias->data_ptr();
  }
  auto stream = at::cuda::getCurrentHIPStream();
  if (splitk == 1) {
    auto kern = kswz ? gemm256_kernel<LAYA, LAYB, false, true>
                     : gemm256_kernel<LAYA, LAYB, false, false>;
    hipLaunchKernelGGL(kern, dim3(nwg), dim3(512), 0, stream,
                       (const ushort*)A.data_ptr(), (const ushort*)B.data_ptr(),
                       bptr, (ushort*)C.data_ptr(), nullptr, M, N, K,
                       lda, ldb, (long)N, 1, K, tiles_n, gsup);
  } else {
    auto slab = torch::empty({splitk, (long)M * N},
                             A.options().dtype(torch::kFloat));
    auto kern = kswz ? gemm256_kernel<LAYA, LAYB, true, true>
                     : gemm256_kernel<LAYA, LAYB, true, false>;
    hipLaunchKernelGGL(kern, dim3(nwg), dim3(512), 0, stream,
                       (const ushort*)A.data_ptr(), (const ushort*)B.data_ptr(),
                       nullptr, nullptr, slab.data_ptr<float>(), M, N, K,
                       lda, ldb, (long)N, splitk, K / splitk, tiles_n, gsup);
    const long total = (long)M * N;
    const long nb = (total / 4 + 255) / 256;
    hipLaunchKernelGGL(splitk_reduce_kernel, dim3(nb), dim3(256), 0, stream,
                       slab.data_ptr<float>(), (ushort*)C.data_ptr(), total,
                       splitk);
  }
  HIP_CHECK_LAST();
  return C;
}

}  // namespace

// y (M,N) = x (M,K) @ w(N,K)^T + bias — nn.Linear forward.
torch::Tensor gemm_fprop(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias) {
  check_bf16_2d(x, "x"); check_bf16_2d(w, "w");
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "shape mismatch");
  return launch_gemm<0, 0>(x, w, bias, M, N, K, K, K, 1, false);
}

// dx (M,Kin) = dy (M,Kout) @ w (Kout,Kin) — nn.Linear input grad.
torch::Tensor gemm_dgrad(torch::Tensor dy, torch::Tensor w, bool kswz) {
  check_bf16_2d(dy, "dy"); check_bf16_2d(w, "w");
  const int M = dy.size(0), K = dy.size(1), N = w.size(1);
  TORCH_CHECK(w.size(0) == K, "shape mismatch");
  return launch_gemm<0, 1>(dy, w, c10::nullopt, M, N, K, K, N, 1, kswz);
}

// dw (Dout,Din) = dy (T,Dout)^T @ x (T,Din) — nn.Linear weight grad,
// K = tokens (the measured-slow hipBLASLt family at K=16384).
torch::Tensor gemm_wgrad(torch::Tensor dy, torch::Tensor x, long splitk,
                         bool kswz) {
  check_bf16_2d(dy, "dy"); check_bf16_2d(x, "x");
  const int K = dy.size(0), M = dy.size(1), N = x.size(1);
  TORCH_CHECK(x.size(0) == K, "shape mismatch");
  return launch_gemm<1, 1>(dy, x, c10::nullopt, M, N, K, M, N,
                           (int)splitk, kswz);
}
