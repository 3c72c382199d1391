// Flash-attention forward v2 for gfx950 — the CDNA4-guide 8-wave structure:
//
//   - workgroup = 8 waves x 32 q-rows = 256-row q-tile (guide §B "8-warp
//     32x32 ladder"); K/V tiles of 64 keys staged in LDS, T14 reg prefetch
//   - SWAPPED QK^T: S^T = mfma_32x32x16(A=K, B=Q^T), so each lane owns ONE
//     q-row (C col = lane&31) and the online softmax is fully in-register:
//     15 local fmax + one shfl_xor(32) per tile — no 16-lane shuffle trees,
//     no LDS round trip for P
//   - P -> PV B-fragments via v_cvt_pk_bf16_f32 (inline asm, no builtin) +
//     __builtin_amdgcn_permlane32_swap half-exchanges (guide T12)
//   - PV computed as O^T = mfma(A=V^T from LDS, B=P^T in-register); O^T
//     accumulators are per-lane columns, so the alpha rescale is one scalar
//     multiply per register
//
// D=128 only (v1 kernel in attention.hip covers D=64); LSE output is
// identical to v1, so the v1 backward kernels consume it unchanged.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(16))) float f32x16;

constexpr int D2 = 128;     // head dim
constexpr int QW = 32;      // q rows per wave
constexpr int NW2 = 8;      // waves per workgroup
constexpr int QT2 = NW2 * QW;   // 256-row q tile
constexpr int KV = 64;      // keys per LDS tile

struct Strides2 {
  long b, h, s;
};

DEVINL bf16x8_v pack8v(const unsigned short* p) {
  return *(const bf16x8_v*)p;
}

// k_lds [KV][D2]: 256B rows; b128 16-lane groups read 16 consecutive rows at
// one col-chunk -> XOR spreads them conflict-free
DEVINL int swzK(int row, int col) {
  return row * D2 + (col ^ ((row & 15) << 3));
}

// vt_lds [D2][KV]: 128B rows; 16-lane groups read 16 consecutive d-rows
DEVINL int swzV(int row, int col) {
  return row * KV + (col ^ ((row & 7) << 3));
}

DEVINL unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2\n\ts_nop 1"
               : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

template <bool CAUSAL>
__launch_bounds__(512)
__global__ void attn_fwd_v2_kernel(const unsigned short* __restrict__ q,
                                   const unsigned short* __restrict__ k,
                                   const unsigned short* __restrict__ v,
                                   unsigned short* __restrict__ o,
                                   float* __restrict__ lse,
                                   Strides2 qs, Strides2 ks, Strides2 vs,
                                   Strides2 os,
                                   int B, int H, int S, float scale,
                                   int q_per_kv) {
  __shared__ __attribute__((aligned(16))) unsigned short k_lds[KV * D2];
  __shared__ __attribute__((aligned(16))) unsigned short vt_lds[D2 * KV];

  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int hkv = hh / q_per_kv;
  const int qbase = blockIdx.x * QT2;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const unsigned short* qp = q + bb * qs.b + hh * qs.h;
  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const unsigned short* vp = v + bb * vs.b + hkv * vs.h;

  // ---- Q fragments: this lane's q-row, 8 chunks of 16 (B-operand layout)
  const int qrow = qbase + wid * QW + l31;
  bf16x8_v q_frag[8];
  {
    long r = (long)(qrow < S ? qrow : S - 1) * qs.s;
#pragma unroll
    for (int c = 0; c < 8; ++c)
      q_frag[c] = pack8v(qp + r + c * 16 + hi * 8);
  }

  // ---- per-lane softmax state (one q-row) + O^T accumulators
  float m_run = -1e30f, l_run = 0.f;
  f32x16 acc[4];   // d-subtiles of 32: O^T[d][q]
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) acc[ds] = (f32x16)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT2) : S;

  // ---- T14 staging: each thread owns two 8-elem chunks of the K/V tile
  // (KV*D2 = 8192 elems, 512 threads)
  bf16x8_v k_reg[2], v_reg[2];
  auto stage_load = [&](int kt0) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int key = idx / D2;
      int col = idx % D2;
      int gkey = kt0 + key;
      if (gkey < S) {
        k_reg[c] = pack8v(kp + (long)gkey * ks.s + col);
        v_reg[c] = pack8v(vp + (long)gkey * vs.s + col);
      } else {
        k_reg[c] = (bf16x8_v)(__bf16)0.f;
        v_reg[c] = (bf16x8_v)(__bf16)0.f;
      }
    }
  };
  stage_load(0);

  for (int kt0 = 0; kt0 < kv_end; kt0 += KV) {
    __syncthreads();
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int key = idx / D2;
      int col = idx % D2;
      *(bf16x8_v*)&k_lds[swzK(key, col)] = k_reg[c];
      const unsigned short* vsrc = (const unsigned short*)&v_reg[c];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_lds[swzV(col + j, key)] = vsrc[j];
    }
    __syncthreads();
    if (kt0 + KV < kv_end) stage_load(kt0 + KV);

    // ---- swapped QK^T: S^T[key][q] for 2 key-subtiles of 32
    f32x16 st[2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      st[kt] = (f32x16)(0.f);
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        // A = K[key = kt*32 + l31][k-chunk c]
        bf16x8_v a_k = pack8v(&k_lds[swzK(kt * 32 + l31, c * 16 + hi * 8)]);
        st[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_k, q_frag[c], st[kt], 0, 0, 0);
      }
    }

    // ---- mask + in-register online softmax (this lane = one q-row)
    float p[2][16];
    float tmax = -1e30f;
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int key = kt0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv = st[kt][r] * scale;
        bool valid = key < S && (!CAUSAL || key <= qrow);
        sv = valid ? sv : -1e30f;
        p[kt][r] = sv;
        tmax = fmaxf(tmax, sv);
      }
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));  // partner holds other keys
    float m_new = fmaxf(m_run, tmax);
    float alpha = __expf(m_run - m_new);
    m_run = m_new;
    float psum = 0.f;
#pragma unroll
    for (int kt = 0; kt < 2; ++kt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float pv = __expf(p[kt][r] - m_new);
        p[kt][r] = pv;
        psum += pv;
      }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;

    // ---- rescale O^T accumulators (per-lane scalar alpha)
#pragma unroll
    for (int ds = 0; ds < 4; ++ds)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[ds][r] *= alpha;

    // ---- P^T B-fragments via cvt_pk + permlane32_swap half-exchange
    bf16x8_v pb[2][2];  // [key-subtile][16-key chunk]
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        // this lane needs octet m = 2*kc + hi of subtile kt
        int m0 = 2 * kc;      // octet when hi==0
        int m1 = 2 * kc + 1;  // octet when hi==1
        unsigned a0 = cvt_pk_bf16(p[kt][4 * m0], p[kt][4 * m0 + 1]);
        unsigned a1 = cvt_pk_bf16(p[kt][4 * m0 + 2], p[kt][4 * m0 + 3]);
        unsigned b0 = cvt_pk_bf16(p[kt][4 * m1], p[kt][4 * m1 + 1]);
        unsigned b1 = cvt_pk_bf16(p[kt][4 * m1 + 2], p[kt][4 * m1 + 3]);
        // exchange halves so every lane holds BOTH hi-halves of its octet:
        auto s0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto s1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        // after swap: s0[0] = [a0(lo lanes) | b0(lo lanes)] -> on lo lanes
        //   own a0, on hi lanes partner's... see header derivation: the
        //   needed octet for lane hi is (2*kc + hi); keys hi*8+j of that
        //   octet come from BOTH source halves:
        //   lo lanes (hi=0, octet m0): keys 8*m0+{0..3} from own a0/a1,
        //     keys 8*m0+{4..7} live on hi lanes' a0/a1 -> arrive in s?[1]
        //   hi lanes (hi=1, octet m1): keys 8*m1+{4..7} own b0/b1, keys
        //     8*m1+{0..3} from lo lanes' b0/b1 -> arrive in s?[0]
        unsigned f0, f1, f2, f3;
        if (hi == 0) {
          f0 = a0;      // keys 8m0+0,1   (own, hi_src=0)
          f1 = a1;      // keys 8m0+2,3
          f2 = s0[1];   // b0 with lo lanes <- a0 hi lanes?  keys 8m0+4,5
          f3 = s1[1];   // keys 8m0+6,7
        } else {
          f0 = s0[0];   // a0 with hi lanes <- b0 lo lanes: keys 8m1+0,1
          f1 = s1[0];   // keys 8m1+2,3
          f2 = b0;      // keys 8m1+4,5 (own)
          f3 = b1;      // keys 8m1+6,7
        }
        unsigned* dst = (unsigned*)&pb[kt][kc];
        dst[0] = f0;
        dst[1] = f1;
        dst[2] = f2;
        dst[3] = f3;
      }
    }

    // ---- O^T += V^T @ P^T   (A = V^T[d][key], B = P^T[key][q])
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          bf16x8_v a_v = pack8v(&vt_lds[swzV(ds * 32 + l31,
                                             kt * 32 + kc * 16 + hi * 8)]);
          acc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_v, pb[kt][kc], acc[ds], 0, 0, 0);
        }
      }
    }
  }

  // ---- epilogue: O[q][d] = O^T / l; LSE
  if (qrow < S) {
    unsigned short* op = o + bb * os.b + hh * os.h + (long)qrow * os.s;
    float inv_l = l_run > 0.f ? 1.f / l_run : 0.f;
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int blk = 0; blk < 4; ++blk) {  // reg groups of 4 -> d+0..3
        int d0 = ds * 32 + 8 * blk + 4 * hi;
        unsigned short out4[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          out4[j] = f2bf(acc[ds][4 * blk + j] * inv_l);
        *(ushort4*)(op + d0) = *(ushort4*)out4;
      }
    }
    if (hi == 0)
      lse[(long)bh * S + qrow] =
          m_run + __logf(l_run > 0.f ? l_run : 1.f);
  }
}

}  // namespace

void attn_fwd_v2(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor o, torch::Tensor lse, bool causal,
                 double scale) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  const int q_per_kv = H / Hkv;
  TORCH_CHECK(D == 128, "attn_fwd_v2: D must be 128");
  auto get = [](const torch::Tensor& t) {
    return Strides2{t.stride(0), t.stride(1), t.stride(2)};
  };
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid((S + QT2 - 1) / QT2, B * H), block(512);
  if (causal)
    hipLaunchKernelGGL((attn_fwd_v2_kernel<true>), grid, block, 0, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (unsigned short*)o.data_ptr(),
                       lse.data_ptr<float>(), get(q), get(k), get(v), get(o),
                       B, H, S, (float)scale, q_per_kv);
  else
    hipLaunchKernelGGL((attn_fwd_v2_kernel<false>), grid, block, 0, stream,
                       (const unsigned short*)q.data_ptr(),
                       (const unsigned short*)k.data_ptr(),
                       (const unsigned short*)v.data_ptr(),
                       (unsigned short*)o.data_ptr(),
                       lse.data_ptr<float>(), get(q), get(k), get(v), get(o),
                       B, H, S, (float)scale, q_per_kv);
  HIP_CHECK_LAST();
}
