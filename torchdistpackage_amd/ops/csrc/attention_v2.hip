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

// vt_lds [D2][KV]: 128B rows.  ds_read_b128 16-lane groups are NOT
// contiguous on CDNA4 — e.g. {0-3,12-15,20-27} — so a (row&7) xor collides
// for rows 8 apart within one group (rows 12 vs 20 etc.), which PMC showed
// as SQ_LDS_BANK_CONFLICT >> SQ_INSTS_MFMA.  slot = (row>>1)^(row>>4) is a
// bijection per parity class on both real group row-sets {0-3,12-15,20-27}
// and {4-11,16-19,28-31} (bank = row&1 ? 32:0 | slot*4), making A-frag reads
// conflict-free; the transpose scatter writes (rows stride 8) hit the 2-way
// minimum an 8-slot row permits.
DEVINL int swzV(int row, int col) {
  return row * KV + (col ^ ((((row >> 1) ^ (row >> 4)) & 7) << 3));
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
  // double-buffered tiles: occupancy is VGPR-bound (1 block/CU at 230
  // VGPRs), so the extra LDS is free and buys one barrier per tile
  __shared__ __attribute__((aligned(16))) unsigned short k_lds[2][KV * D2];
  __shared__ __attribute__((aligned(16))) unsigned short vt_lds[2][D2 * KV];

  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int hkv = hh / q_per_kv;
  // causal: schedule heavy (late) q-tiles first so the tail packs evenly
  const int qtile = CAUSAL ? (gridDim.x - 1 - blockIdx.x) : blockIdx.x;
  const int qbase = qtile * QT2;
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
  // prologue: fill buffer 0
  {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int key = idx / D2;
      int col = idx % D2;
      *(bf16x8_v*)&k_lds[0][swzK(key, col)] = k_reg[c];
      const unsigned short* vsrc = (const unsigned short*)&v_reg[c];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_lds[0][swzV(col + j, key)] = vsrc[j];
    }
    if (KV < kv_end) stage_load(KV);
    __syncthreads();
  }

  int buf = 0;
  for (int kt0 = 0; kt0 < kv_end; kt0 += KV) {
    // ---- swapped QK^T: S^T[key][q] for 2 key-subtiles of 32
    f32x16 st[2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      st[kt] = (f32x16)(0.f);
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        // A = K[key = kt*32 + l31][k-chunk c]
        bf16x8_v a_k = pack8v(
            &k_lds[buf][swzK(kt * 32 + l31, c * 16 + hi * 8)]);
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
          bf16x8_v a_v = pack8v(&vt_lds[buf][swzV(ds * 32 + l31,
                                                  kt * 32 + kc * 16 + hi * 8)]);
          acc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_v, pb[kt][kc], acc[ds], 0, 0, 0);
        }
      }
    }

    // ---- write the prefetched NEXT tile into the other buffer (nobody
    // reads it until after the barrier) and issue the tile-after-next loads
    if (kt0 + KV < kv_end) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        int idx = tid * 8 + c * 4096;
        int key = idx / D2;
        int col = idx % D2;
        *(bf16x8_v*)&k_lds[buf ^ 1][swzK(key, col)] = k_reg[c];
        const unsigned short* vsrc = (const unsigned short*)&v_reg[c];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vt_lds[buf ^ 1][swzV(col + j, key)] = vsrc[j];
      }
      if (kt0 + 2 * KV < kv_end) stage_load(kt0 + 2 * KV);
    }
    __syncthreads();
    buf ^= 1;
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

// ===========================================================================
// Backward v2 (D=128): same swapped-operand structure as the forward.
//
// dq pass: wave owns 32 q-rows (lane = one q-column of the S^T tiles);
//   per 64-key tile: S^T = mfma(K, Q^T), dP^T = mfma(V, dO^T); P and dS stay
//   in registers (lse/delta are per-lane scalars); dq^T += mfma(K^T, dS^T)
//   with the dS^T B-fragments built by the cvt_pk+permlane exchange.
// ===========================================================================

namespace {

template <bool CAUSAL>
__launch_bounds__(512)
__global__ void attn_bwd_dq_v2_kernel(const unsigned short* __restrict__ q,
                                      const unsigned short* __restrict__ k,
                                      const unsigned short* __restrict__ v,
                                      const unsigned short* __restrict__ dout,
                                      const float* __restrict__ lse,
                                      const float* __restrict__ delta,
                                      unsigned short* __restrict__ dq,
                                      Strides2 qs, Strides2 ks, Strides2 vs,
                                      Strides2 dos, Strides2 dqs,
                                      int B, int H, int S, float scale,
                                      int q_per_kv) {
  __shared__ __attribute__((aligned(16))) unsigned short k_lds[2][KV * D2];
  __shared__ __attribute__((aligned(16))) unsigned short v_lds[2][KV * D2];
  __shared__ __attribute__((aligned(16))) unsigned short kt_lds[2][D2 * KV];

  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int hkv = hh / q_per_kv;
  const int qtile = CAUSAL ? (gridDim.x - 1 - blockIdx.x) : blockIdx.x;
  const int qbase = qtile * QT2;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const unsigned short* qp = q + bb * qs.b + hh * qs.h;
  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const unsigned short* vp = v + bb * vs.b + hkv * vs.h;
  const unsigned short* dop = dout + bb * dos.b + hh * dos.h;

  const int qrow = qbase + wid * QW + l31;
  bf16x8_v q_frag[8], do_frag[8];
  float my_lse, my_delta;
  {
    long r = (long)(qrow < S ? qrow : S - 1) * qs.s;
    long rdo = (long)(qrow < S ? qrow : S - 1) * dos.s;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      q_frag[c] = pack8v(qp + r + c * 16 + hi * 8);
      do_frag[c] = pack8v(dop + rdo + c * 16 + hi * 8);
    }
    my_lse = qrow < S ? lse[(long)bh * S + qrow] : 0.f;
    my_delta = qrow < S ? delta[(long)bh * S + qrow] : 0.f;
  }

  f32x16 acc[4];  // dq^T[d][q]
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) acc[ds] = (f32x16)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT2) : S;

  // double-buffered: stage tile t+1 while nobody reads that buffer, one
  // barrier per tile (direct load->write; no prefetch registers: this
  // kernel sits at the 256-VGPR edge)
  auto stage_dq = [&](int kt0, int b) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int key = idx / D2;
      int col = idx % D2;
      int gkey = kt0 + key;
      bf16x8_v kv8, vv8;
      if (gkey < S) {
        kv8 = pack8v(kp + (long)gkey * ks.s + col);
        vv8 = pack8v(vp + (long)gkey * vs.s + col);
      } else {
        kv8 = (bf16x8_v)(__bf16)0.f;
        vv8 = (bf16x8_v)(__bf16)0.f;
      }
      *(bf16x8_v*)&k_lds[b][swzK(key, col)] = kv8;
      *(bf16x8_v*)&v_lds[b][swzK(key, col)] = vv8;
      const unsigned short* ksrc = (const unsigned short*)&kv8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kt_lds[b][swzV(col + j, key)] = ksrc[j];
    }
  };
  stage_dq(0, 0);
  __syncthreads();

  int buf = 0;
  for (int kt0 = 0; kt0 < kv_end; kt0 += KV) {

    // S^T and dP^T for the 2 key-subtiles
    f32x16 st[2], dpt[2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      st[kt] = (f32x16)(0.f);
      dpt[kt] = (f32x16)(0.f);
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        bf16x8_v a_k = pack8v(
            &k_lds[buf][swzK(kt * 32 + l31, c * 16 + hi * 8)]);
        st[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_k, q_frag[c], st[kt], 0, 0, 0);
        bf16x8_v a_v = pack8v(
            &v_lds[buf][swzK(kt * 32 + l31, c * 16 + hi * 8)]);
        dpt[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a_v, do_frag[c], dpt[kt], 0, 0, 0);
      }
    }

    // dS^T = P * (dP^T - delta) * scale, P = exp(S^T*scale - lse)
    // (written back into st[] — fresh arrays would push the kernel past the
    // 256-VGPR budget and spill)
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int key = kt0 + kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        bool valid = key < S && qrow < S && (!CAUSAL || key <= qrow);
        float pv = valid ? __expf(st[kt][r] * scale - my_lse) : 0.f;
        st[kt][r] = pv * (dpt[kt][r] - my_delta) * scale;
      }
    }

    // dS^T B-fragments (same exchange as the forward)
    bf16x8_v db[2][2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        int m0 = 2 * kc, m1 = 2 * kc + 1;
        unsigned a0 = cvt_pk_bf16(st[kt][4 * m0], st[kt][4 * m0 + 1]);
        unsigned a1 = cvt_pk_bf16(st[kt][4 * m0 + 2], st[kt][4 * m0 + 3]);
        unsigned b0 = cvt_pk_bf16(st[kt][4 * m1], st[kt][4 * m1 + 1]);
        unsigned b1 = cvt_pk_bf16(st[kt][4 * m1 + 2], st[kt][4 * m1 + 3]);
        auto s0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto s1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        unsigned f0, f1, f2, f3;
        if (hi == 0) {
          f0 = a0; f1 = a1; f2 = s0[1]; f3 = s1[1];
        } else {
          f0 = s0[0]; f1 = s1[0]; f2 = b0; f3 = b1;
        }
        unsigned* dst = (unsigned*)&db[kt][kc];
        dst[0] = f0; dst[1] = f1; dst[2] = f2; dst[3] = f3;
      }
    }

    // dq^T += K^T @ dS^T
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          bf16x8_v a_kt = pack8v(&kt_lds[buf][swzV(ds * 32 + l31,
                                                   kt * 32 + kc * 16 +
                                                   hi * 8)]);
          acc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_kt, db[kt][kc], acc[ds], 0, 0, 0);
        }
      }
    }

    if (kt0 + KV < kv_end) stage_dq(kt0 + KV, buf ^ 1);
    __syncthreads();
    buf ^= 1;
  }

  if (qrow < S) {
    unsigned short* dqp = dq + bb * dqs.b + hh * dqs.h + (long)qrow * dqs.s;
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int blk = 0; blk < 4; ++blk) {
        int d0 = ds * 32 + 8 * blk + 4 * hi;
        unsigned short out4[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          out4[j] = f2bf(acc[ds][4 * blk + j]);
        *(ushort4*)(dqp + d0) = *(ushort4*)out4;
      }
    }
  }
}

}  // namespace

void attn_bwd_dq_v2(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                    torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                    torch::Tensor dq, bool causal, double scale) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  const int q_per_kv = H / Hkv;
  TORCH_CHECK(D == 128);
  auto get = [](const torch::Tensor& t) {
    return Strides2{t.stride(0), t.stride(1), t.stride(2)};
  };
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid((S + QT2 - 1) / QT2, B * H), block(512);
#define L_DQ2(CC)                                                             \
  hipLaunchKernelGGL((attn_bwd_dq_v2_kernel<CC>), grid, block, 0, stream,     \
                     (const unsigned short*)q.data_ptr(),                     \
                     (const unsigned short*)k.data_ptr(),                     \
                     (const unsigned short*)v.data_ptr(),                     \
                     (const unsigned short*)dout.data_ptr(),                  \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     (unsigned short*)dq.data_ptr(), get(q), get(k), get(v),  \
                     get(dout), get(dq), B, H, S, (float)scale, q_per_kv)
  if (causal) L_DQ2(true); else L_DQ2(false);
#undef L_DQ2
  HIP_CHECK_LAST();
}

// ===========================================================================
// dkdv v2: two passes (dv, then dk), each with the wave owning 32 KEYS
// (lane = one key-column).  A single fused kernel needs both 64-register
// accumulators plus K^T AND V^T fragments and spills ~240 B/lane; splitting
// re-runs the QK^T MFMAs (cheap) but keeps both kernels spill-free.
//   dv pass: S = mfma(Q, K^T regs); P = exp(S*scale - lse[q]);
//            dv^T += mfma(dO^T, P-frag)
//   dk pass: S as above + dP = mfma(dO, V^T regs);
//            dS = P*(dP - delta[q])*scale; dk^T += mfma(Q^T, dS-frag)
// GQA: outputs are per-q-head partials (summed by the host wrapper).
// ===========================================================================

namespace {

// WRITE_DS (dk pass only): also store dS bf16 to ds_out[bh][q][key-padded],
// coalesced (each 32-lane half writes one q-row, 32 consecutive keys).  The
// dq-lite kernel then consumes dS directly instead of recomputing
// S/P/dP per tile (drops 32 of dq's 48 MFMAs plus its V/dO staging).
// PMODE: 0 = self-contained; 1 = dv pass STORES its P register image
// (bf16, per-lane-contiguous 32 B) to p_ws; 2 = dk pass LOADS that image
// (2 x b128, issued before its dP MFMA chain) and SKIPS the 8-MFMA
// S-recompute per q-subtile.  r01's P-store attempt lost 2x to 16 scalar
// per-lane loads; the register-image layout makes the reload two vector
// loads with T14-style early issue.
template <bool CAUSAL, bool DK_PASS, bool WRITE_DS, int PMODE>
__launch_bounds__(512)
__global__ void attn_bwd_dkdv_v2_kernel(
    const unsigned short* __restrict__ q,
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ v,
    const unsigned short* __restrict__ dout,
    const float* __restrict__ lse,
    const float* __restrict__ delta,
    unsigned short* __restrict__ out,   // dk (DK_PASS) or dv
    unsigned short* __restrict__ ds_out, long spad,
    unsigned short* __restrict__ p_ws,
    Strides2 qs, Strides2 ks, Strides2 vs, Strides2 dos, Strides2 outs,
    int B, int H, int S, float scale, int q_per_kv) {
  // LDS: Q and dO tiles (A-operands), plus the transposed tile the pass's
  // final MFMA consumes (dO^T for dv, Q^T for dk)
  __shared__ __attribute__((aligned(16))) unsigned short q_lds[2][KV * D2];
  // dO tile only exists in the dk pass (dv reads dO^T via tr_lds)
  __shared__ __attribute__((aligned(16)))
      unsigned short do_lds[DK_PASS ? 2 * KV * D2 : 8];
  __shared__ __attribute__((aligned(16))) unsigned short tr_lds[2][D2 * KV];
  __shared__ float lse_lds[2][KV];
  __shared__ float del_lds[2][KV];

  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int hkv = hh / q_per_kv;
  const int kbase = blockIdx.x * QT2;
  if (kbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const unsigned short* qp = q + bb * qs.b + hh * qs.h;
  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const unsigned short* vp = v + bb * vs.b + hkv * vs.h;
  const unsigned short* dop = dout + bb * dos.b + hh * dos.h;
  const float* lsep = lse + (long)bh * S;
  const float* delp = delta + (long)bh * S;

  // this wave's 32 keys: K^T fragments always; V^T only in the dk pass
  const int key = kbase + wid * QW + l31;
  bf16x8_v kt_frag[8];
  bf16x8_v vt_frag[DK_PASS ? 8 : 1];
  {
    long r = (long)(key < S ? key : S - 1);
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      kt_frag[c] = pack8v(kp + r * ks.s + c * 16 + hi * 8);
      if (DK_PASS)
        vt_frag[c] = pack8v(vp + r * vs.s + c * 16 + hi * 8);
    }
  }

  f32x16 acc[4];  // dk^T or dv^T (col = key)
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) acc[ds] = (f32x16)(0.f);

  const int qt_start = CAUSAL ? kbase : 0;

  // T14 register staging for the dv pass: global loads for tile t+1 issue
  // right after tile t+1's LDS write, so HBM latency crosses a whole tile
  // of MFMAs (the forward's scheme; measured on it at +?% — see notes).
  // The dk pass sits at 248 VGPRs and keeps the direct stage instead.
  // T14 register staging for the dv pass only: the dk pass sits at 248
  // VGPRs and prefetch regs push it into (measured net-negative) spills.
  bf16x8_v q_reg[DK_PASS ? 1 : 2], do_reg[DK_PASS ? 1 : 2];
  auto load_regs = [&](int qt0) {
    if (DK_PASS) return;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int row = idx / D2;
      int col = idx % D2;
      int grow = qt0 + row;
      if (grow < S) {
        q_reg[DK_PASS ? 0 : c] = pack8v(qp + (long)grow * qs.s + col);
        do_reg[DK_PASS ? 0 : c] = pack8v(dop + (long)grow * dos.s + col);
      } else {
        q_reg[DK_PASS ? 0 : c] = (bf16x8_v)(__bf16)0.f;
        do_reg[DK_PASS ? 0 : c] = (bf16x8_v)(__bf16)0.f;
      }
    }
  };
  auto stage_kv = [&](int qt0, int b) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int row = idx / D2;
      int col = idx % D2;
      int grow = qt0 + row;
      bf16x8_v qv, dv8;
      if (DK_PASS) {
        if (grow < S) {
          qv = pack8v(qp + (long)grow * qs.s + col);
          dv8 = pack8v(dop + (long)grow * dos.s + col);
        } else {
          qv = (bf16x8_v)(__bf16)0.f;
          dv8 = (bf16x8_v)(__bf16)0.f;
        }
      } else {
        qv = q_reg[DK_PASS ? 0 : c];
        dv8 = do_reg[DK_PASS ? 0 : c];
      }
      *(bf16x8_v*)&q_lds[b][swzK(row, col)] = qv;
      if (DK_PASS)
        *(bf16x8_v*)&do_lds[b * KV * D2 + swzK(row, col)] = dv8;
      // transposed tile: dO^T for the dv pass, Q^T for the dk pass
      const unsigned short* tsrc = DK_PASS
          ? (const unsigned short*)&qv : (const unsigned short*)&dv8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        tr_lds[b][swzV(col + j, row)] = tsrc[j];
    }
    if (tid < KV) {
      int grow = qt0 + tid;
      lse_lds[b][tid] = grow < S ? lsep[grow] : 0.f;
      if (DK_PASS)
        del_lds[b][tid] = grow < S ? delp[grow] : 0.f;
    }
  };
  load_regs(qt_start);
  stage_kv(qt_start, 0);
  if (qt_start + KV < S) load_regs(qt_start + KV);
  __syncthreads();

  int buf = 0;
  for (int qt0 = qt_start; qt0 < S; qt0 += KV) {

#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
      // P register-image workspace slot for this (block, q-subtile, lane)
      // q-subtile slots run over the PADDED q range (gridDim.x * 8
      // subtiles of 32): with S not a multiple of 256, (S>>5) under-counts
      // and neighbouring (bh, qsub) slots collide
      unsigned short* pw = (PMODE != 0)
          ? p_ws + (((((long)bh * ((long)gridDim.x * 8) +
                       ((qt0 >> 5) + qt)) *
                      gridDim.x + blockIdx.x) * 512 + tid) * 16)
          : nullptr;
      bf16x8 p_img[2];
      if (PMODE == 2) {
        // issue the P reload FIRST: its latency hides under the dP MFMAs
        p_img[0] = *(const bf16x8*)pw;
        p_img[1] = *(const bf16x8*)(pw + 8);
      }
      f32x16 s_acc = (f32x16)(0.f);
      f32x16 dp_acc;
      if (DK_PASS) dp_acc = (f32x16)(0.f);
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        if (PMODE != 2) {
          bf16x8_v a_q = pack8v(
              &q_lds[buf][swzK(qt * 32 + l31, c * 16 + hi * 8)]);
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_q, kt_frag[c], s_acc, 0, 0, 0);
        }
        if (DK_PASS) {
          bf16x8_v a_do = pack8v(
              &do_lds[buf * KV * D2 + swzK(qt * 32 + l31, c * 16 + hi * 8)]);
          dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_do, vt_frag[c], dp_acc, 0, 0, 0);
        }
      }

      // P into s_acc (dv pass) or dS into s_acc (dk pass)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int qrel = qt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        int qrow = qt0 + qrel;
        float p;
        if (PMODE == 2) {
          p = bf2f(((const unsigned short*)p_img)[r]);
        } else {
          bool valid = key < S && qrow < S && (!CAUSAL || key <= qrow);
          p = valid ?
              __expf(s_acc[r] * scale - lse_lds[buf][qrel]) : 0.f;
        }
        s_acc[r] = DK_PASS
            ? p * (dp_acc[r] - del_lds[buf][qrel]) * scale : p;
        if (DK_PASS && WRITE_DS && qrow < S)
          ds_out[((long)bh * S + qrow) * spad + key] = f2bf(s_acc[r]);
      }
      if (PMODE == 1) {
        unsigned short pb[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) pb[r] = f2bf(s_acc[r]);
        *(bf16x8*)pw = *(const bf16x8*)pb;
        *(bf16x8*)(pw + 8) = *(const bf16x8*)(pb + 8);
      }

      // B-fragments over the q dimension (exchange as in the forward)
      bf16x8_v fb[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        int m0 = 2 * kc, m1 = 2 * kc + 1;
        unsigned a0 = cvt_pk_bf16(s_acc[4 * m0], s_acc[4 * m0 + 1]);
        unsigned a1 = cvt_pk_bf16(s_acc[4 * m0 + 2], s_acc[4 * m0 + 3]);
        unsigned b0 = cvt_pk_bf16(s_acc[4 * m1], s_acc[4 * m1 + 1]);
        unsigned b1 = cvt_pk_bf16(s_acc[4 * m1 + 2], s_acc[4 * m1 + 3]);
        auto s0 = __builtin_amdgcn_permlane32_swap(a0, b0, false, false);
        auto s1 = __builtin_amdgcn_permlane32_swap(a1, b1, false, false);
        unsigned* dst = (unsigned*)&fb[kc];
        if (hi == 0) {
          dst[0] = a0; dst[1] = a1; dst[2] = s0[1]; dst[3] = s1[1];
        } else {
          dst[0] = s0[0]; dst[1] = s1[0]; dst[2] = b0; dst[3] = b1;
        }
      }

      // dv^T += dO^T @ P   or   dk^T += Q^T @ dS
#pragma unroll
      for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          bf16x8_v a_tr = pack8v(&tr_lds[buf][swzV(ds * 32 + l31,
                                                   qt * 32 + kc * 16 +
                                                   hi * 8)]);
          acc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_tr, fb[kc], acc[ds], 0, 0, 0);
        }
      }
    }

    if (qt0 + KV < S) {
      stage_kv(qt0 + KV, buf ^ 1);
      if (qt0 + 2 * KV < S) load_regs(qt0 + 2 * KV);
    }
    __syncthreads();
    buf ^= 1;
  }

  if (key < S) {
    unsigned short* op = out + bb * outs.b + hh * outs.h +
                         (long)key * outs.s;
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int blk = 0; blk < 4; ++blk) {
        int d0 = ds * 32 + 8 * blk + 4 * hi;
        unsigned short o4[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          o4[j] = f2bf(acc[ds][4 * blk + j]);
        *(ushort4*)(op + d0) = *(ushort4*)o4;
      }
    }
  }
}

}  // namespace

void attn_bwd_dkdv_v2(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                      torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                      torch::Tensor dk, torch::Tensor dv, bool causal,
                      double scale) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  const int q_per_kv = H / Hkv;
  TORCH_CHECK(D == 128);
  auto get = [](const torch::Tensor& t) {
    return Strides2{t.stride(0), t.stride(1), t.stride(2)};
  };
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid((S + QT2 - 1) / QT2, B * H), block(512);
#define L_KV2(CC, WDS, DSP, SPAD, PW, PM1, PM2)                               \
  do {                                                                        \
    hipLaunchKernelGGL((attn_bwd_dkdv_v2_kernel<CC, false, false, PM1>),      \
                       grid, block, 0,                                        \
                       stream, (const unsigned short*)q.data_ptr(),           \
                       (const unsigned short*)k.data_ptr(),                   \
                       (const unsigned short*)v.data_ptr(),                   \
                       (const unsigned short*)dout.data_ptr(),                \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                       (unsigned short*)dv.data_ptr(), nullptr, 0, PW,        \
                       get(q),                                                \
                       get(k), get(v), get(dout), get(dv), B, H, S,           \
                       (float)scale, q_per_kv);                               \
    hipLaunchKernelGGL((attn_bwd_dkdv_v2_kernel<CC, true, WDS, PM2>),         \
                       grid, block, 0,                                        \
                       stream, (const unsigned short*)q.data_ptr(),           \
                       (const unsigned short*)k.data_ptr(),                   \
                       (const unsigned short*)v.data_ptr(),                   \
                       (const unsigned short*)dout.data_ptr(),                \
                       lse.data_ptr<float>(), delta.data_ptr<float>(),        \
                       (unsigned short*)dk.data_ptr(), DSP, SPAD, PW,         \
                       get(q),                                                \
                       get(k), get(v), get(dout), get(dk), B, H, S,           \
                       (float)scale, q_per_kv);                               \
  } while (0)
  if (causal) L_KV2(true, false, nullptr, 0, nullptr, 0, 0);
  else L_KV2(false, false, nullptr, 0, nullptr, 0, 0);
  HIP_CHECK_LAST();
}

// ===========================================================================
// dq-lite: consumes the dS tile the dk pass stored instead of recomputing
// S, P and dP per key-tile (48 -> 16 MFMAs per tile; no V/dO staging, no
// exp, no lse/delta).  dS fragments load straight from global: the
// [bh][q][key] layout makes each B-fragment one 16-byte read per lane whose
// element order IS the MFMA k-order (key = kt*32 + kc*16 + hi*8 + j).
// ===========================================================================

namespace {

template <bool CAUSAL>
__launch_bounds__(512)
__global__ void attn_bwd_dq_lite_kernel(
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ dsb,  // [bh][S][spad] bf16
    unsigned short* __restrict__ dq,
    Strides2 ks, Strides2 dqs,
    int B, int H, int S, long spad, int q_per_kv) {
  __shared__ __attribute__((aligned(16))) unsigned short kt_lds[2][D2 * KV];

  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int hkv = hh / q_per_kv;
  const int qtile = CAUSAL ? (gridDim.x - 1 - blockIdx.x) : blockIdx.x;
  const int qbase = qtile * QT2;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const int qrow = qbase + wid * QW + l31;
  const unsigned short* dsp = dsb +
      ((long)bh * S + (qrow < S ? qrow : S - 1)) * spad;

  f32x16 acc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) acc[ds] = (f32x16)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT2) : S;

  bf16x8_v k_reg[2];
  auto load_regs = [&](int kt0) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int gkey = kt0 + idx / D2;
      k_reg[c] = gkey < S ? pack8v(kp + (long)gkey * ks.s + idx % D2)
                          : (bf16x8_v)(__bf16)0.f;
    }
  };
  auto stage_kt = [&](int b) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int key = idx / D2;
      int col = idx % D2;
      const unsigned short* ksrc = (const unsigned short*)&k_reg[c];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kt_lds[b][swzV(col + j, key)] = ksrc[j];
    }
  };
  load_regs(0);
  stage_kt(0);
  if (KV < kv_end) load_regs(KV);
  __syncthreads();

  int buf = 0;
  for (int kt0 = 0; kt0 < kv_end; kt0 += KV) {
    bf16x8_v db[2][2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        db[kt][kc] = pack8v(dsp + kt0 + kt * 32 + kc * 16 + hi * 8);

    // dq^T += K^T @ dS^T
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          bf16x8_v a_kt = pack8v(&kt_lds[buf][swzV(ds * 32 + l31,
                                                   kt * 32 + kc * 16 +
                                                   hi * 8)]);
          acc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_kt, db[kt][kc], acc[ds], 0, 0, 0);
        }
      }
    }

    if (kt0 + KV < kv_end) {
      stage_kt(buf ^ 1);
      if (kt0 + 2 * KV < kv_end) load_regs(kt0 + 2 * KV);
    }
    __syncthreads();
    buf ^= 1;
  }

  if (qrow < S) {
    unsigned short* dqp = dq + bb * dqs.b + hh * dqs.h + (long)qrow * dqs.s;
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int blk = 0; blk < 4; ++blk) {
        int d0 = ds * 32 + 8 * blk + 4 * hi;
        unsigned short out4[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          out4[j] = f2bf(acc[ds][4 * blk + j]);
        *(ushort4*)(dqp + d0) = *(ushort4*)out4;
      }
    }
  }
}

}  // namespace

// ===========================================================================
// dq-lite-tr (default; TDPA_NO_TR16 falls back): same math but the K^T
// A-fragments come from ds_read_b64_tr_b16 over a BLOCKED natural image
// kimg[d/16][key][16] instead of a transposed tile built by a 64-write
// scalar scatter.  Address derivation from the probe-verified gather model
// (see probe.hip): lane l elem j = lds[floor8B(addr(lane ((l>>2)&3)+4j))/2
// + (l&3)]; required element for the A-frag is
//   (d>>4)*1024 + key*16 + (d&15),   d = ds*32 + (l&31)
// which splits into a per-lane base
//   base(k) = 2*[ ((k>>4)&1)*1024 + ((k>>2)&3)*16 + 4*(k&3) + (k>>5)*128 ]
// plus a uniform immediate offset ds*4096 + (kt*32+kc*16+sub*4)*32 bytes.
// ===========================================================================

namespace {

template <bool CAUSAL>
__launch_bounds__(512)
__global__ void attn_bwd_dq_lite_tr_kernel(
    const unsigned short* __restrict__ k,
    const unsigned short* __restrict__ dsb,
    unsigned short* __restrict__ dq,
    Strides2 ks, Strides2 dqs,
    int B, int H, int S, long spad, int q_per_kv) {
  __shared__ __attribute__((aligned(16))) unsigned short kimg[2][8 * KV * 16];

  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int hkv = hh / q_per_kv;
  const int qtile = CAUSAL ? (gridDim.x - 1 - blockIdx.x) : blockIdx.x;
  const int qbase = qtile * QT2;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l31 = lane & 31;
  const int hi = lane >> 5;

  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const int qrow = qbase + wid * QW + l31;
  const unsigned short* dsp = dsb +
      ((long)bh * S + (qrow < S ? qrow : S - 1)) * spad;

  // per-lane tr16 base (bytes, within kimg[0]); buffer 1 adds 16 KiB
  const unsigned lds_base = (unsigned)(unsigned long long)
      (__attribute__((address_space(3))) unsigned short*)&kimg[0][0];
  const unsigned tr_base = lds_base + 2u * (((lane >> 4) & 1) * 1024u +
                                            ((lane >> 2) & 3) * 16u +
                                            (lane & 3) * 4u +
                                            (lane >> 5) * 128u);

  f32x16 acc[4];
#pragma unroll
  for (int ds = 0; ds < 4; ++ds) acc[ds] = (f32x16)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT2) : S;

  bf16x8_v k_reg[2];
  auto load_regs = [&](int kt0) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int gkey = kt0 + idx / D2;
      k_reg[c] = gkey < S ? pack8v(kp + (long)gkey * ks.s + idx % D2)
                          : (bf16x8_v)(__bf16)0.f;
    }
  };
  auto stage_img = [&](int b) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = tid * 8 + c * 4096;
      int key = idx / D2;
      int col = idx % D2;
      // blocked natural image: [col/16][key][col%16], b128 writes
      *(bf16x8_v*)&kimg[b][(col >> 4) * (KV * 16) + key * 16 + (col & 15)] =
          k_reg[c];
    }
  };
  load_regs(0);
  stage_img(0);
  if (KV < kv_end) load_regs(KV);
  __syncthreads();

  int buf = 0;
  for (int kt0 = 0; kt0 < kv_end; kt0 += KV) {
    const unsigned abase = tr_base + (unsigned)(buf * 2 * 8 * KV * 16);
    bf16x8_v db[2][2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        db[kt][kc] = pack8v(dsp + kt0 + kt * 32 + kc * 16 + hi * 8);

#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          // K^T A-frag via two 4-key transpose reads (keys sub*4)
          unsigned long long lo, hic;
          const unsigned off = (unsigned)(ds * 4096 +
                                          (kt * 32 + kc * 16) * 32);
          asm volatile(
              "ds_read_b64_tr_b16 %0, %2 offset:%3\n\t"
              "ds_read_b64_tr_b16 %1, %2 offset:%4\n\t"
              "s_waitcnt lgkmcnt(0)"
              : "=v"(lo), "=v"(hic)
              : "v"(abase + off), "i"(0), "i"(128));
          bf16x8_v a_kt;
          ((unsigned long long*)&a_kt)[0] = lo;
          ((unsigned long long*)&a_kt)[1] = hic;
          acc[ds] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a_kt, db[kt][kc], acc[ds], 0, 0, 0);
        }
      }
    }

    if (kt0 + KV < kv_end) {
      stage_img(buf ^ 1);
      if (kt0 + 2 * KV < kv_end) load_regs(kt0 + 2 * KV);
    }
    __syncthreads();
    buf ^= 1;
  }

  if (qrow < S) {
    unsigned short* dqp = dq + bb * dqs.b + hh * dqs.h + (long)qrow * dqs.s;
#pragma unroll
    for (int ds = 0; ds < 4; ++ds) {
#pragma unroll
      for (int blk = 0; blk < 4; ++blk) {
        int d0 = ds * 32 + 8 * blk + 4 * hi;
        unsigned short out4[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          out4[j] = f2bf(acc[ds][4 * blk + j]);
        *(ushort4*)(dqp + d0) = *(ushort4*)out4;
      }
    }
  }
}

}  // namespace
// Combined backward: dv pass, dk pass storing dS, dq-lite consuming it.
// TDPA_DQ_RECOMPUTE falls back to the independent recomputing dq kernel
// (and no dS workspace) for A/B and as an escape hatch.
void attn_bwd_v2_all(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                     torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                     torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
                     bool causal, double scale) {
  static const bool recompute = std::getenv("TDPA_DQ_RECOMPUTE") != nullptr;
  // The dS-store path buys -13% backward time (r01 v12) at the cost of an
  // O(S^2) bf16 workspace (B*H*S*S_pad).  That grows quadratically with
  // sequence length (536 MB at the bench shape, ~8.6 GB at S=8192/B=16) —
  // above a fixed budget switch to the recompute-dq path, which needs no
  // workspace and loses only the 13% (bound the
  // workspace for the long-context direction).
  constexpr long kMaxDsBytes = 4L << 30;  // 4 GiB cap (S=4096/B=4 stays
                                          // on the fast path, S>=8192 not)
  {
    const long Sq = q.size(2);
    const long spad_est = (Sq + QT2 - 1) / QT2 * QT2;
    const long pmul = std::getenv("TDPA_PSTORE") != nullptr ? 4 : 2;
    const long ds_bytes = (long)q.size(0) * q.size(1) * Sq * spad_est * pmul;
    if (recompute || ds_bytes > kMaxDsBytes) {
      attn_bwd_dq_v2(dout, q, k, v, lse, delta, dq, causal, scale);
      attn_bwd_dkdv_v2(dout, q, k, v, lse, delta, dk, dv, causal, scale);
      return;
    }
  }
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  const int q_per_kv = H / Hkv;
  TORCH_CHECK(D == 128);
  auto get = [](const torch::Tensor& t) {
    return Strides2{t.stride(0), t.stride(1), t.stride(2)};
  };
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid((S + QT2 - 1) / QT2, B * H), block(512);
  const long spad = (long)grid.x * QT2;
  auto dsw = torch::empty({(long)B * H, (long)S, spad}, q.options());
  unsigned short* dsp = (unsigned short*)dsw.data_ptr();
  // P register-image workspace (opt-in, TDPA_PSTORE): the dv pass stores
  // P, the dk pass reloads it (2 early b128/lane) and skips its 8-MFMA S
  // recompute.  Measured NEUTRAL at the bench shape (bwd 891-909 us both
  // ways): the saved MFMAs are repaid by the extra S^2 bf16 write+read —
  // kept for shapes where the balance may differ, default OFF so the
  // backward's workspace stays a single dS buffer.
  static const bool pstore = std::getenv("TDPA_PSTORE") != nullptr;
  torch::Tensor pwt;
  unsigned short* pw = nullptr;
  if (pstore) {
    pwt = torch::empty({(long)B * H, spad, spad}, q.options());
    pw = (unsigned short*)pwt.data_ptr();
  }
  if (causal) {
    if (pw) L_KV2(true, true, dsp, spad, pw, 1, 2);
    else L_KV2(true, true, dsp, spad, nullptr, 0, 0);
  } else {
    if (pw) L_KV2(false, true, dsp, spad, pw, 1, 2);
    else L_KV2(false, true, dsp, spad, nullptr, 0, 0);
  }
#undef L_KV2
  static const bool no_tr16 = std::getenv("TDPA_NO_TR16") != nullptr;
#define L_DQL(CC)                                                             \
  do {                                                                        \
    if (no_tr16)                                                              \
      hipLaunchKernelGGL((attn_bwd_dq_lite_kernel<CC>), grid, block, 0,       \
                         stream, (const unsigned short*)k.data_ptr(), dsp,    \
                         (unsigned short*)dq.data_ptr(), get(k), get(dq),     \
                         B, H, S, spad, q_per_kv);                            \
    else                                                                      \
      hipLaunchKernelGGL((attn_bwd_dq_lite_tr_kernel<CC>), grid, block, 0,    \
                         stream, (const unsigned short*)k.data_ptr(), dsp,    \
                         (unsigned short*)dq.data_ptr(), get(k), get(dq),     \
                         B, H, S, spad, q_per_kv);                            \
  } while (0)
  if (causal) L_DQL(true); else L_DQL(false);
#undef L_DQL
  HIP_CHECK_LAST();
}
