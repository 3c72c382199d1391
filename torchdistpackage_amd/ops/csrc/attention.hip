// Flash-attention forward + backward for gfx950 (CDNA4 MFMA, online softmax).
//
// Blockwise algorithm per the reference's in-repo spec
// (/root/reference/explore/flash-attn/tile_attn.py:100-212): per Q-tile,
// iterate K/V tiles maintaining running row max m and exp-sum l; LSE is saved
// for the backward; backward = two recompute passes (dq; dk+dv) plus a
// delta = rowsum(do*o) preprocess.  No atomics anywhere.
//
// v1 structure (v0 + the two biggest levers from the CDNA4 guide):
//   - LDS XOR swizzles on every MFMA B-fragment tile: without them a
//     ds_read_b128 of 16 rows at one column slot is up to 16-way
//     bank-conflicted (guide 6.4: this one conflict was half an attention
//     kernel's time).  256B-row tiles use col ^= (row & (D/8-1)) << 3,
//     32-short-row tiles use col ^= ((row>>2) & 3) << 3 (short-index space).
//   - Strided global addressing: q/k/v/o/do/dq/dk/dv are (B, H, S, D) VIEWS
//     with arbitrary b/h/s strides (last dim contiguous).  The training path
//     passes views straight into the fused (S, B, 3*H*D) qkv buffer, so no
//     permute-contiguous copies happen anywhere around attention.
//
// Workgroup = 256 threads = 4 waves; wave owns 16 q-rows (fwd/dq) or 16 keys
// (dkdv); K/V tiles of 32 staged in LDS; v_mfma_f32_16x16x32_bf16
// (A/B: 8 bf16/lane K-contiguous; C/D: col=lane&15, row=(lane>>4)*4+reg —
// layout verified on hardware by tests/test_ops_gpu.py::test_mfma_layout_probe).
// Supported: head_dim 64 / 128, any S, causal or full.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"
#include <cstdlib>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int WQ = 16;    // rows per wave
constexpr int KT = 32;    // keys (or q rows, in dkdv) per LDS tile
// waves per workgroup: the block's Q-tile (or key-tile) is NW*WQ rows; K/V
// (or Q/dO) tiles are re-read S/(NW*WQ) times, so wider blocks cut the
// dominant HBM traffic proportionally.
constexpr int NW_FWD = 16;  // 256-row q-tile
constexpr int NW_DQ = 8;
constexpr int NW_DKDV = 8;  // 128-key tile

struct Strides {            // element strides of a (B,H,S,D) view
  long b, h, s;
};

DEVINL bf16x8_v pack8(const unsigned short* p) {
  return *(const bf16x8_v*)p;
}

// swizzled index (in shorts) into a [rows][D] tile with 2*D-byte rows
template <int D>
DEVINL int swzD(int row, int col) {
  return row * D + (col ^ ((row & (D / 8 - 1)) << 3));
}

// swizzled index into a [rows][KT=32] tile (64-byte rows)
DEVINL int swz32(int row, int col) {
  return row * KT + (col ^ (((row >> 2) & 3) << 3));
}

// ---------------------------------------------------------------- forward

template <int D, bool CAUSAL, int NW>
__launch_bounds__(NW * 64)
__global__ void attn_fwd_kernel(const unsigned short* __restrict__ q,
                                const unsigned short* __restrict__ k,
                                const unsigned short* __restrict__ v,
                                unsigned short* __restrict__ o,
                                float* __restrict__ lse,
                                Strides qs, Strides ks, Strides vs, Strides os,
                                int B, int H, int S, float scale,
                                int q_per_kv) {
  constexpr int KC = D / 32;
  constexpr int DC = D / 16;
  __shared__ __attribute__((aligned(16))) unsigned short k_lds[KT * D];        // K[key][d], swzD
  __shared__ __attribute__((aligned(16))) unsigned short vt_lds[D * KT];       // V^T[d][key], swz32
  __shared__ __attribute__((aligned(16))) unsigned short p_lds[NW * 16 * KT];  // P, swz32 per wave

  constexpr int QT = NW * WQ;
  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int qbase = blockIdx.x * QT;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int hkv = hh / q_per_kv;
  const unsigned short* qp = q + bb * qs.b + hh * qs.h;
  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const unsigned short* vp = v + bb * vs.b + hkv * vs.h;

  const int qrow0 = qbase + wid * WQ;
  bf16x8_v a_q[KC];
  {
    int r = qrow0 + l15;
    long rr = (r < S ? r : S - 1) * qs.s;
#pragma unroll
    for (int c = 0; c < KC; ++c)
      a_q[c] = pack8(qp + rr + c * 32 + lg * 8);
  }

  float m_run[4], l_run[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) { m_run[i] = -1e30f; l_run[i] = 0.f; }
  f32x4 acc_o[DC];
#pragma unroll
  for (int d = 0; d < DC; ++d) acc_o[d] = (f32x4)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT) : S;

  // T14 async-STAGE split: each thread owns ONE 8-elem chunk of the K and V
  // tiles (KT*D == NW*64*8 at D=128); the next tile's global loads are
  // issued right after the LDS write barrier, so they fly under the MFMA
  // compute of the current tile (guide 6.15: +17% on attention).
  const int st_idx = tid * 8;
  const int st_key = st_idx / D;
  const int st_col = st_idx % D;
  const bool st_own = st_idx < KT * D;
  bf16x8_v k_reg, v_reg;

  auto stage_load = [&](int kt0) {
    if (!st_own) return;
    int gkey = kt0 + st_key;
    if (gkey < S) {
      k_reg = pack8(kp + (long)gkey * ks.s + st_col);
      v_reg = pack8(vp + (long)gkey * vs.s + st_col);
    } else {
      k_reg = (bf16x8_v)(__bf16)0.f;
      v_reg = (bf16x8_v)(__bf16)0.f;
    }
  };

  stage_load(0);

  for (int kt0 = 0; kt0 < kv_end; kt0 += KT) {
    __syncthreads();
    if (st_own) {
      *(bf16x8_v*)&k_lds[swzD<D>(st_key, st_col)] = k_reg;
      const unsigned short* vsrc = (const unsigned short*)&v_reg;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vt_lds[swz32(st_col + j, st_key)] = vsrc[j];
    }
    __syncthreads();
    if (kt0 + KT < kv_end) stage_load(kt0 + KT);

    f32x4 s_acc[KT / 16];
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg) {
      s_acc[kg] = (f32x4)(0.f);
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_v b_k = pack8(&k_lds[swzD<D>(kg * 16 + l15, c * 32 + lg * 8)]);
        s_acc[kg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_q[c], b_k, s_acc[kg], 0, 0, 0);
      }
    }

    float p_val[KT / 16][4];
    float alpha[4];
    {
      float tile_max[4];
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) tile_max[rr] = -1e30f;
#pragma unroll
      for (int kg = 0; kg < KT / 16; ++kg) {
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int qrow = qrow0 + lg * 4 + rr;
          int key = kt0 + kg * 16 + l15;
          float sv = s_acc[kg][rr] * scale;
          bool valid = key < S && (!CAUSAL || key <= qrow);
          sv = valid ? sv : -1e30f;
          s_acc[kg][rr] = sv;
          tile_max[rr] = fmaxf(tile_max[rr], sv);
        }
      }
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        tile_max[rr] = group16_max(tile_max[rr]);
        float m_new = fmaxf(m_run[rr], tile_max[rr]);
        alpha[rr] = __expf(m_run[rr] - m_new);
        m_run[rr] = m_new;
        float psum = 0.f;
#pragma unroll
        for (int kg = 0; kg < KT / 16; ++kg) {
          float pv = __expf(s_acc[kg][rr] - m_new);
          p_val[kg][rr] = pv;
          psum += pv;
        }
        psum = group16_sum(psum);
        l_run[rr] = l_run[rr] * alpha[rr] + psum;
      }
    }

#pragma unroll
    for (int d = 0; d < DC; ++d)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) acc_o[d][rr] *= alpha[rr];

    unsigned short* pw = &p_lds[wid * 16 * KT];
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        pw[swz32(lg * 4 + rr, kg * 16 + l15)] = f2bf(p_val[kg][rr]);

    bf16x8_v a_p = pack8(&pw[swz32(l15, lg * 8)]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_v = pack8(&vt_lds[swz32(d * 16 + l15, lg * 8)]);
      acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_p, b_v, acc_o[d], 0, 0, 0);
    }
  }

  unsigned short* op = o + bb * os.b + hh * os.h;
  float* lsep = lse + ((long)bh) * S;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int qrow = qrow0 + lg * 4 + rr;
    if (qrow >= S) continue;
    float inv_l = l_run[rr] > 0.f ? 1.f / l_run[rr] : 0.f;
#pragma unroll
    for (int d = 0; d < DC; ++d)
      op[(long)qrow * os.s + d * 16 + l15] = f2bf(acc_o[d][rr] * inv_l);
    if (l15 == 0)
      lsep[qrow] = m_run[rr] + __logf(l_run[rr] > 0.f ? l_run[rr] : 1.f);
  }
}

// ---------------------------------------------------------------- backward

__global__ void attn_delta_kernel(const unsigned short* __restrict__ o,
                                  const unsigned short* __restrict__ dout,
                                  float* __restrict__ delta,
                                  Strides os, Strides ds,
                                  int B, int H, int S, int D) {
  // one wave per row, GRID-STRIDED (a one-row-per-block launch was
  // dispatch-bound: 65k tiny blocks cost 3x the memory floor)
  const long rows = (long)B * H * S;
  const int wpb = blockDim.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long stride = (long)gridDim.x * wpb;
  for (long row = (long)blockIdx.x * wpb + threadIdx.x / WAVE; row < rows;
       row += stride) {
    int s = row % S;
    int h = (row / S) % H;
    int b = row / ((long)S * H);
    const unsigned short* orow = o + b * os.b + h * os.h + (long)s * os.s;
    const unsigned short* drow = dout + b * ds.b + h * ds.h + (long)s * ds.s;
    float acc = 0.f;
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      bf16x8_v ov = pack8(orow + i);
      bf16x8_v dv = pack8(drow + i);
      const unsigned short* op = (const unsigned short*)&ov;
      const unsigned short* dp = (const unsigned short*)&dv;
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += bf2f(op[j]) * bf2f(dp[j]);
    }
    acc = wave_sum(acc);
    if (lane == 0) delta[row] = acc;
  }
}

template <int D, bool CAUSAL, int NW>
__launch_bounds__(NW * 64)
__global__ void attn_bwd_dq_kernel(const unsigned short* __restrict__ q,
                                   const unsigned short* __restrict__ k,
                                   const unsigned short* __restrict__ v,
                                   const unsigned short* __restrict__ dout,
                                   const float* __restrict__ lse,
                                   const float* __restrict__ delta,
                                   unsigned short* __restrict__ dq,
                                   Strides qs, Strides ks, Strides vs,
                                   Strides dos, Strides dqs,
                                   int B, int H, int S, float scale,
                                   int q_per_kv) {
  constexpr int KC = D / 32;
  constexpr int DC = D / 16;
  __shared__ __attribute__((aligned(16))) unsigned short k_lds[KT * D];     // K[key][d], swzD
  __shared__ __attribute__((aligned(16))) unsigned short kt_lds[D * KT];    // K^T[d][key], swz32
  __shared__ __attribute__((aligned(16))) unsigned short v_lds[KT * D];     // V[key][d], swzD
  __shared__ __attribute__((aligned(16))) unsigned short ds_lds[NW * 16 * KT];  // dS, swz32

  constexpr int QT = NW * WQ;
  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int qbase = blockIdx.x * QT;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int hkv = hh / q_per_kv;
  const unsigned short* qp = q + bb * qs.b + hh * qs.h;
  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const unsigned short* vp = v + bb * vs.b + hkv * vs.h;
  const unsigned short* dop = dout + bb * dos.b + hh * dos.h;
  const float* lsep = lse + (long)bh * S;
  const float* delp = delta + (long)bh * S;

  const int qrow0 = qbase + wid * WQ;
  bf16x8_v a_q[KC], a_do[KC];
  float my_lse[4], my_delta[4];
  {
    int r = qrow0 + l15;
    int rc = r < S ? r : S - 1;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      a_q[c] = pack8(qp + (long)rc * qs.s + c * 32 + lg * 8);
      a_do[c] = pack8(dop + (long)rc * dos.s + c * 32 + lg * 8);
    }
#pragma unroll
    for (int rr4 = 0; rr4 < 4; ++rr4) {
      int row = qrow0 + lg * 4 + rr4;
      my_lse[rr4] = row < S ? lsep[row] : 0.f;
      my_delta[rr4] = row < S ? delp[row] : 0.f;
    }
  }

  f32x4 acc_dq[DC];
#pragma unroll
  for (int d = 0; d < DC; ++d) acc_dq[d] = (f32x4)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT) : S;

  // T14 async-STAGE split (see fwd kernel)
  const int st_idx = tid * 8;
  const int st_key = st_idx / D;
  const int st_col = st_idx % D;
  const bool st_own = st_idx < KT * D;
  bf16x8_v k_reg, v_reg;

  auto stage_load = [&](int kt0) {
    if (!st_own) return;
    int gkey = kt0 + st_key;
    if (gkey < S) {
      k_reg = pack8(kp + (long)gkey * ks.s + st_col);
      v_reg = pack8(vp + (long)gkey * vs.s + st_col);
    } else {
      k_reg = (bf16x8_v)(__bf16)0.f;
      v_reg = (bf16x8_v)(__bf16)0.f;
    }
  };

  stage_load(0);

  for (int kt0 = 0; kt0 < kv_end; kt0 += KT) {
    __syncthreads();
    if (st_own) {
      *(bf16x8_v*)&k_lds[swzD<D>(st_key, st_col)] = k_reg;
      *(bf16x8_v*)&v_lds[swzD<D>(st_key, st_col)] = v_reg;
      const unsigned short* ksrc = (const unsigned short*)&k_reg;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kt_lds[swz32(st_col + j, st_key)] = ksrc[j];
    }
    __syncthreads();
    if (kt0 + KT < kv_end) stage_load(kt0 + KT);

    f32x4 s_acc[KT / 16], dp_acc[KT / 16];
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg) {
      s_acc[kg] = (f32x4)(0.f);
      dp_acc[kg] = (f32x4)(0.f);
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_v b_k = pack8(&k_lds[swzD<D>(kg * 16 + l15, c * 32 + lg * 8)]);
        s_acc[kg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_q[c], b_k, s_acc[kg], 0, 0, 0);
        bf16x8_v b_v = pack8(&v_lds[swzD<D>(kg * 16 + l15, c * 32 + lg * 8)]);
        dp_acc[kg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_do[c], b_v, dp_acc[kg], 0, 0, 0);
      }
    }

    unsigned short* dsw = &ds_lds[wid * 16 * KT];
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int qrow = qrow0 + lg * 4 + rr;
        int key = kt0 + kg * 16 + l15;
        bool valid = key < S && qrow < S && (!CAUSAL || key <= qrow);
        float p = valid ?
            __expf(s_acc[kg][rr] * scale - my_lse[rr]) : 0.f;
        float ds = p * (dp_acc[kg][rr] - my_delta[rr]) * scale;
        dsw[swz32(lg * 4 + rr, kg * 16 + l15)] = f2bf(ds);
      }
    }

    bf16x8_v a_ds = pack8(&dsw[swz32(l15, lg * 8)]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_kt = pack8(&kt_lds[swz32(d * 16 + l15, lg * 8)]);
      acc_dq[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_ds, b_kt, acc_dq[d], 0, 0, 0);
    }
  }

  unsigned short* dqp = dq + bb * dqs.b + hh * dqs.h;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int qrow = qrow0 + lg * 4 + rr;
    if (qrow >= S) continue;
#pragma unroll
    for (int d = 0; d < DC; ++d)
      dqp[(long)qrow * dqs.s + d * 16 + l15] = f2bf(acc_dq[d][rr]);
  }
}

template <int D, bool CAUSAL, int NW>
__launch_bounds__(NW * 64)
__global__ void attn_bwd_dkdv_kernel(const unsigned short* __restrict__ q,
                                     const unsigned short* __restrict__ k,
                                     const unsigned short* __restrict__ v,
                                     const unsigned short* __restrict__ dout,
                                     const float* __restrict__ lse,
                                     const float* __restrict__ delta,
                                     unsigned short* __restrict__ dk,
                                     unsigned short* __restrict__ dv,
                                     Strides qs, Strides ks, Strides vs,
                                     Strides dos, Strides dks, Strides dvs,
                                     int B, int H, int S, float scale,
                                     int q_per_kv) {
  constexpr int KC = D / 32;
  constexpr int DC = D / 16;
  __shared__ __attribute__((aligned(16))) unsigned short q_lds[KT * D];     // Q[qrow][d], swzD
  __shared__ __attribute__((aligned(16))) unsigned short qt_lds[D * KT];    // Q^T[d][qrow], swz32
  __shared__ __attribute__((aligned(16))) unsigned short do_lds[KT * D];    // dO[qrow][d], swzD
  __shared__ __attribute__((aligned(16))) unsigned short dot_lds[D * KT];   // dO^T[d][qrow], swz32
  __shared__ __attribute__((aligned(16))) unsigned short st_lds[NW * 16 * KT];  // P^T / dS^T, swz32
  __shared__ __attribute__((aligned(16))) float lse_lds[KT];
  __shared__ __attribute__((aligned(16))) float del_lds[KT];

  constexpr int QT = NW * WQ;
  const int bh = blockIdx.y;
  const int bb = bh / H, hh = bh % H;
  const int kbase = blockIdx.x * QT;
  if (kbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int hkv = hh / q_per_kv;
  const unsigned short* qp = q + bb * qs.b + hh * qs.h;
  const unsigned short* kp = k + bb * ks.b + hkv * ks.h;
  const unsigned short* vp = v + bb * vs.b + hkv * vs.h;
  const unsigned short* dop = dout + bb * dos.b + hh * dos.h;
  const float* lsep = lse + (long)bh * S;
  const float* delp = delta + (long)bh * S;

  const int key0 = kbase + wid * WQ;
  bf16x8_v a_k[KC], a_v[KC];
  {
    int r = key0 + l15;
    long rc = (long)(r < S ? r : S - 1);
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      a_k[c] = pack8(kp + rc * ks.s + c * 32 + lg * 8);
      a_v[c] = pack8(vp + rc * vs.s + c * 32 + lg * 8);
    }
  }

  f32x4 acc_dk[DC], acc_dv[DC];
#pragma unroll
  for (int d = 0; d < DC; ++d) {
    acc_dk[d] = (f32x4)(0.f);
    acc_dv[d] = (f32x4)(0.f);
  }

  const int qt_start = CAUSAL ? kbase : 0;

  // T14 async-STAGE split (see fwd kernel)
  const int st_idx = tid * 8;
  const int st_row = st_idx / D;
  const int st_col = st_idx % D;
  const bool st_own = st_idx < KT * D;
  bf16x8_v q_reg, do_reg;
  float lse_reg = 0.f, del_reg = 0.f;

  auto stage_load = [&](int qt0) {
    if (st_own) {
      int grow = qt0 + st_row;
      if (grow < S) {
        q_reg = pack8(qp + (long)grow * qs.s + st_col);
        do_reg = pack8(dop + (long)grow * dos.s + st_col);
      } else {
        q_reg = (bf16x8_v)(__bf16)0.f;
        do_reg = (bf16x8_v)(__bf16)0.f;
      }
    }
    if (tid < KT) {
      int grow = qt0 + tid;
      lse_reg = grow < S ? lsep[grow] : 0.f;
      del_reg = grow < S ? delp[grow] : 0.f;
    }
  };

  stage_load(qt_start);

  for (int qt0 = qt_start; qt0 < S; qt0 += KT) {
    __syncthreads();
    if (st_own) {
      *(bf16x8_v*)&q_lds[swzD<D>(st_row, st_col)] = q_reg;
      *(bf16x8_v*)&do_lds[swzD<D>(st_row, st_col)] = do_reg;
      const unsigned short* qsrc = (const unsigned short*)&q_reg;
      const unsigned short* dsrc = (const unsigned short*)&do_reg;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        qt_lds[swz32(st_col + j, st_row)] = qsrc[j];
        dot_lds[swz32(st_col + j, st_row)] = dsrc[j];
      }
    }
    if (tid < KT) {
      lse_lds[tid] = lse_reg;
      del_lds[tid] = del_reg;
    }
    __syncthreads();
    if (qt0 + KT < S) stage_load(qt0 + KT);

    f32x4 st_acc[KT / 16], dpt_acc[KT / 16];
#pragma unroll
    for (int qg = 0; qg < KT / 16; ++qg) {
      st_acc[qg] = (f32x4)(0.f);
      dpt_acc[qg] = (f32x4)(0.f);
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_v b_q = pack8(&q_lds[swzD<D>(qg * 16 + l15, c * 32 + lg * 8)]);
        st_acc[qg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_k[c], b_q, st_acc[qg], 0, 0, 0);
        bf16x8_v b_do = pack8(&do_lds[swzD<D>(qg * 16 + l15, c * 32 + lg * 8)]);
        dpt_acc[qg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_v[c], b_do, dpt_acc[qg], 0, 0, 0);
      }
    }

    unsigned short* stw = &st_lds[wid * 16 * KT];
#pragma unroll
    for (int qg = 0; qg < KT / 16; ++qg) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int key = key0 + lg * 4 + rr;
        int qrow = qt0 + qg * 16 + l15;
        bool valid = key < S && qrow < S && (!CAUSAL || key <= qrow);
        float p = valid ?
            __expf(st_acc[qg][rr] * scale - lse_lds[qg * 16 + l15]) : 0.f;
        st_acc[qg][rr] = p;
        stw[swz32(lg * 4 + rr, qg * 16 + l15)] = f2bf(p);
      }
    }

    bf16x8_v a_pt = pack8(&stw[swz32(l15, lg * 8)]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_dot = pack8(&dot_lds[swz32(d * 16 + l15, lg * 8)]);
      acc_dv[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_pt, b_dot, acc_dv[d], 0, 0, 0);
    }

#pragma unroll
    for (int qg = 0; qg < KT / 16; ++qg) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        float ds = st_acc[qg][rr] *
            (dpt_acc[qg][rr] - del_lds[qg * 16 + l15]) * scale;
        stw[swz32(lg * 4 + rr, qg * 16 + l15)] = f2bf(ds);
      }
    }

    bf16x8_v a_dst = pack8(&stw[swz32(l15, lg * 8)]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_qt = pack8(&qt_lds[swz32(d * 16 + l15, lg * 8)]);
      acc_dk[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_dst, b_qt, acc_dk[d], 0, 0, 0);
    }
  }

  unsigned short* dkp = dk + bb * dks.b + hh * dks.h;
  unsigned short* dvp = dv + bb * dvs.b + hh * dvs.h;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int key = key0 + lg * 4 + rr;
    if (key >= S) continue;
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      dkp[(long)key * dks.s + d * 16 + l15] = f2bf(acc_dk[d][rr]);
      dvp[(long)key * dvs.s + d * 16 + l15] = f2bf(acc_dv[d][rr]);
    }
  }
}

Strides get_strides(const torch::Tensor& t) {
  TORCH_CHECK(t.dim() == 4 && t.stride(3) == 1,
              "attention tensors must be 4D (B,H,S,D) with contiguous D");
  return Strides{t.stride(0), t.stride(1), t.stride(2)};
}

}  // namespace

void attn_fwd_v2(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor o, torch::Tensor lse, bool causal,
                 double scale);
void attn_bwd_dq_v2(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                    torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                    torch::Tensor dq, bool causal, double scale);
void attn_bwd_v2_all(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                     torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                     torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
                     bool causal, double scale);
void attn_bwd_dkdv_v2(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                      torch::Tensor v, torch::Tensor lse, torch::Tensor delta,
                      torch::Tensor dk, torch::Tensor dv, bool causal,
                      double scale);

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    bool causal, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16,
              "attn_fwd: bf16 CUDA tensors required");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  TORCH_CHECK(H % Hkv == 0, "n_head must be a multiple of n_kv_head");
  const int q_per_kv = H / Hkv;
  TORCH_CHECK(k.size(2) == S, "cross-attention S_kv != S_q not supported yet");
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));
  static const bool force_v1 = std::getenv("TDPA_ATTN_V1") != nullptr;
  if (D == 128 && !force_v1) {
    attn_fwd_v2(q, k, v, o, lse, causal, scale);
    return {o, lse};
  }
  auto stream = at::cuda::getCurrentHIPStream();
  constexpr int QT = NW_FWD * WQ;
  dim3 grid((S + QT - 1) / QT, B * H), block(NW_FWD * 64);
  Strides qs = get_strides(q), ks = get_strides(k), vs = get_strides(v),
          os = get_strides(o);
  const unsigned short* qp = (const unsigned short*)q.data_ptr();
  const unsigned short* kp = (const unsigned short*)k.data_ptr();
  const unsigned short* vp = (const unsigned short*)v.data_ptr();
  unsigned short* op = (unsigned short*)o.data_ptr();
  float* lp = lse.data_ptr<float>();
#define LAUNCH(DD, CC)                                                        \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, CC, NW_FWD>), grid, block, 0,       \
                     stream, qp, kp, vp, op, lp, qs, ks, vs, os, B, H, S,     \
                     (float)scale, q_per_kv)
  if (D == 128) { if (causal) LAUNCH(128, true); else LAUNCH(128, false); }
  else          { if (causal) LAUNCH(64, true);  else LAUNCH(64, false);  }
#undef LAUNCH
  HIP_CHECK_LAST();
  return {o, lse};
}

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    torch::Tensor dq, torch::Tensor dk,
                                    torch::Tensor dv,
                                    bool causal, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  TORCH_CHECK(H % Hkv == 0);
  const int q_per_kv = H / Hkv;
  // GQA: dk/dv buffers must be per-Q-HEAD partials (B,H,S,D); the wrapper
  // sums groups of q_per_kv afterwards
  TORCH_CHECK(dk.size(1) == H && dv.size(1) == H,
              "dk/dv must be expanded to H q-heads for GQA");
  TORCH_CHECK(D == 64 || D == 128);
  auto delta = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  Strides qs = get_strides(q), ks = get_strides(k), vs = get_strides(v),
          os = get_strides(o), dos = get_strides(dout),
          dqs = get_strides(dq), dks = get_strides(dk), dvs = get_strides(dv);

  {
    long rows = (long)B * H * S;
    long blocks = (rows + 3) / 4;
    if (blocks > 2048) blocks = 2048;   // grid-stride the rest
    hipLaunchKernelGGL(attn_delta_kernel, dim3((unsigned)blocks), dim3(256),
                       0, stream, (const unsigned short*)o.data_ptr(),
                       (const unsigned short*)dout.data_ptr(),
                       delta.data_ptr<float>(), os, dos, B, H, S, D);
  }

  static const bool force_v1b = std::getenv("TDPA_ATTN_V1") != nullptr;
  if (D == 128 && !force_v1b) {
    attn_bwd_v2_all(dout, q, k, v, lse, delta, dq, dk, dv, causal, scale);
    HIP_CHECK_LAST();
    return {dq, dk, dv};
  }
  constexpr int QT_DQ = NW_DQ * WQ;
  constexpr int QT_KV = NW_DKDV * WQ;
  dim3 grid_dq((S + QT_DQ - 1) / QT_DQ, B * H), block_dq(NW_DQ * 64);
  dim3 grid_kv((S + QT_KV - 1) / QT_KV, B * H), block_kv(NW_DKDV * 64);
  const unsigned short* qp = (const unsigned short*)q.data_ptr();
  const unsigned short* kp = (const unsigned short*)k.data_ptr();
  const unsigned short* vp = (const unsigned short*)v.data_ptr();
  const unsigned short* dop = (const unsigned short*)dout.data_ptr();
  const float* lp = lse.data_ptr<float>();
  const float* delp = delta.data_ptr<float>();
#define LAUNCH_BWD(DD, CC)                                                    \
  do {                                                                        \
    hipLaunchKernelGGL((attn_bwd_dq_kernel<DD, CC, NW_DQ>), grid_dq,         \
                       block_dq, 0, stream, qp, kp, vp, dop, lp, delp,        \
                       (unsigned short*)dq.data_ptr(), qs, ks, vs, dos, dqs,  \
                       B, H, S, (float)scale, q_per_kv);                      \
    hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DD, CC, NW_DKDV>), grid_kv,      \
                       block_kv, 0, stream, qp, kp, vp, dop, lp, delp,        \
                       (unsigned short*)dk.data_ptr(),                        \
                       (unsigned short*)dv.data_ptr(), qs, ks, vs, dos,       \
                       dks, dvs, B, H, S, (float)scale, q_per_kv);            \
  } while (0)
  if (D == 128) { if (causal) LAUNCH_BWD(128, true); else LAUNCH_BWD(128, false); }
  else          { if (causal) LAUNCH_BWD(64, true);  else LAUNCH_BWD(64, false);  }
#undef LAUNCH_BWD
  HIP_CHECK_LAST();
  return {dq, dk, dv};
}
