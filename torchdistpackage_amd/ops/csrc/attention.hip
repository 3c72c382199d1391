// Flash-attention forward for gfx950 (CDNA4 MFMA, online softmax).
//
// Blockwise algorithm per the reference's in-repo spec
// (/root/reference/explore/flash-attn/tile_attn.py:100-154): per Q-tile,
// iterate K/V tiles maintaining running row max m and exp-sum l; output is
// rescaled by exp(m_old - m_new) at each tile; LSE = m + log(l) is saved for
// the backward.
//
// v0 structure (correctness-first; optimization ladder applied in-place later):
//   - workgroup = 256 threads = 4 waves; each wave owns 16 q-rows, the block
//     owns a 64-row Q tile of one (batch, head)
//   - K/V tiles of 32 keys staged in LDS (V stored transposed [D][32] so the
//     PV B-fragment reads are contiguous ds_read_b128)
//   - QK^T and PV on v_mfma_f32_16x16x32_bf16 (A/B: 8 bf16/lane, K-contig
//     per lane; C/D: col=lane&15, row=(lane>>4)*4+reg)
//   - P is round-tripped through LDS to convert C-layout -> A-layout
//   - f32 accumulation throughout; bf16 only at memory boundaries
//
// Supported: head_dim 64 / 128, any S (K-tail masked), causal or full.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int QT = 64;    // q rows per workgroup
constexpr int WQ = 16;    // q rows per wave
constexpr int KT = 32;    // keys per kv tile
constexpr int NWAVE = 4;

DEVINL bf16x8_v pack8(const unsigned short* p) {
  // reinterpret 8 contiguous bf16 (16 B, must be 16B-aligned in LDS/global)
  return *(const bf16x8_v*)p;
}

template <int D, bool CAUSAL>
__launch_bounds__(256)
__global__ void attn_fwd_kernel(const unsigned short* __restrict__ q,
                                const unsigned short* __restrict__ k,
                                const unsigned short* __restrict__ v,
                                unsigned short* __restrict__ o,
                                float* __restrict__ lse,
                                int B, int H, int S, float scale) {
  constexpr int KC = D / 32;     // QK^T k-chunks
  constexpr int DC = D / 16;     // PV d-chunks (output col tiles)
  // LDS: K [KT][D] bf16, V^T [D][KT] bf16, P [NWAVE][16][KT+pad?] bf16
  __shared__ unsigned short k_lds[KT][D];
  __shared__ unsigned short vt_lds[D][KT];
  __shared__ unsigned short p_lds[NWAVE][16][KT];

  const int bh = blockIdx.y;          // batch*H + head
  const int qtile = blockIdx.x;
  const int qbase = qtile * QT;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l15 = lane & 15;
  const int lg = lane >> 4;           // 16-lane group id (0..3)

  const long bh_off = (long)bh * S * D;
  const unsigned short* qp = q + bh_off;
  const unsigned short* kp = k + bh_off;
  const unsigned short* vp = v + bh_off;

  // ---- load Q fragments: wave w covers rows qbase + w*16 + (0..15)
  const int qrow0 = qbase + wid * WQ;
  bf16x8_v a_q[KC];
  {
    int r = qrow0 + l15;
    int rr = r < S ? r : S - 1;       // clamp; invalid rows never stored
#pragma unroll
    for (int c = 0; c < KC; ++c)
      a_q[c] = pack8(qp + (long)rr * D + c * 32 + lg * 8);
  }

  // ---- running state: 4 rows per lane (rows lg*4+rr of this wave's 16)
  float m_run[4], l_run[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) { m_run[i] = -1e30f; l_run[i] = 0.f; }
  f32x4 acc_o[DC];
#pragma unroll
  for (int d = 0; d < DC; ++d) acc_o[d] = (f32x4)(0.f);

  const int kv_end = CAUSAL ? min(S, qbase + QT) : S;

  for (int kt0 = 0; kt0 < kv_end; kt0 += KT) {
    // ---- stage K tile [KT][D] and V^T tile [D][KT]
    __syncthreads();
    {
      // 256 threads load KT*D elements; 8 bf16 per thread-step
      const int elems = KT * D;
      for (int idx = tid * 8; idx < elems; idx += 256 * 8) {
        int key = idx / D;
        int col = idx % D;
        int gkey = kt0 + key;
        if (gkey < S) {
          bf16x8_v kv8 = pack8(kp + (long)gkey * D + col);
          *(bf16x8_v*)&k_lds[key][col] = kv8;
          bf16x8_v vv8 = pack8(vp + (long)gkey * D + col);
          // transpose-store V: vt[col + j][key]
          const unsigned short* vsrc = (const unsigned short*)&vv8;
#pragma unroll
          for (int j = 0; j < 8; ++j) vt_lds[col + j][key] = vsrc[j];
        } else {
          // zero-fill tail (scores masked anyway, V contributes 0)
          for (int j = 0; j < 8; ++j) {
            k_lds[key][col + j] = 0;
            vt_lds[col + j][key] = 0;
          }
        }
      }
    }
    __syncthreads();

    // ---- S = scale * Q K^T for this wave's 16 rows x KT keys
    f32x4 s_acc[KT / 16];
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg) {
      s_acc[kg] = (f32x4)(0.f);
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_v b_k = pack8(&k_lds[kg * 16 + l15][c * 32 + lg * 8]);
        s_acc[kg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_q[c], b_k, s_acc[kg], 0, 0, 0);
      }
    }

    // ---- mask + online softmax update
    // C layout: value (row = lg*4 + rr, col = l15) per reg rr
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

    // ---- rescale O accumulators by alpha (acc rows = lg*4+rr)
#pragma unroll
    for (int d = 0; d < DC; ++d)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) acc_o[d][rr] *= alpha[rr];

    // ---- P -> LDS (C layout -> A layout via memory)
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        p_lds[wid][lg * 4 + rr][kg * 16 + l15] = f2bf(p_val[kg][rr]);
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0): LDS writes visible in-wave
    // (wave-private p_lds slice: no cross-wave barrier needed)

    // ---- O += P V  (A = P [16][KT], B = V [KT][16-col chunk])
    bf16x8_v a_p = pack8(&p_lds[wid][l15][lg * 8]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_v = pack8(&vt_lds[d * 16 + l15][lg * 8]);
      acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_p, b_v, acc_o[d], 0, 0, 0);
    }
  }

  // ---- epilogue: O /= l, store bf16; LSE = m + log(l)
  unsigned short* op = o + bh_off;
  float* lsep = lse + (long)bh * S;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int qrow = qrow0 + lg * 4 + rr;
    if (qrow >= S) continue;
    float inv_l = l_run[rr] > 0.f ? 1.f / l_run[rr] : 0.f;
#pragma unroll
    for (int d = 0; d < DC; ++d)
      op[(long)qrow * D + d * 16 + l15] = f2bf(acc_o[d][rr] * inv_l);
    if (l15 == 0)
      lsep[qrow] = m_run[rr] + __logf(l_run[rr] > 0.f ? l_run[rr] : 1.f);
  }
}

}  // namespace

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16,
              "attn_fwd: bf16 CUDA tensors required");
  TORCH_CHECK(q.dim() == 4, "q must be (B,H,S,D)");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(k.size(2) == S, "cross-attention S_kv != S_q not supported yet");
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid((S + QT - 1) / QT, B * H), block(256);
  const unsigned short* qp = (const unsigned short*)q.data_ptr();
  const unsigned short* kp = (const unsigned short*)k.data_ptr();
  const unsigned short* vp = (const unsigned short*)v.data_ptr();
  unsigned short* op = (unsigned short*)o.data_ptr();
  float* lp = lse.data_ptr<float>();
#define LAUNCH(DD, CC)                                                        \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, CC>), grid, block, 0, stream, qp,   \
                     kp, vp, op, lp, B, H, S, (float)scale)
  if (D == 128) { if (causal) LAUNCH(128, true); else LAUNCH(128, false); }
  else          { if (causal) LAUNCH(64, true);  else LAUNCH(64, false);  }
#undef LAUNCH
  HIP_CHECK_LAST();
  return {o, lse};
}

// ===========================================================================
// Flash-attention backward (FA2-style two recompute passes, reference math
// spec: /root/reference/explore/flash-attn/tile_attn.py:156-212).
//   delta kernel: delta = rowsum(do * o)               (memory-bound)
//   dq kernel:    per Q-tile, loop KV tiles: recompute P from LSE,
//                 dp = do V^T, ds = P (dp - delta) scale, dq += ds K
//   dkdv kernel:  per KV-tile, loop Q tiles: recompute P^T,
//                 dv += P^T do, ds^T = P^T (dp^T - delta) scale, dk += ds^T q
// No atomics: each output row is owned by exactly one workgroup.
// ===========================================================================

namespace {

__global__ void attn_delta_kernel(const unsigned short* __restrict__ o,
                                  const unsigned short* __restrict__ dout,
                                  float* __restrict__ delta,
                                  long rows, int D) {
  // one wave per row; vectorized 8-wide
  long row = (long)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (row >= rows) return;
  const int lane = threadIdx.x % WAVE;
  const unsigned short* orow = o + row * D;
  const unsigned short* drow = dout + row * D;
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

template <int D, bool CAUSAL>
__launch_bounds__(256)
__global__ void attn_bwd_dq_kernel(const unsigned short* __restrict__ q,
                                   const unsigned short* __restrict__ k,
                                   const unsigned short* __restrict__ v,
                                   const unsigned short* __restrict__ dout,
                                   const float* __restrict__ lse,
                                   const float* __restrict__ delta,
                                   unsigned short* __restrict__ dq,
                                   int B, int H, int S, float scale) {
  constexpr int KC = D / 32;
  constexpr int DC = D / 16;
  __shared__ unsigned short k_lds[KT][D];     // K[key][d] for QK^T
  __shared__ unsigned short kt_lds[D][KT];    // K^T[d][key] for ds@K
  __shared__ unsigned short v_lds[KT][D];     // V[key][d] for do V^T
  __shared__ unsigned short ds_lds[NWAVE][16][KT];

  const int bh = blockIdx.y;
  const int qtile = blockIdx.x;
  const int qbase = qtile * QT;
  if (qbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const long bh_off = (long)bh * S * D;
  const unsigned short* qp = q + bh_off;
  const unsigned short* kp = k + bh_off;
  const unsigned short* vp = v + bh_off;
  const unsigned short* dop = dout + bh_off;
  const float* lsep = lse + (long)bh * S;
  const float* delp = delta + (long)bh * S;

  const int qrow0 = qbase + wid * WQ;
  bf16x8_v a_q[KC], a_do[KC];
  float my_lse[4], my_delta[4];
  {
    int r = qrow0 + l15;
    int rr = r < S ? r : S - 1;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      a_q[c] = pack8(qp + (long)rr * D + c * 32 + lg * 8);
      a_do[c] = pack8(dop + (long)rr * D + c * 32 + lg * 8);
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

  for (int kt0 = 0; kt0 < kv_end; kt0 += KT) {
    __syncthreads();
    {
      const int elems = KT * D;
      for (int idx = tid * 8; idx < elems; idx += 256 * 8) {
        int key = idx / D;
        int col = idx % D;
        int gkey = kt0 + key;
        if (gkey < S) {
          bf16x8_v kv8 = pack8(kp + (long)gkey * D + col);
          *(bf16x8_v*)&k_lds[key][col] = kv8;
          bf16x8_v vv8 = pack8(vp + (long)gkey * D + col);
          *(bf16x8_v*)&v_lds[key][col] = vv8;
          const unsigned short* ksrc = (const unsigned short*)&kv8;
#pragma unroll
          for (int j = 0; j < 8; ++j) kt_lds[col + j][key] = ksrc[j];
        } else {
          for (int j = 0; j < 8; ++j) {
            k_lds[key][col + j] = 0;
            v_lds[key][col + j] = 0;
            kt_lds[col + j][key] = 0;
          }
        }
      }
    }
    __syncthreads();

    // S and dP tiles for 16 q-rows x KT keys
    f32x4 s_acc[KT / 16], dp_acc[KT / 16];
#pragma unroll
    for (int kg = 0; kg < KT / 16; ++kg) {
      s_acc[kg] = (f32x4)(0.f);
      dp_acc[kg] = (f32x4)(0.f);
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_v b_k = pack8(&k_lds[kg * 16 + l15][c * 32 + lg * 8]);
        s_acc[kg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_q[c], b_k, s_acc[kg], 0, 0, 0);
        bf16x8_v b_v = pack8(&v_lds[kg * 16 + l15][c * 32 + lg * 8]);
        dp_acc[kg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_do[c], b_v, dp_acc[kg], 0, 0, 0);
      }
    }

    // ds = P * (dP - delta) * scale, P = exp(S*scale - lse)
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
        ds_lds[wid][lg * 4 + rr][kg * 16 + l15] = f2bf(ds);
      }
    }

    // dq += ds @ K   (A = ds [16][KT], B = K [KT][16 d-cols])
    bf16x8_v a_ds = pack8(&ds_lds[wid][l15][lg * 8]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_kt = pack8(&kt_lds[d * 16 + l15][lg * 8]);
      acc_dq[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_ds, b_kt, acc_dq[d], 0, 0, 0);
    }
  }

  unsigned short* dqp = dq + bh_off;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int qrow = qrow0 + lg * 4 + rr;
    if (qrow >= S) continue;
#pragma unroll
    for (int d = 0; d < DC; ++d)
      dqp[(long)qrow * D + d * 16 + l15] = f2bf(acc_dq[d][rr]);
  }
}

template <int D, bool CAUSAL>
__launch_bounds__(256)
__global__ void attn_bwd_dkdv_kernel(const unsigned short* __restrict__ q,
                                     const unsigned short* __restrict__ k,
                                     const unsigned short* __restrict__ v,
                                     const unsigned short* __restrict__ dout,
                                     const float* __restrict__ lse,
                                     const float* __restrict__ delta,
                                     unsigned short* __restrict__ dk,
                                     unsigned short* __restrict__ dv,
                                     int B, int H, int S, float scale) {
  constexpr int KC = D / 32;
  constexpr int DC = D / 16;
  // workgroup owns 64 keys (wave w: keys ktile*64 + w*16 + 0..15);
  // loops over q tiles of 32 rows
  __shared__ unsigned short q_lds[KT][D];     // Q[qrow][d]   (KT=32 q rows)
  __shared__ unsigned short qt_lds[D][KT];    // Q^T[d][qrow]
  __shared__ unsigned short do_lds[KT][D];    // dO[qrow][d]
  __shared__ unsigned short dot_lds[D][KT];   // dO^T[d][qrow]
  __shared__ unsigned short st_lds[NWAVE][16][KT];  // P^T / dS^T staging
  __shared__ float lse_lds[KT];
  __shared__ float del_lds[KT];

  const int bh = blockIdx.y;
  const int ktile = blockIdx.x;
  const int kbase = ktile * QT;   // 64 keys per workgroup
  if (kbase >= S) return;
  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid % WAVE;
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const long bh_off = (long)bh * S * D;
  const unsigned short* qp = q + bh_off;
  const unsigned short* kp = k + bh_off;
  const unsigned short* vp = v + bh_off;
  const unsigned short* dop = dout + bh_off;
  const float* lsep = lse + (long)bh * S;
  const float* delp = delta + (long)bh * S;

  // this wave's 16 keys: fragments of K and V (A-operand layout)
  const int key0 = kbase + wid * WQ;
  bf16x8_v a_k[KC], a_v[KC];
  {
    int r = key0 + l15;
    int rr = r < S ? r : S - 1;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      a_k[c] = pack8(kp + (long)rr * D + c * 32 + lg * 8);
      a_v[c] = pack8(vp + (long)rr * D + c * 32 + lg * 8);
    }
  }

  f32x4 acc_dk[DC], acc_dv[DC];
#pragma unroll
  for (int d = 0; d < DC; ++d) {
    acc_dk[d] = (f32x4)(0.f);
    acc_dv[d] = (f32x4)(0.f);
  }

  const int qt_start = CAUSAL ? (kbase / KT) * KT : 0;

  for (int qt0 = qt_start; qt0 < S; qt0 += KT) {
    __syncthreads();
    {
      const int elems = KT * D;
      for (int idx = tid * 8; idx < elems; idx += 256 * 8) {
        int row = idx / D;
        int col = idx % D;
        int grow = qt0 + row;
        if (grow < S) {
          bf16x8_v qv8 = pack8(qp + (long)grow * D + col);
          *(bf16x8_v*)&q_lds[row][col] = qv8;
          bf16x8_v dv8 = pack8(dop + (long)grow * D + col);
          *(bf16x8_v*)&do_lds[row][col] = dv8;
          const unsigned short* qs = (const unsigned short*)&qv8;
          const unsigned short* ds = (const unsigned short*)&dv8;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            qt_lds[col + j][row] = qs[j];
            dot_lds[col + j][row] = ds[j];
          }
        } else {
          for (int j = 0; j < 8; ++j) {
            q_lds[row][col + j] = 0;
            do_lds[row][col + j] = 0;
            qt_lds[col + j][row] = 0;
            dot_lds[col + j][row] = 0;
          }
        }
      }
      if (tid < KT) {
        int grow = qt0 + tid;
        lse_lds[tid] = grow < S ? lsep[grow] : 0.f;
        del_lds[tid] = grow < S ? delp[grow] : 0.f;
      }
    }
    __syncthreads();

    // S^T and dP^T tiles: 16 keys x KT q-rows
    f32x4 st_acc[KT / 16], dpt_acc[KT / 16];
#pragma unroll
    for (int qg = 0; qg < KT / 16; ++qg) {
      st_acc[qg] = (f32x4)(0.f);
      dpt_acc[qg] = (f32x4)(0.f);
#pragma unroll
      for (int c = 0; c < KC; ++c) {
        bf16x8_v b_q = pack8(&q_lds[qg * 16 + l15][c * 32 + lg * 8]);
        st_acc[qg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_k[c], b_q, st_acc[qg], 0, 0, 0);
        bf16x8_v b_do = pack8(&do_lds[qg * 16 + l15][c * 32 + lg * 8]);
        dpt_acc[qg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_v[c], b_do, dpt_acc[qg], 0, 0, 0);
      }
    }

    // P^T = exp(S^T*scale - lse[qcol]); stage P^T for the dv MFMA
#pragma unroll
    for (int qg = 0; qg < KT / 16; ++qg) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int key = key0 + lg * 4 + rr;     // C-layout row = key here
        int qrow = qt0 + qg * 16 + l15;   // C-layout col = q
        bool valid = key < S && qrow < S && (!CAUSAL || key <= qrow);
        float p = valid ?
            __expf(st_acc[qg][rr] * scale - lse_lds[qg * 16 + l15]) : 0.f;
        st_acc[qg][rr] = p;   // reuse as P^T
        st_lds[wid][lg * 4 + rr][qg * 16 + l15] = f2bf(p);
      }
    }

    // dv += P^T @ dO  (A = P^T [16keys][KT q], B = dO [q][16 d-cols])
    bf16x8_v a_pt = pack8(&st_lds[wid][l15][lg * 8]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_dot = pack8(&dot_lds[d * 16 + l15][lg * 8]);
      acc_dv[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_pt, b_dot, acc_dv[d], 0, 0, 0);
    }

    // dS^T = P^T * (dP^T - delta[qcol]) * scale; restage
    __syncthreads();  // st_lds reuse: everyone done reading P^T
#pragma unroll
    for (int qg = 0; qg < KT / 16; ++qg) {
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        float ds = st_acc[qg][rr] *
            (dpt_acc[qg][rr] - del_lds[qg * 16 + l15]) * scale;
        st_lds[wid][lg * 4 + rr][qg * 16 + l15] = f2bf(ds);
      }
    }

    // dk += dS^T @ Q  (A = dS^T [16keys][KT q], B = Q [q][16 d-cols])
    bf16x8_v a_dst = pack8(&st_lds[wid][l15][lg * 8]);
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      bf16x8_v b_qt = pack8(&qt_lds[d * 16 + l15][lg * 8]);
      acc_dk[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_dst, b_qt, acc_dk[d], 0, 0, 0);
    }
  }

  unsigned short* dkp = dk + bh_off;
  unsigned short* dvp = dv + bh_off;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int key = key0 + lg * 4 + rr;
    if (key >= S) continue;
#pragma unroll
    for (int d = 0; d < DC; ++d) {
      dkp[(long)key * D + d * 16 + l15] = f2bf(acc_dk[d][rr]);
      dvp[(long)key * D + d * 16 + l15] = f2bf(acc_dv[d][rr]);
    }
  }
}

}  // namespace

std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal, double scale) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto delta = torch::empty({B, H, S}, q.options().dtype(torch::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();

  {  // delta = rowsum(do * o)
    long rows = (long)B * H * S;
    int waves_per_block = 4;
    long blocks = (rows + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(attn_delta_kernel, dim3((unsigned)blocks), dim3(256),
                       0, stream, (const unsigned short*)o.data_ptr(),
                       (const unsigned short*)dout.contiguous().data_ptr(),
                       delta.data_ptr<float>(), rows, D);
  }

  dim3 grid((S + QT - 1) / QT, B * H), block(256);
  const unsigned short* qp = (const unsigned short*)q.data_ptr();
  const unsigned short* kp = (const unsigned short*)k.data_ptr();
  const unsigned short* vp = (const unsigned short*)v.data_ptr();
  const unsigned short* dop = (const unsigned short*)dout.contiguous().data_ptr();
  const float* lp = lse.data_ptr<float>();
  const float* delp = delta.data_ptr<float>();
#define LAUNCH_BWD(DD, CC)                                                    \
  do {                                                                        \
    hipLaunchKernelGGL((attn_bwd_dq_kernel<DD, CC>), grid, block, 0, stream,  \
                       qp, kp, vp, dop, lp, delp,                             \
                       (unsigned short*)dq.data_ptr(), B, H, S,               \
                       (float)scale);                                         \
    hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DD, CC>), grid, block, 0,        \
                       stream, qp, kp, vp, dop, lp, delp,                     \
                       (unsigned short*)dk.data_ptr(),                        \
                       (unsigned short*)dv.data_ptr(), B, H, S,               \
                       (float)scale);                                         \
  } while (0)
  if (D == 128) { if (causal) LAUNCH_BWD(128, true); else LAUNCH_BWD(128, false); }
  else          { if (causal) LAUNCH_BWD(64, true);  else LAUNCH_BWD(64, false);  }
#undef LAUNCH_BWD
  HIP_CHECK_LAST();
  return {dq, dk, dv};
}
