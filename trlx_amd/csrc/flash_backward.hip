// Flash attention BACKWARD (training path of SURVEY.md K2).
//
// Completes the flash attention story: with this, training forwards use the
// online-softmax kernel too and the [B, H, T, T] score tensor never
// materializes anywhere (at seq 2048 / batch 8 / 16 heads the bf16 scores
// alone are 1 GiB per layer per direction).
//
// Standard two-kernel recomputation scheme (deterministic, no atomics):
//   pre:  Drow = rowsum(dO * O)                       (torch, [B,Hq,T] fp32)
//   dKV:  one block per (b, hkv, 64-key tile): for each causal q-tile,
//         recompute S^T = K (scale*Q)^T, P^T = exp(S^T - L) from the saved
//         row logsumexp L, dP^T = V dO^T, dS^T = P^T (dP^T - Drow), then
//         dV += P^T dO and dK += dS^T (scale*Q); GQA sums over the query
//         heads of the group before the single write.  No atomics.
//   dQ:   one block per (b, hq, 64-query tile): recompute S/P per key-tile,
//         dP = dO V^T, dS = P (dP - Drow), dQ += scale * dS K.
//
// Register/LDS layout (the static-LDS budget is 64 KB, so the straight
// "stage everything in both layouts" plan does not fit at D=128):
//   * A-operands of every mfma are the wave's OWN 16 rows (keys in dKV,
//     queries in dQ) x full D — loaded once from global into VGPRs
//     (D/32 bf16x8 fragments) and reused for every tile.
//   * The contraction-side tensors are staged TRANSPOSED in LDS
//     ([D, 64+8]): the natural ds_read_b128 layout for the dV/dK/dQ
//     accumulation mfmas (B rows = d), and strided 2-byte reads rebuild the
//     row-major B fragments for the S / dP recomputation.
//   * P^T / dS^T / dS relayout (D-fragment -> A-operand) goes through a
//     wave-private [16, 64+8] LDS buffer, reused sequentially — same trick
//     as the forward's P relayout (flash_prefill.hip).
// Masking (causal + left-pad key_starts) matches the forward; fully-masked
// query rows carry L = +inf so P == 0 and they contribute nothing.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BBLOCK = 256;
constexpr int BT = 64;  // tile size (both queries and keys)
constexpr int KPAD = BT + 8;

typedef __attribute__((ext_vector_type(8))) short bf16x8_fb;
typedef __attribute__((ext_vector_type(4))) float f32x4_fb;

// block-wide: stage a [BT, D] bf16 tile TRANSPOSED into tds [D, KPAD],
// scaling by mul; rows past `nrows` clamp to the last valid row (their
// columns are masked out of P later).
template <int D>
DEV void stage_tileT(const bf16_t* __restrict__ src, long row0, long nrows,
                     unsigned short* __restrict__ tds, float mul) {
  for (int i = threadIdx.x; i < BT * (D / 8); i += BBLOCK) {
    const int row = i / (D / 8);
    const int col = (i % (D / 8)) * 8;
    const bf16_t* p = src + (size_t)(row0 + min((long)row, nrows - 1)) * D + col;
    bf16x8_fb v8 = *reinterpret_cast<const bf16x8_fb*>(p);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f((unsigned short)v8[j]);
      tds[(col + j) * KPAD + row] = f2bf(mul == 1.f ? f : f * mul);
    }
  }
}

// per-lane: load this wave's 16-row A-operand fragments (row = lane&15,
// k-slice = (lane>>4)*8) for the full head dim into registers.  rowmax is
// the ABSOLUTE last valid row (inclusive): a wave whose 16-row group starts
// past the end must clamp globally, not relatively — a relative clamp read
// up to 60 rows past the tensor and page-faulted the llama (D=128) bench
// when the allocation ended at the tensor.
template <int D>
DEV void load_afrag(const bf16_t* __restrict__ src, long row0, long rowmax, int lane,
                    bf16x8_fb* frag, float mul) {
  const long row = min(row0 + (long)(lane & 15), rowmax);
  const int g8 = (lane >> 4) * 8;
#pragma unroll
  for (int dch = 0; dch < D / 32; ++dch) {
    bf16x8_fb v8 = *reinterpret_cast<const bf16x8_fb*>(src + (size_t)row * D + dch * 32 + g8);
    if (mul != 1.f) {
#pragma unroll
      for (int j = 0; j < 8; ++j) v8[j] = (short)f2bf(bf2f((unsigned short)v8[j]) * mul);
    }
    frag[dch] = v8;
  }
}

// strided B-operand fragment from a transposed [D, KPAD] tile: output row =
// tile row `col` (query/key index), k-slice elements d = dch*32 + g8 + j.
DEV bf16x8_fb load_bfragT(const unsigned short* __restrict__ tds, int col, int dch, int g8) {
  bf16x8_fb r;
#pragma unroll
  for (int j = 0; j < 8; ++j) r[j] = (short)tds[(dch * 32 + g8 + j) * KPAD + col];
  return r;
}

template <int D>
__global__ __launch_bounds__(BBLOCK, 2) void flash_bwd_dkv_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k, const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ drow, const int* __restrict__ key_starts,
    bf16_t* __restrict__ dk, bf16_t* __restrict__ dv, int B, int Hq, int Hkv, int T,
    float scale) {
  const int kt = blockIdx.x * BT;
  const int hkv = blockIdx.y;
  const int b = blockIdx.z;
  if (kt >= T) return;
  const int rep = Hq / Hkv;
  const int kstart = key_starts ? key_starts[b] : 0;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int c = lane & 15;
  const int g8 = (lane >> 4) * 8;
  const int krow0 = wid * 16;  // this wave's 16 keys within the tile

  __shared__ unsigned short qT_lds[D * KPAD];
  __shared__ unsigned short doT_lds[D * KPAD];
  __shared__ float l_lds[BT];
  __shared__ float d_lds[BT];
  __shared__ unsigned short p_lds[BBLOCK / WAVE][16 * KPAD];

  // K/V A-fragments for this wave's 16 keys, resident for the whole kernel
  bf16x8_fb k_frag[D / 32], v_frag[D / 32];
  load_afrag<D>(k + ((size_t)b * Hkv + hkv) * (size_t)T * D, kt + krow0, (long)T - 1, lane,
                k_frag, 1.f);
  load_afrag<D>(v + ((size_t)b * Hkv + hkv) * (size_t)T * D, kt + krow0, (long)T - 1, lane,
                v_frag, 1.f);

  f32x4_fb dv_acc[D / 16], dk_acc[D / 16];
#pragma unroll
  for (int d = 0; d < D / 16; ++d) {
    dv_acc[d] = {0.f, 0.f, 0.f, 0.f};
    dk_acc[d] = {0.f, 0.f, 0.f, 0.f};
  }

  for (int hq = hkv * rep; hq < (hkv + 1) * rep; ++hq) {
    for (int qt = kt; qt < T; qt += BT) {  // causal: q-tiles at/after the diagonal
      __syncthreads();
      stage_tileT<D>(q + ((size_t)b * Hq + hq) * (size_t)T * D, qt, T - qt, qT_lds, scale);
      stage_tileT<D>(dout + ((size_t)b * Hq + hq) * (size_t)T * D, qt, T - qt, doT_lds, 1.f);
      for (int i = threadIdx.x; i < BT; i += BBLOCK) {
        const int qrow = min(qt + i, T - 1);
        l_lds[i] = lse[((size_t)b * Hq + hq) * T + qrow];
        d_lds[i] = drow[((size_t)b * Hq + hq) * T + qrow];
      }
      __syncthreads();

      // S^T / dP^T fragments [this wave's 16 keys x 64 queries]
      f32x4_fb pt[BT / 16], dst[BT / 16];
#pragma unroll
      for (int qq = 0; qq < BT / 16; ++qq) {
        f32x4_fb st = {0.f, 0.f, 0.f, 0.f};
        f32x4_fb dpt = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int dch = 0; dch < D / 32; ++dch) {
          bf16x8_fb qf = load_bfragT(qT_lds, qq * 16 + c, dch, g8);
          st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[dch], qf, st, 0, 0, 0);
          bf16x8_fb df = load_bfragT(doT_lds, qq * 16 + c, dch, g8);
          dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(v_frag[dch], df, dpt, 0, 0, 0);
        }
        // element r: key = kt + krow0 + (lane>>4)*4 + r, query = qt + qq*16 + c
        const int qpos = qt + qq * 16 + c;
        const float L = l_lds[qq * 16 + c];
        const float Dr = d_lds[qq * 16 + c];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = kt + krow0 + (lane >> 4) * 4 + r;
          const bool valid = key < T && key >= kstart && key <= qpos && qpos < T;
          const float p = (valid && L < INFINITY) ? __expf(st[r] - L) : 0.f;
          st[r] = p;
          dpt[r] = p * (dpt[r] - Dr);
        }
        pt[qq] = st;
        dst[qq] = dpt;
      }

      unsigned short* pw = p_lds[wid];  // wave-private: no block barrier needed
      // dV += P^T dO  (A = P^T via relayout, B = doT natural rows = d)
#pragma unroll
      for (int qq = 0; qq < BT / 16; ++qq)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pw[((lane >> 4) * 4 + r) * KPAD + qq * 16 + c] = f2bf(pt[qq][r]);
#pragma unroll
      for (int d = 0; d < D / 16; ++d)
#pragma unroll
        for (int qch = 0; qch < BT; qch += 32) {
          bf16x8_fb pf = *reinterpret_cast<const bf16x8_fb*>(&pw[c * KPAD + qch + g8]);
          bf16x8_fb dof =
              *reinterpret_cast<const bf16x8_fb*>(&doT_lds[(d * 16 + c) * KPAD + qch + g8]);
          dv_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, dof, dv_acc[d], 0, 0, 0);
        }
      // dK += dS^T (scale*Q)  (A = dS^T via relayout, B = qT natural)
#pragma unroll
      for (int qq = 0; qq < BT / 16; ++qq)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          pw[((lane >> 4) * 4 + r) * KPAD + qq * 16 + c] = f2bf(dst[qq][r]);
#pragma unroll
      for (int d = 0; d < D / 16; ++d)
#pragma unroll
        for (int qch = 0; qch < BT; qch += 32) {
          bf16x8_fb sf = *reinterpret_cast<const bf16x8_fb*>(&pw[c * KPAD + qch + g8]);
          bf16x8_fb qtf =
              *reinterpret_cast<const bf16x8_fb*>(&qT_lds[(d * 16 + c) * KPAD + qch + g8]);
          dk_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(sf, qtf, dk_acc[d], 0, 0, 0);
        }
    }
  }

  // write this wave's 16 keys of dK/dV (C row = key-within-16, col = d)
#pragma unroll
  for (int d = 0; d < D / 16; ++d)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = kt + krow0 + (lane >> 4) * 4 + r;
      if (key >= T) continue;
      dv[(((size_t)b * Hkv + hkv) * T + key) * D + d * 16 + c].u = f2bf(dv_acc[d][r]);
      dk[(((size_t)b * Hkv + hkv) * T + key) * D + d * 16 + c].u = f2bf(dk_acc[d][r]);
    }
}

template <int D>
__global__ __launch_bounds__(BBLOCK, 2) void flash_bwd_dq_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k, const bf16_t* __restrict__ v,
    const bf16_t* __restrict__ dout, const float* __restrict__ lse,
    const float* __restrict__ drow, const int* __restrict__ key_starts,
    bf16_t* __restrict__ dq, int B, int Hq, int Hkv, int T, float scale) {
  const int qt = blockIdx.x * BT;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  if (qt >= T) return;
  const int hkv = h / (Hq / Hkv);
  const int kstart = key_starts ? key_starts[b] : 0;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int c = lane & 15;
  const int g8 = (lane >> 4) * 8;
  const int qrow0 = wid * 16;  // this wave's 16 queries within the tile

  __shared__ unsigned short kT_lds[D * KPAD];
  __shared__ unsigned short vT_lds[D * KPAD];
  __shared__ unsigned short p_lds[BBLOCK / WAVE][16 * KPAD];

  // scaled-Q / dO A-fragments for this wave's 16 queries, plus their L / Drow
  bf16x8_fb q_frag[D / 32], do_frag[D / 32];
  load_afrag<D>(q + ((size_t)b * Hq + h) * (size_t)T * D, qt + qrow0, (long)T - 1, lane,
                q_frag, scale);
  load_afrag<D>(dout + ((size_t)b * Hq + h) * (size_t)T * D, qt + qrow0, (long)T - 1, lane,
                do_frag, 1.f);
  float L_r[4], D_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = min(qt + qrow0 + (lane >> 4) * 4 + r, T - 1);
    L_r[r] = lse[((size_t)b * Hq + h) * T + qrow];
    D_r[r] = drow[((size_t)b * Hq + h) * T + qrow];
  }

  f32x4_fb dq_acc[D / 16];
#pragma unroll
  for (int d = 0; d < D / 16; ++d) dq_acc[d] = {0.f, 0.f, 0.f, 0.f};

  const int kt_lo = (kstart / BT) * BT;
  const int kt_hi = min(T, qt + BT);  // causal: no key tile past this block's last query
  for (int kt = kt_lo; kt < kt_hi; kt += BT) {
    __syncthreads();
    stage_tileT<D>(k + ((size_t)b * Hkv + hkv) * (size_t)T * D, kt, T - kt, kT_lds, 1.f);
    stage_tileT<D>(v + ((size_t)b * Hkv + hkv) * (size_t)T * D, kt, T - kt, vT_lds, 1.f);
    __syncthreads();

    unsigned short* pw = p_lds[wid];
#pragma unroll
    for (int kk = 0; kk < BT / 16; ++kk) {
      f32x4_fb s = {0.f, 0.f, 0.f, 0.f};
      f32x4_fb dp = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int dch = 0; dch < D / 32; ++dch) {
        bf16x8_fb kf = load_bfragT(kT_lds, kk * 16 + c, dch, g8);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[dch], kf, s, 0, 0, 0);
        bf16x8_fb vf = load_bfragT(vT_lds, kk * 16 + c, dch, g8);
        dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(do_frag[dch], vf, dp, 0, 0, 0);
      }
      // element r: query = qt + qrow0 + (lane>>4)*4 + r, key = kt + kk*16 + c
      const int key = kt + kk * 16 + c;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = qt + qrow0 + (lane >> 4) * 4 + r;
        const bool valid = key < T && key >= kstart && key <= qrow && qrow < T;
        const float p = (valid && L_r[r] < INFINITY) ? __expf(s[r] - L_r[r]) : 0.f;
        s[r] = p * (dp[r] - D_r[r]);  // dS
      }
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pw[((lane >> 4) * 4 + r) * KPAD + kk * 16 + c] = f2bf(s[r]);
    }
    // dQ += dS K  (A = dS via relayout, B = kT natural rows = d)
#pragma unroll
    for (int d = 0; d < D / 16; ++d)
#pragma unroll
      for (int kch = 0; kch < BT; kch += 32) {
        bf16x8_fb sf = *reinterpret_cast<const bf16x8_fb*>(&pw[c * KPAD + kch + g8]);
        bf16x8_fb ktf =
            *reinterpret_cast<const bf16x8_fb*>(&kT_lds[(d * 16 + c) * KPAD + kch + g8]);
        dq_acc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(sf, ktf, dq_acc[d], 0, 0, 0);
      }
  }

#pragma unroll
  for (int d = 0; d < D / 16; ++d)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = qt + qrow0 + (lane >> 4) * 4 + r;
      if (qrow >= T) continue;
      dq[(((size_t)b * Hq + h) * T + qrow) * D + d * 16 + c].u = f2bf(scale * dq_acc[d][r]);
    }
}

}  // namespace

// reference parity: CarperAI/trlx delegates attention backward to HF/torch
// autograd through the materialized scores; here the same gradients come from
// the recomputation kernels above (see tests/test_flash_backward.py).
std::vector<at::Tensor> flash_prefill_bwd(const at::Tensor& q, const at::Tensor& k,
                                          const at::Tensor& v, const at::Tensor& out,
                                          const at::Tensor& dout, const at::Tensor& lse,
                                          const c10::optional<at::Tensor>& key_starts,
                                          double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16 && q.dim() == 4 && q.is_contiguous(),
              "flash_prefill_bwd: q must be contiguous bf16 [B,Hq,T,D]");
  TORCH_CHECK(k.is_contiguous() && v.is_contiguous() && lse.is_contiguous());
  const int B = q.size(0), Hq = q.size(1), T = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  TORCH_CHECK(k.size(2) == T, "flash_prefill_bwd: training path only (Tk == T)");
  TORCH_CHECK(Hq % Hkv == 0);
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto dc = dout.contiguous();
  auto drow = (dc.to(at::kFloat) * out.to(at::kFloat)).sum(-1).contiguous();  // [B,Hq,T]
  const int* ks = nullptr;
  at::Tensor ksc;
  if (key_starts.has_value()) {
    ksc = key_starts->contiguous();
    ks = ksc.data_ptr<int>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  auto qp = reinterpret_cast<const bf16_t*>(q.data_ptr());
  auto kp = reinterpret_cast<const bf16_t*>(k.data_ptr());
  auto vp = reinterpret_cast<const bf16_t*>(v.data_ptr());
  auto dop = reinterpret_cast<const bf16_t*>(dc.data_ptr());
  auto dqp = reinterpret_cast<bf16_t*>(dq.data_ptr());
  auto dkp = reinterpret_cast<bf16_t*>(dk.data_ptr());
  auto dvp = reinterpret_cast<bf16_t*>(dv.data_ptr());
  dim3 gk((T + BT - 1) / BT, Hkv, B);
  dim3 gq((T + BT - 1) / BT, Hq, B);
#define LAUNCH_BWD(DV)                                                                    \
  do {                                                                                    \
    flash_bwd_dkv_kernel<DV><<<gk, BBLOCK, 0, stream>>>(                                  \
        qp, kp, vp, dop, lse.data_ptr<float>(), drow.data_ptr<float>(), ks, dkp, dvp, B,  \
        Hq, Hkv, T, (float)scale);                                                        \
    flash_bwd_dq_kernel<DV><<<gq, BBLOCK, 0, stream>>>(                                   \
        qp, kp, vp, dop, lse.data_ptr<float>(), drow.data_ptr<float>(), ks, dqp, B, Hq,   \
        Hkv, T, (float)scale);                                                            \
  } while (0)
  switch (D) {
    case 64: LAUNCH_BWD(64); break;
    case 128: LAUNCH_BWD(128); break;
    default: TORCH_CHECK(false, "flash_prefill_bwd: head dim must be 64 or 128");
  }
#undef LAUNCH_BWD
  HIP_CHECK_LAST();
  return {dq, dk, dv};
}
