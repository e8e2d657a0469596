// Flash prefill/experience attention, forward-only.
//
// The training/prefill path previously materialized the [B, H, T, T] score
// tensor (QK^T GEMM -> fused causal softmax -> PV GEMM) — at seq 1024-2048
// (BASELINE configs #4/#5) the score tensor is the memory wall (VERDICT r01
// "What's missing" #2).  This kernel runs the no_grad experience forward and
// generation prefill with ONLINE softmax: scores live only in registers/LDS
// tiles, O(B*H*T*D) traffic total.
//
// Structure (CDNA4, one 256-thread block per (b, hq, 64-query tile)):
//   - Q tile [64, D] staged to LDS once (pre-scaled);
//   - per K-tile [64, D]: K staged to LDS, V staged TRANSPOSED [D, 64]
//     (the PV MFMA's B-operand wants rows = output dims);
//   - each wave owns 16 query rows: S = Q K^T via mfma_f32_16x16x32_bf16
//     chains, causal + left-pad masking on the fragments, online (m, l)
//     update with row broadcasts over the low lane nibble, P -> LDS
//     (A-operand relayout), O += P V via MFMA with per-row rescale of the
//     accumulator.
// Masks follow the causal_softmax/flash-decode conventions: query at global
// position start_pos + qi sees keys j with key_start[b] <= j <= start_pos+qi.
// GQA maps hq -> hkv = hq / (Hq/Hkv).  Backward stays on the materializing
// path (training attention backward is a later item; the experience forward
// and prefill are no_grad and dominate PPO's attention time).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int FBLOCK = 256;
constexpr int FWAVES = FBLOCK / WAVE;
constexpr int TQ = 64;  // queries per block (16 per wave)
constexpr int TK = 64;  // keys per tile

typedef __attribute__((ext_vector_type(8))) short bf16x8_fp;
typedef __attribute__((ext_vector_type(4))) float f32x4_fp;

template <int D>
__global__ __launch_bounds__(FBLOCK, 2) void flash_prefill_kernel(
    const bf16_t* __restrict__ q, const bf16_t* __restrict__ k, const bf16_t* __restrict__ v,
    const int* __restrict__ key_starts, bf16_t* __restrict__ out,
    float* __restrict__ lse /* [B,Hq,T] logsumexp of the SCALED scores; may be null */,
    int B, int Hq, int Hkv, int T, int Tk,
    int Sk /* allocated token stride of k/v (cache prefill) */, int start_pos, float scale) {
  constexpr int DPAD = D + 8;    // LDS row stride (shorts), bank-staggered
  constexpr int KPAD = TK + 8;

  const int qt = blockIdx.x;          // query tile
  const int h = blockIdx.y;           // hq
  const int b = blockIdx.z;
  const int hkv = h / (Hq / Hkv);
  const int q0 = qt * TQ;
  if (q0 >= T) return;
  const int kstart = key_starts ? key_starts[b] : 0;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;

  __shared__ unsigned short q_lds[TQ * DPAD];
  __shared__ unsigned short k_lds[TK * DPAD];
  __shared__ unsigned short vt_lds[D * KPAD];  // V transposed [d][key]
  __shared__ unsigned short p_lds[FWAVES][16 * KPAD];

  // ---- stage Q tile (pre-scaled) --------------------------------------------
  {
    const bf16_t* qp = q + (((size_t)b * Hq + h) * T) * D;
    for (int i = threadIdx.x; i < TQ * (D / 8); i += FBLOCK) {
      const int row = i / (D / 8);
      const int col = (i % (D / 8)) * 8;
      const int qrow = min(q0 + row, T - 1);
      float vals[8];
      load8<bf16_t>(qp + (size_t)qrow * D + col, vals);
      bf16x8_fp s;
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] = (short)f2bf(vals[j] * scale);
      *reinterpret_cast<bf16x8_fp*>(&q_lds[row * DPAD + col]) = s;
    }
  }
  __syncthreads();

  // this wave's 16 query rows: global positions q0+wid*16 .. +16
  const int qrow0 = wid * 16;
  // BLOCK-uniform causal bound (the K-tile loop contains barriers, so every
  // wave must take the same trip count; early waves just mask harder)
  const int qhi_pos = start_pos + min(q0 + TQ - 1, T - 1);

  float m_row[4], l_row[4];  // per (lane>>4)*4+r fragment rows (valid on all lanes)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_row[r] = -INFINITY;
    l_row[r] = 0.f;
  }
  f32x4_fp oacc[D / 16];  // O fragments [16 q x 16 d] per d-group
#pragma unroll
  for (int d = 0; d < D / 16; ++d) oacc[d] = {0.f, 0.f, 0.f, 0.f};

  const bf16_t* kp = k + (((size_t)b * Hkv + hkv) * (size_t)Sk) * D;
  const bf16_t* vp = v + (((size_t)b * Hkv + hkv) * (size_t)Sk) * D;

  const int kt_lo = (kstart / TK) * TK;
  const int kt_hi = min(Tk, qhi_pos + 1);  // no tile fully above the causal line

  for (int kt = kt_lo; kt < kt_hi; kt += TK) {
    // ---- stage K tile + V tile (transposed) ---------------------------------
    __syncthreads();
    for (int i = threadIdx.x; i < TK * (D / 8); i += FBLOCK) {
      const int row = i / (D / 8);
      const int col = (i % (D / 8)) * 8;
      const int krow = min(kt + row, Tk - 1);
      bf16x8_fp kv8 = *reinterpret_cast<const bf16x8_fp*>(kp + (size_t)krow * D + col);
      *reinterpret_cast<bf16x8_fp*>(&k_lds[row * DPAD + col]) = kv8;
      bf16x8_fp vv8 = *reinterpret_cast<const bf16x8_fp*>(vp + (size_t)krow * D + col);
#pragma unroll
      for (int j = 0; j < 8; ++j) vt_lds[(col + j) * KPAD + row] = (unsigned short)vv8[j];
    }
    __syncthreads();

    // ---- S = Q K^T for this wave's 16 q rows over 64 keys -------------------
    const int c = lane & 15;
    const int g8 = (lane >> 4) * 8;
    f32x4_fp sfrag[TK / 16];
#pragma unroll
    for (int kk = 0; kk < TK / 16; ++kk) {
      f32x4_fp acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int d = 0; d < D; d += 32) {
        bf16x8_fp af = *reinterpret_cast<const bf16x8_fp*>(&q_lds[(qrow0 + c) * DPAD + d + g8]);
        bf16x8_fp bf = *reinterpret_cast<const bf16x8_fp*>(&k_lds[(kk * 16 + c) * DPAD + d + g8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
      }
      sfrag[kk] = acc;
    }

    // ---- masking + online softmax update ------------------------------------
    // fragment element (kk, r): q row = q0 + qrow0 + (lane>>4)*4 + r (global
    // query position start_pos + that), key = kt + kk*16 + (lane&15)
    float mnew[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) mnew[r] = m_row[r];
#pragma unroll
    for (int kk = 0; kk < TK / 16; ++kk) {
      const int key = kt + kk * 16 + c;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + qrow0 + (lane >> 4) * 4 + r;
        const bool valid = key < Tk && key >= kstart && key <= start_pos + qrow && qrow < T;
        if (!valid) sfrag[kk][r] = -INFINITY;
      }
    }
    // row max across the 16 lanes of each key group and the 4 key groups
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mx = -INFINITY;
#pragma unroll
      for (int kk = 0; kk < TK / 16; ++kk) mx = fmaxf(mx, sfrag[kk][r]);
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) mx = fmaxf(mx, __shfl_xor(mx, off));
      mnew[r] = fmaxf(mnew[r], mx);
    }
    // P = exp(S - mnew), row sums, write P to LDS in A-operand layout
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < TK / 16; ++kk) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = (sfrag[kk][r] == -INFINITY || mnew[r] == -INFINITY)
                            ? 0.f
                            : __expf(sfrag[kk][r] - mnew[r]);
        sfrag[kk][r] = p;
        psum[r] += p;
        p_lds[wid][((lane >> 4) * 4 + r) * KPAD + kk * 16 + c] = f2bf(p);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) psum[r] += __shfl_xor(psum[r], off);
    }

    // ---- rescale O, l and accumulate P V ------------------------------------
    // per-row corr = exp(m_old - m_new); O fragment rows match (lane>>4)*4+r
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      corr[r] = (m_row[r] == -INFINITY || mnew[r] == -INFINITY) ? 0.f
                : __expf(m_row[r] - mnew[r]);
      l_row[r] = l_row[r] * corr[r] + psum[r];
      m_row[r] = mnew[r];
    }
#pragma unroll
    for (int d = 0; d < D / 16; ++d) {
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[d][r] *= corr[r];
    }
    // p_lds is wave-private: the compiler orders the ds_writes above against
    // the ds_reads below through the shared-memory dependence
#pragma unroll
    for (int d = 0; d < D / 16; ++d) {
#pragma unroll
      for (int kk2 = 0; kk2 < TK; kk2 += 32) {
        bf16x8_fp pf = *reinterpret_cast<const bf16x8_fp*>(&p_lds[wid][c * KPAD + kk2 + g8]);
        bf16x8_fp vf = *reinterpret_cast<const bf16x8_fp*>(&vt_lds[(d * 16 + c) * KPAD + kk2 + g8]);
        oacc[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf, vf, oacc[d], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: O / l (+ optional logsumexp for the backward) --------------
  const int c = lane & 15;
  if (lse != nullptr && c == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + qrow0 + (lane >> 4) * 4 + r;
      if (qrow < T) {
        // rows with no visible keys store +inf so exp(S - L) == 0 in bwd
        const float L = l_row[r] > 0.f ? m_row[r] + __logf(l_row[r]) : INFINITY;
        lse[((size_t)b * Hq + h) * T + qrow] = L;
      }
    }
  }
#pragma unroll
  for (int d = 0; d < D / 16; ++d) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + qrow0 + (lane >> 4) * 4 + r;
      if (qrow >= T) continue;
      const float denom = l_row[r] > 0.f ? l_row[r] : 1.f;
      const float val = oacc[d][r] / denom;
      out[(((size_t)b * Hq + h) * T + qrow) * D + d * 16 + c].u = f2bf(val);
    }
  }
}

}  // namespace

std::vector<at::Tensor> flash_prefill_lse(const at::Tensor& q, const at::Tensor& k,
                                          const at::Tensor& v,
                                          const c10::optional<at::Tensor>& key_starts,
                                          long start_pos, double scale, long tk, bool want_lse);

at::Tensor flash_prefill(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
                         const c10::optional<at::Tensor>& key_starts, long start_pos,
                         double scale, long tk) {
  return flash_prefill_lse(q, k, v, key_starts, start_pos, scale, tk, false)[0];
}

std::vector<at::Tensor> flash_prefill_lse(const at::Tensor& q, const at::Tensor& k,
                                          const at::Tensor& v,
                                          const c10::optional<at::Tensor>& key_starts,
                                          long start_pos, double scale, long tk,
                                          bool want_lse) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16 && q.dim() == 4 && q.is_contiguous());
  TORCH_CHECK(k.is_contiguous() && v.is_contiguous());
  const int B = q.size(0), Hq = q.size(1), T = q.size(2), D = q.size(3);
  const int Hkv = k.size(1), Sk = k.size(2);
  const int Tk = tk > 0 ? (int)tk : Sk;  // valid keys (<= allocated stride)
  TORCH_CHECK(Tk <= Sk);
  TORCH_CHECK(k.size(3) == D && Hq % Hkv == 0);
  auto out = at::empty_like(q);
  auto lse = want_lse ? at::empty({B, Hq, T}, q.options().dtype(at::kFloat)) : at::Tensor();
  if (q.numel() == 0) return {out, lse};
  const int* ks = nullptr;
  at::Tensor ksc;
  if (key_starts.has_value()) {
    ksc = key_starts->contiguous();
    TORCH_CHECK(ksc.numel() == B && ksc.dtype() == at::kInt);
    ks = ksc.data_ptr<int>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid((T + TQ - 1) / TQ, Hq, B);
  auto qp = reinterpret_cast<const bf16_t*>(q.data_ptr());
  auto kp = reinterpret_cast<const bf16_t*>(k.data_ptr());
  auto vp = reinterpret_cast<const bf16_t*>(v.data_ptr());
  auto op = reinterpret_cast<bf16_t*>(out.data_ptr());
  float* lp = want_lse ? lse.data_ptr<float>() : nullptr;
  switch (D) {
    case 64:
      flash_prefill_kernel<64><<<grid, FBLOCK, 0, stream>>>(
          qp, kp, vp, ks, op, lp, B, Hq, Hkv, T, Tk, Sk, (int)start_pos, (float)scale);
      break;
    case 128:
      flash_prefill_kernel<128><<<grid, FBLOCK, 0, stream>>>(
          qp, kp, vp, ks, op, lp, B, Hq, Hkv, T, Tk, Sk, (int)start_pos, (float)scale);
      break;
    default:
      TORCH_CHECK(false, "flash_prefill: head dim must be 64 or 128, got ", D);
  }
  HIP_CHECK_LAST();
  return {out, lse};
}
