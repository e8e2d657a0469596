// Fused decode attention — the generation hot kernel.
//
// q [B, Hq, 1, D] vs KV cache [B, Hkv, S, D] (bf16, contiguous, GQA-aware),
// per-row valid lengths.  Memory-bound flash-decode: online softmax, zero
// [B, H, S] score materialization, KV loads vectorized at 16 B/lane (guide
// Guideline 13: 64 lanes x 8 bf16 = one 1 KiB transaction covering 512/D keys
// per instruction).  One 256-thread block per (b, h); the 4 waves x
// (512/D key-groups) partial accumulators are merged through LDS.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;

template <int D>
__global__ void attn_decode_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ kc,
                                   const bf16_t* __restrict__ vc, const int* __restrict__ seq_lens,
                                   const int* __restrict__ seq_starts, bf16_t* __restrict__ out,
                                   int Hq, int Hkv, int S, float scale) {
  constexpr int G = D / 8;          // lanes cooperating on one key
  constexpr int KPW = WAVE / G;     // keys per wave per iteration
  constexpr int NPART = NWAVES * KPW;  // independent (m, s, o) partials

  const int b = blockIdx.x / Hq;
  const int hq = blockIdx.x % Hq;
  const int hkv = hq / (Hq / Hkv);
  const int len = seq_lens[b];
  const int kstart = seq_starts ? seq_starts[b] : 0;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int kgrp = lane / G;        // which key within the wave's group
  const int d0 = (lane % G) * 8;    // this lane's slice of D

  const bf16_t* qp = q + ((size_t)b * Hq + hq) * D;
  const bf16_t* kbase = kc + ((size_t)b * Hkv + hkv) * S * D;
  const bf16_t* vbase = vc + ((size_t)b * Hkv + hkv) * S * D;

  // q fragment for this lane's d-slice
  float qf[8];
  load8<bf16_t>(qp + d0, qf);
#pragma unroll
  for (int i = 0; i < 8; ++i) qf[i] *= scale;

  float m = -INFINITY, s = 0.f;
  float o[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) o[i] = 0.f;

  for (int sbase = kstart + wid * KPW; sbase < len; sbase += NWAVES * KPW) {
    const int key = sbase + kgrp;
    if (key < len) {
      float kf[8], vf[8];
      load8<bf16_t>(kbase + (size_t)key * D + d0, kf);
      load8<bf16_t>(vbase + (size_t)key * D + d0, vf);
      float partial = 0.f;
#pragma unroll
      for (int i = 0; i < 8; ++i) partial += qf[i] * kf[i];
      // reduce across the G lanes of this key group (contiguous lanes)
#pragma unroll
      for (int off = G / 2; off > 0; off >>= 1) partial += __shfl_xor(partial, off);
      const float score = partial;
      // branchless online update (first valid key: m=-inf -> corr=0)
      const float mnew = fmaxf(m, score);
      const float corr = (m > -INFINITY) ? __expf(m - mnew) : 0.f;
      const float pw = __expf(score - mnew);
      s = s * corr + pw;
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = o[i] * corr + pw * vf[i];
      m = mnew;
    }
  }

  // merge the NPART partial accumulators via LDS
  __shared__ float ms_buf[NPART][2];
  __shared__ float o_buf[NPART][D];
  const int part = wid * KPW + kgrp;
  if (lane % G == 0) {
    ms_buf[part][0] = m;
    ms_buf[part][1] = s;
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) o_buf[part][d0 + i] = o[i];
  __syncthreads();

  // final combine: thread t handles output dim t (D <= BLOCK)
  if (threadIdx.x < D) {
    float mstar = -INFINITY;
#pragma unroll
    for (int p = 0; p < NPART; ++p) mstar = fmaxf(mstar, ms_buf[p][0]);
    float sstar = 0.f, acc = 0.f;
#pragma unroll
    for (int p = 0; p < NPART; ++p) {
      const float mp = ms_buf[p][0];
      if (mp == -INFINITY) continue;
      const float w = __expf(mp - mstar);
      sstar += ms_buf[p][1] * w;
      acc += o_buf[p][threadIdx.x] * w;
    }
    const float res = (sstar > 0.f) ? acc / sstar : 0.f;
    bf16_t* op = out + ((size_t)b * Hq + hq) * D;
    op[threadIdx.x].u = f2bf(res);
  }
}

// Fused decode_prep + attention for MHA (Hq == Hkv): one kernel per
// (b, h) does the q/k RoPE + cache append for ITS head (3 rows, one wave
// each), a block barrier, then the flash-decode loop — the separate prep
// launch and the [B, Hq, 1, D] q round-trip through HBM disappear.  The
// freshly appended key/value at cache position *cache_idx are read back
// through the same CU's L1 after __syncthreads() (intra-block global RAW).
template <int D, bool INTERLEAVED>
__global__ void fused_decode_attn_kernel(const bf16_t* __restrict__ qkv, bf16_t* __restrict__ kc,
                                         bf16_t* __restrict__ vc,
                                         const int* __restrict__ seq_lens,
                                         const int* __restrict__ seq_starts,
                                         const float* __restrict__ cs,
                                         const float* __restrict__ sn,
                                         const long* __restrict__ cache_idx,
                                         bf16_t* __restrict__ out, int H, int S, int rot,
                                         float scale) {
  constexpr int G = D / 8;
  constexpr int KPW = WAVE / G;
  constexpr int NPART = NWAVES * KPW;

  const int b = blockIdx.x / H;
  const int h = blockIdx.x % H;
  const int len = seq_lens[b];
  const int kstart = seq_starts ? seq_starts[b] : 0;
  const long pos = *cache_idx;

  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;

  __shared__ float q_lds[D];

  // ---- stage 1: split + RoPE + cache append for this head --------------
  {
    const bf16_t* src = qkv + ((size_t)b * 3 * H + (size_t)wid * H + h) * D;
    bf16_t* kdst = kc + (((size_t)b * H + h) * S + pos) * D;
    bf16_t* vdst = vc + (((size_t)b * H + h) * S + pos) * D;
    if (wid < 2 && cs != nullptr) {  // q (wid 0) and k (wid 1): rotate
      const int p = (int)pos - kstart;
      const float* c = cs + (size_t)p * (rot / 2);
      const float* sn_p = sn + (size_t)p * (rot / 2);
      for (int i = lane; i < rot / 2; i += WAVE) {
        const int i1 = INTERLEAVED ? 2 * i : i;
        const int i2 = INTERLEAVED ? 2 * i + 1 : i + rot / 2;
        const float x1 = bf2f(src[i1].u);
        const float x2 = bf2f(src[i2].u);
        const float r1 = x1 * c[i] - x2 * sn_p[i];
        const float r2 = x2 * c[i] + x1 * sn_p[i];
        if (wid == 0) {
          q_lds[i1] = r1 * scale;
          q_lds[i2] = r2 * scale;
        } else {
          kdst[i1].u = f2bf(r1);
          kdst[i2].u = f2bf(r2);
        }
      }
      for (int i = rot + lane; i < D; i += WAVE) {
        if (wid == 0) q_lds[i] = bf2f(src[i].u) * scale;
        else kdst[i] = src[i];
      }
    } else if (wid < 2) {  // no rope (learned/alibi-free path)
      for (int i = lane; i < D; i += WAVE) {
        if (wid == 0) q_lds[i] = bf2f(src[i].u) * scale;
        else kdst[i] = src[i];
      }
    } else if (wid == 2) {  // v: straight copy
      const int D4 = D / 4;
      for (int i = lane; i < D4; i += WAVE)
        reinterpret_cast<short4v*>(vdst)[i] = reinterpret_cast<const short4v*>(src)[i];
    }
  }
  __syncthreads();

  // ---- stage 2: flash-decode over the cache (incl. the new key) --------
  const int kgrp = lane / G;
  const int d0 = (lane % G) * 8;
  const bf16_t* kbase = kc + ((size_t)b * H + h) * S * D;
  const bf16_t* vbase = vc + ((size_t)b * H + h) * S * D;
  float qf[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) qf[i] = q_lds[d0 + i];

  float m = -INFINITY, s = 0.f;
  float o[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) o[i] = 0.f;
  for (int sbase = kstart + wid * KPW; sbase < len; sbase += NWAVES * KPW) {
    const int key = sbase + kgrp;
    if (key < len) {
      float kf[8], vf[8];
      load8<bf16_t>(kbase + (size_t)key * D + d0, kf);
      load8<bf16_t>(vbase + (size_t)key * D + d0, vf);
      float partial = 0.f;
#pragma unroll
      for (int i = 0; i < 8; ++i) partial += qf[i] * kf[i];
#pragma unroll
      for (int off = G / 2; off > 0; off >>= 1) partial += __shfl_xor(partial, off);
      const float score = partial;
      const float mnew = fmaxf(m, score);
      const float corr = (m > -INFINITY) ? __expf(m - mnew) : 0.f;
      const float pw = __expf(score - mnew);
      s = s * corr + pw;
#pragma unroll
      for (int i = 0; i < 8; ++i) o[i] = o[i] * corr + pw * vf[i];
      m = mnew;
    }
  }

  __shared__ float ms_buf[NPART][2];
  __shared__ float o_buf[NPART][D];
  const int part = wid * KPW + kgrp;
  if (lane % G == 0) {
    ms_buf[part][0] = m;
    ms_buf[part][1] = s;
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) o_buf[part][d0 + i] = o[i];
  __syncthreads();
  if (threadIdx.x < D) {
    float mstar = -INFINITY;
#pragma unroll
    for (int p = 0; p < NPART; ++p) mstar = fmaxf(mstar, ms_buf[p][0]);
    float sstar = 0.f, acc = 0.f;
#pragma unroll
    for (int p = 0; p < NPART; ++p) {
      const float mp = ms_buf[p][0];
      if (mp == -INFINITY) continue;
      const float w = __expf(mp - mstar);
      sstar += ms_buf[p][1] * w;
      acc += o_buf[p][threadIdx.x] * w;
    }
    const float res = (sstar > 0.f) ? acc / sstar : 0.f;
    bf16_t* op = out + ((size_t)b * H + h) * D;
    op[threadIdx.x].u = f2bf(res);
  }
}

}  // namespace

at::Tensor fused_decode_attention(const at::Tensor& qkv, at::Tensor& kcache, at::Tensor& vcache,
                                  const at::Tensor& seq_lens,
                                  const c10::optional<at::Tensor>& seq_starts,
                                  const c10::optional<at::Tensor>& cos,
                                  const c10::optional<at::Tensor>& sin,
                                  const at::Tensor& cache_idx, long rot, bool interleaved,
                                  double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == at::kBFloat16 && qkv.is_contiguous());
  const int B = qkv.size(0);
  const int Hkv = kcache.size(1), S = kcache.size(2), D = kcache.size(3);
  const int QKV = qkv.numel() / B;
  TORCH_CHECK(QKV == 3 * Hkv * D, "fused_decode_attention requires Hq == Hkv");
  TORCH_CHECK(seq_lens.dtype() == at::kInt && seq_lens.numel() == B);
  auto out = at::empty({B, Hkv, 1, D}, qkv.options());
  if (B == 0) return out;
  const float* cs = nullptr;
  const float* sn = nullptr;
  if (cos.has_value()) {
    cs = cos->data_ptr<float>();
    sn = sin->data_ptr<float>();
  }
  const int* ss = nullptr;
  at::Tensor ssc;
  if (seq_starts.has_value()) {
    ssc = seq_starts->contiguous();
    TORCH_CHECK(ssc.numel() == B && ssc.dtype() == at::kInt);
    ss = ssc.data_ptr<int>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = B * Hkv;
  auto qp = reinterpret_cast<const bf16_t*>(qkv.data_ptr());
  auto kp = reinterpret_cast<bf16_t*>(kcache.data_ptr());
  auto vp = reinterpret_cast<bf16_t*>(vcache.data_ptr());
  auto op = reinterpret_cast<bf16_t*>(out.data_ptr());
  auto sl = seq_lens.data_ptr<int>();
  auto ci = cache_idx.data_ptr<long>();
#define LAUNCH_FUSED(DV)                                                                         do {                                                                                             if (interleaved)                                                                                 fused_decode_attn_kernel<DV, true><<<grid, BLOCK, 0, stream>>>(                                    qp, kp, vp, sl, ss, cs, sn, ci, op, Hkv, S, (int)rot, (float)scale);                     else                                                                                             fused_decode_attn_kernel<DV, false><<<grid, BLOCK, 0, stream>>>(                                   qp, kp, vp, sl, ss, cs, sn, ci, op, Hkv, S, (int)rot, (float)scale);                   } while (0)
  switch (D) {
    case 32: LAUNCH_FUSED(32); break;
    case 64: LAUNCH_FUSED(64); break;
    case 128: LAUNCH_FUSED(128); break;
    case 256: LAUNCH_FUSED(256); break;
    default: TORCH_CHECK(false, "fused_decode_attention: head dim must be 32/64/128/256");
  }
#undef LAUNCH_FUSED
  HIP_CHECK_LAST();
  return out;
}

at::Tensor attention_decode(const at::Tensor& q, const at::Tensor& kc, const at::Tensor& vc,
                            const at::Tensor& seq_lens, double scale,
                            const c10::optional<at::Tensor>& seq_starts) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == at::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(kc.is_contiguous() && vc.is_contiguous());
  TORCH_CHECK(q.dim() == 4 && q.size(2) == 1, "attention_decode: q must be [B,Hq,1,D]");
  const int B = q.size(0), Hq = q.size(1), D = q.size(3);
  const int Hkv = kc.size(1), S = kc.size(2);
  TORCH_CHECK(kc.size(3) == D && Hq % Hkv == 0);
  TORCH_CHECK(seq_lens.dtype() == at::kInt && seq_lens.numel() == B);
  auto out = at::empty_like(q);
  if (B == 0) return out;
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = B * Hq;
  auto qp = reinterpret_cast<const bf16_t*>(q.data_ptr());
  auto kp = reinterpret_cast<const bf16_t*>(kc.data_ptr());
  auto vp = reinterpret_cast<const bf16_t*>(vc.data_ptr());
  auto op = reinterpret_cast<bf16_t*>(out.data_ptr());
  auto sl = seq_lens.data_ptr<int>();
  const int* ss = nullptr;
  at::Tensor ssc;
  if (seq_starts.has_value()) {
    ssc = seq_starts->contiguous();
    TORCH_CHECK(ssc.numel() == B && ssc.dtype() == at::kInt);
    ss = ssc.data_ptr<int>();
  }
  switch (D) {
    case 32:
      attn_decode_kernel<32><<<grid, BLOCK, 0, stream>>>(qp, kp, vp, sl, ss, op, Hq, Hkv, S, (float)scale);
      break;
    case 64:
      attn_decode_kernel<64><<<grid, BLOCK, 0, stream>>>(qp, kp, vp, sl, ss, op, Hq, Hkv, S, (float)scale);
      break;
    case 128:
      attn_decode_kernel<128><<<grid, BLOCK, 0, stream>>>(qp, kp, vp, sl, ss, op, Hq, Hkv, S, (float)scale);
      break;
    case 256:
      attn_decode_kernel<256><<<grid, BLOCK, 0, stream>>>(qp, kp, vp, sl, ss, op, Hq, Hkv, S, (float)scale);
      break;
    default:
      TORCH_CHECK(false, "attention_decode: head dim must be 32/64/128/256, got ", D);
  }
  HIP_CHECK_LAST();
  return out;
}
