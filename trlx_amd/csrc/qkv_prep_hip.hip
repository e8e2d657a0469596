#include "hip/hip_runtime.h"
// Fused multi-token QKV preparation, forward + backward.
//
// Training/prefill (T>1) eager path cost per layer: 3 slice+transpose copies
// out of the fused qkv GEMM, 2 RoPE launches, 1 q*scale elementwise — and
// their mirror images in backward (profile tprof: ~9% of the train step in
// copies).  This pair fuses it: ONE kernel splits [B,T,(Hq+2Hkv)*D] into
// q/k/v [B,H,T,D], applies RoPE at per-(b,t) positions, and folds the
// 1/sqrt(D) query scale; the backward applies the inverse rotation to the
// incoming gradients and re-packs them into the fused-GEMM layout.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

// FWD=true: qkv -> (q,k,v);  FWD=false: (dq,dk,dv) -> dqkv with inverse RoPE
template <bool INTERLEAVED, bool FWD>
__global__ void qkv_prep_kernel(bf16_t* __restrict__ qkv, bf16_t* __restrict__ q,
                                bf16_t* __restrict__ k, bf16_t* __restrict__ v,
                                const float* __restrict__ cs, const float* __restrict__ sn,
                                const int* __restrict__ pos, float qscale, int B, int T, int Hq,
                                int Hkv, int D, int rot, int QKV) {
  const int waves_per_block = blockDim.x / WAVE;
  const int slots = Hq + 2 * Hkv;
  const long idx = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  if (idx >= (long)B * T * slots) return;
  const int lane = threadIdx.x % WAVE;
  const int slot = idx % slots;
  const int t = (idx / slots) % T;
  const long b = idx / ((long)T * slots);

  bf16_t* packed = qkv + ((size_t)b * T + t) * QKV + (size_t)slot * D;
  bf16_t* split;
  bool rope = (cs != nullptr);
  float scale = 1.f;
  if (slot < Hq) {
    split = q + (((size_t)b * Hq + slot) * T + t) * D;
    scale = qscale;
  } else if (slot < Hq + Hkv) {
    split = k + (((size_t)b * Hkv + (slot - Hq)) * T + t) * D;
  } else {
    split = v + (((size_t)b * Hkv + (slot - Hq - Hkv)) * T + t) * D;
    rope = false;
  }
  const bf16_t* src = FWD ? packed : split;
  bf16_t* dst = FWD ? split : packed;

  if (rope) {
    const int p = pos[b * T + t];
    const float* c = cs + (size_t)p * (rot / 2);
    const float* s = sn + (size_t)p * (rot / 2);
    for (int i = lane; i < rot / 2; i += WAVE) {
      const float ci = c[i];
      const float si = FWD ? s[i] : -s[i];  // backward = inverse rotation
      const int i1 = INTERLEAVED ? 2 * i : i;
      const int i2 = INTERLEAVED ? 2 * i + 1 : i + rot / 2;
      const float x1 = bf2f(src[i1].u) * scale;
      const float x2 = bf2f(src[i2].u) * scale;
      dst[i1].u = f2bf(x1 * ci - x2 * si);
      dst[i2].u = f2bf(x2 * ci + x1 * si);
    }
    for (int i = rot + lane; i < D; i += WAVE) dst[i].u = f2bf(bf2f(src[i].u) * scale);
  } else if (scale != 1.f) {
    for (int i = lane; i < D; i += WAVE) dst[i].u = f2bf(bf2f(src[i].u) * scale);
  } else {
    const int D4 = D / 4;
    for (int i = lane; i < D4; i += WAVE) {
      reinterpret_cast<short4v*>(dst)[i] = reinterpret_cast<const short4v*>(src)[i];
    }
  }
}

// Vectorized variant: one THREAD per 8-element chunk (16 B loads/stores,
// guide Guideline 13) instead of one wave per D-row with scalar accesses —
// the wave-per-row version measured 35 us at [3360 tokens, gpt2] = 0.9 TB/s
// (half the lanes idle in the RoPE loop + 2 B scalar transactions).
// Requires D % 8 == 0 and rot % 16 == 0 (each chunk stays inside one
// rotation half); otherwise the row kernel above runs.
template <bool INTERLEAVED, bool FWD>
__global__ void qkv_prep_vec_kernel(bf16_t* __restrict__ qkv, bf16_t* __restrict__ q,
                                    bf16_t* __restrict__ k, bf16_t* __restrict__ v,
                                    const float* __restrict__ cs, const float* __restrict__ sn,
                                    const int* __restrict__ pos, float qscale, int B, int T,
                                    int Hq, int Hkv, int D, int rot, int QKV) {
  const int slots = Hq + 2 * Hkv;
  const long nchunks = (long)B * T * slots * (D / 8);
  const long cidx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (cidx >= nchunks) return;
  const int off = (int)(cidx % (D / 8)) * 8;  // chunk start within the row
  const long row = cidx / (D / 8);
  const int slot = (int)(row % slots);
  const int t = (int)((row / slots) % T);
  const long b = row / ((long)T * slots);

  bf16_t* packed = qkv + ((size_t)b * T + t) * QKV + (size_t)slot * D;
  bf16_t* split;
  bool rope = (cs != nullptr);
  float scale = 1.f;
  if (slot < Hq) {
    split = q + (((size_t)b * Hq + slot) * T + t) * D;
    scale = qscale;
  } else if (slot < Hq + Hkv) {
    split = k + (((size_t)b * Hkv + (slot - Hq)) * T + t) * D;
  } else {
    split = v + (((size_t)b * Hkv + (slot - Hq - Hkv)) * T + t) * D;
    rope = false;
  }
  const bf16_t* src = FWD ? packed : split;
  bf16_t* dst = FWD ? split : packed;

  float x[8];
  load8<bf16_t>(src + off, x);
#pragma unroll
  for (int j = 0; j < 8; ++j) x[j] *= scale;
  if (rope && off < rot) {
    const int p = pos[b * T + t];
    const float* c = cs + (size_t)p * (rot / 2);
    const float* s = sn + (size_t)p * (rot / 2);
    float o[8];
    if (INTERLEAVED) {
      // pairs (2i, 2i+1) live inside the chunk
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int i = off / 2 + j;
        const float ci = c[i];
        const float si = FWD ? s[i] : -s[i];
        o[2 * j] = x[2 * j] * ci - x[2 * j + 1] * si;
        o[2 * j + 1] = x[2 * j + 1] * ci + x[2 * j] * si;
      }
    } else {
      // pair partner sits rot/2 away: second 16 B read
      const bool first_half = off < rot / 2;
      float y[8];
      load8<bf16_t>(src + (first_half ? off + rot / 2 : off - rot / 2), y);
#pragma unroll
      for (int j = 0; j < 8; ++j) y[j] *= scale;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int i = (first_half ? off : off - rot / 2) + j;
        const float ci = c[i];
        const float si = FWD ? s[i] : -s[i];
        // x1' = x1*c - x2*s ; x2' = x2*c + x1*s
        o[j] = first_half ? (x[j] * ci - y[j] * si) : (x[j] * ci + y[j] * si);
      }
    }
    store8<bf16_t>(dst + off, o);
  } else {
    store8<bf16_t>(dst + off, x);
  }
}


void launch(bool fwd, at::Tensor& qkv, at::Tensor& q, at::Tensor& k, at::Tensor& v,
            const c10::optional<at::Tensor>& cos, const c10::optional<at::Tensor>& sin,
            const c10::optional<at::Tensor>& pos, double qscale, long rot, bool interleaved) {
  const int B = qkv.size(0), T = qkv.size(1);
  const int QKV = qkv.size(2);
  const int Hq = q.size(1), Hkv = k.size(1), D = q.size(3);
  TORCH_CHECK(QKV == (Hq + 2 * Hkv) * D, "qkv_prep: width mismatch");
  const float* cs = nullptr;
  const float* sn = nullptr;
  const int* pp = nullptr;
  if (cos.has_value()) {
    cs = cos->data_ptr<float>();
    sn = sin->data_ptr<float>();
    TORCH_CHECK(pos.has_value() && pos->dtype() == at::kInt && pos->is_contiguous());
    pp = pos->data_ptr<int>();
  }
  constexpr int BLOCK = 256;
  const int wpb = BLOCK / WAVE;
  const long rows = (long)B * T * (Hq + 2 * Hkv);
  const long grid = (rows + wpb - 1) / wpb;
  auto stream = c10::hip::getCurrentHIPStream();
  auto qkvp = reinterpret_cast<bf16_t*>(qkv.data_ptr());
  auto qp = reinterpret_cast<bf16_t*>(q.data_ptr());
  auto kp = reinterpret_cast<bf16_t*>(k.data_ptr());
  auto vp = reinterpret_cast<bf16_t*>(v.data_ptr());
  if (D % 8 == 0 && (rot % 16 == 0 || cs == nullptr)) {
    const long nchunks = rows * (D / 8);
    const long vgrid = (nchunks + BLOCK - 1) / BLOCK;
#define LAUNCH_QKV_VEC(IL, F)                                                             \
 hipLaunchKernelGGL(( qkv_prep_vec_kernel<IL, F>), dim3(vgrid), dim3(BLOCK), 0, stream, qkvp, qp, kp, vp, cs, sn, pp,   \
                                                          (float)qscale, B, T, Hq, Hkv,   \
                                                          D, (int)rot, QKV)
    if (interleaved) {
      if (fwd) LAUNCH_QKV_VEC(true, true);
      else LAUNCH_QKV_VEC(true, false);
    } else {
      if (fwd) LAUNCH_QKV_VEC(false, true);
      else LAUNCH_QKV_VEC(false, false);
    }
#undef LAUNCH_QKV_VEC
    HIP_CHECK_LAST();
    return;
  }
#define LAUNCH_QKV(IL, F)                                                                 \
 hipLaunchKernelGGL(( qkv_prep_kernel<IL, F>), dim3(grid), dim3(BLOCK), 0, stream, qkvp, qp, kp, vp, cs, sn, pp,        \
                                                     (float)qscale, B, T, Hq, Hkv, D,     \
                                                     (int)rot, QKV)
  if (interleaved) {
    if (fwd) LAUNCH_QKV(true, true);
    else LAUNCH_QKV(true, false);
  } else {
    if (fwd) LAUNCH_QKV(false, true);
    else LAUNCH_QKV(false, false);
  }
#undef LAUNCH_QKV
  HIP_CHECK_LAST();
}

}  // namespace

std::vector<at::Tensor> qkv_prep_fwd(const at::Tensor& qkv, long num_heads, long num_kv_heads,
                                     long head_dim, const c10::optional<at::Tensor>& cos,
                                     const c10::optional<at::Tensor>& sin,
                                     const c10::optional<at::Tensor>& pos, double qscale,
                                     long rot, bool interleaved) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == at::kBFloat16 && qkv.dim() == 3 && qkv.is_contiguous());
  const int B = qkv.size(0), T = qkv.size(1);
  auto q = at::empty({B, num_heads, T, head_dim}, qkv.options());
  auto k = at::empty({B, num_kv_heads, T, head_dim}, qkv.options());
  auto v = at::empty({B, num_kv_heads, T, head_dim}, qkv.options());
  auto qkv_ = const_cast<at::Tensor&>(qkv);
  launch(true, qkv_, q, k, v, cos, sin, pos, qscale, rot, interleaved);
  return {q, k, v};
}

at::Tensor qkv_prep_bwd(const at::Tensor& dq, const at::Tensor& dk, const at::Tensor& dv,
                        const c10::optional<at::Tensor>& cos, const c10::optional<at::Tensor>& sin,
                        const c10::optional<at::Tensor>& pos, double qscale, long rot,
                        bool interleaved) {
  TORCH_CHECK(dq.is_cuda() && dq.dtype() == at::kBFloat16 && dq.is_contiguous());
  TORCH_CHECK(dk.is_contiguous() && dv.is_contiguous());
  const int B = dq.size(0), Hq = dq.size(1), T = dq.size(2), D = dq.size(3);
  const int Hkv = dk.size(1);
  auto dqkv = at::empty({B, T, (Hq + 2 * Hkv) * D}, dq.options());
  auto dq_ = const_cast<at::Tensor&>(dq);
  auto dk_ = const_cast<at::Tensor&>(dk);
  auto dv_ = const_cast<at::Tensor&>(dv);
  launch(false, dqkv, dq_, dk_, dv_, cos, sin, pos, qscale, rot, interleaved);
  return dqkv;
}
