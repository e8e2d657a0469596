#include "hip/hip_runtime.h"
// Fused decode-step QKV preparation.
//
// For a single-token decode step the eager path costs 7 kernels per layer:
// 3 slice-contiguous copies (q/k/v out of the fused qkv GEMM), 2 RoPE
// launches, 2 KV-cache slice writes (profile r01: 10.9% of cycle time in
// copies).  This kernel does all of it in ONE launch: split the fused
// [B, 1, (Hq+2*Hkv)*D] projection, rotate q/k at each row's position
// (pos = *cache_idx - key_starts[b]), scatter k/v into the cache at
// *cache_idx, and emit q in [B, Hq, 1, D] layout for the decode-attention
// kernel.  cache_idx is a DEVICE scalar so the whole step is hipGraph-
// capturable (replays advance the index without host involvement).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

template <bool INTERLEAVED>
__global__ void decode_prep_kernel(const bf16_t* __restrict__ qkv, bf16_t* __restrict__ q_out,
                                   bf16_t* __restrict__ kc, bf16_t* __restrict__ vc,
                                   const float* __restrict__ cs, const float* __restrict__ sn,
                                   const int* __restrict__ key_starts,
                                   const long* __restrict__ cache_idx, int B, int Hq, int Hkv,
                                   int S, int D, int rot, int QKV) {
  const int waves_per_block = blockDim.x / WAVE;
  const int slots = Hq + 2 * Hkv;
  const long idx = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  if (idx >= (long)B * slots) return;
  const int lane = threadIdx.x % WAVE;
  const int b = idx / slots;
  const int slot = idx % slots;
  const long pos = *cache_idx;

  const bf16_t* src = qkv + (size_t)b * QKV + (size_t)slot * D;
  bf16_t* dst;
  bool rope = (cs != nullptr);
  if (slot < Hq) {
    dst = q_out + ((size_t)b * Hq + slot) * D;
  } else if (slot < Hq + Hkv) {
    dst = kc + (((size_t)b * Hkv + (slot - Hq)) * S + pos) * D;
  } else {
    dst = vc + (((size_t)b * Hkv + (slot - Hq - Hkv)) * S + pos) * D;
    rope = false;
  }

  if (rope) {
    const int p = (int)pos - (key_starts ? key_starts[b] : 0);
    const float* c = cs + (size_t)p * (rot / 2);
    const float* s = sn + (size_t)p * (rot / 2);
    for (int i = lane; i < rot / 2; i += WAVE) {
      const int i1 = INTERLEAVED ? 2 * i : i;
      const int i2 = INTERLEAVED ? 2 * i + 1 : i + rot / 2;
      const float x1 = bf2f(src[i1].u);
      const float x2 = bf2f(src[i2].u);
      dst[i1].u = f2bf(x1 * c[i] - x2 * s[i]);
      dst[i2].u = f2bf(x2 * c[i] + x1 * s[i]);
    }
    for (int i = rot + lane; i < D; i += WAVE) dst[i] = src[i];
  } else {
    // vectorized 4-element copy (D is a multiple of 4 for all supported archs)
    const int D4 = D / 4;
    for (int i = lane; i < D4; i += WAVE) {
      reinterpret_cast<short4v*>(dst)[i] = reinterpret_cast<const short4v*>(src)[i];
    }
  }
}

}  // namespace

at::Tensor decode_prep(const at::Tensor& qkv, at::Tensor& kcache, at::Tensor& vcache,
                       const c10::optional<at::Tensor>& cos, const c10::optional<at::Tensor>& sin,
                       const c10::optional<at::Tensor>& key_starts, const at::Tensor& cache_idx,
                       long num_heads, long rot, bool interleaved) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == at::kBFloat16 && qkv.is_contiguous());
  const int B = qkv.size(0);
  const int QKV = qkv.numel() / B;
  const int Hkv = kcache.size(1), S = kcache.size(2), D = kcache.size(3);
  const int Hq = (int)num_heads;
  TORCH_CHECK(QKV == (Hq + 2 * Hkv) * D, "decode_prep: qkv width mismatch");
  TORCH_CHECK(cache_idx.is_cuda() && cache_idx.dtype() == at::kLong);
  auto q_out = at::empty({B, Hq, 1, D}, qkv.options());
  const float* cs = nullptr;
  const float* sn = nullptr;
  if (cos.has_value()) {
    TORCH_CHECK(cos->is_contiguous() && sin->is_contiguous());
    cs = cos->data_ptr<float>();
    sn = sin->data_ptr<float>();
  }
  const int* ks = nullptr;
  at::Tensor ksc;
  if (key_starts.has_value()) {
    ksc = key_starts->contiguous();
    TORCH_CHECK(ksc.dtype() == at::kInt);
    ks = ksc.data_ptr<int>();
  }
  constexpr int BLOCK = 256;
  const int wpb = BLOCK / WAVE;
  const long rows = (long)B * (Hq + 2 * Hkv);
  const long grid = (rows + wpb - 1) / wpb;
  auto stream = c10::hip::getCurrentHIPStream();
  auto launch = [&](auto interleaved_tag) {
    constexpr bool IL = decltype(interleaved_tag)::value;
   hipLaunchKernelGGL(( decode_prep_kernel<IL>), dim3(grid), dim3(BLOCK), 0, stream, 
        reinterpret_cast<const bf16_t*>(qkv.data_ptr()),
        reinterpret_cast<bf16_t*>(q_out.data_ptr()), reinterpret_cast<bf16_t*>(kcache.data_ptr()),
        reinterpret_cast<bf16_t*>(vcache.data_ptr()), cs, sn, ks, cache_idx.data_ptr<long>(), B,
        Hq, Hkv, S, D, (int)rot, QKV);
  };
  if (interleaved)
    launch(std::true_type{});
  else
    launch(std::false_type{});
  HIP_CHECK_LAST();
  return q_out;
}
