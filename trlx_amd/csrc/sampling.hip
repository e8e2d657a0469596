// Fused categorical sampling via Gumbel-max.
//
// Samples exactly from softmax(logits / T) in one pass over [B, V]: per
// element a counter-based uniform -> Gumbel key, argmax-reduced.  The grid
// splits each row's vocab across V-chunks (B=128 rows alone would leave the
// 256-CU chip half idle — profile r01) and combines chunk winners with a
// packed (orderable-float | index) u64 atomicMax; a tiny second kernel
// unpacks the winners.  RNG is keyed (seed, *offset, row, col); the offset
// lives in DEVICE memory so hipGraph replays draw fresh noise, and per-DP-rank
// seeds decorrelate rollouts (reference modeling_nemo_ppo.py:384-393).
// An optional per-row threshold (k-th largest from a library top-k) masks
// the tail for top_k sampling in the same pass.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;
constexpr int CHUNK = 8192;  // vocab elements per block

// map float to an order-preserving u32 (works for -inf / inf too)
DEV unsigned int float_orderable(float x) {
  unsigned int u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

template <typename T>
__global__ void gumbel_sample_kernel(const T* __restrict__ logits,
                                     const float* __restrict__ thresholds,
                                     unsigned long long* __restrict__ packed, int V, float invTemp,
                                     unsigned long long seed, const long* __restrict__ offset_ptr,
                                     long host_offset) {
  // grid: (chunks, rows)
  const long row = blockIdx.y;
  const int lo = blockIdx.x * CHUNK;
  const int hi = min(lo + CHUNK, V);
  const T* x = logits + (size_t)row * V;
  const float thr = thresholds ? thresholds[row] : -INFINITY;
  const unsigned long long off = offset_ptr ? (unsigned long long)(*offset_ptr) : (unsigned long long)host_offset;
  const unsigned long long key = splitmix64(seed ^ (0x9e3779b97f4a7c15ull * (off + 1)));

  float best = -INFINITY;
  int best_i = lo;
  for (int i = lo + threadIdx.x; i < hi; i += BLOCK) {
    const float xi = ScalarIO<T>::load(x + i);
    if (xi < thr || xi == -INFINITY) continue;
    const float u = rng_uniform(key, (unsigned long long)row, (unsigned long long)i);
    const float g = -__logf(-__logf(u));
    const float v = xi * invTemp + g;
    if (v > best) {
      best = v;
      best_i = i;
    }
  }
#pragma unroll
  for (int off2 = 32; off2 > 0; off2 >>= 1) {
    const float ov = __shfl_xor(best, off2);
    const int oi = __shfl_xor(best_i, off2);
    if (ov > best || (ov == best && oi < best_i)) {
      best = ov;
      best_i = oi;
    }
  }
  __shared__ float wv[NWAVES];
  __shared__ int wi[NWAVES];
  const int wid = threadIdx.x / WAVE;
  if (threadIdx.x % WAVE == 0) {
    wv[wid] = best;
    wi[wid] = best_i;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
#pragma unroll
    for (int w = 1; w < NWAVES; ++w) {
      if (wv[w] > best || (wv[w] == best && wi[w] < best_i)) {
        best = wv[w];
        best_i = wi[w];
      }
    }
    // pack: high 32 bits = orderable float, low 32 = ~index (ties -> lower index)
    const unsigned long long p =
        ((unsigned long long)float_orderable(best) << 32) | (unsigned int)(~(unsigned int)best_i);
    atomicMax(&packed[row], p);
  }
}

__global__ void unpack_kernel(const unsigned long long* __restrict__ packed, long* __restrict__ out,
                              int B) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b < B) out[b] = (long)(~(unsigned int)(packed[b] & 0xffffffffu));
}

}  // namespace

at::Tensor gumbel_sample_impl(const at::Tensor& logits, double temperature,
                              const c10::optional<at::Tensor>& thresholds, long seed,
                              const long* offset_ptr, long host_offset, hipStream_t stream) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous() &&
              (logits.dtype() == at::kFloat || logits.dtype() == at::kBFloat16));
  const long B = logits.size(0);
  const int V = logits.size(1);
  auto out = at::empty({B}, logits.options().dtype(at::kLong));
  if (B == 0) return out;
  auto packed = at::zeros({B}, logits.options().dtype(at::kLong));
  const float* thr = nullptr;
  at::Tensor thrc;
  if (thresholds.has_value()) {
    thrc = thresholds->contiguous();
    TORCH_CHECK(thrc.numel() == B && thrc.dtype() == at::kFloat);
    thr = thrc.data_ptr<float>();
  }
  const int nchunks = (V + CHUNK - 1) / CHUNK;
  dim3 grid(nchunks, B);
  if (logits.dtype() == at::kBFloat16)
    gumbel_sample_kernel<bf16_t><<<grid, BLOCK, 0, stream>>>(
        reinterpret_cast<const bf16_t*>(logits.data_ptr()), thr,
        reinterpret_cast<unsigned long long*>(packed.data_ptr<long>()), V,
        1.0f / (float)std::max(temperature, 1e-6), (unsigned long long)seed, offset_ptr,
        host_offset);
  else
    gumbel_sample_kernel<float><<<grid, BLOCK, 0, stream>>>(
        logits.data_ptr<float>(), thr,
        reinterpret_cast<unsigned long long*>(packed.data_ptr<long>()), V,
        1.0f / (float)std::max(temperature, 1e-6), (unsigned long long)seed, offset_ptr,
        host_offset);
  const int ub = (int)((B + 255) / 256);
  unpack_kernel<<<ub, 256, 0, stream>>>(
      reinterpret_cast<const unsigned long long*>(packed.data_ptr<long>()), out.data_ptr<long>(),
      (int)B);
  HIP_CHECK_LAST();
  return out;
}

at::Tensor gumbel_sample(const at::Tensor& logits, double temperature,
                         const c10::optional<at::Tensor>& thresholds, long seed, long offset) {
  auto stream = c10::hip::getCurrentHIPStream();
  return gumbel_sample_impl(logits, temperature, thresholds, seed, nullptr, offset, stream);
}

at::Tensor gumbel_sample_dev(const at::Tensor& logits, double temperature,
                             const c10::optional<at::Tensor>& thresholds, long seed,
                             const at::Tensor& offset) {
  TORCH_CHECK(offset.is_cuda() && offset.dtype() == at::kLong);
  auto stream = c10::hip::getCurrentHIPStream();
  return gumbel_sample_impl(logits, temperature, thresholds, seed, offset.data_ptr<long>(), 0,
                            stream);
}
