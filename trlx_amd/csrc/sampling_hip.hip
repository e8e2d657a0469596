#include "hip/hip_runtime.h"
// Fused categorical sampling via Gumbel-max.
//
// One pass over [B, V] logits: per-element counter-based RNG -> Gumbel noise,
// argmax reduce.  Samples exactly from softmax(logits / T) with NO softmax /
// cumulative-sum materialization — the hot PPO rollout path (do_sample,
// top_k=0, top_p=1) is a single memory-bound kernel.  An optional per-row
// threshold (k-th largest, computed by a library top-k when top_k>0) masks
// the tail in the same pass.  RNG is keyed (seed, step-offset, row, col) so
// per-DP-rank rollout streams are decorrelated and reproducible (reference
// forks RNG per DP rank — modeling_nemo_ppo.py:384-393).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;

struct ArgMax {
  float v;
  int i;
};

DEV ArgMax amax_combine(ArgMax a, ArgMax b) {
  // deterministic tie-break on lower index
  if (b.v > a.v || (b.v == a.v && b.i < a.i)) return b;
  return a;
}

__global__ void gumbel_sample_kernel(const float* __restrict__ logits,
                                     const float* __restrict__ thresholds, long* __restrict__ out,
                                     int V, float invTemp, unsigned long long key) {
  const long row = blockIdx.x;
  const float* x = logits + (size_t)row * V;
  const float thr = thresholds ? thresholds[row] : -INFINITY;
  ArgMax best{-INFINITY, 0};
  for (int i = threadIdx.x; i < V; i += BLOCK) {
    const float xi = x[i];
    if (xi < thr || xi == -INFINITY) continue;
    const float u = rng_uniform(key, (unsigned long long)row, (unsigned long long)i);
    const float g = -logf(-logf(u));
    best = amax_combine(best, ArgMax{xi * invTemp + g, i});
  }
  // wave argmax
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    ArgMax o;
    o.v = __shfl_xor(best.v, off);
    o.i = __shfl_xor(best.i, off);
    best = amax_combine(best, o);
  }
  __shared__ ArgMax wbuf[NWAVES];
  const int wid = threadIdx.x / WAVE;
  if (threadIdx.x % WAVE == 0) wbuf[wid] = best;
  __syncthreads();
  if (threadIdx.x == 0) {
    ArgMax total = wbuf[0];
#pragma unroll
    for (int i = 1; i < NWAVES; ++i) total = amax_combine(total, wbuf[i]);
    out[row] = total.i;
  }
}

}  // namespace

at::Tensor gumbel_sample(const at::Tensor& logits, double temperature,
                         const c10::optional<at::Tensor>& thresholds, long seed, long offset) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.dtype() == at::kFloat &&
              logits.is_contiguous());
  const long B = logits.size(0);
  const int V = logits.size(1);
  auto out = at::empty({B}, logits.options().dtype(at::kLong));
  if (B == 0) return out;
  const float* thr = nullptr;
  at::Tensor thrc;
  if (thresholds.has_value()) {
    thrc = thresholds->contiguous();
    TORCH_CHECK(thrc.numel() == B && thrc.dtype() == at::kFloat);
    thr = thrc.data_ptr<float>();
  }
  const unsigned long long key =
      splitmix64_host((unsigned long long)seed ^ (0x9e3779b97f4a7c15ull * (unsigned long long)(offset + 1)));
  auto stream = c10::hip::getCurrentHIPStream();
 hipLaunchKernelGGL(( gumbel_sample_kernel), dim3(B), dim3(BLOCK), 0, stream, logits.data_ptr<float>(), thr,
                                                out.data_ptr<long>(), V,
                                                1.0f / (float)std::max(temperature, 1e-6), key);
  HIP_CHECK_LAST();
  return out;
}
