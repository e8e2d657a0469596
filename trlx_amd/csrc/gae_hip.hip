#include "hip/hip_runtime.h"
// GAE reverse scan + whitening statistics (SURVEY.md K6, K11's whiten).
//
// The reference computes GAE with a Python reverse loop over T
// (trlx/models/modeling_ppo.py:136-173).  Here the first-order linear
// recurrence A_t = delta_t + (gamma*lam) * A_{t+1} is evaluated as a
// wave-parallel scan of affine maps: each lane serially composes its chunk of
// reversed time into (A, B) with u_out = A*u_in + B, a log2(64)-step shfl_up
// scan composes lanes, then each lane replays its chunk with the correct
// incoming accumulator.  One wave per batch row.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

__global__ void gae_kernel(const float* __restrict__ values, const float* __restrict__ rewards,
                           float* __restrict__ adv, float* __restrict__ ret, int B, int T,
                           float gamma, float lam) {
  const int waves_per_block = blockDim.x / WAVE;
  const int row = blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  if (row >= B) return;
  const int lane = threadIdx.x % WAVE;
  const float c = gamma * lam;
  const int L = (T + WAVE - 1) / WAVE;
  const float* vr = values + (size_t)row * T;
  const float* rr = rewards + (size_t)row * T;

  // pass 1: compose this lane's chunk (reversed-time indices [lane*L, ...))
  float A = 1.f, Bc = 0.f;
  for (int j = 0; j < L; ++j) {
    const int i = lane * L + j;
    if (i < T) {
      const int t = T - 1 - i;
      const float nextv = (t < T - 1) ? vr[t + 1] : 0.f;
      const float delta = rr[t] + gamma * nextv - vr[t];
      A = c * A;
      Bc = delta + c * Bc;
    }
  }
  // inclusive scan over lanes (composition in lane order: lane l after lanes <l)
  float sA = A, sB = Bc;
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    const float pA = __shfl_up(sA, off);
    const float pB = __shfl_up(sB, off);
    if (lane >= off) {
      sB = sA * pB + sB;
      sA = sA * pA;
    }
  }
  // exclusive: incoming accumulator for this lane's chunk
  const float prevB = __shfl_up(sB, 1);
  float u = (lane == 0) ? 0.f : prevB;

  // pass 2: replay the chunk emitting advantages/returns
  for (int j = 0; j < L; ++j) {
    const int i = lane * L + j;
    if (i < T) {
      const int t = T - 1 - i;
      const float nextv = (t < T - 1) ? vr[t + 1] : 0.f;
      const float delta = rr[t] + gamma * nextv - vr[t];
      u = delta + c * u;
      adv[(size_t)row * T + t] = u;
      ret[(size_t)row * T + t] = u + vr[t];
    }
  }
}

// sum + sum-of-squares for whitening; the exact count is set host-side.
__global__ void sum_sq_kernel(const float* __restrict__ x, float* __restrict__ stats, long n) {
  __shared__ float rbuf[4];
  float s = 0.f, ss = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    const float v = x[i];
    s += v;
    ss += v * v;
  }
  s = block_sum<4>(s, rbuf);
  ss = block_sum<4>(ss, rbuf);
  if (threadIdx.x == 0) {
    atomicAdd(&stats[0], s);
    atomicAdd(&stats[1], ss);
  }
}

__global__ void normalize_kernel(const float* __restrict__ x, float* __restrict__ y,
                                 const float* __restrict__ mean, const float* __restrict__ var,
                                 bool shift_mean, long n) {
  const float mu = *mean;
  const float r = rsqrtf(*var + 1e-8f);
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n; i += (long)gridDim.x * blockDim.x) {
    float v = (x[i] - mu) * r;
    if (!shift_mean) v += mu;
    y[i] = v;
  }
}

}  // namespace

std::vector<at::Tensor> gae(const at::Tensor& values, const at::Tensor& rewards, double gamma,
                            double lam) {
  TORCH_CHECK(values.is_cuda() && values.dim() == 2 && values.is_contiguous());
  TORCH_CHECK(rewards.sizes() == values.sizes() && rewards.is_contiguous());
  TORCH_CHECK(values.dtype() == at::kFloat && rewards.dtype() == at::kFloat);
  const int B = values.size(0), T = values.size(1);
  auto adv = at::empty_like(values);
  auto ret = at::empty_like(values);
  if (B == 0 || T == 0) return {adv, ret};
  constexpr int BLOCK = 256;
  const int wpb = BLOCK / WAVE;
  const int grid = (B + wpb - 1) / wpb;
  auto stream = c10::hip::getCurrentHIPStream();
 hipLaunchKernelGGL(( gae_kernel), dim3(grid), dim3(BLOCK), 0, stream, values.data_ptr<float>(), rewards.data_ptr<float>(),
                                         adv.data_ptr<float>(), ret.data_ptr<float>(), B, T,
                                         (float)gamma, (float)lam);
  HIP_CHECK_LAST();
  return {adv, ret};
}

at::Tensor sum_count(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kFloat && x.is_contiguous());
  const long n = x.numel();
  auto stats = at::zeros({3}, x.options());
  stats[2] = (float)n;
  if (n == 0) return stats;
  constexpr int BLOCK = 256;
  const int grid = (int)std::min<long>((n + BLOCK - 1) / BLOCK, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
 hipLaunchKernelGGL(( sum_sq_kernel), dim3(grid), dim3(BLOCK), 0, stream, x.data_ptr<float>(), stats.data_ptr<float>(), n);
  HIP_CHECK_LAST();
  return stats;
}

at::Tensor normalize(const at::Tensor& x, const at::Tensor& mean, const at::Tensor& var,
                     bool shift_mean) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == at::kFloat && x.is_contiguous());
  auto y = at::empty_like(x);
  const long n = x.numel();
  if (n == 0) return y;
  constexpr int BLOCK = 256;
  const int grid = (int)std::min<long>((n + BLOCK - 1) / BLOCK, 2048);
  auto stream = c10::hip::getCurrentHIPStream();
  auto mc = mean.contiguous();
  auto vc = var.contiguous();
 hipLaunchKernelGGL(( normalize_kernel), dim3(grid), dim3(BLOCK), 0, stream, x.data_ptr<float>(), y.data_ptr<float>(),
                                               mc.data_ptr<float>(), vc.data_ptr<float>(),
                                               shift_mean, n);
  HIP_CHECK_LAST();
  return y;
}
