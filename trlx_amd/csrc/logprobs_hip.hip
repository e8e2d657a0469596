#include "hip/hip_runtime.h"
// Fused per-token log-prob gather.
//
// Replaces the reference's log_softmax + gather (trlx/utils/modeling.py:213-219)
// which materializes a [N, V] logprob tensor in HBM.  Here: one block per row,
// single pass online logsumexp (vectorized 16 B/lane loads), then a one-element
// gather — the [N, V] intermediate never exists.  Backward recomputes softmax
// from the saved logsumexp.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include <torch/library.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;

template <typename T>
__global__ void logprobs_fwd_kernel(const T* __restrict__ logits, const long* __restrict__ labels,
                                    float* __restrict__ out, float* __restrict__ lse, int V) {
  const long row = blockIdx.x;
  const T* x = logits + (size_t)row * V;
  MS ms{-INFINITY, 0.f};
  const int V8 = V & ~7;
  for (int base = threadIdx.x * 8; base < V8; base += BLOCK * 8) {
    float v[8];
    load8<T>(x + base, v);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float xi = v[i];
      if (xi > ms.m) {
        ms.s = ms.s * expf(ms.m - xi) + 1.f;
        ms.m = xi;
      } else {
        ms.s += expf(xi - ms.m);
      }
    }
  }
  for (int i = V8 + threadIdx.x; i < V; i += BLOCK) {
    float xi = ScalarIO<T>::load(x + i);
    if (xi > ms.m) {
      ms.s = ms.s * expf(ms.m - xi) + 1.f;
      ms.m = xi;
    } else {
      ms.s += expf(xi - ms.m);
    }
  }
  ms = wave_ms(ms);
  __shared__ MS wbuf[NWAVES];
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (lane == 0) wbuf[wid] = ms;
  __syncthreads();
  if (threadIdx.x == 0) {
    MS total = wbuf[0];
#pragma unroll
    for (int i = 1; i < NWAVES; ++i) total = ms_combine(total, wbuf[i]);
    float l = total.m + logf(total.s);
    float xl = ScalarIO<T>::load(x + labels[row]);
    out[row] = xl - l;
    lse[row] = l;
  }
}

template <typename T>
__global__ void logprobs_bwd_kernel(const T* __restrict__ logits, const long* __restrict__ labels,
                                    const float* __restrict__ lse, const float* __restrict__ gout,
                                    T* __restrict__ gin, int V) {
  const long row = blockIdx.x;
  const T* x = logits + (size_t)row * V;
  T* gx = gin + (size_t)row * V;
  const float g = gout[row];
  const float l = lse[row];
  const long lab = labels[row];
  const int V8 = V & ~7;
  for (int base = threadIdx.x * 8; base < V8; base += BLOCK * 8) {
    float v[8];
    load8<T>(x + base, v);
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      float gi = -g * expf(v[i] - l);
      if (base + i == lab) gi += g;
      v[i] = gi;
    }
    store8<T>(gx + base, v);
  }
  for (int i = V8 + threadIdx.x; i < V; i += BLOCK) {
    float gi = -g * expf(ScalarIO<T>::load(x + i) - l);
    if (i == lab) gi += g;
    ScalarIO<T>::store(gx + i, gi);
  }
}

}  // namespace

std::vector<at::Tensor> logprobs_fwd(const at::Tensor& logits, const at::Tensor& labels) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(labels.dtype() == at::kLong);
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto out = at::empty({N}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({N}, logits.options().dtype(at::kFloat));
  if (N == 0) return {out, lse};
  auto stream = c10::hip::getCurrentHIPStream();
  if (logits.dtype() == at::kBFloat16) {
   hipLaunchKernelGGL(( logprobs_fwd_kernel<bf16_t>), dim3(N), dim3(BLOCK), 0, stream, 
        reinterpret_cast<const bf16_t*>(logits.data_ptr()), labels.data_ptr<long>(),
        out.data_ptr<float>(), lse.data_ptr<float>(), V);
  } else if (logits.dtype() == at::kFloat) {
   hipLaunchKernelGGL(( logprobs_fwd_kernel<float>), dim3(N), dim3(BLOCK), 0, stream, 
        logits.data_ptr<float>(), labels.data_ptr<long>(), out.data_ptr<float>(),
        lse.data_ptr<float>(), V);
  } else {
    TORCH_CHECK(false, "logprobs_fwd: unsupported dtype");
  }
  HIP_CHECK_LAST();
  return {out, lse};
}

at::Tensor logprobs_bwd(const at::Tensor& logits, const at::Tensor& labels, const at::Tensor& lse,
                        const at::Tensor& gout) {
  const long N = logits.size(0);
  const int V = logits.size(1);
  auto gin = at::empty_like(logits);
  if (N == 0) return gin;
  auto stream = c10::hip::getCurrentHIPStream();
  auto g = gout.to(at::kFloat).contiguous();
  if (logits.dtype() == at::kBFloat16) {
   hipLaunchKernelGGL(( logprobs_bwd_kernel<bf16_t>), dim3(N), dim3(BLOCK), 0, stream, 
        reinterpret_cast<const bf16_t*>(logits.data_ptr()), labels.data_ptr<long>(),
        lse.data_ptr<float>(), g.data_ptr<float>(), reinterpret_cast<bf16_t*>(gin.data_ptr()), V);
  } else {
   hipLaunchKernelGGL(( logprobs_bwd_kernel<float>), dim3(N), dim3(BLOCK), 0, stream, 
        logits.data_ptr<float>(), labels.data_ptr<long>(), lse.data_ptr<float>(),
        g.data_ptr<float>(), gin.data_ptr<float>(), V);
  }
  HIP_CHECK_LAST();
  return gin;
}
