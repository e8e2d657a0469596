#include "hip/hip_runtime.h"
// Fused causal-masked softmax for training attention (SURVEY.md K2's
// softmax/mask half; score and PV GEMMs ride rocBLAS batched MFMA GEMMs).
//
// Replaces Apex's scaled_upper_triang_masked_softmax: the causal mask is
// never materialized — each row's valid prefix length is computed from the
// row index, the online max/sum runs in fp32 over vectorized bf16 loads, and
// masked positions are written as exact zeros (so the PV GEMM sees zeros).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;

template <typename T>
__global__ void causal_softmax_fwd_kernel(const T* __restrict__ scores, T* __restrict__ probs,
                                          const int* __restrict__ key_starts, int HTq, int Tq,
                                          int Tk, int start_pos, long rows) {
  __shared__ MS wbuf[NWAVES];
  __shared__ float bshare[2];
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const int tq = row % Tq;
    const int kstart = key_starts ? key_starts[row / HTq] : 0;
    const int valid = min(Tk, start_pos + tq + 1);
    const T* xr = scores + (size_t)row * Tk;
    T* pr = probs + (size_t)row * Tk;
    MS ms{-INFINITY, 0.f};
    const int T8 = Tk & ~7;
    for (int base = threadIdx.x * 8; base < T8; base += BLOCK * 8) {
      if (base + 8 <= kstart || base >= valid) continue;
      float v[8];
      load8<T>(xr + base, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int j = base + i;
        if (j < kstart || j >= valid) continue;
        const float xi = v[i];
        if (xi > ms.m) {
          ms.s = ms.s * expf(ms.m - xi) + 1.f;
          ms.m = xi;
        } else {
          ms.s += expf(xi - ms.m);
        }
      }
    }
    for (int i = T8 + threadIdx.x; i < valid; i += BLOCK) {
      if (i < kstart) continue;
      const float xi = ScalarIO<T>::load(xr + i);
      if (xi > ms.m) {
        ms.s = ms.s * expf(ms.m - xi) + 1.f;
        ms.m = xi;
      } else {
        ms.s += expf(xi - ms.m);
      }
    }
    ms = wave_ms(ms);
    const int wid = threadIdx.x / WAVE;
    if (threadIdx.x % WAVE == 0) wbuf[wid] = ms;
    __syncthreads();
    if (threadIdx.x == 0) {
      MS total = wbuf[0];
#pragma unroll
      for (int i = 1; i < NWAVES; ++i) total = ms_combine(total, wbuf[i]);
      bshare[0] = total.m;
      bshare[1] = (total.s > 0.f) ? 1.0f / total.s : 0.f;
    }
    __syncthreads();
    const float m = bshare[0];
    const float rs = bshare[1];
    for (int base = threadIdx.x * 8; base < T8; base += BLOCK * 8) {
      float v[8];
      load8<T>(xr + base, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int j = base + i;
        v[i] = (j >= kstart && j < valid) ? expf(v[i] - m) * rs : 0.f;
      }
      store8<T>(pr + base, v);
    }
    for (int i = T8 + threadIdx.x; i < Tk; i += BLOCK) {
      float p = 0.f;
      if (i >= kstart && i < valid) p = expf(ScalarIO<T>::load(xr + i) - m) * rs;
      ScalarIO<T>::store(pr + i, p);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void causal_softmax_bwd_kernel(const T* __restrict__ probs, const T* __restrict__ dprobs,
                                          T* __restrict__ dscores, int Tk, long rows) {
  __shared__ float rbuf[NWAVES];
  __shared__ float bshare;
  for (long row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* pr = probs + (size_t)row * Tk;
    const T* dpr = dprobs + (size_t)row * Tk;
    T* dsr = dscores + (size_t)row * Tk;
    float dot = 0.f;
    const int T8 = Tk & ~7;
    for (int base = threadIdx.x * 8; base < T8; base += BLOCK * 8) {
      float p[8], dp[8];
      load8<T>(pr + base, p);
      load8<T>(dpr + base, dp);
#pragma unroll
      for (int i = 0; i < 8; ++i) dot += p[i] * dp[i];
    }
    for (int i = T8 + threadIdx.x; i < Tk; i += BLOCK) {
      dot += ScalarIO<T>::load(pr + i) * ScalarIO<T>::load(dpr + i);
    }
    dot = block_sum<NWAVES>(dot, rbuf);
    if (threadIdx.x == 0) bshare = dot;
    __syncthreads();
    const float d = bshare;
    for (int base = threadIdx.x * 8; base < T8; base += BLOCK * 8) {
      float p[8], dp[8];
      load8<T>(pr + base, p);
      load8<T>(dpr + base, dp);
#pragma unroll
      for (int i = 0; i < 8; ++i) p[i] = p[i] * (dp[i] - d);
      store8<T>(dsr + base, p);
    }
    for (int i = T8 + threadIdx.x; i < Tk; i += BLOCK) {
      const float p = ScalarIO<T>::load(pr + i);
      ScalarIO<T>::store(dsr + i, p * (ScalarIO<T>::load(dpr + i) - d));
    }
    __syncthreads();
  }
}

// Wave-per-row variants for short rows (Tk <= 1024): one 64-lane wave owns a
// row — no __syncthreads, 4 rows in flight per block, scalar loads (rows this
// short are latency- not bandwidth-bound).
template <typename T>
__global__ void causal_softmax_fwd_wave(const T* __restrict__ scores, T* __restrict__ probs,
                                        const int* __restrict__ key_starts, int HTq, int Tq,
                                        int Tk, int start_pos, long rows) {
  // single pass over the row: each lane caches its <=16 strided elements in
  // registers (Tk <= 1024 guaranteed by the caller), so the normalize step
  // re-reads nothing; fully unrolled so xv[] stays in VGPRs.  The previous
  // two-pass scalar version measured 24.3 us (0.7 TB/s) at the bench shape.
  const int wpb = blockDim.x / WAVE;
  const long row0 = (long)blockIdx.x * wpb + threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long stride = (long)gridDim.x * wpb;
  for (long row = row0; row < rows; row += stride) {
    const int tq = row % Tq;
    const int kstart = key_starts ? key_starts[row / HTq] : 0;
    const int valid = min(Tk, start_pos + tq + 1);
    const T* xr = scores + (size_t)row * Tk;
    T* pr = probs + (size_t)row * Tk;
    float xv[16];
    MS ms{-INFINITY, 0.f};
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int i = lane + j * WAVE;
      if (i < Tk) {
        const float xi = ScalarIO<T>::load(xr + i);
        xv[j] = xi;
        if (i >= kstart && i < valid) {
          if (xi > ms.m) {
            ms.s = (ms.m > -INFINITY ? ms.s * __expf(ms.m - xi) : 0.f) + 1.f;
            ms.m = xi;
          } else {
            ms.s += __expf(xi - ms.m);
          }
        }
      }
    }
    ms = wave_ms(ms);
    const float m = ms.m;
    const float rs = (ms.s > 0.f) ? 1.0f / ms.s : 0.f;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      const int i = lane + j * WAVE;
      if (i < Tk) {
        float pv = 0.f;
        if (i >= kstart && i < valid) pv = __expf(xv[j] - m) * rs;
        ScalarIO<T>::store(pr + i, pv);
      }
    }
  }
}

template <typename T>
__global__ void causal_softmax_bwd_wave(const T* __restrict__ probs, const T* __restrict__ dprobs,
                                        T* __restrict__ dscores, int Tk, long rows) {
  const int wpb = blockDim.x / WAVE;
  const long row0 = (long)blockIdx.x * wpb + threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long stride = (long)gridDim.x * wpb;
  for (long row = row0; row < rows; row += stride) {
    const T* pr = probs + (size_t)row * Tk;
    const T* dpr = dprobs + (size_t)row * Tk;
    T* dsr = dscores + (size_t)row * Tk;
    float dot = 0.f;
    for (int i = lane; i < Tk; i += WAVE) {
      dot += ScalarIO<T>::load(pr + i) * ScalarIO<T>::load(dpr + i);
    }
    dot = wave_sum(dot);
    for (int i = lane; i < Tk; i += WAVE) {
      const float p = ScalarIO<T>::load(pr + i);
      ScalarIO<T>::store(dsr + i, p * (ScalarIO<T>::load(dpr + i) - dot));
    }
  }
}

}  // namespace

at::Tensor causal_softmax_fwd(const at::Tensor& scores, long start_pos,
                              const c10::optional<at::Tensor>& key_starts) {
  TORCH_CHECK(scores.is_cuda() && scores.dim() == 4 && scores.is_contiguous());
  const long B = scores.size(0), H = scores.size(1), Tq = scores.size(2), Tk = scores.size(3);
  auto probs = at::empty_like(scores);
  const long rows = B * H * Tq;
  if (rows == 0) return probs;
  const int grid = (int)std::min<long>(rows, 8192);
  auto stream = c10::hip::getCurrentHIPStream();
  const int* ks = nullptr;
  at::Tensor ksc;
  if (key_starts.has_value()) {
    ksc = key_starts->contiguous();
    TORCH_CHECK(ksc.numel() == B && ksc.dtype() == at::kInt);
    ks = ksc.data_ptr<int>();
  }
  const int HTq = (int)(H * Tq);
  const bool short_rows = Tk <= 1024;
  const int wave_grid = (int)std::min<long>((rows + 3) / 4, 16384);
  if (scores.dtype() == at::kBFloat16) {
    auto sp = reinterpret_cast<const bf16_t*>(scores.data_ptr());
    auto pp = reinterpret_cast<bf16_t*>(probs.data_ptr());
    if (short_rows)
     hipLaunchKernelGGL(( causal_softmax_fwd_wave<bf16_t>), dim3(wave_grid), dim3(BLOCK), 0, stream, sp, pp, ks, HTq, (int)Tq,
                                                                       (int)Tk, (int)start_pos, rows);
    else
     hipLaunchKernelGGL(( causal_softmax_fwd_kernel<bf16_t>), dim3(grid), dim3(BLOCK), 0, stream, sp, pp, ks, HTq, (int)Tq,
                                                                    (int)Tk, (int)start_pos, rows);
  } else if (scores.dtype() == at::kFloat) {
    if (short_rows)
     hipLaunchKernelGGL(( causal_softmax_fwd_wave<float>), dim3(wave_grid), dim3(BLOCK), 0, stream, 
          scores.data_ptr<float>(), probs.data_ptr<float>(), ks, HTq, (int)Tq, (int)Tk,
          (int)start_pos, rows);
    else
     hipLaunchKernelGGL(( causal_softmax_fwd_kernel<float>), dim3(grid), dim3(BLOCK), 0, stream, 
          scores.data_ptr<float>(), probs.data_ptr<float>(), ks, HTq, (int)Tq, (int)Tk,
          (int)start_pos, rows);
  } else {
    TORCH_CHECK(false, "causal_softmax: unsupported dtype");
  }
  HIP_CHECK_LAST();
  return probs;
}

at::Tensor causal_softmax_bwd(const at::Tensor& probs, const at::Tensor& dprobs) {
  const long B = probs.size(0), H = probs.size(1), Tq = probs.size(2), Tk = probs.size(3);
  auto dscores = at::empty_like(probs);
  const long rows = B * H * Tq;
  if (rows == 0) return dscores;
  const int grid = (int)std::min<long>(rows, 8192);
  auto stream = c10::hip::getCurrentHIPStream();
  const bool short_rows = Tk <= 1024;
  const int wave_grid = (int)std::min<long>((rows + 3) / 4, 16384);
  if (probs.dtype() == at::kBFloat16) {
    auto pp = reinterpret_cast<const bf16_t*>(probs.data_ptr());
    auto dpp = reinterpret_cast<const bf16_t*>(dprobs.data_ptr());
    auto dsp = reinterpret_cast<bf16_t*>(dscores.data_ptr());
    if (short_rows)
     hipLaunchKernelGGL(( causal_softmax_bwd_wave<bf16_t>), dim3(wave_grid), dim3(BLOCK), 0, stream, pp, dpp, dsp, (int)Tk, rows);
    else
     hipLaunchKernelGGL(( causal_softmax_bwd_kernel<bf16_t>), dim3(grid), dim3(BLOCK), 0, stream, pp, dpp, dsp, (int)Tk, rows);
  } else {
    if (short_rows)
     hipLaunchKernelGGL(( causal_softmax_bwd_wave<float>), dim3(wave_grid), dim3(BLOCK), 0, stream, 
          probs.data_ptr<float>(), dprobs.data_ptr<float>(), dscores.data_ptr<float>(), (int)Tk, rows);
    else
     hipLaunchKernelGGL(( causal_softmax_bwd_kernel<float>), dim3(grid), dim3(BLOCK), 0, stream, 
          probs.data_ptr<float>(), dprobs.data_ptr<float>(), dscores.data_ptr<float>(), (int)Tk,
          rows);
  }
  HIP_CHECK_LAST();
  return dscores;
}
