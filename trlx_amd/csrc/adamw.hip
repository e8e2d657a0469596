// Fused AdamW over flat parameter arenas.
//
// The native optimizer keeps every trainable parameter as a view into one
// contiguous arena per (dtype, weight-decay) group: bf16 params, fp32 master
// weights, fp32 moments, and a flat grad buffer that autograd accumulates
// into directly.  The whole optimizer step is then ONE grid-stride kernel per
// arena (vs. one launch per tensor), the gradient 1/N DP scaling is fused in
// (grad_div_ar_fusion, megatron_20b.yaml:72), and bf16 params are re-cast
// from the fp32 masters in the same pass (fp32-master mixed precision, K13).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

// PT: parameter storage type (bf16_t with separate fp32 master, or float
// where master IS the parameter).  G: gradient type.
template <typename PT, typename G>
__global__ void adamw_kernel(PT* __restrict__ p, float* __restrict__ master,
                             const G* __restrict__ g, float* __restrict__ m,
                             float* __restrict__ v, long n, float lr, float b1, float b2,
                             float eps, float wd, float bc1, float bc2, float gscale) {
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4; base < n; base += stride) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long i = base + j;
      if (i >= n) break;
      const float gi = ScalarIO<G>::load(g + i) * gscale;
      const float mi = b1 * m[i] + (1.f - b1) * gi;
      const float vi = b2 * v[i] + (1.f - b2) * gi * gi;
      m[i] = mi;
      v[i] = vi;
      float w = master[i];
      w -= lr * wd * w;
      w -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
      master[i] = w;
      ScalarIO<PT>::store(p + i, w);  // aliased (fp32, non-ZeRO) or sharded
    }
  }
}

}  // namespace

void fused_adamw(at::Tensor& p, at::Tensor& master, const at::Tensor& g, at::Tensor& m,
                 at::Tensor& v, long step, double lr, double beta1, double beta2, double eps,
                 double weight_decay, double grad_scale) {
  TORCH_CHECK(p.is_cuda() && p.is_contiguous() && master.is_contiguous() && g.is_contiguous());
  TORCH_CHECK(master.dtype() == at::kFloat && m.dtype() == at::kFloat && v.dtype() == at::kFloat);
  const long n = p.numel();
  TORCH_CHECK(master.numel() == n && g.numel() == n && m.numel() == n && v.numel() == n);
  if (n == 0) return;
  const float bc1 = 1.f - powf((float)beta1, (float)step);
  const float bc2 = 1.f - powf((float)beta2, (float)step);
  constexpr int BLOCK = 256;
  const int grid = (int)std::min<long>((n + BLOCK * 4 - 1) / (BLOCK * 4), 4096);
  auto stream = c10::hip::getCurrentHIPStream();

#define LAUNCH_ADAMW(PT, GT, PP, GP)                                                        \
  adamw_kernel<PT, GT><<<grid, BLOCK, 0, stream>>>(                                         \
      PP, master.data_ptr<float>(), GP, m.data_ptr<float>(), v.data_ptr<float>(), n,        \
      (float)lr, (float)beta1, (float)beta2, (float)eps, (float)weight_decay, bc1, bc2,     \
      (float)grad_scale)

  if (p.dtype() == at::kBFloat16) {
    auto pp = reinterpret_cast<bf16_t*>(p.data_ptr());
    if (g.dtype() == at::kBFloat16) {
      LAUNCH_ADAMW(bf16_t, bf16_t, pp, reinterpret_cast<const bf16_t*>(g.data_ptr()));
    } else {
      LAUNCH_ADAMW(bf16_t, float, pp, g.data_ptr<float>());
    }
  } else if (p.dtype() == at::kFloat) {
    auto pp = p.data_ptr<float>();
    if (g.dtype() == at::kBFloat16) {
      LAUNCH_ADAMW(float, bf16_t, pp, reinterpret_cast<const bf16_t*>(g.data_ptr()));
    } else {
      LAUNCH_ADAMW(float, float, pp, g.data_ptr<float>());
    }
  } else {
    TORCH_CHECK(false, "fused_adamw: unsupported param dtype");
  }
#undef LAUNCH_ADAMW
  HIP_CHECK_LAST();
}
