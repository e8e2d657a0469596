#include "hip/hip_runtime.h"
// Rotary positional embedding.
//
// Host-precomputed cos/sin tables (guide Appendix B: on-device trig turns a
// memory-bound op VALU-bound), per-(batch,token) position indices so
// left-padded prompts get correct rotations.  Backward = same rotation with
// the angle negated (INVERSE template arg).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

template <typename T, bool INTERLEAVED, bool INVERSE>
__global__ void rope_kernel(const T* __restrict__ x, T* __restrict__ y,
                            const float* __restrict__ cs, const float* __restrict__ sn,
                            const int* __restrict__ pos, long rows, int Hh, int T_, int D,
                            int rot) {
  const int waves_per_block = blockDim.x / WAVE;
  long row = (long)blockIdx.x * waves_per_block + threadIdx.x / WAVE;
  if (row >= rows) return;
  const int lane = threadIdx.x % WAVE;
  const int t = row % T_;
  const long b = row / ((long)Hh * T_);
  const int p = pos[b * T_ + t];
  const float* c = cs + (size_t)p * (rot / 2);
  const float* s = sn + (size_t)p * (rot / 2);
  const T* xr = x + (size_t)row * D;
  T* yr = y + (size_t)row * D;
  for (int i = lane; i < rot / 2; i += WAVE) {
    const float ci = c[i];
    const float si = INVERSE ? -s[i] : s[i];
    const int i1 = INTERLEAVED ? 2 * i : i;
    const int i2 = INTERLEAVED ? 2 * i + 1 : i + rot / 2;
    const float x1 = ScalarIO<T>::load(xr + i1);
    const float x2 = ScalarIO<T>::load(xr + i2);
    ScalarIO<T>::store(yr + i1, x1 * ci - x2 * si);
    ScalarIO<T>::store(yr + i2, x2 * ci + x1 * si);
  }
  // pass-through tail (partial-rotary archs: GPT-J/NeoX rotary_pct < 1)
  for (int i = rot + lane; i < D; i += WAVE) {
    ScalarIO<T>::store(yr + i, ScalarIO<T>::load(xr + i));
  }
}

}  // namespace

at::Tensor rope_fwd(const at::Tensor& x, const at::Tensor& cos, const at::Tensor& sin,
                    const at::Tensor& pos, bool interleaved, bool inverse, long rot) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous(), "rope: x must be [B,H,T,D] contiguous");
  TORCH_CHECK(cos.dtype() == at::kFloat && sin.dtype() == at::kFloat && cos.is_contiguous() && sin.is_contiguous());
  TORCH_CHECK(pos.dtype() == at::kInt && pos.is_contiguous());
  const long B = x.size(0), Hh = x.size(1), T_ = x.size(2), D = x.size(3);
  if (rot <= 0) rot = D;
  TORCH_CHECK(rot % 2 == 0 && rot <= D);
  TORCH_CHECK(cos.size(-1) == rot / 2 && sin.size(-1) == rot / 2, "rope: table width must be rot/2");
  auto y = at::empty_like(x);
  const long rows = B * Hh * T_;
  if (rows == 0) return y;
  constexpr int BLOCK = 256;
  const int wpb = BLOCK / WAVE;
  const long grid = (rows + wpb - 1) / wpb;
  auto stream = c10::hip::getCurrentHIPStream();

#define LAUNCH_ROPE(T, IL, INV, XP, YP)                                                   \
 hipLaunchKernelGGL(( rope_kernel<T, IL, INV>), dim3(grid), dim3(BLOCK), 0, stream, XP, YP, cos.data_ptr<float>(),      \
                                                      sin.data_ptr<float>(),              \
                                                      pos.data_ptr<int>(), rows, (int)Hh, \
                                                      (int)T_, (int)D, (int)rot)
  if (x.dtype() == at::kBFloat16) {
    auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
    auto yp = reinterpret_cast<bf16_t*>(y.data_ptr());
    if (interleaved) {
      if (inverse) LAUNCH_ROPE(bf16_t, true, true, xp, yp);
      else LAUNCH_ROPE(bf16_t, true, false, xp, yp);
    } else {
      if (inverse) LAUNCH_ROPE(bf16_t, false, true, xp, yp);
      else LAUNCH_ROPE(bf16_t, false, false, xp, yp);
    }
  } else if (x.dtype() == at::kFloat) {
    auto xp = x.data_ptr<float>();
    auto yp = y.data_ptr<float>();
    if (interleaved) {
      if (inverse) LAUNCH_ROPE(float, true, true, xp, yp);
      else LAUNCH_ROPE(float, true, false, xp, yp);
    } else {
      if (inverse) LAUNCH_ROPE(float, false, true, xp, yp);
      else LAUNCH_ROPE(float, false, false, xp, yp);
    }
  } else {
    TORCH_CHECK(false, "rope: unsupported dtype");
  }
#undef LAUNCH_ROPE
  HIP_CHECK_LAST();
  return y;
}
