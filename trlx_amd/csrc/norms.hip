// Fused RMSNorm / LayerNorm fwd+bwd.
//
// One block per row (grid-strided), vectorized 16 B/lane bf16 loads (guide
// Guideline 13), fp32 accumulation, saved row statistics for backward.
// Weight-grad accumulation goes through an LDS per-block partial (each column
// owned by exactly one thread, race-free) followed by one global atomicAdd per
// column per block — not per row.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BLOCK = 256;
constexpr int NWAVES = BLOCK / WAVE;

// HAS_RES: s = x + res is written to sout and normalized (residual-add
// fused into the norm — saves the separate elementwise-add kernel and one
// full stream read per residual connection; profile r01: adds were 5.6% of
// cycle kernels).
template <typename T, bool HAS_RES>
__global__ void rmsnorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                   const T* __restrict__ w, T* __restrict__ y,
                                   T* __restrict__ sout, float* __restrict__ invr, int H,
                                   float eps, long N) {
  __shared__ float rbuf[NWAVES];
  const int H8 = H & ~7;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (size_t)row * H;
    const T* rr = HAS_RES ? res + (size_t)row * H : nullptr;
    T* sr = HAS_RES ? sout + (size_t)row * H : nullptr;
    T* yr = y + (size_t)row * H;
    float ss = 0.f;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float v[8];
      load8<T>(xr + base, v);
      if (HAS_RES) {
        float rv[8];
        load8<T>(rr + base, rv);
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] += rv[i];
        store8<T>(sr + base, v);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) ss += v[i] * v[i];
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      float xi = ScalarIO<T>::load(xr + i);
      if (HAS_RES) {
        xi += ScalarIO<T>::load(rr + i);
        ScalarIO<T>::store(sr + i, xi);
      }
      ss += xi * xi;
    }
    ss = block_sum<NWAVES>(ss, rbuf);
    const float r = rsqrtf(ss / H + eps);
    if (threadIdx.x == 0) invr[row] = r;
    const T* src = HAS_RES ? sr : xr;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float v[8], wv[8];
      load8<T>(src + base, v);
      load8<T>(w + base, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) v[i] = v[i] * r * wv[i];
      store8<T>(yr + base, v);
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      ScalarIO<T>::store(yr + i, ScalarIO<T>::load(src + i) * r * ScalarIO<T>::load(w + i));
    }
    __syncthreads();
  }
}

// HAS_DRES: dres (the residual-stream gradient from downstream) is added
// into dx — the backward half of the fused residual+norm.
template <typename T, bool HAS_DRES>
__global__ void rmsnorm_bwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                   const float* __restrict__ invr, const T* __restrict__ dy,
                                   const T* __restrict__ dres, T* __restrict__ dx,
                                   float* __restrict__ dw, int H, long N) {
  extern __shared__ float dwacc[];  // H floats
  __shared__ float rbuf[NWAVES];
  for (int i = threadIdx.x; i < H; i += BLOCK) dwacc[i] = 0.f;
  __syncthreads();
  const int H8 = H & ~7;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (size_t)row * H;
    const T* dyr = dy + (size_t)row * H;
    const T* drr = HAS_DRES ? dres + (size_t)row * H : nullptr;
    T* dxr = dx + (size_t)row * H;
    const float r = invr[row];
    float c = 0.f;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float xv[8], dv[8], wv[8];
      load8<T>(xr + base, xv);
      load8<T>(dyr + base, dv);
      load8<T>(w + base, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) c += dv[i] * wv[i] * xv[i];
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK)
      c += ScalarIO<T>::load(dyr + i) * ScalarIO<T>::load(w + i) * ScalarIO<T>::load(xr + i);
    c = block_sum<NWAVES>(c, rbuf);
    const float k = c * r * r * r / H;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float xv[8], dv[8], wv[8], o[8];
      load8<T>(xr + base, xv);
      load8<T>(dyr + base, dv);
      load8<T>(w + base, wv);
      float ev[8];
      if (HAS_DRES) load8<T>(drr + base, ev);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        o[i] = r * wv[i] * dv[i] - k * xv[i];
        if (HAS_DRES) o[i] += ev[i];
        dwacc[base + i] += dv[i] * xv[i] * r;
      }
      store8<T>(dxr + base, o);
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      float xv = ScalarIO<T>::load(xr + i);
      float dv = ScalarIO<T>::load(dyr + i);
      float o = r * ScalarIO<T>::load(w + i) * dv - k * xv;
      if (HAS_DRES) o += ScalarIO<T>::load(drr + i);
      ScalarIO<T>::store(dxr + i, o);
      dwacc[i] += dv * xv * r;
    }
    __syncthreads();
  }
  for (int i = threadIdx.x; i < H; i += BLOCK) dw[(size_t)blockIdx.x * H + i] = dwacc[i];
}

template <typename T, bool HAS_BIAS, bool HAS_RES>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                     const T* __restrict__ w, const T* __restrict__ b,
                                     T* __restrict__ y, T* __restrict__ sout,
                                     float* __restrict__ mean, float* __restrict__ invstd, int H,
                                     float eps, long N) {
  __shared__ float rbuf[NWAVES];
  const int H8 = H & ~7;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (size_t)row * H;
    const T* rr = HAS_RES ? res + (size_t)row * H : nullptr;
    T* sr = HAS_RES ? sout + (size_t)row * H : nullptr;
    T* yr = y + (size_t)row * H;
    float s = 0.f, ss = 0.f;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float v[8];
      load8<T>(xr + base, v);
      if (HAS_RES) {
        float rv[8];
        load8<T>(rr + base, rv);
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] += rv[i];
        store8<T>(sr + base, v);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s += v[i];
        ss += v[i] * v[i];
      }
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      float xi = ScalarIO<T>::load(xr + i);
      if (HAS_RES) {
        xi += ScalarIO<T>::load(rr + i);
        ScalarIO<T>::store(sr + i, xi);
      }
      s += xi;
      ss += xi * xi;
    }
    s = block_sum<NWAVES>(s, rbuf);
    ss = block_sum<NWAVES>(ss, rbuf);
    const float mu = s / H;
    const float var = fmaxf(ss / H - mu * mu, 0.f);
    const float istd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean[row] = mu;
      invstd[row] = istd;
    }
    const T* src = HAS_RES ? sr : xr;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float v[8], wv[8], bv[8];
      load8<T>(src + base, v);
      load8<T>(w + base, wv);
      if (HAS_BIAS) load8<T>(b + base, bv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float o = (v[i] - mu) * istd * wv[i];
        if (HAS_BIAS) o += bv[i];
        v[i] = o;
      }
      store8<T>(yr + base, v);
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      float o = (ScalarIO<T>::load(src + i) - mu) * istd * ScalarIO<T>::load(w + i);
      if (HAS_BIAS) o += ScalarIO<T>::load(b + i);
      ScalarIO<T>::store(yr + i, o);
    }
    __syncthreads();
  }
}

template <typename T, bool HAS_DRES>
__global__ void layernorm_bwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd, const T* __restrict__ dy,
                                     const T* __restrict__ dres, T* __restrict__ dx,
                                     float* __restrict__ dw, float* __restrict__ db, int H,
                                     long N) {
  extern __shared__ float acc[];  // 2*H floats: [dw | db]
  __shared__ float rbuf[NWAVES];
  float* dwacc = acc;
  float* dbacc = acc + H;
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    dwacc[i] = 0.f;
    dbacc[i] = 0.f;
  }
  __syncthreads();
  const int H8 = H & ~7;
  for (long row = blockIdx.x; row < N; row += gridDim.x) {
    const T* xr = x + (size_t)row * H;
    const T* dyr = dy + (size_t)row * H;
    const T* drr = HAS_DRES ? dres + (size_t)row * H : nullptr;
    T* dxr = dx + (size_t)row * H;
    const float mu = mean[row];
    const float istd = invstd[row];
    float s1 = 0.f, s2 = 0.f;  // sum(dy*w), sum(dy*w*xhat)
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float xv[8], dv[8], wv[8];
      load8<T>(xr + base, xv);
      load8<T>(dyr + base, dv);
      load8<T>(w + base, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float dyw = dv[i] * wv[i];
        s1 += dyw;
        s2 += dyw * (xv[i] - mu) * istd;
      }
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      float dyw = ScalarIO<T>::load(dyr + i) * ScalarIO<T>::load(w + i);
      s1 += dyw;
      s2 += dyw * (ScalarIO<T>::load(xr + i) - mu) * istd;
    }
    s1 = block_sum<NWAVES>(s1, rbuf) / H;
    s2 = block_sum<NWAVES>(s2, rbuf) / H;
    for (int base = threadIdx.x * 8; base < H8; base += BLOCK * 8) {
      float xv[8], dv[8], wv[8], o[8];
      load8<T>(xr + base, xv);
      load8<T>(dyr + base, dv);
      load8<T>(w + base, wv);
      float ev[8];
      if (HAS_DRES) load8<T>(drr + base, ev);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float xhat = (xv[i] - mu) * istd;
        o[i] = istd * (dv[i] * wv[i] - s1 - xhat * s2);
        if (HAS_DRES) o[i] += ev[i];
        dwacc[base + i] += dv[i] * xhat;
        dbacc[base + i] += dv[i];
      }
      store8<T>(dxr + base, o);
    }
    for (int i = H8 + threadIdx.x; i < H; i += BLOCK) {
      float dv = ScalarIO<T>::load(dyr + i);
      float xhat = (ScalarIO<T>::load(xr + i) - mu) * istd;
      float o = istd * (dv * ScalarIO<T>::load(w + i) - s1 - xhat * s2);
      if (HAS_DRES) o += ScalarIO<T>::load(drr + i);
      ScalarIO<T>::store(dxr + i, o);
      dwacc[i] += dv * xhat;
      dbacc[i] += dv;
    }
    __syncthreads();
  }
  // per-block partial slabs (plain stores; v1's atomicAdd tail serialized
  // ~2048 adds per address = ~60 us/launch — profile r02)
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    dw[(size_t)blockIdx.x * H + i] = dwacc[i];
    db[(size_t)blockIdx.x * H + i] = dbacc[i];
  }
}

// Wave-per-row forward variants for small/medium H (decode shapes: H=768,
// N=128 rows).  The block kernels above give each 768-wide row a 256-thread
// block: only 96 lanes load anything and the row pays two LDS+barrier
// reduction rounds.  Here each WAVE owns a row — shfl-only reductions, zero
// LDS, zero barriers — and a block carries NWAVES independent rows.
// Measured (bench cycle profile pf3): layernorm_fwd was 8.1 us/call avg and
// the top non-GEMM kernel at 185 ms of a 3.1 s GPU cycle.
template <typename T, bool HAS_RES>
__global__ void rmsnorm_fwd_wave_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                        const T* __restrict__ w, T* __restrict__ y,
                                        T* __restrict__ sout, float* __restrict__ invr, int H,
                                        float eps, long N) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int H8 = H & ~7;
  for (long row = (long)blockIdx.x * NWAVES + wid; row < N;
       row += (long)gridDim.x * NWAVES) {
    const T* xr = x + (size_t)row * H;
    const T* rr = HAS_RES ? res + (size_t)row * H : nullptr;
    T* sr = HAS_RES ? sout + (size_t)row * H : nullptr;
    T* yr = y + (size_t)row * H;
    float ss = 0.f;
    for (int base = lane * 8; base < H8; base += WAVE * 8) {
      float v[8];
      load8<T>(xr + base, v);
      if (HAS_RES) {
        float rv[8];
        load8<T>(rr + base, rv);
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] += rv[i];
        store8<T>(sr + base, v);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) ss += v[i] * v[i];
    }
    for (int i = H8 + lane; i < H; i += WAVE) {
      float xi = ScalarIO<T>::load(xr + i);
      if (HAS_RES) {
        xi += ScalarIO<T>::load(rr + i);
        ScalarIO<T>::store(sr + i, xi);
      }
      ss += xi * xi;
    }
    ss = wave_sum(ss);
    const float r = rsqrtf(ss / H + eps);
    if (lane == 0) invr[row] = r;
    const T* src = HAS_RES ? sr : xr;
    for (int base = lane * 8; base < H8; base += WAVE * 8) {
      float v[8], wv[8];
      load8<T>(src + base, v);
      load8<T>(w + base, wv);
#pragma unroll
      for (int i = 0; i < 8; ++i) v[i] = v[i] * r * wv[i];
      store8<T>(yr + base, v);
    }
    for (int i = H8 + lane; i < H; i += WAVE)
      ScalarIO<T>::store(yr + i, ScalarIO<T>::load(src + i) * r * ScalarIO<T>::load(w + i));
  }
}

template <typename T, bool HAS_BIAS, bool HAS_RES>
__global__ void layernorm_fwd_wave_kernel(const T* __restrict__ x, const T* __restrict__ res,
                                          const T* __restrict__ w, const T* __restrict__ b,
                                          T* __restrict__ y, T* __restrict__ sout,
                                          float* __restrict__ mean,
                                          float* __restrict__ invstd, int H, float eps,
                                          long N) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int H8 = H & ~7;
  for (long row = (long)blockIdx.x * NWAVES + wid; row < N;
       row += (long)gridDim.x * NWAVES) {
    const T* xr = x + (size_t)row * H;
    const T* rr = HAS_RES ? res + (size_t)row * H : nullptr;
    T* sr = HAS_RES ? sout + (size_t)row * H : nullptr;
    T* yr = y + (size_t)row * H;
    float s = 0.f, ss = 0.f;
    for (int base = lane * 8; base < H8; base += WAVE * 8) {
      float v[8];
      load8<T>(xr + base, v);
      if (HAS_RES) {
        float rv[8];
        load8<T>(rr + base, rv);
#pragma unroll
        for (int i = 0; i < 8; ++i) v[i] += rv[i];
        store8<T>(sr + base, v);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s += v[i];
        ss += v[i] * v[i];
      }
    }
    for (int i = H8 + lane; i < H; i += WAVE) {
      float xi = ScalarIO<T>::load(xr + i);
      if (HAS_RES) {
        xi += ScalarIO<T>::load(rr + i);
        ScalarIO<T>::store(sr + i, xi);
      }
      s += xi;
      ss += xi * xi;
    }
    s = wave_sum(s);
    ss = wave_sum(ss);
    const float mu = s / H;
    const float var = fmaxf(ss / H - mu * mu, 0.f);
    const float istd = rsqrtf(var + eps);
    if (lane == 0) {
      mean[row] = mu;
      invstd[row] = istd;
    }
    const T* src = HAS_RES ? sr : xr;
    for (int base = lane * 8; base < H8; base += WAVE * 8) {
      float v[8], wv[8], bv[8];
      load8<T>(src + base, v);
      load8<T>(w + base, wv);
      if (HAS_BIAS) load8<T>(b + base, bv);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float o = (v[i] - mu) * istd * wv[i];
        if (HAS_BIAS) o += bv[i];
        v[i] = o;
      }
      store8<T>(yr + base, v);
    }
    for (int i = H8 + lane; i < H; i += WAVE) {
      float o = (ScalarIO<T>::load(src + i) - mu) * istd * ScalarIO<T>::load(w + i);
      if (HAS_BIAS) o += ScalarIO<T>::load(b + i);
      ScalarIO<T>::store(yr + i, o);
    }
  }
}

int pick_grid(long n) { return (int)std::min<long>(n, 2048); }

// wave-per-row dispatch bound: a wave covers H<=4096 in <=8 vec8 strides
constexpr int WAVE_H_MAX = 4096;
int pick_wave_grid(long n) {
  return (int)std::min<long>((n + NWAVES - 1) / NWAVES, 2048);
}
bool wave_norm_enabled() {
  static const bool on = [] {
    const char* e = getenv("TRLX_AMD_NO_WAVE_NORM");
    return !(e && e[0] == '1');
  }();
  return on;
}

}  // namespace

std::vector<at::Tensor> rmsnorm_fwd(const at::Tensor& x, const at::Tensor& w, double eps,
                                    const c10::optional<at::Tensor>& res) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(w.dtype() == x.dtype());
  const long N = x.size(0);
  const int H = x.size(1);
  const bool has_res = res.has_value();
  auto y = at::empty_like(x);
  auto invr = at::empty({N}, x.options().dtype(at::kFloat));
  at::Tensor sout, rc;
  if (has_res) {
    rc = res->contiguous();
    TORCH_CHECK(rc.sizes() == x.sizes() && rc.dtype() == x.dtype());
    sout = at::empty_like(x);
  }
  if (N == 0) return has_res ? std::vector<at::Tensor>{y, invr, sout}
                             : std::vector<at::Tensor>{y, invr};
  auto stream = c10::hip::getCurrentHIPStream();
  auto wc = w.contiguous();
  const int grid = pick_grid(N);
  const bool wavep = H <= WAVE_H_MAX && wave_norm_enabled();
  const int wgrid = pick_wave_grid(N);
#define LAUNCH_RMS_FWD(T, HR, XP, RP, WP, YP, SP)                                            \
  do {                                                                                       \
    if (wavep)                                                                               \
      rmsnorm_fwd_wave_kernel<T, HR><<<wgrid, BLOCK, 0, stream>>>(                           \
          XP, RP, WP, YP, SP, invr.data_ptr<float>(), H, (float)eps, N);                     \
    else                                                                                     \
      rmsnorm_fwd_kernel<T, HR><<<grid, BLOCK, 0, stream>>>(                                 \
          XP, RP, WP, YP, SP, invr.data_ptr<float>(), H, (float)eps, N);                     \
  } while (0)
  if (x.dtype() == at::kBFloat16) {
    auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
    auto wp = reinterpret_cast<const bf16_t*>(wc.data_ptr());
    auto yp = reinterpret_cast<bf16_t*>(y.data_ptr());
    if (has_res)
      LAUNCH_RMS_FWD(bf16_t, true, xp, reinterpret_cast<const bf16_t*>(rc.data_ptr()), wp, yp,
                     reinterpret_cast<bf16_t*>(sout.data_ptr()));
    else
      LAUNCH_RMS_FWD(bf16_t, false, xp, nullptr, wp, yp, nullptr);
  } else {
    if (has_res)
      LAUNCH_RMS_FWD(float, true, x.data_ptr<float>(), rc.data_ptr<float>(),
                     wc.data_ptr<float>(), y.data_ptr<float>(), sout.data_ptr<float>());
    else
      LAUNCH_RMS_FWD(float, false, x.data_ptr<float>(), nullptr, wc.data_ptr<float>(),
                     y.data_ptr<float>(), nullptr);
  }
#undef LAUNCH_RMS_FWD
  HIP_CHECK_LAST();
  return has_res ? std::vector<at::Tensor>{y, invr, sout} : std::vector<at::Tensor>{y, invr};
}

std::vector<at::Tensor> rmsnorm_bwd(const at::Tensor& x, const at::Tensor& w,
                                    const at::Tensor& invr, const at::Tensor& dy,
                                    const c10::optional<at::Tensor>& dres) {
  const long N = x.size(0);
  const int H = x.size(1);
  TORCH_CHECK(H <= 16384, "rmsnorm_bwd: H too large for LDS accumulation");
  const bool has_dres = dres.has_value();
  at::Tensor drc;
  if (has_dres) drc = dres->contiguous();
  auto dx = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  auto wc = w.contiguous();
  auto dyc = dy.contiguous();
  const int grid = pick_grid(N);
  auto dwp = at::empty({std::max(grid, 1), H}, x.options().dtype(at::kFloat));
  if (N == 0) dwp.zero_();
  const size_t lds = (size_t)H * sizeof(float);
  if (N > 0) {
    if (x.dtype() == at::kBFloat16) {
      auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
      auto wp = reinterpret_cast<const bf16_t*>(wc.data_ptr());
      auto dyp = reinterpret_cast<const bf16_t*>(dyc.data_ptr());
      auto dxp = reinterpret_cast<bf16_t*>(dx.data_ptr());
      if (has_dres)
        rmsnorm_bwd_kernel<bf16_t, true><<<grid, BLOCK, lds, stream>>>(
            xp, wp, invr.data_ptr<float>(), dyp,
            reinterpret_cast<const bf16_t*>(drc.data_ptr()), dxp, dwp.data_ptr<float>(), H, N);
      else
        rmsnorm_bwd_kernel<bf16_t, false><<<grid, BLOCK, lds, stream>>>(
            xp, wp, invr.data_ptr<float>(), dyp, nullptr, dxp, dwp.data_ptr<float>(), H, N);
    } else {
      if (has_dres)
        rmsnorm_bwd_kernel<float, true><<<grid, BLOCK, lds, stream>>>(
            x.data_ptr<float>(), wc.data_ptr<float>(), invr.data_ptr<float>(),
            dyc.data_ptr<float>(), drc.data_ptr<float>(), dx.data_ptr<float>(),
            dwp.data_ptr<float>(), H, N);
      else
        rmsnorm_bwd_kernel<float, false><<<grid, BLOCK, lds, stream>>>(
            x.data_ptr<float>(), wc.data_ptr<float>(), invr.data_ptr<float>(),
            dyc.data_ptr<float>(), nullptr, dx.data_ptr<float>(), dwp.data_ptr<float>(), H, N);
    }
    HIP_CHECK_LAST();
  }
  auto dwf = dwp.sum(0);
  return {dx, dwf.to(w.dtype())};
}

std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x, const at::Tensor& w,
                                      const c10::optional<at::Tensor>& b, double eps,
                                      const c10::optional<at::Tensor>& res) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  const long N = x.size(0);
  const int H = x.size(1);
  const bool has_res = res.has_value();
  auto y = at::empty_like(x);
  auto mean = at::empty({N}, x.options().dtype(at::kFloat));
  auto invstd = at::empty({N}, x.options().dtype(at::kFloat));
  at::Tensor sout, rc;
  if (has_res) {
    rc = res->contiguous();
    TORCH_CHECK(rc.sizes() == x.sizes() && rc.dtype() == x.dtype());
    sout = at::empty_like(x);
  }
  if (N == 0) return has_res ? std::vector<at::Tensor>{y, mean, invstd, sout}
                             : std::vector<at::Tensor>{y, mean, invstd};
  auto stream = c10::hip::getCurrentHIPStream();
  auto wc = w.contiguous();
  const bool has_bias = b.has_value();
  at::Tensor bc;
  if (has_bias) bc = b->contiguous();
  const int grid = pick_grid(N);
  const bool wavep = H <= WAVE_H_MAX && wave_norm_enabled();
  const int wgrid = pick_wave_grid(N);
#define LAUNCH_LN_FWD(T, HB, HR, XP, RP, WP, BP, YP, SP)                                    \
  do {                                                                                      \
    if (wavep)                                                                              \
      layernorm_fwd_wave_kernel<T, HB, HR><<<wgrid, BLOCK, 0, stream>>>(                    \
          XP, RP, WP, BP, YP, SP, mean.data_ptr<float>(), invstd.data_ptr<float>(), H,      \
          (float)eps, N);                                                                   \
    else                                                                                    \
      layernorm_fwd_kernel<T, HB, HR><<<grid, BLOCK, 0, stream>>>(                          \
          XP, RP, WP, BP, YP, SP, mean.data_ptr<float>(), invstd.data_ptr<float>(), H,      \
          (float)eps, N);                                                                   \
  } while (0)
  if (x.dtype() == at::kBFloat16) {
    auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
    auto wp = reinterpret_cast<const bf16_t*>(wc.data_ptr());
    auto yp = reinterpret_cast<bf16_t*>(y.data_ptr());
    auto bp = has_bias ? reinterpret_cast<const bf16_t*>(bc.data_ptr()) : nullptr;
    auto rp = has_res ? reinterpret_cast<const bf16_t*>(rc.data_ptr()) : nullptr;
    auto sp = has_res ? reinterpret_cast<bf16_t*>(sout.data_ptr()) : nullptr;
    if (has_bias && has_res) LAUNCH_LN_FWD(bf16_t, true, true, xp, rp, wp, bp, yp, sp);
    else if (has_bias) LAUNCH_LN_FWD(bf16_t, true, false, xp, nullptr, wp, bp, yp, nullptr);
    else if (has_res) LAUNCH_LN_FWD(bf16_t, false, true, xp, rp, wp, nullptr, yp, sp);
    else LAUNCH_LN_FWD(bf16_t, false, false, xp, nullptr, wp, nullptr, yp, nullptr);
  } else {
    auto xp = x.data_ptr<float>();
    auto wp = wc.data_ptr<float>();
    auto yp = y.data_ptr<float>();
    auto bp = has_bias ? bc.data_ptr<float>() : nullptr;
    auto rp = has_res ? rc.data_ptr<float>() : nullptr;
    auto sp = has_res ? sout.data_ptr<float>() : nullptr;
    if (has_bias && has_res) LAUNCH_LN_FWD(float, true, true, xp, rp, wp, bp, yp, sp);
    else if (has_bias) LAUNCH_LN_FWD(float, true, false, xp, nullptr, wp, bp, yp, nullptr);
    else if (has_res) LAUNCH_LN_FWD(float, false, true, xp, rp, wp, nullptr, yp, sp);
    else LAUNCH_LN_FWD(float, false, false, xp, nullptr, wp, nullptr, yp, nullptr);
  }
#undef LAUNCH_LN_FWD
  HIP_CHECK_LAST();
  return has_res ? std::vector<at::Tensor>{y, mean, invstd, sout}
                 : std::vector<at::Tensor>{y, mean, invstd};
}

std::vector<at::Tensor> layernorm_bwd(const at::Tensor& x, const at::Tensor& w,
                                      const at::Tensor& mean, const at::Tensor& invstd,
                                      const at::Tensor& dy,
                                      const c10::optional<at::Tensor>& dres) {
  const long N = x.size(0);
  const int H = x.size(1);
  TORCH_CHECK(H <= 16384, "layernorm_bwd: H too large for LDS accumulation");
  const bool has_dres = dres.has_value();
  at::Tensor drc;
  if (has_dres) drc = dres->contiguous();
  auto dx = at::empty_like(x);
  auto stream = c10::hip::getCurrentHIPStream();
  auto wc = w.contiguous();
  auto dyc = dy.contiguous();
  const int grid = pick_grid(N);
  auto dwp = at::empty({std::max(grid, 1), H}, x.options().dtype(at::kFloat));
  auto dbp = at::empty({std::max(grid, 1), H}, x.options().dtype(at::kFloat));
  if (N == 0) {
    dwp.zero_();
    dbp.zero_();
  }
  const size_t lds = 2 * (size_t)H * sizeof(float);
  if (N > 0) {
    if (x.dtype() == at::kBFloat16) {
      auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
      auto wp = reinterpret_cast<const bf16_t*>(wc.data_ptr());
      auto dyp = reinterpret_cast<const bf16_t*>(dyc.data_ptr());
      auto dxp = reinterpret_cast<bf16_t*>(dx.data_ptr());
      if (has_dres)
        layernorm_bwd_kernel<bf16_t, true><<<grid, BLOCK, lds, stream>>>(
            xp, wp, mean.data_ptr<float>(), invstd.data_ptr<float>(), dyp,
            reinterpret_cast<const bf16_t*>(drc.data_ptr()), dxp, dwp.data_ptr<float>(),
            dbp.data_ptr<float>(), H, N);
      else
        layernorm_bwd_kernel<bf16_t, false><<<grid, BLOCK, lds, stream>>>(
            xp, wp, mean.data_ptr<float>(), invstd.data_ptr<float>(), dyp, nullptr, dxp,
            dwp.data_ptr<float>(), dbp.data_ptr<float>(), H, N);
    } else {
      if (has_dres)
        layernorm_bwd_kernel<float, true><<<grid, BLOCK, lds, stream>>>(
            x.data_ptr<float>(), wc.data_ptr<float>(), mean.data_ptr<float>(),
            invstd.data_ptr<float>(), dyc.data_ptr<float>(), drc.data_ptr<float>(),
            dx.data_ptr<float>(), dwp.data_ptr<float>(), dbp.data_ptr<float>(), H, N);
      else
        layernorm_bwd_kernel<float, false><<<grid, BLOCK, lds, stream>>>(
            x.data_ptr<float>(), wc.data_ptr<float>(), mean.data_ptr<float>(),
            invstd.data_ptr<float>(), dyc.data_ptr<float>(), nullptr, dx.data_ptr<float>(),
            dwp.data_ptr<float>(), dbp.data_ptr<float>(), H, N);
    }
    HIP_CHECK_LAST();
  }
  auto dwf = dwp.sum(0);
  auto dbf = dbp.sum(0);
  return {dx, dwf.to(w.dtype()), dbf.to(w.dtype())};
}
