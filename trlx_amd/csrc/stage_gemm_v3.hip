// stage_gemm v3 — W-stationary skinny GEMM for the fused decode stages.
//
// The v1/v2 measurements (tools/bench_stage_gemm.py) converge at ~10 us for
// the big-N stages no matter how loads are shaped.  The resolution: per-tile
// kernels re-read each 16-row W panel once per M-tile — 8x the unique weight
// bytes at M=128 (qkv 3.5 MB -> 28 MB demand, lm_head 77 MB -> 616 MB).  At
// that demand the kernels were ALREADY near the streaming limit; the wall is
// the re-read, not the access pattern.
//
// v3 makes W stationary: a wave loads one W fragment and feeds it to MT
// MFMAs (all M-tiles), so unique W bytes stream exactly once per kernel.  A
// is re-read per N-panel instead — it is KB-sized and L2-resident, the right
// asymmetry.  Norm statistics flow through PARTIAL-STAT SLABS
// (pstats[nt][row][2], plain stores by the producer's epilogue, reduced in
// the consumer's pre-phase) — deterministic, and without the producer-side
// atomicAdd tail that cost ~9 us per stage in v1.
//
//   C = act( norm(A) @ W^T + bias )        [the qkv / fc stages]
//   lm_sample_v3: the same loop + per-row gumbel-argmax winners
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int V3BLOCK = 256;
constexpr int V3WAVES = V3BLOCK / WAVE;

typedef __attribute__((ext_vector_type(8))) short bf16x8_v3;
typedef __attribute__((ext_vector_type(4))) float f32x4_v3;

DEV float v3_act(float x, int act) {
  switch (act) {
    case 1:
      return 0.5f * x * (1.f + erff(x * 0.70710678118654752f));
    case 2: {
      const float c = 0.797884560802865f;
      return 0.5f * x * (1.f + tanhf(c * (x + 0.044715f * x * x * x)));
    }
    case 3:
      return fmaxf(x, 0.f);
    case 4:
      return x / (1.f + __expf(-x));
    default:
      return x;
  }
}

DEV unsigned int v3_float_orderable(float x) {
  unsigned int u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

// consumer pre-phase: reduce the producer's nparts partial (sum, sumsq) slabs
// into LDS mean/rstd per row.  pstats layout [nparts][M][2].
template <bool RMS>
DEV void reduce_stats(const float* __restrict__ pstats, int nparts, int M, int K, float eps,
                      float* __restrict__ stat_lds /* [M][2] -> (mu, rstd) */) {
  const float inv_nK = 1.f / K;
  for (int row = threadIdx.x; row < M; row += V3BLOCK) {
    float s = 0.f, s2 = 0.f;
    for (int p = 0; p < nparts; ++p) {
      s += pstats[((size_t)p * M + row) * 2];
      s2 += pstats[((size_t)p * M + row) * 2 + 1];
    }
    float mu = 0.f, rstd;
    if (RMS) {
      rstd = __frsqrt_rn(s2 * inv_nK + eps);
    } else {
      mu = s * inv_nK;
      rstd = __frsqrt_rn(fmaxf(s2 * inv_nK - mu * mu, 0.f) + eps);
    }
    stat_lds[row * 2] = mu;
    stat_lds[row * 2 + 1] = rstd;
  }
  __syncthreads();
}

// MT = number of 16-row M-tiles each wave carries (M = MT*16 exactly).
template <int MT, bool NORM, bool RMS>
__global__ __launch_bounds__(V3BLOCK, 4) void stage_gemm_v3_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ W, const bf16_t* __restrict__ bias,
    bf16_t* __restrict__ C, int M, int N, int K, const float* __restrict__ pstats, int nparts,
    const bf16_t* __restrict__ nw, const bf16_t* __restrict__ nb, float eps, int act,
    float* __restrict__ pstats_out) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nN = (N + 15) >> 4;
  const int kq = ((K / 32 + V3WAVES - 1) / V3WAVES) * 32;

  __shared__ float stat_lds[256 * 2];
  __shared__ float red[V3WAVES * WAVE * 4];

  if (NORM) reduce_stats<RMS>(pstats, nparts, M, K, eps, stat_lds);

  const int c = lane & 15;
  const int k8 = (lane >> 4) * 8;

  for (int nt = blockIdx.x; nt < nN; nt += gridDim.x) {
    const int n0 = min(nt * 16, N - 16);
    const bf16_t* wp = W + (size_t)(n0 + c) * K + k8;
    const bf16_t* ap = A + (size_t)c * K + k8;  // row c of tile m: + m*16*K

    const int k0 = wid * kq;
    const int k1 = min(K, k0 + kq);

    f32x4_v3 acc[MT];
#pragma unroll
    for (int m = 0; m < MT; ++m) acc[m] = {0.f, 0.f, 0.f, 0.f};

    for (int k = k0; k < k1; k += 32) {
      bf16x8_v3 wv = *reinterpret_cast<const bf16x8_v3*>(wp + k);
#pragma unroll
      for (int m = 0; m < MT; ++m) {
        bf16x8_v3 av = *reinterpret_cast<const bf16x8_v3*>(ap + (size_t)m * 16 * K + k);
        if (NORM) {
          const float mu = stat_lds[(m * 16 + c) * 2];
          const float rstd = stat_lds[(m * 16 + c) * 2 + 1];
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float v = (bf2f((unsigned short)av[i]) - mu) * rstd * bf2f(nw[k + k8 + i].u);
            if (nb) v += bf2f(nb[k + k8 + i].u);
            av[i] = (short)f2bf(v);
          }
        }
        acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, wv, acc[m], 0, 0, 0);
      }
    }

    // epilogue per M-tile: reduce the 4 K-quarters, store, emit partial stats
#pragma unroll
    for (int m = 0; m < MT; ++m) {
#pragma unroll
      for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[m][r];
      __syncthreads();
      if (wid == 0) {
        const int ccol = n0 + c;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int crow = m * 16 + (lane >> 4) * 4 + r;
          float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] +
                    red[(2 * WAVE + lane) * 4 + r] + red[(3 * WAVE + lane) * 4 + r];
          if (bias) v += bf2f(bias[ccol].u);
          v = v3_act(v, act);
          const unsigned short vb = f2bf(v);
          if (crow < M && ccol < N) C[(size_t)crow * N + ccol].u = vb;
          if (pstats_out) {
            const float vr = bf2f(vb);
            float s = vr, s2 = vr * vr;
#pragma unroll
            for (int off = 8; off > 0; off >>= 1) {
              s += __shfl_xor(s, off);
              s2 += __shfl_xor(s2, off);
            }
            if (c == 0 && crow < M) {
              pstats_out[((size_t)nt * M + crow) * 2] = s;
              pstats_out[((size_t)nt * M + crow) * 2 + 1] = s2;
            }
          }
        }
      }
      __syncthreads();
    }
  }
}

// lm_head + gumbel-max, W-stationary.  Block-local per-row winners live in
// LDS across the block's N-panels; ONE packed atomicMax per row per block.
template <int MT, bool RMS, bool SAMPLE>
__global__ __launch_bounds__(V3BLOCK, 4) void lm_sample_v3_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ W, const bf16_t* __restrict__ blm,
    const float* __restrict__ pstats, int nparts, const bf16_t* __restrict__ nw,
    const bf16_t* __restrict__ nb, unsigned long long* __restrict__ packed, int M, int N, int K,
    float eps, float inv_temp, unsigned long long seed, const long* __restrict__ offset_ptr) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nN = (N + 15) >> 4;
  const int kq = ((K / 32 + V3WAVES - 1) / V3WAVES) * 32;
  const unsigned long long off = (unsigned long long)(*offset_ptr);
  const unsigned long long key = splitmix64(seed ^ (0x9e3779b97f4a7c15ull * (off + 1)));

  __shared__ float stat_lds[256 * 2];
  __shared__ float red[V3WAVES * WAVE * 4];
  __shared__ float bestv_lds[256];
  __shared__ int bestc_lds[256];

  reduce_stats<RMS>(pstats, nparts, M, K, eps, stat_lds);
  for (int row = threadIdx.x; row < M; row += V3BLOCK) {
    bestv_lds[row] = -INFINITY;
    bestc_lds[row] = 0;
  }
  __syncthreads();

  const int c = lane & 15;
  const int k8 = (lane >> 4) * 8;

  for (int nt = blockIdx.x; nt < nN; nt += gridDim.x) {
    const int n0 = min(nt * 16, N - 16);
    const bf16_t* wp = W + (size_t)(n0 + c) * K + k8;
    const bf16_t* ap = A + (size_t)c * K + k8;
    const int k0 = wid * kq;
    const int k1 = min(K, k0 + kq);

    f32x4_v3 acc[MT];
#pragma unroll
    for (int m = 0; m < MT; ++m) acc[m] = {0.f, 0.f, 0.f, 0.f};
    for (int k = k0; k < k1; k += 32) {
      bf16x8_v3 wv = *reinterpret_cast<const bf16x8_v3*>(wp + k);
#pragma unroll
      for (int m = 0; m < MT; ++m) {
        bf16x8_v3 av = *reinterpret_cast<const bf16x8_v3*>(ap + (size_t)m * 16 * K + k);
        const float mu = stat_lds[(m * 16 + c) * 2];
        const float rstd = stat_lds[(m * 16 + c) * 2 + 1];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float v = (bf2f((unsigned short)av[i]) - mu) * rstd * bf2f(nw[k + k8 + i].u);
          if (nb) v += bf2f(nb[k + k8 + i].u);
          av[i] = (short)f2bf(v);
        }
        acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, wv, acc[m], 0, 0, 0);
      }
    }

#pragma unroll
    for (int m = 0; m < MT; ++m) {
#pragma unroll
      for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[m][r];
      __syncthreads();
      if (wid == 0) {
        const int ccol = n0 + c;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int crow = m * 16 + (lane >> 4) * 4 + r;
          float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] +
                    red[(2 * WAVE + lane) * 4 + r] + red[(3 * WAVE + lane) * 4 + r];
          float val = -INFINITY;
          if (crow < M && ccol < N) {
            if (blm) v += bf2f(blm[ccol].u);
            v = bf2f(f2bf(v));  // engine parity: sampler sees bf16 logits
            if (SAMPLE) {
              const float u =
                  rng_uniform(key, (unsigned long long)crow, (unsigned long long)ccol);
              val = v * inv_temp + (-__logf(-__logf(u)));
            } else {
              val = v;
            }
          }
          float wv2 = val;
          int wc = ccol;
#pragma unroll
          for (int o = 8; o > 0; o >>= 1) {
            const float ov = __shfl_xor(wv2, o);
            const int oc = __shfl_xor(wc, o);
            if (ov > wv2 || (ov == wv2 && oc < wc)) {
              wv2 = ov;
              wc = oc;
            }
          }
          if (c == 0 && crow < M) {
            if (wv2 > bestv_lds[crow] ||
                (wv2 == bestv_lds[crow] && wc < bestc_lds[crow])) {
              bestv_lds[crow] = wv2;
              bestc_lds[crow] = wc;
            }
          }
        }
      }
      __syncthreads();
    }
  }

  // one flush per row per block
  for (int row = threadIdx.x; row < M; row += V3BLOCK) {
    if (bestv_lds[row] > -INFINITY) {
      const unsigned long long p = ((unsigned long long)v3_float_orderable(bestv_lds[row]) << 32) |
                                   (unsigned int)(~(unsigned int)bestc_lds[row]);
      atomicMax(&packed[row], p);
    }
  }
}

int pick_mt(int M) {
  if (M % 16 != 0) return 0;
  const int mt = M / 16;
  return (mt == 1 || mt == 2 || mt == 4 || mt == 8 || mt == 16) ? mt : 0;
}

}  // namespace

void stage_gemm_v3(const at::Tensor& a, const at::Tensor& w, const c10::optional<at::Tensor>& bias,
                   at::Tensor& c, const c10::optional<at::Tensor>& pstats, long nparts,
                   const c10::optional<at::Tensor>& nw, const c10::optional<at::Tensor>& nb,
                   bool norm_rms, double eps, long act,
                   const c10::optional<at::Tensor>& pstats_out) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16 && a.dim() == 2 && a.is_contiguous());
  TORCH_CHECK(w.dtype() == at::kBFloat16 && w.is_contiguous());
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && K % 32 == 0 && N >= 16);
  const int mt = pick_mt(M);
  TORCH_CHECK(mt > 0 && M <= 256, "stage_gemm_v3: M must be 16/32/64/128/256");
  const bool norm = pstats.has_value();
  auto bp = bias.has_value() ? reinterpret_cast<const bf16_t*>(bias->data_ptr()) : nullptr;
  auto psp = norm ? pstats->data_ptr<float>() : nullptr;
  auto nwp = nw.has_value() ? reinterpret_cast<const bf16_t*>(nw->data_ptr()) : nullptr;
  auto nbp = nb.has_value() ? reinterpret_cast<const bf16_t*>(nb->data_ptr()) : nullptr;
  auto pso = pstats_out.has_value() ? pstats_out->data_ptr<float>() : nullptr;
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = min((N + 15) / 16, 2048);

#define LAUNCH_V3(MTV, NORMV, RMSV)                                                        \
  stage_gemm_v3_kernel<MTV, NORMV, RMSV><<<grid, V3BLOCK, 0, stream>>>(                    \
      reinterpret_cast<const bf16_t*>(a.data_ptr()),                                       \
      reinterpret_cast<const bf16_t*>(w.data_ptr()), bp,                                   \
      reinterpret_cast<bf16_t*>(c.data_ptr()), M, N, K, psp, (int)nparts, nwp, nbp,        \
      (float)eps, (int)act, pso)
#define DISPATCH_MT(NORMV, RMSV)                                                           \
  do {                                                                                     \
    switch (mt) {                                                                          \
      case 1: LAUNCH_V3(1, NORMV, RMSV); break;                                            \
      case 2: LAUNCH_V3(2, NORMV, RMSV); break;                                            \
      case 4: LAUNCH_V3(4, NORMV, RMSV); break;                                            \
      case 8: LAUNCH_V3(8, NORMV, RMSV); break;                                            \
      case 16: LAUNCH_V3(16, NORMV, RMSV); break;                                          \
    }                                                                                      \
  } while (0)
  if (norm) {
    if (norm_rms) DISPATCH_MT(true, true);
    else DISPATCH_MT(true, false);
  } else {
    DISPATCH_MT(false, false);
  }
#undef DISPATCH_MT
#undef LAUNCH_V3
  HIP_CHECK_LAST();
}

void lm_sample_v3(const at::Tensor& x, const at::Tensor& wlm,
                  const c10::optional<at::Tensor>& blm, const at::Tensor& pstats, long nparts,
                  const at::Tensor& nw, const c10::optional<at::Tensor>& nb, at::Tensor& packed,
                  bool norm_rms, double eps, double temperature, long seed,
                  const at::Tensor& rng_offset) {
  const int M = x.size(0), K = x.size(1), N = wlm.size(0);
  TORCH_CHECK(wlm.size(1) == K && K % 32 == 0 && N >= 16);
  const int mt = pick_mt(M);
  TORCH_CHECK(mt > 0 && M <= 256, "lm_sample_v3: M must be 16/32/64/128/256");
  auto stream = c10::hip::getCurrentHIPStream();
  // fewer blocks than stage_gemm: each extra block is M more same-address
  // atomics at the end; 768 keeps the chip fed and the flush tail short
  const int grid = min((N + 15) / 16, 768);
  const float inv_temp = temperature == 0.0 ? 0.f : (float)(1.0 / temperature);
  const bool sample = temperature != 0.0;
  auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
  auto wp = reinterpret_cast<const bf16_t*>(wlm.data_ptr());
  auto bp = blm.has_value() ? reinterpret_cast<const bf16_t*>(blm->data_ptr()) : nullptr;
  auto nwp = reinterpret_cast<const bf16_t*>(nw.data_ptr());
  auto nbp = nb.has_value() ? reinterpret_cast<const bf16_t*>(nb->data_ptr()) : nullptr;
  auto pk = reinterpret_cast<unsigned long long*>(packed.data_ptr<long>());

#define LAUNCH_LM3(MTV, RMSV, SV)                                                          \
  lm_sample_v3_kernel<MTV, RMSV, SV><<<grid, V3BLOCK, 0, stream>>>(                        \
      xp, wp, bp, pstats.data_ptr<float>(), (int)nparts, nwp, nbp, pk, M, N, K,            \
      (float)eps, inv_temp, (unsigned long long)seed, rng_offset.data_ptr<long>())
#define DISPATCH_LM(RMSV, SV)                                                              \
  do {                                                                                     \
    switch (mt) {                                                                          \
      case 1: LAUNCH_LM3(1, RMSV, SV); break;                                              \
      case 2: LAUNCH_LM3(2, RMSV, SV); break;                                              \
      case 4: LAUNCH_LM3(4, RMSV, SV); break;                                              \
      case 8: LAUNCH_LM3(8, RMSV, SV); break;                                              \
      case 16: LAUNCH_LM3(16, RMSV, SV); break;                                            \
    }                                                                                      \
  } while (0)
  if (norm_rms) {
    if (sample) DISPATCH_LM(true, true);
    else DISPATCH_LM(true, false);
  } else {
    if (sample) DISPATCH_LM(false, true);
    else DISPATCH_LM(false, false);
  }
#undef DISPATCH_LM
#undef LAUNCH_LM3
  HIP_CHECK_LAST();
}
