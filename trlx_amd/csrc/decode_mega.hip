// Fused decode-stage kernels (gfx950) — the per-token decode step as FIVE
// kernels per layer instead of ~12 (plus embed / lm_head+sample / advance),
// each stage folding its neighbours' elementwise work:
//
//   stage_gemm:  C = act( norm(A) @ W^T + bias ) (+residual), with the row
//                (sum, sumsq) statistics of the NEXT norm accumulated in the
//                epilogue — kills the separate LayerNorm / bias / GELU /
//                residual-add kernels that were ~7 us each in profile r01.
//                Wave-level 16x16 MFMA tiles stream W with the skinny_gemm
//                4-deep register pipeline; the 4 waves of a block split K and
//                combine through LDS (4x busy waves vs plain tiling — these
//                skinny GEMMs are latency-bound, not bandwidth-bound).
//   embed_stats: token+position embedding lookup + row stats + buffer zeroing
//                (replaces 3-4 elementwise kernels per token).
//   lm_sample:   final-norm-folded lm_head GEMM whose epilogue feeds logits
//                (rounded through bf16 for engine parity) straight into the
//                Gumbel-max packed-u64 argmax of sampling.hip — the [B, V]
//                logits never round-trip HBM.
//
// Rationale (NOTES_ROUND2 item 1 + this round's coop_microbench): decode is
// kernel-latency bound (~150 kernels x 5-12 us at B=128); a grid.sync
// megakernel is NOT viable on MI355X (measured 33-40 us per fenced grid-wide
// barrier at 256 blocks — 63 barriers/token would cost 2 ms), so the fusion
// happens at kernel granularity and the per-token step stays hipGraph-
// captured (63 launches instead of ~150, each doing real work).
//
// The decode attention stage is csrc/attn_decode.hip's fused_decode_attn
// (unchanged); state advancement is csrc/decode_advance.hip with a variant
// here that consumes the sampler's packed winners directly.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int SBLOCK = 256;
constexpr int SWAVES = SBLOCK / WAVE;

typedef __attribute__((ext_vector_type(8))) short bf16x8_st;
typedef __attribute__((ext_vector_type(4))) float f32x4_st;

DEV float st_act(float x, int act) {
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

DEV unsigned int st_float_orderable(float x) {
  unsigned int u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

// ---- fused projection stage -------------------------------------------------
// NORM: fold (x - mu) * rstd * nw + nb into the A-fragment (mu/rstd from the
// producing stage's accumulated row stats). RESID/OSTATS: epilogue extras.
template <bool NORM, bool RMS, bool RESID, bool OSTATS>
__global__ __launch_bounds__(SBLOCK) void stage_gemm_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ W, const bf16_t* __restrict__ bias,
    bf16_t* __restrict__ C, int M, int N, int K, const float* __restrict__ nstats,
    const bf16_t* __restrict__ nw, const bf16_t* __restrict__ nb, float eps, float inv_nK,
    int act, const bf16_t* __restrict__ resid, float* __restrict__ out_stats) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nM = (M + 15) >> 4;
  const int nN = (N + 15) >> 4;
  const int ntiles = nM * nN;
  const int kq = ((K / 32 + SWAVES - 1) / SWAVES) * 32;
  const int k0 = wid * kq;
  const int k1 = min(K, k0 + kq);
  __shared__ float red[SWAVES * WAVE * 4];

  for (int tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int mt = tile / nN;  // mt-major: A fragments stay L1-hot across nt
    const int nt = tile % nN;
    const int arow = min(mt * 16 + (lane & 15), M - 1);
    const int wrow = min(nt * 16 + (lane & 15), N - 1);
    const int k8 = (lane >> 4) * 8;
    const bf16_t* ap = A + (size_t)arow * K + k8;
    const bf16_t* wp = W + (size_t)wrow * K + k8;

    float mu = 0.f, rstd = 1.f;
    if (NORM) {
      const float s1 = nstats[arow * 2];
      const float s2 = nstats[arow * 2 + 1];
      if (RMS) {
        rstd = __frsqrt_rn(s2 * inv_nK + eps);
      } else {
        mu = s1 * inv_nK;
        rstd = __frsqrt_rn(fmaxf(s2 * inv_nK - mu * mu, 0.f) + eps);
      }
    }

    f32x4_st acc = {0.f, 0.f, 0.f, 0.f};
    int k = k0;
    if (!NORM && k + 128 <= k1) {
      // raw-A fast path: the skinny_gemm 4-deep register pipeline
      bf16x8_st a0 = *reinterpret_cast<const bf16x8_st*>(ap + k);
      bf16x8_st b0 = *reinterpret_cast<const bf16x8_st*>(wp + k);
      bf16x8_st a1 = *reinterpret_cast<const bf16x8_st*>(ap + k + 32);
      bf16x8_st b1 = *reinterpret_cast<const bf16x8_st*>(wp + k + 32);
      bf16x8_st a2 = *reinterpret_cast<const bf16x8_st*>(ap + k + 64);
      bf16x8_st b2 = *reinterpret_cast<const bf16x8_st*>(wp + k + 64);
      bf16x8_st a3 = *reinterpret_cast<const bf16x8_st*>(ap + k + 96);
      bf16x8_st b3 = *reinterpret_cast<const bf16x8_st*>(wp + k + 96);
      for (k += 128; k + 128 <= k1; k += 128) {
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
        a0 = *reinterpret_cast<const bf16x8_st*>(ap + k);
        b0 = *reinterpret_cast<const bf16x8_st*>(wp + k);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
        a1 = *reinterpret_cast<const bf16x8_st*>(ap + k + 32);
        b1 = *reinterpret_cast<const bf16x8_st*>(wp + k + 32);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, acc, 0, 0, 0);
        a2 = *reinterpret_cast<const bf16x8_st*>(ap + k + 64);
        b2 = *reinterpret_cast<const bf16x8_st*>(wp + k + 64);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, b3, acc, 0, 0, 0);
        a3 = *reinterpret_cast<const bf16x8_st*>(ap + k + 96);
        b3 = *reinterpret_cast<const bf16x8_st*>(wp + k + 96);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, acc, 0, 0, 0);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, b3, acc, 0, 0, 0);
    }
    for (; k < k1; k += 32) {
      bf16x8_st av = *reinterpret_cast<const bf16x8_st*>(ap + k);
      bf16x8_st wv = *reinterpret_cast<const bf16x8_st*>(wp + k);
      if (NORM) {
        bf16x8_st nwv = *reinterpret_cast<const bf16x8_st*>(nw + k + k8);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float xv = (bf2f((unsigned short)av[i]) - mu) * rstd * bf2f((unsigned short)nwv[i]);
          if (nb) xv += bf2f(nb[k + k8 + i].u);
          av[i] = (short)f2bf(xv);
        }
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, wv, acc, 0, 0, 0);
    }

#pragma unroll
    for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[r];
    __syncthreads();
    if (wid == 0) {
      const int ccol = nt * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int crow = mt * 16 + (lane >> 4) * 4 + r;
        float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] + red[(2 * WAVE + lane) * 4 + r] +
                  red[(3 * WAVE + lane) * 4 + r];
        const bool live = crow < M && ccol < N;
        float vr = 0.f;
        if (live) {
          if (bias) v += bf2f(bias[ccol].u);
          v = st_act(v, act);
          if (RESID) v += bf2f(resid[(size_t)crow * N + ccol].u);
          const unsigned short vb = f2bf(v);
          C[(size_t)crow * N + ccol].u = vb;
          vr = bf2f(vb);  // stats of the STORED value
        }
        if (OSTATS) {
          float s = vr, s2 = vr * vr;
#pragma unroll
          for (int off = 8; off > 0; off >>= 1) {
            s += __shfl_xor(s, off);
            s2 += __shfl_xor(s2, off);
          }
          if ((lane & 15) == 0 && crow < M) {
            atomicAdd(&out_stats[crow * 2], s);
            atomicAdd(&out_stats[crow * 2 + 1], s2);
          }
        }
      }
    }
    __syncthreads();
  }
}

// ---- embed + row stats + stat-slot zeroing ----------------------------------
// ``stats`` is [nslots, B, 2]: one (sum, sumsq) slot PER NORM INSTANCE
// (ln1/ln2 of every layer + the final norm) so no per-layer zero kernels are
// needed — this kernel writes slot 0 (the first ln1) and zeroes the rest.
__global__ void embed_stats_kernel(const bf16_t* __restrict__ wte, const bf16_t* __restrict__ wpe,
                                   const long* __restrict__ cur_tok,
                                   const int* __restrict__ pos_ids, int pos_offset,
                                   bf16_t* __restrict__ x, float* __restrict__ stats, int nslots,
                                   unsigned long long* __restrict__ packed, int B, int H) {
  __shared__ float buf[2 * SWAVES];
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const bf16_t* te = wte + (size_t)cur_tok[b] * H;
    const bf16_t* pe = wpe ? wpe + (size_t)(pos_ids[b] + pos_offset) * H : nullptr;
    float ssum = 0.f, ssq = 0.f;
    for (int i = threadIdx.x; i < H; i += SBLOCK) {
      float v = bf2f(te[i].u);
      if (pe) v += bf2f(pe[i].u);
      const unsigned short vb = f2bf(v);
      x[(size_t)b * H + i].u = vb;
      const float vr = bf2f(vb);
      ssum += vr;
      ssq += vr * vr;
    }
    const float bs = block_sum<SWAVES>(ssum, buf);
    const float bq = block_sum<SWAVES>(ssq, buf + SWAVES);
    for (int sl = 1 + threadIdx.x; sl < nslots; sl += SBLOCK) {
      stats[((size_t)sl * B + b) * 2] = 0.f;
      stats[((size_t)sl * B + b) * 2 + 1] = 0.f;
    }
    if (threadIdx.x == 0) {
      stats[b * 2] = bs;
      stats[b * 2 + 1] = bq;
      packed[b] = 0ull;
    }
    __syncthreads();
  }
}

// ---- final-norm lm_head GEMM + gumbel-max sampling --------------------------
// Block assignment is mt-major (a block's tiles all share the same 16 rows)
// so each block keeps a RUNNING per-row winner in registers and issues ONE
// packed atomicMax per row at the end — the first cut atomicMax'd every
// element ([B, V] = 6.4M same-address atomics per token; measured 10x
// regression on the whole bench).
template <bool RMS, bool SAMPLE>
__global__ __launch_bounds__(SBLOCK) void lm_sample_kernel(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ wlm, const bf16_t* __restrict__ blm,
    const float* __restrict__ nstats, const bf16_t* __restrict__ nw,
    const bf16_t* __restrict__ nb, unsigned long long* __restrict__ packed, int M, int N, int K,
    float eps, float inv_temp, unsigned long long seed, const long* __restrict__ offset_ptr,
    int ncolblocks) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nN = (N + 15) >> 4;
  const int kq = ((K / 32 + SWAVES - 1) / SWAVES) * 32;
  const int k0 = wid * kq;
  const int k1 = min(K, k0 + kq);
  const float inv_nK = 1.f / K;
  const unsigned long long off = (unsigned long long)(*offset_ptr);
  const unsigned long long key = splitmix64(seed ^ (0x9e3779b97f4a7c15ull * (off + 1)));
  __shared__ float red[SWAVES * WAVE * 4];

  const int mt = blockIdx.x / ncolblocks;
  const int c0 = blockIdx.x % ncolblocks;
  const int arow = min(mt * 16 + (lane & 15), M - 1);
  const int k8 = (lane >> 4) * 8;
  const bf16_t* ap = x + (size_t)arow * K + k8;

  const float s1 = nstats[arow * 2];
  const float s2 = nstats[arow * 2 + 1];
  float mu = 0.f, rstd;
  if (RMS) {
    rstd = __frsqrt_rn(s2 * inv_nK + eps);
  } else {
    mu = s1 * inv_nK;
    rstd = __frsqrt_rn(fmaxf(s2 * inv_nK - mu * mu, 0.f) + eps);
  }
  float bestv[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  int bestc[4] = {0, 0, 0, 0};

  for (int nt = c0; nt < nN; nt += ncolblocks) {
    const int wrow = min(nt * 16 + (lane & 15), N - 1);
    const bf16_t* wp = wlm + (size_t)wrow * K + k8;
    f32x4_st acc = {0.f, 0.f, 0.f, 0.f};
    for (int k = k0; k < k1; k += 32) {
      bf16x8_st av = *reinterpret_cast<const bf16x8_st*>(ap + k);
      bf16x8_st wv = *reinterpret_cast<const bf16x8_st*>(wp + k);
      bf16x8_st nwv = *reinterpret_cast<const bf16x8_st*>(nw + k + k8);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float xv = (bf2f((unsigned short)av[i]) - mu) * rstd * bf2f((unsigned short)nwv[i]);
        if (nb) xv += bf2f(nb[k + k8 + i].u);
        av[i] = (short)f2bf(xv);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, wv, acc, 0, 0, 0);
    }

#pragma unroll
    for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[r];
    __syncthreads();
    if (wid == 0) {
      const int ccol = nt * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int crow = mt * 16 + (lane >> 4) * 4 + r;
        float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] + red[(2 * WAVE + lane) * 4 + r] +
                  red[(3 * WAVE + lane) * 4 + r];
        float val = -INFINITY;
        if (crow < M && ccol < N) {
          if (blm) v += bf2f(blm[ccol].u);
          v = bf2f(f2bf(v));  // engine parity: sampler sees bf16 logits
          if (SAMPLE) {
            const float u = rng_uniform(key, (unsigned long long)crow, (unsigned long long)ccol);
            val = v * inv_temp + (-__logf(-__logf(u)));
          } else {
            val = v;
          }
        }
        // per-row winner across the 16 columns (low 4 lane bits), ties to
        // the LOWER column (matches the packed ~index ordering)
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
        if ((lane & 15) == 0) {
          if (wv2 > bestv[r] || (wv2 == bestv[r] && wc < bestc[r])) {
            bestv[r] = wv2;
            bestc[r] = wc;
          }
        }
      }
    }
    __syncthreads();
  }

  if (wid == 0 && (lane & 15) == 0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int crow = mt * 16 + (lane >> 4) * 4 + r;
      if (crow < M && bestv[r] > -INFINITY) {
        const unsigned long long p = ((unsigned long long)st_float_orderable(bestv[r]) << 32) |
                                     (unsigned int)(~(unsigned int)bestc[r]);
        atomicMax(&packed[crow], p);
      }
    }
  }
}

// ---- advance variant consuming the packed winners ---------------------------
__global__ void advance_packed_kernel(const unsigned long long* __restrict__ packed,
                                      long* __restrict__ out_tokens, long* __restrict__ cur_tok,
                                      bool* __restrict__ finished, long* __restrict__ rng_offset,
                                      long* __restrict__ step_col, long* __restrict__ cache_idx,
                                      int* __restrict__ seq_lens, int* __restrict__ pos_ids,
                                      const int* __restrict__ key_starts, int B, int max_new,
                                      long eos, long pad) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  const long col = *step_col;
  const long new_cache = *cache_idx + 1;
  if (b < B) {
    long t = (long)(~(unsigned int)(packed[b] & 0xffffffffu));
    if (eos >= 0) {
      if (finished[b]) t = pad;
      finished[b] = finished[b] || (t == eos);
    }
    if (col >= 0 && col < max_new) out_tokens[(size_t)b * max_new + col] = t;
    cur_tok[b] = t;
    seq_lens[b] += 1;
    const int ks = key_starts ? key_starts[b] : 0;
    pos_ids[b] = (int)(new_cache - ks);
  }
  if (b == 0) {
    *rng_offset += 1;
    *step_col = col + 1;
    *cache_idx = new_cache;
  }
}

int stage_grid(int M, int N) {
  const int ntiles = ((M + 15) / 16) * ((N + 15) / 16);
  return min(ntiles, 2048);
}

}  // namespace

void stage_gemm(const at::Tensor& a, const at::Tensor& w, const c10::optional<at::Tensor>& bias,
                at::Tensor& c, const c10::optional<at::Tensor>& nstats,
                const c10::optional<at::Tensor>& nw, const c10::optional<at::Tensor>& nb,
                bool norm_rms, double eps, long act, const c10::optional<at::Tensor>& resid,
                const c10::optional<at::Tensor>& out_stats) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16 && a.dim() == 2 && a.is_contiguous());
  TORCH_CHECK(w.dtype() == at::kBFloat16 && w.is_contiguous());
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && K % 32 == 0, "stage_gemm: K must be a multiple of 32");
  TORCH_CHECK(c.size(0) == M && c.size(1) == N && c.is_contiguous());
  const bool has_norm = nstats.has_value();
  const bool has_res = resid.has_value();
  const bool has_os = out_stats.has_value();
  auto bp = bias.has_value() ? reinterpret_cast<const bf16_t*>(bias->data_ptr()) : nullptr;
  auto np = has_norm ? nstats->data_ptr<float>() : nullptr;
  auto nwp = nw.has_value() ? reinterpret_cast<const bf16_t*>(nw->data_ptr()) : nullptr;
  auto nbp = nb.has_value() ? reinterpret_cast<const bf16_t*>(nb->data_ptr()) : nullptr;
  auto rp = has_res ? reinterpret_cast<const bf16_t*>(resid->data_ptr()) : nullptr;
  auto op = has_os ? out_stats->data_ptr<float>() : nullptr;
  auto stream = c10::hip::getCurrentHIPStream();
  const int grid = stage_grid(M, N);
  const float inv_nK = 1.f / K;

#define LAUNCH_SG(NORMV, RMSV, RESV, OSV)                                                     \
  stage_gemm_kernel<NORMV, RMSV, RESV, OSV><<<grid, SBLOCK, 0, stream>>>(                     \
      reinterpret_cast<const bf16_t*>(a.data_ptr()),                                          \
      reinterpret_cast<const bf16_t*>(w.data_ptr()), bp,                                      \
      reinterpret_cast<bf16_t*>(c.data_ptr()), M, N, K, np, nwp, nbp, (float)eps, inv_nK,     \
      (int)act, rp, op)
  if (has_norm) {
    if (norm_rms) {
      if (has_res && has_os) LAUNCH_SG(true, true, true, true);
      else if (has_res) LAUNCH_SG(true, true, true, false);
      else if (has_os) LAUNCH_SG(true, true, false, true);
      else LAUNCH_SG(true, true, false, false);
    } else {
      if (has_res && has_os) LAUNCH_SG(true, false, true, true);
      else if (has_res) LAUNCH_SG(true, false, true, false);
      else if (has_os) LAUNCH_SG(true, false, false, true);
      else LAUNCH_SG(true, false, false, false);
    }
  } else {
    if (has_res && has_os) LAUNCH_SG(false, false, true, true);
    else if (has_res) LAUNCH_SG(false, false, true, false);
    else if (has_os) LAUNCH_SG(false, false, false, true);
    else LAUNCH_SG(false, false, false, false);
  }
#undef LAUNCH_SG
  HIP_CHECK_LAST();
}

void embed_stats(const at::Tensor& wte, const c10::optional<at::Tensor>& wpe,
                 const at::Tensor& cur_tok, const at::Tensor& pos_ids, long pos_offset,
                 at::Tensor& x, at::Tensor& stats, at::Tensor& packed) {
  const int B = x.size(0), H = x.size(1);
  TORCH_CHECK(stats.dim() == 3 && stats.size(1) == B && stats.size(2) == 2);
  auto stream = c10::hip::getCurrentHIPStream();
  embed_stats_kernel<<<min(B, 256), SBLOCK, 0, stream>>>(
      reinterpret_cast<const bf16_t*>(wte.data_ptr()),
      wpe.has_value() ? reinterpret_cast<const bf16_t*>(wpe->data_ptr()) : nullptr,
      cur_tok.data_ptr<long>(), pos_ids.data_ptr<int>(), (int)pos_offset,
      reinterpret_cast<bf16_t*>(x.data_ptr()), stats.data_ptr<float>(), (int)stats.size(0),
      reinterpret_cast<unsigned long long*>(packed.data_ptr<long>()), B, H);
  HIP_CHECK_LAST();
}

void lm_sample(const at::Tensor& x, const at::Tensor& wlm, const c10::optional<at::Tensor>& blm,
               const at::Tensor& nstats, const at::Tensor& nw,
               const c10::optional<at::Tensor>& nb, at::Tensor& packed, bool norm_rms,
               double eps, double temperature, long seed, const at::Tensor& rng_offset) {
  const int M = x.size(0), K = x.size(1), N = wlm.size(0);
  TORCH_CHECK(wlm.size(1) == K && K % 32 == 0);
  auto stream = c10::hip::getCurrentHIPStream();
  const int nM = (M + 15) / 16;
  const int ncolblocks = max(1, min((N + 15) / 16, 2048 / nM));
  const int grid = nM * ncolblocks;
  const float inv_temp = temperature == 0.0 ? 0.f : (float)(1.0 / temperature);
  const bool sample = temperature != 0.0;
  auto xp = reinterpret_cast<const bf16_t*>(x.data_ptr());
  auto wp = reinterpret_cast<const bf16_t*>(wlm.data_ptr());
  auto bp = blm.has_value() ? reinterpret_cast<const bf16_t*>(blm->data_ptr()) : nullptr;
  auto nwp = reinterpret_cast<const bf16_t*>(nw.data_ptr());
  auto nbp = nb.has_value() ? reinterpret_cast<const bf16_t*>(nb->data_ptr()) : nullptr;
  auto pk = reinterpret_cast<unsigned long long*>(packed.data_ptr<long>());
#define LAUNCH_LM(RMSV, SV)                                                                \
  lm_sample_kernel<RMSV, SV><<<grid, SBLOCK, 0, stream>>>(                                 \
      xp, wp, bp, nstats.data_ptr<float>(), nwp, nbp, pk, M, N, K, (float)eps, inv_temp,   \
      (unsigned long long)seed, rng_offset.data_ptr<long>(), ncolblocks)
  if (norm_rms) {
    if (sample) LAUNCH_LM(true, true);
    else LAUNCH_LM(true, false);
  } else {
    if (sample) LAUNCH_LM(false, true);
    else LAUNCH_LM(false, false);
  }
#undef LAUNCH_LM
  HIP_CHECK_LAST();
}

void advance_packed(const at::Tensor& packed, at::Tensor& out_tokens, at::Tensor& cur_tok,
                    at::Tensor& finished, at::Tensor& rng_offset, at::Tensor& step_col,
                    at::Tensor& cache_idx, at::Tensor& seq_lens, at::Tensor& pos_ids,
                    const c10::optional<at::Tensor>& key_starts, long eos, long pad) {
  const int B = cur_tok.numel();
  const int max_new = out_tokens.size(1);
  const int* ks = nullptr;
  if (key_starts.has_value()) ks = key_starts->data_ptr<int>();
  auto stream = c10::hip::getCurrentHIPStream();
  advance_packed_kernel<<<(B + 255) / 256, 256, 0, stream>>>(
      reinterpret_cast<const unsigned long long*>(packed.data_ptr<long>()),
      out_tokens.data_ptr<long>(), cur_tok.data_ptr<long>(), finished.data_ptr<bool>(),
      rng_offset.data_ptr<long>(), step_col.data_ptr<long>(), cache_idx.data_ptr<long>(),
      seq_lens.data_ptr<int>(), pos_ids.data_ptr<int>(), ks, B, max_new, eos, pad);
  HIP_CHECK_LAST();
}
