// stage_gemm v2 — LDS-staged skinny GEMM for the fused decode stages.
//
// v1 (decode_mega.hip stage_gemm_kernel) loads MFMA fragments straight from
// global memory: each wave-instruction touches 16 rows x 64 B — SIXTEEN
// scattered segments — and measured 0.2-0.5 TB/s even L3-hot (qkv 9.6 us,
// down 13.3 us standalone; tools/bench_stage_gemm.py).  v2 is the guide's §5
// shape:
//   - global loads are COALESCED: one wave-instruction covers 4 rows x 256 B
//     contiguous (4 segments instead of 16);
//   - chunks go global -> VGPR -> LDS (async-STAGE split: the NEXT chunk's
//     loads are issued before the CURRENT chunk's MFMAs, so HBM latency
//     hides under compute; buffers are wave-private -> no barriers in the
//     K-loop);
//   - MFMA fragments come from LDS (ds_read_b128, +8 B row stagger for
//     conflict-free banks).
//
// The norm fold also moves here: the CONSUMER's pre-phase streams its 16 A
// rows once (coalesced) to compute the row (mean, rstd) directly — the v1
// producer-side atomicAdd statistics (+9 us per stage, order-
// nondeterministic) are gone entirely, and so is the stats tensor.
//
//   C[M,N](bf16) = act( norm(A)[M,K] @ W[N,K]^T + bias ) (+residual)
//
// Tile: one 16x16 output tile per BLOCK; the 4 waves split K in quarters and
// combine through LDS (tile counts stay high for the skinny decode shapes).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int V2BLOCK = 256;
constexpr int V2WAVES = V2BLOCK / WAVE;
constexpr int BK = 128;        // K elements staged per chunk
constexpr int LROW = BK + 8;   // LDS row stride (shorts): +16 B -> 4-bank row stagger (b128 reads tile the banks)

typedef __attribute__((ext_vector_type(8))) short bf16x8_v2;
typedef __attribute__((ext_vector_type(4))) float f32x4_v2;

DEV float v2_act(float x, int act) {
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

// one wave's in-flight chunk: 4 coalesced instructions each for A and W.
// instruction j covers rows [4j, 4j+4) x 256 B: lane -> row 4j + (lane>>4),
// elements [(lane&15)*8, +8).
struct ChunkRegs {
  bf16x8_v2 a[4], w[4];
};

DEV void issue_loads(ChunkRegs& r, const bf16_t* __restrict__ A, const bf16_t* __restrict__ W,
                     size_t strideA, size_t strideW, int k) {
  const int lane = threadIdx.x % WAVE;
  const int rr = lane >> 4;            // row within the 4-row group
  const int e = (lane & 15) * 8;       // element slice
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    r.a[j] = *reinterpret_cast<const bf16x8_v2*>(A + (size_t)(j * 4 + rr) * strideA + k + e);
    r.w[j] = *reinterpret_cast<const bf16x8_v2*>(W + (size_t)(j * 4 + rr) * strideW + k + e);
  }
}

template <bool NORM>
DEV void write_chunk(unsigned short* __restrict__ a_lds, unsigned short* __restrict__ w_lds,
                     ChunkRegs& r, const float* __restrict__ stat_buf,
                     const bf16_t* __restrict__ nw, const bf16_t* __restrict__ nb, int k,
                     float inv_nK, float eps, bool rms) {
  const int lane = threadIdx.x % WAVE;
  const int rr = lane >> 4;
  const int e = (lane & 15) * 8;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    if (NORM) {
      const int row = j * 4 + rr;
      const float s = stat_buf[row * 2];
      const float s2 = stat_buf[row * 2 + 1];
      float rm = 0.f, rstd;
      if (rms) {
        rstd = __frsqrt_rn(s2 * inv_nK + eps);
      } else {
        rm = s * inv_nK;
        rstd = __frsqrt_rn(fmaxf(s2 * inv_nK - rm * rm, 0.f) + eps);
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = (bf2f((unsigned short)r.a[j][i]) - rm) * rstd * bf2f(nw[k + e + i].u);
        if (nb) v += bf2f(nb[k + e + i].u);
        r.a[j][i] = (short)f2bf(v);
      }
    }
    *reinterpret_cast<bf16x8_v2*>(a_lds + (j * 4 + rr) * LROW + e) = r.a[j];
    *reinterpret_cast<bf16x8_v2*>(w_lds + (j * 4 + rr) * LROW + e) = r.w[j];
  }
}

template <bool NORM, bool RMS, bool RESID>
__global__ __launch_bounds__(V2BLOCK, 2) void stage_gemm_v2_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ W, const bf16_t* __restrict__ bias,
    bf16_t* __restrict__ C, int M, int N, int K, const bf16_t* __restrict__ nw,
    const bf16_t* __restrict__ nb, float eps, int act, const bf16_t* __restrict__ resid,
    float* __restrict__ pstats_out) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nM = (M + 15) >> 4;
  const int nN = (N + 15) >> 4;
  const int ntiles = nM * nN;
  const int kq = (((K / BK) + V2WAVES - 1) / V2WAVES) * BK;  // K per wave, BK-aligned
  const float inv_nK = 1.f / K;

  __shared__ unsigned short a_lds[V2WAVES][2][16 * LROW];
  __shared__ unsigned short w_lds[V2WAVES][2][16 * LROW];
  __shared__ float red[V2WAVES * WAVE * 4];
  __shared__ float stat_buf[2 * 16];

  for (int tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int mt = tile / nN;
    const int nt = tile % nN;
    // edge tiles slide back so every tile is FULL (M, N >= 16 checked by the
    // launcher); overlapped rows/cols are written twice with identical values
    const int m0 = min(mt * 16, M - 16);
    const int n0 = min(nt * 16, N - 16);

    if (NORM) {
      // pre-phase: block streams its 16 A rows coalesced -> row stats
      const int r = threadIdx.x >> 4;
      const int t16 = threadIdx.x & 15;
      const bf16_t* ap = A + (size_t)min(m0 + r, M - 1) * K;
      float s = 0.f, s2 = 0.f;
      for (int k = t16 * 8; k < K; k += 16 * 8) {
        float v[8];
        load8<bf16_t>(ap + k, v);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          s += v[i];
          s2 += v[i] * v[i];
        }
      }
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) {
        s += __shfl_xor(s, off);
        s2 += __shfl_xor(s2, off);
      }
      if (t16 == 0) {
        stat_buf[r * 2] = s;
        stat_buf[r * 2 + 1] = s2;
      }
      __syncthreads();
    }

    const bf16_t* Abase = A + (size_t)m0 * K;
    const bf16_t* Wbase = W + (size_t)n0 * K;

    const int k0 = wid * kq;
    const int k1 = min(K, k0 + kq);
    const int nchunks = (k1 - k0 + BK - 1) / BK;

    unsigned short* al0 = a_lds[wid][0];
    unsigned short* al1 = a_lds[wid][1];
    unsigned short* wl0 = w_lds[wid][0];
    unsigned short* wl1 = w_lds[wid][1];

    f32x4_v2 acc = {0.f, 0.f, 0.f, 0.f};
    ChunkRegs cur, nxt;
    if (nchunks > 0) issue_loads(cur, Abase, Wbase, K, K, k0);
    for (int ci = 0; ci < nchunks; ++ci) {
      if (ci + 1 < nchunks) issue_loads(nxt, Abase, Wbase, K, K, k0 + (ci + 1) * BK);
      unsigned short* al = (ci & 1) ? al1 : al0;
      unsigned short* wl = (ci & 1) ? wl1 : wl0;
      write_chunk<NORM>(al, wl, cur, stat_buf, nw, nb, k0 + ci * BK, inv_nK, eps, RMS);
      const int g8 = (lane >> 4) * 8;
      const int c = lane & 15;
#pragma unroll
      for (int kk = 0; kk < BK; kk += 32) {
        bf16x8_v2 af = *reinterpret_cast<const bf16x8_v2*>(al + c * LROW + kk + g8);
        bf16x8_v2 wf = *reinterpret_cast<const bf16x8_v2*>(wl + c * LROW + kk + g8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, wf, acc, 0, 0, 0);
      }
      cur = nxt;
    }

    // combine the 4 K-quarter partials through LDS
#pragma unroll
    for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[r];
    __syncthreads();
    if (wid == 0) {
      const int ccol = n0 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int crow = m0 + (lane >> 4) * 4 + r;
        float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] + red[(2 * WAVE + lane) * 4 + r] +
                  red[(3 * WAVE + lane) * 4 + r];
        if (bias) v += bf2f(bias[ccol].u);
        v = v2_act(v, act);
        if (RESID) v += bf2f(resid[(size_t)crow * N + ccol].u);
        const unsigned short vb = f2bf(v);
        C[(size_t)crow * N + ccol].u = vb;
        if (pstats_out) {
          // per-row 16-col partial (sum, sumsq) of the STORED value — plain
          // slab store, reduced by the consumer (stage_gemm_v3 pre-phase)
          const float vr = bf2f(vb);
          float s = vr, s2 = vr * vr;
#pragma unroll
          for (int off = 8; off > 0; off >>= 1) {
            s += __shfl_xor(s, off);
            s2 += __shfl_xor(s2, off);
          }
          if ((lane & 15) == 0) {
            pstats_out[((size_t)nt * M + crow) * 2] = s;
            pstats_out[((size_t)nt * M + crow) * 2 + 1] = s2;
          }
        }
      }
    }
    __syncthreads();
  }
}


DEV unsigned int v2_float_orderable(float x) {
  unsigned int u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

// final-norm lm_head GEMM + gumbel-max sampling, v2-staged.  Block layout is
// mt-major (ncolblocks blocks per row-tile) so each block keeps RUNNING
// per-row winners and issues one packed atomicMax per row at the end.
template <bool RMS, bool SAMPLE>
__global__ __launch_bounds__(V2BLOCK, 2) void lm_sample_v2_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ W, const bf16_t* __restrict__ blm,
    const bf16_t* __restrict__ nw, const bf16_t* __restrict__ nb,
    unsigned long long* __restrict__ packed, int M, int N, int K, float eps, float inv_temp,
    unsigned long long seed, const long* __restrict__ offset_ptr, int ncolblocks) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nN = (N + 15) >> 4;
  const int kq = (((K / BK) + V2WAVES - 1) / V2WAVES) * BK;
  const float inv_nK = 1.f / K;
  const unsigned long long off = (unsigned long long)(*offset_ptr);
  const unsigned long long key = splitmix64(seed ^ (0x9e3779b97f4a7c15ull * (off + 1)));

  __shared__ unsigned short a_lds[V2WAVES][2][16 * LROW];
  __shared__ unsigned short w_lds[V2WAVES][2][16 * LROW];
  __shared__ float red[V2WAVES * WAVE * 4];
  __shared__ float stat_buf[2 * 16];

  const int mt = blockIdx.x / ncolblocks;
  const int c0 = blockIdx.x % ncolblocks;
  const int m0 = min(mt * 16, M - 16);

  {
    // pre-phase: row stats of the block's 16 A rows (the final norm)
    const int r = threadIdx.x >> 4;
    const int t16 = threadIdx.x & 15;
    const bf16_t* ap = A + (size_t)(m0 + r) * K;
    float s = 0.f, s2 = 0.f;
    for (int k = t16 * 8; k < K; k += 16 * 8) {
      float v[8];
      load8<bf16_t>(ap + k, v);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        s += v[i];
        s2 += v[i] * v[i];
      }
    }
#pragma unroll
    for (int o = 8; o > 0; o >>= 1) {
      s += __shfl_xor(s, o);
      s2 += __shfl_xor(s2, o);
    }
    if (t16 == 0) {
      stat_buf[r * 2] = s;
      stat_buf[r * 2 + 1] = s2;
    }
    __syncthreads();
  }

  const bf16_t* Abase = A + (size_t)m0 * K;
  const int k0 = wid * kq;
  const int k1 = min(K, k0 + kq);
  const int nchunks = (k1 - k0 + BK - 1) / BK;

  float bestv[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
  int bestc[4] = {0, 0, 0, 0};

  for (int nt = c0; nt < nN; nt += ncolblocks) {
    const int n0 = min(nt * 16, N - 16);
    const bf16_t* Wbase = W + (size_t)n0 * K;
    unsigned short* al0 = a_lds[wid][0];
    unsigned short* al1 = a_lds[wid][1];
    unsigned short* wl0 = w_lds[wid][0];
    unsigned short* wl1 = w_lds[wid][1];

    f32x4_v2 acc = {0.f, 0.f, 0.f, 0.f};
    ChunkRegs cur, nxt;
    if (nchunks > 0) issue_loads(cur, Abase, Wbase, K, K, k0);
    for (int ci = 0; ci < nchunks; ++ci) {
      if (ci + 1 < nchunks) issue_loads(nxt, Abase, Wbase, K, K, k0 + (ci + 1) * BK);
      unsigned short* al = (ci & 1) ? al1 : al0;
      unsigned short* wl = (ci & 1) ? wl1 : wl0;
      write_chunk<true>(al, wl, cur, stat_buf, nw, nb, k0 + ci * BK, inv_nK, eps, RMS);
      const int g8 = (lane >> 4) * 8;
      const int c = lane & 15;
#pragma unroll
      for (int kk = 0; kk < BK; kk += 32) {
        bf16x8_v2 af = *reinterpret_cast<const bf16x8_v2*>(al + c * LROW + kk + g8);
        bf16x8_v2 wf = *reinterpret_cast<const bf16x8_v2*>(wl + c * LROW + kk + g8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, wf, acc, 0, 0, 0);
      }
      cur = nxt;
    }

#pragma unroll
    for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[r];
    __syncthreads();
    if (wid == 0) {
      const int ccol = n0 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] + red[(2 * WAVE + lane) * 4 + r] +
                  red[(3 * WAVE + lane) * 4 + r];
        const int crow = m0 + (lane >> 4) * 4 + r;
        if (blm) v += bf2f(blm[ccol].u);
        v = bf2f(f2bf(v));  // engine parity: sampler sees bf16 logits
        float val;
        if (SAMPLE) {
          const float u = rng_uniform(key, (unsigned long long)crow, (unsigned long long)ccol);
          val = v * inv_temp + (-__logf(-__logf(u)));
        } else {
          val = v;
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
      const int crow = m0 + (lane >> 4) * 4 + r;
      if (bestv[r] > -INFINITY) {
        const unsigned long long p = ((unsigned long long)v2_float_orderable(bestv[r]) << 32) |
                                     (unsigned int)(~(unsigned int)bestc[r]);
        atomicMax(&packed[crow], p);
      }
    }
  }
}

}  // namespace

void stage_gemm_v2(const at::Tensor& a, const at::Tensor& w, const c10::optional<at::Tensor>& bias,
                   at::Tensor& c, bool norm, const c10::optional<at::Tensor>& nw,
                   const c10::optional<at::Tensor>& nb, bool norm_rms, double eps, long act,
                   const c10::optional<at::Tensor>& resid,
                   const c10::optional<at::Tensor>& pstats_out) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16 && a.dim() == 2 && a.is_contiguous());
  TORCH_CHECK(w.dtype() == at::kBFloat16 && w.is_contiguous());
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && K % BK == 0, "stage_gemm_v2: K must be a multiple of 128");
  TORCH_CHECK(M >= 16 && N >= 16, "stage_gemm_v2: M and N must be >= 16");
  TORCH_CHECK(c.size(0) == M && c.size(1) == N && c.is_contiguous());
  const bool has_res = resid.has_value();
  TORCH_CHECK(!pstats_out.has_value() || (M % 16 == 0 && N % 16 == 0),
              "stage_gemm_v2: pstats_out needs M,N multiples of 16");
  auto pso = pstats_out.has_value() ? pstats_out->data_ptr<float>() : (float*)nullptr;
  auto bp = bias.has_value() ? reinterpret_cast<const bf16_t*>(bias->data_ptr()) : nullptr;
  auto nwp = nw.has_value() ? reinterpret_cast<const bf16_t*>(nw->data_ptr()) : nullptr;
  auto nbp = nb.has_value() ? reinterpret_cast<const bf16_t*>(nb->data_ptr()) : nullptr;
  auto rp = has_res ? reinterpret_cast<const bf16_t*>(resid->data_ptr()) : nullptr;
  auto stream = c10::hip::getCurrentHIPStream();
  const int ntiles = ((M + 15) / 16) * ((N + 15) / 16);
  const int grid = min(ntiles, 2048);

#define LAUNCH_V2(NORMV, RMSV, RESV)                                                      \
  stage_gemm_v2_kernel<NORMV, RMSV, RESV><<<grid, V2BLOCK, 0, stream>>>(                  \
      reinterpret_cast<const bf16_t*>(a.data_ptr()),                                      \
      reinterpret_cast<const bf16_t*>(w.data_ptr()), bp,                                  \
      reinterpret_cast<bf16_t*>(c.data_ptr()), M, N, K, nwp, nbp, (float)eps, (int)act, rp, pso)
  if (norm) {
    if (norm_rms) {
      if (has_res) LAUNCH_V2(true, true, true);
      else LAUNCH_V2(true, true, false);
    } else {
      if (has_res) LAUNCH_V2(true, false, true);
      else LAUNCH_V2(true, false, false);
    }
  } else {
    if (has_res) LAUNCH_V2(false, false, true);
    else LAUNCH_V2(false, false, false);
  }
#undef LAUNCH_V2
  HIP_CHECK_LAST();
}


void lm_sample_v2(const at::Tensor& x, const at::Tensor& wlm,
                  const c10::optional<at::Tensor>& blm, const at::Tensor& nw,
                  const c10::optional<at::Tensor>& nb, at::Tensor& packed, bool norm_rms,
                  double eps, double temperature, long seed, const at::Tensor& rng_offset) {
  const int M = x.size(0), K = x.size(1), N = wlm.size(0);
  TORCH_CHECK(wlm.size(1) == K && K % 128 == 0 && M >= 16 && N >= 16);
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
#define LAUNCH_LMV2(RMSV, SV)                                                             \
  lm_sample_v2_kernel<RMSV, SV><<<grid, V2BLOCK, 0, stream>>>(                            \
      xp, wp, bp, nwp, nbp, pk, M, N, K, (float)eps, inv_temp, (unsigned long long)seed,  \
      rng_offset.data_ptr<long>(), ncolblocks)
  if (norm_rms) {
    if (sample) LAUNCH_LMV2(true, true);
    else LAUNCH_LMV2(true, false);
  } else {
    if (sample) LAUNCH_LMV2(false, true);
    else LAUNCH_LMV2(false, false);
  }
#undef LAUNCH_LMV2
  HIP_CHECK_LAST();
}
