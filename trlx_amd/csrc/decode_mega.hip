// Whole-decode-step megakernel (gfx950, cooperative launch).
//
// ONE kernel generates up to n_tokens tokens: per token it runs embed ->
// L x (qkv GEMM -> decode attention -> o GEMM -> fc GEMM -> down GEMM) ->
// lm_head GEMM fused with Gumbel-max sampling -> state advance, with
// cg::grid_group::sync() between dependent stages (63 grid syncs per token
// for GPT-2's 12 layers).  Rationale (profile r01 + NOTES_ROUND2 item 1):
// the per-token hipGraph replay serializes ~150 tiny kernels at a 5-12 us
// launch/latency floor each (~0.9 ms/token, chip 99% idle) while the weight
// traffic floor is ~40 us/token; in-kernel barriers replace the per-kernel
// latency with ~1 us grid syncs and keep all loop state device-resident.
//
// Design notes:
// - GEMM stages: wave-level 16x16 MFMA tiles streaming W (the skinny_gemm
//   pipeline), with IN-WORKGROUP split-K: the 4 waves of a block compute the
//   same tile over K quarters and combine through LDS — deterministic, no
//   global slabs, and 4x the busy waves of plain tiling (the decode GEMMs
//   are latency-bound at low waves/CU — skinny_gemm.hip header).
// - LayerNorm/RMSNorm is FOLDED into the consuming GEMM's A-fragment load;
//   row (sum, sumsq) statistics are accumulated by the producing stage's
//   epilogue via fp32 atomicAdd (order-nondeterministic in the last ulp;
//   the fp32 mean/rstd are insensitive at 1e-7 relative).
// - Attention reuses the fused_decode_attn structure (RoPE/copy append +
//   flash-decode per (b, head) unit, one block per unit, looped).
// - lm_head epilogue feeds logits straight into the Gumbel-max argmax
//   (packed (orderable-float|~index) u64 atomicMax, sampling.hip's scheme,
//   with the logit rounded through bf16 so tokens match the non-mega
//   engine's bf16 logits bit-for-bit given the same seed/offset).
// - Cross-XCD visibility between stages relies on grid.sync()'s device-scope
//   release/acquire fences (validated by tools/coop_microbench.hip).
//
// Eligibility is decided host-side (mega_decode_supported): MHA, head_dim
// 64/128, learned or rope positions, no ALiBi, no parallel residual, no
// pre-embed norm, B <= 256, top_k/top_p off.  Everything else stays on the
// hipGraph engine.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_cooperative_groups.h>

#include "common.h"

namespace cg = cooperative_groups;

namespace {

constexpr int MBLOCK = 256;
constexpr int MWAVES = MBLOCK / WAVE;

typedef __attribute__((ext_vector_type(8))) short bf16x8_mk;
typedef __attribute__((ext_vector_type(4))) float f32x4_mk;

struct MegaCfg {
  int B, H, I, V, L, heads, D, S;
  int max_new;
  float eps;
  int act;           // sk_act code for the MLP activation
  int norm_rms;      // 1 = rmsnorm (no mean, no bias)
  int pos_kind;      // 0 = learned embedding, 1 = rope
  int pos_offset;    // learned-position offset (OPT)
  int rot;           // rope rotary dims
  int interleaved;   // rope interleaved pairing
  float scale;       // attention q scale
  float inv_temp;    // 0 => greedy argmax
  unsigned long long seed;
  long eos, pad;     // eos < 0 => no eos handling
  int n_tokens;
};

// per-layer pointer-table slots (bf16 weights; bias slots may be 0)
constexpr int PW_LN1_W = 0, PW_LN1_B = 1, PW_QKV = 2, PW_QKV_B = 3, PW_O = 4, PW_O_B = 5,
              PW_LN2_W = 6, PW_LN2_B = 7, PW_FC = 8, PW_FC_B = 9, PW_DOWN = 10, PW_DOWN_B = 11,
              PW_KC = 12, PW_VC = 13, PW_PER_LAYER = 14;
// global slots before the per-layer table
constexpr int PG_WTE = 0, PG_WPE = 1, PG_RCOS = 2, PG_RSIN = 3, PG_LNF_W = 4, PG_LNF_B = 5,
              PG_WLM = 6, PG_WLM_B = 7, PG_N = 8;

DEV float mk_act(float x, int act) {
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

DEV unsigned int mk_float_orderable(float x) {
  unsigned int u = __float_as_uint(x);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

template <typename T>
DEV const T* PTR(const unsigned long long* p, int slot) {
  return reinterpret_cast<const T*>(p[slot]);
}

template <typename T>
DEV T* PTRW(const unsigned long long* p, int slot) {
  return reinterpret_cast<T*>(p[slot]);
}

// ---- GEMM stage -------------------------------------------------------------
// C[M,N](bf16) = act( normA(A)[M,K] @ W[N,K]^T + bias ) (+resid)
// with optional row-stat accumulation of C+resid into out_stats[M][2].
// All 4 waves of a block compute the same 16x16 tile over K quarters.
DEV void mega_gemm(const bf16_t* __restrict__ A, const bf16_t* __restrict__ W,
                   const bf16_t* __restrict__ bias, bf16_t* __restrict__ C, int M, int N, int K,
                   const float* __restrict__ nstats, const bf16_t* __restrict__ nw,
                   const bf16_t* __restrict__ nb, int norm_rms, float eps, float inv_nH, int act,
                   const bf16_t* __restrict__ resid, float* __restrict__ out_stats,
                   float* __restrict__ smem) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nM = (M + 15) >> 4;
  const int nN = (N + 15) >> 4;
  const int ntiles = nM * nN;
  // K quarters, 32-aligned (K is a multiple of 32 for all supported archs)
  const int kq = ((K / 32 + MWAVES - 1) / MWAVES) * 32;
  const int k0 = wid * kq;
  const int k1 = min(K, k0 + kq);

  for (int tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int mt = tile / nN;  // mt-major: A fragments stay L1-hot across nt
    const int nt = tile % nN;
    const int arow = min(mt * 16 + (lane & 15), M - 1);
    const int wrow = min(nt * 16 + (lane & 15), N - 1);
    const int k8 = (lane >> 4) * 8;
    const bf16_t* ap = A + (size_t)arow * K + k8;
    const bf16_t* wp = W + (size_t)wrow * K + k8;

    float mu = 0.f, rstd = 1.f;
    if (nstats) {
      const float s1 = nstats[arow * 2];
      const float s2 = nstats[arow * 2 + 1];
      if (norm_rms) {
        rstd = __frsqrt_rn(s2 * inv_nH + eps);
      } else {
        mu = s1 * inv_nH;
        rstd = __frsqrt_rn(fmaxf(s2 * inv_nH - mu * mu, 0.f) + eps);
      }
    }

    f32x4_mk acc = {0.f, 0.f, 0.f, 0.f};
    for (int k = k0; k < k1; k += 32) {
      bf16x8_mk av = *reinterpret_cast<const bf16x8_mk*>(ap + k);
      bf16x8_mk wv = *reinterpret_cast<const bf16x8_mk*>(wp + k);
      if (nstats) {
        // fold the producing stage's norm into the fragment
        bf16x8_mk nwv = *reinterpret_cast<const bf16x8_mk*>(nw + k + k8);
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float x = (bf2f((unsigned short)av[i]) - mu) * rstd * bf2f((unsigned short)nwv[i]);
          if (nb) x += bf2f((unsigned short)nb[k + k8 + i].u);
          av[i] = (short)f2bf(x);
        }
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, wv, acc, 0, 0, 0);
    }

    // combine the 4 K-quarter partials through LDS: smem[wid][lane][4]
    float* red = smem;  // MWAVES * WAVE * 4 floats = 4 KB
#pragma unroll
    for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[r];
    __syncthreads();
    if (wid == 0) {
      const int ccol = nt * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int crow = mt * 16 + (lane >> 4) * 4 + r;
        if (crow >= M || ccol >= N) continue;
        float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] + red[(2 * WAVE + lane) * 4 + r] +
                  red[(3 * WAVE + lane) * 4 + r];
        if (bias) v += bf2f(bias[ccol].u);
        v = mk_act(v, act);
        if (resid) v += bf2f(resid[(size_t)crow * N + ccol].u);
        const unsigned short vb = f2bf(v);
        C[(size_t)crow * N + ccol].u = vb;
        if (out_stats) {
          // per-row partial sums over this tile's 16 columns (lanes sharing
          // (lane>>4, r) hold the same row across the low 4 lane bits)
          float vr = bf2f(vb);  // stats of the STORED bf16 value
          float s = vr, s2 = vr * vr;
#pragma unroll
          for (int off = 8; off > 0; off >>= 1) {
            s += __shfl_xor(s, off);
            s2 += __shfl_xor(s2, off);
          }
          if ((lane & 15) == 0 && ccol < N) {
            atomicAdd(&out_stats[crow * 2], s);
            atomicAdd(&out_stats[crow * 2 + 1], s2);
          }
        }
      }
    }
    __syncthreads();
  }
}

// ---- decode attention stage (one block per (b, h) unit, looped) ------------
template <int D>
DEV void mega_attn(const bf16_t* __restrict__ qkv, bf16_t* __restrict__ kc,
                   bf16_t* __restrict__ vc, bf16_t* __restrict__ attn_out,
                   const int* __restrict__ seq_lens, const int* __restrict__ key_starts,
                   const float* __restrict__ rcos, const float* __restrict__ rsin, long pos,
                   const MegaCfg cfg, float* __restrict__ smem) {
  constexpr int G = D / 8;
  constexpr int KPW = WAVE / G;
  constexpr int NPART = MWAVES * KPW;
  const int H = cfg.heads;
  const int units = cfg.B * H;
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;

  // smem layout: q[D] | ms[NPART][2] | o[NPART][D]
  float* q_lds = smem;
  float* ms_buf = smem + D;
  float* o_buf = ms_buf + NPART * 2;

  for (int u = blockIdx.x; u < units; u += gridDim.x) {
    const int b = u / H;
    const int h = u % H;
    const int len = seq_lens[b];
    const int kstart = key_starts ? key_starts[b] : 0;

    // stage 1: split + rope + cache append for this head (3 rows)
    {
      const bf16_t* src = qkv + ((size_t)b * 3 * H + (size_t)wid * H + h) * D;
      bf16_t* kdst = kc + (((size_t)b * H + h) * cfg.S + pos) * D;
      bf16_t* vdst = vc + (((size_t)b * H + h) * cfg.S + pos) * D;
      if (wid < 2 && rcos != nullptr) {
        const int p = (int)pos - kstart;
        const float* c = rcos + (size_t)p * (cfg.rot / 2);
        const float* sn_p = rsin + (size_t)p * (cfg.rot / 2);
        for (int i = lane; i < cfg.rot / 2; i += WAVE) {
          const int i1 = cfg.interleaved ? 2 * i : i;
          const int i2 = cfg.interleaved ? 2 * i + 1 : i + cfg.rot / 2;
          const float x1 = bf2f(src[i1].u);
          const float x2 = bf2f(src[i2].u);
          const float r1 = x1 * c[i] - x2 * sn_p[i];
          const float r2 = x2 * c[i] + x1 * sn_p[i];
          if (wid == 0) {
            q_lds[i1] = r1 * cfg.scale;
            q_lds[i2] = r2 * cfg.scale;
          } else {
            kdst[i1].u = f2bf(r1);
            kdst[i2].u = f2bf(r2);
          }
        }
        for (int i = cfg.rot + lane; i < D; i += WAVE) {
          if (wid == 0) q_lds[i] = bf2f(src[i].u) * cfg.scale;
          else kdst[i] = src[i];
        }
      } else if (wid < 2) {
        for (int i = lane; i < D; i += WAVE) {
          if (wid == 0) q_lds[i] = bf2f(src[i].u) * cfg.scale;
          else kdst[i] = src[i];
        }
      } else if (wid == 2) {
        const int D4 = D / 4;
        for (int i = lane; i < D4; i += WAVE)
          reinterpret_cast<short4v*>(vdst)[i] = reinterpret_cast<const short4v*>(src)[i];
      }
    }
    __syncthreads();

    // stage 2: flash-decode over the cache (incl. the fresh key)
    const int kgrp = lane / G;
    const int d0 = (lane % G) * 8;
    const bf16_t* kbase = kc + ((size_t)b * H + h) * cfg.S * D;
    const bf16_t* vbase = vc + ((size_t)b * H + h) * cfg.S * D;
    float qf[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) qf[i] = q_lds[d0 + i];

    float m = -INFINITY, s = 0.f;
    float o[8];
#pragma unroll
    for (int i = 0; i < 8; ++i) o[i] = 0.f;
    for (int sbase = kstart + wid * KPW; sbase < len; sbase += MWAVES * KPW) {
      const int key = sbase + kgrp;
      if (key < len) {
        float kf[8], vf[8];
        load8<bf16_t>(kbase + (size_t)key * D + d0, kf);
        load8<bf16_t>(vbase + (size_t)key * D + d0, vf);
        float partial = 0.f;
#pragma unroll
        for (int i = 0; i < 8; ++i) partial += qf[i] * kf[i];
#pragma unroll
        for (int off = G / 2; off > 0; off >>= 1) partial += __shfl_xor(partial, off);
        const float score = partial;
        const float mnew = fmaxf(m, score);
        const float corr = (m > -INFINITY) ? __expf(m - mnew) : 0.f;
        const float pw = __expf(score - mnew);
        s = s * corr + pw;
#pragma unroll
        for (int i = 0; i < 8; ++i) o[i] = o[i] * corr + pw * vf[i];
        m = mnew;
      }
    }

    const int part = wid * KPW + kgrp;
    if (lane % G == 0) {
      ms_buf[part * 2] = m;
      ms_buf[part * 2 + 1] = s;
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) o_buf[part * D + d0 + i] = o[i];
    __syncthreads();
    if (threadIdx.x < D) {
      float mstar = -INFINITY;
#pragma unroll
      for (int p = 0; p < NPART; ++p) mstar = fmaxf(mstar, ms_buf[p * 2]);
      float sstar = 0.f, acc = 0.f;
#pragma unroll
      for (int p = 0; p < NPART; ++p) {
        const float mp = ms_buf[p * 2];
        if (mp == -INFINITY) continue;
        const float w = __expf(mp - mstar);
        sstar += ms_buf[p * 2 + 1] * w;
        acc += o_buf[p * D + threadIdx.x] * w;
      }
      const float res = (sstar > 0.f) ? acc / sstar : 0.f;
      // attn_out is [B, H*D] row-major = [B][h][d]
      attn_out[((size_t)b * H + h) * D + threadIdx.x].u = f2bf(res);
    }
    __syncthreads();
  }
}

// ---- lm_head + gumbel-argmax stage -----------------------------------------
DEV void mega_lm_sample(const bf16_t* __restrict__ x, const bf16_t* __restrict__ wlm,
                        const bf16_t* __restrict__ blm, const float* __restrict__ nstats,
                        const bf16_t* __restrict__ nw, const bf16_t* __restrict__ nb,
                        unsigned long long* __restrict__ packed, const MegaCfg cfg,
                        long rng_offset, float* __restrict__ smem) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int M = cfg.B, N = cfg.V, K = cfg.H;
  const int nM = (M + 15) >> 4;
  const int nN = (N + 15) >> 4;
  const int ntiles = nM * nN;
  const int kq = ((K / 32 + MWAVES - 1) / MWAVES) * 32;
  const int k0 = wid * kq;
  const int k1 = min(K, k0 + kq);
  const float inv_nH = 1.f / K;
  const unsigned long long key =
      splitmix64(cfg.seed ^ (0x9e3779b97f4a7c15ull * ((unsigned long long)rng_offset + 1)));
  const bool sample = cfg.inv_temp != 0.f;

  for (int tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int mt = tile / nN;
    const int nt = tile % nN;
    const int arow = min(mt * 16 + (lane & 15), M - 1);
    const int wrow = min(nt * 16 + (lane & 15), N - 1);
    const int k8 = (lane >> 4) * 8;
    const bf16_t* ap = x + (size_t)arow * K + k8;
    const bf16_t* wp = wlm + (size_t)wrow * K + k8;

    const float s1 = nstats[arow * 2];
    const float s2 = nstats[arow * 2 + 1];
    float mu = 0.f, rstd;
    if (cfg.norm_rms) {
      rstd = __frsqrt_rn(s2 * inv_nH + cfg.eps);
    } else {
      mu = s1 * inv_nH;
      rstd = __frsqrt_rn(fmaxf(s2 * inv_nH - mu * mu, 0.f) + cfg.eps);
    }

    f32x4_mk acc = {0.f, 0.f, 0.f, 0.f};
    for (int k = k0; k < k1; k += 32) {
      bf16x8_mk av = *reinterpret_cast<const bf16x8_mk*>(ap + k);
      bf16x8_mk wv = *reinterpret_cast<const bf16x8_mk*>(wp + k);
      bf16x8_mk nwv = *reinterpret_cast<const bf16x8_mk*>(nw + k + k8);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float xv = (bf2f((unsigned short)av[i]) - mu) * rstd * bf2f((unsigned short)nwv[i]);
        if (nb) xv += bf2f(nb[k + k8 + i].u);
        av[i] = (short)f2bf(xv);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, wv, acc, 0, 0, 0);
    }

    float* red = smem;
#pragma unroll
    for (int r = 0; r < 4; ++r) red[(wid * WAVE + lane) * 4 + r] = acc[r];
    __syncthreads();
    if (wid == 0) {
      const int ccol = nt * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int crow = mt * 16 + (lane >> 4) * 4 + r;
        if (crow >= M || ccol >= N) continue;
        float v = red[lane * 4 + r] + red[(WAVE + lane) * 4 + r] + red[(2 * WAVE + lane) * 4 + r] +
                  red[(3 * WAVE + lane) * 4 + r];
        if (blm) v += bf2f(blm[ccol].u);
        // round through bf16: token choice matches the non-mega engine's
        // bf16 logits exactly (generation.py feeds bf16 logits to the sampler)
        v = bf2f(f2bf(v));
        float val;
        if (sample) {
          const float u = rng_uniform(key, (unsigned long long)crow, (unsigned long long)ccol);
          val = v * cfg.inv_temp + (-__logf(-__logf(u)));
        } else {
          val = v;
        }
        const unsigned long long p =
            ((unsigned long long)mk_float_orderable(val) << 32) | (unsigned int)(~(unsigned int)ccol);
        atomicMax(&packed[crow], p);
      }
    }
    __syncthreads();
  }
}

// ---- the megakernel ---------------------------------------------------------
template <int D>
__global__ __launch_bounds__(MBLOCK, 4) void mega_decode_kernel(
    const unsigned long long* __restrict__ ptrs,  // PG_N + L*PW_PER_LAYER entries
    bf16_t* __restrict__ x, bf16_t* __restrict__ x1, bf16_t* __restrict__ qkv_buf,
    bf16_t* __restrict__ attn_buf, bf16_t* __restrict__ act_buf,
    float* __restrict__ stats,  // [2][B][2] (s0 = ln1/lnf, s1 = ln2)
    long* __restrict__ cur_tok, long* __restrict__ out_tokens, bool* __restrict__ finished,
    long* __restrict__ rng_offset, long* __restrict__ step_col, long* __restrict__ cache_idx,
    int* __restrict__ seq_lens, int* __restrict__ pos_ids, const int* __restrict__ key_starts,
    unsigned long long* __restrict__ packed, int* __restrict__ n_fin, MegaCfg cfg) {
  cg::grid_group grid = cg::this_grid();
  __shared__ float smem[4096];  // 16 KB, unioned across stages

  const int B = cfg.B, H = cfg.H;
  float* s0 = stats;
  float* s1 = stats + 2 * B;
  const float inv_H = 1.f / H;
  const float inv_I = 1.f / cfg.I;
  const float* rcos = cfg.pos_kind == 1 ? PTR<float>(ptrs, PG_RCOS) : nullptr;
  const float* rsin = cfg.pos_kind == 1 ? PTR<float>(ptrs, PG_RSIN) : nullptr;

  for (int t = 0; t < cfg.n_tokens; ++t) {
    // ---- embed: x[b,:] = wte[cur_tok[b]] (+ wpe[pos]); row stats -> s0 ----
    for (int b = blockIdx.x; b < B; b += gridDim.x) {
      const bf16_t* wte = PTR<bf16_t>(ptrs, PG_WTE) + (size_t)cur_tok[b] * H;
      const bf16_t* wpe = nullptr;
      if (cfg.pos_kind == 0)
        wpe = PTR<bf16_t>(ptrs, PG_WPE) + (size_t)(pos_ids[b] + cfg.pos_offset) * H;
      float ssum = 0.f, ssq = 0.f;
      for (int i = threadIdx.x; i < H; i += MBLOCK) {
        float v = bf2f(wte[i].u);
        if (wpe) v += bf2f(wpe[i].u);
        const unsigned short vb = f2bf(v);
        x[(size_t)b * H + i].u = vb;
        const float vr = bf2f(vb);
        ssum += vr;
        ssq += vr * vr;
      }
      const float bs = block_sum<MWAVES>(ssum, smem);
      const float bq = block_sum<MWAVES>(ssq, smem + MWAVES);
      if (threadIdx.x == 0) {
        s0[b * 2] = bs;
        s0[b * 2 + 1] = bq;
        s1[b * 2] = 0.f;
        s1[b * 2 + 1] = 0.f;
        packed[b] = 0ull;
      }
    }
    grid.sync();

    const long pos = *cache_idx;  // uniform: set by the previous advance
    bf16_t* xa = x;   // layer input / residual stream
    bf16_t* xb = x1;  // o-proj output stream
    for (int l = 0; l < cfg.L; ++l) {
      const unsigned long long* lp = ptrs + PG_N + l * PW_PER_LAYER;
      // qkv = ln1(xa) @ Wqkv^T + b
      mega_gemm(xa, PTR<bf16_t>(lp, PW_QKV), PTR<bf16_t>(lp, PW_QKV_B), qkv_buf, B, 3 * H, H,
                s0, PTR<bf16_t>(lp, PW_LN1_W), PTR<bf16_t>(lp, PW_LN1_B), cfg.norm_rms, cfg.eps,
                inv_H, 0, nullptr, nullptr, smem);
      grid.sync();
      // attention (writes attn_buf) + zero s0 for the down-stage reuse
      if (blockIdx.x == 0 && threadIdx.x < 2 * B) {
        // safe: s0 consumed by the qkv stage above
        reinterpret_cast<float*>(s0)[threadIdx.x] = 0.f;
      }
      mega_attn<D>(qkv_buf, PTRW<bf16_t>(lp, PW_KC), PTRW<bf16_t>(lp, PW_VC), attn_buf, seq_lens,
                   key_starts, rcos, rsin, pos, cfg, smem);
      grid.sync();
      // xb = xa + attn_buf @ Wo^T + b ; accumulate ln2 stats -> s1
      mega_gemm(attn_buf, PTR<bf16_t>(lp, PW_O), PTR<bf16_t>(lp, PW_O_B), xb, B, H, H, nullptr,
                nullptr, nullptr, 0, 0.f, 0.f, 0, xa, s1, smem);
      grid.sync();
      // act_buf = act( ln2(xb) @ Wfc^T + b )
      mega_gemm(xb, PTR<bf16_t>(lp, PW_FC), PTR<bf16_t>(lp, PW_FC_B), act_buf, B, cfg.I, H, s1,
                PTR<bf16_t>(lp, PW_LN2_W), PTR<bf16_t>(lp, PW_LN2_B), cfg.norm_rms, cfg.eps,
                inv_H, cfg.act, nullptr, nullptr, smem);
      grid.sync();
      // s1 consumed; zero it for the next layer before the down-stage sync
      if (blockIdx.x == 1 % gridDim.x && threadIdx.x < 2 * B) {
        reinterpret_cast<float*>(s1)[threadIdx.x] = 0.f;
      }
      // xa' = xb + act_buf @ Wdown^T + b ; accumulate next ln1 stats -> s0
      mega_gemm(act_buf, PTR<bf16_t>(lp, PW_DOWN), PTR<bf16_t>(lp, PW_DOWN_B), xa, B, H, cfg.I,
                nullptr, nullptr, nullptr, 0, 0.f, inv_I, 0, xb, s0, smem);
      grid.sync();
      // xa now holds the layer output (residual stream continues in xa)
    }

    // ---- lm_head + sampling ----
    mega_lm_sample(xa, PTR<bf16_t>(ptrs, PG_WLM), PTR<bf16_t>(ptrs, PG_WLM_B), s0,
                   PTR<bf16_t>(ptrs, PG_LNF_W), PTR<bf16_t>(ptrs, PG_LNF_B), packed, cfg,
                   *rng_offset, smem);
    grid.sync();

    // ---- advance (block 0), decode_advance semantics ----
    if (blockIdx.x == 0) {
      const long col = *step_col;
      const long new_cache = *cache_idx + 1;
      for (int b = threadIdx.x; b < B; b += MBLOCK) {
        long tok = (long)(~(unsigned int)(packed[b] & 0xffffffffu));
        if (cfg.eos >= 0) {
          if (finished[b]) tok = cfg.pad;
          finished[b] = finished[b] || (tok == cfg.eos);
        }
        if (col >= 0 && col < cfg.max_new) out_tokens[(size_t)b * cfg.max_new + col] = tok;
        cur_tok[b] = tok;
        seq_lens[b] += 1;
        const int ks = key_starts ? key_starts[b] : 0;
        pos_ids[b] = (int)(new_cache - ks);
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        *rng_offset += 1;
        *step_col = col + 1;
        *cache_idx = new_cache;
        int nf = 0;
        if (cfg.eos >= 0)
          for (int b = 0; b < B; ++b) nf += finished[b] ? 1 : 0;
        *n_fin = nf;
      }
    }
    grid.sync();
    if (cfg.eos >= 0 && *n_fin == B) break;
  }
}

}  // namespace

// host-side launcher ----------------------------------------------------------

long mega_decode(const at::Tensor& ptrs, at::Tensor& x, at::Tensor& x1, at::Tensor& qkv_buf,
                 at::Tensor& attn_buf, at::Tensor& act_buf, at::Tensor& stats,
                 at::Tensor& cur_tok, at::Tensor& out_tokens, at::Tensor& finished,
                 at::Tensor& rng_offset, at::Tensor& step_col, at::Tensor& cache_idx,
                 at::Tensor& seq_lens, at::Tensor& pos_ids, const at::Tensor& key_starts,
                 at::Tensor& packed, at::Tensor& n_fin,
                 long B, long H, long I, long V, long L, long heads, long D, long S,
                 long max_new, double eps, long act, long norm_rms, long pos_kind,
                 long pos_offset, long rot, long interleaved, double scale, double temperature,
                 long seed, long eos, long pad, long n_tokens) {
  MegaCfg cfg;
  cfg.B = (int)B;
  cfg.H = (int)H;
  cfg.I = (int)I;
  cfg.V = (int)V;
  cfg.L = (int)L;
  cfg.heads = (int)heads;
  cfg.D = (int)D;
  cfg.S = (int)S;
  cfg.max_new = (int)max_new;
  cfg.eps = (float)eps;
  cfg.act = (int)act;
  cfg.norm_rms = (int)norm_rms;
  cfg.pos_kind = (int)pos_kind;
  cfg.pos_offset = (int)pos_offset;
  cfg.rot = (int)rot;
  cfg.interleaved = (int)interleaved;
  cfg.scale = (float)scale;
  cfg.inv_temp = temperature == 0.0 ? 0.f : (float)(1.0 / temperature);
  cfg.seed = (unsigned long long)seed;
  cfg.eos = eos;
  cfg.pad = pad;
  cfg.n_tokens = (int)n_tokens;

  TORCH_CHECK(H % 32 == 0 && I % 32 == 0, "mega_decode: H and I must be multiples of 32");
  TORCH_CHECK(heads * D == H, "mega_decode: MHA only (heads*D == H)");
  TORCH_CHECK(B <= 256, "mega_decode: B must be <= 256");

  const void* kern = nullptr;
  switch (D) {
    case 64: kern = (const void*)mega_decode_kernel<64>; break;
    case 128: kern = (const void*)mega_decode_kernel<128>; break;
    default: TORCH_CHECK(false, "mega_decode: head dim must be 64 or 128");
  }
  int dev = 0;
  TORCH_CHECK(hipGetDevice(&dev) == hipSuccess);
  hipDeviceProp_t prop;
  TORCH_CHECK(hipGetDeviceProperties(&prop, dev) == hipSuccess);
  TORCH_CHECK(prop.cooperativeLaunch, "mega_decode: device lacks cooperative launch");
  int maxActive = 0;
  TORCH_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(&maxActive, kern, MBLOCK, 0) ==
              hipSuccess);
  TORCH_CHECK(maxActive >= 1, "mega_decode: kernel cannot be co-resident (occupancy 0)");
  // fill the chip, capped at guaranteed co-residency (deadlock-safe)
  const int grid = min(maxActive, 4) * prop.multiProcessorCount;

  auto stream = c10::hip::getCurrentHIPStream();
  auto pp = reinterpret_cast<const unsigned long long*>(ptrs.data_ptr<long>());
  auto xb_ = reinterpret_cast<bf16_t*>(x.data_ptr());
  auto x1_ = reinterpret_cast<bf16_t*>(x1.data_ptr());
  auto qk_ = reinterpret_cast<bf16_t*>(qkv_buf.data_ptr());
  auto at_ = reinterpret_cast<bf16_t*>(attn_buf.data_ptr());
  auto ac_ = reinterpret_cast<bf16_t*>(act_buf.data_ptr());
  auto st_ = stats.data_ptr<float>();
  auto ct_ = cur_tok.data_ptr<long>();
  auto ot_ = out_tokens.data_ptr<long>();
  auto fin_ = finished.data_ptr<bool>();
  auto ro_ = rng_offset.data_ptr<long>();
  auto sc_ = step_col.data_ptr<long>();
  auto ci_ = cache_idx.data_ptr<long>();
  auto sl_ = seq_lens.data_ptr<int>();
  auto pi_ = pos_ids.data_ptr<int>();
  auto ks_ = key_starts.defined() ? key_starts.data_ptr<int>() : (int*)nullptr;
  auto pk_ = reinterpret_cast<unsigned long long*>(packed.data_ptr<long>());
  auto nf_ = n_fin.data_ptr<int>();

  void* args[] = {&pp, &xb_, &x1_, &qk_, &at_, &ac_, &st_, &ct_, &ot_, &fin_, &ro_, &sc_,
                  &ci_, &sl_, &pi_, &ks_, &pk_, &nf_, &cfg};
  hipError_t e = hipLaunchCooperativeKernel(kern, dim3(grid), dim3(MBLOCK), args, 0,
                                            stream.stream());
  TORCH_CHECK(e == hipSuccess, "mega_decode launch failed: ", hipGetErrorString(e));
  HIP_CHECK_LAST();
  return grid;
}
