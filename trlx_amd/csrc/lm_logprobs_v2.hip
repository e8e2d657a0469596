// Fused LM-head GEMM + online logsumexp + label gather, 8-phase pipelined
// (SURVEY.md K1+K5; supersedes the single-buffered tile kernel in
// lm_logprobs.hip for H % 64 == 0).
//
// Structure (CDNA4 guide "minimum 2-phase" T3 recipe + T1/T2/T5):
//   - 256x256 output tile, BK=64, 8 waves (2M x 4N), per-wave C = 128x64
//     held in 128 accumulator VGPRs (the [N, V] logits never exist).
//   - Double-buffered A/B LDS (128 KiB dynamic) staged with 16-byte
//     global_load_lds.  Per K-tile: {issue next tile's 8 loads; ds_read
//     k2-half fragments; lgkmcnt(0); setprio(1); 32 MFMA; setprio(0)} x2,
//     then ONE vmcnt(0)+barrier.  Intra-tile, waves only READ the shared
//     buffer — no hazard — so they skew freely and one wave's ds_reads
//     overlap the others' MFMAs.  (A fully barriered 8-phase variant
//     measured 465 TF with SQ_WAIT_ANY ~= MFMA cycles: CU-wide lockstep
//     serialized the LDS and MFMA pipes.)
//   - 3-bit XOR LDS swizzle (16B-chunk ^= row&7): applied as a PRE-SWIZZLED
//     GLOBAL source column on the store side — global_load_lds ignores
//     per-lane LDS addresses (lowered readfirstlane->M0, hardware writes
//     M0 + laneId*16) — and as swizzled ds_read addresses.  The linear
//     [row][64] layout is 16-way bank conflicted (row stride 128 B);
//     measured 44M -> 6M SQ_LDS_BANK_CONFLICT.
//   - XCD-aware bijective workgroup remap (8 XCDs, each a private L2),
//     mt-major so co-resident workgroups share a B (weight) tile.
//   - Epilogue: acc -> LDS fp32 tile (two 128-row halves, padded stride
//     257), then per-row two-pass scan (max, then sum-exp with 4
//     independent accumulator chains) by 4 threads x 64 cols per row.
//     The register-direct epilogue (per-fragment shfl_xor chains +
//     predicated stores) cost 432 us of an 870 us kernel; this one ~75 us.
//
// Measured (N=5248, V=50257, H=768, MI355X): 511 us = 792 TF vs 685 us for
// hipBLASLt GEMM + fused logprob-gather (1.34x); N=1312: 157 vs 221 us
// (1.41x).  Numerics exact vs fp32 torch reference (max err 2.9e-6).
//
// Shapes: hidden [N, H] bf16 (H % 64 == 0), weight [V, H] bf16 row-major
// (NT GEMM), labels [N] i64 -> out [N] f32.  Inference-only.
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BM2 = 256;   // rows (tokens) per workgroup
constexpr int BN2 = 256;   // vocab columns per workgroup
constexpr int BK2 = 64;    // K-step
constexpr int BLOCK2 = 512;  // 8 waves: 2 (M) x 4 (N)

typedef __attribute__((ext_vector_type(8))) short bf16x8_v2;
typedef __attribute__((ext_vector_type(4))) float f32x4_v2;

// byte layout of one A or B buffer: [256 rows][64 cols] bf16 = 32 KiB
constexpr int TILE_BYTES = BM2 * BK2 * 2;      // 32768
constexpr int HALF_BYTES = TILE_BYTES / 2;     // 16384 (128 rows)
constexpr int ROW_BYTES = BK2 * 2;             // 128

__device__ __forceinline__ int swz(int byte_off) {
  // 3-bit XOR swizzle: 16B-chunk index (byte bits 4-6) ^= row low bits
  // (byte bits 7-9).  Fragment ds_reads walk 16 consecutive rows at a fixed
  // column chunk (16-way bank conflicted in a linear [row][64] layout — row
  // stride 128 B = all 32 banks apart); the XOR spreads them across all 8
  // chunk positions -> 2 lanes/bank, which CDNA4 serves conflict-free
  // (guide: 2-way is 1.02x).  Stronger than st_16x32's single-bit XOR
  // (8-way): measured 44M SQ_LDS_BANK_CONFLICT with 1-bit vs the MFMA
  // cycles it gated.
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

// cooperative issue of one half-tile (128 rows x 64 cols bf16) = 2
// global_load_lds instructions; gr0 = first global row, rows clamped to the
// last valid row (duplicates are masked in the epilogue).
//
// The hardware derives each lane's LDS byte as M0 + laneId*16 (the compiler
// lowers the LDS operand via readfirstlane -> M0; per-lane destination bits
// are DISCARDED), so the st_16x32 swizzle cannot be applied on the store
// side.  Instead the GLOBAL source column is pre-swizzled: the lane whose
// (implicit, linear) LDS slot l should hold the element that a swizzled
// ds_read at swz(b)=l expects loads that element directly —
// src_col = tcol ^ ((dst_row & 7) * 8)  (element-granular form of swz).
__device__ __forceinline__ void issue_half(const bf16_t* __restrict__ src, int src_stride,
                                           int gr0, int max_row, int k0, char* lds_base,
                                           int dst_row0, int tid) {
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int off = c * 4096 + tid * 8;  // element offset within [128][64]
    const int trow = off >> 6;
    const int tcol = off & 63;
    const int dst_row = dst_row0 + trow;
    const int src_col = tcol ^ ((dst_row & 7) << 3);  // pre-swizzle
    const int grow = min(gr0 + trow, max_row);
    const int dst_byte = dst_row * ROW_BYTES + tcol * 2;  // linear (hw layout)
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)(src + (size_t)grow * src_stride +
                                                                k0 + src_col),
        (__attribute__((address_space(3))) unsigned int*)(lds_base + dst_byte), 16, 0, 0);
  }
}

struct MSv2 {
  float m, s;
};

// mode: perf-probe ablation bits (0 = normal). 1: skip epilogue, 2: skip
// ds_reads, 4: skip MFMAs — numerically invalid, used to bisect where the
// kernel's time goes on real hardware.
template <bool NOWAIT>
__global__ __launch_bounds__(BLOCK2) void lm_logprobs_v2_kernel(
    const bf16_t* __restrict__ hidden, const bf16_t* __restrict__ weight,
    float* __restrict__ partials,  // [nV][N][2]
    float* __restrict__ label_logit, const long* __restrict__ labels, int N, int H, int V,
    int nV, int nM, int mode) {
  extern __shared__ char smem[];
  // K-loop:   [ A buf0 | A buf1 | B buf0 | B buf1 | ... | labels 2K ]
  // epilogue: [ c_lds fp32 [128][260] | cpart [128][4][2] | labels 2K ]
  // (c_lds overwrites the A/B buffers — time-disjoint; lab_s sits above both)
  char* a_base = smem;
  char* b_base = smem + 2 * TILE_BYTES;
  float* c_lds = reinterpret_cast<float*>(smem);
  float* cpart = reinterpret_cast<float*>(smem + 131584);   // [128][4][2]
  long* lab_s = reinterpret_cast<long*>(smem + 135680);  // [256]

  // XCD-aware bijective remap of the flattened workgroup id (8 XCDs)
  const int nwg = nV * nM;
  int wg = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int xcd = wg % 8;
    const int q8 = nwg / 8, r8 = nwg % 8;
    wg = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + wg / 8;
  }
  // mt-major: consecutive workgroup ids (contiguous per XCD after the remap)
  // share one B (weight) tile and walk the M tiles, so each XCD's L2 holds
  // its hot B tile + the full A K-slice; vt-major left every resident WG
  // streaming a distinct 384 KB B tile (measured: vmem-wait dominated).
  const int vt = wg / nM;
  const int mt = wg % nM;
  const int row0 = mt * BM2;
  const int col0 = vt * BN2;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wave_m = wid >> 2;          // 0..1 -> rows wave_m*128
  const int wave_n = wid & 3;           // 0..3 -> cols wave_n*64

  // stage labels for this row tile
  for (int i = tid; i < BM2; i += BLOCK2) {
    const int n = row0 + i;
    lab_s[i] = (n < N) ? labels[n] : -1;
  }

  f32x4_v2 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nK = H / BK2;
  const int max_arow = N - 1;
  const int max_brow = V - 1;

  // ---- prologue: stage K-tile 0, drain, sync
  issue_half(weight, H, col0, max_brow, 0, b_base, 0, tid);
  issue_half(weight, H, col0 + 128, max_brow, 0, b_base, 128, tid);
  issue_half(hidden, H, row0, max_arow, 0, a_base, 0, tid);
  issue_half(hidden, H, row0 + 128, max_arow, 0, a_base, 128, tid);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  // fragment addressing (byte offsets, swizzled at use)
  const int frow = lane & 15;           // row/col within a 16-wide fragment
  const int fk8 = (lane >> 4) * 8;      // k-slice start within a 32-wide K block

  bf16x8_v2 bfrag[4];  // [j] — one half-K (32) at a time
  bf16x8_v2 afrag[8];  // [i]
#pragma unroll
  for (int j = 0; j < 4; ++j) bfrag[j] = bf16x8_v2{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
  for (int i = 0; i < 8; ++i) afrag[i] = bf16x8_v2{0, 0, 0, 0, 0, 0, 0, 0};

  // K-loop, minimum-2-phase shape (guide T3 recipe): per K-tile
  //   {issue next tile's 8 loads; ds_read k2=0 frags; lgkm; MFMA*32;
  //    ds_read k2=1 frags; lgkm; MFMA*32; vmcnt(0); barrier; flip}
  // ONE barrier per K-tile: intra-tile, waves only READ the same buffer
  // (no hazard), so they skew freely and each wave's ds_reads overlap the
  // other waves' MFMAs on the SIMD.  The earlier 8-barriers-per-K-tile
  // lockstep serialized the LDS and MFMA pipes CU-wide (measured 465 TF,
  // SQ_WAIT_ANY ~= MFMA cycles); this shape measured +40% (guide m230/m248).
  for (int m = 0; m < nK; ++m) {
    const int p = m & 1;
    char* a_buf = a_base + p * TILE_BYTES;
    char* b_buf = b_base + p * TILE_BYTES;
    if (m + 1 < nK) {
      char* a_nxt = a_base + (p ^ 1) * TILE_BYTES;
      char* b_nxt = b_base + (p ^ 1) * TILE_BYTES;
      const int k0 = (m + 1) * BK2;
      issue_half(weight, H, col0, max_brow, k0, b_nxt, 0, tid);
      issue_half(weight, H, col0 + 128, max_brow, k0, b_nxt, 128, tid);
      issue_half(hidden, H, row0, max_arow, k0, a_nxt, 0, tid);
      issue_half(hidden, H, row0 + 128, max_arow, k0, a_nxt, 128, tid);
    }
#pragma unroll
    for (int k2 = 0; k2 < 2; ++k2) {
      if (!(mode & 2)) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int brow = wave_n * 64 + j * 16 + frow;
          bfrag[j] = *reinterpret_cast<const bf16x8_v2*>(
              b_buf + swz(brow * ROW_BYTES + (k2 * 32 + fk8) * 2));
        }
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const int arow = wave_m * 128 + i * 16 + frow;
          afrag[i] = *reinterpret_cast<const bf16x8_v2*>(
              a_buf + swz(arow * ROW_BYTES + (k2 * 32 + fk8) * 2));
        }
        asm volatile("s_waitcnt lgkmcnt(0)");
      }
      if (!(mode & 4)) {
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[i], bfrag[j], acc[i][j],
                                                                0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
    if constexpr (!NOWAIT) {
      asm volatile("s_waitcnt vmcnt(0)");
    }
    __builtin_amdgcn_s_barrier();
  }

  if (mode & 1) {  // probe: keep acc observable, skip the epilogue
    if (tid == 0 && mode & 8) partials[0] = acc[0][0][0];
    return;
  }
  // ---- epilogue: acc -> LDS transpose (two row-halves), then a
  // cooperative per-row scan.  The register-direct version (per-row 4-step
  // shfl_xor chains + a predicated global store per fragment value) measured
  // 432 us of the 870 us kernel at N=5248 — half the runtime; this layout
  // costs ~2 LDS round-trips of the C tile and scans rows with plain
  // contiguous reads (4 threads x 64 cols per row).
  const int n_valid_cols = V - col0;    // may exceed 256 (then all valid)
  constexpr int CPAD = 257;             // fp32 row stride: 257 % 32 == 1, so
                                        // row-per-lane reads walk all banks
  const int row_in_half = tid & 127;    // one row per lane across 2 waves-of-rows
  const int qc0 = (tid >> 7) * 64;      // this thread's column span
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    __builtin_amdgcn_s_barrier();
    if (wave_m == half) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int rr = i * 16 + (lane >> 4) * 4 + r;
            const int cc = wave_n * 64 + j * 16 + (lane & 15);
            c_lds[rr * CPAD + cc] = acc[i][j][r];
          }
    }
    __builtin_amdgcn_s_barrier();
    const int tile_row = half * 128 + row_in_half;
    const int n = row0 + tile_row;
    const long lab = lab_s[tile_row];
    const float* row = c_lds + row_in_half * CPAD;
    const int cend = min(qc0 + 64, n_valid_cols);
    // pass 1: max (4 independent chains) + label gather
    float mx0 = -INFINITY, mx1 = -INFINITY, mx2 = -INFINITY, mx3 = -INFINITY;
    for (int c = qc0; c + 4 <= cend; c += 4) {
      mx0 = fmaxf(mx0, row[c]);
      mx1 = fmaxf(mx1, row[c + 1]);
      mx2 = fmaxf(mx2, row[c + 2]);
      mx3 = fmaxf(mx3, row[c + 3]);
    }
    for (int c = qc0 + ((cend - qc0) & ~3); c < cend; ++c) mx0 = fmaxf(mx0, row[c]);
    const float mx = fmaxf(fmaxf(mx0, mx1), fmaxf(mx2, mx3));
    if (lab >= col0 + qc0 && lab < col0 + cend && n < N)
      label_logit[n] = row[lab - col0];
    // pass 2: sum exp(v - mx), 4 independent accumulator chains
    float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
    for (int c = qc0; c + 4 <= cend; c += 4) {
      s0 += __expf(row[c] - mx);
      s1 += __expf(row[c + 1] - mx);
      s2 += __expf(row[c + 2] - mx);
      s3 += __expf(row[c + 3] - mx);
    }
    for (int c = qc0 + ((cend - qc0) & ~3); c < cend; ++c) s0 += __expf(row[c] - mx);
    MSv2 ms{cend > qc0 ? mx : -INFINITY, (s0 + s1) + (s2 + s3)};
    // combine the 4 spans per row (threads tid, tid+128, ... — different
    // waves, so a block barrier orders the exchange)
    cpart[(row_in_half * 4 + (tid >> 7)) * 2] = ms.m;
    cpart[(row_in_half * 4 + (tid >> 7)) * 2 + 1] = ms.s;
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)");
    if (tid < 128 && n < N) {
      MSv2 t{-INFINITY, 0.f};
#pragma unroll
      for (int q4 = 0; q4 < 4; ++q4) {
        MSv2 o{cpart[(row_in_half * 4 + q4) * 2], cpart[(row_in_half * 4 + q4) * 2 + 1]};
        if (o.m > t.m) {
          t.s = (t.m > -INFINITY ? t.s * __expf(t.m - o.m) : 0.f) + o.s;
          t.m = o.m;
        } else if (o.m > -INFINITY) {
          t.s += o.s * __expf(o.m - t.m);
        }
      }
      float* dst = partials + ((size_t)vt * N + n) * 2;
      dst[0] = t.m;
      dst[1] = t.s;
    }
  }
}

__global__ void lm_logprobs_v2_reduce(const float* __restrict__ partials,
                                      const float* __restrict__ label_logit,
                                      float* __restrict__ out, float* __restrict__ lse_out,
                                      int N, int nV) {
  const int wpb = blockDim.x / WAVE;
  const int n = blockIdx.x * wpb + threadIdx.x / WAVE;
  if (n >= N) return;
  const int lane = threadIdx.x % WAVE;
  MSv2 ms{-INFINITY, 0.f};
  for (int t = lane; t < nV; t += WAVE) {
    const float* p = partials + ((size_t)t * N + n) * 2;
    MSv2 o{p[0], p[1]};
    if (o.m > ms.m) {
      ms.s = (ms.m > -INFINITY ? ms.s * __expf(ms.m - o.m) : 0.f) + o.s;
      ms.m = o.m;
    } else if (o.m > -INFINITY) {
      ms.s += o.s * __expf(o.m - ms.m);
    }
  }
#pragma unroll
  for (int d = 1; d < WAVE; d <<= 1) {
    MSv2 o;
    o.m = __shfl_xor(ms.m, d, WAVE);
    o.s = __shfl_xor(ms.s, d, WAVE);
    if (o.m > ms.m) {
      ms.s = (ms.m > -INFINITY ? ms.s * __expf(ms.m - o.m) : 0.f) + o.s;
      ms.m = o.m;
    } else if (o.m > -INFINITY) {
      ms.s += o.s * __expf(o.m - ms.m);
    }
  }
  if (lane == 0) {
    const float lse = ms.m + logf(ms.s);
    out[n] = label_logit[n] - lse;
    if (lse_out) lse_out[n] = lse;
  }
}

constexpr int V2_LDS = 4 * TILE_BYTES + 8192 + BM2 * (int)sizeof(long);  // 139264

}  // namespace

std::vector<at::Tensor> lm_logprobs_v2_with_lse(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels, bool want_lse);

at::Tensor lm_logprobs_v2(const at::Tensor& hidden, const at::Tensor& weight,
                          const at::Tensor& labels) {
  return lm_logprobs_v2_with_lse(hidden, weight, labels, false)[0];
}

std::vector<at::Tensor> lm_logprobs_v2_with_lse(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels, bool want_lse) {
  const int mode = [] {
    const char* e = getenv("TRLX_AMD_LMLP_MODE");
    return e ? atoi(e) : 0;
  }();
  TORCH_CHECK(hidden.is_cuda() && hidden.dtype() == at::kBFloat16 && hidden.dim() == 2 &&
              hidden.is_contiguous());
  TORCH_CHECK(weight.dtype() == at::kBFloat16 && weight.is_contiguous());
  TORCH_CHECK(labels.dtype() == at::kLong && labels.is_contiguous());
  const int N = hidden.size(0);
  const int H = hidden.size(1);
  const int V = weight.size(0);
  TORCH_CHECK(weight.size(1) == H && labels.numel() == N);
  TORCH_CHECK(H % BK2 == 0, "lm_logprobs_v2: hidden size must be a multiple of 64");
  auto out = at::empty({N}, hidden.options().dtype(at::kFloat));
  auto lse = want_lse ? at::empty({N}, hidden.options().dtype(at::kFloat))
                      : at::empty({0}, hidden.options().dtype(at::kFloat));
  if (N == 0) return {out, lse};
  const int nV = (V + BN2 - 1) / BN2;
  const int nM = (N + BM2 - 1) / BM2;
  auto partials = at::empty({nV, (long)N, 2}, hidden.options().dtype(at::kFloat));
  auto label_logit = at::zeros({N}, hidden.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  const bool nowait = [] {
    const char* e = getenv("TRLX_AMD_LMLP_NOWAIT");
    return e && e[0] == '1';
  }();
  static bool lds_configured = false;
  if (!lds_configured) {
    (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&lm_logprobs_v2_kernel<false>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, V2_LDS);
    (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&lm_logprobs_v2_kernel<true>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, V2_LDS);
    lds_configured = true;
  }
  dim3 grid(nV, nM);
  if (nowait)
    lm_logprobs_v2_kernel<true><<<grid, BLOCK2, V2_LDS, stream>>>(
        reinterpret_cast<const bf16_t*>(hidden.data_ptr()),
        reinterpret_cast<const bf16_t*>(weight.data_ptr()), partials.data_ptr<float>(),
        label_logit.data_ptr<float>(), labels.data_ptr<long>(), N, H, V, nV, nM, mode);
  else
    lm_logprobs_v2_kernel<false><<<grid, BLOCK2, V2_LDS, stream>>>(
        reinterpret_cast<const bf16_t*>(hidden.data_ptr()),
        reinterpret_cast<const bf16_t*>(weight.data_ptr()), partials.data_ptr<float>(),
        label_logit.data_ptr<float>(), labels.data_ptr<long>(), N, H, V, nV, nM, mode);
  const int wpb = 256 / WAVE;
  lm_logprobs_v2_reduce<<<(N + wpb - 1) / wpb, 256, 0, stream>>>(
      partials.data_ptr<float>(), label_logit.data_ptr<float>(), out.data_ptr<float>(),
      want_lse ? lse.data_ptr<float>() : nullptr, N, nV);
  HIP_CHECK_LAST();
  return {out, lse};
}
