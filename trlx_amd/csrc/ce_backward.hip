// Backward of the fused lm_head+logprobs training op (SURVEY.md K1+K5):
// dlogits[n,v] = dlp[n] * (1{v==labels[n]} - softmax(logits)[n,v]),
// with logits RECOMPUTED tile-by-tile from (hidden, weight) and the saved
// per-row logsumexp — the forward (lm_logprobs_v2) never materialized them.
// The caller contracts dlogits with W / hidden via hipBLASLt for dh and dW.
//
// Same 256x256x64 minimum-2-phase MFMA pipeline + 3-bit LDS swizzle as
// lm_logprobs_v2.hip (see there for the schedule derivation); the epilogue
// writes the bf16 dlogits tile straight from the accumulators (adjacent
// lanes hold adjacent columns -> 32 B-coalesced stores).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BM3 = 256;
constexpr int BN3 = 256;
constexpr int BK3 = 64;
constexpr int BLOCK3 = 512;

typedef __attribute__((ext_vector_type(8))) short bf16x8_ce;
typedef __attribute__((ext_vector_type(4))) float f32x4_ce;

constexpr int TILE_B = BM3 * BK3 * 2;  // 32 KiB per A/B buffer
constexpr int ROW_B = BK3 * 2;

__device__ __forceinline__ int swz3(int byte_off) {
  return byte_off ^ (((byte_off >> 7) & 7) << 4);
}

__device__ __forceinline__ void stage_half(const bf16_t* __restrict__ src, int stride, int gr0,
                                           int max_row, int k0, char* lds_base, int dst_row0,
                                           int tid) {
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int off = c * 4096 + tid * 8;
    const int trow = off >> 6;
    const int tcol = off & 63;
    const int dst_row = dst_row0 + trow;
    const int src_col = tcol ^ ((dst_row & 7) << 3);
    const int grow = min(gr0 + trow, max_row);
    const int dst_byte = dst_row * ROW_B + tcol * 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)(src + (size_t)grow * stride + k0 +
                                                                src_col),
        (__attribute__((address_space(3))) unsigned int*)(lds_base + dst_byte), 16, 0, 0);
  }
}

__global__ __launch_bounds__(BLOCK3) void ce_dlogits_kernel(
    const bf16_t* __restrict__ hidden, const bf16_t* __restrict__ weight,
    const long* __restrict__ labels, const float* __restrict__ lse,
    const float* __restrict__ dlp, bf16_t* __restrict__ dlogits, int N, int H, int V, int nV,
    int nM) {
  extern __shared__ char smem[];
  char* a_base = smem;
  char* b_base = smem + 2 * TILE_B;
  long* lab_s = reinterpret_cast<long*>(smem + 4 * TILE_B);           // [256]
  float* row_s = reinterpret_cast<float*>(smem + 4 * TILE_B + 2048);  // [256][2]: lse, dlp

  const int nwg = nV * nM;
  int wg = blockIdx.y * gridDim.x + blockIdx.x;
  {
    const int xcd = wg % 8;
    const int q8 = nwg / 8, r8 = nwg % 8;
    wg = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + wg / 8;
  }
  const int vt = wg / nM;
  const int mt = wg % nM;
  const int row0 = mt * BM3;
  const int col0 = vt * BN3;

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wave_m = wid >> 2;
  const int wave_n = wid & 3;

  for (int i = tid; i < BM3; i += BLOCK3) {
    const int n = row0 + i;
    lab_s[i] = (n < N) ? labels[n] : -1;
    row_s[i * 2] = (n < N) ? lse[n] : 0.f;
    row_s[i * 2 + 1] = (n < N) ? dlp[n] : 0.f;
  }

  f32x4_ce acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nK = H / BK3;
  const int max_ar = N - 1;
  const int max_br = V - 1;

  stage_half(weight, H, col0, max_br, 0, b_base, 0, tid);
  stage_half(weight, H, col0 + 128, max_br, 0, b_base, 128, tid);
  stage_half(hidden, H, row0, max_ar, 0, a_base, 0, tid);
  stage_half(hidden, H, row0 + 128, max_ar, 0, a_base, 128, tid);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  const int frow = lane & 15;
  const int fk8 = (lane >> 4) * 8;
  bf16x8_ce bfrag[4];
  bf16x8_ce afrag[8];

  for (int m = 0; m < nK; ++m) {
    const int p = m & 1;
    char* a_buf = a_base + p * TILE_B;
    char* b_buf = b_base + p * TILE_B;
    if (m + 1 < nK) {
      const int k0 = (m + 1) * BK3;
      stage_half(weight, H, col0, max_br, k0, b_base + (p ^ 1) * TILE_B, 0, tid);
      stage_half(weight, H, col0 + 128, max_br, k0, b_base + (p ^ 1) * TILE_B, 128, tid);
      stage_half(hidden, H, row0, max_ar, k0, a_base + (p ^ 1) * TILE_B, 0, tid);
      stage_half(hidden, H, row0 + 128, max_ar, k0, a_base + (p ^ 1) * TILE_B, 128, tid);
    }
#pragma unroll
    for (int k2 = 0; k2 < 2; ++k2) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int brow = wave_n * 64 + j * 16 + frow;
        bfrag[j] = *reinterpret_cast<const bf16x8_ce*>(
            b_buf + swz3(brow * ROW_B + (k2 * 32 + fk8) * 2));
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int arow = wave_m * 128 + i * 16 + frow;
        afrag[i] = *reinterpret_cast<const bf16x8_ce*>(
            a_buf + swz3(arow * ROW_B + (k2 * 32 + fk8) * 2));
      }
      asm volatile("s_waitcnt lgkmcnt(0)");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag[i], bfrag[j], acc[i][j],
                                                              0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: dlogits = dlp * (1{label} - exp(logit - lse)), straight from acc
  const int cgrp = lane >> 4;
  const int ccol0 = wave_n * 64 + (lane & 15);
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int trow = wave_m * 128 + i * 16 + cgrp * 4 + r;
      const int n = row0 + trow;
      if (n >= N) continue;
      const float l = row_s[trow * 2];
      const float g = row_s[trow * 2 + 1];
      const long lab = lab_s[trow];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int v = col0 + ccol0 + j * 16;
        if (v >= V) continue;
        float p = __expf(acc[i][j][r] - l);
        float d = g * ((v == lab ? 1.f : 0.f) - p);
        ScalarIO<bf16_t>::store(dlogits + (size_t)n * V + v, d);
      }
    }
  }
}

constexpr int CE_LDS = 4 * TILE_B + 2048 + BM3 * 2 * (int)sizeof(float);

}  // namespace

at::Tensor ce_dlogits(const at::Tensor& hidden, const at::Tensor& weight,
                      const at::Tensor& labels, const at::Tensor& lse, const at::Tensor& dlp) {
  TORCH_CHECK(hidden.is_cuda() && hidden.dtype() == at::kBFloat16 && hidden.dim() == 2 &&
              hidden.is_contiguous());
  TORCH_CHECK(weight.dtype() == at::kBFloat16 && weight.is_contiguous());
  const int N = hidden.size(0);
  const int H = hidden.size(1);
  const int V = weight.size(0);
  TORCH_CHECK(weight.size(1) == H && labels.numel() == N && lse.numel() == N &&
              dlp.numel() == N);
  TORCH_CHECK(H % BK3 == 0, "ce_dlogits: hidden size must be a multiple of 64");
  auto dlogits = at::empty({N, V}, hidden.options());
  if (N == 0) return dlogits;
  const int nV = (V + BN3 - 1) / BN3;
  const int nM = (N + BM3 - 1) / BM3;
  auto stream = c10::hip::getCurrentHIPStream();
  static bool configured = false;
  if (!configured) {
    (void)hipFuncSetAttribute(reinterpret_cast<const void*>(&ce_dlogits_kernel),
                              hipFuncAttributeMaxDynamicSharedMemorySize, CE_LDS);
    configured = true;
  }
  dim3 grid(nV, nM);
  ce_dlogits_kernel<<<grid, BLOCK3, CE_LDS, stream>>>(
      reinterpret_cast<const bf16_t*>(hidden.data_ptr()),
      reinterpret_cast<const bf16_t*>(weight.data_ptr()), labels.data_ptr<long>(),
      lse.data_ptr<float>(), dlp.contiguous().data_ptr<float>(),
      reinterpret_cast<bf16_t*>(dlogits.data_ptr()), N, H, V, nV, nM);
  HIP_CHECK_LAST();
  return dlogits;
}
