// Skinny-M streaming GEMM for the decode hot path:
// C[M,N] = A[M,K] @ W[N,K]^T (+bias, + optional fused activation), M <= 256.
//
// During KV-cached decode every projection is a [B, hidden] x [out, hidden]
// GEMM with B ~ 128: hipBLASLt's tiles run it at ~0.4 TB/s effective weight
// bandwidth (profile r01: ~9.3 us in-graph for a 3.5 MB weight read; the
// four per-layer projections ARE the decode time).  Weight-streaming shape:
//
//   - ONE 16x16 output tile per WAVE via an mfma_16x16x32_bf16 chain:
//     lane l holds A row (l&15) / W row (l&15) at k-slice (l>>4)*8.  No
//     LDS, no barriers, ~40 VGPRs -> high occupancy.
//   - 4-deep software-pipelined register loads: the first cut issued 2
//     loads per 32-K step back-to-back with their MFMA (measured 268 GB/s —
//     latency-bound at ~4 waves/CU); keeping 8 independent loads in flight
//     amortizes the ~600-cycle memory latency.
//   - split-K when the tile count is too small to fill 256 CUs (e.g.
//     down_proj [128, 768, 3072]: 384 waves): S K-slices write fp32 slabs
//     [S, M, N], a tiny epilogue kernel sums + bias + activation + casts.
//   - nt-major within a block: the 4 waves of a block compute 4 consecutive
//     M-tiles of the SAME W panel (one weight fetch per block); A (tiny)
//     stays L2-resident.
//   - bias + activation (gelu-erf / gelu-tanh / relu / silu) fused: kills
//     the separate per-projection elementwise kernel (gelu was 3.4% of
//     cycle kernels, all in decode).
//
// Inference-only (decode runs under no_grad; training uses hipBLASLt).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_sk;
typedef __attribute__((ext_vector_type(4))) float f32x4_sk;

constexpr int SK_BLOCK = 256;  // 4 waves, 4 tiles per block

__device__ __forceinline__ float sk_act(float x, int act) {
  switch (act) {
    case 1:  // gelu (erf form, HF "gelu")
      return 0.5f * x * (1.f + erff(x * 0.70710678118654752f));
    case 2: {  // gelu_new / gelu_pytorch_tanh
      const float c = 0.797884560802865f;  // sqrt(2/pi)
      return 0.5f * x * (1.f + tanhf(c * (x + 0.044715f * x * x * x)));
    }
    case 3:  // relu
      return fmaxf(x, 0.f);
    case 4:  // silu
      return x / (1.f + __expf(-x));
    default:
      return x;
  }
}

// SPLIT: C is fp32 slabs [S, M, N] (epilogue applied separately);
// !SPLIT: C is bf16 [M, N] with bias+act fused.
template <bool SPLIT>
__global__ __launch_bounds__(SK_BLOCK) void skinny_gemm_kernel(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ W,
    const bf16_t* __restrict__ bias, void* __restrict__ Cout, int M, int N, int K, int act,
    int nsplit) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nM = (M + 15) >> 4;
  const int nN = (N + 15) >> 4;
  int tile = blockIdx.x * 4 + wid;
  const int ntile2 = nM * nN;
  if (tile >= ntile2 * nsplit) return;
  const int ks = tile / ntile2;  // K-split index
  tile -= ks * ntile2;
  // nt-major: the block's 4 waves walk M-tiles of one W panel
  const int nt = tile / nM;
  const int mt = tile % nM;

  const int arow = min(mt * 16 + (lane & 15), M - 1);
  const int wrow = min(nt * 16 + (lane & 15), N - 1);
  const int k8 = (lane >> 4) * 8;
  // this split's K range (multiples of 32; last split takes the remainder)
  const int kchunk = ((K / 32) / nsplit) * 32;
  const int k0 = ks * kchunk;
  const int k1 = (ks == nsplit - 1) ? K : k0 + kchunk;
  const bf16_t* ap = A + (size_t)arow * K + k8;
  const bf16_t* wp = W + (size_t)wrow * K + k8;

  f32x4_sk acc = {0.f, 0.f, 0.f, 0.f};
  int k = k0;
  // 4-deep register pipeline (8 loads in flight)
  if (k + 128 <= k1) {
    bf16x8_sk a0 = *reinterpret_cast<const bf16x8_sk*>(ap + k);
    bf16x8_sk b0 = *reinterpret_cast<const bf16x8_sk*>(wp + k);
    bf16x8_sk a1 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 32);
    bf16x8_sk b1 = *reinterpret_cast<const bf16x8_sk*>(wp + k + 32);
    bf16x8_sk a2 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 64);
    bf16x8_sk b2 = *reinterpret_cast<const bf16x8_sk*>(wp + k + 64);
    bf16x8_sk a3 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 96);
    bf16x8_sk b3 = *reinterpret_cast<const bf16x8_sk*>(wp + k + 96);
    for (k += 128; k + 128 <= k1; k += 128) {
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
      a0 = *reinterpret_cast<const bf16x8_sk*>(ap + k);
      b0 = *reinterpret_cast<const bf16x8_sk*>(wp + k);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
      a1 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 32);
      b1 = *reinterpret_cast<const bf16x8_sk*>(wp + k + 32);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, acc, 0, 0, 0);
      a2 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 64);
      b2 = *reinterpret_cast<const bf16x8_sk*>(wp + k + 64);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, b3, acc, 0, 0, 0);
      a3 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 96);
      b3 = *reinterpret_cast<const bf16x8_sk*>(wp + k + 96);
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, b2, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, b3, acc, 0, 0, 0);
  }
  for (; k < k1; k += 32) {
    bf16x8_sk a = *reinterpret_cast<const bf16x8_sk*>(ap + k);
    bf16x8_sk b = *reinterpret_cast<const bf16x8_sk*>(wp + k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

  // D layout: row = (lane>>4)*4 + r (M dim), col = lane&15 (N dim)
  const int ccol = nt * 16 + (lane & 15);
  if (ccol >= N) return;
  if (SPLIT) {
    float* c = reinterpret_cast<float*>(Cout) + (size_t)ks * M * N;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int crow = mt * 16 + (lane >> 4) * 4 + r;
      if (crow < M) c[(size_t)crow * N + ccol] = acc[r];
    }
  } else {
    bf16_t* c = reinterpret_cast<bf16_t*>(Cout);
    float bv = bias ? ScalarIO<bf16_t>::load(bias + ccol) : 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int crow = mt * 16 + (lane >> 4) * 4 + r;
      if (crow < M) {
        ScalarIO<bf16_t>::store(c + (size_t)crow * N + ccol, sk_act(acc[r] + bv, act));
      }
    }
  }
}

__global__ void skinny_epilogue_kernel(const float* __restrict__ ws,
                                       const bf16_t* __restrict__ bias, bf16_t* __restrict__ C,
                                       long MN, int N, int act, int nsplit) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= MN) return;
  float v = 0.f;
  for (int s = 0; s < nsplit; ++s) v += ws[(size_t)s * MN + i];
  if (bias) v += ScalarIO<bf16_t>::load(bias + (i % N));
  ScalarIO<bf16_t>::store(C + i, sk_act(v, act));
}

// ---- fp8 (e4m3) weight-only variant ---------------------------------------
// The decode GEMMs are pinned at an L2-access-rate wall proportional to the
// UNIQUE bytes demanded (profiles/r02_notes.md): fp8 weights halve the weight
// stream.  W is stored as e4m3 bytes [N, K] with one fp32 scale per output
// row; fragments dequantize in registers to feed the same bf16 MFMA chain
// (A stays bf16 — weight-only quantization, activations untouched), and the
// row scale multiplies the fp32 accumulator once at the epilogue (exact).

typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_sk;
typedef __attribute__((ext_vector_type(2))) float f32x2_sk;

// hardware v_cvt_pk_f32_fp8 (gfx940+): 2 e4m3 bytes -> 2 floats per
// instruction; the first cut's scalar ldexpf chain made the kernel
// ALU-bound and erased the bandwidth win
DEV bf16x8_sk dequant8(u32x2_sk raw) {
  bf16x8_sk r;
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    f32x2_sk lo = __builtin_amdgcn_cvt_pk_f32_fp8(raw[h], false);
    f32x2_sk hi = __builtin_amdgcn_cvt_pk_f32_fp8(raw[h], true);
    r[4 * h + 0] = (short)f2bf(lo[0]);
    r[4 * h + 1] = (short)f2bf(lo[1]);
    r[4 * h + 2] = (short)f2bf(hi[0]);
    r[4 * h + 3] = (short)f2bf(hi[1]);
  }
  return r;
}

template <bool SPLIT>
__global__ __launch_bounds__(SK_BLOCK) void skinny_gemm_fp8_kernel(
    const bf16_t* __restrict__ A, const unsigned char* __restrict__ W,
    const float* __restrict__ wscale, const bf16_t* __restrict__ bias, void* __restrict__ Cout,
    int M, int N, int K, int act, int nsplit) {
  const int lane = threadIdx.x % WAVE;
  const int wid = threadIdx.x / WAVE;
  const int nM = (M + 15) >> 4;
  const int nN = (N + 15) >> 4;
  int tile = blockIdx.x * 4 + wid;
  const int ntile2 = nM * nN;
  if (tile >= ntile2 * nsplit) return;
  const int ks = tile / ntile2;
  tile -= ks * ntile2;
  const int nt = tile / nM;
  const int mt = tile % nM;

  const int arow = min(mt * 16 + (lane & 15), M - 1);
  const int wrow = min(nt * 16 + (lane & 15), N - 1);
  const int k8 = (lane >> 4) * 8;
  const int kchunk = ((K / 32) / nsplit) * 32;
  const int k0 = ks * kchunk;
  const int k1 = (ks == nsplit - 1) ? K : k0 + kchunk;
  const bf16_t* ap = A + (size_t)arow * K + k8;
  const unsigned char* wp = W + (size_t)wrow * K + k8;

  f32x4_sk acc = {0.f, 0.f, 0.f, 0.f};
  int k = k0;
  if (k + 128 <= k1) {
    bf16x8_sk a0 = *reinterpret_cast<const bf16x8_sk*>(ap + k);
    u32x2_sk b0 = *reinterpret_cast<const u32x2_sk*>(wp + k);
    bf16x8_sk a1 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 32);
    u32x2_sk b1 = *reinterpret_cast<const u32x2_sk*>(wp + k + 32);
    bf16x8_sk a2 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 64);
    u32x2_sk b2 = *reinterpret_cast<const u32x2_sk*>(wp + k + 64);
    bf16x8_sk a3 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 96);
    u32x2_sk b3 = *reinterpret_cast<const u32x2_sk*>(wp + k + 96);
    for (k += 128; k + 128 <= k1; k += 128) {
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, dequant8(b0), acc, 0, 0, 0);
      a0 = *reinterpret_cast<const bf16x8_sk*>(ap + k);
      b0 = *reinterpret_cast<const u32x2_sk*>(wp + k);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, dequant8(b1), acc, 0, 0, 0);
      a1 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 32);
      b1 = *reinterpret_cast<const u32x2_sk*>(wp + k + 32);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, dequant8(b2), acc, 0, 0, 0);
      a2 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 64);
      b2 = *reinterpret_cast<const u32x2_sk*>(wp + k + 64);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, dequant8(b3), acc, 0, 0, 0);
      a3 = *reinterpret_cast<const bf16x8_sk*>(ap + k + 96);
      b3 = *reinterpret_cast<const u32x2_sk*>(wp + k + 96);
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, dequant8(b0), acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, dequant8(b1), acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a2, dequant8(b2), acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a3, dequant8(b3), acc, 0, 0, 0);
  }
  for (; k < k1; k += 32) {
    bf16x8_sk a = *reinterpret_cast<const bf16x8_sk*>(ap + k);
    u32x2_sk b = *reinterpret_cast<const u32x2_sk*>(wp + k);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, dequant8(b), acc, 0, 0, 0);
  }

  const int ccol = nt * 16 + (lane & 15);
  if (ccol >= N) return;
  const float sc = wscale[ccol];
  if (SPLIT) {
    float* c = reinterpret_cast<float*>(Cout) + (size_t)ks * M * N;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int crow = mt * 16 + (lane >> 4) * 4 + r;
      if (crow < M) c[(size_t)crow * N + ccol] = acc[r] * sc;
    }
  } else {
    bf16_t* c = reinterpret_cast<bf16_t*>(Cout);
    float bv = bias ? ScalarIO<bf16_t>::load(bias + ccol) : 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int crow = mt * 16 + (lane >> 4) * 4 + r;
      if (crow < M) {
        ScalarIO<bf16_t>::store(c + (size_t)crow * N + ccol, sk_act(acc[r] * sc + bv, act));
      }
    }
  }
}

}  // namespace

at::Tensor skinny_gemm(const at::Tensor& a, const at::Tensor& w,
                       const c10::optional<at::Tensor>& bias, long act) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16 && a.dim() == 2 && a.is_contiguous());
  TORCH_CHECK(w.dtype() == at::kBFloat16 && w.dim() == 2 && w.is_contiguous());
  const int M = a.size(0);
  const int K = a.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K && K % 32 == 0, "skinny_gemm: K must be a multiple of 32");
  auto c = at::empty({M, N}, a.options());
  if (M == 0) return c;
  const bf16_t* bp = nullptr;
  at::Tensor bc;
  if (bias.has_value() && bias->defined()) {
    bc = bias->contiguous();
    TORCH_CHECK(bc.numel() == N && bc.dtype() == at::kBFloat16);
    bp = reinterpret_cast<const bf16_t*>(bc.data_ptr());
  }
  const int ntiles = ((M + 15) / 16) * ((N + 15) / 16);
  // fill the chip AND the memory pipeline: ~4 waves/CU only keeps ~0.5 KB
  // of loads in flight per CU (13 us for a 3.5 MB weight read); target
  // >= 4096 waves (16/CU) via K-splits so the latency-bound stream overlaps
  int nsplit = 1;
  while (nsplit < 8 && ntiles * nsplit < 4096 && (K / 32) / (nsplit * 2) >= 2) nsplit *= 2;
  auto stream = c10::hip::getCurrentHIPStream();
  const auto ap = reinterpret_cast<const bf16_t*>(a.data_ptr());
  const auto wp = reinterpret_cast<const bf16_t*>(w.data_ptr());
  if (nsplit == 1) {
    const int grid = (ntiles + 3) / 4;
    skinny_gemm_kernel<false><<<grid, SK_BLOCK, 0, stream>>>(
        ap, wp, bp, c.data_ptr(), M, N, K, (int)act, 1);
  } else {
    auto ws = at::empty({nsplit, (long)M, (long)N}, a.options().dtype(at::kFloat));
    const int grid = (ntiles * nsplit + 3) / 4;
    skinny_gemm_kernel<true><<<grid, SK_BLOCK, 0, stream>>>(
        ap, wp, nullptr, ws.data_ptr(), M, N, K, (int)act, nsplit);
    const long MN = (long)M * N;
    skinny_epilogue_kernel<<<(int)((MN + 255) / 256), 256, 0, stream>>>(
        ws.data_ptr<float>(), bp, reinterpret_cast<bf16_t*>(c.data_ptr()), MN, N, (int)act,
        nsplit);
  }
  HIP_CHECK_LAST();
  return c;
}

at::Tensor skinny_gemm_fp8(const at::Tensor& a, const at::Tensor& w8, const at::Tensor& wscale,
                           const c10::optional<at::Tensor>& bias, long act) {
  TORCH_CHECK(a.is_cuda() && a.dtype() == at::kBFloat16 && a.dim() == 2 && a.is_contiguous());
  TORCH_CHECK(w8.dtype() == at::kByte && w8.dim() == 2 && w8.is_contiguous(),
              "skinny_gemm_fp8: weight must be contiguous e4m3 bytes viewed as uint8");
  const int M = a.size(0);
  const int K = a.size(1);
  const int N = w8.size(0);
  TORCH_CHECK(w8.size(1) == K && K % 32 == 0, "skinny_gemm_fp8: K must be a multiple of 32");
  TORCH_CHECK(wscale.dtype() == at::kFloat && wscale.numel() == N && wscale.is_contiguous());
  auto c = at::empty({M, N}, a.options());
  if (M == 0) return c;
  const bf16_t* bp = nullptr;
  at::Tensor bc;
  if (bias.has_value() && bias->defined()) {
    bc = bias->contiguous();
    TORCH_CHECK(bc.numel() == N && bc.dtype() == at::kBFloat16);
    bp = reinterpret_cast<const bf16_t*>(bc.data_ptr());
  }
  const int ntiles = ((M + 15) / 16) * ((N + 15) / 16);
  int nsplit = 1;
  while (nsplit < 8 && ntiles * nsplit < 4096 && (K / 32) / (nsplit * 2) >= 2) nsplit *= 2;
  auto stream = c10::hip::getCurrentHIPStream();
  const auto ap = reinterpret_cast<const bf16_t*>(a.data_ptr());
  const auto wp = w8.data_ptr<unsigned char>();
  const auto sp = wscale.data_ptr<float>();
  if (nsplit == 1) {
    const int grid = (ntiles + 3) / 4;
    skinny_gemm_fp8_kernel<false><<<grid, SK_BLOCK, 0, stream>>>(
        ap, wp, sp, bp, c.data_ptr(), M, N, K, (int)act, 1);
  } else {
    auto ws = at::empty({nsplit, (long)M, (long)N}, a.options().dtype(at::kFloat));
    const int grid = (ntiles * nsplit + 3) / 4;
    skinny_gemm_fp8_kernel<true><<<grid, SK_BLOCK, 0, stream>>>(
        ap, wp, sp, nullptr, ws.data_ptr(), M, N, K, (int)act, nsplit);
    const long MN = (long)M * N;
    skinny_epilogue_kernel<<<(int)((MN + 255) / 256), 256, 0, stream>>>(
        ws.data_ptr<float>(), bp, reinterpret_cast<bf16_t*>(c.data_ptr()), MN, N, (int)act,
        nsplit);
  }
  HIP_CHECK_LAST();
  return c;
}
