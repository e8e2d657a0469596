// Fused LM-head GEMM + online logsumexp + label gather (SURVEY.md K1+K5).
//
// The PPO experience pass needs ONLY per-token logprobs of the sampled
// labels, but the eager path materializes [N, V] logits in HBM (hundreds of
// MB), reads them back for logsumexp, then throws them away — twice (policy
// and reference).  This kernel computes logits tile-by-tile with hand-written
// MFMA (v_mfma_f32_16x16x32_bf16, LDS-staged A/W tiles via 16-byte
// global_load_lds, per the CDNA4 guide's GEMM anatomy) and immediately
// reduces each tile to per-row (max, sumexp) partials + the label logit; a
// small second kernel combines partials into
// logprob[n] = logit[label] - logsumexp.  The [N, V] intermediate never
// exists.  The C tile aliases the staging LDS (disjoint in time).
//
// Shapes: hidden [N, H] bf16 (H % 32 == 0), weight [V, H] bf16 row-major
// (the natural lm_head layout — an NT GEMM), labels [N] i64 -> out [N] f32.
// Inference-only (the training path keeps logits for backward).
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

constexpr int BM = 128;  // rows (tokens) per workgroup
constexpr int BN = 128;  // vocab columns per workgroup
constexpr int BK = 32;   // K-step (one mfma_..x32 per fragment pair)
constexpr int BLOCK = 256;  // 4 waves; each owns a 64x64 quadrant

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ __launch_bounds__(BLOCK) void lm_logprobs_tile_kernel(
    const bf16_t* __restrict__ hidden, const bf16_t* __restrict__ weight,
    float* __restrict__ partials,  // [nV, N, 2]
    float* __restrict__ label_logit, const long* __restrict__ labels, int N, int H, int V,
    int nV) {
  // grid: (nV, nM)
  const int vt = blockIdx.x;
  const int mt = blockIdx.y;
  const int row0 = mt * BM;
  const int col0 = vt * BN;

  // 64 KiB shared: [a_tile 8K | b_tile 8K | ...] during the K loop,
  // then the full fp32 C tile [128][128] afterwards (time-disjoint).
  __shared__ float smem[BM * BN];
  bf16_t* a_lds = reinterpret_cast<bf16_t*>(smem);
  bf16_t* b_lds = a_lds + BM * BK;
  float (*c_lds)[BN] = reinterpret_cast<float(*)[BN]>(smem);

  const int tid = threadIdx.x;
  const int lane = tid % WAVE;
  const int wid = tid / WAVE;
  const int wr = (wid >> 1) * 64;  // wave quadrant row
  const int wc = (wid & 1) * 64;   // wave quadrant col

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // staging: the [128, BK] bf16 tile is 8 KiB = 256 threads x 2 chunks x 16 B.
  // chunk c covers linear elements [c*2048 + tid*8, +8): row = off/BK, k = off%BK.
  // LDS dest (wave-uniform base + lane*16B) matches row-major [128][BK] exactly.
  for (int k0 = 0; k0 < H; k0 += BK) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int off = c * 2048 + tid * 8;
      const int trow = off / BK;
      const int tk = off % BK;
      const int ga_row = min(row0 + trow, N - 1);
      const int gb_row = min(col0 + trow, V - 1);
      const size_t lds_off = (size_t)c * 2048 + (size_t)wid * 512 + (size_t)lane * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(hidden + (size_t)ga_row * H + k0 + tk),
          (__attribute__((address_space(3))) unsigned int*)(a_lds + lds_off), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(weight + (size_t)gb_row * H + k0 + tk),
          (__attribute__((address_space(3))) unsigned int*)(b_lds + lds_off), 16, 0, 0);
    }
    __syncthreads();  // barrier drains vmcnt (guide §5 m97 analysis)

    // fragment layout (mfma_f32_16x16x32_bf16): lane l holds rows l%16,
    // k-slice (l/16)*8 for A; columns l%16 (= W rows) for B.
    const int frow = lane & 15;
    const int fk = (lane >> 4) * 8;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(&a_lds[(wr + i * 16 + frow) * BK + fk]);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(&b_lds[(wc + j * 16 + frow) * BK + fk]);
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[i][j], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // write accumulators: C/D layout col = lane&15, row = (lane>>4)*4 + r
  const int c_col = lane & 15;
  const int c_row = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        c_lds[wr + i * 16 + c_row + r][wc + j * 16 + c_col] = acc[i][j][r];
  __syncthreads();

  // per-row online (max, sumexp) over this tile's valid columns + label
  // gather.  4 waves x 32 rows; reads run along rows (conflict-free).
  const int valid_cols = min(BN, V - col0);
  for (int rr = wid * 32; rr < wid * 32 + 32; ++rr) {
    const int n = row0 + rr;
    if (n >= N) break;
    MS ms{-INFINITY, 0.f};
    for (int c = lane; c < valid_cols; c += WAVE) {
      const float x = c_lds[rr][c];
      if (x > ms.m) {
        ms.s = ms.s * expf(ms.m - x) + 1.f;
        ms.m = x;
      } else {
        ms.s += expf(x - ms.m);
      }
    }
    ms = wave_ms(ms);
    if (lane == 0) {
      float* p = partials + ((size_t)vt * N + n) * 2;
      p[0] = ms.m;
      p[1] = ms.s;
      const long lab = labels[n];
      if (lab >= col0 && lab < col0 + valid_cols) {
        label_logit[n] = c_lds[rr][lab - col0];
      }
    }
  }
}

__global__ void lm_logprobs_reduce_kernel(const float* __restrict__ partials,
                                          const float* __restrict__ label_logit,
                                          float* __restrict__ out, int N, int nV) {
  // one wave per row; combine nV (m, s) partials
  const int wpb = blockDim.x / WAVE;
  const int n = blockIdx.x * wpb + threadIdx.x / WAVE;
  if (n >= N) return;
  const int lane = threadIdx.x % WAVE;
  MS ms{-INFINITY, 0.f};
  for (int t = lane; t < nV; t += WAVE) {
    const float* p = partials + ((size_t)t * N + n) * 2;
    MS o{p[0], p[1]};
    ms = ms_combine(ms, o);
  }
  ms = wave_ms(ms);
  if (lane == 0) out[n] = label_logit[n] - (ms.m + logf(ms.s));
}

}  // namespace

at::Tensor lm_logprobs(const at::Tensor& hidden, const at::Tensor& weight,
                       const at::Tensor& labels) {
  TORCH_CHECK(hidden.is_cuda() && hidden.dtype() == at::kBFloat16 && hidden.dim() == 2 &&
              hidden.is_contiguous());
  TORCH_CHECK(weight.dtype() == at::kBFloat16 && weight.is_contiguous());
  TORCH_CHECK(labels.dtype() == at::kLong && labels.is_contiguous());
  const int N = hidden.size(0);
  const int H = hidden.size(1);
  const int V = weight.size(0);
  TORCH_CHECK(weight.size(1) == H && labels.numel() == N);
  TORCH_CHECK(H % BK == 0, "lm_logprobs: hidden size must be a multiple of 32");
  auto out = at::empty({N}, hidden.options().dtype(at::kFloat));
  if (N == 0) return out;
  const int nV = (V + BN - 1) / BN;
  const int nM = (N + BM - 1) / BM;
  auto partials = at::empty({nV, (long)N, 2}, hidden.options().dtype(at::kFloat));
  auto label_logit = at::empty({N}, hidden.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid(nV, nM);
  lm_logprobs_tile_kernel<<<grid, BLOCK, 0, stream>>>(
      reinterpret_cast<const bf16_t*>(hidden.data_ptr()),
      reinterpret_cast<const bf16_t*>(weight.data_ptr()), partials.data_ptr<float>(),
      label_logit.data_ptr<float>(), labels.data_ptr<long>(), N, H, V, nV);
  const int wpb = 256 / WAVE;
  lm_logprobs_reduce_kernel<<<(N + wpb - 1) / wpb, 256, 0, stream>>>(
      partials.data_ptr<float>(), label_logit.data_ptr<float>(), out.data_ptr<float>(), N, nV);
  HIP_CHECK_LAST();
  return out;
}
