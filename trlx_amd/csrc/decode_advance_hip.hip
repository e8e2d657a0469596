#include "hip/hip_runtime.h"
// Fused decode-loop bookkeeping: one kernel replaces the
// ~11 elementwise launches per generated token that advance the hipGraph
// decode engine's device-resident state (eos masking, output write, counter
// bumps, next-step position ids).  Each is ~3 us of launch/latency floor at
// B=128 — together they were ~30 us of the ~700 us token step.
//
// Semantics (must match generation.py DecodeEngine._advance + the _step
// prologue exactly, in order):
//   tok      = finished ? pad : tok
//   finished |= tok == eos                  (skipped when eos < 0)
//   out_tokens[b, step_col] = tok
//   cur_tok[b, 0] = tok
//   rng_offset += 1; step_col += 1; cache_idx += 1; seq_lens[b] += 1
//   pos_ids[b, 0] = cache_idx_new - key_starts[b]   (for the NEXT step)
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>

#include "common.h"

namespace {

__global__ void decode_advance_kernel(const long* __restrict__ tok, long* __restrict__ out_tokens,
                                      long* __restrict__ cur_tok, bool* __restrict__ finished,
                                      long* __restrict__ rng_offset, long* __restrict__ step_col,
                                      long* __restrict__ cache_idx, int* __restrict__ seq_lens,
                                      int* __restrict__ pos_ids, const int* __restrict__ key_starts,
                                      int B, int max_new, long eos, long pad) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  const long col = *step_col;
  const long new_cache = *cache_idx + 1;
  if (b < B) {
    long t = tok[b];
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

}  // namespace

void decode_advance(const at::Tensor& tok, at::Tensor& out_tokens, at::Tensor& cur_tok,
                    at::Tensor& finished, at::Tensor& rng_offset, at::Tensor& step_col,
                    at::Tensor& cache_idx, at::Tensor& seq_lens, at::Tensor& pos_ids,
                    const c10::optional<at::Tensor>& key_starts, long eos, long pad) {
  const int B = tok.numel();
  const int max_new = out_tokens.size(1);
  TORCH_CHECK(tok.is_cuda() && tok.dtype() == at::kLong && tok.is_contiguous());
  TORCH_CHECK(out_tokens.size(0) == B && cur_tok.numel() == B && finished.numel() == B);
  TORCH_CHECK(seq_lens.dtype() == at::kInt && pos_ids.dtype() == at::kInt);
  const int* ks = nullptr;
  if (key_starts.has_value()) {
    TORCH_CHECK(key_starts->dtype() == at::kInt && key_starts->is_contiguous());
    ks = key_starts->data_ptr<int>();
  }
  auto stream = c10::hip::getCurrentHIPStream();
  const int block = 256;
 hipLaunchKernelGGL(( decode_advance_kernel), dim3((B + block - 1) / block), dim3(block), 0, stream, 
      tok.data_ptr<long>(), out_tokens.data_ptr<long>(), cur_tok.data_ptr<long>(),
      finished.data_ptr<bool>(), rng_offset.data_ptr<long>(), step_col.data_ptr<long>(),
      cache_idx.data_ptr<long>(), seq_lens.data_ptr<int>(), pos_ids.data_ptr<int>(), ks, B,
      max_new, eos, pad);
  HIP_CHECK_LAST();
}
