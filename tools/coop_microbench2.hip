// Hand-rolled grid barrier cost vs the 33-153 us cg::grid_group::sync()
// (coop_microbench.hip round 1).  Variants:
//   A. two-phase sense-reversing barrier, __threadfence() before arrive and
//      after release (the visibility the megakernel stages need)
//   B. same, no fences (algorithm floor)
//   C. fence-only loop (__threadfence cost alone)
// plus a producer/consumer visibility re-check under barrier A.
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/coop_microbench2.hip -o /tmp/coopbench2
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define CHECK(x)                                                           \
  do {                                                                     \
    hipError_t e = (x);                                                    \
    if (e != hipSuccess) {                                                 \
      printf("HIP error %s at line %d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                             \
    }                                                                      \
  } while (0)

// two-phase barrier: generation counter + arrive count, device-scope atomics.
// All threads of the block participate via __syncthreads around thread 0's
// atomic.  gen/cnt live in one cacheline-separated pair.
struct GridBar {
  unsigned int cnt;  // add-only arrival counter (never reset: reset races
                     // with the next round's atomics across per-XCD L2s)
  unsigned int pad[31];
  unsigned int gen;
  unsigned int pad2[31];
  unsigned int fail;
};

template <bool FENCE>
__device__ __forceinline__ void grid_bar(GridBar* b, unsigned int nblocks,
                                         unsigned int* local_gen) {
  if (FENCE) __threadfence();
  __syncthreads();
  if (threadIdx.x == 0) {
    const unsigned int g = *local_gen;
    const unsigned int arrived = __atomic_fetch_add(&b->cnt, 1u, __ATOMIC_ACQ_REL) + 1;
    if (arrived == (g + 1) * nblocks) {
      __atomic_store_n(&b->gen, g + 1, __ATOMIC_RELEASE);
    } else {
      long spins = 0;
      while (__atomic_load_n(&b->gen, __ATOMIC_ACQUIRE) < g + 1) {
        __builtin_amdgcn_s_sleep(8);
        if (++spins > (1L << 24)) {  // bounded: a logic bug must not hang the box
          atomicAdd(&b->fail, 1u);
          break;
        }
      }
    }
    *local_gen = g + 1;
  }
  __syncthreads();
  if (FENCE) __threadfence();
}

template <bool FENCE>
__global__ void bar_cost_kernel(GridBar* bar, int iters, float* dummy) {
  // continue from the persistent generation: back-to-back launches share the
  // bar without a host-side reset (cnt is add-only)
  unsigned int local_gen = __atomic_load_n(&bar->gen, __ATOMIC_ACQUIRE);
  for (int i = 0; i < iters; ++i) grid_bar<FENCE>(bar, gridDim.x, &local_gen);
  if (blockIdx.x == 0 && threadIdx.x == 0) *dummy = local_gen;
}

__global__ void fence_cost_kernel(int iters, float* dummy) {
  for (int i = 0; i < iters; ++i) __threadfence();
  if (blockIdx.x == 0 && threadIdx.x == 0) *dummy = iters;
}

// visibility under barrier A (same pattern as round 1 but hand barrier)
__global__ void vis_kernel(GridBar* bar, int iters, unsigned int* buf, long n, int* errors) {
  unsigned int local_gen = __atomic_load_n(&bar->gen, __ATOMIC_ACQUIRE);
  const long stride = (long)gridDim.x * blockDim.x;
  const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (int i = 0; i < iters; ++i) {
    const long shift = (long)(i + 1) * 7919 * blockDim.x;
    for (long j = tid; j < n; j += stride) {
      const long src = (j + shift) % n;
      if (buf[src] != (unsigned int)i) atomicAdd(errors, 1);
    }
    grid_bar<true>(bar, gridDim.x, &local_gen);
    for (long j = tid; j < n; j += stride) {
      const long src = (j + shift) % n;
      buf[src] = (unsigned int)(i + 1);
    }
    grid_bar<true>(bar, gridDim.x, &local_gen);
  }
}

static float time_kernel(const void* kern, dim3 grid, dim3 block, void** args, int reps) {
  hipEvent_t t0, t1;
  CHECK(hipEventCreate(&t0));
  CHECK(hipEventCreate(&t1));
  CHECK(hipLaunchCooperativeKernel(kern, grid, block, args, 0, 0));
  CHECK(hipDeviceSynchronize());
  CHECK(hipEventRecord(t0));
  for (int r = 0; r < reps; ++r) CHECK(hipLaunchCooperativeKernel(kern, grid, block, args, 0, 0));
  CHECK(hipEventRecord(t1));
  CHECK(hipDeviceSynchronize());
  float ms;
  CHECK(hipEventElapsedTime(&ms, t0, t1));
  CHECK(hipEventDestroy(t0));
  CHECK(hipEventDestroy(t1));
  return ms / reps;
}

int main() {
  GridBar* bar;
  CHECK(hipMalloc(&bar, sizeof(GridBar)));
  CHECK(hipMemset(bar, 0, sizeof(GridBar)));
  float* dummy;
  CHECK(hipMalloc(&dummy, 4));

  int iters = 2000;
  for (int nblk : {64, 128, 256, 512, 1024}) {
    void* args[] = {&bar, &iters, &dummy};
    CHECK(hipMemset(bar, 0, sizeof(GridBar)));
    float a = time_kernel((const void*)bar_cost_kernel<true>, dim3(nblk), dim3(256), args, 3);
    CHECK(hipMemset(bar, 0, sizeof(GridBar)));
    float b = time_kernel((const void*)bar_cost_kernel<false>, dim3(nblk), dim3(256), args, 3);
    printf("handbar: grid %4d x 256 -> fenced %.3f us  unfenced %.3f us\n", nblk,
           a * 1000.f / iters, b * 1000.f / iters);
  }
  {
    void* args[] = {&iters, &dummy};
    float c = time_kernel((const void*)fence_cost_kernel, dim3(256), dim3(256), args, 3);
    printf("threadfence alone: %.3f us\n", c * 1000.f / iters);
  }
  {
    const long n = 32 * 1024 * 1024 / 4;
    unsigned int* buf;
    int* errors;
    CHECK(hipMalloc(&buf, n * 4));
    CHECK(hipMemset(buf, 0, n * 4));
    CHECK(hipMalloc(&errors, 4));
    CHECK(hipMemset(errors, 0, 4));
    CHECK(hipMemset(bar, 0, sizeof(GridBar)));
    int it2 = 64;
    long nn = n;
    void* args[] = {&bar, &it2, &buf, &nn, &errors};
    CHECK(hipLaunchCooperativeKernel((const void*)vis_kernel, dim3(256), dim3(256), args, 0, 0));
    CHECK(hipDeviceSynchronize());
    int h_err = -1;
    CHECK(hipMemcpy(&h_err, errors, 4, hipMemcpyDeviceToHost));
    printf("visibility under hand barrier: %s (%d stale)\n", h_err == 0 ? "OK" : "FAIL", h_err);
  }
  printf("DONE\n");
  return 0;
}
