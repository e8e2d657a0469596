// Shared device helpers for the trlx_amd gfx950 kernels.
//
// Written natively for CDNA4 (wave64, 32-bank LDS, MFMA) per
// /opt/skills/guides/cdna_hip_programming.md — no CUDA-compat shims.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(2))) short short2v;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((ext_vector_type(4))) float float4v;

// ---- bf16 <-> f32 -----------------------------------------------------------

DEV float bf2f(unsigned short u) {
  union {
    float f;
    unsigned int i;
  } x;
  x.i = ((unsigned int)u) << 16;
  return x.f;
}

DEV unsigned short f2bf(float f) {
  union {
    float f;
    unsigned int i;
  } x;
  x.f = f;
  if ((x.i & 0x7fffffffu) > 0x7f800000u) return 0x7fc0;  // NaN
  unsigned int lsb = (x.i >> 16) & 1u;
  x.i += 0x7fffu + lsb;  // round to nearest even
  return (unsigned short)(x.i >> 16);
}

// ---- scalar load/store traits ----------------------------------------------

template <typename T>
struct ScalarIO;

template <>
struct ScalarIO<float> {
  DEV static float load(const float* p) { return *p; }
  DEV static void store(float* p, float v) { *p = v; }
};

struct bf16_t {
  unsigned short u;
};

template <>
struct ScalarIO<bf16_t> {
  DEV static float load(const bf16_t* p) { return bf2f(p->u); }
  DEV static void store(bf16_t* p, float v) { p->u = f2bf(v); }
};

// vectorized 8-element load of T into float[8]
template <typename T>
DEV void load8(const T* p, float* out);

template <>
DEV void load8<float>(const float* p, float* out) {
  const float4v* v = reinterpret_cast<const float4v*>(p);
  float4v a = v[0], b = v[1];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    out[i] = a[i];
    out[i + 4] = b[i];
  }
}

template <>
DEV void load8<bf16_t>(const bf16_t* p, float* out) {
  short8v s = *reinterpret_cast<const short8v*>(p);
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf2f((unsigned short)s[i]);
}

template <typename T>
DEV void store8(T* p, const float* in);

template <>
DEV void store8<float>(float* p, const float* in) {
  float4v a, b;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    a[i] = in[i];
    b[i] = in[i + 4];
  }
  float4v* v = reinterpret_cast<float4v*>(p);
  v[0] = a;
  v[1] = b;
}

template <>
DEV void store8<bf16_t>(bf16_t* p, const float* in) {
  short8v s;
#pragma unroll
  for (int i = 0; i < 8; ++i) s[i] = (short)f2bf(in[i]);
  *reinterpret_cast<short8v*>(p) = s;
}

// ---- wave / block reductions ------------------------------------------------

DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// block-level sum over NWAVES waves (block size = NWAVES*64); buf: NWAVES floats
template <int NWAVES>
DEV float block_sum(float v, float* buf) {
  v = wave_sum(v);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (lane == 0) buf[wid] = v;
  __syncthreads();
  float r = 0.f;
#pragma unroll
  for (int i = 0; i < NWAVES; ++i) r += buf[i];
  __syncthreads();
  return r;
}

template <int NWAVES>
DEV float block_max(float v, float* buf) {
  v = wave_max(v);
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  if (lane == 0) buf[wid] = v;
  __syncthreads();
  float r = -INFINITY;
#pragma unroll
  for (int i = 0; i < NWAVES; ++i) r = fmaxf(r, buf[i]);
  __syncthreads();
  return r;
}

// online-logsumexp pair (m = running max, s = sum of exp(x - m))
struct MS {
  float m, s;
};

DEV MS ms_combine(MS a, MS b) {
  MS r;
  r.m = fmaxf(a.m, b.m);
  // exp(-inf - -inf) guard: if both -inf, s stays 0
  float ea = (a.m == -INFINITY) ? 0.f : __expf(a.m - r.m);
  float eb = (b.m == -INFINITY) ? 0.f : __expf(b.m - r.m);
  r.s = a.s * ea + b.s * eb;
  return r;
}

DEV MS wave_ms(MS v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    MS o;
    o.m = __shfl_xor(v.m, off);
    o.s = __shfl_xor(v.s, off);
    v = ms_combine(v, o);
  }
  return v;
}

// ---- counter-based RNG (splitmix64) -----------------------------------------

DEV unsigned long long splitmix64(unsigned long long z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

// uniform in (0, 1): 24 high bits, never exactly 0
DEV float rng_uniform(unsigned long long seed, unsigned long long a, unsigned long long b) {
  unsigned long long h = splitmix64(seed ^ splitmix64(a ^ splitmix64(b)));
  return ((h >> 40) + 1.0f) * (1.0f / 16777217.0f);
}

inline unsigned long long splitmix64_host(unsigned long long z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

#define HIP_CHECK_LAST()                                           \
  do {                                                             \
    hipError_t e = hipGetLastError();                              \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e)); \
  } while (0)
