// Common helpers for the CDNA4 (gfx950) kernel library.
// Wave size is 64 on CDNA — every cross-lane idiom below assumes that.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

using bf16 = __hip_bfloat16;

// vector types for wide loads (guide G13: always vectorize bf16 loads)
typedef short s16x2 __attribute__((ext_vector_type(2)));
typedef short s16x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef float f32x2 __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

DEV_INLINE float bf2f(bf16 x) { return __bfloat162float(x); }
DEV_INLINE bf16 f2bf(float x) { return __float2bfloat16(x); }

DEV_INLINE float bfbits2f(short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)(unsigned short)u) << 16;
  return c.f;
}
DEV_INLINE short f2bfbits(float x) {
  union { float f; unsigned int i; } c;
  c.f = x;
  unsigned int lsb = (c.i >> 16) & 1u;       // round-to-nearest-even
  c.i += 0x7fffu + lsb;
  return (short)(c.i >> 16);
}

// ---- wave reductions (64-lane) --------------------------------------------
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}
DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

#define HIP_CHECK_LAUNCH()                                            \
  do {                                                                \
    hipError_t e = hipGetLastError();                                 \
    if (e != hipSuccess) {                                            \
      printf("kernel launch failed: %s\n", hipGetErrorString(e));     \
    }                                                                 \
  } while (0)

constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }

// Last-arriver gate (guide G16 counter form) for single-launch
// partial-reduce + finalize kernels.  Call with ALL of the block's threads
// AFTER the block's device-scope atomicAdds.  Returns true in exactly one
// (the last-arriving) block, whose threads may then read every block's
// atomics with plain loads: each wave drains its own adds (vmcnt) before
// the one-lane relaxed agent ticket, and the winner takes a one-lane
// agent acquire (drops this CU's stale L1 lines — the accumulator may
// have been read by a PREVIOUS call's finalize on this CU) followed by a
// barrier.  The caller re-zeroes *cnt (any store: gfx950 stores write
// through to the coherent point; the next launch's atomics see it).
DEV_INLINE bool last_arriver(unsigned* cnt, unsigned total) {
  __shared__ unsigned la_flag;
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned old = __hip_atomic_fetch_add(cnt, 1u, __ATOMIC_RELAXED,
                                          __HIP_MEMORY_SCOPE_AGENT);
    la_flag = (old + 1 == total) ? 1u : 0u;
  }
  __syncthreads();
  if (la_flag == 0) return false;
  if (threadIdx.x == 0)
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  __syncthreads();
  return true;
}
