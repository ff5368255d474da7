// Shared CDNA4 device helpers (gfx950: wave64, 4x SIMD-32 per CU).
#pragma once

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define WAVE 64

// The HIP stream torch has current for this thread (per-job streams are set
// from python with torch.cuda.stream(...)).
inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

#define CHECK_IN(x) TORCH_CHECK((x).is_cuda() && (x).is_contiguous(), #x " must be contiguous on device")

// Full-wave sum: butterfly over 64 lanes; every lane ends with the total.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int m = 32; m > 0; m >>= 1) v += __shfl_xor(v, m, WAVE);
  return v;
}

// Inclusive prefix sum across 64 lanes (Hillis-Steele via shfl_up).
__device__ __forceinline__ float wave_inclusive_scan(float v) {
#pragma unroll
  for (int d = 1; d < WAVE; d <<= 1) {
    float o = __shfl_up(v, d, WAVE);
    if ((threadIdx.x & (WAVE - 1)) >= d) v += o;
  }
  return v;
}

// Counter-based RNG: two murmur3 finalizer rounds over (seed, ctr).
// Deterministic and reproduced bit-exactly by the torch reference
// (harmony_amd/ops/rng.py) so CPU<->GPU numerics tests can compare samples.
__device__ __forceinline__ unsigned int rng_u32(unsigned int seed,
                                                unsigned int ctr) {
  unsigned int h = seed ^ (ctr * 2654435761u);
#pragma unroll
  for (int r = 0; r < 2; ++r) {
    h ^= h >> 16; h *= 0x85ebca6bu;
    h ^= h >> 13; h *= 0xc2b2ae35u;
    h ^= h >> 16;
  }
  return h;
}

__device__ __forceinline__ float rng_uniform(unsigned int seed,
                                             unsigned int ctr) {
  return (rng_u32(seed, ctr) + 0.5f) * 2.3283064365386963e-10f;  // 2^-32
}
