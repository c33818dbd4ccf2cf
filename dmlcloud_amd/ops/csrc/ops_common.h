// Shared helpers for the dmlcloud_amd gfx950 kernels.
//
// Target: MI355X (CDNA4, gfx950) only. Wavefront = 64 lanes; block size is
// always a multiple of 64. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>
#include <limits>

namespace dmlamd {

constexpr int kWave = 64; // CDNA wavefront width (not 32!)
constexpr int kBlock = 256; // 4 waves per workgroup
constexpr int kMaxGrid = 2048; // 256 CUs x 8 blocks: grid-stride beyond this

// Reduction op codes — keep in sync with ops/_reference.py
enum ReduceOp : int { OP_SUM = 0, OP_MIN = 1, OP_MAX = 2 };

template <typename A, int OP>
struct Combine;

template <typename A>
struct Combine<A, OP_SUM> {
  static __device__ __forceinline__ A apply(A a, A b) { return a + b; }
  static __host__ __device__ A identity() { return A(0); }
};

template <typename A>
struct Combine<A, OP_MIN> {
  static __device__ __forceinline__ A apply(A a, A b) { return a < b ? a : b; }
  static __host__ __device__ A identity() {
    return std::numeric_limits<A>::has_infinity ? std::numeric_limits<A>::infinity()
                                                : std::numeric_limits<A>::max();
  }
};

template <typename A>
struct Combine<A, OP_MAX> {
  static __device__ __forceinline__ A apply(A a, A b) { return a > b ? a : b; }
  static __host__ __device__ A identity() {
    return std::numeric_limits<A>::has_infinity ? -std::numeric_limits<A>::infinity()
                                                : std::numeric_limits<A>::lowest();
  }
};

// Cross-lane reduction over the full 64-lane wave via shfl_down.
template <typename A, int OP>
__device__ __forceinline__ A wave_reduce(A v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) {
    v = Combine<A, OP>::apply(v, __shfl_down(v, off, kWave));
  }
  return v;
}

// Block-level reduction: wave shuffle + LDS partials (kBlock/kWave waves).
// Returns the block total in thread 0 (other threads: undefined).
template <typename A, int OP>
__device__ __forceinline__ A block_reduce(A v) {
  __shared__ A lds[kBlock / kWave];
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  v = wave_reduce<A, OP>(v);
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  if (threadIdx.x == 0) {
    A total = lds[0];
#pragma unroll
    for (int w = 1; w < kBlock / kWave; ++w) total = Combine<A, OP>::apply(total, lds[w]);
    return total;
  }
  return Combine<A, OP>::identity();
}

inline int grid_for(int64_t work_items, int per_block) {
  int64_t blocks = (work_items + per_block - 1) / per_block;
  if (blocks < 1) blocks = 1;
  if (blocks > kMaxGrid) blocks = kMaxGrid;
  return static_cast<int>(blocks);
}

} // namespace dmlamd
