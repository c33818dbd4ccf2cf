// Chunked gather/scatter copy engine for gfx950.
//
// One kernel serves three framework features that are Python loops over
// slice copies in the reference:
//   - checkpoint tensor pack/unpack (flatten a model/optimizer state into
//     one flat HBM buffer for a single D2H + file write) — the weight
//     saving the reference declares but never implements (reference
//     dmlcloud/pipeline.py:61-64),
//   - batch interleaving (reference dmlcloud/util/data.py:266-341, a
//     Python loop of N^2 strided slice copies),
//   - flat-parameter replication setup (parallel/flat.py).
//
// Host splits the work into `CopyUnit{src, dst, nbytes}` descriptors of at
// most 1 MiB each; blocks grid-stride over units; within a unit lanes copy
// 16 B each (dwordx4) when both pointers are 16-byte congruent, else 4 B,
// else single bytes. HBM3E-bound: the 16 B/lane path is the coalescing
// sweet spot (guide §2).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "ops_common.h"

namespace dmlamd {

struct CopyUnit {
  const void* src;
  void* dst;
  int64_t nbytes;
};

__global__ void __launch_bounds__(kBlock) chunked_copy_kernel(
    const CopyUnit* __restrict__ units, int64_t nunits) {
  for (int64_t u = blockIdx.x; u < nunits; u += gridDim.x) {
    const CopyUnit unit = units[u];
    const char* src = (const char*)unit.src;
    char* dst = (char*)unit.dst;
    const int64_t n = unit.nbytes;

    const bool algn16 = ((((uintptr_t)src) | ((uintptr_t)dst)) & 15) == 0;
    const bool algn4 = ((((uintptr_t)src) | ((uintptr_t)dst)) & 3) == 0;

    if (algn16) {
      const int64_t nvec = n / 16;
      const uint4* s4 = (const uint4*)src;
      uint4* d4 = (uint4*)dst;
      for (int64_t i = threadIdx.x; i < nvec; i += kBlock) d4[i] = s4[i];
      for (int64_t i = nvec * 16 + threadIdx.x; i < n; i += kBlock) dst[i] = src[i];
    } else if (algn4) {
      const int64_t nvec = n / 4;
      const uint32_t* s1 = (const uint32_t*)src;
      uint32_t* d1 = (uint32_t*)dst;
      for (int64_t i = threadIdx.x; i < nvec; i += kBlock) d1[i] = s1[i];
      for (int64_t i = nvec * 4 + threadIdx.x; i < n; i += kBlock) dst[i] = src[i];
    } else {
      for (int64_t i = threadIdx.x; i < n; i += kBlock) dst[i] = src[i];
    }
  }
}

// units_blob: int64 tensor on DEVICE of shape [nunits, 3] holding
// {src_ptr, dst_ptr, nbytes} rows (built host-side, uploaded once).
void chunked_copy(at::Tensor units_blob, int64_t nunits) {
  TORCH_CHECK(units_blob.is_cuda(), "unit table must be on device");
  TORCH_CHECK(units_blob.scalar_type() == at::kLong && units_blob.is_contiguous(),
              "unit table must be contiguous int64");
  TORCH_CHECK(units_blob.numel() >= nunits * 3, "unit table too small");
  if (nunits == 0) return;
  auto stream = c10::hip::getCurrentHIPStream();
  int blocks = (int)std::min<int64_t>(nunits, kMaxGrid);
  hipLaunchKernelGGL(chunked_copy_kernel, dim3(blocks), dim3(kBlock), 0, stream,
                     (const CopyUnit*)units_blob.data_ptr<int64_t>(), nunits);
}

} // namespace dmlamd
