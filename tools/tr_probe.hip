// Empirical probe of ds_read_b64_tr_b16 addressing on gfx950.
// Fills LDS with element-index values and dumps what each lane receives
// for (a) wave-uniform address, (b) per-lane address = lane*8 bytes.
//
//   hipcc --offload-arch=gfx950 -O2 tools/tr_probe.hip -o tr_probe && ./tr_probe

#include <hip/hip_runtime.h>
#include <cstdio>

typedef short short4v __attribute__((ext_vector_type(4)));

__global__ void probe(short* out_uniform, short* out_lane) {
  __shared__ short lds[512];
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 512; i += 64) lds[i] = (short)i;
  __syncthreads();

  // (a) uniform address = &lds[0]
  short4v a = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4v*)&lds[0]);
#pragma unroll
  for (int j = 0; j < 4; ++j) out_uniform[lane * 4 + j] = a[j];

  // (b) per-lane address = &lds[lane*4] (8 bytes per lane)
  short4v b = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4v*)&lds[lane * 4]);
#pragma unroll
  for (int j = 0; j < 4; ++j) out_lane[lane * 4 + j] = b[j];
}

int main() {
  short *du, *dl;
  hipMalloc(&du, 64 * 4 * sizeof(short));
  hipMalloc(&dl, 64 * 4 * sizeof(short));
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, du, dl);
  hipDeviceSynchronize();
  short hu[256], hl[256];
  hipMemcpy(hu, du, sizeof(hu), hipMemcpyDeviceToHost);
  hipMemcpy(hl, dl, sizeof(hl), hipMemcpyDeviceToHost);
  printf("uniform addr (&lds[0]): lane: elems\n");
  for (int l = 0; l < 20; ++l)
    printf("  l%02d: %4d %4d %4d %4d\n", l, hu[l * 4], hu[l * 4 + 1], hu[l * 4 + 2], hu[l * 4 + 3]);
  printf("  l16..18:\n");
  for (int l = 16; l < 19; ++l)
    printf("  l%02d: %4d %4d %4d %4d\n", l, hu[l * 4], hu[l * 4 + 1], hu[l * 4 + 2], hu[l * 4 + 3]);
  printf("per-lane addr (&lds[lane*4]):\n");
  for (int l = 0; l < 20; ++l)
    printf("  l%02d: %4d %4d %4d %4d\n", l, hl[l * 4], hl[l * 4 + 1], hl[l * 4 + 2], hl[l * 4 + 3]);
  return 0;
}
