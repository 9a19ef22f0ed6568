// Probe: global_load_lds semantics on gfx950 — verifies that the intrinsic
// writes lane l's 16 bytes at (wave-uniform LDS base) + l*16, by round-
// tripping a pattern. Run: hipcc --offload-arch=gfx950 glds_probe.hip -o t && ./t
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned int u32;

__global__ void glds_rt(const u32* __restrict__ src, u32* __restrict__ dst) {
  __shared__ u32 lds[64 * 4];   // one wave: 64 lanes x 16B
  const int lane = threadIdx.x & 63;
  auto g = (const __attribute__((address_space(1))) u32*)(src + lane * 4);
  auto l = (__attribute__((address_space(3))) u32*)(&lds[0]);
  __builtin_amdgcn_global_load_lds(g, l, 16, 0, 0);
  __builtin_amdgcn_s_waitcnt(0);   // vmcnt(0)
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x) dst[i] = lds[i];
}

int main() {
  u32 *src, *dst;
  hipMalloc(&src, 256 * 4);
  hipMalloc(&dst, 256 * 4);
  u32 h[256];
  for (int i = 0; i < 256; ++i) h[i] = i * 7 + 3;
  hipMemcpy(src, h, sizeof(h), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(glds_rt, dim3(1), dim3(64), 0, 0, src, dst);
  hipDeviceSynchronize();
  u32 out[256];
  hipMemcpy(out, dst, sizeof(out), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 256; ++i)
    if (out[i] != h[i]) { if (bad < 4) printf("mismatch %d: %u vs %u\n", i, out[i], h[i]); ++bad; }
  printf(bad ? "GLDS PROBE FAILED (%d)\n" : "GLDS PROBE OK\n", bad);
  return bad != 0;
}
