// Standalone probe: empirical lane->element mapping of ds_read_b64_tr_b16.
// Not part of the extension; built and run ad hoc on a GPU box:
//   hipcc --offload-arch=gfx950 -O2 probe_tr.hip -o /tmp/probe_tr && /tmp/probe_tr
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(unsigned short* out, int mode) {
  __shared__ unsigned short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  int lane = threadIdx.x & 63;
  unsigned short* addr;
  if (mode == 0) {
    addr = &lds[0];                 // uniform base
  } else if (mode == 1) {
    addr = &lds[lane * 4];          // per-lane 8B stride
  } else {
    addr = &lds[(lane & 15) * 4 + (lane >> 4) * 256];
  }
  // 64-bit transpose read, 16-bit elements
  using v4s = __attribute__((ext_vector_type(4))) short;
  v4s v = __builtin_amdgcn_ds_read_tr16_b64_v4i16((__attribute__((address_space(3))) v4s*)addr);
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = ((unsigned short*)&v)[j];
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * sizeof(unsigned short));
  unsigned short h[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2],
             h[l*4+3]);
    }
  }
  return 0;
}
