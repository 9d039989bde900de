// Empirical semantics probe for gfx950 ds_read_b64_tr_b16.
//
// Fills LDS with s[i] = i (u16), has each lane issue the transpose read
// at a chosen per-lane address, and prints which LDS element index each
// (lane, elem) slot received. Run on an MI355X box:
//   hipcc --offload-arch=gfx950 tools/trprobe.hip -o gpurun_out/trprobe
//   ./gpurun_out/trprobe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short s16x4;

__global__ void probe(const int* addrs, unsigned short* out) {
  __shared__ unsigned short s[4096];
  int t = threadIdx.x;
  for (int i = t; i < 4096; i += 64) s[i] = (unsigned short)i;
  __syncthreads();
  int a = addrs[t];  // element index this lane passes as its address
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4*)&s[a]);
#pragma unroll
  for (int j = 0; j < 4; ++j) out[t * 4 + j] = (unsigned short)v[j];
}

static void run(const char* name, int* addrs_h) {
  int* addrs;
  unsigned short* out;
  (void)hipMalloc(&addrs, 64 * sizeof(int));
  (void)hipMalloc(&out, 256 * sizeof(unsigned short));
  (void)hipMemcpy(addrs, addrs_h, 64 * sizeof(int), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, addrs, out);
  unsigned short out_h[256];
  (void)hipMemcpy(out_h, out, sizeof(out_h), hipMemcpyDeviceToHost);
  (void)hipDeviceSynchronize();
  printf("=== %s\n", name);
  for (int l = 0; l < 64; ++l) {
    printf("lane %2d addr %4d -> %4d %4d %4d %4d\n", l, addrs_h[l],
           out_h[l * 4], out_h[l * 4 + 1], out_h[l * 4 + 2],
           out_h[l * 4 + 3]);
  }
  (void)hipFree(addrs);
  (void)hipFree(out);
}

int main() {
  int a[64];
  for (int l = 0; l < 64; ++l) a[l] = 0;          // uniform base
  run("uniform0", a);
  for (int l = 0; l < 64; ++l) a[l] = l;          // addr = lane (elems)
  run("lane-linear", a);
  for (int l = 0; l < 64; ++l) a[l] = (l & 15) + (l >> 4) * 64;
  run("guide-pattern", a);                        // (l&15) + (l>>4)*64
  for (int l = 0; l < 64; ++l) a[l] = l * 4;      // b64-style
  run("4l", a);
  return 0;
}
