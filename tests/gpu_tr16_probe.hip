// ds_read_b64_tr_b16 mapping probe: stage LDS with value = linear index
// (int16, exact), read with the tr builtin under several per-lane address
// patterns, dump lane -> 4 delivered values. Build:
//   hipcc --offload-arch=gfx950 tests/gpu_tr16_probe.hip -o gpurun_out/tr16probe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short s16x4;

extern "C" __global__ void k_trprobe(short* out, int mode) {
  __shared__ short s[64 * 128];  // natural [q 64][d 128], value = q*128+d
  int tid = threadIdx.x;
  for (int i = tid; i < 64 * 128; i += 64) s[i] = (short)i;
  __syncthreads();
  int addr;
  switch (mode) {
    case 0: addr = tid * 8; break;                     // contiguous 8B/lane
    case 1: addr = (tid & 3) * 256 + (tid >> 2) * 8; break;  // 4 q rows x 16 lanes
    case 2: addr = (tid & 15) * 256 + (tid >> 4) * 8; break; // 16 q rows x 4 lanes
    default: addr = (tid >> 4) * 256 + (tid & 15) * 8; break; // 4 lanes/q ordered
  }
  typedef __attribute__((address_space(3))) s16x4 as3_s16x4;
  auto* base = (__attribute__((address_space(3))) char*)s;
  s16x4 r = __builtin_amdgcn_ds_read_tr16_b64_v4i16((as3_s16x4*)(base + addr));
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = r[j];
}

int main() {
  short* out;
  (void)hipMalloc(&out, 64 * 4 * sizeof(short));
  short h[256];
  for (int mode = 0; mode < 4; ++mode) {
    hipLaunchKernelGGL(k_trprobe, dim3(1), dim3(64), 0, 0, out, mode);
    (void)hipMemcpy(h, out, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d (addr pattern %d):\n", mode, mode);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d:", l);
      for (int j = 0; j < 4; ++j) {
        int v = h[l * 4 + j];
        printf(" (q%2d,d%3d)", v / 128, v % 128);
      }
      printf("\n");
    }
  }
  return 0;
}
