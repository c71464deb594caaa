// Empirical map of ds_read_b64_tr_b16 on gfx950: fill LDS with element
// indices, each lane passes addr = base + lane*8B, record what each lane's
// 4 elements are. Also variant with uniform address.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));

__global__ void probe(unsigned short* out_laneaddr, unsigned short* out_uniform) {
  __shared__ unsigned short s[1024];
  const int l = threadIdx.x;
  for (int i = l; i < 1024; i += blockDim.x) s[i] = (unsigned short)i;
  __syncthreads();
  if (l < 64) {
    auto* p = (__attribute__((address_space(3))) bf16x4*)(s + l * 4);  // lane*8B
    bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
    for (int j = 0; j < 4; ++j)
      out_laneaddr[l * 4 + j] = ((unsigned short*)&v)[j];
    auto* q = (__attribute__((address_space(3))) bf16x4*)(s);  // uniform base
    bf16x4 u = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(q);
    for (int j = 0; j < 4; ++j)
      out_uniform[l * 4 + j] = ((unsigned short*)&u)[j];
  }
}

int main() {
  unsigned short *a, *b;
  hipMalloc(&a, 256 * 2); hipMalloc(&b, 256 * 2);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, a, b);
  hipDeviceSynchronize();
  unsigned short ha[256], hb[256];
  hipMemcpy(ha, a, 512, hipMemcpyDeviceToHost);
  hipMemcpy(hb, b, 512, hipMemcpyDeviceToHost);
  printf("lane-addr (addr = base + lane*8B):\n");
  for (int l = 0; l < 64; ++l) {
    printf("lane %2d: %4d %4d %4d %4d\n", l, ha[l*4], ha[l*4+1], ha[l*4+2], ha[l*4+3]);
  }
  printf("uniform base:\n");
  for (int l = 0; l < 16; ++l)
    printf("lane %2d: %4d %4d %4d %4d\n", l, hb[l*4], hb[l*4+1], hb[l*4+2], hb[l*4+3]);
  return 0;
}
