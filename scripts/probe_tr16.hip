// Probe ds_read_b64_tr_b16 semantics on gfx950: fill LDS with
// identifiable u16 values (value == linear element index), read with
// the transpose instruction at several address patterns, dump what
// each lane received. Build+run standalone:
//   hipcc --offload-arch=gfx950 -o /tmp/probe scripts/probe_tr16.hip
//   ./probe
#include <hip/hip_runtime.h>

#include <cstdio>

__global__ void probe(unsigned short* out, int pattern) {
  __shared__ unsigned short lds[2048];  // 4 KiB
  int l = threadIdx.x;
  for (int i = l; i < 2048; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  int addr_b;
  switch (pattern) {
    case 0: addr_b = l * 8; break;                       // flat 8B/lane
    case 1: addr_b = (l & 15) * 8 + (l >> 4) * 128; break;
    case 2: addr_b = (l >> 4) * 8 + (l & 15) * 32; break;
    case 3: addr_b = (l & 3) * 8 + (l >> 2) * 32; break;
    default: addr_b = l * 8;
  }
  unsigned long long r;
  unsigned int a32 = (unsigned int)(size_t)(
      reinterpret_cast<const char*>(lds) + addr_b);
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r)
               : "v"(a32));
  out[l * 4 + 0] = (unsigned short)(r & 0xffff);
  out[l * 4 + 1] = (unsigned short)((r >> 16) & 0xffff);
  out[l * 4 + 2] = (unsigned short)((r >> 32) & 0xffff);
  out[l * 4 + 3] = (unsigned short)((r >> 48) & 0xffff);
}

int main() {
  unsigned short* d;
  (void)hipMalloc(&d, 64 * 4 * 2);
  unsigned short h[256];
  for (int pat = 0; pat < 4; ++pat) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, pat);
    (void)hipDeviceSynchronize();
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("pattern %d:\n", pat);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d (addr elems %4d..): got %4d %4d %4d %4d\n", l,
             0, h[l * 4], h[l * 4 + 1], h[l * 4 + 2], h[l * 4 + 3]);
      if (l == 19 && pat > 0) { l = 47; }  // print lanes 0..19, 48..63
    }
  }
  return 0;
}
