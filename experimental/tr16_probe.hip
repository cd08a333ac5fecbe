// Hardware probe: ds_read_b64_tr_b16 lane->element mapping on gfx950.
// Fills LDS with u16 value == element index, reads through the transpose
// instruction with several per-lane addressing schemes, dumps what each
// lane's 4 elements actually were. Run once, derive the mapping, use it
// for the prefill PV B-fragment (guide T10: layout must be built for the
// read's gather; wrong alignment returns wrong data silently).
#include <hip/hip_runtime.h>
typedef unsigned short u16;
typedef short s16x4 __attribute__((ext_vector_type(4)));

__global__ void k_tr16_probe(u16* __restrict__ out, int scheme) {
  __shared__ u16 buf[1024];          // elements 0..1023, value == index
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 1024; i += 64) buf[i] = (u16)i;
  __syncthreads();
  const int l15 = lane & 15, lg = lane >> 4;
  int elem;                           // per-lane element offset
  switch (scheme) {
    case 0: elem = 0; break;                         // uniform
    case 1: elem = l15; break;                       // column-per-lane
    case 2: elem = l15 + lg * 64; break;             // doc's canonical
    case 3: elem = l15 + lg * 128; break;            // 32-row fragment
    case 4: elem = (lane & 3) * 4; break;            // 8B-aligned var
    default: elem = lane; break;
  }
  const unsigned addr = (unsigned)(unsigned long)&buf[elem];   // LDS byte addr
  s16x4 r0, r4;
  asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
               "ds_read_b64_tr_b16 %1, %2 offset:128\n\t"
               "s_waitcnt lgkmcnt(0)"
               : "=&v"(r0), "=&v"(r4) : "v"(addr));
  __builtin_amdgcn_sched_barrier(0);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    out[(long)lane * 8 + j] = (u16)r0[j];
    out[(long)lane * 8 + 4 + j] = (u16)r4[j];
  }
}

extern "C" int tr16_probe_run() {
  u16* d;
  if (hipMalloc(&d, 6 * 64 * 8 * sizeof(u16)) != hipSuccess) return 1;
  for (int s = 0; s < 6; ++s)
    hipLaunchKernelGGL(k_tr16_probe, dim3(1), dim3(64), 0, 0, d + s * 512, s);
  if (hipDeviceSynchronize() != hipSuccess) return 2;
  u16 h[6 * 512];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int s = 0; s < 6; ++s) {
    printf("scheme %d:\n", s);
    for (int lane = 0; lane < 32; ++lane) {
      printf("  lane %2d:", lane);
      for (int j = 0; j < 8; ++j) printf(" %4d", h[s * 512 + lane * 8 + j]);
      printf("\n");
    }
  }
  hipFree(d);
  return 0;
}

int main() { return tr16_probe_run(); }
