// Probe: exact semantics of ds_read_b64_tr_b16 on gfx950.
// Fills LDS with element index values, issues the transpose-read with
// per-lane addresses addr = (lane&15)*2 + (lane>>4)*128 (bytes) and prints
// which LDS element index each lane's 4 dest elements received, for a few
// addressing conventions.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef short s16x4 __attribute__((ext_vector_type(4)));
#define AS3 __attribute__((address_space(3)))

__global__ void probe(short* out, int mode) {
  __shared__ short lds[1024];
  int t = threadIdx.x;
  for (int i = t; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  int lane = t & 63;
  int addr_elems;
  if (mode == 0)        // my assumed convention: (l&15) + (l>>4)*64
    addr_elems = (lane & 15) + (lane >> 4) * 64;
  else if (mode == 1)   // natural b64 addressing: lane*4 elems
    addr_elems = lane * 4;
  else                  // uniform base 0
    addr_elems = 0;
  AS3 s16x4* p = (AS3 s16x4*)((AS3 char*)lds + addr_elems * 2);
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(p);
  for (int j = 0; j < 4; ++j) out[t * 4 + j] = v[j];
}

int main() {
  short* out;
  (void)hipMalloc(&out, 64 * 4 * sizeof(short));
  short host[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out, mode);
    (void)hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l)
      printf("  lane %2d: %4d %4d %4d %4d\n", l, host[l * 4], host[l * 4 + 1],
             host[l * 4 + 2], host[l * 4 + 3]);
  }
  (void)hipFree(out);
  return 0;
}
