// Empirical probe of gfx950's ds_read_b64_tr_b16 lane<->data mapping.
//
// The CDNA4 guide gives two partial descriptions of the transpose read
// (uniform-address layout formula vs per-lane 8-B slices); which one the
// hardware implements decides how the GEMM's B-operand LDS image must be
// built. Same methodology as scripts/mfma_f64_probe.cpp (which caught a
// wrong C/D row map that symmetric tests missed): fill LDS with
// identity-coded u16 values (lds16[i] = i), run the instruction under
// several addressing modes, print the exact u16 index each lane's 4
// result elements came from.
//
// Build: hipcc --offload-arch=gfx950 -O3 -o /tmp/tr16_probe scripts/tr16_probe.cpp
// Run (GPU box): /tmp/tr16_probe

#include <hip/hip_runtime.h>

#include <cstdio>

typedef unsigned short u16;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
using lds_bf16x4 = __attribute__((address_space(3))) bf16x4;

__global__ void probe(u16* out, int mode) {
  __shared__ u16 lds[4096];
  int tid = threadIdx.x;
  for (int i = tid; i < 4096; i += 64) lds[i] = (u16)i;
  __syncthreads();
  unsigned byte_off;
  switch (mode) {
    case 0: byte_off = 0; break;                      // uniform base
    case 1: byte_off = (tid & 15) * 8; break;         // 8-B slice per lane in group
    case 2: byte_off = tid * 8; break;                // fully lane-linear 8-B
    case 3: byte_off = (tid >> 4) * 128; break;       // per-group 128-B blocks
    case 4: byte_off = 256; break;                    // uniform, nonzero base
    default: byte_off = 0;
  }
  bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16x4*)((char*)lds + byte_off));
  for (int j = 0; j < 4; j++) out[tid * 4 + j] = ((u16*)&v)[j];
}

int main() {
  u16* out;
  (void)hipMalloc(&out, 64 * 4 * sizeof(u16));
  u16 host[256];
  const char* names[] = {"uniform base 0", "lane=(l&15)*8B", "lane=l*8B",
                         "group=(l>>4)*128B", "uniform base 256B"};
  for (int mode = 0; mode <= 4; mode++) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out, mode);
    (void)hipMemcpy(host, out, sizeof host, hipMemcpyDeviceToHost);
    printf("== mode %d (%s): lane -> [u16 indices of 4 elems]\n", mode,
           names[mode]);
    for (int l = 0; l < 64; l++) {
      printf("l%02d:[%4d %4d %4d %4d]%s", l, host[l * 4], host[l * 4 + 1],
             host[l * 4 + 2], host[l * 4 + 3], (l % 4 == 3) ? "\n" : "  ");
    }
  }
  (void)hipFree(out);
  return 0;
}
