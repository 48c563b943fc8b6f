// Empirical layout probe for v_mfma_f64_16x16x4_f64 on gfx950.
// Feeds known per-lane A/B operands, dumps the 4 accumulator regs per
// lane, and infers the (lane,reg) -> (row,col) mapping plus the A/B
// lane -> (index,k) mapping. Build: hipcc --offload-arch=gfx950 -O2.

#include <hip/hip_runtime.h>

#include <cstdio>

using f64x4 = __attribute__((__vector_size__(4 * sizeof(double)))) double;

__global__ void probe(const double* a_in, const double* b_in, double* d_out) {
  int l = threadIdx.x;
  f64x4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a_in[l], b_in[l], acc, 0, 0, 0);
  for (int r = 0; r < 4; r++) d_out[l * 4 + r] = acc[r];
}

int main() {
  double ha[64], hb[64], hd[256];
  double *da, *db, *dd;
  (void)hipMalloc(&da, sizeof ha);
  (void)hipMalloc(&db, sizeof hb);
  (void)hipMalloc(&dd, sizeof hd);

  auto run = [&](const char* name) {
    (void)hipMemcpy(da, ha, sizeof ha, hipMemcpyHostToDevice);
    (void)hipMemcpy(db, hb, sizeof hb, hipMemcpyHostToDevice);
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, da, db, dd);
    (void)hipDeviceSynchronize();
    (void)hipMemcpy(hd, dd, sizeof hd, hipMemcpyDeviceToHost);
    printf("== %s ==\n", name);
    for (int l = 0; l < 64; l++) {
      printf("lane %2d: %8.1f %8.1f %8.1f %8.1f\n", l, hd[l * 4], hd[l * 4 + 1],
             hd[l * 4 + 2], hd[l * 4 + 3]);
    }
  };

  // probe 1: a encodes candidate row (l&15), b = 1
  // if A map is A[i=l&15][k=l>>4] and B[k][j]=1: D[i][j] = 4*i
  for (int l = 0; l < 64; l++) {
    ha[l] = (double)(l & 15);
    hb[l] = 1.0;
  }
  run("A=(l&15), B=1  -> D[i][j]=4*i if A[i=l&15][k=l>>4]");

  // probe 2: b encodes candidate col, a = 1 -> D[i][j] = 4*j
  for (int l = 0; l < 64; l++) {
    ha[l] = 1.0;
    hb[l] = (double)(l & 15);
  }
  run("A=1, B=(l&15)  -> D[i][j]=4*j if B[k=l>>4][j=l&15]");

  // probe 3: a encodes k-slot as 10^k, b = 1 -> D[i][j] = 1111 if k=l>>4
  for (int l = 0; l < 64; l++) {
    double p = 1;
    for (int t = 0; t < (l >> 4); t++) p *= 10;
    ha[l] = p;
    hb[l] = 1.0;
  }
  run("A=10^(l>>4), B=1 -> D=1111 everywhere if k=l>>4");

  // probe 4: full lane id in a, b = 1 -> D[i][j] = sum of lane ids of row i
  for (int l = 0; l < 64; l++) {
    ha[l] = (double)l;
    hb[l] = 1.0;
  }
  run("A=l, B=1       -> D[i][*] = sum of the 4 lane ids feeding row i");
  return 0;
}
