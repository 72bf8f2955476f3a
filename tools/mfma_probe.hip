// Empirical probe of the v_mfma_f64_16x16x4_f64 operand/result lane
// mapping on gfx950.  Computes D = A*B for A[16][4], B[4][16] filled with
// distinct values under an ASSUMED mapping (lane l supplies
// A[l&15][l>>4] and B[l>>4][l&15]; lane l reg v holds D[(l>>4)*4+v][l&15])
// and compares against the host triple loop; prints a per-assumption
// verdict plus, on mismatch, the inferred true mapping.
// RESULT (gfx950, ROCm 7.2): A/B mapping as assumed; the D map is
// col = lane&15, row = 4*reg + (lane>>4)  (NOT (lane>>4)*4 + reg).
//   build & run (GPU box):  hipcc --offload-arch=gfx950 tools/mfma_probe.hip
//                           -o /tmp/mfma_probe && /tmp/mfma_probe
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>

typedef double d4 __attribute__((ext_vector_type(4)));

__global__ void kProbe(const double* A, const double* B, double* D) {
  const int l = threadIdx.x;
  const int i = l & 15;
  const int k = l >> 4;
  const double av = A[i * 4 + k];   // assumed: lane holds A[row=l&15][k=l>>4]
  const double bv = B[k * 16 + i];  // assumed: lane holds B[k=l>>4][col=l&15]
  d4 acc = {0, 0, 0, 0};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(av, bv, acc, 0, 0, 0);
  // assumed D map: row=(l>>4)*4+v, col=l&15
  for (int v = 0; v < 4; ++v) D[((l >> 4) * 4 + v) * 16 + (l & 15)] = acc[v];
}

int main() {
  double hA[64], hB[64], hD[256], ref[256];
  // asymmetric fills so transposes are visible
  for (int m = 0; m < 16; ++m)
    for (int k = 0; k < 4; ++k) hA[m * 4 + k] = 1.0 + m * 0.25 + k * 7.0;
  for (int k = 0; k < 4; ++k)
    for (int n = 0; n < 16; ++n) hB[k * 16 + n] = 0.5 + k * 3.0 + n * 0.125;
  for (int m = 0; m < 16; ++m)
    for (int n = 0; n < 16; ++n) {
      double s = 0;
      for (int k = 0; k < 4; ++k) s += hA[m * 4 + k] * hB[k * 16 + n];
      ref[m * 16 + n] = s;
    }
  double *dA, *dB, *dD;
  (void)hipMalloc(&dA, sizeof(hA));
  (void)hipMalloc(&dB, sizeof(hB));
  (void)hipMalloc(&dD, sizeof(hD));
  (void)hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(kProbe, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  (void)hipMemcpy(hD, dD, sizeof(hD), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 256; ++i)
    if (std::fabs(hD[i] - ref[i]) > 1e-9) ++bad;
  printf("assumed mapping: %s (%d/256 mismatched)\n",
         bad == 0 ? "CORRECT" : "WRONG", bad);
  if (bad) {
    // is it the transpose?
    int badT = 0;
    for (int m = 0; m < 16; ++m)
      for (int n = 0; n < 16; ++n)
        if (std::fabs(hD[n * 16 + m] - ref[m * 16 + n]) > 1e-9) ++badT;
    printf("transposed D map: %s (%d/256)\n",
           badT == 0 ? "CORRECT" : "also wrong", badT);
    printf("D[0][0..7]:   ");
    for (int n = 0; n < 8; ++n) printf("%8.2f ", hD[n]);
    printf("\nref[0][0..7]: ");
    for (int n = 0; n < 8; ++n) printf("%8.2f ", ref[n]);
    printf("\nD[1][0..7]:   ");
    for (int n = 0; n < 8; ++n) printf("%8.2f ", hD[16 + n]);
    printf("\nref[1][0..7]: ");
    for (int n = 0; n < 8; ++n) printf("%8.2f ", ref[16 + n]);
    printf("\n");
    // dump full D and ref for offline analysis
    FILE* f = fopen("gpurun_out/mfma_probe_dump.txt", "w");
    if (f) {
      for (int i = 0; i < 256; ++i)
        fprintf(f, "%d %.17g %.17g\n", i, hD[i], ref[i]);
      fclose(f);
    }
  }
  return bad ? 1 : 0;
}
