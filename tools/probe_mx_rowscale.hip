// probe_mx_rowscale.hip — validate ROW-wise e8m0 block scales on the gfx950
// MX-scaled MFMA (v_mfma_scale_f32_16x16x128_f8f6f4).
//
// Claim under test: for the 16x16x128 shape each lane holds exactly one
// 32-element K-block of A (row = lane&15, K-block = lane>>4) and one of B
// (col = lane&15), so a PER-ROW scale s_i = 2^k (e8m0 byte 127+k, replicated
// across the scale dword so any byte-select matches) makes the MFMA emit the
// EXACT scaled dot  Σ_k (qa·2^ka)·(qb·2^kb)  — i.e. hardware dequantization
// with per-row dynamic range instead of the production kernels' per-tensor
// fold into the temperature.  A standalone parity check against an fp32
// host reference; adopting it in the production epilogue is a round-3 item
// (see NOTES_ROUND3.md).
//
// Build+run (GPU box):
//   hipcc --offload-arch=gfx950 -O3 tools/probe_mx_rowscale.hip -o /tmp/mxp && /tmp/mxp

#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define CHK(x) do { hipError_t e = (x); if (e) { \
  printf("HIP error %d at %d\n", (int)e, __LINE__); exit(1); } } while (0)

// One wave computes a 16x16 tile over K=128: A (16,128) e4m3 + row scales,
// B (16,128) e4m3 + row scales (B "rows" are output columns).
__global__ void mx_tile_kernel(const unsigned char* qa,
                               const unsigned char* qb,
                               const unsigned char* ea,
                               const unsigned char* eb,
                               float* out /*16x16*/) {
  const int lane = threadIdx.x & 63;
  const int row = lane & 15;
  const int kb = lane >> 4;          // K-block 0..3 (32 elements each)
  i32x8 af, bf;
  const unsigned char* pa = qa + row * 128 + kb * 32;
  const unsigned char* pb = qb + row * 128 + kb * 32;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    af[i] = reinterpret_cast<const int*>(pa)[i];
    bf[i] = reinterpret_cast<const int*>(pb)[i];
  }
  const unsigned sa = 0x01010101u * ea[row];
  const unsigned sb = 0x01010101u * eb[row];
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      af, bf, acc, 0, 0, 0, (int)sa, 0, (int)sb);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = (lane >> 4) * 4 + r;
    const int ocol = lane & 15;
    out[orow * 16 + ocol] = acc[r];
  }
}

int main() {
  srand(7);
  const int R = 16, K = 128;
  std::vector<float> A(R * K), B(R * K);
  std::vector<unsigned char> qa(R * K), qb(R * K), ea(R), eb(R);
  // Rows with wildly different magnitudes (the per-tensor fold's weak spot).
  auto quant_row = [](float* x, unsigned char* q, int n, unsigned char* e8) {
    float amax = 0.f;
    for (int i = 0; i < n; ++i) amax = fmaxf(amax, fabsf(x[i]));
    int k = (int)ceilf(log2f(fmaxf(amax, 1e-30f) / 448.0f));
    if (k < -126) k = -126;
    *e8 = (unsigned char)(127 + k);
    const float r = exp2f((float)-k);
    for (int i = 0; i < n; ++i) {
      __hip_fp8_e4m3 v(x[i] * r);
      q[i] = v.__x;
      x[i] = (float)v * exp2f((float)k);   // reference sees quantized value
    }
  };
  for (int r = 0; r < R; ++r) {
    const float mag_a = exp2f((float)(rand() % 24 - 12));
    const float mag_b = exp2f((float)(rand() % 24 - 12));
    for (int k = 0; k < K; ++k) {
      A[r * K + k] = mag_a * ((rand() / (float)RAND_MAX) * 2 - 1);
      B[r * K + k] = mag_b * ((rand() / (float)RAND_MAX) * 2 - 1);
    }
    quant_row(&A[r * K], &qa[r * K], K, &ea[r]);
    quant_row(&B[r * K], &qb[r * K], K, &eb[r]);
  }
  unsigned char *dqa, *dqb, *dea, *deb;
  float* dout;
  CHK(hipMalloc(&dqa, R * K));
  CHK(hipMalloc(&dqb, R * K));
  CHK(hipMalloc(&dea, R));
  CHK(hipMalloc(&deb, R));
  CHK(hipMalloc(&dout, R * 16 * sizeof(float)));
  CHK(hipMemcpy(dqa, qa.data(), R * K, hipMemcpyHostToDevice));
  CHK(hipMemcpy(dqb, qb.data(), R * K, hipMemcpyHostToDevice));
  CHK(hipMemcpy(dea, ea.data(), R, hipMemcpyHostToDevice));
  CHK(hipMemcpy(deb, eb.data(), R, hipMemcpyHostToDevice));
  hipLaunchKernelGGL(mx_tile_kernel, dim3(1), dim3(64), 0, 0, dqa, dqb, dea,
                     deb, dout);
  CHK(hipDeviceSynchronize());
  std::vector<float> out(R * 16);
  CHK(hipMemcpy(out.data(), dout, out.size() * 4, hipMemcpyDeviceToHost));
  // fp32 reference on the dequantized values: exact match expected up to
  // fp32 accumulation order.
  // Error normalized by Σ|a·b| (the accumulation's natural scale) — plain
  // rel-to-ref explodes on cancellation-heavy dots.
  double max_rel = 0.0;
  for (int i = 0; i < R; ++i)
    for (int j = 0; j < R; ++j) {
      double ref = 0, mag = 0;
      for (int k = 0; k < K; ++k) {
        const double p = (double)A[i * K + k] * B[j * K + k];
        ref += p;
        mag += fabs(p);
      }
      const double got = out[i * 16 + j];
      const double rel = fabs(got - ref) / fmax(mag, 1e-20);
      if (rel > max_rel) max_rel = rel;
    }
  printf("max |got-ref|/sum|a*b| = %.3e over 16x16 "
         "(row magnitudes spanning 2^-12..2^12)\n", max_rel);
  printf(max_rel < 1e-5 ? "MX ROW-SCALE PARITY: PASS\n"
                        : "MX ROW-SCALE PARITY: FAIL\n");
  return max_rel < 1e-5 ? 0 : 1;
}
