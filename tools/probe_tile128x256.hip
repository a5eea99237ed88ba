// probe_tile128x256.hip — occupancy experiment for the fwd fused loss.
//
// Hypothesis (from the CDNA4 ladder data): staging bandwidth per CU scales
// with blocks/CU because co-resident blocks' L2/L3 hits overlap — the
// guide's 128² kernels reach ~56 GB/s/CU of staging at 2-3 blocks/CU while
// our 1-block/CU 256² kernel gets ~26.  A 128×256 tile with 64-register
// accumulators (wave tile 64×64) fits 2 blocks/CU at __launch_bounds__(512,4).
//
// G0 = production 256² geometry (1 block/CU), G1 = 128×256 @ 4 waves/SIMD.
// Interleaved A/B rounds; loss value printed as a checksum (both must agree).
//
// Build+run: hipcc --offload-arch=gfx950 -O3 -std=c++17
//            tools/probe_tile128x256.hip -o /tmp/probe_t && /tmp/probe_t

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef const __attribute__((address_space(1))) unsigned int* gas_ptr;
typedef __attribute__((address_space(3))) unsigned int* las_ptr;

__device__ __forceinline__ float softplus_f(float x) {
  return fmaxf(x, 0.0f) + __logf(1.0f + __expf(-fabsf(x)));
}
__device__ __forceinline__ int kmask(int r) {
  return (((r >> 1) & 1) << 2) | ((r >> 2) & 3);
}
__device__ __forceinline__ int kmask2(int r) { return (r >> 2) & 3; }

// ------------ G0: production 256², BK=64, 8 waves, 1 block/CU ------------
__launch_bounds__(512) __global__ void fwd_g0(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, int b, int n, int d) {
  constexpr int BM = 256, FM = 8, FN = 4, ROW = 128, TILE = 256 * 128;
  __shared__ char smem[4 * TILE];
  const int row_base = blockIdx.x * BM, col_base = blockIdx.y * 256;
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  const int wrow = (wave >> 2) * 128, wcol = (wave & 3) * 64;
  f32x4 acc[FM][FN];
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) acc[mi][ni] = f32x4{0, 0, 0, 0};
  const int ktiles = d / 64;
  const int fr = lane & 15, qbase = lane >> 4;
  const int mk = kmask(fr);
  int aAddr[2], bAddr[2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int ch = (kk * 4 + qbase) ^ mk;
    aAddr[kk] = (wrow + fr) * ROW + ch * 16;
    bAddr[kk] = TILE + (wcol + fr) * ROW + ch * 16;
  }
  const int rsub = lane >> 3, cch = lane & 7;
  int va[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int rloc = (wave * 4 + j) * 8 + rsub;
    va[j] = rloc * d * 2 + ((cch ^ kmask(rloc)) * 16);
  }
  const char* abase = zimg + (size_t)row_base * d * 2;
  const char* bbase = ztxt + (size_t)col_base * d * 2;
  auto stage = [&](int buf) {
    const int lb = __builtin_amdgcn_readfirstlane(wave * 4096) +
        buf * (2 * TILE);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      __builtin_amdgcn_global_load_lds((gas_ptr)(abase + va[j]),
                                       (las_ptr)(smem + lb + j * 1024), 16,
                                       0, 0);
      __builtin_amdgcn_global_load_lds(
          (gas_ptr)(bbase + va[j]), (las_ptr)(smem + lb + TILE + j * 1024),
          16, 0, 0);
    }
    abase += 128;
    bbase += 128;
  };
  stage(0);
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) {
      stage((kt + 1) & 1);
      asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[FM], bf[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
        af[mi] = *reinterpret_cast<const bf16x8*>(smem + aAddr[kk] +
                                                  mi * (16 * ROW));
#pragma unroll
      for (int ni = 0; ni < FN; ++ni)
        bf[ni] = *reinterpret_cast<const bf16x8*>(smem + bAddr[kk] +
                                                  ni * (16 * ROW));
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    aAddr[0] ^= 2 * TILE; aAddr[1] ^= 2 * TILE;
    bAddr[0] ^= 2 * TILE; bAddr[1] ^= 2 * TILE;
    asm volatile("s_barrier" ::: "memory");
  }
  float s0 = 0.f;
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        s0 += softplus_f(acc[mi][ni][reg] * t + bias);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) s0 += __shfl_down(s0, off);
  if (lane == 0) atomicAdd(out, s0);
}

// -------- G1: 128×256 tile, BK=32 (64-B rows), 8 waves, 2 blocks/CU -------
// Wave grid 2(M)×4(N), wave tile 64×64 → acc 4×4 fragments (64 VGPRs).
__launch_bounds__(512, 4) __global__ void fwd_g1(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, int b, int n, int d) {
  constexpr int FM = 4, FN = 4, ROW = 64;
  constexpr int A_BYTES = 128 * ROW;          //  8 KiB
  constexpr int B_BYTES = 256 * ROW;          // 16 KiB
  constexpr int STEP = A_BYTES + B_BYTES;     // 24 KiB per K-step buffer
  __shared__ char smem[2 * STEP];             // 48 KiB → 3 LDS-fit, 2 by VGPR
  const int row_base = blockIdx.x * 128, col_base = blockIdx.y * 256;
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  const int wrow = (wave >> 2) * 64, wcol = (wave & 3) * 64;
  f32x4 acc[FM][FN];
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) acc[mi][ni] = f32x4{0, 0, 0, 0};
  const int ktiles = d / 32;
  const int fr = lane & 15, q = lane >> 4;
  int aAddr = (wrow + fr) * ROW + ((q ^ kmask2(wrow + fr)) * 16);
  int bAddr = A_BYTES + (wcol + fr) * ROW + ((q ^ kmask2(wcol + fr)) * 16);
  // DMA: 24 KiB per K-step = 24 wave-instrs; wave w issues 3:
  //   id = w*3+j; ids 0..7 → A rows id*16..+16; ids 8..23 → B rows
  //   (id-8)*16..+16.
  const int rsub = lane >> 2, cch = lane & 3;
  int vsrc[3], ldst[3];
  bool is_a[3];
#pragma unroll
  for (int j = 0; j < 3; ++j) {
    const int id = wave * 3 + j;
    const bool a = id < 8;
    const int rloc = (a ? id : id - 8) * 16 + rsub;
    is_a[j] = a;
    vsrc[j] = rloc * d * 2 + ((cch ^ kmask2(rloc)) * 16);
    ldst[j] = a ? id * 1024 : A_BYTES + (id - 8) * 1024;
  }
  const char* abase = zimg + (size_t)row_base * d * 2;
  const char* bbase = ztxt + (size_t)col_base * d * 2;
  auto stage = [&](int buf) {
    const int boff = buf * STEP;
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      const char* src = (is_a[j] ? abase : bbase) + vsrc[j];
      const int lb = __builtin_amdgcn_readfirstlane(ldst[j]) + boff;
      __builtin_amdgcn_global_load_lds((gas_ptr)src, (las_ptr)(smem + lb),
                                       16, 0, 0);
    }
    abase += ROW;
    bbase += ROW;
  };
  stage(0);
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 1 < ktiles) {
      stage((kt + 1) & 1);
      asm volatile("s_waitcnt vmcnt(3)\n\ts_barrier" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
    }
    {
      bf16x8 af[FM], bf[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
        af[mi] = *reinterpret_cast<const bf16x8*>(smem + aAddr +
                                                  mi * (16 * ROW));
#pragma unroll
      for (int ni = 0; ni < FN; ++ni)
        bf[ni] = *reinterpret_cast<const bf16x8*>(smem + bAddr +
                                                  ni * (16 * ROW));
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    aAddr ^= STEP;
    bAddr ^= STEP;
    asm volatile("s_barrier" ::: "memory");
  }
  float s0 = 0.f;
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        s0 += softplus_f(acc[mi][ni][reg] * t + bias);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) s0 += __shfl_down(s0, off);
  if (lane == 0) atomicAdd(out, s0);
}

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { \
  printf("HIP error %d at %d\n", e, __LINE__); exit(1); } } while (0)

template <typename K>
float run(K kern, int bx, int by, const char* zi, const char* zt, float* out,
          int b, int n, int d, int iters, float* loss) {
  dim3 grid(bx, by);
  hipEvent_t e0, e1;
  HIP_CHECK(hipEventCreate(&e0));
  HIP_CHECK(hipEventCreate(&e1));
  HIP_CHECK(hipMemset(out, 0, 4));
  hipLaunchKernelGGL(kern, grid, dim3(512), 0, 0, zi, zt, 10.f, -10.f, out,
                     b, n, d);
  HIP_CHECK(hipMemcpy(loss, out, 4, hipMemcpyDeviceToHost));
  for (int i = 0; i < 2; ++i)
    hipLaunchKernelGGL(kern, grid, dim3(512), 0, 0, zi, zt, 10.f, -10.f, out,
                       b, n, d);
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(e0));
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL(kern, grid, dim3(512), 0, 0, zi, zt, 10.f, -10.f, out,
                       b, n, d);
  HIP_CHECK(hipEventRecord(e1));
  HIP_CHECK(hipDeviceSynchronize());
  float ms;
  HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
  return ms / iters;
}

int main() {
  const int b = 16384, n = 16384, d = 768, iters = 10, rounds = 3;
  size_t bytes = (size_t)b * d * 2;
  char *zi, *zt;
  float* out;
  HIP_CHECK(hipMalloc(&zi, bytes));
  HIP_CHECK(hipMalloc(&zt, bytes));
  HIP_CHECK(hipMalloc(&out, 4));
  std::vector<unsigned short> host(b * (size_t)d);
  srand(42);
  for (auto& v : host) {
    float f = (rand() / (float)RAND_MAX - 0.5f) * 0.07f;
    unsigned u;
    __builtin_memcpy(&u, &f, 4);
    v = (unsigned short)(u >> 16);
  }
  HIP_CHECK(hipMemcpy(zi, host.data(), bytes, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(zt, host.data(), bytes, hipMemcpyHostToDevice));
  const double flops = 2.0 * b * n * d;
  float l0 = 0, l1 = 0, b0 = 1e9f, b1 = 1e9f;
  for (int r = 0; r < rounds; ++r) {
    b0 = fminf(b0, run(fwd_g0, b / 256, n / 256, zi, zt, out, b, n, d, iters,
                       &l0));
    b1 = fminf(b1, run(fwd_g1, b / 128, n / 256, zi, zt, out, b, n, d, iters,
                       &l1));
  }
  printf("G0 256x256 (1 blk/CU): %8.3f ms  %7.1f TF/s  loss %.1f\n", b0,
         flops / b0 / 1e9, l0);
  printf("G1 128x256 (2 blk/CU): %8.3f ms  %7.1f TF/s  loss %.1f\n", b1,
         flops / b1 / 1e9, l1);
  printf("rel err %.2e\n", fabsf(l0 - l1) / fabsf(l0));
  return 0;
}
