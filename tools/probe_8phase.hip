// probe_8phase.hip — 8-phase fragment-unit pipeline vs the production loop.
//
// P8 structure (derived to satisfy the CDNA4 guide's 8-phase constraints):
//   - iteration = 2 K-tiles (K advance 128 bf16); 4 phases per K-tile window,
//     phase q = quadrant (mh=q>>1, nh=q&1) of every wave's 128×64 sub-tile.
//   - LDS = 8 fragment-aligned 16-KiB units (2 tiles resident):
//       A-unit[h] = zimg rows {r : (r>>6)&1 == h}  (the rows the waves' mi
//                   half h fragments read), 128 rows × 128 B;
//       B-unit[h] = ztxt rows {r : (r>>5)&1 == h}, same shape.
//   - stage stream, one unit per phase, lagging retirement by one barrier:
//       window of tile τ stages [B0(τ+1), B1(τ+1), A1(τ+1), A0(τ+2)].
//   - counted s_waitcnt vmcnt(2) only at window boundaries (phases 3/7);
//     vmcnt(0) at the last two boundaries (tail stages nothing).
//   - phase = [s_barrier][ds_read frags][stage unit][boundary wait][16 MFMA];
//     A fragments persist across the two nh phases of their mh half.
//
// Prints G0 (production) and P8 times + loss checksums (must agree).

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

// ------------------------- G0: production loop ---------------------------
__launch_bounds__(512) __global__ void fwd_g0(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, int b, int n, int d) {
  constexpr int FM = 8, FN = 4, ROW = 128, TILE = 256 * 128;
  __shared__ char smem[4 * TILE];
  const int row_base = blockIdx.x * 256, col_base = blockIdx.y * 256;
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

// --------------------------- P8: 8-phase pipeline ------------------------
constexpr int UNIT = 128 * 128;       // 16 KiB fragment-aligned unit
// smem layout: 8 slots of UNIT: slot(unit, par) where unit∈{A0,A1,B0,B1}:
//   base = par*4*UNIT + unit_index*UNIT, unit_index: A0=0,A1=1,B0=2,B1=3.

__launch_bounds__(512) __global__ void fwd_p8(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, int b, int n, int d) {
  constexpr int FM = 8, FN = 4, ROW = 128;
  __shared__ char smem[8 * UNIT];
  const int row_base = blockIdx.x * 256, col_base = blockIdx.y * 256;
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  const int wrow = (wave >> 2) * 128, wcol = (wave & 3) * 64;
  f32x4 acc[FM][FN];
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) acc[mi][ni] = f32x4{0, 0, 0, 0};

  const int ntiles = d / 64;          // K-tiles (BK=64); ntiles even req'd
  const int fr = lane & 15, qbase = lane >> 4;
  const int mk = kmask(fr);
  // Fragment read bases within a unit:
  //   A: global row r = wrow + mh*64 + mi'*16 + fr → unit_row =
  //      mi'*16 + fr + (wrow>>1)   (wrow∈{0,128} → +0/+64)
  //   B: r = wcol + nh*32 + ni'*16 + fr → unit_row =
  //      ni'*16 + fr + (wcol>>6)*32
  // kmask depends on r&15 = fr&15 → unchanged under the unit remap of
  // bits ≥4?  unit_row&15 = fr&15 ✓ (mi'*16, offsets multiples of 16).
  int aBase[2], bBase[2];               // per kk chunk
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int ch = (kk * 4 + qbase) ^ mk;
    aBase[kk] = (fr + (wrow >> 1)) * ROW + ch * 16;
    bBase[kk] = (fr + (wcol >> 6) * 32) * ROW + ch * 16;
  }

  // Staging: one unit = 16 wave-instrs = 2 per wave; lane covers 8 rows of
  // 128 B: unit rows (wave*2+j)*8 + (lane>>3), chunk lane&7 (pre-swizzled).
  const int rsub = lane >> 3, cch = lane & 7;
  int ur[2], swz[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    ur[j] = (wave * 2 + j) * 8 + rsub;
    swz[j] = (cch ^ kmask(ur[j])) * 16;   // unit_row&15 == global r&15? see below
  }
  // NOTE: global row r for A-unit h: r = (ur&63) + ((ur>>6)<<7) + h*64.
  // r&15 == ur&15 ✓ so the same kmask applies on both sides.
  const char* zi0 = zimg + (size_t)row_base * d * 2;
  const char* zt0 = ztxt + (size_t)col_base * d * 2;

  int koffA = 0, koffB = 0;  // byte K-offsets advance per staged tile-unit
  // Stage unit: ut 0=A0 1=A1 2=B0 3=B1, tile parity par, K byte offset koff.
  auto stage_unit = [&](int ut, int par, int koff) {
    const int lb0 = __builtin_amdgcn_readfirstlane(wave * 2048);
    const int base = par * 4 * UNIT + ut * UNIT + lb0;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int gr;
      const char* src;
      if (ut < 2) {
        gr = (ur[j] & 63) + ((ur[j] >> 6) << 7) + (ut & 1) * 64;
        src = zi0 + (size_t)gr * d * 2 + koff + swz[j];
      } else {
        gr = (ur[j] & 31) + ((ur[j] >> 5) << 6) + (ut & 1) * 32;
        src = zt0 + (size_t)gr * d * 2 + koff + swz[j];
      }
      __builtin_amdgcn_global_load_lds((gas_ptr)src,
                                       (las_ptr)(smem + base + j * 1024),
                                       16, 0, 0);
    }
  };

  // Prologue: tile0's 4 units + A0(tile1); wait all but the last in flight.
  stage_unit(0, 0, 0);                 // A0(0)
  stage_unit(2, 0, 0);                 // B0(0)
  stage_unit(3, 0, 0);                 // B1(0)
  stage_unit(1, 0, 0);                 // A1(0)
  stage_unit(0, 1, 128);               // A0(1)
  asm volatile("s_waitcnt vmcnt(2)\n\ts_barrier" ::: "memory");

  bf16x8 af[4][2];                     // A fragments persist across nh pair
  for (int kt = 0; kt < ntiles; ++kt) {
    const int par = kt & 1;
    const int kA = (kt + 2) * 128;     // K byte offset of A0(τ+2)
    const int kN = (kt + 1) * 128;     // of τ+1's units
    const bool s1 = kt + 1 < ntiles;   // stage τ+1 units?
    const bool s2 = kt + 2 < ntiles;   // stage A0(τ+2)?
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int mh = q >> 1, nh = q & 1;
      // ds_read fragments: A (8) on nh==0, B (4) every phase.
      if (nh == 0) {
#pragma unroll
        for (int mi2 = 0; mi2 < 4; ++mi2)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            af[mi2][kk] = *reinterpret_cast<const bf16x8*>(
                smem + (par * 4 + mh) * UNIT + aBase[kk] +
                mi2 * (16 * ROW));
      }
      bf16x8 bf[2][2];
#pragma unroll
      for (int ni2 = 0; ni2 < 2; ++ni2)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bf[ni2][kk] = *reinterpret_cast<const bf16x8*>(
              smem + (par * 4 + 2 + nh) * UNIT + bBase[kk] +
              ni2 * (16 * ROW));
      // stage stream: q0→B0(τ+1), q1→B1(τ+1), q2→A1(τ+1), q3→A0(τ+2)
      if (q == 0 && s1) stage_unit(2, par ^ 1, kN);
      else if (q == 1 && s1) stage_unit(3, par ^ 1, kN);
      else if (q == 2 && s1) stage_unit(1, par ^ 1, kN);
      else if (q == 3 && s2) stage_unit(0, par, kA);
      if (q == 3) {
        if (s2) {
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        } else {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi2 = 0; mi2 < 4; ++mi2)
#pragma unroll
        for (int ni2 = 0; ni2 < 2; ++ni2)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[mh * 4 + mi2][nh * 2 + ni2] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af[mi2][kk], bf[ni2][kk],
                    acc[mh * 4 + mi2][nh * 2 + ni2], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      asm volatile("s_barrier" ::: "memory");
    }
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
float run(K kern, const char* zi, const char* zt, float* out, int b, int n,
          int d, int iters, float* loss) {
  dim3 grid(b / 256, n / 256);
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

int main(int argc, char** argv) {
  const int b = 16384, n = 16384, iters = 10, rounds = 4;
  const int d = argc > 1 ? atoi(argv[1]) : 768;
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
    b0 = fminf(b0, run(fwd_g0, zi, zt, out, b, n, d, iters, &l0));
    b1 = fminf(b1, run(fwd_p8, zi, zt, out, b, n, d, iters, &l1));
  }
  printf("G0 production : %8.3f ms  %7.1f TF/s  loss %.2f\n", b0,
         flops / b0 / 1e9, l0);
  printf("P8 8-phase    : %8.3f ms  %7.1f TF/s  loss %.2f\n", b1,
         flops / b1 / 1e9, l1);
  printf("rel err %.3e  %s\n", fabsf(l0 - l1) / fabsf(l0),
         fabsf(l0 - l1) / fabsf(l0) < 1e-4 ? "MATCH" : "MISMATCH");
  return 0;
}
