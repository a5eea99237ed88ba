// ablate_fwd.hip — standalone ablation probe for the fwd fused-loss main loop.
//
// Variants (template<int ABL>) of the 256²/BK=64/8-wave interior loop:
//   0 = full kernel (stage + wait/barrier + ds_read + MFMA + barrier)
//   1 = no second barrier            (cost of the post-compute barrier)
//   2 = no staging/waits             (ds_read + MFMA on stale LDS)
//   3 = MFMA only                    (fragments kept live via asm)
//   4 = no MFMA                      (stage + waits + ds_read only)
//   5 = decoupled DMA                (stage into unread dummy region; compute
//                                     on static buffers — co-run without the
//                                     producer-consumer coupling)
//   6 = register DMA                 (plain global_load to discarded regs; no
//                                     LDS writes — HBM read + MFMA co-run)
//   7 = dummy DMA + MFMA-only        (no ds_reads at all: separates LDS-port
//                                     contention from clock/issue co-run)
// Results are wrong for ABL>0 — perf-diagnostic only.  Prints ms and TF/s
// per variant, interleaved rounds (guide §5.4 rule 24).
//
// Build+run: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/ablate_fwd.hip
//            -o /tmp/ablate_fwd && /tmp/ablate_fwd

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef const __attribute__((address_space(1))) unsigned int* gas_ptr;
typedef __attribute__((address_space(3))) unsigned int* las_ptr;

constexpr int BM = 256, BN = 256, THREADS = 512, FM = 8, FN = 4;
constexpr int ROW_BYTES = 128;
constexpr int TILE_BYTES = BM * ROW_BYTES;

__device__ __forceinline__ float softplus_f(float x) {
  return fmaxf(x, 0.0f) + __logf(1.0f + __expf(-fabsf(x)));
}
__device__ __forceinline__ int kmask(int r) {
  return (((r >> 1) & 1) << 2) | ((r >> 2) & 3);
}

template <int ABL>
__launch_bounds__(THREADS) __global__ void fwd_kernel(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, int b, int n, int d) {
  __shared__ char smem[4 * TILE_BYTES];
  const int bx = blockIdx.x, by = blockIdx.y;
  const int row_base = bx * BM, col_base = by * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
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
    aAddr[kk] = (wrow + fr) * ROW_BYTES + ch * 16;
    bAddr[kk] = TILE_BYTES + (wcol + fr) * ROW_BYTES + ch * 16;
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
        buf * (2 * TILE_BYTES);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      __builtin_amdgcn_global_load_lds((gas_ptr)(abase + va[j]),
                                       (las_ptr)(smem + lb + j * 1024), 16,
                                       0, 0);
      __builtin_amdgcn_global_load_lds(
          (gas_ptr)(bbase + va[j]), (las_ptr)(smem + lb + TILE_BYTES + j * 1024),
          16, 0, 0);
    }
    abase += 128;
    bbase += 128;
  };

  auto stage_dummy = [&]() {
    // DMA the same bytes into buffer 1's region, never read by compute.
    const int lb = __builtin_amdgcn_readfirstlane(wave * 4096) +
        2 * TILE_BYTES;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      __builtin_amdgcn_global_load_lds((gas_ptr)(abase + va[j]),
                                       (las_ptr)(smem + lb + j * 1024), 16,
                                       0, 0);
      __builtin_amdgcn_global_load_lds(
          (gas_ptr)(bbase + va[j]), (las_ptr)(smem + lb + TILE_BYTES + j * 1024),
          16, 0, 0);
    }
    abase += 128;
    bbase += 128;
  };
  auto load_regs = [&]() {
    // Same HBM traffic, straight to registers, discarded.
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      uint4 va_ = *reinterpret_cast<const uint4*>(abase + va[j]);
      uint4 vb_ = *reinterpret_cast<const uint4*>(bbase + va[j]);
      asm volatile("" :: "v"(va_.x), "v"(va_.w), "v"(vb_.x), "v"(vb_.w));
    }
    abase += 128;
    bbase += 128;
  };

  if (ABL != 2 && ABL != 3 && ABL != 5 && ABL != 6 && ABL != 7) stage(0);
  for (int kt = 0; kt < ktiles; ++kt) {
    if (ABL == 5 || ABL == 7) {
      stage_dummy();
      asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
    } else if (ABL == 6) {
      load_regs();
      asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
    } else if (ABL != 2 && ABL != 3) {
      if (kt + 1 < ktiles) {
        stage((kt + 1) & 1);
        asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
      }
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 afrag[FM], bfrag[FN];
      if (ABL != 3 && ABL != 7) {
#pragma unroll
        for (int mi = 0; mi < FM; ++mi)
          afrag[mi] = *reinterpret_cast<const bf16x8*>(
              smem + aAddr[kk] + mi * (16 * ROW_BYTES));
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          bfrag[ni] = *reinterpret_cast<const bf16x8*>(
              smem + bAddr[kk] + ni * (16 * ROW_BYTES));
      } else {
        // fragments from accumulator bits; keep live without reads
#pragma unroll
        for (int mi = 0; mi < FM; ++mi) {
          afrag[mi] = bf16x8{};
          asm volatile("" : "+v"(afrag[mi]));
        }
#pragma unroll
        for (int ni = 0; ni < FN; ++ni) {
          bfrag[ni] = bf16x8{};
          asm volatile("" : "+v"(bfrag[ni]));
        }
      }
      if (ABL != 4) {
#pragma unroll
        for (int mi = 0; mi < FM; ++mi)
#pragma unroll
          for (int ni = 0; ni < FN; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
      } else {
#pragma unroll
        for (int mi = 0; mi < FM; ++mi) {
          asm volatile("" : "+v"(afrag[mi]));
        }
      }
    }
    aAddr[0] ^= 2 * TILE_BYTES; aAddr[1] ^= 2 * TILE_BYTES;
    bAddr[0] ^= 2 * TILE_BYTES; bAddr[1] ^= 2 * TILE_BYTES;
    if (ABL == 0 || ABL == 5 || ABL == 6 || ABL == 7)
      asm volatile("s_barrier" ::: "memory");
    else if (ABL == 2 || ABL == 3 || ABL == 4) __syncthreads();
    // ABL 1: no post-compute barrier
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

template <int ABL>
float run(const char* zi, const char* zt, float* out, int b, int n, int d,
          int iters) {
  dim3 grid(b / BM, n / BN);
  hipEvent_t e0, e1;
  HIP_CHECK(hipEventCreate(&e0));
  HIP_CHECK(hipEventCreate(&e1));
  for (int i = 0; i < 2; ++i)
    hipLaunchKernelGGL((fwd_kernel<ABL>), grid, dim3(THREADS), 0, 0, zi, zt,
                       10.f, -10.f, out, b, n, d);
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipEventRecord(e0));
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL((fwd_kernel<ABL>), grid, dim3(THREADS), 0, 0, zi, zt,
                       10.f, -10.f, out, b, n, d);
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
  const char* names[] = {"full", "no-2nd-barrier", "no-stage", "mfma-only",
                         "no-mfma", "decoupled-dma", "reg-dma",
                         "dma+mfma-no-lds"};
  float best[8] = {1e9f, 1e9f, 1e9f, 1e9f, 1e9f, 1e9f, 1e9f, 1e9f};
  for (int r = 0; r < rounds; ++r) {
    best[0] = fminf(best[0], run<0>(zi, zt, out, b, n, d, iters));
    best[1] = fminf(best[1], run<1>(zi, zt, out, b, n, d, iters));
    best[2] = fminf(best[2], run<2>(zi, zt, out, b, n, d, iters));
    best[3] = fminf(best[3], run<3>(zi, zt, out, b, n, d, iters));
    best[4] = fminf(best[4], run<4>(zi, zt, out, b, n, d, iters));
    best[5] = fminf(best[5], run<5>(zi, zt, out, b, n, d, iters));
    best[6] = fminf(best[6], run<6>(zi, zt, out, b, n, d, iters));
    best[7] = fminf(best[7], run<7>(zi, zt, out, b, n, d, iters));
  }
  for (int v = 0; v < 8; ++v)
    printf("%-16s %8.3f ms  %7.1f TF/s\n", names[v], best[v],
           flops / best[v] / 1e9);
  return 0;
}
