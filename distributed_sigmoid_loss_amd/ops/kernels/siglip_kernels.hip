// siglip_kernels.hip — fused SigLIP sigmoid-loss kernels for MI355X (gfx950, CDNA4).
//
// Implements, per (b, n) block of image×text embeddings (bf16, row-major,
// K = emb dim contiguous):
//
//   forward:  loss += sum_ij softplus(-l_ij * (t * <zimg_i, ztxt_j> + bias))
//   backward: g_ij  = -l_ij * sigmoid(-l_ij * z_ij)   (written as bf16 slab)
//             scal[0] += sum g_ij * <zimg_i, ztxt_j>   (for dt')
//             scal[1] += sum g_ij                      (for dbias)
//
// with labels l_ij = +1 iff j == i + diag_offset (an index predicate — the
// (b,n) label matrix of the reference, distributed_sigmoid_loss.py:28-30 and
// rwightman_sigmoid_loss.py:43-47, is never materialized), l_ij = -1 else.
// The (b,n) logits matrix never leaves the MFMA accumulators in forward; the
// backward recomputes it tile-by-tile and emits only the g slab consumed by
// the two rocBLAS GEMMs (dzimg = t·g@ztxt, dztxt = t·gᵀ@zimg) on the Python
// side (ops/__init__.py).
//
// Kernel structure (one template, two epilogues):
//   - 128×128 output tile per 256-thread (4-wave) workgroup,
//     64×64 per wave as 4×4 fragments of v_mfma_f32_16x16x32_bf16.
//   - K-loop BK=64, double-buffered LDS (write-after-barrier pipeline:
//     global→reg loads issued before the MFMA phase, reg→LDS writes after).
//   - LDS rows padded to 144 B (BK*2 + 16) so the 16-lane ds_read_b128
//     fragment reads are bank-conflict-free (rows r: dword bank 36r mod 64,
//     all 16 distinct slots) while keeping 16-B alignment.
//   - fp32 accumulation throughout; scalar results via per-wave shuffle
//     reduction + one atomicAdd per wave.
//
// Requirements: d % 8 == 0 (16-byte K-vectors); b, n arbitrary (guarded).
// Compile: hipcc --offload-arch=gfx950 -O3 -shared -fPIC.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <climits>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace {

constexpr int BM = 128;        // image rows per block
constexpr int BN = 128;        // text rows (logit cols) per block
constexpr int BK = 64;         // K (emb dim) step
constexpr int THREADS = 256;   // 4 waves
constexpr int LDS_ROW = BK * 2 + 16;          // 144 B padded row
constexpr int LDS_TILE = BM * LDS_ROW;        // 18432 B per operand tile
constexpr int DIAG_NONE = INT_MIN;

__device__ __forceinline__ float softplus_f(float x) {
  // log(1 + e^x), stable for all x.
  return fmaxf(x, 0.0f) + log1pf(__expf(-fabsf(x)));
}

struct StageRegs {
  uint4 a[4];
  uint4 b[4];
};

// Issue the global loads for K-tile kt into registers (zero-filled outside
// [b|n, d)).  Thread t covers row t>>1 of each tile, 32 K-elements starting at
// (t&1)*32 — four 16-B vectors, coalesced along K.
__device__ __forceinline__ void stage_load(
    StageRegs& r, const __bf16* __restrict__ zimg,
    const __bf16* __restrict__ ztxt, int row_base, int col_base, int b, int n,
    int d, int k0) {
  const int t = threadIdx.x;
  const int row = t >> 1;
  const int kh = (t & 1) * 32;
  const int ga = row_base + row;
  const int gb = col_base + row;
  const uint4 zero = {0u, 0u, 0u, 0u};
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int k = k0 + kh + i * 8;
    r.a[i] = (ga < b && k < d)
        ? *reinterpret_cast<const uint4*>(zimg + (size_t)ga * d + k) : zero;
    r.b[i] = (gb < n && k < d)
        ? *reinterpret_cast<const uint4*>(ztxt + (size_t)gb * d + k) : zero;
  }
}

__device__ __forceinline__ void stage_write(const StageRegs& r, char* As,
                                            char* Bs) {
  const int t = threadIdx.x;
  const int row = t >> 1;
  const int off = row * LDS_ROW + (t & 1) * 64;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    *reinterpret_cast<uint4*>(As + off + i * 16) = r.a[i];
    *reinterpret_cast<uint4*>(Bs + off + i * 16) = r.b[i];
  }
}

// MODE 0: forward loss.  MODE 1: backward g-slab + scalar partials.
template <int MODE>
__launch_bounds__(THREADS) __global__ void siglip_tile_kernel(
    const __bf16* __restrict__ zimg, const __bf16* __restrict__ ztxt,
    const float* __restrict__ t_prime, const float* __restrict__ bias_p,
    float* __restrict__ out,          // MODE 0: loss[1].  MODE 1: scal[2].
    __bf16* __restrict__ g_out,       // MODE 1 only; leading dim = n
    int b, int n, int d, int diag) {
  __shared__ char smem[4 * LDS_TILE];

  const int row_base = blockIdx.x * BM;
  const int col_base = blockIdx.y * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wrow = (wave >> 1) * 64;   // wave's 64×64 sub-tile origin
  const int wcol = (wave & 1) * 64;

  const float t = __expf(*t_prime);
  const float bias = *bias_p;

  f32x4 acc[4][4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ktiles = (d + BK - 1) / BK;
  StageRegs regs;
  stage_load(regs, zimg, ztxt, row_base, col_base, b, n, d, 0);
  stage_write(regs, smem, smem + LDS_TILE);
  __syncthreads();

  for (int kt = 0; kt < ktiles; ++kt) {
    char* As = smem + (kt & 1) * (2 * LDS_TILE);
    char* Bs = As + LDS_TILE;
    const bool have_next = kt + 1 < ktiles;
    if (have_next)
      stage_load(regs, zimg, ztxt, row_base, col_base, b, n, d,
                 (kt + 1) * BK);

    // frag row/col within the wave tile: lane&15 selects the 16-row group
    // element; lane>>4 selects the 8-wide K-subgroup.
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        afrag[mi] = *reinterpret_cast<const bf16x8*>(
            As + (wrow + mi * 16 + fr) * LDS_ROW + (kk * 32 + fk) * 2);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(
            Bs + (wcol + ni * 16 + fr) * LDS_ROW + (kk * 32 + fk) * 2);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }

    __syncthreads();
    if (have_next) {
      char* An = smem + ((kt + 1) & 1) * (2 * LDS_TILE);
      stage_write(regs, An, An + LDS_TILE);
      __syncthreads();
    }
  }

  // Epilogue.  C/D fragment layout of mfma_f32_16x16x32_bf16:
  //   col = lane & 15, row = (lane >> 4) * 4 + reg.
  float s0 = 0.f, s1 = 0.f;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int grow = row_base + wrow + mi * 16 + (lane >> 4) * 4 + reg;
        const int gcol = col_base + wcol + ni * 16 + (lane & 15);
        if (grow < b && gcol < n) {
          const float dot = acc[mi][ni][reg];
          const float z = dot * t + bias;
          const bool pos = (diag != DIAG_NONE) && (gcol == grow + diag);
          if (MODE == 0) {
            s0 += softplus_f(pos ? -z : z);
          } else {
            // g = -l * sigmoid(-l z) = -l / (1 + exp(l z))
            const float g = pos ? (-1.0f / (1.0f + __expf(z)))
                                : (1.0f / (1.0f + __expf(-z)));
            g_out[(size_t)grow * n + gcol] = (__bf16)g;
            s0 += g * dot;
            s1 += g;
          }
        }
      }
    }
  }

  // Wave-level tree reduction, one atomicAdd per wave per scalar.
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s0 += __shfl_down(s0, off);
    if (MODE == 1) s1 += __shfl_down(s1, off);
  }
  if (lane == 0) {
    atomicAdd(&out[0], s0);
    if (MODE == 1) atomicAdd(&out[1], s1);
  }
}

inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

}  // namespace

extern "C" {

int siglip_ext_abi(void) { return 1; }

int siglip_fwd_bf16(uintptr_t stream, const void* zimg, const void* ztxt,
                    const void* t_prime, const void* bias, void* loss_out,
                    int b, int n, int d, int diag) {
  if (b <= 0 || n <= 0) return (int)hipSuccess;
  if (d % 8 != 0) return (int)hipErrorInvalidValue;
  dim3 grid(ceil_div(b, BM), ceil_div(n, BN));
  hipLaunchKernelGGL((siglip_tile_kernel<0>), grid, dim3(THREADS), 0,
                     (hipStream_t)stream,
                     (const __bf16*)zimg, (const __bf16*)ztxt,
                     (const float*)t_prime, (const float*)bias,
                     (float*)loss_out, (__bf16*)nullptr, b, n, d, diag);
  return (int)hipGetLastError();
}

int siglip_bwd_g_bf16(uintptr_t stream, const void* zimg, const void* ztxt,
                      const void* t_prime, const void* bias, void* g_out,
                      void* scal, int b, int n, int d, int diag) {
  if (b <= 0 || n <= 0) return (int)hipSuccess;
  if (d % 8 != 0) return (int)hipErrorInvalidValue;
  dim3 grid(ceil_div(b, BM), ceil_div(n, BN));
  hipLaunchKernelGGL((siglip_tile_kernel<1>), grid, dim3(THREADS), 0,
                     (hipStream_t)stream,
                     (const __bf16*)zimg, (const __bf16*)ztxt,
                     (const float*)t_prime, (const float*)bias,
                     (float*)scal, (__bf16*)g_out, b, n, d, diag);
  return (int)hipGetLastError();
}

}  // extern "C"
