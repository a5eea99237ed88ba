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
// Kernel structure (one template; MODE 0 = fwd, MODE 1 = bwd-g):
//   - 256×256 output tile per 512-thread (8-wave, 2×4) workgroup; each wave
//     owns a 128×64 sub-tile as 8×4 fragments of v_mfma_f32_16x16x32_bf16,
//     fp32 accumulate.  The 256² tile (vs 128²) halves the staged bytes per
//     FLOP — this op is staging-bandwidth-bound at large batch.
//   - K-loop BK=64, double-buffered LDS (4×32 KiB) staged by
//     global_load_lds_dwordx4 (direct HBM→LDS DMA, no VGPR round trip); the
//     DMA for tile k+1 is issued before the MFMA phase of tile k.
//   - LDS image is lane-linear (glds requirement), so the bank swizzle is
//     applied to the per-lane *source* address and the ds_read offset
//     (both-sides rule): chunk' = chunk ^ ((row&7 + row>>3&1) & 7), making
//     the 16-lane ds_read_b128 fragment reads bank-conflict-free.
//   - Edge blocks (ragged b/n, d%64≠0) take a register-staged path writing
//     the same swizzled LDS image with zero-fill guards; interior blocks
//     (the entire grid at benchmark shapes) take the DMA fast path.
//   - Block-id remap for cache locality (flags): bit0 = XCD-contiguous
//     remap, bit1 = grouped column-major walk (8 block-rows per group) so
//     temporally-close blocks share operand panels in L2/L3.
//   - MODE 1 stores the g tile through LDS (two 128-row passes, padded rows)
//     so global writes are 16-B vectors instead of 2-B scatters.
//
// Requirements: d % 8 == 0 (16-byte K-vectors); b, n arbitrary (guarded).
// Compile: hipcc --offload-arch=gfx950 -O3 -shared -fPIC.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <climits>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef const __attribute__((address_space(1))) unsigned int* gas_ptr;
typedef __attribute__((address_space(3))) unsigned int* las_ptr;

namespace {

constexpr int BM = 256;          // image rows per block
constexpr int BN = 256;          // text rows (logit cols) per block
constexpr int BK = 64;           // K (emb dim) step
constexpr int THREADS = 512;     // 8 waves as 2(M)×4(N)
constexpr int FM = 8;            // M fragments per wave (128 rows)
constexpr int FN = 4;            // N fragments per wave (64 cols)
constexpr int ROW_BYTES = BK * 2;            // 128 B linear row
constexpr int TILE_BYTES = BM * ROW_BYTES;   // 32 KiB per operand tile
constexpr int DIAG_NONE = INT_MIN;
constexpr int NXCD = 8;
constexpr int GROUP_M = 8;       // block-rows per locality group (bit1)

// g-tile epilogue staging: 128 rows of BN bf16, rows padded to 528 B.
constexpr int G_ROW = BN * 2 + 16;

// softplus via the inlined fast log/exp (log1pf is a device-lib CALL —
// measured as s_getpc/s_setpc pairs inside the epilogue).  For y = e^{-|x|}
// ∈ (0,1] the naive log(1+y) loses only ~y·ε absolute accuracy near y→0,
// where the term itself vanishes — fine at bf16-class tolerances.
__device__ __forceinline__ float softplus_f(float x) {
  return fmaxf(x, 0.0f) + __logf(1.0f + __expf(-fabsf(x)));
}

__device__ __forceinline__ float sigmoid_fast(float negz) {
  // 1/(1+e^{negz}) via the single-instruction v_rcp_f32.
  return __builtin_amdgcn_rcpf(1.0f + __expf(negz));
}

// Per-row XOR mask on the 16-B chunk index.  Depends only on r&15 (so
// fragment reads at row = base + mi*16 + fr share one per-lane mask) and is
// conflict-free for ds_read_b128 under both contiguous and interleaved
// 16-lane servicing groups: same-parity rows differing by 2 always differ
// in mask bit 2, and a fixed-q column read over 16 rows sees all 8 masks.
__device__ __forceinline__ int kmask(int r) {
  return (((r >> 1) & 1) << 2) | ((r >> 2) & 3);
}

// HBM→LDS DMA staging of one 256×64 bf16 tile (32 KiB).  Each of the 8 waves
// issues 4 global_load_lds_dwordx4: LDS dest = wave-uniform base + lane*16
// (lane-linear), per-lane source address carries the inverse swizzle.
__device__ __forceinline__ void stage_glds(const __bf16* __restrict__ gsrc,
                                           char* lds, int row0, int d,
                                           int k0) {
  const int w = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int rsub = l >> 3;         // row within the 8-row group
  const int c = l & 7;             // 16-B chunk within the row
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int rloc = (w * 4 + j) * 8 + rsub;
    const int sc = c ^ kmask(rloc);
    const __bf16* src = gsrc + (size_t)(row0 + rloc) * d + k0 + sc * 8;
    // The LDS base is wave-uniform in value but threadIdx-derived, which the
    // compiler treats as divergent — readfirstlane makes uniformity provable
    // (avoids a waterfall loop around each DMA).
    const int lbase = __builtin_amdgcn_readfirstlane((w * 4 + j) * 1024);
    __builtin_amdgcn_global_load_lds((gas_ptr)src, (las_ptr)(lds + lbase),
                                     16, 0, 0);
  }
}

// Register-staged fallback for edge blocks: same swizzled LDS image,
// zero-filled outside [rows, d).
__device__ __forceinline__ void stage_guarded(const __bf16* __restrict__ gsrc,
                                              char* lds, int row0, int rows,
                                              int d, int k0) {
  const int t = threadIdx.x;
  const int row = t >> 1;          // 0..255
  const int ch0 = (t & 1) * 4;
  const int gr = row0 + row;
  const uint4 zero = {0u, 0u, 0u, 0u};
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int c = ch0 + i;
    const int k = k0 + c * 8;
    uint4 v = (gr < rows && k < d)
        ? *reinterpret_cast<const uint4*>(gsrc + (size_t)gr * d + k) : zero;
    *reinterpret_cast<uint4*>(lds + row * ROW_BYTES +
                              ((c ^ kmask(row)) * 16)) = v;
  }
}

template <int MODE, bool INTERIOR>
__device__ __forceinline__ void tile_body(
    const __bf16* __restrict__ zimg, const __bf16* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, __bf16* __restrict__ g_out,
    int b, int n, int d, int diag, int row_base, int col_base, char* smem) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wrow = (wave >> 2) * 128;   // wave sub-tile origin: 2×4 grid
  const int wcol = (wave & 3) * 64;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ktiles = (d + BK - 1) / BK;
  const int fr = lane & 15;
  const int qbase = lane >> 4;          // K subgroup 0..3

  if (INTERIOR) {
    // ---- Per-lane precomputed addressing (the naive form recomputed every
    // glds source with a 64-bit multiply and every fragment address with
    // 2-3 VALU ops per read — measured VALU-bound, VALUBusy ≈ 5×MfmaUtil).
    const int w = threadIdx.x >> 6;
    const int rsub = lane >> 3;          // glds: row within an 8-row group
    const int cch = lane & 7;            // glds: 16-B chunk within the row
    // glds source voffsets (bytes) for the 4 DMA issues of this wave, per
    // operand; the K advance lives in the uniform base pointers below.
    int va[4], vb[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int rloc = (w * 4 + j) * 8 + rsub;
      const int sc = cch ^ kmask(rloc);
      va[j] = rloc * d * 2 + sc * 16;
      vb[j] = va[j];                     // same tile geometry for B
    }
    // Fragment ds_read addresses: chunk mask depends only on fr = r&15, so
    // each (operand, kk) needs one per-lane base; mi/ni go into the 16-bit
    // instruction offset (mi*2048 B).  Buffer double-toggle = one v_xor.
    const int mk = kmask(fr);
    int aAddr[2], bAddr[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int ch = (kk * 4 + qbase) ^ mk;
      aAddr[kk] = (wrow + fr) * ROW_BYTES + ch * 16;
      bAddr[kk] = TILE_BYTES + (wcol + fr) * ROW_BYTES + ch * 16;
    }
    const char* abase = reinterpret_cast<const char*>(zimg) +
        (size_t)row_base * d * 2;
    const char* bbase = reinterpret_cast<const char*>(ztxt) +
        (size_t)col_base * d * 2;

    auto stage = [&](int buf) {
      const int lb = __builtin_amdgcn_readfirstlane((w * 4) * 1024) +
          buf * (2 * TILE_BYTES);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        __builtin_amdgcn_global_load_lds(
            (gas_ptr)(abase + va[j]), (las_ptr)(smem + lb + j * 1024),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (gas_ptr)(bbase + vb[j]),
            (las_ptr)(smem + lb + TILE_BYTES + j * 1024), 16, 0, 0);
      }
      abase += BK * 2;                   // uniform K advance (SALU)
      bbase += BK * 2;
    };

    stage(0);
    for (int kt = 0; kt < ktiles; ++kt) {
      if (kt + 1 < ktiles) {
        // Safe to overwrite buf[(kt+1)&1]: the barrier ending iteration
        // kt-1 fenced every wave's reads of tile kt-1 from it.
        stage((kt + 1) & 1);
        // Wait + barrier in ONE asm statement with a "memory" clobber: the
        // plain s_barrier builtin is not a compiler memory fence, and the
        // scheduler was observed hoisting the fragment ds_reads above it —
        // reading rows another wave's DMA had not landed yet.  vmcnt(8):
        // this wave's 8 DMAs for tile kt landed, tile kt+1's still flying.
        asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
      } else {
        asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
      }
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 afrag[FM], bfrag[FN];
#pragma unroll
        for (int mi = 0; mi < FM; ++mi)
          afrag[mi] = *reinterpret_cast<const bf16x8*>(
              smem + aAddr[kk] + mi * (16 * ROW_BYTES));
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          bfrag[ni] = *reinterpret_cast<const bf16x8*>(
              smem + bAddr[kk] + ni * (16 * ROW_BYTES));
#pragma unroll
        for (int mi = 0; mi < FM; ++mi)
#pragma unroll
          for (int ni = 0; ni < FN; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
      }
      aAddr[0] ^= 2 * TILE_BYTES; aAddr[1] ^= 2 * TILE_BYTES;
      bAddr[0] ^= 2 * TILE_BYTES; bAddr[1] ^= 2 * TILE_BYTES;
      // Fence reads of buf[kt&1] before the next iteration's DMA overwrites.
      asm volatile("s_barrier" ::: "memory");
    }
  } else {
    auto compute_ktile = [&](char* As, char* Bs) {
#pragma unroll
      for (int kk = 0; kk < BK / 32; ++kk) {
        const int q = kk * 4 + qbase;   // 16-B chunk index within the row
        bf16x8 afrag[FM], bfrag[FN];
#pragma unroll
        for (int mi = 0; mi < FM; ++mi) {
          const int r = wrow + mi * 16 + fr;
          afrag[mi] = *reinterpret_cast<const bf16x8*>(
              As + r * ROW_BYTES + ((q ^ kmask(r)) * 16));
        }
#pragma unroll
        for (int ni = 0; ni < FN; ++ni) {
          const int r = wcol + ni * 16 + fr;
          bfrag[ni] = *reinterpret_cast<const bf16x8*>(
              Bs + r * ROW_BYTES + ((q ^ kmask(r)) * 16));
        }
#pragma unroll
        for (int mi = 0; mi < FM; ++mi)
#pragma unroll
          for (int ni = 0; ni < FN; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
      }
    };
    stage_guarded(zimg, smem, row_base, b, d, 0);
    stage_guarded(ztxt, smem + TILE_BYTES, col_base, n, d, 0);
    __syncthreads();
    for (int kt = 0; kt < ktiles; ++kt) {
      char* As = smem + (kt & 1) * (2 * TILE_BYTES);
      char* Bs = As + TILE_BYTES;
      if (kt + 1 < ktiles) {
        char* An = smem + ((kt + 1) & 1) * (2 * TILE_BYTES);
        stage_guarded(zimg, An, row_base, b, d, (kt + 1) * BK);
        stage_guarded(ztxt, An + TILE_BYTES, col_base, n, d, (kt + 1) * BK);
      }
      compute_ktile(As, Bs);
      __syncthreads();
    }
  }

  // Epilogue.  C/D layout of mfma_f32_16x16x32_bf16:
  //   col = lane&15, row = (lane>>4)*4 + reg.
  float s0 = 0.f, s1 = 0.f;
  if (MODE == 0) {
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int grow = row_base + wrow + mi * 16 + (lane >> 4) * 4 + reg;
          const int gcol = col_base + wcol + ni * 16 + (lane & 15);
          if (INTERIOR || (grow < b && gcol < n)) {
            const float z = acc[mi][ni][reg] * t + bias;
            const bool pos = (diag != DIAG_NONE) && (gcol == grow + diag);
            s0 += softplus_f(pos ? -z : z);
          }
        }
      }
    }
  } else {
    // One per-lane base offset + per-(mi,reg) scalar row offset keeps the
    // store addressing affine — per-element (size_t)grow*n math made the
    // allocator hoist 128 addresses and spill.  Caller guarantees
    // b*n*2 < 2^32 (ops/__init__.py column-chunks the slab).
    __bf16* gb = g_out + (size_t)row_base * n + col_base;
    const unsigned lane_off =
        (unsigned)(wrow + (lane >> 4) * 4) * (unsigned)n
        + (unsigned)(wcol + (lane & 15));
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const unsigned row_off = (unsigned)(mi * 16 + reg) * (unsigned)n;
        const int grow = row_base + wrow + mi * 16 + (lane >> 4) * 4 + reg;
#pragma unroll
        for (int ni = 0; ni < FN; ++ni) {
          const int gcol = col_base + wcol + ni * 16 + (lane & 15);
          if (INTERIOR || (grow < b && gcol < n)) {
            const float dot = acc[mi][ni][reg];
            const float z = dot * t + bias;
            const bool pos = (diag != DIAG_NONE) && (gcol == grow + diag);
            const float gv = sigmoid_fast(pos ? z : -z);
            const float g = pos ? -gv : gv;
            gb[lane_off + row_off + ni * 16] = (__bf16)g;
            s0 += g * dot;
            s1 += g;
          }
        }
      }
      __builtin_amdgcn_sched_barrier(0);  // cap epilogue register pressure
    }
  }

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

__device__ __forceinline__ void remap_block(int flags, int& bx, int& by) {
  const int gx = gridDim.x, gy = gridDim.y;
  int id = blockIdx.y * gx + blockIdx.x;
  if (flags & 1) {
    const int nwg = gx * gy;
    const int q = nwg / NXCD, r = nwg % NXCD;
    const int xcd = id % NXCD, idx = id / NXCD;
    id = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  if (flags & 2) {
    const int group = id / (GROUP_M * gy);
    const int within = id % (GROUP_M * gy);
    const int gm = min(GROUP_M, gx - group * GROUP_M);
    bx = group * GROUP_M + within % gm;
    by = within / gm;
  } else {
    bx = id % gx;
    by = id / gx;
  }
}

// Interior-only kernel: every tile full, d%64==0, n%8==0 — checked by the
// host launcher.  Separate from the general kernel so the hot path's
// register allocation is not inflated by the guarded path.
template <int MODE>
__launch_bounds__(THREADS) __global__ void siglip_tile_kernel_interior(
    const __bf16* __restrict__ zimg, const __bf16* __restrict__ ztxt,
    const float* __restrict__ t_prime, const float* __restrict__ bias_p,
    float* __restrict__ out, __bf16* __restrict__ g_out,
    int b, int n, int d, int diag, int flags) {
  __shared__ char smem[4 * TILE_BYTES];
  int bx, by;
  remap_block(flags, bx, by);
  const float t = __expf(*t_prime);
  const float bias = *bias_p;
  tile_body<MODE, true>(zimg, ztxt, t, bias, out, g_out, b, n, d, diag,
                        bx * BM, by * BN, smem);
}

// MODE 0: forward loss.  MODE 1: backward g-slab + scalar partials.
template <int MODE>
__launch_bounds__(THREADS) __global__ void siglip_tile_kernel(
    const __bf16* __restrict__ zimg, const __bf16* __restrict__ ztxt,
    const float* __restrict__ t_prime, const float* __restrict__ bias_p,
    float* __restrict__ out,          // MODE 0: loss[1].  MODE 1: scal[2].
    __bf16* __restrict__ g_out,       // MODE 1 only; leading dim = n
    int b, int n, int d, int diag, int flags) {
  __shared__ char smem[4 * TILE_BYTES];
  int bx, by;
  remap_block(flags, bx, by);
  const int row_base = bx * BM;
  const int col_base = by * BN;

  const float t = __expf(*t_prime);
  const float bias = *bias_p;

  const bool interior = (row_base + BM <= b) && (col_base + BN <= n) &&
      (d % BK == 0) && (n % 8 == 0);
  if (interior)
    tile_body<MODE, true>(zimg, ztxt, t, bias, out, g_out, b, n, d, diag,
                          row_base, col_base, smem);
  else
    tile_body<MODE, false>(zimg, ztxt, t, bias, out, g_out, b, n, d, diag,
                           row_base, col_base, smem);
}

inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

}  // namespace

extern "C" {

int siglip_ext_abi(void) { return 3; }

static inline bool all_interior(int b, int n, int d) {
  return (b % BM == 0) && (n % BN == 0) && (d % BK == 0);
}

int siglip_fwd_bf16(uintptr_t stream, const void* zimg, const void* ztxt,
                    const void* t_prime, const void* bias, void* loss_out,
                    int b, int n, int d, int diag, int flags) {
  if (b <= 0 || n <= 0) return (int)hipSuccess;
  if (d % 8 != 0) return (int)hipErrorInvalidValue;
  dim3 grid(ceil_div(b, BM), ceil_div(n, BN));
  if (all_interior(b, n, d))
    hipLaunchKernelGGL((siglip_tile_kernel_interior<0>), grid, dim3(THREADS),
                       0, (hipStream_t)stream,
                       (const __bf16*)zimg, (const __bf16*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)loss_out, (__bf16*)nullptr, b, n, d, diag,
                       flags);
  else
    hipLaunchKernelGGL((siglip_tile_kernel<0>), grid, dim3(THREADS), 0,
                       (hipStream_t)stream,
                       (const __bf16*)zimg, (const __bf16*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)loss_out, (__bf16*)nullptr, b, n, d, diag,
                       flags);
  return (int)hipGetLastError();
}

int siglip_bwd_g_bf16(uintptr_t stream, const void* zimg, const void* ztxt,
                      const void* t_prime, const void* bias, void* g_out,
                      void* scal, int b, int n, int d, int diag, int flags) {
  if (b <= 0 || n <= 0) return (int)hipSuccess;
  if (d % 8 != 0) return (int)hipErrorInvalidValue;
  dim3 grid(ceil_div(b, BM), ceil_div(n, BN));
  if (all_interior(b, n, d))
    hipLaunchKernelGGL((siglip_tile_kernel_interior<1>), grid, dim3(THREADS),
                       0, (hipStream_t)stream,
                       (const __bf16*)zimg, (const __bf16*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)scal, (__bf16*)g_out, b, n, d, diag, flags);
  else
    hipLaunchKernelGGL((siglip_tile_kernel<1>), grid, dim3(THREADS), 0,
                       (hipStream_t)stream,
                       (const __bf16*)zimg, (const __bf16*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)scal, (__bf16*)g_out, b, n, d, diag, flags);
  return (int)hipGetLastError();
}

}  // extern "C"
