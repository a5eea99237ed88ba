// siglip_kernels.hip — fused SigLIP sigmoid-loss kernels for MI355X (gfx950, CDNA4).
//
// Implements, per (b, n) block of image×text embeddings (row-major, K = emb
// dim contiguous; bf16 or OCP fp8-e4m3):
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
// Kernel structure (templates: MODE 0 = fwd, 1 = bwd-g; EB = element bytes,
// 2 = bf16 via v_mfma_f32_16x16x32_bf16, 1 = fp8-e4m3 via the MX-scaled
// v_mfma_scale_f32_16x16x128_f8f6f4 with unit block scales — 2× the bf16
// MFMA rate and half the staged bytes; per-tensor scales are folded into the
// temperature by the Python wrapper):
//   - 256×256 output tile per 512-thread (8-wave, 2×4) workgroup; each wave
//     owns a 128×64 sub-tile as 8×4 fragments, fp32 accumulate.
//   - K-loop in 128-byte K-steps (64 bf16 / 128 fp8 elements), double-
//     buffered LDS (4×32 KiB) staged by global_load_lds_dwordx4 (direct
//     HBM→LDS DMA); the DMA for tile k+1 is issued before the MFMA phase of
//     tile k and stays in flight across raw s_barriers under a counted
//     s_waitcnt vmcnt(8) (a __syncthreads would drain it).
//   - LDS image is lane-linear (glds requirement); the bank swizzle lives on
//     the per-lane *source* address and the ds_read chunk index (both-sides
//     rule): chunk' = chunk ^ kmask(row&15), conflict-free for the b128
//     fragment reads under both contiguous and interleaved lane grouping.
//   - All addressing is precomputed per lane before the loop (recomputing it
//     per tile measured VALU-bound: VALUBusy ≈ 5×MfmaUtil); the K advance
//     rides the uniform base pointers (SALU), the buffer toggle is one XOR.
//   - Edge blocks (ragged b/n, short d) take a register-staged path writing
//     the same swizzled LDS image with zero-fill guards; the interior kernel
//     is compiled separately so the hot path's register allocation is not
//     inflated by the guarded path (spill-free at 168-190 VGPRs).
//   - Block-id remap (flags): bit0 = XCD-contiguous spans, bit1 = grouped
//     column-major walk — temporally-close blocks share operand panels in
//     the per-XCD L2 (measured best together: +4%).
//
// Requirements: bf16 d%8==0, fp8 d%16==0; b, n arbitrary (guarded path).
// Compile: hipcc --offload-arch=gfx950 -O3 -shared -fPIC.

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>
#include <climits>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef const __attribute__((address_space(1))) unsigned int* gas_ptr;
typedef __attribute__((address_space(3))) unsigned int* las_ptr;

namespace {

constexpr int BM = 256;          // image rows per block
constexpr int BN = 256;          // text rows (logit cols) per block
constexpr int THREADS = 512;     // 8 waves as 2(M)×4(N)
constexpr int FM = 8;            // M fragments per wave (128 rows)
constexpr int FN = 4;            // N fragments per wave (64 cols)
constexpr int ROW_BYTES = 128;   // K-step bytes per row (one LDS row)
constexpr int TILE_BYTES = BM * ROW_BYTES;   // 32 KiB per operand tile
constexpr int DIAG_NONE = INT_MIN;
constexpr int NXCD = 8;
constexpr int GROUP_M = 8;       // block-rows per locality group (bit1)

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
// in mask bit 2, and a fixed-chunk column read over 16 rows sees all 8 masks.
__device__ __forceinline__ int kmask(int r) {
  return (((r >> 1) & 1) << 2) | ((r >> 2) & 3);
}

// Register-staged fallback for edge blocks: same swizzled LDS image,
// zero-filled outside [rows, d).  Byte-based; k0/d in elements.
template <int EB>
__device__ __forceinline__ void stage_guarded(const char* __restrict__ gsrc,
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
    const int k = k0 + c * (16 / EB);        // element index of this chunk
    uint4 v = (gr < rows && k < d)
        ? *reinterpret_cast<const uint4*>(gsrc + (size_t)gr * d * EB + k * EB)
        : zero;
    *reinterpret_cast<uint4*>(lds + row * ROW_BYTES +
                              ((c ^ kmask(row)) * 16)) = v;
  }
}

__device__ __forceinline__ i32x8 pack8(const uint4 lo, const uint4 hi) {
  return i32x8{(int)lo.x, (int)lo.y, (int)lo.z, (int)lo.w,
               (int)hi.x, (int)hi.y, (int)hi.z, (int)hi.w};
}

// One K-step of MFMAs from the (already swizzled) LDS images at byte
// addresses aAddr/bAddr (per-lane; fragment index rides the 16-bit
// instruction offset).
template <int EB>
__device__ __forceinline__ void mma_ktile(const char* smem, const int* aAddr,
                                          const int* bAddr,
                                          f32x4 (&acc)[FM][FN]) {
  if constexpr (EB == 2) {
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
  } else {
    // fp8: one MX-scaled MFMA covers the whole 128-element K-step.  Unit
    // block scales (e8m0 = 127 → 1.0); per-tensor scaling is folded into
    // the temperature upstream.
    uint4 blo[FN], bhi[FN];
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) {
      blo[ni] = *reinterpret_cast<const uint4*>(
          smem + bAddr[0] + ni * (16 * ROW_BYTES));
      bhi[ni] = *reinterpret_cast<const uint4*>(
          smem + bAddr[1] + ni * (16 * ROW_BYTES));
    }
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
      const uint4 alo = *reinterpret_cast<const uint4*>(
          smem + aAddr[0] + mi * (16 * ROW_BYTES));
      const uint4 ahi = *reinterpret_cast<const uint4*>(
          smem + aAddr[1] + mi * (16 * ROW_BYTES));
      const i32x8 af = pack8(alo, ahi);
#pragma unroll
      for (int ni = 0; ni < FN; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            af, pack8(blo[ni], bhi[ni]), acc[mi][ni], 0, 0,
            0, 0x7f7f7f7f, 0, 0x7f7f7f7f);
    }
  }
}

template <int MODE, bool INTERIOR, int EB>
__device__ __forceinline__ void tile_body(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
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

  constexpr int KTE = ROW_BYTES / EB;   // K elements per K-step
  const int ktiles = (d + KTE - 1) / KTE;
  const int fr = lane & 15;
  const int qbase = lane >> 4;          // K subgroup 0..3

  // Fragment ds_read chunk pair for this lane: bf16 reads 16 B per kk-half
  // of its 16-element K-slice; fp8 reads the two 16-B halves of its
  // 32-element K-slice.
  const int mk = kmask(fr);
  int aAddr[2], bAddr[2];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk) {
    const int ch = (EB == 2) ? ((kk * 4 + qbase) ^ mk)
                             : ((qbase * 2 + kk) ^ mk);
    aAddr[kk] = (wrow + fr) * ROW_BYTES + ch * 16;
    bAddr[kk] = TILE_BYTES + (wcol + fr) * ROW_BYTES + ch * 16;
  }

  if (INTERIOR) {
    // Per-lane precomputed DMA source offsets; K advance rides the uniform
    // base pointers (SALU), buffer toggle is one XOR per address.
    const int w = threadIdx.x >> 6;
    const int rsub = lane >> 3;          // glds: row within an 8-row group
    const int cch = lane & 7;            // glds: 16-B chunk within the row
    int va[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int rloc = (w * 4 + j) * 8 + rsub;
      const int sc = cch ^ kmask(rloc);
      va[j] = rloc * d * EB + sc * 16;
    }
    const char* abase = zimg + (size_t)row_base * d * EB;
    const char* bbase = ztxt + (size_t)col_base * d * EB;

    auto stage = [&](int buf) {
      const int lb = __builtin_amdgcn_readfirstlane((w * 4) * 1024) +
          buf * (2 * TILE_BYTES);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        __builtin_amdgcn_global_load_lds(
            (gas_ptr)(abase + va[j]), (las_ptr)(smem + lb + j * 1024),
            16, 0, 0);
        __builtin_amdgcn_global_load_lds(
            (gas_ptr)(bbase + va[j]),
            (las_ptr)(smem + lb + TILE_BYTES + j * 1024), 16, 0, 0);
      }
      abase += ROW_BYTES;                // uniform K advance (SALU)
      bbase += ROW_BYTES;
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
      mma_ktile<EB>(smem, aAddr, bAddr, acc);
      aAddr[0] ^= 2 * TILE_BYTES; aAddr[1] ^= 2 * TILE_BYTES;
      bAddr[0] ^= 2 * TILE_BYTES; bAddr[1] ^= 2 * TILE_BYTES;
      // Fence reads of buf[kt&1] before the next iteration's DMA overwrites.
      asm volatile("s_barrier" ::: "memory");
    }
  } else {
    stage_guarded<EB>(zimg, smem, row_base, b, d, 0);
    stage_guarded<EB>(ztxt, smem + TILE_BYTES, col_base, n, d, 0);
    __syncthreads();
    for (int kt = 0; kt < ktiles; ++kt) {
      if (kt + 1 < ktiles) {
        char* An = smem + ((kt + 1) & 1) * (2 * TILE_BYTES);
        stage_guarded<EB>(zimg, An, row_base, b, d, (kt + 1) * KTE);
        stage_guarded<EB>(ztxt, An + TILE_BYTES, col_base, n, d,
                          (kt + 1) * KTE);
      }
      mma_ktile<EB>(smem, aAddr, bAddr, acc);
      aAddr[0] ^= 2 * TILE_BYTES; aAddr[1] ^= 2 * TILE_BYTES;
      bAddr[0] ^= 2 * TILE_BYTES; bAddr[1] ^= 2 * TILE_BYTES;
      __syncthreads();
    }
  }

  // Epilogue.  C/D layout (shape-determined, dtype-independent on gfx950):
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
  } else if (INTERIOR) {
    // Stage this wave's 128×64 g sub-tile in its own LDS slice (the staging
    // buffers are free after the main loop; no cross-wave barrier needed),
    // then store it as 16 rounds of 8 fully-coalesced 128-B rows — 16
    // dwordx4 stores per lane instead of 128 scattered 2-B stores.
    char* wlds = smem + wave * (128 * 128);   // 16 KiB per wave
#pragma unroll
    for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
      for (int ni = 0; ni < FN; ++ni) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int lr = mi * 16 + (lane >> 4) * 4 + reg;   // 0..127
          const int ec = ni * 16 + (lane & 15);             // 0..63
          const int grow = row_base + wrow + lr;
          const int gcol = col_base + wcol + ec;
          const float dot = acc[mi][ni][reg];
          const float z = dot * t + bias;
          const bool pos = (diag != DIAG_NONE) && (gcol == grow + diag);
          const float gv = sigmoid_fast(pos ? z : -z);
          const float g = pos ? -gv : gv;
          // Swizzled 16-B chunk index so the readback is conflict-free.
          *reinterpret_cast<__bf16*>(
              wlds + lr * 128 + (((ec >> 3) ^ kmask(lr)) * 16)
              + (ec & 7) * 2) = (__bf16)g;
          s0 += g * dot;
          s1 += g;
        }
      }
    }
    // Own-wave LDS round trip: the compiler's lgkmcnt tracking orders the
    // reads after the writes; other waves never touch this slice.
    {
      const int rsub8 = lane >> 3;     // 8 rows per round
      const int c8 = lane & 7;         // 16-B chunk within the row
      __bf16* gb2 = g_out + (size_t)(row_base + wrow) * n + col_base + wcol;
#pragma unroll
      for (int r0 = 0; r0 < 128; r0 += 8) {
        const int lr = r0 + rsub8;
        const uint4 v = *reinterpret_cast<const uint4*>(
            wlds + lr * 128 + ((c8 ^ kmask(lr)) * 16));
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(gb2) + (size_t)lr * n * 2 + c8 * 16) = v;
      }
    }
  } else {
    // One per-lane base offset + per-(mi,reg) scalar row offset keeps the
    // store addressing affine.  (Edge blocks only.)
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
          if (grow < b && gcol < n) {
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

// ---------------- v10: 256² bf16 tile, 3-buffer depth-2 DMA ----------------
// Same 8-wave 256×256 tile, but the K-step drops to 64 B rows (32 bf16) so
// THREE K-step buffers fit LDS (3 × 32 KiB = 96 KiB): the DMA for tile k+2
// is issued while tile k computes, giving each transfer a two-compute-phase
// window instead of one — the measured per-iteration vmcnt wait disappears.
// bf16 interior only; fp8 and ragged shapes use the kernels above.
constexpr int VROW = 64;                     // bytes per LDS row (32 bf16)
constexpr int VTILE = BM * VROW;             // 16 KiB per operand tile
constexpr int VBUF = 2 * VTILE;              // 32 KiB per K-step buffer

// 2-bit chunk mask for 64-B rows: rows sharing a bank base (r ≡ r' mod 4)
// get distinct chunks; interleaved groups are conflict-free by row spacing.
__device__ __forceinline__ int kmask2(int r) { return (r >> 2) & 3; }

template <int MODE>
__device__ __forceinline__ void tile_body_v10(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    float t, float bias, float* __restrict__ out, __bf16* __restrict__ g_out,
    int b, int n, int d, int diag, int row_base, int col_base, char* smem) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wrow = (wave >> 2) * 128;
  const int wcol = (wave & 3) * 64;

  f32x4 acc[FM][FN];
#pragma unroll
  for (int mi = 0; mi < FM; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int ktiles = d / 32;                  // launcher enforces d%64==0
  const int fr = lane & 15;
  const int q = lane >> 4;                    // 16-B chunk 0..3

  int aAddr = (wrow + fr) * VROW + ((q ^ kmask2(wrow + fr)) * 16);
  int bAddr = VTILE + (wcol + fr) * VROW + ((q ^ kmask2(wcol + fr)) * 16);

  // DMA: 16 KiB per operand per K-step = 16 wave-instructions; wave w
  // issues 2 per operand (rows (w*2+j)*16 .. +16; 4 lanes per 64-B row).
  const int rsub = lane >> 2;
  const int cch = lane & 3;
  int va[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int rloc = (wave * 2 + j) * 16 + rsub;
    va[j] = rloc * d * 2 + ((cch ^ kmask2(rloc)) * 16);
  }
  const char* abase = zimg + (size_t)row_base * d * 2;
  const char* bbase = ztxt + (size_t)col_base * d * 2;

  auto stage = [&](int buf) {
    const int lb = __builtin_amdgcn_readfirstlane(wave * 2048) + buf * VBUF;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      __builtin_amdgcn_global_load_lds(
          (gas_ptr)(abase + va[j]), (las_ptr)(smem + lb + j * 1024),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (gas_ptr)(bbase + va[j]),
          (las_ptr)(smem + lb + VTILE + j * 1024), 16, 0, 0);
    }
    abase += VROW;
    bbase += VROW;
  };

  stage(0);
  if (ktiles > 1) stage(1);
  int bufoff = 0;
  for (int kt = 0; kt < ktiles; ++kt) {
    if (kt + 2 < ktiles) {
      // Buffer (kt+2)%3 held tile kt-1; its reads were fenced by the
      // barrier that ended iteration kt-1.
      stage((kt + 2) % 3);
      asm volatile("s_waitcnt vmcnt(8)\n\ts_barrier" ::: "memory");
    } else if (kt + 1 < ktiles) {
      asm volatile("s_waitcnt vmcnt(4)\n\ts_barrier" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)\n\ts_barrier" ::: "memory");
    }
    {
      const int aCur = aAddr + bufoff;
      const int bCur = bAddr + bufoff;
      bf16x8 afrag[FM], bfrag[FN];
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
        afrag[mi] = *reinterpret_cast<const bf16x8*>(
            smem + aCur + mi * (16 * VROW));
#pragma unroll
      for (int ni = 0; ni < FN; ++ni)
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(
            smem + bCur + ni * (16 * VROW));
#pragma unroll
      for (int mi = 0; mi < FM; ++mi)
#pragma unroll
        for (int ni = 0; ni < FN; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
    bufoff = (bufoff == 2 * VBUF) ? 0 : bufoff + VBUF;
    asm volatile("s_barrier" ::: "memory");
  }

  // Epilogue (identical math to the 256² kernels above).
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
          const float z = acc[mi][ni][reg] * t + bias;
          const bool pos = (diag != DIAG_NONE) && (gcol == grow + diag);
          s0 += softplus_f(pos ? -z : z);
        }
      }
    }
  } else {
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
      __builtin_amdgcn_sched_barrier(0);
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

template <int MODE>
__launch_bounds__(THREADS) __global__ void siglip_tile_kernel_v10(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    const float* __restrict__ t_prime, const float* __restrict__ bias_p,
    float* __restrict__ out, __bf16* __restrict__ g_out,
    int b, int n, int d, int diag, int flags) {
  __shared__ char smem[3 * VBUF];
  int bx, by;
  remap_block(flags, bx, by);
  const float t = __expf(*t_prime);
  const float bias = *bias_p;
  tile_body_v10<MODE>(zimg, ztxt, t, bias, out, g_out, b, n, d, diag,
                      bx * BM, by * BN, smem);
}

// Interior-only kernel: every tile full, d % K-step == 0, n%8==0 — checked
// by the host launcher.  Separate from the general kernel so the hot path's
// register allocation is not inflated by the guarded path.
template <int MODE, int EB>
__launch_bounds__(THREADS) __global__ void siglip_tile_kernel_interior(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    const float* __restrict__ t_prime, const float* __restrict__ bias_p,
    float* __restrict__ out, __bf16* __restrict__ g_out,
    int b, int n, int d, int diag, int flags) {
  __shared__ char smem[4 * TILE_BYTES];
  int bx, by;
  remap_block(flags, bx, by);
  const float t = __expf(*t_prime);
  const float bias = *bias_p;
  tile_body<MODE, true, EB>(zimg, ztxt, t, bias, out, g_out, b, n, d, diag,
                            bx * BM, by * BN, smem);
}

// General kernel: interior blocks take the DMA fast path, edge blocks the
// guarded register-staged path.  MODE 0: forward loss.  MODE 1: backward
// g-slab + scalar partials.
template <int MODE, int EB>
__launch_bounds__(THREADS) __global__ void siglip_tile_kernel(
    const char* __restrict__ zimg, const char* __restrict__ ztxt,
    const float* __restrict__ t_prime, const float* __restrict__ bias_p,
    float* __restrict__ out, __bf16* __restrict__ g_out,
    int b, int n, int d, int diag, int flags) {
  __shared__ char smem[4 * TILE_BYTES];
  int bx, by;
  remap_block(flags, bx, by);
  const int row_base = bx * BM;
  const int col_base = by * BN;

  const float t = __expf(*t_prime);
  const float bias = *bias_p;

  const bool interior = (row_base + BM <= b) && (col_base + BN <= n) &&
      (d % (ROW_BYTES / EB) == 0) && (n % 8 == 0);
  if (interior)
    tile_body<MODE, true, EB>(zimg, ztxt, t, bias, out, g_out, b, n, d, diag,
                              row_base, col_base, smem);
  else
    tile_body<MODE, false, EB>(zimg, ztxt, t, bias, out, g_out, b, n, d,
                               diag, row_base, col_base, smem);
}

inline int ceil_div(int a, int b) { return (a + b - 1) / b; }

inline bool siglip_use_v10() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("SIGLIP_V10");
    v = (e == nullptr || e[0] != '0') ? 1 : 0;   // default on; =0 for A/B
  }
  return v == 1;
}

template <int MODE, int EB>
int launch(uintptr_t stream, const void* zimg, const void* ztxt,
           const void* t_prime, const void* bias, void* out, void* g_out,
           int b, int n, int d, int diag, int flags) {
  if (b <= 0 || n <= 0) return (int)hipSuccess;
  if (d % (16 / EB) != 0) return (int)hipErrorInvalidValue;
  dim3 grid(ceil_div(b, BM), ceil_div(n, BN));
  if (EB == 2 && (b % BM == 0) && (n % BN == 0) && (d % 64 == 0) &&
      siglip_use_v10()) {
    hipLaunchKernelGGL((siglip_tile_kernel_v10<MODE>), grid, dim3(THREADS),
                       0, (hipStream_t)stream,
                       (const char*)zimg, (const char*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)out, (__bf16*)g_out, b, n, d, diag, flags);
    return (int)hipGetLastError();
  }
  const bool interior = (b % BM == 0) && (n % BN == 0) &&
      (d % (ROW_BYTES / EB) == 0);
  if (interior)
    hipLaunchKernelGGL((siglip_tile_kernel_interior<MODE, EB>), grid,
                       dim3(THREADS), 0, (hipStream_t)stream,
                       (const char*)zimg, (const char*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)out, (__bf16*)g_out, b, n, d, diag, flags);
  else
    hipLaunchKernelGGL((siglip_tile_kernel<MODE, EB>), grid, dim3(THREADS),
                       0, (hipStream_t)stream,
                       (const char*)zimg, (const char*)ztxt,
                       (const float*)t_prime, (const float*)bias,
                       (float*)out, (__bf16*)g_out, b, n, d, diag, flags);
  return (int)hipGetLastError();
}

}  // namespace

extern "C" {

int siglip_ext_abi(void) { return 4; }

int siglip_fwd_bf16(uintptr_t stream, const void* zimg, const void* ztxt,
                    const void* t_prime, const void* bias, void* loss_out,
                    int b, int n, int d, int diag, int flags) {
  return launch<0, 2>(stream, zimg, ztxt, t_prime, bias, loss_out, nullptr,
                      b, n, d, diag, flags);
}

int siglip_bwd_g_bf16(uintptr_t stream, const void* zimg, const void* ztxt,
                      const void* t_prime, const void* bias, void* g_out,
                      void* scal, int b, int n, int d, int diag, int flags) {
  return launch<1, 2>(stream, zimg, ztxt, t_prime, bias, scal, g_out,
                      b, n, d, diag, flags);
}

int siglip_fwd_fp8(uintptr_t stream, const void* zimg, const void* ztxt,
                   const void* t_prime, const void* bias, void* loss_out,
                   int b, int n, int d, int diag, int flags) {
  return launch<0, 1>(stream, zimg, ztxt, t_prime, bias, loss_out, nullptr,
                      b, n, d, diag, flags);
}

int siglip_bwd_g_fp8(uintptr_t stream, const void* zimg, const void* ztxt,
                     const void* t_prime, const void* bias, void* g_out,
                     void* scal, int b, int n, int d, int diag, int flags) {
  return launch<1, 1>(stream, zimg, ztxt, t_prime, bias, scal, g_out,
                      b, n, d, diag, flags);
}

}  // extern "C"
