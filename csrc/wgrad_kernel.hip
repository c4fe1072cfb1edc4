// Hand-written MFMA wgrad kernel for MI355X (gfx950).
//
// dW[N,K] = dy[M,N]^T @ x[M,K] with M = batch (e.g. 250k) >> N,K <= 512 —
// the tall-K weight-gradient GEMM of the benchmark train step. hipBLASLt
// runs this shape ~10x off the memory roofline even tuned
// (profiles/PERF.md "GEMM layout findings"). This kernel does the split-M
// reduction natively:
//
//   grid = (N/64 x K/64 tiles) x SPLIT_M; each 256-thread WG owns a 64x64
//   dW tile and a contiguous m-chunk. Stages of 64 m-rows are staged
//   TRANSPOSED in double-buffered LDS ([col][m], +8-halfword row pad) so:
//     * global loads are coalesced (each lane: 4x 8-B loads covering 4
//       consecutive columns of a row pair, 128-B lines per 16 lanes),
//     * LDS stores are paired ds_write_b32 of (m, m+1) halfwords —
//       conflict-free (banks spread by col*20+m/2 over 16 of 32 banks),
//     * MFMA A/B fragments are contiguous 16-B ds_read_b128 within a
//       padded row (conflict-free, 16-B aligned).
//   v_mfma_f32_32x32x16_bf16 fragment maps (probe-verified,
//   tools/mfma_probe.hip):
//     A[i = lane&31][k_frag = (lane>>5)*8 + j]   (i -> n, k_frag -> m)
//     B[k_frag][n = lane&31]                      (n -> k)
//     D col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//   Four waves tile the 64x64 output 2x2; the stage loop double-buffers
//   (write stage s -> barrier -> issue stage s+1 global loads -> MFMA on
//   stage s), one barrier per stage. Partials accumulate into the fp32 dW
//   with global atomicAdd (dW <= 0.5 MB so SPLIT_M-way atomic traffic is a
//   few MB); db[n] = sum_m dy[m,n] is fused on k-block-0 WGs.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace rsdl {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16;

#define WG_TILE 64
#define MT 64                  // m-rows per stage
#define LDS_STRIDE (MT + 8)    // halfwords per [col] row (16-B aligned: 144B)

struct StageRegs {
  // 4 row-pairs x 4 cols per operand, as packed (lo,hi) halfword pairs.
  uint32_t dy_v[4][4];
  uint32_t x_v[4][4];
};

// Thread t covers 4 rows x 4 cols per operand: rows (t>>4)*2 + {0,1} +
// 32*p (p = 0,1), cols c0 = (t&15)*4 .. +3. The fast path loads each row's
// 4 consecutive bf16 as one 8-B vector load, UNGUARDED — per-element
// bounds ternaries make hipcc branch around each load and wait vmcnt(0)
// per element (serialized L2 round trips; cdna_hip_programming.md §5
// trap 4(c)), which cost this kernel ~5x.
typedef __attribute__((__vector_size__(4 * sizeof(short)))) short bf16x4;

__device__ __forceinline__ void load_stage_fast(
    const short* __restrict__ dy, const short* __restrict__ x, int32_t N,
    int32_t K, int32_t n0, int32_t k0, int64_t m0, int32_t tid,
    StageRegs& r) {
  const int32_t c0 = (tid & 15) * 4;
  const int32_t rp = (tid >> 4) * 2;
  // Issue all 8 independent 8-B loads up front (separate destination and
  // address registers), THEN pack: lets the compiler keep every load in
  // flight with one counted wait instead of serializing on reused regs.
  bf16x4 d[2][2], xv[2][2];
  #pragma unroll
  for (int p = 0; p < 2; p++) {
    #pragma unroll
    for (int q = 0; q < 2; q++) {
      const int64_t m = m0 + rp + 32 * p + q;
      d[p][q] = *reinterpret_cast<const bf16x4*>(&dy[m * N + n0 + c0]);
      xv[p][q] = *reinterpret_cast<const bf16x4*>(&x[m * K + k0 + c0]);
    }
  }
  #pragma unroll
  for (int p = 0; p < 2; p++) {
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      r.dy_v[p * 2][j] = (uint32_t)(uint16_t)d[p][0][j] |
                         (((uint32_t)(uint16_t)d[p][1][j]) << 16);
      r.x_v[p * 2][j] = (uint32_t)(uint16_t)xv[p][0][j] |
                        (((uint32_t)(uint16_t)xv[p][1][j]) << 16);
    }
  }
}

__device__ void load_stage_guarded(
    const short* __restrict__ dy, const short* __restrict__ x,
    int32_t N, int32_t K, int32_t n0, int32_t k0, int64_t m0, int64_t m_hi,
    int32_t tid, StageRegs& r) {
  const int32_t c0 = (tid & 15) * 4;
  const int32_t rp = (tid >> 4) * 2;
  #pragma unroll
  for (int p = 0; p < 2; p++) {
    #pragma unroll
    for (int q = 0; q < 2; q++) {
      const int64_t m = m0 + rp + 32 * p + q;
      const bool ok = m < m_hi;
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        const int32_t n = n0 + c0 + j;
        const int32_t k = k0 + c0 + j;
        const uint16_t dv =
            (ok && n < N) ? (uint16_t)dy[m * N + n] : (uint16_t)0;
        const uint16_t xv =
            (ok && k < K) ? (uint16_t)x[m * K + k] : (uint16_t)0;
        if (q == 0) {
          r.dy_v[p * 2][j] = dv;
          r.x_v[p * 2][j] = xv;
        } else {
          r.dy_v[p * 2][j] |= ((uint32_t)dv) << 16;
          r.x_v[p * 2][j] |= ((uint32_t)xv) << 16;
        }
      }
    }
  }
}

__device__ __forceinline__ void write_stage(short* dyT, short* xT,
                                            int32_t tid,
                                            const StageRegs& r) {
  const int32_t c0 = (tid & 15) * 4;
  const int32_t rp = (tid >> 4) * 2;
  #pragma unroll
  for (int p = 0; p < 2; p++) {
    const int32_t m = rp + 32 * p;
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      *reinterpret_cast<uint32_t*>(&dyT[(c0 + j) * LDS_STRIDE + m]) =
          r.dy_v[p * 2][j];
      *reinterpret_cast<uint32_t*>(&xT[(c0 + j) * LDS_STRIDE + m]) =
          r.x_v[p * 2][j];
    }
  }
}

__global__ void __launch_bounds__(256)
wgrad_bf16_kernel(const short* __restrict__ dy,  // [M, N] bf16 bits
                  const short* __restrict__ x,   // [M, K]
                  float* __restrict__ dW,        // [N, K] fp32 (zeroed)
                  float* __restrict__ db,        // [N] fp32 (zeroed) or null
                  int64_t M, int32_t N, int32_t K, int32_t split_m) {
  __shared__ short lds[2][2][WG_TILE * LDS_STRIDE];  // [buf][op][col*stride]

  const int32_t kblocks = (K + WG_TILE - 1) / WG_TILE;
  const int32_t n0 = (blockIdx.x / kblocks) * WG_TILE;
  const int32_t k0 = (blockIdx.x % kblocks) * WG_TILE;
  const int64_t chunk = (M + split_m - 1) / split_m;
  const int64_t m_lo = (int64_t)blockIdx.y * chunk;
  const int64_t m_hi = min(m_lo + chunk, M);

  const int32_t tid = threadIdx.x;
  const int32_t lane = tid & 63;
  const int32_t wave = tid >> 6;
  const int32_t wn = (wave & 1) * 32;
  const int32_t wk = (wave >> 1) * 32;

  const bool full_nk = (n0 + WG_TILE <= N) && (k0 + WG_TILE <= K);
  f32x16 acc = {};
  float bias_acc = 0.0f;
  // Bias grad comes straight from the A fragments already in registers
  // (each lane's a[] covers dy[mfrag..mfrag+7][n = wn + lane&31]; summed
  // over stages + the lane-half split this is exactly sum_m dy[m,n]) —
  // no LDS traffic, no bank conflicts.
  const bool do_bias = (db != nullptr) && (k0 == 0) && (wk == 0);

  // Two-deep software pipeline with STATIC register sets (a regs[parity]
  // runtime index would push the arrays to scratch): the loop body is
  // unrolled over two stages, r0/r1 alternating; loads for stage s issue
  // at stage s-2 (1-deep measured 73% SQ_WAIT_ANY).
  // Stage plan: n_full full stages run the UNGUARDED vector loads only
  // (per-element guard branches in the hot loop serialize every load with
  // a vmcnt(0) — seen in the disassembly); at most one guarded tail stage
  // is peeled to the end.
  const int64_t span = m_hi - m_lo;
  const int64_t n_full = full_nk ? span / MT : 0;
  const int64_t m_full_end = m_lo + n_full * MT;

  StageRegs r0, r1;
  int buf = 0;
  if (m_lo < m_full_end) {
    load_stage_fast(dy, x, N, K, n0, k0, m_lo, tid, r0);
  }
  if (m_lo + MT < m_full_end) {
    load_stage_fast(dy, x, N, K, n0, k0, m_lo + MT, tid, r1);
  }

  auto compute_stage = [&](int32_t b) {
    const short* dyT = lds[b][0];
    const short* xT = lds[b][1];
    #pragma unroll
    for (int32_t ms = 0; ms < MT; ms += 16) {
      const int32_t mfrag = ms + ((lane >> 5) * 8);
      bf16x8 a, bfr;
      *reinterpret_cast<uint4*>(&a) = *reinterpret_cast<const uint4*>(
          &dyT[(wn + (lane & 31)) * LDS_STRIDE + mfrag]);
      *reinterpret_cast<uint4*>(&bfr) = *reinterpret_cast<const uint4*>(
          &xT[(wk + (lane & 31)) * LDS_STRIDE + mfrag]);
      if (do_bias) {
        #pragma unroll
        for (int j = 0; j < 8; j++) {
          __hip_bfloat16 h;
          short sv = a[j];
          *reinterpret_cast<short*>(&h) = sv;
          bias_acc += __bfloat162float(h);
        }
      }
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, bfr, acc, 0, 0, 0);
    }
  };

  for (int64_t m0 = m_lo; m0 < m_full_end; m0 += 2 * MT) {
    // stage A (r0)
    write_stage(lds[buf][0], lds[buf][1], tid, r0);
    __syncthreads();
    if (m0 + 2 * MT < m_full_end) {
      load_stage_fast(dy, x, N, K, n0, k0, m0 + 2 * MT, tid, r0);
    }
    compute_stage(buf);
    buf ^= 1;
    __syncthreads();
    // stage B (r1)
    if (m0 + MT < m_full_end) {
      write_stage(lds[buf][0], lds[buf][1], tid, r1);
      __syncthreads();
      if (m0 + 3 * MT < m_full_end) {
        load_stage_fast(dy, x, N, K, n0, k0, m0 + 3 * MT, tid, r1);
      }
      compute_stage(buf);
      buf ^= 1;
      __syncthreads();
    }
  }
  // Peeled guarded tail (runs at most ceil((span - n_full*MT)/MT) stages;
  // only edge WGs or the last m-chunk reach it).
  for (int64_t m0 = m_full_end; m0 < m_hi; m0 += MT) {
    load_stage_guarded(dy, x, N, K, n0, k0, m0, m_hi, tid, r0);
    write_stage(lds[buf][0], lds[buf][1], tid, r0);
    __syncthreads();
    compute_stage(buf);
    buf ^= 1;
    __syncthreads();
  }

  #pragma unroll
  for (int reg = 0; reg < 16; reg++) {
    const int32_t col = lane & 31;
    const int32_t row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    const int32_t n = n0 + wn + row;
    const int32_t k = k0 + wk + col;
    if (n < N && k < K) atomicAdd(&dW[(int64_t)n * K + k], acc[reg]);
  }
  if (do_bias) {
    const int32_t n = n0 + wn + (lane & 31);
    if (n < N) atomicAdd(&db[n], bias_acc);
  }
}

void launch_wgrad_bf16(const void* dy, const void* x, float* dW, float* db,
                       int64_t M, int32_t N, int32_t K, int32_t split_m,
                       hipStream_t stream) {
  const int32_t nblocks = (N + WG_TILE - 1) / WG_TILE;
  const int32_t kblocks = (K + WG_TILE - 1) / WG_TILE;
  dim3 grid(nblocks * kblocks, split_m);
  hipLaunchKernelGGL(wgrad_bf16_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const short*>(dy),
                     reinterpret_cast<const short*>(x), dW, db, M, N, K,
                     split_m);
}

}  // namespace rsdl
