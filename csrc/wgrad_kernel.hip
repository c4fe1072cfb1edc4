// Hand-written MFMA wgrad kernel for MI355X (gfx950).
//
// dW[N,K] = dy[M,N]^T @ x[M,K] with M = batch (e.g. 250k) >> N,K <= 512 —
// the tall-K weight-gradient GEMM of the benchmark train step. hipBLASLt
// runs this shape ~10x off the memory roofline even tuned
// (profiles/PERF.md "GEMM layout findings"); a chunked-bmm workaround
// reaches 2.7 TB/s. This kernel does the split-M reduction natively:
//
//   grid = (N/64 x K/64 tiles) x SPLIT_M; each 256-thread WG owns a 64x64
//   dW tile and a contiguous m-chunk. Per 32-row stage both operands are
//   staged TRANSPOSED in LDS ([n][m] / [k][m], padded rows) so each wave's
//   MFMA A/B fragments are contiguous 16-byte ds_read_b128 reads:
//     v_mfma_f32_32x32x16_bf16 maps (probe-verified, tools/mfma_probe.hip)
//       A[i = lane&31][k_frag = (lane>>5)*8 + j]   (i -> n, k_frag -> m)
//       B[k_frag][n = lane&31]                      (n -> k)
//       D col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//   Four waves tile the 64x64 output 2x2. Partial tiles accumulate into
//   the fp32 dW with global atomicAdd (dW is tiny: N*K*4B <= 0.5 MB, so
//   SPLIT_M-way atomic traffic is a few MB). db[n] = sum_m dy[m,n] is fused
//   (k-block-0 WGs accumulate their dy tile's row sums).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace rsdl {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16;

// LDS tile geometry: [64 rows][MT columns(m)] of bf16, row padded to avoid
// bank conflicts on the transposed writes (stride 40 halfwords = 80 B;
// 80/4 = 20 dwords, gcd(20,32) = 4 -> <=4-way on 2-B scatter writes,
// conflict-free 16-B fragment reads from within one padded row).
#define WG_TILE 64
#define MT 32
#define LDS_STRIDE (MT + 8)  // halfwords

__global__ void __launch_bounds__(256)
wgrad_bf16_kernel(const short* __restrict__ dy,  // [M, N] bf16 bits
                  const short* __restrict__ x,   // [M, K]
                  float* __restrict__ dW,        // [N, K] fp32 (zeroed)
                  float* __restrict__ db,        // [N] fp32 (zeroed) or null
                  int64_t M, int32_t N, int32_t K, int32_t split_m) {
  __shared__ short dyT[WG_TILE * LDS_STRIDE];
  __shared__ short xT[WG_TILE * LDS_STRIDE];

  const int32_t kblocks = (K + WG_TILE - 1) / WG_TILE;
  const int32_t n0 = (blockIdx.x / kblocks) * WG_TILE;
  const int32_t k0 = (blockIdx.x % kblocks) * WG_TILE;
  // m-chunk for this split
  const int64_t chunk = (M + split_m - 1) / split_m;
  const int64_t m_lo = (int64_t)blockIdx.y * chunk;
  const int64_t m_hi = min(m_lo + chunk, M);

  const int32_t tid = threadIdx.x;
  const int32_t lane = tid & 63;
  const int32_t wave = tid >> 6;  // 0..3 -> 2x2 over the 64x64 tile
  const int32_t wn = (wave & 1) * 32;  // n offset within tile
  const int32_t wk = (wave >> 1) * 32; // k offset within tile

  f32x16 acc = {};
  float bias_acc = 0.0f;

  // Stage loop: MT(=32) m-rows at a time.
  for (int64_t m0 = m_lo; m0 < m_hi; m0 += MT) {
    const int32_t rows = (int32_t)min((int64_t)MT, m_hi - m0);
    // ---- load + transpose into LDS -----------------------------------
    // 256 threads cover 32(m) x 64(col) elements per operand: thread t
    // handles m = t >> 3, cols c0 = (t & 7) * 8 .. +7 (one 16-B global
    // load of 8 bf16), written as 8 transposed LDS halfword stores.
    {
      const int32_t mi = tid >> 3;
      const int32_t c0 = (tid & 7) * 8;
      // dy operand
      short vals[8];
      if (mi < rows) {
        const int64_t row = m0 + mi;
        #pragma unroll
        for (int j = 0; j < 8; j++) {
          const int32_t n = n0 + c0 + j;
          vals[j] = (n < N) ? dy[row * N + n] : (short)0;
        }
      } else {
        #pragma unroll
        for (int j = 0; j < 8; j++) vals[j] = 0;
      }
      #pragma unroll
      for (int j = 0; j < 8; j++)
        dyT[(c0 + j) * LDS_STRIDE + mi] = vals[j];
      // x operand
      if (mi < rows) {
        const int64_t row = m0 + mi;
        #pragma unroll
        for (int j = 0; j < 8; j++) {
          const int32_t k = k0 + c0 + j;
          vals[j] = (k < K) ? x[row * K + k] : (short)0;
        }
      } else {
        #pragma unroll
        for (int j = 0; j < 8; j++) vals[j] = 0;
      }
      #pragma unroll
      for (int j = 0; j < 8; j++)
        xT[(c0 + j) * LDS_STRIDE + mi] = vals[j];
    }
    __syncthreads();

    // ---- MFMA: two k_frag(m) steps of 16 ------------------------------
    #pragma unroll
    for (int32_t ms = 0; ms < MT; ms += 16) {
      const int32_t mfrag = ms + ((lane >> 5) * 8);
      bf16x8 a, b;
      // contiguous 16-B reads within a padded LDS row
      *reinterpret_cast<uint4*>(&a) = *reinterpret_cast<const uint4*>(
          &dyT[(wn + (lane & 31)) * LDS_STRIDE + mfrag]);
      *reinterpret_cast<uint4*>(&b) = *reinterpret_cast<const uint4*>(
          &xT[(wk + (lane & 31)) * LDS_STRIDE + mfrag]);
      acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
    }

    // ---- fused bias grad: waves with wk==0 sum dy rows ----------------
    if (db != nullptr && k0 == 0 && wk == 0) {
      // lane covers n = wn + (lane&31); halves (lane>=32) split the m range
      const int32_t n = wn + (lane & 31);
      const int32_t mb = (lane >> 5) * (MT / 2);
      float s = 0.0f;
      #pragma unroll
      for (int32_t m = 0; m < MT / 2; m++) {
        __hip_bfloat16 h;
        *reinterpret_cast<short*>(&h) = dyT[n * LDS_STRIDE + mb + m];
        s += __bfloat162float(h);
      }
      bias_acc += s;
    }
    __syncthreads();
  }

  // ---- epilogue: atomic accumulate into dW (and db) -------------------
  #pragma unroll
  for (int reg = 0; reg < 16; reg++) {
    const int32_t col = lane & 31;                       // k within wave tile
    const int32_t row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);  // n
    const int32_t n = n0 + wn + row;
    const int32_t k = k0 + wk + col;
    if (n < N && k < K) atomicAdd(&dW[(int64_t)n * K + k], acc[reg]);
  }
  if (db != nullptr && k0 == 0 && wk == 0) {
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
