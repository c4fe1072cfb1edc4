// Wide-tile MFMA wgrad v2 for MI355X (gfx950): cuts the tile-level operand
// re-reads that bounded csrc/wgrad_kernel.hip (dW tile 64x64 re-read dy
// once per k-block and x once per n-block). Here a 256-thread WG owns a
// (NW x KW) = (256x128) or (128x256) dW tile, so for the benchmark layer
// shapes one operand is read exactly once and the other at most 4x:
//
//   layer (N,K)        tile      dy reads  x reads   traffic (M=250k)
//   L1 (512,128p)      256x128   1x        2x        ~360 MB
//   L2 (256,512)       256x128   4x        1x        ~770 MB
//   L3 (128,256)       128x256   2x(->1)   1x        ~320 MB
//
// Same verified MFMA maps, transposed LDS staging and 2-deep static
// register pipeline as wgrad_kernel.hip; stages are MT=16 m-rows (LDS
// budget: (NW+KW)*(MT+8)*2B*2buf = 36.9 KB -> 4 WGs/CU). N and K must be
// multiples of 64 (the torch binding pads and narrows).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace rsdl {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short w_bf16x8;
typedef __attribute__((__vector_size__(4 * sizeof(short)))) short w_bf16x4;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float w_f32x16;

#define WMT 32
#define WLDS_STRIDE (WMT + 8)  // WMT+8 halfwords per [col] row (pad keeps 16-B reads conflict-free)

// Per-operand stage registers: operand with C columns contributes
// C*WMT/256 elements per thread = C/32 packed row-pair words per thread
// at 4 cols per group -> C/128 groups of 4 words.
#define WG_GROUPS(C) (((C / 4) * (WMT / 2) + 511) / 512)

template <int C>
struct WStage {
  uint32_t v[WG_GROUPS(C)][4];  // [group][col j] packed (m,m+1) halfwords
};

#define W_THREADS 512

template <int C>
__device__ __forceinline__ void w_load(const short* __restrict__ src,
                                       int64_t ld, int32_t c_base,
                                       int64_t m0, int32_t tid,
                                       WStage<C>& r) {
  // unit u = g*W_THREADS + tid covers row-pair pr = u / (C/4),
  // cols (u % (C/4))*4. For C*WMT/8 < W_THREADS the tail threads idle
  // (wave-uniform for the shapes used).
  w_bf16x4 lo[WG_GROUPS(C)], hi[WG_GROUPS(C)];
  #pragma unroll
  for (int g = 0; g < WG_GROUPS(C); g++) {
    const int32_t u = g * W_THREADS + tid;
    if (u >= (C / 4) * (WMT / 2)) break;
    const int32_t c4 = u % (C / 4);
    const int32_t pr = u / (C / 4);
    const int64_t m = m0 + pr * 2;
    lo[g] = *reinterpret_cast<const w_bf16x4*>(&src[m * ld + c_base + c4 * 4]);
    hi[g] = *reinterpret_cast<const w_bf16x4*>(
        &src[(m + 1) * ld + c_base + c4 * 4]);
  }
  #pragma unroll
  for (int g = 0; g < WG_GROUPS(C); g++) {
    const int32_t u = g * W_THREADS + tid;
    if (u >= (C / 4) * (WMT / 2)) break;
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      r.v[g][j] = (uint32_t)(uint16_t)lo[g][j] |
                  (((uint32_t)(uint16_t)hi[g][j]) << 16);
    }
  }
}

template <int C>
__device__ __forceinline__ void w_load_guarded(
    const short* __restrict__ src, int64_t ld, int32_t c_base, int64_t m0,
    int64_t m_hi, int32_t tid, WStage<C>& r) {
  #pragma unroll
  for (int g = 0; g < WG_GROUPS(C); g++) {
    const int32_t u = g * W_THREADS + tid;
    if (u >= (C / 4) * (WMT / 2)) break;
    const int32_t c4 = u % (C / 4);
    const int32_t pr = u / (C / 4);
    const int64_t m = m0 + pr * 2;
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      const uint16_t lo =
          (m < m_hi) ? (uint16_t)src[m * ld + c_base + c4 * 4 + j] : 0;
      const uint16_t hi =
          (m + 1 < m_hi)
              ? (uint16_t)src[(m + 1) * ld + c_base + c4 * 4 + j]
              : 0;
      r.v[g][j] = (uint32_t)lo | (((uint32_t)hi) << 16);
    }
  }
}

template <int C>
__device__ __forceinline__ void w_write(short* lds_t, int32_t tid,
                                        const WStage<C>& r) {
  #pragma unroll
  for (int g = 0; g < WG_GROUPS(C); g++) {
    const int32_t u = g * W_THREADS + tid;
    if (u >= (C / 4) * (WMT / 2)) break;
    const int32_t c4 = u % (C / 4);
    const int32_t pr = u / (C / 4);
    #pragma unroll
    for (int j = 0; j < 4; j++) {
      *reinterpret_cast<uint32_t*>(
          &lds_t[(c4 * 4 + j) * WLDS_STRIDE + pr * 2]) = r.v[g][j];
    }
  }
}

// NW x KW dW tile per WG; per wave: WN x WK subtile.
template <int NW, int KW, int WN, int WK>
__global__ void __launch_bounds__(W_THREADS)
wgrad_wide_kernel(const short* __restrict__ dy, const short* __restrict__ x,
                  float* __restrict__ dW, float* __restrict__ db, int64_t M,
                  int32_t N, int32_t K, int32_t split_m) {
  __shared__ short dyT[2][NW * WLDS_STRIDE];
  __shared__ short xT[2][KW * WLDS_STRIDE];

  const int32_t kblocks = K / KW;
  const int32_t n0 = (blockIdx.x / kblocks) * NW;
  const int32_t k0 = (blockIdx.x % kblocks) * KW;
  const int64_t chunk64 =
      ((M + split_m - 1) / split_m + WMT - 1) / WMT * WMT;
  const int64_t m_lo = (int64_t)blockIdx.y * chunk64;
  const int64_t m_hi = min(m_lo + chunk64, M);

  const int32_t tid = threadIdx.x;
  const int32_t lane = tid & 63;
  const int32_t wave = tid >> 6;
  // wave subtile offsets within the WG tile ((NW/WN) x (KW/WK) == 8 waves)
  const int32_t wn = (wave % (NW / WN)) * WN;
  const int32_t wk = (wave / (NW / WN)) * WK;

  w_f32x16 acc[(WN / 32) * (WK / 32)] = {};
  // Per-A-fragment bias partials: afr[ai] covers n = wn + ai*32 + (lane&31)
  // (both lane halves hit the same n with different m slices; the atomics
  // combine them).
  float bias_acc[WN / 32] = {};
  const bool do_bias = (db != nullptr) && (k0 == 0) && (wk == 0);

  const int64_t span = m_hi - m_lo;
  const int64_t n_full = span / WMT;
  const int64_t m_full_end = m_lo + n_full * WMT;

  WStage<NW> dy0, dy1;
  WStage<KW> x0, x1;
  int buf = 0;
  if (m_lo < m_full_end) {
    w_load<NW>(dy, N, n0, m_lo, tid, dy0);
    w_load<KW>(x, K, k0, m_lo, tid, x0);
  }
  if (m_lo + WMT < m_full_end) {
    w_load<NW>(dy, N, n0, m_lo + WMT, tid, dy1);
    w_load<KW>(x, K, k0, m_lo + WMT, tid, x1);
  }

  auto compute = [&](int32_t b) {
    const short* dT = dyT[b];
    const short* xTb = xT[b];
    #pragma unroll
    for (int32_t ms = 0; ms < WMT; ms += 16) {
      const int32_t mfrag = ms + (lane >> 5) * 8;
      w_bf16x8 afr[WN / 32];
      w_bf16x8 bfr[WK / 32];
      #pragma unroll
      for (int ai = 0; ai < WN / 32; ai++) {
        *reinterpret_cast<uint4*>(&afr[ai]) =
            *reinterpret_cast<const uint4*>(
                &dT[(wn + ai * 32 + (lane & 31)) * WLDS_STRIDE + mfrag]);
      }
      #pragma unroll
      for (int bi = 0; bi < WK / 32; bi++) {
        *reinterpret_cast<uint4*>(&bfr[bi]) =
            *reinterpret_cast<const uint4*>(
                &xTb[(wk + bi * 32 + (lane & 31)) * WLDS_STRIDE + mfrag]);
      }
      if (do_bias) {
        #pragma unroll
        for (int ai = 0; ai < WN / 32; ai++) {
          #pragma unroll
          for (int j = 0; j < 8; j++) {
            __hip_bfloat16 h;
            short sv = afr[ai][j];
            *reinterpret_cast<short*>(&h) = sv;
            bias_acc[ai] += __bfloat162float(h);
          }
        }
      }
      #pragma unroll
      for (int ai = 0; ai < WN / 32; ai++) {
        #pragma unroll
        for (int bi = 0; bi < WK / 32; bi++) {
          acc[ai * (WK / 32) + bi] =
              __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  afr[ai], bfr[bi], acc[ai * (WK / 32) + bi], 0, 0, 0);
        }
      }
    }
  };

  for (int64_t m0 = m_lo; m0 < m_full_end; m0 += 2 * WMT) {
    w_write<NW>(dyT[buf], tid, dy0);
    w_write<KW>(xT[buf], tid, x0);
    __syncthreads();
    if (m0 + 2 * WMT < m_full_end) {
      w_load<NW>(dy, N, n0, m0 + 2 * WMT, tid, dy0);
      w_load<KW>(x, K, k0, m0 + 2 * WMT, tid, x0);
    }
    compute(buf);
    buf ^= 1;
    __syncthreads();
    if (m0 + WMT < m_full_end) {
      w_write<NW>(dyT[buf], tid, dy1);
      w_write<KW>(xT[buf], tid, x1);
      __syncthreads();
      if (m0 + 3 * WMT < m_full_end) {
        w_load<NW>(dy, N, n0, m0 + 3 * WMT, tid, dy1);
        w_load<KW>(x, K, k0, m0 + 3 * WMT, tid, x1);
      }
      compute(buf);
      buf ^= 1;
      __syncthreads();
    }
  }
  // Peeled guarded tail (last partial stage of the last chunk only).
  for (int64_t m0 = m_full_end; m0 < m_hi; m0 += WMT) {
    w_load_guarded<NW>(dy, N, n0, m0, m_hi, tid, dy0);
    w_load_guarded<KW>(x, K, k0, m0, m_hi, tid, x0);
    w_write<NW>(dyT[buf], tid, dy0);
    w_write<KW>(xT[buf], tid, x0);
    __syncthreads();
    compute(buf);
    buf ^= 1;
    __syncthreads();
  }

  #pragma unroll
  for (int ai = 0; ai < WN / 32; ai++) {
    #pragma unroll
    for (int bi = 0; bi < WK / 32; bi++) {
      const w_f32x16 a = acc[ai * (WK / 32) + bi];
      #pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const int32_t col = lane & 31;
        const int32_t row =
            (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        const int32_t n = n0 + wn + ai * 32 + row;
        const int32_t k = k0 + wk + bi * 32 + col;
        atomicAdd(&dW[(int64_t)n * K + k], a[reg]);
      }
    }
  }
  if (do_bias) {
    #pragma unroll
    for (int ai = 0; ai < WN / 32; ai++) {
      const int32_t n = n0 + wn + ai * 32 + (lane & 31);
      atomicAdd(&db[n], bias_acc[ai]);
    }
  }
}

void launch_wgrad_wide(const void* dy, const void* x, float* dW, float* db,
                       int64_t M, int32_t N, int32_t K, int32_t split_m,
                       hipStream_t stream) {
  // Tile choice: cover N fully when N >= 256, else favor K coverage.
  if (N % 256 == 0 && K % 128 == 0) {
    dim3 grid((N / 256) * (K / 128), split_m);
    hipLaunchKernelGGL((wgrad_wide_kernel<256, 128, 64, 64>), grid,
                       dim3(W_THREADS), 0, stream,
                       reinterpret_cast<const short*>(dy),
                       reinterpret_cast<const short*>(x), dW, db, M, N, K,
                       split_m);
  } else if (N % 128 == 0 && K % 256 == 0) {
    dim3 grid((N / 128) * (K / 256), split_m);
    hipLaunchKernelGGL((wgrad_wide_kernel<128, 256, 64, 64>), grid,
                       dim3(W_THREADS), 0, stream,
                       reinterpret_cast<const short*>(dy),
                       reinterpret_cast<const short*>(x), dW, db, M, N, K,
                       split_m);
  } else {
    dim3 grid((N / 128) * (K / 128), split_m);
    hipLaunchKernelGGL((wgrad_wide_kernel<128, 128, 32, 64>), grid,
                       dim3(W_THREADS), 0, stream,
                       reinterpret_cast<const short*>(dy),
                       reinterpret_cast<const short*>(x), dW, db, M, N, K,
                       split_m);
  }
}

}  // namespace rsdl
