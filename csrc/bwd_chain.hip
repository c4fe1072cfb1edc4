// EXPERIMENTAL (round-2 WIP, see docs/MEGAKERNEL_PLAN.md): fused backward
// chain for the flagship TabularMLP on MI355X (gfx950). Companion to
// csrc/fwd_chain.hip.
//
// Given dy = dLoss/dout [M,1] and the saved activations a1/a2/a3:
//   da3 = dy * w4          (outer product)      dz3 = da3 * (a3 > 0)
//   da2 = dz3 @ W3         ([M,128] -> [M,256]) dz2 = da2 * (a2 > 0)
//   da1 = dz2 @ W2         ([M,256] -> [M,512]) dz1 = da1 * (a1 > 0)
//   db_l = sum_m dz_l (l = 1..3), db4 = sum_m dy
// One 64-row slab per workgroup; the a_l tiles are loaded once and
// overwritten IN PLACE by their dz_l (the mask consumes the value it
// replaces), so LDS peaks at ~118 KB. dz1/dz2/dz3 are written to global
// for the (separate, reduction-shaped) wgrad kernels; bias partials go to
// per-workgroup slabs finalized by one at::sum.
//
// dgrad orientation: da[m,c] = sum_n dz[m,n] W[n,c] — the contraction is
// over the layer's OUTPUT index n, so the MFMA B fragment (contiguous
// over the contraction) needs memory laid out [c][n]: the host passes
// PRE-TRANSPOSED weights WT = W.t().contiguous(), shape
// [in_features, out_features] (cheap once per step; both fragments are
// then contiguous row segments — same A/B/D maps as fwd_chain,
// probe-verified in tools/mfma_probe.hip). In bc_layer terms WT is
// [N, K]: N = dgrad output width (in_features), K = contraction width.
//
// Exercised only by the RSDL_EXPERIMENTAL=1 GPU test; not on any default
// path.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace rsdl {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bc_bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float bc_f32x16;
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int bc_u32x4;

#define BC_MT 32
#define BC_MTILES (BC_MT / 32)
#define BC_N1 512
#define BC_N2 256
#define BC_N3 128
#define BC_S1 (BC_N1 + 8)
#define BC_S2 (BC_N2 + 8)
#define BC_S3 (BC_N3 + 8)

__device__ __forceinline__ float bc_b2f(short s) {
  __hip_bfloat16 h;
  *reinterpret_cast<short*>(&h) = s;
  return __bfloat162float(h);
}

__device__ __forceinline__ short bc_f2b(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// Load a [BC_MT][N] global activation tile into LDS (stride S), 16-B
// vectors, rows past M zero-filled (their dz contributions are masked by
// the guarded final stores and zero bias partials).
template <int N, int S>
__device__ void bc_load_tile(const short* __restrict__ g, short* lds,
                             int64_t m0, int64_t M, int32_t tid) {
  constexpr int VPR = N / 8;
  for (int32_t u = tid; u < BC_MT * VPR; u += 256) {
    const int32_t m = u / VPR;
    const int32_t c = (u % VPR) * 8;
    bc_u32x4 v = {0, 0, 0, 0};
    if (m0 + m < M) {
      // Non-temporal: the activation stream (449 MB/step) must not evict
      // the L2-resident transposed weights every workgroup re-reads.
      v = __builtin_nontemporal_load(
          reinterpret_cast<const bc_u32x4*>(&g[(m0 + m) * N + c]));
    }
    *reinterpret_cast<bc_u32x4*>(&lds[m * S + c]) = v;
  }
}

template <int N, int S>
__device__ void bc_store_tile(const short* __restrict__ lds, short* out,
                              int64_t m0, int64_t M, int32_t tid) {
  constexpr int VPR = N / 8;
  for (int32_t u = tid; u < BC_MT * VPR; u += 256) {
    const int32_t m = u / VPR;
    const int32_t c = (u % VPR) * 8;
    if (m0 + m < M) {
      __builtin_nontemporal_store(
          *reinterpret_cast<const bc_u32x4*>(&lds[m * S + c]),
          reinterpret_cast<bc_u32x4*>(&out[(m0 + m) * N + c]));
    }
  }
}

// One dgrad layer: da = dz_src @ W (via WT [N,K] contiguous), then
// dz_dst = da * (a_dst > 0) written IN PLACE over the a_dst tile.
// K = dz_src width (contraction), N = output width. The layer's bias
// partials db[n] = sum_m dz[m,n] are folded into the epilogue (the
// masked fp32 values are already in registers; lanes l/l+32 share a
// column, one shfl_xor combines them) — the old per-column scalar-LDS
// reduction pass (~112 LDS reads/thread) is gone.
template <int K, int N, int SRC_S, int DST_S>
__device__ void bc_layer(const short* __restrict__ dz_src,
                         const short* __restrict__ WT,
                         short* __restrict__ a_dst,
                         float* __restrict__ db_out, int32_t wave,
                         int32_t lane) {
  constexpr int NT = N / 128;
  constexpr int ITERS = K / 16;
  const int32_t n_base = wave * (N / 4);
  const int32_t frag_k0 = (lane >> 5) * 8;
  const int32_t ml = lane & 31;

  // Software-pipelined k-loop (same rationale as fc_layer: 1 wave/SIMD —
  // the next fragments must be in flight during the current MFMAs).
  const short* srcA[BC_MTILES];
  #pragma unroll
  for (int mt = 0; mt < BC_MTILES; mt++) {
    srcA[mt] = &dz_src[(mt * 32 + ml) * SRC_S + frag_k0];
  }
  // Fragment-major swizzled weights (see fwd_chain.hip): block
  // (ntile, kc) = 512 contiguous halfwords, lane slice at lane*8.
  const short* srcB[NT];
  #pragma unroll
  for (int nt = 0; nt < NT; nt++) {
    srcB[nt] = &WT[((int64_t)(wave * NT + nt) * ITERS) * 512 + lane * 8];
  }

  bc_f32x16 acc[BC_MTILES][NT] = {};
  bc_bf16x8 a[2][BC_MTILES], b[2][NT];

  #pragma unroll
  for (int mt = 0; mt < BC_MTILES; mt++) {
    *reinterpret_cast<uint4*>(&a[0][mt]) =
        *reinterpret_cast<const uint4*>(srcA[mt]);
  }
  #pragma unroll
  for (int nt = 0; nt < NT; nt++) {
    *reinterpret_cast<uint4*>(&b[0][nt]) =
        *reinterpret_cast<const uint4*>(srcB[nt]);
  }
  #pragma unroll
  for (int i = 0; i < ITERS; i++) {
    const int cur = i & 1;
    const int nxt = cur ^ 1;
    if (i + 1 < ITERS) {
      const int32_t k = (i + 1) * 16;
      #pragma unroll
      for (int mt = 0; mt < BC_MTILES; mt++) {
        *reinterpret_cast<uint4*>(&a[nxt][mt]) =
            *reinterpret_cast<const uint4*>(&srcA[mt][k]);
      }
      #pragma unroll
      for (int nt = 0; nt < NT; nt++) {
        *reinterpret_cast<uint4*>(&b[nxt][nt]) =
            *reinterpret_cast<const uint4*>(&srcB[nt][(i + 1) * 512]);
      }
    }
    #pragma unroll
    for (int mt = 0; mt < BC_MTILES; mt++) {
      #pragma unroll
      for (int nt = 0; nt < NT; nt++) {
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a[cur][mt], b[cur][nt], acc[mt][nt], 0, 0, 0);
      }
    }
  }
  // No barrier needed between the k-loop and the epilogue: each wave
  // reads only dz_src (a different buffer) and masks/writes only its own
  // n-range of a_dst; inter-layer ordering is handled by the
  // __syncthreads() between bc_layer calls in the kernel body.
  #pragma unroll
  for (int nt = 0; nt < NT; nt++) {
    const int32_t n = n_base + nt * 32 + ml;
    float colsum = 0.f;
    #pragma unroll
    for (int mt = 0; mt < BC_MTILES; mt++) {
      #pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const int32_t mrow =
            mt * 32 + (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
        short* cell = &a_dst[mrow * DST_S + n];
        const float live = bc_b2f(*cell) > 0.f ? 1.f : 0.f;
        const float v = acc[mt][nt][reg] * live;
        *cell = bc_f2b(v);
        colsum += v;
      }
    }
    colsum += __shfl_xor(colsum, 32);
    if (lane < 32) db_out[n] = colsum;
  }
}

__global__ void __launch_bounds__(256) bwd_chain_kernel(
    const short* __restrict__ dy,    // [M,1] bf16 (head grad)
    const short* __restrict__ a1,    // [M,512] saved activations
    const short* __restrict__ a2,    // [M,256]
    const short* __restrict__ a3,    // [M,128]
    const short* __restrict__ w4,    // [128]
    const short* __restrict__ W3T,   // swizzled W3^T fragments
    const short* __restrict__ W2T,   // swizzled W2^T fragments
    short* __restrict__ dz1, short* __restrict__ dz2,
    short* __restrict__ dz3,
    // [grid][512+256+128+1+256]: db1|db2|db3|db4|dW4 partials. dW4[k] =
    // sum_m dy[m]*a3[m,k] folded into the dz3 seed loop (a3 is already
    // in LDS there); the two per-k thread partials (even/odd m) are
    // summed host-side.
    float* __restrict__ db_part,
    int64_t M) {
  // LDS is the occupancy lever: t3 (dz3) ALIASES t1's storage — a1 is
  // loaded only after dz3 has been stored to global, so t1+t2+epsilon
  // = ~50 KB -> 3 workgroups/CU (the monolithic t1+t2+t3 layout was
  // 59 KB -> 2).
  __shared__ __align__(16) char smem[BC_MT * BC_S1 * 2 + BC_MT * BC_S2 * 2 +
                                     BC_MT * 4 + 256 * 4];
  short* t1 = reinterpret_cast<short*>(smem);
  short* t3 = t1;  // aliased: live ranges disjoint (barrier-ordered)
  short* t2 = reinterpret_cast<short*>(smem + BC_MT * BC_S1 * 2);
  float* dyf = reinterpret_cast<float*>(smem + BC_MT * BC_S1 * 2 +
                                        BC_MT * BC_S2 * 2);
  float* s3sh = dyf + BC_MT;

  const int64_t m0 = (int64_t)blockIdx.x * BC_MT;
  const int32_t tid = threadIdx.x;
  const int32_t wave = tid >> 6;
  const int32_t lane = tid & 63;
  float* part =
      &db_part[(int64_t)blockIdx.x * (BC_N1 + BC_N2 + BC_N3 + 1 + 256)];

  bc_load_tile<BC_N2, BC_S2>(a2, t2, m0, M, tid);
  bc_load_tile<BC_N3, BC_S3>(a3, t3, m0, M, tid);
  if (tid < BC_MT) {
    dyf[tid] = (m0 + tid < M) ? bc_b2f(dy[m0 + tid]) : 0.f;
  }
  __syncthreads();

  // dz3 = (dy * w4) * (a3 > 0), in place over t3; also db4 + dW4 + db3
  // partials (dy and a3 are at hand here).
  if (tid == 0) {
    float s = 0.f;
    for (int32_t m = 0; m < BC_MT; m++) s += dyf[m];
    part[BC_N1 + BC_N2 + BC_N3] = s;
  }
  {
    float s4 = 0.f;  // dW4 partial: this thread's k is constant (tid%128)
    float s3 = 0.f;  // db3 partial for the same k
    for (int32_t u = tid; u < BC_MT * BC_N3; u += 256) {
      const int32_t m = u / BC_N3;
      const int32_t k = u % BC_N3;
      short* cell = &t3[m * BC_S3 + k];
      const float a3v = bc_b2f(*cell);
      s4 += dyf[m] * a3v;
      const float live = a3v > 0.f ? 1.f : 0.f;
      const float dz3v = dyf[m] * bc_b2f(w4[k]) * live;
      *cell = bc_f2b(dz3v);
      s3 += dz3v;
    }
    part[BC_N1 + BC_N2 + BC_N3 + 1 + tid] = s4;
    s3sh[tid] = s3;
  }
  __syncthreads();
  if (tid < BC_N3) {
    part[BC_N1 + BC_N2 + tid] = s3sh[tid] + s3sh[tid + BC_N3];
  }

  // da2 = dz3 @ W3 (via W3T), mask by a2 -> dz2 in place over t2;
  // db2 partials written by the epilogue.
  bc_layer<BC_N3, BC_N2, BC_S3, BC_S2>(t3, W3T, t2, &part[BC_N1], wave,
                                       lane);
  __syncthreads();
  bc_store_tile<BC_N3, BC_S3>(t3, dz3, m0, M, tid);
  bc_store_tile<BC_N2, BC_S2>(t2, dz2, m0, M, tid);
  __syncthreads();  // t3 fully read before a1 overwrites its storage
  bc_load_tile<BC_N1, BC_S1>(a1, t1, m0, M, tid);
  __syncthreads();
  // da1 = dz2 @ W2 (via W2T), mask by a1 -> dz1 in place over t1; db1.
  bc_layer<BC_N2, BC_N1, BC_S2, BC_S1>(t2, W2T, t1, &part[0], wave,
                                       lane);
  __syncthreads();
  bc_store_tile<BC_N1, BC_S1>(t1, dz1, m0, M, tid);
}

int64_t bwd_chain_grid(int64_t M) { return (M + BC_MT - 1) / BC_MT; }

void launch_bwd_chain(const void* dy, const void* a1, const void* a2,
                      const void* a3, const void* w4, const void* W3T,
                      const void* W2T, void* dz1, void* dz2, void* dz3,
                      float* db_part, int64_t M, hipStream_t stream) {
  const int32_t grid = (int32_t)((M + BC_MT - 1) / BC_MT);
  hipLaunchKernelGGL(bwd_chain_kernel, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const short*>(dy),
                     reinterpret_cast<const short*>(a1),
                     reinterpret_cast<const short*>(a2),
                     reinterpret_cast<const short*>(a3),
                     reinterpret_cast<const short*>(w4),
                     reinterpret_cast<const short*>(W3T),
                     reinterpret_cast<const short*>(W2T),
                     reinterpret_cast<short*>(dz1),
                     reinterpret_cast<short*>(dz2),
                     reinterpret_cast<short*>(dz3), db_part, M);
}

}  // namespace rsdl
