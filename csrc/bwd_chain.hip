// Fused backward chain for the flagship TabularMLP on MI355X (gfx950) —
// the DEFAULT bench train-step backward. Companion to csrc/fwd_chain.hip.
//
// Given dy = dLoss/dout [M,1] and the saved activations a1/a2/a3:
//   da3 = dy * w4          (outer product)      dz3 = da3 * (a3 > 0)
//   da2 = dz3 @ W3         ([M,128] -> [M,256]) dz2 = da2 * (a2 > 0)
//   da1 = dz2 @ W2         ([M,256] -> [M,512]) dz1 = da1 * (a1 > 0)
//   db_l = sum_m dz_l (l = 1..3), db4 = sum_m dy
// One 32-row slab per workgroup. The relu masks arrive as 1-bit words
// from the forward (a1/a2 VALUES are never read); dz1/dz2/dz3 leave
// ONLY as wgrad fragment-major transposes, emitted straight from the
// epilogue registers; bias/dW4 partials go to per-workgroup slabs
// reduced host-side by the slab-reduce kernel.
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
// Validated by tests/test_gpu_kernels.py and tests/test_chain_sim.py.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace rsdl {

// A/B'd: nontemporal streaming accesses measure ~4% faster than
// cacheable (0.885 vs 0.919 ms fused step) — the activation streams
// evicting L2 hurts more than write-combining helps, even with the
// weight set fragment-swizzled. Keep nontemporal.
#ifndef RSDL_PLAIN_STREAMS
#define RSDL_PLAIN_STREAMS 0
#endif
template <typename T>
__device__ __forceinline__ void _rsdl_store(T v, T* p) {
#if RSDL_PLAIN_STREAMS
  *p = v;
#else
  __builtin_nontemporal_store(v, p);
#endif
}
template <typename T>
__device__ __forceinline__ T _rsdl_load(const T* p) {
#if RSDL_PLAIN_STREAMS
  return *p;
#else
  return __builtin_nontemporal_load(p);
#endif
}


typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bc_bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float bc_f32x16;
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int bc_u32x4;
typedef __attribute__((__vector_size__(2 * sizeof(float)))) float bc_f32x2;

__device__ __forceinline__ uint32_t bc_cvt_pk_bf16(bc_f32x2 v) {
  uint32_t p;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(p) : "v"(v[0]), "v"(v[1]));
  return p;
}

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
      v = _rsdl_load(
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
      _rsdl_store(
          *reinterpret_cast<const bc_u32x4*>(&lds[m * S + c]),
          reinterpret_cast<bc_u32x4*>(&out[(m0 + m) * N + c]));
    }
  }
}

// One dgrad layer: da = dz_src @ W (via swizzled WT fragments), then
// dz_dst = da * relu_mask. The mask arrives as one 32-bit word per
// column (maskT[m_tile][n], produced by the forward epilogue) — the
// backward never reads activation VALUES. dz leaves two ways:
//   * WRITE_LDS: into the dst LDS tile (stride DST_S) when the next
//     dgrad layer consumes it as its A source (dz2);
//   * always: TRANSPOSED wgrad-fragment-major runs straight from
//     registers to global (same shfl_xor half-row exchange as the
//     forward emitter) — what the wgrad kernel reads, coalesced.
// Bias partials db[n] fold into the epilogue as before.
template <int K, int N, int SRC_S, int DST_S, bool WRITE_LDS,
          bool PI16 = false>
__device__ void bc_layer(const short* __restrict__ dz_src,
                         const short* __restrict__ WT,
                         const uint32_t* __restrict__ mask_row,
                         short* __restrict__ dst_lds,
                         short* __restrict__ dzt_out,
                         float* __restrict__ db_out, int64_t mchunks,
                         int64_t mc0, int32_t wave, int32_t lane) {
  constexpr int NT = N / 128;
  constexpr int ITERS = K / 16;
  static_assert(BC_MTILES == 1, "transposed emission assumes 32-row slabs");
  const int32_t n_base = wave * (N / 4);
  const int32_t frag_k0 = (lane >> 5) * 8;
  const int32_t ml = lane & 31;
  const int32_t h = lane >> 5;

  const short* srcA = &dz_src[ml * SRC_S + frag_k0];
  const short* srcB[NT];
  #pragma unroll
  for (int nt = 0; nt < NT; nt++) {
    srcB[nt] = &WT[((int64_t)(wave * NT + nt) * ITERS) * 512 + lane * 8];
  }

  bc_f32x16 acc[NT] = {};
  bc_bf16x8 a[2], b[2][NT];

  *reinterpret_cast<uint4*>(&a[0]) =
      *reinterpret_cast<const uint4*>(srcA);
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
      *reinterpret_cast<uint4*>(&a[nxt]) =
          *reinterpret_cast<const uint4*>(&srcA[(i + 1) * 16]);
      #pragma unroll
      for (int nt = 0; nt < NT; nt++) {
        *reinterpret_cast<uint4*>(&b[nxt][nt]) =
            *reinterpret_cast<const uint4*>(&srcB[nt][(i + 1) * 512]);
      }
    }
    #pragma unroll
    for (int nt = 0; nt < NT; nt++) {
      acc[nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          a[cur], b[cur][nt], acc[nt], 0, 0, 0);
    }
  }

  #pragma unroll
  for (int nt = 0; nt < NT; nt++) {
    const int32_t n = n_base + nt * 32 + ml;
    const uint32_t mw = mask_row[n];
    uint32_t p[8];
    bc_f32x2 colsum2 = {0.f, 0.f};
    short* colbase =
        WRITE_LDS ? dst_lds + 4 * h * DST_S + n : nullptr;
    #pragma unroll
    for (int q = 0; q < 8; q++) {
      const int32_t mrow = ((2 * q) & 3) + 8 * (q >> 1) + 4 * h;
      const int32_t roff = (((2 * q) & 3) + 8 * (q >> 1)) * DST_S;
      const bc_f32x2 v2 = {
          ((mw >> mrow) & 1u) ? acc[nt][2 * q] : 0.f,
          ((mw >> (mrow + 1)) & 1u) ? acc[nt][2 * q + 1] : 0.f};
      colsum2 += v2;
      const uint32_t pk = bc_cvt_pk_bf16(v2);
      p[q] = pk;
      if (WRITE_LDS) {
        colbase[roff] = (short)(pk & 0xFFFFu);
        colbase[roff + DST_S] = (short)(pk >> 16);
      }
    }
    float colsum = colsum2[0] + colsum2[1];
    colsum += __shfl_xor(colsum, 32);
    if (lane < 32) db_out[n] = colsum;
    // Transposed fragment emission (see fwd_chain fc_layer EMIT_T):
    // PI16 writes each half-wave's own packed pairs as the runs (the
    // consumers agree on the pi16 intra-chunk M-permutation); default
    // is the packed-pair half-row exchange.
    bc_u32x4 run0, run1;
    if (PI16) {
      #pragma unroll
      for (int j = 0; j < 4; j++) {
        run0[j] = p[j];
        run1[j] = p[4 + j];
      }
    } else {
      uint32_t rx[4];
      #pragma unroll
      for (int j = 0; j < 2; j++) {
        rx[j] = __shfl_xor((int)(h == 0 ? p[2 + j] : p[j]), 32);
        rx[2 + j] = __shfl_xor((int)(h == 0 ? p[6 + j] : p[4 + j]), 32);
      }
      run0[0] = h == 0 ? p[0] : rx[0];
      run0[1] = h == 0 ? p[1] : rx[1];
      run0[2] = h == 0 ? rx[0] : p[2];
      run0[3] = h == 0 ? rx[1] : p[3];
      run1[0] = h == 0 ? p[4] : rx[2];
      run1[1] = h == 0 ? p[5] : rx[3];
      run1[2] = h == 0 ? rx[2] : p[6];
      run1[3] = h == 0 ? rx[3] : p[7];
    }
    const int64_t nt_g = (int64_t)(n_base + nt * 32) >> 5;
    short* blk0 = dzt_out + ((nt_g * mchunks + mc0) * 512) + h * 256 +
                  ml * 8;
    short* blk1 = dzt_out + ((nt_g * mchunks + mc0 + 1) * 512) + h * 256 +
                  ml * 8;
    _rsdl_store(run0, reinterpret_cast<bc_u32x4*>(blk0));
    _rsdl_store(run1, reinterpret_cast<bc_u32x4*>(blk1));
  }
}

template <bool PI16 = false>
__global__ void __launch_bounds__(256, 3) bwd_chain_kernel(
    const short* __restrict__ dy,     // [M,1] bf16 (head grad)
    const short* __restrict__ a3,     // [M,128] saved activations
    const uint32_t* __restrict__ mask1,  // [m_tiles][512] relu-mask words
    const uint32_t* __restrict__ mask2,  // [m_tiles][256]
    const short* __restrict__ w4,     // [128]
    const short* __restrict__ W3T,    // swizzled W3^T fragments
    const short* __restrict__ W2T,    // swizzled W2^T fragments
    short* __restrict__ dz1t,         // wgrad fragment-major dz^T outputs
    short* __restrict__ dz2t, short* __restrict__ dz3t,
    // [grid][512+256+128+1+256]: db1|db2|db3|db4|dW4 partials. dW4[k] =
    // sum_m dy[m]*a3[m,k] folded into the dz3 seed loop (a3 is already
    // in LDS there); the two per-k thread partials (even/odd m) are
    // summed host-side.
    float* __restrict__ db_part,
    int64_t M, int64_t mchunks) {
  // The backward never reads a1/a2 (relu masks arrive as bit-words from
  // the forward) and dz1 never touches LDS (emitted transposed straight
  // from registers), so LDS is just t2 + t3 + partial scratch ~27 KB.
  __shared__ __align__(16) short t2[BC_MT * BC_S2];
  __shared__ __align__(16) short t3[BC_MT * BC_S3];
  __shared__ float dyf[BC_MT];
  __shared__ float s3sh[256];

  const int64_t m0 = (int64_t)blockIdx.x * BC_MT;
  const int32_t tid = threadIdx.x;
  const int32_t wave = tid >> 6;
  const int32_t lane = tid & 63;
  const int64_t mc0 = (int64_t)blockIdx.x * 2;
  // stride padded to a 4-multiple for the host-side float4 slab reduce
  float* part =
      &db_part[(int64_t)blockIdx.x *
               ((BC_N1 + BC_N2 + BC_N3 + 1 + 256 + 3) / 4 * 4)];

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
  // dz3^T emission: one 8-row run per work item read out of t3 (the
  // wgrad kernel's A layout; 512 runs over 256 threads).
  for (int32_t r = tid; r < (BC_N3 / 32) * 2 * 2 * 32; r += 256) {
    const int32_t mlr = r & 31;
    const int32_t hr = (r >> 5) & 1;
    const int32_t mcl = (r >> 6) & 1;
    const int32_t ntr = r >> 7;
    const int32_t k = ntr * 32 + mlr;
    const int32_t mbase = mcl * 16 + hr * 8;
    short run[8];
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      const int32_t mr = PI16
          ? (mcl * 16 + (j & 3) + ((j & 4) << 1) + 4 * hr)
          : (mbase + j);
      run[j] = t3[mr * BC_S3 + k];
    }
    short* blk = dz3t + (((int64_t)ntr * mchunks + mc0 + mcl) * 512) +
                 hr * 256 + mlr * 8;
    _rsdl_store(*reinterpret_cast<bc_u32x4*>(run),
                                reinterpret_cast<bc_u32x4*>(blk));
  }
  __syncthreads();

  // da2 = dz3 @ W3 (via W3T), masked -> dz2: LDS tile (next layer's A
  // source) + transposed emission + db2 partials.
  bc_layer<BC_N3, BC_N2, BC_S3, BC_S2, true, PI16>(
      t3, W3T, &mask2[(int64_t)blockIdx.x * BC_N2], t2, dz2t,
      &part[BC_N1], mchunks, mc0, wave, lane);
  __syncthreads();
  // da1 = dz2 @ W2 (via W2T), masked -> dz1: transposed emission only.
  bc_layer<BC_N2, BC_N1, BC_S2, BC_S1, false, PI16>(
      t2, W2T, &mask1[(int64_t)blockIdx.x * BC_N1], nullptr, dz1t,
      &part[0], mchunks, mc0, wave, lane);
}

int64_t bwd_chain_grid(int64_t M) { return (M + BC_MT - 1) / BC_MT; }

void launch_bwd_chain(const void* dy, const void* a3, const void* mask1,
                      const void* mask2, const void* w4, const void* W3T,
                      const void* W2T, void* dz1t, void* dz2t, void* dz3t,
                      float* db_part, int64_t M, int pi16,
                      hipStream_t stream) {
  const int32_t grid = (int32_t)((M + BC_MT - 1) / BC_MT);
  const int64_t mchunks = (int64_t)grid * 2;
  auto kern = pi16 ? bwd_chain_kernel<true> : bwd_chain_kernel<false>;
  hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const short*>(dy),
                     reinterpret_cast<const short*>(a3),
                     reinterpret_cast<const uint32_t*>(mask1),
                     reinterpret_cast<const uint32_t*>(mask2),
                     reinterpret_cast<const short*>(w4),
                     reinterpret_cast<const short*>(W3T),
                     reinterpret_cast<const short*>(W2T),
                     reinterpret_cast<short*>(dz1t),
                     reinterpret_cast<short*>(dz2t),
                     reinterpret_cast<short*>(dz3t), db_part, M, mchunks);
}

}  // namespace rsdl
