// Fused forward chain for the flagship TabularMLP on MI355X (gfx950) —
// the DEFAULT bench train-step forward (models/fused_step.py;
// RSDL_FUSED_STEP=0 reverts to eager).
//
//   a1 = relu(x0 @ W1^T + b1)   [M,100] -> [M,512]
//   a2 = relu(a1 @ W2^T + b2)   -> [M,256]
//   a3 = relu(a2 @ W3^T + b3)   -> [M,128]
//   out = a3 @ w4^T + b4        -> [M,1]
//
// One 256-thread workgroup carries an FC_MT-row slab through the whole
// chain: the slab's activations live in LDS between layers, weights are
// read from global (L2-resident, ~0.44 MB total) per MFMA step, and only
// x0 (read) and the a1/a2/a3/out tiles (written once for backward) touch
// HBM. Eliminates the inter-layer activation re-reads of the eager path
// (~448 MB/step at the bench shape; measured eager fwd 0.25 ms vs a
// ~0.1 ms fused roofline).
//
// FC_MT = 32 (round-2 tuning): the 64-row variant used ~133 KB LDS ->
// 1 workgroup/CU = 1 wave/SIMD, leaving the L2 weight-fragment latency
// of the k-loops unhidden even with software pipelining (measured
// 512 us vs the ~100 us roofline). 32-row tiles take ~67 KB -> 2
// workgroups/CU, so one workgroup's MFMAs cover the other's loads.
//
// MFMA orientation (v_mfma_f32_32x32x16_bf16, probe-verified maps in
// tools/mfma_probe.hip / csrc/wgrad_kernel.hip):
//   D[mrow][ncol] = A[mrow][k] x B[k][ncol], per lane:
//     A[m = lane&31][k = (lane>>5)*8 + j]   j = 0..7 of a bf16x8
//     B[k = (lane>>5)*8 + j][n = lane&31]
//     D col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// For y[m,n] = sum_k x[m,k] W[n,k] the A fragment is a contiguous row
// segment of the LDS tile. The B (weight) fragment is served from a
// FRAGMENT-MAJOR swizzled weight layout
//   SW[n_tile][k_chunk][h][ml][8]  (n_tile = n/32, k_chunk = k/16,
//   h = lane>>5, ml = lane&31)
// so one wave's 64 16-B fragment loads are a single contiguous 1 KB
// block. In the natural [N][K] layout each B load touched 32 cache
// lines 1 KB apart (rows), which bound the kernel on L1/TCP line
// processing (~16 GB of line traffic per launch) — the swizzle makes
// weight traffic coalesced and line-minimal. The host binding performs
// the swizzle per call (weights are 0.44 MB total; a few us).
//
// Validated by tests/test_gpu_kernels.py (chain numerics, layout
// oracles, whole-step parity) and tests/test_chain_sim.py (lane-level
// index simulation).

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


typedef __attribute__((__vector_size__(8 * sizeof(short)))) short fc_bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float fc_f32x16;
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int fc_u32x4;
typedef __attribute__((__vector_size__(2 * sizeof(unsigned int)))) unsigned int fc_u32x2;
typedef __attribute__((__vector_size__(2 * sizeof(float)))) float fc_f32x2;

// Packed pair helpers (one VALU op per TWO elements on gfx950).
__device__ __forceinline__ fc_f32x2 fc_pk_max0(fc_f32x2 v) {
  // no packed f32 max on gfx950; two v_max_f32
  fc_f32x2 r = {fmaxf(v[0], 0.f), fmaxf(v[1], 0.f)};
  return r;
}
__device__ __forceinline__ uint32_t fc_cvt_pk_bf16(fc_f32x2 v) {
  uint32_t p;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(p) : "v"(v[0]), "v"(v[1]));
  return p;
}

#define FC_MT 32            // rows per workgroup slab
#define FC_MTILES (FC_MT / 32)  // 32-row MFMA m-tiles per slab
#define FC_K0 100           // input feature count
#define FC_K0P 112          // padded to a 16-multiple for the k-loop
#define FC_N1 512
#define FC_N2 256
#define FC_N3 128
// LDS halfword strides (16-B aligned: multiples of 8; +8 pad de-banks the
// column writes of the D->tile stores).
#define FC_S0 (FC_K0P + 8)
#define FC_S1 (FC_N1 + 8)
#define FC_S2 (FC_N2 + 8)
#define FC_S3 (FC_N3 + 8)

__device__ __forceinline__ float fc_b2f(short s) {
  __hip_bfloat16 h;
  *reinterpret_cast<short*>(&h) = s;
  return __bfloat162float(h);
}

__device__ __forceinline__ short fc_f2b(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// One GEMM layer of the chain: src LDS tile [FC_MT][K] (stride SRC_S
// halfwords) x W[N][K] global -> relu(.+bias) -> dst LDS tile (stride
// DST_S). N_WAVE = output columns per wave (N/4). Each wave owns m-tiles
// {0,32} x its n-range; accumulators are static f32x16 arrays.
// A_FRAGMAJOR: src is a GLOBAL fragment-major m-tile block (layout as
// the swizzled weights: [kc][h][ml][8], 512 halfwords per k-chunk) —
// used for layer 1, whose input x is pre-swizzled host-side; SRC_S is
// ignored. Otherwise src is the LDS tile of the previous layer.
// EMIT_T: besides the LDS tile, the epilogue emits this m-tile's
// activations TRANSPOSED in wgrad fragment-major layout
// ([N/32][mchunks][2][32][8]; see csrc/wgrad_frag.hip) plus one 32-bit
// relu-mask word per column (maskT[m_tile][n], bit i = row i alive) —
// the backward chain then never reads the activations at all (masks
// only) and the wgrad kernel gets coalesced B fragments for free.
// Lanes l/l+32 hold complementary 4-row runs of a column; one shfl_xor
// per 4 values assembles the 8-row fragment runs in registers.
template <int K, int N, int SRC_S, int DST_S, bool RELU,
          bool A_FRAGMAJOR = false, bool EMIT_T = false, bool PI16 = false>
__device__ void fc_layer(const short* __restrict__ src_lds,
                         const short* __restrict__ W,
                         const float* __restrict__ bias,
                         short* __restrict__ dst_lds, int32_t wave,
                         int32_t lane, short* __restrict__ at_out = nullptr,
                         uint32_t* __restrict__ mask_row = nullptr,
                         int64_t mchunks = 0, int64_t mc0 = 0) {
  constexpr int NT = N / 128;  // n-tiles of 32 per wave (4 waves)
  constexpr int ITERS = K / 16;
  const int32_t n_base = wave * (N / 4);
  const int32_t frag_k0 = (lane >> 5) * 8;
  const int32_t ml = lane & 31;  // A row within m-tile / D col (n)

  // Software-pipelined k-loop: LDS occupancy pins this kernel at 1
  // wave/SIMD, so cross-wave latency hiding does not exist — the next
  // iteration's A (LDS) and B (global/L2) fragments must be IN FLIGHT
  // while the current MFMAs run. Fully unrolled (compile-time ITERS)
  // double-buffered prefetch: buf indices become constants after unroll,
  // so the fragment arrays stay in registers.
  const short* srcA[FC_MTILES];
  #pragma unroll
  for (int mt = 0; mt < FC_MTILES; mt++) {
    srcA[mt] = A_FRAGMAJOR
                   ? &src_lds[(int64_t)mt * ITERS * 512 + lane * 8]
                   : &src_lds[(mt * 32 + ml) * SRC_S + frag_k0];
  }
  // Swizzled weight base for this wave's n-tiles: block (ntile, kc) is
  // 512 contiguous halfwords; lane's 16-B slice at lane*8.
  const short* srcB[NT];
  #pragma unroll
  for (int nt = 0; nt < NT; nt++) {
    srcB[nt] = &W[((int64_t)(wave * NT + nt) * ITERS) * 512 + lane * 8];
  }

  fc_f32x16 acc[FC_MTILES][NT] = {};
  fc_bf16x8 a[2][FC_MTILES], b[2][NT];

  #pragma unroll
  for (int mt = 0; mt < FC_MTILES; mt++) {
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
      for (int mt = 0; mt < FC_MTILES; mt++) {
        *reinterpret_cast<uint4*>(&a[nxt][mt]) =
            *reinterpret_cast<const uint4*>(
                &srcA[mt][A_FRAGMAJOR ? (i + 1) * 512 : k]);
      }
      #pragma unroll
      for (int nt = 0; nt < NT; nt++) {
        *reinterpret_cast<uint4*>(&b[nxt][nt]) =
            *reinterpret_cast<const uint4*>(&srcB[nt][(i + 1) * 512]);
      }
    }
    #pragma unroll
    for (int mt = 0; mt < FC_MTILES; mt++) {
      #pragma unroll
      for (int nt = 0; nt < NT; nt++) {
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a[cur][mt], b[cur][nt], acc[mt][nt], 0, 0, 0);
      }
    }
  }
  // Epilogue: bias + (relu) + cast + store the D fragments into the dst
  // tile at [mrow][n] (b16 column writes; DST_S padding spreads banks);
  // with EMIT_T also emit the transposed fragment runs + mask word.
  // All arithmetic runs on packed row pairs (v_pk_add/max_f32 +
  // v_cvt_pk_bf16_f32): regs 2q/2q+1 are consecutive rows, and the
  // packed bf16 word IS the emission payload.
  static_assert(!EMIT_T || FC_MTILES == 1, "EMIT_T assumes 32-row slabs");
  const int32_t h = lane >> 5;
  #pragma unroll
  for (int mt = 0; mt < FC_MTILES; mt++) {
    #pragma unroll
    for (int nt = 0; nt < NT; nt++) {
      const int32_t n = n_base + nt * 32 + ml;
      const fc_f32x2 bv2 = {bias[n], bias[n]};
      // One base address per column; row offsets are compile-time, so
      // the b16 stores use ds_write immediate offsets instead of ~1.5
      // VALU address ops each.
      short* colbase = dst_lds + (mt * 32 + 4 * h) * DST_S + n;
      uint32_t p[8];
      #pragma unroll
      for (int q = 0; q < 8; q++) {
        const int32_t roff = (((2 * q) & 3) + 8 * (q >> 1)) * DST_S;
        fc_f32x2 v2 = {acc[mt][nt][2 * q], acc[mt][nt][2 * q + 1]};
        v2 += bv2;
        if (RELU) v2 = fc_pk_max0(v2);
        const uint32_t pk = fc_cvt_pk_bf16(v2);
        p[q] = pk;
        colbase[roff] = (short)(pk & 0xFFFFu);
        colbase[roff + DST_S] = (short)(pk >> 16);
      }
      if (EMIT_T) {
        // Mask word for column n (bit i = bf16 value of row i > 0): a
        // positive nonzero bf16 is 0 < bits < 0x8000.
        uint32_t w = 0;
        #pragma unroll
        for (int q = 0; q < 8; q++) {
          const int32_t mrow = ((2 * q) & 3) + 8 * (q >> 1) + 4 * h;
          const uint32_t lo = p[q] & 0xFFFFu;
          const uint32_t hi = p[q] >> 16;
          w |= (uint32_t)(lo - 1u < 0x7FFFu) << mrow;
          w |= (uint32_t)(hi - 1u < 0x7FFFu) << (mrow + 1);
        }
        // PI16: every producer/consumer of the transposed layout agrees
        // on the pi16 intra-chunk M-permutation (swap bits 2<->3 of the
        // 16-row position), under which each half-wave's own packed
        // pairs ARE the 8-element runs — the 4-shfl half-row exchange
        // and its slot selects disappear. M is the contraction dim of
        // every consumer (wgrad_frag), so dW is invariant.
        // Default (!PI16): exchange the half-rows as packed ints (4
        // shfls); lane h=0 assembles rows {0-7},{16-23}, h=1
        // {8-15},{24-31}.
        fc_u32x4 run0, run1;
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
        short* blk0 = at_out + ((nt_g * mchunks + mc0) * 512) + h * 256 +
                      ml * 8;
        short* blk1 = at_out + ((nt_g * mchunks + mc0 + 1) * 512) +
                      h * 256 + ml * 8;
        _rsdl_store(run0,
                                    reinterpret_cast<fc_u32x4*>(blk0));
        _rsdl_store(run1,
                                    reinterpret_cast<fc_u32x4*>(blk1));
        w |= __shfl_xor(w, 32);
        if (h == 0) mask_row[n] = w;
      }
    }
  }
}

// Copy an LDS activation tile [FC_MT][N] (stride S halfwords) to the
// global row-major [M, N] tensor, vectorized 16 B.
template <int N, int S>
__device__ void fc_store_tile(const short* __restrict__ lds, short* out,
                              int64_t m0, int64_t M, int32_t tid) {
  constexpr int VPR = N / 8;  // uint4 vectors per row
  for (int32_t u = tid; u < FC_MT * VPR; u += 256) {
    const int32_t m = u / VPR;
    const int32_t c = (u % VPR) * 8;
    if (m0 + m < M) {
      // Non-temporal: activations are written once and re-read only by the
      // backward kernel much later — keeping them OUT of L2 preserves the
      // weight working set (which every workgroup re-reads).
      _rsdl_store(
          *reinterpret_cast<const fc_u32x4*>(&lds[m * S + c]),
          reinterpret_cast<fc_u32x4*>(&out[(m0 + m) * N + c]));
    }
  }
}

template <bool PI16 = false>
__global__ void __launch_bounds__(256) fwd_chain_kernel(
    const short* __restrict__ x0s,  // fragment-major swizzled x (see note)
    const short* __restrict__ W1, const float* __restrict__ b1,
    const short* __restrict__ W2, const float* __restrict__ b2,
    const short* __restrict__ W3, const float* __restrict__ b3,
    const short* __restrict__ w4, const float* __restrict__ b4,
    short* __restrict__ a1t,        // wgrad fragment-major a1^T
    uint32_t* __restrict__ mask1,   // [m_tiles][512] relu-mask words
    short* __restrict__ a2t,        // wgrad fragment-major a2^T
    uint32_t* __restrict__ mask2,   // [m_tiles][256]
    short* __restrict__ a3, short* __restrict__ out,
    // Optional fused MSE epilogue (target != nullptr): dyb[m] =
    // (2/M)*(out[m]-target[m]) in bf16 and loss_part[blockIdx] =
    // sum_m (out[m]-target[m])^2 — removes the eager loss/grad kernel
    // chain from the fused train step.
    const float* __restrict__ target, short* __restrict__ dyb,
    float* __restrict__ loss_part, float inv_m, int64_t M,
    int64_t mchunks) {
  // LDS budget is the occupancy lever: t1 + t2 + lsum = ~50 KB -> 3
  // workgroups/CU. x needs no tile (pre-swizzled fragment-major in
  // global, coalesced A loads), and t3 ALIASES t1's storage — layer 3
  // runs after layer 2 consumed t1 (a1 itself leaves through the
  // EMIT_T epilogue, transposed, plus mask words — no row-major a1/a2
  // tensors exist at all).
  __shared__ __align__(16) char smem[FC_MT * FC_S1 * 2 + FC_MT * FC_S2 * 2 +
                                     FC_MT * 4];
  short* t1 = reinterpret_cast<short*>(smem);
  short* t2 = reinterpret_cast<short*>(smem + FC_MT * FC_S1 * 2);
  short* t3 = t1;  // aliased: live ranges are disjoint (barrier-ordered)
  float* lsum = reinterpret_cast<float*>(smem + FC_MT * FC_S1 * 2 +
                                         FC_MT * FC_S2 * 2);

  const int64_t m0 = (int64_t)blockIdx.x * FC_MT;
  const int32_t tid = threadIdx.x;
  const int32_t wave = tid >> 6;
  const int32_t lane = tid & 63;
  const int64_t mc0 = (int64_t)blockIdx.x * 2;

  // Layer 1: A fragments straight from the swizzled global x block.
  const short* xblk = &x0s[(int64_t)blockIdx.x * (FC_K0P / 16) * 512];
  fc_layer<FC_K0P, FC_N1, 0, FC_S1, true, true, true, PI16>(
      xblk, W1, b1, t1, wave, lane, a1t, &mask1[(int64_t)blockIdx.x * FC_N1],
      mchunks, mc0);
  __syncthreads();
  fc_layer<FC_N1, FC_N2, FC_S1, FC_S2, true, false, true, PI16>(
      t1, W2, b2, t2, wave, lane, a2t, &mask2[(int64_t)blockIdx.x * FC_N2],
      mchunks, mc0);
  __syncthreads();
  fc_layer<FC_N2, FC_N3, FC_S2, FC_S3, true>(t2, W3, b3, t3, wave, lane);
  __syncthreads();

  // Head: out[m] = sum_k a3[m,k] * w4[k] + b4. 256/FC_MT threads per row,
  // pair-wise LDS-free reduce via wave shuffles (partners are adjacent
  // lanes).
  {
    constexpr int TPR = 256 / FC_MT;       // threads per row
    constexpr int KPT = FC_N3 / TPR;       // k per thread
    const int32_t m = tid / TPR;
    const int32_t part = tid % TPR;
    float s = 0.f;
    #pragma unroll
    for (int32_t kk = 0; kk < KPT; kk++) {
      const int32_t k = part * KPT + kk;
      s += fc_b2f(t3[m * FC_S3 + k]) * fc_b2f(w4[k]);
    }
    #pragma unroll
    for (int32_t d = 1; d < TPR; d <<= 1) {
      s += __shfl_down(s, d);
    }
    if (part == 0) {
      const float o = s + b4[0];
      float d2 = 0.f;
      if (m0 + m < M) {
        out[m0 + m] = fc_f2b(o);
        if (target != nullptr) {
          const float diff = o - target[m0 + m];
          dyb[m0 + m] = fc_f2b(2.f * inv_m * diff);
          d2 = diff * diff;
        }
      }
      if (target != nullptr) lsum[m] = d2;
    }
  }
  if (target != nullptr) {
    __syncthreads();
    if (tid == 0) {
      float s = 0.f;
      #pragma unroll
      for (int32_t m = 0; m < FC_MT; m++) s += lsum[m];
      loss_part[blockIdx.x] = s;
    }
  }

  fc_store_tile<FC_N3, FC_S3>(t3, a3, m0, M, tid);
}

// x [M,100] bf16 -> fragment-major [ceil(M/32)][7][2][32][8] with zero
// padding (cols 100..111 and rows past M). One thread per 8-halfword
// fragment slice; replaces a 3-kernel pad+permute+contiguous chain.
__global__ void __launch_bounds__(256) swizzle_x_kernel(
    const short* __restrict__ x, short* __restrict__ out, int64_t M,
    int64_t total_blocks) {
  const int64_t b = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (b >= total_blocks) return;
  // b = ((mt*7 + kc)*2 + h)*32 + ml
  const int32_t ml = (int32_t)(b & 31);
  int64_t r = b >> 5;
  const int32_t h = (int32_t)(r & 1);
  r >>= 1;
  const int32_t kc = (int32_t)(r % 7);
  const int64_t mt = r / 7;
  const int64_t m = mt * 32 + ml;
  const int32_t k0 = kc * 16 + h * 8;
  short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (m < M) {
    const short* row = &x[m * FC_K0];
    if (k0 + 8 <= FC_K0) {
      // rows are 8-B aligned (100 * 2 B stride), not 16-B: two 8-B loads
      *reinterpret_cast<fc_u32x2*>(&v[0]) = _rsdl_load(
          reinterpret_cast<const fc_u32x2*>(&row[k0]));
      *reinterpret_cast<fc_u32x2*>(&v[4]) = _rsdl_load(
          reinterpret_cast<const fc_u32x2*>(&row[k0 + 4]));
    } else {
      #pragma unroll
      for (int j = 0; j < 8; j++) {
        if (k0 + j < FC_K0) v[j] = row[k0 + j];
      }
    }
  }
  _rsdl_store(*reinterpret_cast<fc_u32x4*>(v),
                              reinterpret_cast<fc_u32x4*>(&out[b * 8]));
}

void launch_swizzle_x(const void* x, void* out, int64_t M,
                      hipStream_t stream) {
  const int64_t mtiles = (M + FC_MT - 1) / FC_MT;
  // fragment blocks per 32-row m-tile: 7 kc * 2 h * 32 ml
  const int64_t total = mtiles * (FC_K0P / 16) * 2 * 32;
  const int64_t grid = (total + 255) / 256;
  hipLaunchKernelGGL(swizzle_x_kernel, dim3((uint32_t)grid), dim3(256), 0,
                     stream, reinterpret_cast<const short*>(x),
                     reinterpret_cast<short*>(out), M, total);
}

// Combined swizzle: one pass over x emits BOTH the forward fragment
// layout (xs: [mt][7][2][32][8] over k-contraction) and the wgrad
// fragment layout (xt: [4][mchunks][2][32][8] over m-contraction).
// One thread per 16-B fwd block (8 k for one row m); the wgrad-side
// elements it holds are scattered, so the wgrad half works via LDS: the
// workgroup stages a [32 m x 128 k] tile and re-emits it transposed.
template <bool PI16 = false>
__global__ void __launch_bounds__(256) swizzle_x_both_kernel(
    const short* __restrict__ x, short* __restrict__ xs,
    short* __restrict__ xt, int64_t M, int64_t mtiles) {
  __shared__ short tile[32 * 136];  // [m][k] stride 136 (16-B rows)
  const int64_t mt = blockIdx.x;
  if (mt >= mtiles) return;
  const int32_t tid = threadIdx.x;
  const int64_t m0 = mt * 32;
  // stage: 32 rows x 100 cols, 8-B vectors (25 uint2 slots per row)
  for (int32_t u = tid; u < 32 * 25; u += 256) {
    const int32_t m = u / 25;
    const int32_t c = (u % 25) * 4;
    fc_u32x2 v = {0, 0};
    if (m0 + m < M) {
      v = _rsdl_load(
          reinterpret_cast<const fc_u32x2*>(&x[(m0 + m) * FC_K0 + c]));
    }
    *reinterpret_cast<fc_u32x2*>(&tile[m * 136 + c]) = v;
  }
  // zero-pad cols 100..127 (fwd uses 100..111; wgrad k-tiles reach 127)
  for (int32_t u = tid; u < 32 * 7; u += 256) {
    const int32_t m = u / 7;
    const int32_t c = 100 + (u % 7) * 4;
    fc_u32x2 z = {0, 0};
    *reinterpret_cast<fc_u32x2*>(&tile[m * 136 + c]) = z;
  }
  __syncthreads();
  // fwd layout: block (kc, h, ml) -> 8 k of row ml
  for (int32_t u = tid; u < 7 * 2 * 32; u += 256) {
    const int32_t ml = u & 31;
    const int32_t h = (u >> 5) & 1;
    const int32_t kc = u >> 6;
    const int32_t k0 = kc * 16 + h * 8;
    fc_u32x4 v = *reinterpret_cast<const fc_u32x4*>(&tile[ml * 136 + k0]);
    _rsdl_store(
        v, reinterpret_cast<fc_u32x4*>(
               &xs[(mt * 14 + (int64_t)u / 32) * 256 + ml * 8]));
  }
  // wgrad layout: block (kt, mc_local, h, ml) -> 8 m of column kt*32+ml
  const int64_t mchunks = mtiles * 2;
  for (int32_t u = tid; u < 4 * 2 * 2 * 32; u += 256) {
    const int32_t ml = u & 31;
    const int32_t h = (u >> 5) & 1;
    const int32_t mcl = (u >> 6) & 1;
    const int32_t kt = u >> 7;
    const int32_t k = kt * 32 + ml;
    const int32_t mb = mcl * 16 + h * 8;
    fc_u32x4 pack;
    short* vp = reinterpret_cast<short*>(&pack);
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      // PI16 intra-chunk M-permutation (pos -> row = swap bits 2<->3):
      // slice h's positions hold rows {0-3,8-11} (h=0) / {4-7,12-15}.
      const int32_t mr = PI16
          ? (mcl * 16 + (j & 3) + ((j & 4) << 1) + 4 * h)
          : (mb + j);
      vp[j] = tile[mr * 136 + k];
    }
    _rsdl_store(
        pack, reinterpret_cast<fc_u32x4*>(
                  &xt[(((int64_t)kt * mchunks + mt * 2 + mcl) * 2 + h) *
                          256 +
                      ml * 8]));
  }
}

void launch_swizzle_x_both(const void* x, void* xs, void* xt, int64_t M,
                           int pi16, hipStream_t stream) {
  const int64_t mtiles = (M + FC_MT - 1) / FC_MT;
  auto kern =
      pi16 ? swizzle_x_both_kernel<true> : swizzle_x_both_kernel<false>;
  hipLaunchKernelGGL(kern, dim3((uint32_t)mtiles),
                     dim3(256), 0, stream,
                     reinterpret_cast<const short*>(x),
                     reinterpret_cast<short*>(xs),
                     reinterpret_cast<short*>(xt), M, mtiles);
}

// x [M,100] bf16 -> wgrad fragment-major x^T:
// [128/32][mchunks][2][32][8] with zero pads (cols 100..127, rows >= M).
// One thread per 16-B out block = 8 consecutive m's of one column k;
// writes fully coalesced, reads gather 8 row-strided elements (L1-local:
// neighboring threads read the same 8 rows).
template <bool PI16 = false>
__global__ void __launch_bounds__(256) swizzle_xt_kernel(
    const short* __restrict__ x, short* __restrict__ out, int64_t M,
    int64_t total_blocks) {
  const int64_t b = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (b >= total_blocks) return;
  // b = ((kt*mchunks + mc)*2 + h)*32 + ml
  const int32_t ml = (int32_t)(b & 31);
  int64_t r = b >> 5;
  const int32_t h = (int32_t)(r & 1);
  r >>= 1;
  // total = 4 (kt) * mchunks * 2 (h) * 32 (ml) = 256 * mchunks
  const int64_t mchunks = total_blocks >> 8;
  const int64_t mc = r % mchunks;
  const int32_t kt = (int32_t)(r / mchunks);
  const int32_t k = kt * 32 + ml;
  const int64_t m0 = mc * 16 + h * 8;
  short v[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  if (k < FC_K0) {
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      const int64_t m = PI16
          ? (mc * 16 + (j & 3) + ((j & 4) << 1) + 4 * h)
          : (m0 + j);
      if (m < M) v[j] = x[m * FC_K0 + k];
    }
  }
  _rsdl_store(*reinterpret_cast<fc_u32x4*>(v),
                              reinterpret_cast<fc_u32x4*>(&out[b * 8]));
}

void launch_swizzle_xt(const void* x, void* out, int64_t M, int pi16,
                       hipStream_t stream) {
  const int64_t mtiles = (M + FC_MT - 1) / FC_MT;
  const int64_t mchunks = mtiles * 2;
  const int64_t total = 4 * mchunks * 2 * 32;  // kt * mc * h * ml
  const int64_t grid = (total + 255) / 256;
  auto kern = pi16 ? swizzle_xt_kernel<true> : swizzle_xt_kernel<false>;
  hipLaunchKernelGGL(kern, dim3((uint32_t)grid), dim3(256), 0,
                     stream, reinterpret_cast<const short*>(x),
                     reinterpret_cast<short*>(out), M, total);
}

int64_t fwd_chain_grid(int64_t M) { return (M + FC_MT - 1) / FC_MT; }

void launch_fwd_chain(const void* x0s, const void* W1, const float* b1,
                      const void* W2, const float* b2, const void* W3,
                      const float* b3, const void* w4, const float* b4,
                      void* a1t, uint32_t* mask1, void* a2t,
                      uint32_t* mask2, void* a3, void* out,
                      const float* target, void* dyb, float* loss_part,
                      int64_t M, int pi16, hipStream_t stream) {
  const int32_t grid = (int32_t)((M + FC_MT - 1) / FC_MT);
  const int64_t mchunks = (int64_t)grid * 2;
  auto kern = pi16 ? fwd_chain_kernel<true> : fwd_chain_kernel<false>;
  hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, stream,
                     reinterpret_cast<const short*>(x0s),
                     reinterpret_cast<const short*>(W1), b1,
                     reinterpret_cast<const short*>(W2), b2,
                     reinterpret_cast<const short*>(W3), b3,
                     reinterpret_cast<const short*>(w4), b4,
                     reinterpret_cast<short*>(a1t), mask1,
                     reinterpret_cast<short*>(a2t), mask2,
                     reinterpret_cast<short*>(a3),
                     reinterpret_cast<short*>(out), target,
                     reinterpret_cast<short*>(dyb), loss_part,
                     target ? 1.f / (float)M : 0.f, M, mchunks);
}

}  // namespace rsdl
