// Fragment-major weight-gradient kernel for MI355X (gfx950).
//
//   dW[n][k] = sum_m dz[m][n] * src[m][k]
//
// Both inputs arrive TRANSPOSED in the fragment-major layout the MFMA
// wants (same block structure as the chain kernels' swizzled weights,
// csrc/fwd_chain.hip, but over the M contraction):
//
//   AT = dz^T  fragments: [N/32][Mp/16][2][32][8]   (Mp = M padded to 16)
//   BT = src^T fragments: [K/32][Mp/16][2][32][8]
//
// so every wave's fragment load is a contiguous 1 KB block — the round-1
// wgrad kernel (csrc/wgrad_kernel.hip) read m-strided fragments and was
// bound on L1/TCC line processing at ~1.3-1.6 TB/s; the library split-K
// bmm ran ~3x off the stream roofline. The producers emit these layouts
// for free: bwd_chain writes dz^T straight from its epilogue registers
// and fwd_chain/swizzle_x emit src^T (round-2 stage 2/3).
//
// Structure: NO LDS, no barriers — per m-chunk (16 rows) each wave loads
// its (NT_W + KT_W) fragments directly from global (coalesced; waves of
// one workgroup share B blocks -> L1 hits) and issues NT_W*KT_W MFMAs,
// 3-deep prefetched, consumed in reverse load order so the final wait
// leaves newer prefetches outstanding. The workgroup owns an output
// block of [128*NT_W n x 32*KT_W k] and one m-slab; fp32 partials land
// in per-slab buffers reduced by slab_reduce_kernel (global atomics at
// ~16M adds/launch measured 35-65 us of contention).
//
// XCD-aware decode: consecutive blockIdx round-robin across the 8 XCDs,
// so the grid is decoded as (xcd, seq) with all output-blocks of one
// m-slab given to ONE xcd — their shared input slab stays in that XCD's
// L2 instead of being refetched from HBM per block.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <cstdlib>

namespace rsdl {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short wf_bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float wf_f32x16;

template <int NT_W, int KT_W, int MIN_WAVES = 2, int SCHED = 0>
__global__ void __launch_bounds__(256, MIN_WAVES) wgrad_frag_kernel(
    const short* __restrict__ AT, const short* __restrict__ BT,
    float* __restrict__ dW,  // [N,K] fp32, pre-zeroed
    int32_t N, int32_t K, int64_t mchunks, int32_t nblk_n, int32_t nblk_k,
    int64_t nslabs, int64_t chunks_per_slab) {
  const int32_t NBLK = (nblk_n < 0 ? -nblk_n : nblk_n) * nblk_k;
  int64_t slab;
  int32_t blk;
  if (nblk_n < 0) {  // linear decode (A/B: RSDL_WGRAD_LINEAR)
    nblk_n = -nblk_n;
    blk = (int32_t)(blockIdx.x % (uint32_t)NBLK);
    slab = blockIdx.x / (uint32_t)NBLK;
  } else {
    const int32_t xcd = blockIdx.x & 7;
    const int32_t g = blockIdx.x >> 3;
    blk = g % NBLK;
    slab = (int64_t)(g / NBLK) * 8 + xcd;
  }
  if (slab >= nslabs) return;
  const int32_t bn = blk / nblk_k;
  const int32_t bk = blk % nblk_k;
  const int64_t c0 = slab * chunks_per_slab;
  const int64_t cend = min(c0 + chunks_per_slab, mchunks);
  if (c0 >= cend) return;
  const int64_t iters = cend - c0;

  const int32_t wave = threadIdx.x >> 6;
  const int32_t lane = threadIdx.x & 63;
  const int32_t nt0 = bn * (4 * NT_W) + wave * NT_W;
  const int32_t kt0 = bk * KT_W;

  const short* aptr[NT_W];
  const short* bptr[KT_W];
  #pragma unroll
  for (int nt = 0; nt < NT_W; nt++) {
    aptr[nt] = AT + ((int64_t)(nt0 + nt) * mchunks + c0) * 512 + lane * 8;
  }
  #pragma unroll
  for (int kt = 0; kt < KT_W; kt++) {
    bptr[kt] = BT + ((int64_t)(kt0 + kt) * mchunks + c0) * 512 + lane * 8;
  }

  wf_f32x16 acc[NT_W][KT_W] = {};
  wf_bf16x8 a0[NT_W], b0[KT_W], a1[NT_W], b1[KT_W];
  wf_bf16x8 a2[NT_W], b2[KT_W];

#define WF_LOAD(abuf, bbuf, i)                                             \
  {                                                                        \
    _Pragma("unroll") for (int nt = 0; nt < NT_W; nt++) {                  \
      *reinterpret_cast<uint4*>(&abuf[nt]) =                               \
          *reinterpret_cast<const uint4*>(&aptr[nt][(i)*512]);             \
    }                                                                      \
    _Pragma("unroll") for (int kt = 0; kt < KT_W; kt++) {                  \
      *reinterpret_cast<uint4*>(&bbuf[kt]) =                               \
          *reinterpret_cast<const uint4*>(&bptr[kt][(i)*512]);             \
    }                                                                      \
  }
// Consume fragments in REVERSE load order: the final MFMA of a batch
// then depends on the FIRST-issued load, so the backend's waitcnt for
// it leaves the newer prefetch loads outstanding (forward order ended
// each batch with s_waitcnt vmcnt(0), draining the pipeline).
#define WF_MFMA(abuf, bbuf)                                                \
  {                                                                        \
    _Pragma("unroll") for (int nt = NT_W - 1; nt >= 0; nt--) {             \
      _Pragma("unroll") for (int kt = KT_W - 1; kt >= 0; kt--) {           \
        acc[nt][kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(             \
            abuf[nt], bbuf[kt], acc[nt][kt], 0, 0, 0);                     \
      }                                                                    \
    }                                                                      \
  }

// SCHED=1: pin the load/MFMA group order with sched_barrier(0).
// Without it the backend SINKS one batch's loads to just before their
// consuming MFMAs (seen in the gfx950 asm: a global_load_dwordx4 group
// at offset:2048 immediately followed by s_waitcnt vmcnt(4)/(2)/(0)
// into MFMAs on those registers), collapsing the 3-deep pipeline once
// per 3 chunks and exposing a full HBM latency. With the fences every
// batch's consume sits >= 2 load-groups (12 loads) behind its issue.
// Static property verified by disassembly (profiles/r02/
// wgrad_sched_asm.md); default OFF until measured on hardware.
#define WF_FENCE()                                                         \
  if (SCHED) __builtin_amdgcn_sched_barrier(0)
  // 3-deep prefetch: at depth 1 only ~6 KB/wave was in flight and the
  // kernel ran latency-bound at ~half the stream roofline (depth 4
  // spills the <1,8> config's registers).
  int64_t i = 0;
  // Fenced variants: depth 3 fits <2,4>'s registers; <1,8> (9-quad load
  // groups) spills 36 VGPRs at depth 3 once the fences extend the
  // buffer live ranges, so it runs the 2-buffer depth-2 loop instead.
  // Both hoist an UNCONDITIONAL 2-chunk preheader under an iters>=2
  // guard: with the shared `if (1 < iters)` preheader below, the loop
  // header has a predecessor path with a different outstanding-load
  // count and the waitcnt pass merges to vmcnt(0).
  constexpr bool SCHED3 = SCHED && (NT_W + KT_W <= 6);
  constexpr bool SCHED2 = SCHED && (NT_W + KT_W > 6);
  if (!SCHED) {
    WF_LOAD(a0, b0, 0);
    if (1 < iters) WF_LOAD(a1, b1, 1);
  } else if (iters < 2) {
    WF_LOAD(a0, b0, 0);
    WF_MFMA(a0, b0);
  }
  if (SCHED && iters < 2) {
    // handled above
  } else if (SCHED3) {
    WF_LOAD(a0, b0, 0);
    WF_LOAD(a1, b1, 1);
    // Straight-line steady loop: every prefetch is UNCONDITIONAL (the
    // loop bound guarantees i+4 < iters), so the outstanding-load count
    // is path-independent and the backend can emit precise partial
    // vmcnt waits instead of the vmcnt(0) it is forced to at the
    // control-flow joins of the conditional-load loop below. Fences pin
    // the load/MFMA group order so no load sinks to its consumer.
    while (i + 5 <= iters) {
      WF_LOAD(a2, b2, i + 2);
      WF_FENCE();
      WF_MFMA(a0, b0);
      WF_FENCE();
      WF_LOAD(a0, b0, i + 3);
      WF_FENCE();
      WF_MFMA(a1, b1);
      WF_FENCE();
      WF_LOAD(a1, b1, i + 4);
      WF_FENCE();
      WF_MFMA(a2, b2);
      WF_FENCE();
      i += 3;
    }
    // Drain: rem = iters - i in [1,4]; a0/a1 hold chunks i, i+1.
    if (i + 2 < iters) WF_LOAD(a2, b2, i + 2);
    if (i < iters) WF_MFMA(a0, b0);
    if (i + 3 < iters) WF_LOAD(a0, b0, i + 3);
    if (i + 1 < iters) WF_MFMA(a1, b1);
    if (i + 2 < iters) WF_MFMA(a2, b2);
    if (i + 3 < iters) WF_MFMA(a0, b0);
  } else if (SCHED2) {
    // Same idea at prefetch depth 2 (two buffers): each load group sits
    // one consume-group (NT_W*KT_W MFMAs) ahead of its use, waits stay
    // at vmcnt(NT_W+KT_W) in a single-block loop.
    WF_LOAD(a0, b0, 0);
    WF_LOAD(a1, b1, 1);
    while (i + 4 <= iters) {
      WF_FENCE();
      WF_MFMA(a0, b0);
      WF_FENCE();
      WF_LOAD(a0, b0, i + 2);
      WF_FENCE();
      WF_MFMA(a1, b1);
      WF_FENCE();
      WF_LOAD(a1, b1, i + 3);
      WF_FENCE();
      i += 2;
    }
    // Drain: rem = iters - i in [1,3]; a0/a1 hold chunks i, i+1.
    if (i + 2 < iters) WF_LOAD(a2, b2, i + 2);
    if (i < iters) WF_MFMA(a0, b0);
    if (i + 1 < iters) WF_MFMA(a1, b1);
    if (i + 2 < iters) WF_MFMA(a2, b2);
  } else {
    while (i + 3 <= iters) {
      WF_LOAD(a2, b2, i + 2);
      WF_MFMA(a0, b0);
      if (i + 3 < iters) WF_LOAD(a0, b0, i + 3);
      WF_MFMA(a1, b1);
      if (i + 4 < iters) WF_LOAD(a1, b1, i + 4);
      WF_MFMA(a2, b2);
      i += 3;
    }
    if (i < iters) WF_MFMA(a0, b0);
    if (i + 1 < iters) WF_MFMA(a1, b1);
  }
#undef WF_LOAD
#undef WF_MFMA
#undef WF_FENCE

  // Epilogue: D[row = n-in-tile][col = k-in-tile]; 32 lanes write 32
  // consecutive k's. Plain per-slab partial stores (a second-stage
  // reduce sums the slabs) — global fp32 atomics at ~16M adds per
  // launch cost 35-65 us of contention.
  float* part = dW + slab * (int64_t)N * K;
  #pragma unroll
  for (int nt = 0; nt < NT_W; nt++) {
    #pragma unroll
    for (int kt = 0; kt < KT_W; kt++) {
      const int32_t k = (kt0 + kt) * 32 + (lane & 31);
      #pragma unroll
      for (int reg = 0; reg < 16; reg++) {
        const int32_t n = (nt0 + nt) * 32 + (reg & 3) + 8 * (reg >> 2) +
                          4 * (lane >> 5);
        part[(int64_t)n * K + k] = acc[nt][kt][reg];
      }
    }
  }
}

// out[i] = sum_s part[s][i] — the slab-partial reduction (fp32, float4).
// 2-D grid: y splits the slab range so small outputs still fill the
// chip; the few split partials combine with fp32 atomics (<= NSPLIT
// writers per line — negligible contention). out must be zeroed.
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float wf_f32x4;
template <int SCHED = 0>
__global__ void __launch_bounds__(256) slab_reduce_kernel(
    const float* __restrict__ part, float* __restrict__ out, int64_t nk4,
    int64_t nslabs, int64_t s1) {
  const int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (i >= nk4) return;
  const int64_t lo = (int64_t)blockIdx.y * s1;
  const int64_t hi = min(lo + s1, nslabs);
  // 4 independent accumulators with grouped loads: a single add-chain
  // loop serializes on one outstanding load per iteration (the backend
  // waits vmcnt(0) before each accumulate).
  // Plain (cacheable) loads: the partials were JUST written by the
  // producer kernel and are partially L2-resident; nontemporal loads
  // bypassed that and measured ~1.6-2.9 TB/s.
  wf_f32x4 s0 = {0.f, 0.f, 0.f, 0.f}, s1v = s0, s2 = s0, s3 = s0;
  int64_t sl = lo;
  if (SCHED) {
    // Double-buffered 4-load groups, 2x-unrolled so the buffer roles
    // swap instead of copying (a la[j]=lb[j] rotation materializes as
    // 12 v_movs + a vmcnt(0)): the next group is UNCONDITIONALLY in
    // flight while the current one accumulates, waits sit at vmcnt(4)
    // in a single-block loop.
    wf_f32x4 la[4], lb[4];
#define SR_LOAD(buf, base)                                                 \
  {                                                                        \
    _Pragma("unroll") for (int j = 0; j < 4; j++) {                        \
      buf[j] = *reinterpret_cast<const wf_f32x4*>(                         \
          part + (((base) + j) * nk4 + i) * 4);                            \
    }                                                                      \
  }
#define SR_ACC(buf)                                                        \
  {                                                                        \
    s0 += buf[0];                                                          \
    s1v += buf[1];                                                         \
    s2 += buf[2];                                                          \
    s3 += buf[3];                                                          \
  }
    if (sl + 4 <= hi) {
      SR_LOAD(la, sl);
      while (sl + 12 <= hi) {
        SR_LOAD(lb, sl + 4);
        __builtin_amdgcn_sched_barrier(0);
        SR_ACC(la);
        __builtin_amdgcn_sched_barrier(0);
        SR_LOAD(la, sl + 8);
        __builtin_amdgcn_sched_barrier(0);
        SR_ACC(lb);
        __builtin_amdgcn_sched_barrier(0);
        sl += 8;
      }
      // rem groups: la holds sl; possibly one more unloaded group.
      if (sl + 8 <= hi) {
        SR_LOAD(lb, sl + 4);
        SR_ACC(la);
        SR_ACC(lb);
        sl += 8;
      } else {
        SR_ACC(la);
        sl += 4;
      }
    }
#undef SR_LOAD
#undef SR_ACC
  } else {
    for (; sl + 4 <= hi; sl += 4) {
      const wf_f32x4 l0 = *reinterpret_cast<const wf_f32x4*>(
          part + ((sl + 0) * nk4 + i) * 4);
      const wf_f32x4 l1 = *reinterpret_cast<const wf_f32x4*>(
          part + ((sl + 1) * nk4 + i) * 4);
      const wf_f32x4 l2 = *reinterpret_cast<const wf_f32x4*>(
          part + ((sl + 2) * nk4 + i) * 4);
      const wf_f32x4 l3 = *reinterpret_cast<const wf_f32x4*>(
          part + ((sl + 3) * nk4 + i) * 4);
      s0 += l0;
      s1v += l1;
      s2 += l2;
      s3 += l3;
    }
  }
  for (; sl < hi; sl++) {
    s0 += *reinterpret_cast<const wf_f32x4*>(part + (sl * nk4 + i) * 4);
  }
  const wf_f32x4 s = (s0 + s1v) + (s2 + s3);
  float* o = out + i * 4;
  unsafeAtomicAdd(&o[0], s[0]);
  unsafeAtomicAdd(&o[1], s[1]);
  unsafeAtomicAdd(&o[2], s[2]);
  unsafeAtomicAdd(&o[3], s[3]);
}

// layer configs: 1 -> NT_W=2,KT_W=4 (dW1 [512,128pad]); 2 -> 1,8
// (dW2 [256,512]); 3 -> 1,8 (dW3 [128,256]).
int64_t wgrad_frag_nslabs(int64_t mchunks, int32_t N, int32_t K,
                          int32_t nt_w, int32_t kt_w);

void launch_slab_reduce(const float* part, float* out, int64_t nk,
                        int64_t nslabs, hipStream_t stream) {
  const int64_t nk4 = nk / 4;
  const int64_t gx = (nk4 + 255) / 256;
  // aim for >= 512 workgroups total
  int64_t nsplit = (512 + gx - 1) / gx;
  if (nsplit > nslabs) nsplit = nslabs;
  const int64_t s1 = (nslabs + nsplit - 1) / nsplit;
  nsplit = (nslabs + s1 - 1) / s1;
  static bool sched = [] {
    const char* e = std::getenv("RSDL_WGRAD_SCHED");
    return e && e[0] == '1';
  }();
  auto kern = sched ? slab_reduce_kernel<1> : slab_reduce_kernel<0>;
  hipLaunchKernelGGL(kern,
                     dim3((uint32_t)gx, (uint32_t)nsplit), dim3(256), 0,
                     stream, part, out, nk4, nslabs, s1);
}

static int64_t frag_target(int32_t K, int32_t nt_w) {
  static int64_t env_target = [] {
    const char* e = std::getenv("RSDL_WGRAD_WGS");
    return e ? atoll(e) : 0;
  }();
  if (env_target > 0) return env_target;
  return (nt_w == 2) ? 256 : (K >= 512 ? 512 : 256);
}

int64_t wgrad_frag_nslabs(int64_t mchunks, int32_t N, int32_t K,
                          int32_t nt_w, int32_t kt_w) {
  const int32_t nblk = (N / (nt_w * 128)) * (K / (kt_w * 32));
  int64_t nslabs = (frag_target(K, nt_w) + nblk - 1) / nblk;
  if (nslabs > mchunks) nslabs = mchunks;
  const int64_t cps = (mchunks + nslabs - 1) / nslabs;
  return (mchunks + cps - 1) / cps;
}

void launch_wgrad_frag(const void* AT, const void* BT, float* dW, int32_t N,
                       int32_t K, int64_t mchunks, int32_t nt_w,
                       int32_t kt_w, hipStream_t stream) {
  const int32_t nblk_n = N / (nt_w * 128);
  const int32_t nblk_k = K / (kt_w * 32);
  const int32_t nblk = nblk_n * nblk_k;
  // Workgroup-count target: more slabs = more stream parallelism but a
  // bigger partial buffer to reduce. Measured optima at the flagship
  // shapes (profiles/PERF.md). RSDL_WGRAD_WGS overrides for tuning.
  int64_t target = frag_target(K, nt_w);
  int64_t nslabs = (target + nblk - 1) / nblk;
  if (nslabs > mchunks) nslabs = mchunks;
  const int64_t chunks_per_slab = (mchunks + nslabs - 1) / nslabs;
  // re-derive so every slab is non-empty (partial rows must all be
  // written; there is no zero-init)
  nslabs = (mchunks + chunks_per_slab - 1) / chunks_per_slab;
  static bool linear = std::getenv("RSDL_WGRAD_LINEAR") != nullptr;
  // Pinned-schedule variant (see WF_FENCE above); flip on for A/B runs.
  static bool sched = [] {
    const char* e = std::getenv("RSDL_WGRAD_SCHED");
    return e && e[0] == '1';
  }();
  const int64_t grid =
      linear ? nslabs * nblk : ((nslabs + 7) / 8) * 8 * nblk;
  const int32_t nb_n = linear ? -nblk_n : nblk_n;
  if (nt_w == 2 && kt_w == 4) {
    auto kern = sched ? wgrad_frag_kernel<2, 4, 2, 1>
                      : wgrad_frag_kernel<2, 4>;
    hipLaunchKernelGGL(kern, dim3((uint32_t)grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const short*>(AT),
                       reinterpret_cast<const short*>(BT), dW, N, K, mchunks,
                       nb_n, nblk_k, nslabs, chunks_per_slab);
  } else if (nt_w == 1 && kt_w == 8) {
    auto kern = sched ? wgrad_frag_kernel<1, 8, 2, 1>
                      : wgrad_frag_kernel<1, 8>;
    hipLaunchKernelGGL(kern, dim3((uint32_t)grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const short*>(AT),
                       reinterpret_cast<const short*>(BT), dW, N, K, mchunks,
                       nb_n, nblk_k, nslabs, chunks_per_slab);
  } else if (nt_w == 1 && kt_w == 4) {
    // Small-tile config (RSDL_WGRAD_SMALL_TILES): 64 acc regs -> 3
    // waves/SIMD, and NT_W+KT_W<=6 takes the depth-3 fenced loop when
    // RSDL_WGRAD_SCHED is also set.
    auto kern = sched ? wgrad_frag_kernel<1, 4, 3, 1>
                      : wgrad_frag_kernel<1, 4, 3>;
    hipLaunchKernelGGL(kern, dim3((uint32_t)grid),
                       dim3(256), 0, stream,
                       reinterpret_cast<const short*>(AT),
                       reinterpret_cast<const short*>(BT), dW, N, K, mchunks,
                       nb_n, nblk_k, nslabs, chunks_per_slab);
  }
}

}  // namespace rsdl
