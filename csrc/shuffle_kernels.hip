// MI355X (gfx950, CDNA4) shuffle kernels.
//
// These are the native replacements for the reference's pandas/numpy hot ops
// (reference: ray_shuffling_data_loader/shuffle.py:156-194 random assignment +
// mask-partition + concat + sample(frac=1); torch_dataset.py:204-236 column
// cast/pack):
//
//   * gather_rows_u4     — full row permutation of a packed row-major byte
//                          matrix (the reducer-side `sample(frac=1)` +
//                          implicit `pd.concat`, fused into ONE gather pass).
//   * unpack_permute     — fused gather-permute + per-column cast + pack of
//                          packed rows into contiguous per-column torch
//                          tensors (the `convert_to_tensor` replacement).
//   * pack_columns       — inverse: per-column cast + interleave into packed
//                          rows (map-side, feeds the RCCL all-to-all).
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//   - All three are memory-bound; the job is coalescing + enough waves in
//     flight to hide HBM latency (~900 cyc misses on the random-row gathers).
//   - Rows are padded to 16 B (Schema.row_stride), so row copies are uint4
//     (dwordx4) moves: 16 B/lane/instruction.
//   - Writes are always coalesced (consecutive lanes -> consecutive output
//     addresses); reads on the permuted side are row-random but contiguous
//     within a row, which the L2/L3 absorb at dataset-shard scale.
//   - Grids are sized ≫ 256 workgroups so all 8 XCDs fill; the default
//     blockIdx -> data mapping already round-robins XCDs (block b runs on
//     XCD b%8), which spreads the random-gather traffic across all 8 L2s.
//   - Wave width 64 is assumed (gfx950); no warp-32 idioms.

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define RSDL_MAX_COLS 128

namespace rsdl {

// ---------------------------------------------------------------------------
// gather_rows: dst[i][:] = src[perm[i]][:], row_stride multiple of 16 B.
// One uint4 (16 B) per thread; consecutive threads cover one row then the
// next, so global stores are perfectly coalesced and loads are contiguous
// segments of row_stride bytes at a random row base.
// ---------------------------------------------------------------------------
__global__ void gather_rows_u4_kernel(
    const uint4* __restrict__ src,
    uint4* __restrict__ dst,
    const int64_t* __restrict__ perm,
    int64_t n_rows,
    int64_t row_u4) {
  const int64_t total = n_rows * row_u4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t row = i / row_u4;
    const int64_t j = i - row * row_u4;
    dst[i] = src[perm[row] * row_u4 + j];
  }
}

// Identity-permutation variant (pure packed copy, used for the exchange
// staging path): dst[i] = src[sel[i]] with sel given as 32-bit indices.
__global__ void gather_rows_u4_idx32_kernel(
    const uint4* __restrict__ src,
    uint4* __restrict__ dst,
    const int32_t* __restrict__ perm,
    int64_t n_rows,
    int64_t row_u4) {
  const int64_t total = n_rows * row_u4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int64_t row = i / row_u4;
    const int64_t j = i - row * row_u4;
    dst[i] = src[(int64_t)perm[row] * row_u4 + j];
  }
}

// ---------------------------------------------------------------------------
// Fused unpack+permute+cast and pack+cast.
//
// Column descriptors are passed as a kernel argument struct (lands in
// scalar registers / constant memory — no extra global loads).
// ---------------------------------------------------------------------------

enum DType : int32_t {
  DT_F32 = 0,
  DT_F64 = 1,
  DT_I32 = 2,
  DT_I64 = 3,
  DT_F16 = 4,
  DT_BF16 = 5,
  DT_U8 = 6,
};

struct ColDesc {
  int64_t col_ptr;     // device pointer to the contiguous column tensor
  int32_t packed_off;  // byte offset of this column inside a packed row
  int32_t src_dtype;   // dtype inside the packed row
  int32_t dst_dtype;   // dtype of the column tensor
  int32_t numel;       // elements per row for this column
};

struct ColTable {
  ColDesc cols[RSDL_MAX_COLS];
};

template <typename S, typename D>
__device__ __forceinline__ D cast_elem(S v) {
  return (D)v;
}
template <>
__device__ __forceinline__ __hip_bfloat16 cast_elem(float v) {
  return __float2bfloat16(v);
}
template <>
__device__ __forceinline__ __hip_bfloat16 cast_elem(double v) {
  return __float2bfloat16((float)v);
}
template <>
__device__ __forceinline__ __hip_bfloat16 cast_elem(int32_t v) {
  return __float2bfloat16((float)v);
}
template <>
__device__ __forceinline__ __hip_bfloat16 cast_elem(int64_t v) {
  return __float2bfloat16((float)v);
}
template <>
__device__ __forceinline__ float cast_elem(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <>
__device__ __forceinline__ double cast_elem(__hip_bfloat16 v) {
  return (double)__bfloat162float(v);
}

// Vectorized unpack specializations for the hot flagship combos: the
// feature-matrix column is (f32 -> f32) or (f32 -> bf16) with numel % 4 == 0
// and a 16-B-aligned packed offset, so each lane moves one float4 (16 B
// load) instead of four scalar dwords. ~2x on the fused unpack
// (profiles/microbench: scalar 2.6 TB/s vs gather roofline 4.8 TB/s).
__device__ __forceinline__ void unpack_vec4_f32_f32(
    const uint8_t* __restrict__ packed, int64_t row_stride,
    const int64_t* __restrict__ perm, float* __restrict__ dst,
    int32_t packed_off, int32_t numel, int64_t n_rows, int64_t tid,
    int64_t nthreads) {
  const int32_t groups = numel >> 2;
  const int64_t total = n_rows * groups;
  for (int64_t i = tid; i < total; i += nthreads) {
    const int64_t row = i / groups;
    const int64_t g = i - row * groups;
    const int64_t srow = perm ? perm[row] : row;
    const float4 v = *reinterpret_cast<const float4*>(
        packed + srow * row_stride + packed_off + (g << 4));
    *reinterpret_cast<float4*>(dst + row * numel + (g << 2)) = v;
  }
}

__device__ __forceinline__ void unpack_vec4_f32_bf16(
    const uint8_t* __restrict__ packed, int64_t row_stride,
    const int64_t* __restrict__ perm, __hip_bfloat16* __restrict__ dst,
    int32_t packed_off, int32_t numel, int64_t n_rows, int64_t tid,
    int64_t nthreads) {
  const int32_t groups = numel >> 2;
  const int64_t total = n_rows * groups;
  for (int64_t i = tid; i < total; i += nthreads) {
    const int64_t row = i / groups;
    const int64_t g = i - row * groups;
    const int64_t srow = perm ? perm[row] : row;
    const float4 v = *reinterpret_cast<const float4*>(
        packed + srow * row_stride + packed_off + (g << 4));
    union {
      __hip_bfloat16 h[4];
      uint2 u;
    } o;
    o.h[0] = __float2bfloat16(v.x);
    o.h[1] = __float2bfloat16(v.y);
    o.h[2] = __float2bfloat16(v.z);
    o.h[3] = __float2bfloat16(v.w);
    *reinterpret_cast<uint2*>(dst + row * numel + (g << 2)) = o.u;
  }
}

// Unpack direction: column[i*numel + e] = cast(packed[perm[i]*stride + off +
// e*sizeof(S)]). Writes coalesced; reads random-row.
template <typename S, typename D>
__device__ void unpack_col_loop(
    const uint8_t* __restrict__ packed,
    int64_t row_stride,
    const int64_t* __restrict__ perm,
    D* __restrict__ dst,
    int32_t packed_off,
    int32_t numel,
    int64_t n_rows,
    int64_t tid,
    int64_t nthreads) {
  const int64_t total = n_rows * numel;
  for (int64_t i = tid; i < total; i += nthreads) {
    const int64_t row = i / numel;
    const int64_t e = i - row * numel;
    const int64_t srow = perm ? perm[row] : row;
    const S* sp = reinterpret_cast<const S*>(
        packed + srow * row_stride + packed_off);
    dst[i] = cast_elem<S, D>(sp[e]);
  }
}

// Pack direction: packed[perm[i]*stride + off + e] = cast(column[i*numel+e])
// (perm==null => packed[i*...]). Reads coalesced; writes strided by row.
template <typename S, typename D>
__device__ void pack_col_loop(
    uint8_t* __restrict__ packed,
    int64_t row_stride,
    const int64_t* __restrict__ perm,
    const S* __restrict__ src,
    int32_t packed_off,
    int32_t numel,
    int64_t n_rows,
    int64_t tid,
    int64_t nthreads) {
  const int64_t total = n_rows * numel;
  for (int64_t i = tid; i < total; i += nthreads) {
    const int64_t row = i / numel;
    const int64_t e = i - row * numel;
    const int64_t drow = perm ? perm[row] : row;
    D* dp = reinterpret_cast<D*>(packed + drow * row_stride + packed_off);
    dp[e] = cast_elem<S, D>(src[i]);
  }
}

#define RSDL_DISPATCH_DST(S_CTYPE, SRC_DT, DST_DT, CALL)                   \
  switch (DST_DT) {                                                        \
    case DT_F32: { using D = float; CALL; break; }                         \
    case DT_F64: { using D = double; CALL; break; }                        \
    case DT_I32: { using D = int32_t; CALL; break; }                       \
    case DT_I64: { using D = int64_t; CALL; break; }                       \
    case DT_BF16: { using D = __hip_bfloat16; CALL; break; }               \
    default: break;                                                        \
  }

#define RSDL_DISPATCH_PAIR(SRC_DT, DST_DT, CALL)                           \
  switch (SRC_DT) {                                                        \
    case DT_F32: { using S = float; RSDL_DISPATCH_DST(float, SRC_DT,       \
                   DST_DT, CALL); break; }                                 \
    case DT_F64: { using S = double; RSDL_DISPATCH_DST(double, SRC_DT,     \
                   DST_DT, CALL); break; }                                 \
    case DT_I32: { using S = int32_t; RSDL_DISPATCH_DST(int32_t, SRC_DT,   \
                   DST_DT, CALL); break; }                                 \
    case DT_I64: { using S = int64_t; RSDL_DISPATCH_DST(int64_t, SRC_DT,   \
                   DST_DT, CALL); break; }                                 \
    default: break;                                                        \
  }

// Work is partitioned by column via blockIdx.y; blockIdx.x strides within a
// column. Columns have very different sizes, but the grid-stride loop keeps
// all blocks busy until their column is finished; with >=1024 blocks per
// column the chip stays full.
__global__ void unpack_permute_kernel(
    const uint8_t* __restrict__ packed,
    int64_t row_stride,
    const int64_t* __restrict__ perm,
    ColTable table,
    int32_t num_cols,
    int64_t n_rows) {
  const int32_t c = blockIdx.y;
  if (c >= num_cols) return;
  const ColDesc d = table.cols[c];
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  // Hot-path vector specializations (flagship feature matrix).
  if (d.src_dtype == DT_F32 && (d.numel & 3) == 0 &&
      (d.packed_off & 15) == 0) {
    if (d.dst_dtype == DT_F32) {
      unpack_vec4_f32_f32(packed, row_stride, perm,
                          reinterpret_cast<float*>(d.col_ptr), d.packed_off,
                          d.numel, n_rows, tid, nthreads);
      return;
    }
    if (d.dst_dtype == DT_BF16) {
      unpack_vec4_f32_bf16(packed, row_stride, perm,
                           reinterpret_cast<__hip_bfloat16*>(d.col_ptr),
                           d.packed_off, d.numel, n_rows, tid, nthreads);
      return;
    }
  }
  RSDL_DISPATCH_PAIR(
      d.src_dtype, d.dst_dtype,
      (unpack_col_loop<S, D>(packed, row_stride, perm,
                             reinterpret_cast<D*>(d.col_ptr), d.packed_off,
                             d.numel, n_rows, tid, nthreads)));
}

__global__ void pack_columns_kernel(
    uint8_t* __restrict__ packed,
    int64_t row_stride,
    const int64_t* __restrict__ perm,
    ColTable table,
    int32_t num_cols,
    int64_t n_rows) {
  const int32_t c = blockIdx.y;
  if (c >= num_cols) return;
  const ColDesc d = table.cols[c];
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t nthreads = (int64_t)gridDim.x * blockDim.x;
  RSDL_DISPATCH_PAIR(
      d.dst_dtype, d.src_dtype,
      (pack_col_loop<S, D>(packed, row_stride, perm,
                           reinterpret_cast<const S*>(d.col_ptr),
                           d.packed_off, d.numel, n_rows, tid, nthreads)));
}

// ---------------------------------------------------------------------------
// Radix partition by destination id (replaces torch argsort+bincount on the
// map side; reference shuffle.py:156-161 boolean-mask loop). Three phases:
//   1. per-block LDS histogram of dest ids -> block_counts[B][T]
//   2. exclusive scan on host side (tiny [B,T] tensor, torch cumsum)
//   3. rank+scatter: per-block LDS running counters assign each row its
//      output slot; writes the GATHER permutation (perm[dst] = src row),
//      which then drives the roofline-speed gather_rows kernel.
// Within-destination order is block-local-nondeterministic, which is fine:
// a full random permutation is applied downstream either way.
// ---------------------------------------------------------------------------

#define RSDL_PART_MAX_DESTS 1024

__global__ void partition_hist_kernel(
    const int32_t* __restrict__ dest,
    int32_t* __restrict__ block_counts,  // [gridDim.x][num_dests]
    int64_t n,
    int32_t num_dests) {
  extern __shared__ int32_t lds_counts[];
  for (int32_t d = threadIdx.x; d < num_dests; d += blockDim.x)
    lds_counts[d] = 0;
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t lo = (int64_t)blockIdx.x * chunk;
  const int64_t hi = min(lo + chunk, n);
  for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x)
    atomicAdd(&lds_counts[dest[i]], 1);
  __syncthreads();
  for (int32_t d = threadIdx.x; d < num_dests; d += blockDim.x)
    block_counts[(int64_t)blockIdx.x * num_dests + d] = lds_counts[d];
}

__global__ void partition_scatter_kernel(
    const int32_t* __restrict__ dest,
    const int32_t* __restrict__ block_base,  // [gridDim.x][num_dests]
    int32_t* __restrict__ perm_out,          // [n] gather indices
    int64_t n,
    int32_t num_dests) {
  extern __shared__ int32_t lds_next[];
  for (int32_t d = threadIdx.x; d < num_dests; d += blockDim.x)
    lds_next[d] = block_base[(int64_t)blockIdx.x * num_dests + d];
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t lo = (int64_t)blockIdx.x * chunk;
  const int64_t hi = min(lo + chunk, n);
  for (int64_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    const int32_t pos = atomicAdd(&lds_next[dest[i]], 1);
    perm_out[pos] = (int32_t)i;
  }
}

// ---------------------------------------------------------------------------
// Tiled pack for the map side: C scalar/vector columns -> packed rows.
// The naive per-element kernel stores 4-8 B at row_stride intervals
// (~0.5 TB/s); this one stages a TILE_R-row tile in LDS (per-column
// coalesced loads, conflict-spread by the padded row stride) and then
// streams the tile out as one contiguous row-major block (16-B lanes,
// perfectly coalesced).
// ---------------------------------------------------------------------------

__global__ void pack_tiled_kernel(
    uint8_t* __restrict__ packed,
    int64_t row_stride,
    ColTable table,
    int32_t num_cols,
    int64_t n_rows,
    int32_t tile_rows,
    int32_t n8) {  // n8 >= 0: fast path; first n8 columns are 8-B scalars,
                   // the rest 4-B scalars, no casts. n8 < 0: generic path.
  extern __shared__ uint8_t lds_tile[];  // [tile_rows][lds_stride]
  // +8 B pad keeps 8-B column alignment inside LDS and makes the dword
  // stride ≡ 2 (mod 4): per-column lane strides cover 16 of 32 banks
  // (gcd=2 -> at worst 2-way, which is free for ds_write_b32).
  const int64_t lds_stride = row_stride + 8;
  const int64_t row0 = (int64_t)blockIdx.x * tile_rows;
  const int32_t rows_here =
      (int32_t)min((int64_t)tile_rows, n_rows - row0);
  if (rows_here <= 0) return;

  if (n8 >= 0) {
    // Fast phase 1: column plan staged in LDS-adjacent static shared
    // memory; flat (column-major) element loop, no per-element dispatch.
    // Loads are perfectly coalesced (consecutive lanes -> consecutive rows
    // of one column); each wave stays within one column.
    __shared__ int64_t s_ptr[RSDL_MAX_COLS];
    __shared__ int32_t s_off[RSDL_MAX_COLS];
    for (int32_t c = threadIdx.x; c < num_cols; c += blockDim.x) {
      s_ptr[c] = table.cols[c].col_ptr;
      s_off[c] = table.cols[c].packed_off;
    }
    __syncthreads();
    const int64_t total8 = (int64_t)rows_here * n8;
    for (int64_t t = threadIdx.x; t < total8; t += blockDim.x) {
      const int32_t c = (int32_t)(t / rows_here);
      const int32_t r = (int32_t)(t - (int64_t)c * rows_here);
      const uint64_t v =
          reinterpret_cast<const uint64_t*>(s_ptr[c])[row0 + r];
      *reinterpret_cast<uint64_t*>(
          lds_tile + (int64_t)r * lds_stride + s_off[c]) = v;
    }
    const int32_t n4 = num_cols - n8;
    const int64_t total4 = (int64_t)rows_here * n4;
    for (int64_t t = threadIdx.x; t < total4; t += blockDim.x) {
      const int32_t c = n8 + (int32_t)(t / rows_here);
      const int32_t r = (int32_t)(t - (int64_t)(c - n8) * rows_here);
      const uint32_t v =
          reinterpret_cast<const uint32_t*>(s_ptr[c])[row0 + r];
      *reinterpret_cast<uint32_t*>(
          lds_tile + (int64_t)r * lds_stride + s_off[c]) = v;
    }
  } else {
    // Generic phase 1: per-column loop with dtype-cast dispatch.
    for (int32_t c = 0; c < num_cols; ++c) {
      const ColDesc d = table.cols[c];
      const int64_t elems = (int64_t)rows_here * d.numel;
      for (int64_t t = threadIdx.x; t < elems; t += blockDim.x) {
        const int32_t r = (int32_t)(t / d.numel);
        const int32_t e = (int32_t)(t - (int64_t)r * d.numel);
        uint8_t* dst_base = lds_tile + (int64_t)r * lds_stride + d.packed_off;
        RSDL_DISPATCH_PAIR(
            d.dst_dtype, d.src_dtype,
            (reinterpret_cast<D*>(dst_base)[e] = cast_elem<S, D>(
                 reinterpret_cast<const S*>(d.col_ptr)[row0 * d.numel + t])));
      }
    }
  }
  __syncthreads();

  // Phase 2: stream the tile to global as contiguous row-major dwords
  // (consecutive lanes -> consecutive banks and fully coalesced stores;
  // the padded LDS stride rules out 16-B-aligned reads).
  const int64_t row_dw = row_stride >> 2;
  const int64_t lds_stride_dw = lds_stride >> 2;
  const int64_t total_dw = (int64_t)rows_here * row_dw;
  uint32_t* gout = reinterpret_cast<uint32_t*>(packed + row0 * row_stride);
  const uint32_t* lin = reinterpret_cast<const uint32_t*>(lds_tile);
  for (int64_t t = threadIdx.x; t < total_dw; t += blockDim.x) {
    const int64_t r = t / row_dw;
    const int64_t o = t - r * row_dw;
    gout[t] = lin[r * lds_stride_dw + o];
  }
}

// ---------------------------------------------------------------------------
// Host-side launchers (called from shuffle_ops.cpp).
// ---------------------------------------------------------------------------

void launch_partition_hist(
    const int32_t* dest, int32_t* block_counts, int64_t n,
    int32_t num_dests, int32_t num_blocks, hipStream_t stream) {
  hipLaunchKernelGGL(partition_hist_kernel, dim3(num_blocks), dim3(256),
                     num_dests * sizeof(int32_t), stream, dest, block_counts,
                     n, num_dests);
}

void launch_partition_scatter(
    const int32_t* dest, const int32_t* block_base, int32_t* perm_out,
    int64_t n, int32_t num_dests, int32_t num_blocks, hipStream_t stream) {
  hipLaunchKernelGGL(partition_scatter_kernel, dim3(num_blocks), dim3(256),
                     num_dests * sizeof(int32_t), stream, dest, block_base,
                     perm_out, n, num_dests);
}

void launch_pack_tiled(
    void* packed, int64_t row_stride, const ColTable& table,
    int32_t num_cols, int64_t n_rows, int32_t tile_rows, int64_t lds_bytes,
    int32_t n8, hipStream_t stream) {
  const int64_t blocks = (n_rows + tile_rows - 1) / tile_rows;
  hipLaunchKernelGGL(pack_tiled_kernel, dim3((uint32_t)blocks), dim3(256),
                     (uint32_t)lds_bytes, stream,
                     reinterpret_cast<uint8_t*>(packed), row_stride, table,
                     num_cols, n_rows, tile_rows, n8);
}

void launch_gather_rows(
    const void* src, void* dst, const int64_t* perm, int64_t n_rows,
    int64_t row_bytes, hipStream_t stream) {
  const int64_t row_u4 = row_bytes / 16;
  const int64_t total = n_rows * row_u4;
  const int threads = 256;
  int64_t blocks = (total + threads - 1) / threads;
  if (blocks > 65535 * 16) blocks = 65535 * 16;  // grid-stride covers rest
  hipLaunchKernelGGL(gather_rows_u4_kernel, dim3((uint32_t)blocks),
                     dim3(threads), 0, stream,
                     reinterpret_cast<const uint4*>(src),
                     reinterpret_cast<uint4*>(dst), perm, n_rows, row_u4);
}

void launch_gather_rows_idx32(
    const void* src, void* dst, const int32_t* perm, int64_t n_rows,
    int64_t row_bytes, hipStream_t stream) {
  const int64_t row_u4 = row_bytes / 16;
  const int64_t total = n_rows * row_u4;
  const int threads = 256;
  int64_t blocks = (total + threads - 1) / threads;
  if (blocks > 65535 * 16) blocks = 65535 * 16;
  hipLaunchKernelGGL(gather_rows_u4_idx32_kernel, dim3((uint32_t)blocks),
                     dim3(threads), 0, stream,
                     reinterpret_cast<const uint4*>(src),
                     reinterpret_cast<uint4*>(dst), perm, n_rows, row_u4);
}

void launch_unpack_permute(
    const void* packed, int64_t row_stride, const int64_t* perm,
    const ColTable& table, int32_t num_cols, int64_t n_rows,
    hipStream_t stream) {
  const int threads = 256;
  // ~4096 blocks per column keeps 256 CUs busy even for a single column.
  int64_t bx = (n_rows + threads - 1) / threads;
  if (bx > 4096) bx = 4096;
  if (bx < 1) bx = 1;
  dim3 grid((uint32_t)bx, (uint32_t)num_cols);
  hipLaunchKernelGGL(unpack_permute_kernel, grid, dim3(threads), 0, stream,
                     reinterpret_cast<const uint8_t*>(packed), row_stride,
                     perm, table, num_cols, n_rows);
}

void launch_pack_columns(
    void* packed, int64_t row_stride, const int64_t* perm,
    const ColTable& table, int32_t num_cols, int64_t n_rows,
    hipStream_t stream) {
  const int threads = 256;
  int64_t bx = (n_rows + threads - 1) / threads;
  if (bx > 4096) bx = 4096;
  if (bx < 1) bx = 1;
  dim3 grid((uint32_t)bx, (uint32_t)num_cols);
  hipLaunchKernelGGL(pack_columns_kernel, grid, dim3(threads), 0, stream,
                     reinterpret_cast<uint8_t*>(packed), row_stride, perm,
                     table, num_cols, n_rows);
}

}  // namespace rsdl
