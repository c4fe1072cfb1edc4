// Torch bindings for the MI355X shuffle kernels (csrc/shuffle_kernels.hip).
//
// Exposes the fused reducer-side ops (gather-permute / unpack+cast+pack /
// pack) to Python as ray_shuffling_data_loader_amd._rsdl_hip. All ops run on
// the CALLER's current HIP stream, so the shuffle engine's side-stream
// discipline (double-buffering against the trainer step) is controlled from
// Python via torch.cuda.Stream.

#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <vector>

namespace rsdl {

// Mirror of the definitions in shuffle_kernels.hip.
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
  int64_t col_ptr;
  int32_t packed_off;
  int32_t src_dtype;
  int32_t dst_dtype;
  int32_t numel;
};

struct ColTable {
  ColDesc cols[128];
};

void launch_gather_rows(const void* src, void* dst, const int64_t* perm,
                        int64_t n_rows, int64_t row_bytes, hipStream_t stream);
void launch_gather_rows_idx32(const void* src, void* dst, const int32_t* perm,
                              int64_t n_rows, int64_t row_bytes,
                              hipStream_t stream);
void launch_unpack_permute(const void* packed, int64_t row_stride,
                           const int64_t* perm, const ColTable& table,
                           int32_t num_cols, int64_t n_rows,
                           hipStream_t stream);
void launch_pack_columns(void* packed, int64_t row_stride, const int64_t* perm,
                         const ColTable& table, int32_t num_cols,
                         int64_t n_rows, hipStream_t stream);
void launch_partition_hist(const int32_t* dest, int32_t* block_counts,
                           int64_t n, int32_t num_dests, int32_t num_blocks,
                           hipStream_t stream);
void launch_partition_scatter(const int32_t* dest, const int32_t* block_base,
                              int32_t* perm_out, int64_t n, int32_t num_dests,
                              int32_t num_blocks, hipStream_t stream);
void launch_pack_tiled(void* packed, int64_t row_stride, const ColTable& table,
                       int32_t num_cols, int64_t n_rows, int32_t tile_rows,
                       int64_t lds_bytes, int32_t n8, hipStream_t stream);
void launch_wgrad_bf16(const void* dy, const void* x, float* dW, float* db,
                       int64_t M, int32_t N, int32_t K, int32_t split_m,
                       hipStream_t stream);
void launch_wgrad_wide(const void* dy, const void* x, float* dW, float* db,
                       int64_t M, int32_t N, int32_t K, int32_t split_m,
                       hipStream_t stream);
void launch_relu_bwd_bias(const void* dy, const void* y, void* dx,
                          float* db_part, int64_t nvec, int32_t N,
                          int32_t grid, hipStream_t stream);
int64_t fwd_chain_grid(int64_t M);
void launch_swizzle_x(const void* x, void* out, int64_t M,
                      hipStream_t stream);
void launch_swizzle_xt(const void* x, void* out, int64_t M, int pi16,
                       hipStream_t stream);
void launch_swizzle_x_both(const void* x, void* xs, void* xt, int64_t M,
                           int pi16, hipStream_t stream);
void launch_fwd_chain(const void* x0s, const void* W1, const float* b1,
                      const void* W2, const float* b2, const void* W3,
                      const float* b3, const void* w4, const float* b4,
                      void* a1t, uint32_t* mask1, void* a2t,
                      uint32_t* mask2, void* a3, void* out,
                      const float* target, void* dyb, float* loss_part,
                      int64_t M, int pi16, hipStream_t stream);
int64_t bwd_chain_grid(int64_t M);
void launch_wgrad_frag(const void* AT, const void* BT, float* dW, int32_t N,
                       int32_t K, int64_t mchunks, int32_t nt_w,
                       int32_t kt_w, hipStream_t stream);
int64_t wgrad_frag_nslabs(int64_t mchunks, int32_t N, int32_t K,
                          int32_t nt_w, int32_t kt_w);
void launch_slab_reduce(const float* part, float* out, int64_t nk,
                        int64_t nslabs, hipStream_t stream);
void launch_bwd_chain(const void* dy, const void* a3, const void* mask1,
                      const void* mask2, const void* w4, const void* W3T,
                      const void* W2T, void* dz1t, void* dz2t, void* dz3t,
                      float* db_part, int64_t M, int pi16,
                      hipStream_t stream);

namespace {

int32_t dtype_code(at::ScalarType st) {
  switch (st) {
    case at::kFloat: return DT_F32;
    case at::kDouble: return DT_F64;
    case at::kInt: return DT_I32;
    case at::kLong: return DT_I64;
    case at::kHalf: return DT_F16;
    case at::kBFloat16: return DT_BF16;
    case at::kByte: return DT_U8;
    default:
      TORCH_CHECK(false, "unsupported dtype for shuffle ops: ", st);
  }
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

int64_t row_bytes_of(const at::Tensor& t) {
  TORCH_CHECK(t.dim() == 2, "packed tensor must be 2-D [rows, row_bytes]");
  TORCH_CHECK(t.is_contiguous(), "packed tensor must be contiguous");
  int64_t rb = t.size(1) * t.element_size();
  TORCH_CHECK(rb % 16 == 0,
              "packed row byte width must be a multiple of 16, got ", rb);
  return rb;
}

// dst[i,:] = src[perm[i],:]
at::Tensor gather_rows(const at::Tensor& src, const at::Tensor& perm) {
  TORCH_CHECK(src.is_cuda() && perm.is_cuda(), "gather_rows: inputs on GPU");
  TORCH_CHECK(perm.scalar_type() == at::kLong || perm.scalar_type() == at::kInt,
              "perm must be int64 or int32");
  TORCH_CHECK(perm.is_contiguous(), "perm must be contiguous");
  int64_t rb = row_bytes_of(src);
  int64_t n = perm.numel();
  auto dst = at::empty({n, src.size(1)}, src.options());
  if (n == 0) return dst;
  if (perm.scalar_type() == at::kLong) {
    launch_gather_rows(src.data_ptr(), dst.data_ptr(),
                       perm.data_ptr<int64_t>(), n, rb, current_stream());
  } else {
    launch_gather_rows_idx32(src.data_ptr(), dst.data_ptr(),
                             perm.data_ptr<int32_t>(), n, rb,
                             current_stream());
  }
  return dst;
}

void gather_rows_out(const at::Tensor& src, const at::Tensor& perm,
                     at::Tensor& dst) {
  TORCH_CHECK(src.is_cuda() && perm.is_cuda() && dst.is_cuda(),
              "gather_rows_out: inputs on GPU");
  int64_t rb = row_bytes_of(src);
  TORCH_CHECK(row_bytes_of(dst) == rb, "dst row width mismatch");
  TORCH_CHECK(dst.size(0) >= perm.numel(), "dst too small");
  TORCH_CHECK(perm.is_contiguous(), "perm must be contiguous");
  int64_t n = perm.numel();
  if (n == 0) return;
  if (perm.scalar_type() == at::kLong) {
    launch_gather_rows(src.data_ptr(), dst.data_ptr(),
                       perm.data_ptr<int64_t>(), n, rb, current_stream());
  } else if (perm.scalar_type() == at::kInt) {
    launch_gather_rows_idx32(src.data_ptr(), dst.data_ptr(),
                             perm.data_ptr<int32_t>(), n, rb,
                             current_stream());
  } else {
    TORCH_CHECK(false, "perm must be int64 or int32");
  }
}

// outs[c][i*numel+e] = cast(packed[perm[i]*stride + off_c + e*esz])
void unpack_permute(const at::Tensor& packed,
                    const c10::optional<at::Tensor>& perm,
                    const std::vector<at::Tensor>& outs,
                    const std::vector<int64_t>& packed_offsets,
                    const std::vector<int64_t>& src_dtype_codes) {
  TORCH_CHECK(packed.is_cuda(), "packed must be on GPU");
  TORCH_CHECK(packed.scalar_type() == at::kByte, "packed must be uint8");
  TORCH_CHECK(outs.size() == packed_offsets.size() &&
                  outs.size() == src_dtype_codes.size(),
              "descriptor length mismatch");
  TORCH_CHECK(outs.size() <= 128, "at most 128 columns per launch");
  int64_t stride = packed.size(1);
  int64_t n_rows;
  const int64_t* perm_ptr = nullptr;
  if (perm.has_value()) {
    TORCH_CHECK(perm->is_cuda() && perm->scalar_type() == at::kLong &&
                    perm->is_contiguous(),
                "perm must be contiguous int64 on GPU");
    n_rows = perm->numel();
    perm_ptr = perm->data_ptr<int64_t>();
  } else {
    n_rows = packed.size(0);
  }
  if (n_rows == 0 || outs.empty()) return;
  ColTable table{};
  for (size_t c = 0; c < outs.size(); ++c) {
    const auto& o = outs[c];
    TORCH_CHECK(o.is_cuda() && o.is_contiguous(),
                "output column must be contiguous GPU tensor");
    TORCH_CHECK(o.size(0) == n_rows, "output rows mismatch");
    table.cols[c] = ColDesc{
        reinterpret_cast<int64_t>(o.data_ptr()),
        (int32_t)packed_offsets[c],
        (int32_t)src_dtype_codes[c],
        dtype_code(o.scalar_type()),
        (int32_t)(o.numel() / n_rows),
    };
  }
  launch_unpack_permute(packed.data_ptr(), stride, perm_ptr, table,
                        (int32_t)outs.size(), n_rows, current_stream());
}

// packed[perm[i]*stride + off_c + e*esz] = cast(cols[c][i*numel+e])
at::Tensor pack_columns(const std::vector<at::Tensor>& cols,
                        const std::vector<int64_t>& packed_offsets,
                        const std::vector<int64_t>& packed_dtype_codes,
                        int64_t row_stride,
                        const c10::optional<at::Tensor>& perm) {
  TORCH_CHECK(!cols.empty(), "need at least one column");
  TORCH_CHECK(cols.size() <= 128, "at most 128 columns per launch");
  TORCH_CHECK(row_stride % 16 == 0, "row_stride must be multiple of 16");
  int64_t n_rows = cols[0].size(0);
  const int64_t* perm_ptr = nullptr;
  if (perm.has_value()) {
    TORCH_CHECK(perm->is_cuda() && perm->scalar_type() == at::kLong &&
                    perm->is_contiguous(),
                "perm must be contiguous int64 on GPU");
    TORCH_CHECK(perm->numel() == n_rows, "perm length mismatch");
    perm_ptr = perm->data_ptr<int64_t>();
  }
  auto packed = at::empty(
      {n_rows, row_stride},
      cols[0].options().dtype(at::kByte));
  ColTable table{};
  for (size_t c = 0; c < cols.size(); ++c) {
    const auto& t = cols[c];
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "input column must be contiguous GPU tensor");
    TORCH_CHECK(t.size(0) == n_rows, "column rows mismatch");
    table.cols[c] = ColDesc{
        reinterpret_cast<int64_t>(t.data_ptr()),
        (int32_t)packed_offsets[c],
        (int32_t)packed_dtype_codes[c],
        dtype_code(t.scalar_type()),
        (int32_t)(t.numel() / n_rows),
    };
  }
  if (n_rows > 0) {
    launch_pack_columns(packed.data_ptr(), row_stride, perm_ptr, table,
                        (int32_t)cols.size(), n_rows, current_stream());
  }
  return packed;
}

// Build the gather permutation that groups rows by destination id.
// Returns (perm int32 [n], counts int64 [num_dests]).
std::vector<at::Tensor> partition_build_perm(const at::Tensor& dest,
                                             int64_t num_dests) {
  TORCH_CHECK(dest.is_cuda() && dest.is_contiguous(),
              "dest must be contiguous GPU tensor");
  TORCH_CHECK(num_dests >= 1 && num_dests <= 1024,
              "num_dests must be in [1, 1024]");
  auto dest32 = dest.scalar_type() == at::kInt ? dest : dest.to(at::kInt);
  int64_t n = dest32.numel();
  int32_t num_blocks = 1024;
  if (n < (int64_t)num_blocks * 256) {
    num_blocks = (int32_t)std::max<int64_t>(1, (n + 255) / 256);
  }
  auto opts32 = dest32.options();
  auto block_counts = at::empty({num_blocks, num_dests}, opts32);
  auto stream = current_stream();
  if (n > 0) {
    launch_partition_hist(dest32.data_ptr<int32_t>(),
                          block_counts.data_ptr<int32_t>(), n,
                          (int32_t)num_dests, num_blocks, stream);
  } else {
    block_counts.zero_();
  }
  // Exclusive scan (dest-major, then block-major within a dest), done with
  // torch ops on the same stream: tiny [B, T] tensor.
  auto counts64 = block_counts.to(at::kLong);
  auto per_dest = counts64.sum(0);                    // [T]
  auto dest_base = per_dest.cumsum(0) - per_dest;     // exclusive [T]
  auto within = counts64.cumsum(0) - counts64;        // exclusive over blocks
  auto base = (within + dest_base.unsqueeze(0)).to(at::kInt).contiguous();
  auto perm = at::empty({n}, opts32);
  if (n > 0) {
    launch_partition_scatter(dest32.data_ptr<int32_t>(),
                             base.data_ptr<int32_t>(),
                             perm.data_ptr<int32_t>(), n, (int32_t)num_dests,
                             num_blocks, stream);
  }
  return {perm, per_dest};
}

// LDS-tiled pack (no scatter perm): cols -> packed rows. ``out`` may be a
// contiguous [n_rows, row_stride] uint8 view (e.g. a row-slice of a larger
// preallocated source tensor) to pack in place.
at::Tensor pack_columns_tiled(const std::vector<at::Tensor>& cols,
                              const std::vector<int64_t>& packed_offsets,
                              const std::vector<int64_t>& packed_dtype_codes,
                              int64_t row_stride,
                              const c10::optional<at::Tensor>& out) {
  TORCH_CHECK(!cols.empty() && cols.size() <= 128, "1..128 columns");
  TORCH_CHECK(row_stride % 16 == 0, "row_stride must be multiple of 16");
  int64_t n_rows = cols[0].size(0);
  at::Tensor packed;
  if (out.has_value()) {
    packed = *out;
    TORCH_CHECK(packed.is_cuda() && packed.is_contiguous() &&
                    packed.scalar_type() == at::kByte &&
                    packed.size(0) == n_rows &&
                    packed.size(1) == row_stride,
                "out must be contiguous uint8 [n_rows, row_stride]");
  } else {
    packed =
        at::empty({n_rows, row_stride}, cols[0].options().dtype(at::kByte));
  }
  ColTable table{};
  for (size_t c = 0; c < cols.size(); ++c) {
    const auto& t = cols[c];
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "input column must be contiguous GPU tensor");
    TORCH_CHECK(t.size(0) == n_rows, "column rows mismatch");
    table.cols[c] = ColDesc{
        reinterpret_cast<int64_t>(t.data_ptr()),
        (int32_t)packed_offsets[c],
        (int32_t)packed_dtype_codes[c],
        dtype_code(t.scalar_type()),
        (int32_t)(t.numel() / n_rows),
    };
  }
  const int64_t lds_stride = row_stride + 8;
  int32_t tile_rows = (int32_t)std::min<int64_t>(128, 65536 / lds_stride);
  TORCH_CHECK(tile_rows >= 1, "row_stride too large for tiled pack");
  // Fast-path eligibility: scalar columns, no casts, 8-B columns first
  // (guaranteed by Schema's descending-size packing) then 4-B columns.
  int32_t n8 = 0;
  bool fast = true;
  bool in8 = true;
  for (size_t c = 0; c < cols.size(); ++c) {
    const ColDesc& d = table.cols[c];
    if (d.numel != 1 || d.src_dtype != d.dst_dtype) { fast = false; break; }
    int esz = (d.src_dtype == DT_F64 || d.src_dtype == DT_I64) ? 8
              : (d.src_dtype == DT_F32 || d.src_dtype == DT_I32) ? 4 : 0;
    if (esz == 0) { fast = false; break; }
    if (esz == 8) {
      if (!in8) { fast = false; break; }
      n8++;
    } else {
      in8 = false;
    }
  }
  if (n_rows > 0) {
    launch_pack_tiled(packed.data_ptr(), row_stride, table,
                      (int32_t)cols.size(), n_rows, tile_rows,
                      lds_stride * tile_rows, fast ? n8 : -1,
                      current_stream());
  }
  return packed;
}

// MFMA split-M weight gradient: dW = dy^T @ x (+ db = dy.sum(0)).
// Returns (dW fp32 [N,K], db fp32 [N] or empty).
std::vector<at::Tensor> wgrad_bf16(const at::Tensor& dy, const at::Tensor& x,
                                   bool with_bias) {
  TORCH_CHECK(dy.is_cuda() && x.is_cuda(), "wgrad: inputs on GPU");
  TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                  x.scalar_type() == at::kBFloat16,
              "wgrad: bf16 inputs only");
  TORCH_CHECK(dy.dim() == 2 && x.dim() == 2 && dy.size(0) == x.size(0),
              "wgrad: dy [M,N], x [M,K]");
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous(),
              "wgrad: contiguous inputs");
  int64_t M = dy.size(0);
  int32_t N = (int32_t)dy.size(1);
  int32_t K = (int32_t)x.size(1);
  bool wide = getenv("RSDL_WGRAD_WIDE") != nullptr && N >= 64 && K >= 64 &&
              M >= 4096;
  if (wide) {
    // Wide-tile kernel wants 128-multiples; pad (cheap relative to the
    // re-read amplification it removes) and narrow the result.
    auto dyp = (N % 128) ? at::constant_pad_nd(dy, {0, 128 - N % 128})
                         : dy.contiguous();
    auto xp = (K % 128) ? at::constant_pad_nd(x, {0, 128 - K % 128})
                        : x.contiguous();
    int32_t Np = (int32_t)dyp.size(1), Kp = (int32_t)xp.size(1);
    auto dWp = at::zeros({Np, Kp}, dy.options().dtype(at::kFloat));
    auto dbp = with_bias ? at::zeros({Np}, dy.options().dtype(at::kFloat))
                         : at::Tensor();
    int32_t tiles = (Np / 128) * (Kp / 128);
    int32_t split = (int32_t)std::min<int64_t>(
        std::max<int64_t>(1, 2048 / std::max(1, tiles)),
        std::max<int64_t>(1, M / 64));
    launch_wgrad_wide(dyp.data_ptr(), xp.data_ptr(), dWp.data_ptr<float>(),
                      with_bias ? dbp.data_ptr<float>() : nullptr, M, Np, Kp,
                      split, current_stream());
    auto dW = dWp.narrow(0, 0, N).narrow(1, 0, K).contiguous();
    auto db = with_bias ? dbp.narrow(0, 0, N).contiguous() : at::Tensor();
    return {dW, db};
  }
  // 64-tile kernel: pad N/K to 64-multiples so the hot loop runs only the
  // unguarded vector-load stages (per-element guards serialize loads).
  auto dyp = (N % 64) ? at::constant_pad_nd(dy, {0, 64 - N % 64})
                      : dy.contiguous();
  auto xp = (K % 64) ? at::constant_pad_nd(x, {0, 64 - K % 64})
                     : x.contiguous();
  int32_t Np = (int32_t)dyp.size(1), Kp = (int32_t)xp.size(1);
  auto dWp = at::zeros({Np, Kp}, dy.options().dtype(at::kFloat));
  auto dbp = with_bias ? at::zeros({Np}, dy.options().dtype(at::kFloat))
                       : at::Tensor();
  int32_t tiles = (Np / 64) * (Kp / 64);
  int32_t split = (int32_t)std::min<int64_t>(
      std::max<int64_t>(1, 2048 / tiles), std::max<int64_t>(1, M / 64));
  if (const char* ov = getenv("RSDL_WGRAD_SPLIT")) {
    split = std::max(1, atoi(ov));
  }
  if (M > 0) {
    launch_wgrad_bf16(dyp.data_ptr(), xp.data_ptr(), dWp.data_ptr<float>(),
                      with_bias ? dbp.data_ptr<float>() : nullptr, M, Np, Kp,
                      split, current_stream());
  }
  auto dW = (Np == N && Kp == K)
                ? dWp
                : dWp.narrow(0, 0, N).narrow(1, 0, K).contiguous();
  auto db = with_bias
                ? ((Np == N) ? dbp : dbp.narrow(0, 0, N).contiguous())
                : at::Tensor();
  return {dW, db};
}


// Fused ReLU-backward + bias gradient: dx = dy * (y > 0), db = dx.sum(0).
// One pass over HBM instead of threshold_backward + a separate column
// reduce (csrc/relu_bwd.hip). Requires bf16, contiguous, N a power of two
// >= 8 (the LinearReLU widths); callers fall back to torch otherwise.
std::vector<at::Tensor> relu_bwd_bias(const at::Tensor& dy,
                                      const at::Tensor& y) {
  TORCH_CHECK(dy.is_cuda() && y.is_cuda(), "relu_bwd_bias: inputs on GPU");
  TORCH_CHECK(dy.scalar_type() == at::kBFloat16 &&
                  y.scalar_type() == at::kBFloat16,
              "relu_bwd_bias: bf16 inputs only");
  TORCH_CHECK(dy.dim() == 2 && dy.sizes() == y.sizes(),
              "relu_bwd_bias: dy and y must be equal 2-D shapes");
  TORCH_CHECK(dy.is_contiguous() && y.is_contiguous(),
              "relu_bwd_bias: contiguous inputs");
  const int64_t M = dy.size(0);
  const int64_t N = dy.size(1);
  TORCH_CHECK(N >= 8 && N <= 2048 && (N & (N - 1)) == 0,
              "relu_bwd_bias: N must be a power of two in [8, 2048]");
  auto dx = at::empty_like(dy);
  const int64_t nvec = M * N / 8;
  int32_t grid = (int32_t)std::min<int64_t>(
      1024, std::max<int64_t>(1, nvec / (256 * 8)));
  auto db_part = at::empty({grid, N}, dy.options().dtype(at::kFloat));
  if (nvec > 0) {
    launch_relu_bwd_bias(dy.data_ptr(), y.data_ptr(), dx.data_ptr(),
                         db_part.data_ptr<float>(), nvec, (int32_t)N, grid,
                         current_stream());
  } else {
    db_part.zero_();
  }
  return {dx, db_part.sum(0)};
}


// EXPERIMENTAL round-2 fused forward chain (csrc/fwd_chain.hip): fixed
// TabularMLP(100-512-256-128-1) architecture, bf16. Returns
// (a1, a2, a3, out). Exercised only by the RSDL_EXPERIMENTAL=1 GPU test.
// Fragment-major weight swizzle for the chain kernels (see
// csrc/fwd_chain.hip): [N,K] -> [N/32][K/16][2][32][8] flat, so one
// wave's B-fragment load is a contiguous 1 KB block.
static at::Tensor swizzle_frag(const at::Tensor& W) {
  int64_t N = W.size(0), K = W.size(1);
  TORCH_CHECK(N % 32 == 0 && K % 16 == 0, "swizzle_frag: bad shape");
  return W.view({N / 32, 32, K / 16, 2, 8})
      .permute({0, 2, 3, 1, 4})
      .contiguous()
      .view({-1});
}

std::vector<at::Tensor> fwd_chain_bf16(
    const at::Tensor& x, const at::Tensor& W1, const at::Tensor& b1,
    const at::Tensor& W2, const at::Tensor& b2, const at::Tensor& W3,
    const at::Tensor& b3, const at::Tensor& w4, const at::Tensor& b4,
    const c10::optional<at::Tensor>& target,
    const c10::optional<at::Tensor>& xt_out, bool pi16) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
                  x.dim() == 2 && x.size(1) == 100 && x.is_contiguous(),
              "fwd_chain: x must be contiguous bf16 [M,100]");
  auto chk_w = [](const at::Tensor& W, int64_t n, int64_t k,
                  const char* nm) {
    TORCH_CHECK(W.is_cuda() && W.scalar_type() == at::kBFloat16 &&
                    W.dim() == 2 && W.size(0) == n && W.size(1) == k &&
                    W.is_contiguous(),
                "fwd_chain: ", nm, " must be contiguous bf16 [", n, ",", k,
                "]");
  };
  // W1 may arrive pre-padded [512,112] (cached by the fused-step driver)
  // or raw [512,100] (padded here, one kernel per call).
  const bool w1_padded = W1.size(1) == 112;
  chk_w(W1, 512, w1_padded ? 112 : 100, "W1");
  chk_w(W2, 256, 512, "W2");
  chk_w(W3, 128, 256, "W3");
  TORCH_CHECK(w4.numel() == 128 && w4.scalar_type() == at::kBFloat16 &&
                  w4.is_contiguous(),
              "fwd_chain: w4 must be contiguous bf16 [128]");
  auto fb = [](const at::Tensor& b, int64_t n) {
    TORCH_CHECK(b.numel() == n, "fwd_chain: bias size mismatch");
    return b.to(at::kFloat).contiguous();
  };
  auto b1f = fb(b1, 512), b2f = fb(b2, 256), b3f = fb(b3, 128),
       b4f = fb(b4, 1);
  const int64_t M = x.size(0);
  // The kernel's A-fragment loads need W1's k-dim padded 100 -> 112.
  auto W1p = w1_padded ? W1 : at::constant_pad_nd(W1, {0, 12}).contiguous();
  // a1/a2 exist only TRANSPOSED in wgrad fragment-major layout (plus
  // 32-bit relu-mask words per column) — the backward chain consumes
  // masks, the wgrad kernel consumes the fragments; nothing reads
  // row-major a1/a2.
  const int64_t mtiles = std::max<int64_t>((M + 31) / 32, 1);
  const int64_t mchunks = mtiles * 2;
  auto a1t = at::empty({16 * mchunks * 512}, x.options());
  auto a2t = at::empty({8 * mchunks * 512}, x.options());
  auto mask1 = at::empty({mtiles, 512}, x.options().dtype(at::kInt));
  auto mask2 = at::empty({mtiles, 256}, x.options().dtype(at::kInt));
  auto a3 = at::empty({M, 128}, x.options());
  auto out = at::empty({M, 1}, x.options());
  // Fused MSE epilogue: with a target, the kernel also emits the bf16
  // loss gradient dyb = (2/M)(out - target) and per-block squared-error
  // partials (loss = loss_part.sum()/M) — no eager loss/grad kernels.
  const bool with_loss = target.has_value();
  at::Tensor tgt, dyb, loss_part;
  const float* tgt_ptr = nullptr;
  void* dyb_ptr = nullptr;
  float* lp_ptr = nullptr;
  if (with_loss) {
    tgt = target->reshape({-1}).to(at::kFloat).contiguous();
    TORCH_CHECK(tgt.numel() == M, "fwd_chain: target size mismatch");
    dyb = at::empty({M, 1}, x.options());
    loss_part = at::empty({std::max<int64_t>(fwd_chain_grid(M), 1)},
                          x.options().dtype(at::kFloat));
    tgt_ptr = tgt.data_ptr<float>();
    dyb_ptr = dyb.data_ptr();
    lp_ptr = loss_part.data_ptr<float>();
  }
  if (M > 0) {
    auto W1s = swizzle_frag(W1p);
    auto W2s = swizzle_frag(W2);
    auto W3s = swizzle_frag(W3);
    // x pre-swizzled to the same fragment-major layout (per 32-row
    // m-tile): layer 1's A fragments then come straight from global,
    // coalesced, and the kernel needs no x LDS tile (occupancy 2 -> 3
    // workgroups/CU). Pads are zeroed by the swizzle kernel
    // (uninitialized bf16 could be NaN; NaN * W1pad(0) would poison
    // real rows).
    const int64_t Mp = (M + 31) / 32 * 32;
    auto xs = at::empty({Mp * 112}, x.options());
    if (xt_out.has_value()) {
      // one pass over x emits both the forward and the wgrad layouts
      launch_swizzle_x_both(x.data_ptr(), xs.data_ptr(),
                            xt_out->data_ptr(), M, pi16 ? 1 : 0,
                            current_stream());
    } else {
      launch_swizzle_x(x.data_ptr(), xs.data_ptr(), M, current_stream());
    }
    launch_fwd_chain(xs.data_ptr(), W1s.data_ptr(), b1f.data_ptr<float>(),
                     W2s.data_ptr(), b2f.data_ptr<float>(), W3s.data_ptr(),
                     b3f.data_ptr<float>(), w4.data_ptr(),
                     b4f.data_ptr<float>(), a1t.data_ptr(),
                     reinterpret_cast<uint32_t*>(mask1.data_ptr<int32_t>()),
                     a2t.data_ptr(),
                     reinterpret_cast<uint32_t*>(mask2.data_ptr<int32_t>()),
                     a3.data_ptr(), out.data_ptr(), tgt_ptr, dyb_ptr,
                     lp_ptr, M, pi16 ? 1 : 0, current_stream());
  }
  if (with_loss) return {a1t, mask1, a2t, mask2, a3, out, dyb, loss_part};
  return {a1t, mask1, a2t, mask2, a3, out};
}


// EXPERIMENTAL round-2 fused backward chain (csrc/bwd_chain.hip):
// dgrad + relu mask + bias partials for the fixed TabularMLP
// architecture. Returns (dz1, dz2, dz3, db1, db2, db3, db4); the wgrads
// (dW_l = dz_l^T @ a_{l-1}) remain the caller's streaming kernels.
// Swizzle of W^T computed directly from model-layout W [K,N] (single
// strided copy, no intermediate transpose materialization).
static at::Tensor swizzle_frag_T(const at::Tensor& W) {
  int64_t K = W.size(0), N = W.size(1);
  TORCH_CHECK(K % 16 == 0 && N % 32 == 0, "swizzle_frag_T: bad shape");
  return W.view({K / 16, 2, 8, N / 32, 32})
      .permute({3, 0, 1, 4, 2})
      .contiguous()
      .view({-1});
}

std::vector<at::Tensor> bwd_chain_bf16(
    const at::Tensor& dy, const at::Tensor& a3, const at::Tensor& mask1,
    const at::Tensor& mask2, const at::Tensor& w4, const at::Tensor& W3,
    const at::Tensor& W2, bool pi16) {
  const int64_t M = dy.size(0);
  const int64_t mtiles = std::max<int64_t>((M + 31) / 32, 1);
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16 &&
                  dy.numel() == M && dy.is_contiguous(),
              "bwd_chain: dy must be contiguous bf16 [M] or [M,1]");
  auto chk_a = [M](const at::Tensor& a, int64_t n, const char* nm) {
    TORCH_CHECK(a.is_cuda() && a.scalar_type() == at::kBFloat16 &&
                    a.dim() == 2 && a.size(0) == M && a.size(1) == n &&
                    a.is_contiguous(),
                "bwd_chain: ", nm, " must be contiguous bf16 [M,", n, "]");
  };
  chk_a(a3, 128, "a3");
  TORCH_CHECK(mask1.is_cuda() && mask1.scalar_type() == at::kInt &&
                  mask1.is_contiguous() && mask1.numel() == mtiles * 512,
              "bwd_chain: mask1 must be int32 [mtiles,512]");
  TORCH_CHECK(mask2.is_cuda() && mask2.scalar_type() == at::kInt &&
                  mask2.is_contiguous() && mask2.numel() == mtiles * 256,
              "bwd_chain: mask2 must be int32 [mtiles,256]");
  TORCH_CHECK(w4.numel() == 128 && w4.scalar_type() == at::kBFloat16,
              "bwd_chain: w4 must be bf16 [128]");
  // Weights may arrive pre-transposed ([256,128] / [512,256], cached by
  // the fused-step driver) or in model layout (transposed here per call).
  // The shapes are unambiguous between the two layouts.
  TORCH_CHECK(W3.scalar_type() == at::kBFloat16 &&
                  ((W3.size(0) == 128 && W3.size(1) == 256) ||
                   (W3.size(0) == 256 && W3.size(1) == 128)),
              "bwd_chain: W3 must be bf16 [128,256] or W3^T [256,128]");
  TORCH_CHECK(W2.scalar_type() == at::kBFloat16 &&
                  ((W2.size(0) == 256 && W2.size(1) == 512) ||
                   (W2.size(0) == 512 && W2.size(1) == 256)),
              "bwd_chain: W2 must be bf16 [256,512] or W2^T [512,256]");
  auto w4c = w4.contiguous();
  // Swizzled W^T fragments, computed directly from whichever layout was
  // passed (model [out,in] or pre-transposed [in,out]).
  auto W3Ts = (W3.size(0) == 128) ? swizzle_frag_T(W3.contiguous())
                                  : swizzle_frag(W3.contiguous());
  auto W2Ts = (W2.size(0) == 256) ? swizzle_frag_T(W2.contiguous())
                                  : swizzle_frag(W2.contiguous());
  // dz outputs exist only transposed, in wgrad fragment-major layout.
  const int64_t mchunks = mtiles * 2;
  auto dz1t = at::empty({16 * mchunks * 512}, dy.options());
  auto dz2t = at::empty({8 * mchunks * 512}, dy.options());
  auto dz3t = at::empty({4 * mchunks * 512}, dy.options());
  const int64_t grid = bwd_chain_grid(M);
  constexpr int64_t kPartW = 512 + 256 + 128 + 1 + 256;
  // Width padded to a 4-multiple for the float4 slab reduce; the 3 pad
  // columns are never read back.
  constexpr int64_t kPartWPad = (kPartW + 3) / 4 * 4;
  auto db_part = at::empty({std::max<int64_t>(grid, 1), kPartWPad},
                           dy.options().dtype(at::kFloat));
  if (M == 0) db_part.zero_();
  if (M > 0) {
    launch_bwd_chain(dy.data_ptr(), a3.data_ptr(), mask1.data_ptr(),
                     mask2.data_ptr(), w4c.data_ptr(), W3Ts.data_ptr(),
                     W2Ts.data_ptr(), dz1t.data_ptr(), dz2t.data_ptr(),
                     dz3t.data_ptr(), db_part.data_ptr<float>(), M,
                     pi16 ? 1 : 0, current_stream());
  }
  // Column reduction with the split-slab reduce kernel (the GEMV ran at
  // ~0.9 TB/s; this streams the 28 MB slab near roofline). Pad columns
  // may sum garbage; they are never read.
  auto db_full = at::zeros({kPartWPad}, dy.options().dtype(at::kFloat));
  launch_slab_reduce(db_part.data_ptr<float>(), db_full.data_ptr<float>(),
                     kPartWPad, db_part.size(0), current_stream());
  auto db = db_full.narrow(0, 0, kPartW);
  auto dw4 = (db.narrow(0, 897, 128) + db.narrow(0, 897 + 128, 128))
                 .reshape({1, 128});
  return {dz1t,
          dz2t,
          dz3t,
          db.narrow(0, 0, 512),
          db.narrow(0, 512, 256),
          db.narrow(0, 512 + 256, 128),
          db.narrow(0, 512 + 256 + 128, 1),
          dw4};
}

}  // namespace
// x [M,100] -> wgrad fragment-major x^T ([128/32][mchunks][2][32][8],
// zero-padded cols 100..127 and rows past M) for the dW1 wgrad.
at::Tensor swizzle_xt_bf16(const at::Tensor& x, bool pi16) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 &&
                  x.dim() == 2 && x.size(1) == 100 && x.is_contiguous(),
              "swizzle_xt: x must be contiguous bf16 [M,100]");
  const int64_t M = x.size(0);
  const int64_t mtiles = std::max<int64_t>((M + 31) / 32, 1);
  auto out = at::empty({4 * mtiles * 2 * 512}, x.options());
  if (M > 0) {
    launch_swizzle_xt(x.data_ptr(), out.data_ptr(), M, pi16 ? 1 : 0,
                      current_stream());
  }
  return out;
}

// Fragment-major wgrad (csrc/wgrad_frag.hip): dW = dz^T @ src with both
// inputs pre-swizzled to [C/32][Mp/16][2][32][8] fragment layout.
at::Tensor wgrad_frag_bf16(const at::Tensor& AT, const at::Tensor& BT,
                           int64_t N, int64_t K, int64_t mchunks,
                           int64_t nt_w, int64_t kt_w) {
  TORCH_CHECK(AT.is_cuda() && AT.scalar_type() == at::kBFloat16 &&
                  AT.is_contiguous() && BT.is_cuda() &&
                  BT.scalar_type() == at::kBFloat16 && BT.is_contiguous(),
              "wgrad_frag: inputs must be contiguous bf16 on GPU");
  TORCH_CHECK(AT.numel() == (N / 32) * mchunks * 512,
              "wgrad_frag: AT numel mismatch");
  TORCH_CHECK(BT.numel() == (K / 32) * mchunks * 512,
              "wgrad_frag: BT numel mismatch");
  TORCH_CHECK((nt_w == 2 && kt_w == 4) || (nt_w == 1 && kt_w == 8),
              "wgrad_frag: unsupported tile config");
  TORCH_CHECK(N % (nt_w * 128) == 0 && K % (kt_w * 32) == 0,
              "wgrad_frag: N/K not divisible by block shape");
  auto dW = at::zeros({N, K}, AT.options().dtype(at::kFloat));
  if (mchunks > 0) {
    // Per-slab fp32 partials + split-slab reduce (atomic combine over
    // <= a handful of writers per line).
    const int64_t nslabs = wgrad_frag_nslabs(mchunks, (int32_t)N,
                                             (int32_t)K, (int32_t)nt_w,
                                             (int32_t)kt_w);
    auto part =
        at::empty({nslabs, N * K}, AT.options().dtype(at::kFloat));
    launch_wgrad_frag(AT.data_ptr(), BT.data_ptr(), part.data_ptr<float>(),
                      (int32_t)N, (int32_t)K, mchunks, (int32_t)nt_w,
                      (int32_t)kt_w, current_stream());
    launch_slab_reduce(part.data_ptr<float>(), dW.data_ptr<float>(),
                       N * K, nslabs, current_stream());
  }
  return dW;
}

}  // namespace rsdl

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X (gfx950) fused shuffle kernels";
  m.def("gather_rows", &rsdl::gather_rows, "row gather of packed rows",
        py::arg("src"), py::arg("perm"));
  m.def("gather_rows_out", &rsdl::gather_rows_out, py::arg("src"),
        py::arg("perm"), py::arg("dst"));
  m.def("unpack_permute", &rsdl::unpack_permute, py::arg("packed"),
        py::arg("perm"), py::arg("outs"), py::arg("packed_offsets"),
        py::arg("src_dtype_codes"));
  m.def("pack_columns", &rsdl::pack_columns, py::arg("cols"),
        py::arg("packed_offsets"), py::arg("packed_dtype_codes"),
        py::arg("row_stride"), py::arg("perm") = c10::nullopt);
  m.def("pack_columns_tiled", &rsdl::pack_columns_tiled, py::arg("cols"),
        py::arg("packed_offsets"), py::arg("packed_dtype_codes"),
        py::arg("row_stride"), py::arg("out") = c10::nullopt);
  m.def("partition_build_perm", &rsdl::partition_build_perm, py::arg("dest"),
        py::arg("num_dests"));
  m.def("wgrad_bf16", &rsdl::wgrad_bf16, py::arg("dy"), py::arg("x"),
        py::arg("with_bias") = true);
  m.def("relu_bwd_bias", &rsdl::relu_bwd_bias, py::arg("dy"), py::arg("y"));
  m.def("swizzle_xt_bf16", &rsdl::swizzle_xt_bf16, py::arg("x"),
        py::arg("pi16") = false);
  m.def("wgrad_frag_bf16", &rsdl::wgrad_frag_bf16, py::arg("AT"),
        py::arg("BT"), py::arg("N"), py::arg("K"), py::arg("mchunks"),
        py::arg("nt_w"), py::arg("kt_w"));
  m.def("fwd_chain_bf16", &rsdl::fwd_chain_bf16, py::arg("x"),
        py::arg("W1"), py::arg("b1"), py::arg("W2"), py::arg("b2"),
        py::arg("W3"), py::arg("b3"), py::arg("w4"), py::arg("b4"),
        py::arg("target") = c10::nullopt,
        py::arg("xt_out") = c10::nullopt, py::arg("pi16") = false);
  m.def("bwd_chain_bf16", &rsdl::bwd_chain_bf16, py::arg("dy"),
        py::arg("a3"), py::arg("mask1"), py::arg("mask2"), py::arg("w4"),
        py::arg("W3"), py::arg("W2"), py::arg("pi16") = false);
  m.attr("DT_F32") = (int)rsdl::DT_F32;
  m.attr("DT_F64") = (int)rsdl::DT_F64;
  m.attr("DT_I32") = (int)rsdl::DT_I32;
  m.attr("DT_I64") = (int)rsdl::DT_I64;
  m.attr("DT_F16") = (int)rsdl::DT_F16;
  m.attr("DT_BF16") = (int)rsdl::DT_BF16;
  m.attr("DT_U8") = (int)rsdl::DT_U8;
}
