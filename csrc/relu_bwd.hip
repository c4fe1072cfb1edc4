// Fused ReLU-backward + bias-gradient kernel for MI355X (gfx950).
//
// Computes, in ONE pass over the [M, N] bf16 activation gradient:
//   dx[m, n] = y[m, n] > 0 ? dy[m, n] : 0        (aten::threshold_backward)
//   db[n]    = sum_m dx[m, n]                     (bias gradient)
// The unfused sequence (threshold_backward kernel + dx.sum(0) reduce)
// re-reads the whole dx tensor from HBM for the column sum; at the bench
// shapes (250k x 512/256/128 bf16) that second pass plus the multi-stage
// reduce kernels is ~0.1 ms/step of pure bandwidth this kernel removes
// (profiles/PERF.md). Replaces the reference's per-batch CPU autograd path
// (it never trains for real — its example mocks the step with a sleep).
//
// Layout: row-major [M, N], N a power of two >= 8 (the LinearReLU widths
// 512/256/128; host falls back to torch otherwise). Each thread walks
// vec8 (16 B) slots with a grid-wide stride; because N divides the element
// stride (grid*256*8), a thread touches the SAME 8 columns every
// iteration, so the bias partials live in 8 VGPRs (no atomics in the
// loop). Per-element masking is v_cmp/v_cndmask on loaded vectors — no
// guarded loads (cdna_hip_programming.md trap 4c). Per-WG column partials
// are LDS-reduced then written to db_part[grid][N]; the tiny final
// sum-over-WGs runs as one at::sum on the same stream.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

namespace rsdl {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short rb_bf16x8;

#define RB_THREADS 256

__global__ void __launch_bounds__(RB_THREADS) relu_bwd_bias_kernel(
    const short* __restrict__ dy, const short* __restrict__ y,
    short* __restrict__ dx, float* __restrict__ db_part, int64_t nvec,
    int32_t N) {
  const int32_t tid = threadIdx.x;
  const int64_t start = (int64_t)blockIdx.x * RB_THREADS + tid;
  const int64_t step = (int64_t)gridDim.x * RB_THREADS;

  float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
  for (int64_t v = start; v < nvec; v += step) {
    const rb_bf16x8 d8 = *reinterpret_cast<const rb_bf16x8*>(&dy[v * 8]);
    const rb_bf16x8 y8 = *reinterpret_cast<const rb_bf16x8*>(&y[v * 8]);
    rb_bf16x8 o8;
    #pragma unroll
    for (int j = 0; j < 8; j++) {
      __hip_bfloat16 yh, dh;
      *reinterpret_cast<short*>(&yh) = y8[j];
      *reinterpret_cast<short*>(&dh) = d8[j];
      const bool live = __bfloat162float(yh) > 0.f;
      o8[j] = live ? d8[j] : (short)0;
      acc[j] += live ? __bfloat162float(dh) : 0.f;
    }
    *reinterpret_cast<rb_bf16x8*>(&dx[v * 8]) = o8;
  }

  // Reduce the per-thread column partials across the WG. Thread t's 8
  // columns start at (t*8) % N, so threads t, t + N/8, t + 2N/8, ... share
  // columns (N | grid*256*8 guaranteed by the host gate).
  __shared__ float lds[RB_THREADS * 8];
  #pragma unroll
  for (int j = 0; j < 8; j++) lds[tid * 8 + j] = acc[j];
  __syncthreads();
  const int32_t groups = N / 8;  // <= 256 and divides RB_THREADS
  if (tid < groups) {
    float s[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int32_t r = tid; r < RB_THREADS; r += groups) {
      #pragma unroll
      for (int j = 0; j < 8; j++) s[j] += lds[r * 8 + j];
    }
    float* out = &db_part[(int64_t)blockIdx.x * N + tid * 8];
    #pragma unroll
    for (int j = 0; j < 8; j++) out[j] = s[j];
  }
}

void launch_relu_bwd_bias(const void* dy, const void* y, void* dx,
                          float* db_part, int64_t nvec, int32_t N,
                          int32_t grid, hipStream_t stream) {
  hipLaunchKernelGGL(relu_bwd_bias_kernel, dim3(grid), dim3(RB_THREADS), 0,
                     stream, reinterpret_cast<const short*>(dy),
                     reinterpret_cast<const short*>(y),
                     reinterpret_cast<short*>(dx), db_part, nvec, N);
}

}  // namespace rsdl
