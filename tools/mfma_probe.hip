// MFMA fragment-layout probe for v_mfma_f32_32x32x16_bf16 on gfx950.
//
// Decodes the lane->element mapping of the A and B fragments empirically:
//   probe A: B := "identity" built via the recovered-by-construction trick
//            is unnecessary — instead we set each lane's A registers to
//            known codes and multiply by a B that selects columns, reading
//            the product through the documented C/D layout
//            (col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)).
//
// Concretely:
//   Run 1: every lane sets its 8 A-register bf16 slots to code = lane,
//          and B slots to (k==n selector built the same empirical way is
//          circular), so instead B is loaded from MEMORY as the identity
//          using the mapping under test in reverse... To stay assumption-
//          free we probe with MEMORY-free one-hot accumulation:
//          For each (lane_sel, reg_sel): A regs = 1 iff (lane==lane_sel &&
//          reg==reg_sel) else 0; B regs = all 1. Then D[i][j] = sum_k A[i][k]
//          = 1 exactly for the row i that (lane_sel, reg_sel) maps to (all
//          j). That reveals A's (lane,reg)->i row map but not k. For k, the
//          second pass sets B regs = 1 iff reg==r2 && lane==l2, giving
//          D[i][j] = A[i][k(l2,r2)] * [j == j(l2,r2)] — nonzero only if
//          the probed A slot coincides in k with B's slot k.
//
// Simpler and sufficient: exhaustively one-hot A-slot (64 lanes x 8 regs is
// too many launches) — so we use code-valued probes with EXACT bf16 values:
//   A slot value = 2^(reg)  (1..128, exact in bf16), B slot value =
//   3^0..  no — products must be decodable...
//
// Final scheme actually implemented (2 launches, exact decode):
//   Launch A-probe: A[lane][reg] = (reg == r) ? 1 : 0 for r = 0..7 in 8
//   separate MFMAs accumulating into 8 separate D buffers; B[lane][reg] =
//   lane_code where lane_code = lane+1 (<= 64, exact). Then
//   D_r[i][j] = sum over the k-slot A(.,r) covers of B[k][j] — since A's
//   one-hot row has a single 1 at (i(lane), k(lane,r)), each output row i
//   equals B[k(lane,r)][j] = (lane_B(k,j)+1). Reading D pins, for every
//   (lane, r): which row i it contributes and which B lane covers its k.
//   Launch B-probe symmetric. Host prints the maps.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16;

__device__ inline short f2b(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// For r in 0..7: A one-hot at reg r (value 1), B[lane][reg] = lane+1.
// D_r stored [8][32][32].
__global__ void probeA(float* out) {
  int lane = threadIdx.x;  // 64 threads, one wave
  bf16x8 b;
  for (int j = 0; j < 8; j++) b[j] = f2b((float)(lane + 1));
  for (int r = 0; r < 8; r++) {
    bf16x8 a = {};
    a[r] = f2b(1.0f);
    f32x16 d = {};
    d = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, d, 0, 0, 0);
    // store via documented C/D layout: col=lane&31,
    // row=(reg&3)+8*(reg>>2)+4*(lane>>5)
    for (int reg = 0; reg < 16; reg++) {
      int col = lane & 31;
      int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      out[(r * 32 + row) * 32 + col] = d[reg];
    }
  }
}

// Symmetric: B one-hot at reg r (value 1), A[lane][reg] = lane+1.
__global__ void probeB(float* out) {
  int lane = threadIdx.x;
  bf16x8 a;
  for (int j = 0; j < 8; j++) a[j] = f2b((float)(lane + 1));
  for (int r = 0; r < 8; r++) {
    bf16x8 b = {};
    b[r] = f2b(1.0f);
    f32x16 d = {};
    d = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, d, 0, 0, 0);
    for (int reg = 0; reg < 16; reg++) {
      int col = lane & 31;
      int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
      out[(r * 32 + row) * 32 + col] = d[reg];
    }
  }
}

// Full random check once maps are known: C = A*B vs CPU.
__global__ void gemm_tile(const short* A, const short* B, float* C,
                          int ai_stride, int bk_stride) {
  int lane = threadIdx.x;
  // Presumed maps (verified by probes): A[i = lane&31][k = (lane>>5)*8 + j]
  // B[k = (lane>>5)*8 + j][n = lane&31]
  bf16x8 a, b;
  for (int j = 0; j < 8; j++) {
    int i = lane & 31, k = (lane >> 5) * 8 + j;
    a[j] = A[i * ai_stride + k];
    b[j] = B[k * bk_stride + (lane & 31)];
  }
  f32x16 d = {};
  d = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, d, 0, 0, 0);
  for (int reg = 0; reg < 16; reg++) {
    int col = lane & 31;
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (lane >> 5);
    C[row * 32 + col] = d[reg];
  }
}

int main() {
  float* dout;
  hipMalloc(&dout, 8 * 32 * 32 * sizeof(float));
  float host[8 * 32 * 32];

  hipLaunchKernelGGL(probeA, dim3(1), dim3(64), 0, 0, dout);
  hipMemcpy(host, dout, sizeof(host), hipMemcpyDeviceToHost);
  printf("== A probe: for A-reg r, D[i][j] = lane_B_code covering k ==\n");
  for (int r = 0; r < 8; r++) {
    // Find rows with nonzero entries; print (i, value at j=0).
    for (int i = 0; i < 32; i++) {
      float v = host[(r * 32 + i) * 32 + 0];
      if (v != 0.0f) {
        // all j should have same value pattern; print rows containing
        // contributions: row i gets hit by A-lanes with i(lane)=i.
        printf("r=%d i=%2d Bcode[j0..3]= %3.0f %3.0f %3.0f %3.0f\n", r, i,
               host[(r * 32 + i) * 32 + 0], host[(r * 32 + i) * 32 + 1],
               host[(r * 32 + i) * 32 + 2], host[(r * 32 + i) * 32 + 31]);
        break;  // one row example per r is enough to identify pattern
      }
    }
  }

  hipLaunchKernelGGL(probeB, dim3(1), dim3(64), 0, 0, dout);
  hipMemcpy(host, dout, sizeof(host), hipMemcpyDeviceToHost);
  printf("== B probe: for B-reg r, D[i][j] = A lane code covering k ==\n");
  for (int r = 0; r < 8; r++) {
    for (int j = 0; j < 32; j++) {
      float v = host[(r * 32 + 0) * 32 + j];
      if (v != 0.0f) {
        printf("r=%d j=%2d Acode[i0..3]= %3.0f %3.0f %3.0f %3.0f\n", r, j,
               host[(r * 32 + 0) * 32 + j], host[(r * 32 + 1) * 32 + j],
               host[(r * 32 + 2) * 32 + j], host[(r * 32 + 31) * 32 + j]);
        break;
      }
    }
  }

  // Random full-tile check with the presumed maps.
  short hA[32 * 16], hB[16 * 32];
  float ref[32 * 32] = {};
  auto b2f = [](short s) {
    __hip_bfloat16 h = *reinterpret_cast<__hip_bfloat16*>(&s);
    return __bfloat162float(h);
  };
  srand(7);
  for (int i = 0; i < 32 * 16; i++) {
    float v = (float)((rand() % 17) - 8);
    __hip_bfloat16 h = __float2bfloat16(v);
    hA[i] = *reinterpret_cast<short*>(&h);
  }
  for (int i = 0; i < 16 * 32; i++) {
    float v = (float)((rand() % 13) - 6);
    __hip_bfloat16 h = __float2bfloat16(v);
    hB[i] = *reinterpret_cast<short*>(&h);
  }
  for (int i = 0; i < 32; i++)
    for (int j = 0; j < 32; j++)
      for (int k = 0; k < 16; k++)
        ref[i * 32 + j] += b2f(hA[i * 16 + k]) * b2f(hB[k * 32 + j]);
  short *dA, *dB;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(gemm_tile, dim3(1), dim3(64), 0, 0, dA, dB, dout, 16,
                     32);
  hipMemcpy(host, dout, 32 * 32 * sizeof(float), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int i = 0; i < 32 * 32 && bad < 5; i++)
    if (host[i] != ref[i]) {
      printf("MISMATCH at %d: got %f want %f\n", i, host[i], ref[i]);
      bad++;
    }
  printf(bad ? "GEMM TILE CHECK: FAIL\n" : "GEMM TILE CHECK: PASS\n");
  return bad ? 1 : 0;
}
