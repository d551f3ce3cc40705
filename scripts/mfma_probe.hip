// Empirical MFMA fragment-layout probe for gfx950 mfma_f32_16x16x32_bf16.
//
// For each (lane, reg) slot of the A fragment, set ONLY that slot to 1 (rest
// 0), B[k][n] = k*1000 + n (asymmetric), run the MFMA, and read C through the
// documented C/D layout (col = lane&15, row = (lane>>4)*4 + reg; HW-verified
// per the CDNA4 guide §3). The nonzero C row r, value k*1000+n tells us slot
// (lane, reg) of A holds element (r, k). Same trick for B with A asymmetric.
//
// Build: hipcc --offload-arch=gfx950 -O3 scripts/mfma_probe.hip -o /tmp/mfma_probe
// Output: amap[lane][reg] = r*64 + k ; bmap[lane][reg] = k*64 + n

#include <hip/hip_runtime.h>
#include <cstdio>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ short f2bf(float f) {
  union { float f; unsigned int i; } v;
  v.f = f;
  unsigned int lsb = (v.i >> 16) & 1;
  return (short)((v.i + 0x7FFFu + lsb) >> 16);
}

__device__ __forceinline__ float bf2f(short u) {
  union { float f; unsigned int i; } v;
  v.i = ((unsigned int)(unsigned short)u) << 16;
  return v.f;
}

// probe A: for each slot (l0, r0), A-frag has 1 only in lane l0 reg r0.
// B-frag filled from hypothesis-free "slot id" values: B slot (lane, reg) = lane*8+reg+1.
// Then C[r][n] = sum_k A[r][k] * B[k][n]; with A one-hot = value of B at (k, n)...
// but we don't know B's (k,n) mapping either. Instead: run TWO probes.
// Probe 1 (A-map): B slots hold slotid -> C row r gets, for each n, the B slot
//   value that multiplies A's 1. Record C[r0-row][all 16 cols] per (l0,r0):
//   the 16 nonzero values are the B slot ids at (k0, n=0..15). This
//   simultaneously reveals: A slot (l0,r0) -> row r (which C row is nonzero),
//   and B slots holding k0 for every n.
__global__ void probe_a(float* out /* [64*8][16] : per A-slot, C values for its row */,
                        int* rows /* [64*8] : which C row was nonzero */) {
  const int lane = threadIdx.x;
  for (int slot = 0; slot < 64 * 8; ++slot) {
    const int l0 = slot / 8, r0 = slot % 8;
    bf16x8 a = {0, 0, 0, 0, 0, 0, 0, 0};
    if (lane == l0) a[r0] = f2bf(1.0f);
    bf16x8 b;
#pragma unroll
    for (int r = 0; r < 8; ++r) b[r] = f2bf((float)(lane * 8 + r + 1));
    f32x4 c = {0.f, 0.f, 0.f, 0.f};
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
    // C layout: col = lane&15, row = (lane>>4)*4 + reg
    __shared__ float cfull[16][16];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) cfull[(lane >> 4) * 4 + reg][lane & 15] = c[reg];
    __syncthreads();
    if (lane == 0) {
      int nz_row = -1;
      for (int r = 0; r < 16 && nz_row < 0; ++r)
        for (int n = 0; n < 16; ++n)
          if (cfull[r][n] != 0.f) { nz_row = r; break; }
      rows[slot] = nz_row;
      for (int n = 0; n < 16; ++n) out[slot * 16 + n] = nz_row >= 0 ? cfull[nz_row][n] : -1.f;
    }
    __syncthreads();
  }
}

int main() {
  float* d_out;
  int* d_rows;
  hipMalloc(&d_out, 512 * 16 * sizeof(float));
  hipMalloc(&d_rows, 512 * sizeof(int));
  hipLaunchKernelGGL(probe_a, dim3(1), dim3(64), 0, 0, d_out, d_rows);
  hipDeviceSynchronize();
  float* out = new float[512 * 16];
  int* rows = new int[512];
  hipMemcpy(out, d_out, 512 * 16 * sizeof(float), hipMemcpyDeviceToHost);
  hipMemcpy(rows, d_rows, 512 * sizeof(int), hipMemcpyDeviceToHost);

  // decode: A slot (lane, reg) -> row = rows[slot]; k is implied by which B
  // slots appear: B slot id s = b_lane*8 + b_reg + 1 at value out[slot][n].
  // For each A slot print: row, and the (b_lane, b_reg) that maps to (k, n=0)
  printf("A-frag map: lane reg -> C_row ; B slot feeding n=0 (b_lane,b_reg)\n");
  for (int lane = 0; lane < 64; ++lane) {
    for (int reg = 0; reg < 8; ++reg) {
      const int slot = lane * 8 + reg;
      const int s0 = (int)out[slot * 16 + 0] - 1;  // B slot id at col 0
      printf("A[%2d][%d] -> row %2d ; k-partner B slot lane=%2d reg=%d\n",
             lane, reg, rows[slot], s0 >= 0 ? s0 / 8 : -1, s0 >= 0 ? s0 % 8 : -1);
    }
    if (lane >= 2 && lane <= 60) { lane = 60; printf("  ... (middle lanes elided)\n"); }
  }
  // full dump for offline analysis
  FILE* f = fopen("gpurun_out/mfma_amap.txt", "w");
  if (f) {
    for (int slot = 0; slot < 512; ++slot) {
      fprintf(f, "%d %d %d", slot / 8, slot % 8, rows[slot]);
      for (int n = 0; n < 16; ++n) fprintf(f, " %d", (int)out[slot * 16 + n]);
      fprintf(f, "\n");
    }
    fclose(f);
  }
  printf("done\n");
  return 0;
}
