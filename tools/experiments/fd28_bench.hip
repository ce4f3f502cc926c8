// standalone micro-benchmark: dependent-chain throughput of the current
// 4x64 CIOS fd_mul vs the carry-chain-free radix-2^28 fd28_mul on gfx950.
// Each thread runs ITER dependent multiplies; wall time / (threads*ITER)
// gives the amortized per-mul cost under full occupancy.
#include <cstdio>
#include <hip/hip_runtime.h>
#include "fd28.hpp"
using namespace taiga;

constexpr int ITER = 512;

__global__ void __launch_bounds__(256) kb_fd64(Fp* io, u64 n) {
  u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x;
  if (i >= n) return;
  Fp a = io[i], b = io[(i + 1) % n];
  for (int k = 0; k < ITER; k++) a = fd_mul(a, b);
  io[i] = a;
}

__global__ void __launch_bounds__(256) kb_fd28(Fp* io, u64 n) {
  u64 i = blockIdx.x * (u64)blockDim.x + threadIdx.x;
  if (i >= n) return;
  Fd28<FpCfg> a = fd28_from<FpCfg>(fd_from_mont(io[i]));
  Fd28<FpCfg> b = fd28_from<FpCfg>(fd_from_mont(io[(i + 1) % n]));
  for (int k = 0; k < ITER; k++) {
    a = fd28_mul(a, b);
    // renormalize digits to < 2^28 for the next round (mirrors what a
    // point formula needs between chained muls): cheap linear pass
    u64 carry = 0;
#pragma unroll
    for (int j = 0; j < FD28_ND; j++) {
      u64 v = a.d[j] + carry;
      a.d[j] = v & FD28_MASK;
      carry = v >> 28;
    }
  }
  io[i] = fd28_norm(a);
}

int main() {
  const u64 n = 1 << 20;
  Fp* d;
  hipMalloc(&d, n * sizeof(Fp));
  hipMemset(d, 0x35, n * sizeof(Fp));
  // clear top bits for canonical-ish values
  dim3 grid((unsigned)((n + 255) / 256)), block(256);
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  for (int variant = 0; variant < 2; variant++) {
    // warmup
    if (variant == 0) hipLaunchKernelGGL(kb_fd64, grid, block, 0, 0, d, n);
    else hipLaunchKernelGGL(kb_fd28, grid, block, 0, 0, d, n);
    hipDeviceSynchronize();
    hipEventRecord(e0);
    for (int r = 0; r < 3; r++) {
      if (variant == 0) hipLaunchKernelGGL(kb_fd64, grid, block, 0, 0, d, n);
      else hipLaunchKernelGGL(kb_fd28, grid, block, 0, 0, d, n);
    }
    hipEventRecord(e1);
    hipEventSynchronize(e1);
    float ms;
    hipEventElapsedTime(&ms, e0, e1);
    double per_mul_ns = (double)ms * 1e6 / (3.0 * n * ITER);
    printf("%s: %.3f ms for 3x%llux%d muls = %.3f ns/mul/thread\n",
           variant == 0 ? "fd64 (current)" : "fd28 (lazy-carry)",
           ms, (unsigned long long)n, ITER, per_mul_ns);
  }
  return 0;
}
