// pack_kernels.hip — hand-written CDNA4 (gfx950) copy/pack kernels for the
// staging path (staging.h).  These move bytes between HBM and pinned host
// memory as an alternative to the SDMA copy engines (BNET_STAGE_KERNEL=1),
// and back the torch-side `baguanet.ops` staging benchmarks.
//
// Design for MI355X: 64-wide wavefronts, 16 B/lane vectorized
// loads/stores (uint4 → 1 KiB per wave per instruction), grid-stride so a
// single launch fills all 256 CUs when the payload is large, plain
// byte loop only for the unaligned head/tail.  No LDS round-trip: a pure
// copy gains nothing from LDS staging (guide §5 "GEMV / streamed once"
// row) — global→register→global is the roofline path.

#include <hip/hip_runtime.h>

namespace baguanet {

__global__ void copy_bytes_kernel(char* __restrict__ dst,
                                  const char* __restrict__ src,
                                  size_t nbytes) {
  size_t tid = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t nthreads = (size_t)gridDim.x * blockDim.x;

  // Split: [head | body16 | tail], body aligned to 16 B relative to dst.
  uintptr_t daddr = (uintptr_t)dst;
  size_t head = (16 - (daddr & 15)) & 15;
  head = head > nbytes ? nbytes : head;
  size_t body = (nbytes - head) & ~size_t(15);
  size_t tail = nbytes - head - body;

  // Head / tail: first wave handles them (tiny).
  if (tid < head) dst[tid] = src[tid];
  if (tid < tail) {
    size_t off = head + body + tid;
    dst[off] = src[off];
  }

  // Body: 16 B per lane, vectorized when src is co-aligned, else 4×u32.
  char* dbody = dst + head;
  const char* sbody = src + head;
  size_t nvec = body >> 4;
  if ((((uintptr_t)sbody) & 15) == 0) {
    uint4* d4 = (uint4*)dbody;
    const uint4* s4 = (const uint4*)sbody;
    for (size_t i = tid; i < nvec; i += nthreads) d4[i] = s4[i];
  } else if ((((uintptr_t)sbody) & 3) == 0) {
    uint* d1 = (uint*)dbody;
    const uint* s1 = (const uint*)sbody;
    size_t n1 = body >> 2;
    for (size_t i = tid; i < n1; i += nthreads) d1[i] = s1[i];
  } else {
    for (size_t i = tid; i < body; i += nthreads) dbody[i] = sbody[i];
  }
}

void launch_copy_kernel(void* dst, const void* src, size_t bytes,
                        hipStream_t stream) {
  if (bytes == 0) return;
  const int block = 256;
  // ≥ 2048 workgroups fills 256 CUs × 8 waves; small copies use fewer.
  size_t want = (bytes + (block * 16) - 1) / (block * 16);
  int grid = (int)(want < 1 ? 1 : want > 4096 ? 4096 : want);
  hipLaunchKernelGGL(copy_bytes_kernel, dim3(grid), dim3(block), 0, stream,
                     (char*)dst, (const char*)src, bytes);
}

}  // namespace baguanet
