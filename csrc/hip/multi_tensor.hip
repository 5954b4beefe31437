// multi_tensor.hip — fused multi-tensor pack/unpack kernels (gfx950).
//
// Packs N gradient tensors into one flat fusion buffer (and scatters back)
// in a single launch — the bucket-fusion hot path of BucketedDDP when
// gradient views are not applicable, and the stripe gather/scatter
// primitive of the staging path.  MI355X design: 64-wide wavefronts,
// uint4 (16 B/lane) vectorized moves, one (tensor, tile) work item per
// workgroup with enough workgroups to fill 256 CUs across 8 XCDs.

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstring>

namespace baguanet {

constexpr int kBlock = 256;
constexpr uint32_t kMaxTile = 256 * 1024;   // upper bound per work item
constexpr uint32_t kMinTile = 64 * 1024;    // lower bound per work item
constexpr int kMaxDesc = 512;               // descriptors per launch

struct PackDesc {
  const void* src;  // tensor data (pack) — or dst for unpack
  void* dst;        // flat + offset (pack) — or flat + offset for unpack
  uint32_t bytes;
};

// Work item table lives in device memory; each entry names a descriptor and
// a tile within it.
struct WorkItem {
  uint16_t desc;
  uint16_t tile;
};

__device__ inline void copy_span(char* __restrict__ dst,
                                 const char* __restrict__ src,
                                 uint32_t bytes) {
  uint32_t tid = threadIdx.x;
  // Both sides are torch tensor allocations (256-B aligned) offset by
  // element-size multiples; handle arbitrary alignment anyway.
  if ((((uintptr_t)dst | (uintptr_t)src) & 15) == 0) {
    uint32_t nvec = bytes >> 4;
    uint4* d4 = (uint4*)dst;
    const uint4* s4 = (const uint4*)src;
    for (uint32_t i = tid; i < nvec; i += kBlock) d4[i] = s4[i];
    for (uint32_t i = (nvec << 4) + tid; i < bytes; i += kBlock)
      dst[i] = src[i];
  } else if ((((uintptr_t)dst | (uintptr_t)src) & 3) == 0) {
    uint32_t n1 = bytes >> 2;
    uint* d1 = (uint*)dst;
    const uint* s1 = (const uint*)src;
    for (uint32_t i = tid; i < n1; i += kBlock) d1[i] = s1[i];
    for (uint32_t i = (n1 << 2) + tid; i < bytes; i += kBlock)
      dst[i] = src[i];
  } else {
    for (uint32_t i = tid; i < bytes; i += kBlock) dst[i] = src[i];
  }
}

__global__ void multi_copy_kernel(const PackDesc* __restrict__ descs,
                                  const WorkItem* __restrict__ items,
                                  uint32_t nitems, uint32_t tile_bytes) {
  for (uint32_t w = blockIdx.x; w < nitems; w += gridDim.x) {
    WorkItem it = items[w];
    PackDesc d = descs[it.desc];
    uint32_t off = (uint32_t)it.tile * tile_bytes;
    uint32_t n = d.bytes - off < tile_bytes ? d.bytes - off : tile_bytes;
    copy_span((char*)d.dst + off, (const char*)d.src + off, n);
  }
}

// Host-side launcher: builds the work-item table on the host; `scratch` is
// a device buffer (>= bytes_needed) the caller provides, `staging` is the
// matching host buffer.  Returns bytes needed when scratch is too small.
size_t multi_copy_launch(const PackDesc* host_descs, int ndesc, void* scratch,
                         size_t scratch_bytes, void* staging,
                         hipStream_t stream) {
  // Pick the tile so the launch has >= ~2048 work items (fills 256 CUs x
  // several blocks) without exceeding the per-item upper bound.
  size_t total = 0;
  for (int i = 0; i < ndesc; i++) total += host_descs[i].bytes;
  uint32_t tile = kMaxTile;
  while (tile > kMinTile && total / tile < 2048) tile /= 2;

  // layout: [ndesc PackDesc][nitems WorkItem]
  uint32_t nitems = 0;
  for (int i = 0; i < ndesc; i++) {
    uint32_t t = (host_descs[i].bytes + tile - 1) / tile;
    // WorkItem.tile is uint16: indices 0..65535.  With uint32 byte counts
    // and tile >= 64 KiB this cannot overflow, but guard anyway.
    if (t > 65536u) return SIZE_MAX;
    nitems += t;
  }
  size_t need = sizeof(PackDesc) * ndesc + sizeof(WorkItem) * nitems;
  if (need > scratch_bytes || ndesc > kMaxDesc) return need;

  char* h = (char*)staging;
  std::memcpy(h, host_descs, sizeof(PackDesc) * ndesc);
  WorkItem* hitems = (WorkItem*)(h + sizeof(PackDesc) * ndesc);
  uint32_t w = 0;
  for (int i = 0; i < ndesc; i++) {
    uint32_t t = (host_descs[i].bytes + tile - 1) / tile;
    for (uint32_t j = 0; j < t; j++) hitems[w++] = {(uint16_t)i, (uint16_t)j};
  }
  (void)hipMemcpyAsync(scratch, staging, need, hipMemcpyHostToDevice, stream);
  const PackDesc* ddescs = (const PackDesc*)scratch;
  const WorkItem* ditems =
      (const WorkItem*)((char*)scratch + sizeof(PackDesc) * ndesc);
  uint32_t grid = nitems < 4096 ? (nitems ? nitems : 1) : 4096;
  hipLaunchKernelGGL(multi_copy_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     ddescs, ditems, nitems, tile);
  return 0;
}

// ---------------------------------------------------------------------------
// Fused multi-tensor SGD (momentum, weight decay, nesterov) — the whole
// model's parameter update in ONE launch.  Matches torch.optim.SGD
// semantics exactly:
//   g' = g + wd * p
//   m  = mu * m + g'          (buffer pre-initialized to 0: first step m=g')
//   u  = nesterov ? g' + mu * m : m
//   p -= lr * u
// fp32; 4 elements/lane vectorized when aligned.

struct SgdDesc {
  float* p;
  const float* g;
  float* m;
  uint32_t numel;
};

__global__ void multi_sgd_kernel(const SgdDesc* __restrict__ descs,
                                 const WorkItem* __restrict__ items,
                                 uint32_t nitems, uint32_t tile_elems,
                                 float lr, float mu, float wd, int nesterov) {
  for (uint32_t w = blockIdx.x; w < nitems; w += gridDim.x) {
    WorkItem it = items[w];
    SgdDesc d = descs[it.desc];
    uint32_t start = (uint32_t)it.tile * tile_elems;
    uint32_t n = d.numel - start < tile_elems ? d.numel - start : tile_elems;
    float* p = d.p + start;
    const float* g = d.g + start;
    float* m = d.m + start;
    uint32_t tid = threadIdx.x;
    if ((((uintptr_t)p | (uintptr_t)g | (uintptr_t)m) & 15) == 0) {
      uint32_t n4 = n >> 2;
      float4* p4 = (float4*)p;
      const float4* g4 = (const float4*)g;
      float4* m4 = (float4*)m;
      for (uint32_t i = tid; i < n4; i += kBlock) {
        float4 pv = p4[i], gv = g4[i], mv = m4[i];
        float gx = gv.x + wd * pv.x, gy = gv.y + wd * pv.y,
              gz = gv.z + wd * pv.z, gw = gv.w + wd * pv.w;
        mv.x = mu * mv.x + gx;
        mv.y = mu * mv.y + gy;
        mv.z = mu * mv.z + gz;
        mv.w = mu * mv.w + gw;
        float ux = nesterov ? gx + mu * mv.x : mv.x;
        float uy = nesterov ? gy + mu * mv.y : mv.y;
        float uz = nesterov ? gz + mu * mv.z : mv.z;
        float uw = nesterov ? gw + mu * mv.w : mv.w;
        pv.x -= lr * ux;
        pv.y -= lr * uy;
        pv.z -= lr * uz;
        pv.w -= lr * uw;
        p4[i] = pv;
        m4[i] = mv;
      }
      for (uint32_t i = (n4 << 2) + tid; i < n; i += kBlock) {
        float gg = g[i] + wd * p[i];
        m[i] = mu * m[i] + gg;
        p[i] -= lr * (nesterov ? gg + mu * m[i] : m[i]);
      }
    } else {
      for (uint32_t i = tid; i < n; i += kBlock) {
        float gg = g[i] + wd * p[i];
        m[i] = mu * m[i] + gg;
        p[i] -= lr * (nesterov ? gg + mu * m[i] : m[i]);
      }
    }
  }
}

size_t multi_sgd_launch(const SgdDesc* host_descs, int ndesc, void* scratch,
                        size_t scratch_bytes, void* staging, float lr,
                        float mu, float wd, int nesterov,
                        hipStream_t stream) {
  constexpr uint32_t kTileElems = 64 * 1024;  // 256 KiB of fp32 per item
  uint32_t nitems = 0;
  for (int i = 0; i < ndesc; i++) {
    uint32_t t = (host_descs[i].numel + kTileElems - 1) / kTileElems;
    if (t > 65536u) return SIZE_MAX;  // uint16 tile index guard
    nitems += t;
  }
  size_t need = sizeof(SgdDesc) * ndesc + sizeof(WorkItem) * nitems;
  if (need > scratch_bytes || ndesc > kMaxDesc) return need;

  char* h = (char*)staging;
  std::memcpy(h, host_descs, sizeof(SgdDesc) * ndesc);
  WorkItem* hitems = (WorkItem*)(h + sizeof(SgdDesc) * ndesc);
  uint32_t w = 0;
  for (int i = 0; i < ndesc; i++) {
    uint32_t t = (host_descs[i].numel + kTileElems - 1) / kTileElems;
    for (uint32_t j = 0; j < t; j++) hitems[w++] = {(uint16_t)i, (uint16_t)j};
  }
  (void)hipMemcpyAsync(scratch, staging, need, hipMemcpyHostToDevice, stream);
  const SgdDesc* ddescs = (const SgdDesc*)scratch;
  const WorkItem* ditems =
      (const WorkItem*)((char*)scratch + sizeof(SgdDesc) * ndesc);
  uint32_t grid = nitems < 2048 ? (nitems ? nitems : 1) : 2048;
  hipLaunchKernelGGL(multi_sgd_kernel, dim3(grid), dim3(kBlock), 0, stream,
                     ddescs, ditems, nitems, kTileElems, lr, mu, wd,
                     nesterov);
  return 0;
}

}  // namespace baguanet
