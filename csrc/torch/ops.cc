// ops.cc — torch extension exposing the HIP staging/pack kernels
// (baguanet.ops).  Used by GPU tests (numerics vs plain torch), the staging
// microbenchmarks, and BucketedDDP's non-view fusion path.

#include <c10/cuda/CUDAStream.h>
#include <hip/hip_runtime.h>
#include <torch/extension.h>

#include <vector>

namespace baguanet {

struct PackDesc {
  const void* src;
  void* dst;
  uint32_t bytes;
};

struct SgdDesc {
  float* p;
  const float* g;
  float* m;
  uint32_t numel;
};

void launch_copy_kernel(void* dst, const void* src, size_t bytes,
                        hipStream_t stream);
size_t multi_copy_launch(const PackDesc* host_descs, int ndesc, void* scratch,
                         size_t scratch_bytes, void* staging,
                         hipStream_t stream);
size_t multi_sgd_launch(const SgdDesc* host_descs, int ndesc, void* scratch,
                        size_t scratch_bytes, void* staging, float lr,
                        float mu, float wd, int nesterov,
                        hipStream_t stream);

namespace {

hipStream_t current_stream() {
  return (hipStream_t)c10::cuda::getCurrentCUDAStream().stream();
}

// Per-launch descriptor-table buffers, in a ring: a single shared pinned
// buffer could be overwritten by the CPU while a previous launch's
// hipMemcpyAsync of it was still queued (the GPU runs behind the CPU) —
// torn descriptor tables, silent corruption.  Each slot carries an event
// recorded after its launch; a slot is reused only after that event lands.
struct Scratch {
  void* dev = nullptr;
  void* host = nullptr;
  size_t size = 0;
  hipEvent_t ev = nullptr;
  bool busy = false;
  void ensure(size_t need) {
    if (need <= size) return;
    size_t sz = std::max(need, (size_t)64 * 1024);
    if (dev) (void)hipFree(dev);
    if (host) (void)hipHostFree(host);
    TORCH_CHECK(hipMalloc(&dev, sz) == hipSuccess, "scratch hipMalloc");
    TORCH_CHECK(hipHostMalloc(&host, sz, hipHostMallocDefault) == hipSuccess,
                "scratch hipHostMalloc");
    size = sz;
  }
};
Scratch& scratch() {
  static Scratch ring[4];
  static int next = 0;
  Scratch& s = ring[next];
  next = (next + 1) % 4;
  if (s.busy) {
    TORCH_CHECK(hipEventSynchronize(s.ev) == hipSuccess, "scratch event");
    s.busy = false;
  }
  return s;
}

// Record the in-flight marker after the launch that consumed `s`.
void scratch_commit(Scratch& s, hipStream_t stream) {
  if (!s.ev)
    TORCH_CHECK(
        hipEventCreateWithFlags(&s.ev, hipEventDisableTiming) == hipSuccess,
        "scratch event create");
  TORCH_CHECK(hipEventRecord(s.ev, stream) == hipSuccess, "scratch record");
  s.busy = true;
}

void copy_bytes(torch::Tensor dst, torch::Tensor src) {
  TORCH_CHECK(dst.is_contiguous() && src.is_contiguous(),
              "contiguous tensors required");
  TORCH_CHECK(dst.nbytes() == src.nbytes(), "size mismatch");
  TORCH_CHECK(dst.is_cuda() || src.is_cuda(),
              "at least one tensor must be on the GPU");
  launch_copy_kernel(dst.data_ptr(), src.data_ptr(), dst.nbytes(),
                     current_stream());
}

void build_descs(torch::Tensor& flat, std::vector<torch::Tensor>& tensors,
                 bool pack, std::vector<PackDesc>& descs) {
  TORCH_CHECK(flat.is_cuda() && flat.is_contiguous(), "flat must be CUDA");
  int64_t off = 0;
  char* base = (char*)flat.data_ptr();
  for (auto& t : tensors) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "bucket tensors must be contiguous CUDA tensors");
    TORCH_CHECK(t.scalar_type() == flat.scalar_type(), "dtype mismatch");
    int64_t nb = t.nbytes();
    TORCH_CHECK(nb <= (int64_t)UINT32_MAX,
                "tensor too large for 32-bit descriptor (", nb, " bytes)");
    TORCH_CHECK(off + nb <= (int64_t)flat.nbytes(), "flat too small");
    if (pack)
      descs.push_back({t.data_ptr(), base + off, (uint32_t)nb});
    else
      descs.push_back({base + off, t.data_ptr(), (uint32_t)nb});
    off += nb;
  }
}

void run_multi(std::vector<PackDesc>& descs) {
  auto& s = scratch();
  size_t need = multi_copy_launch(descs.data(), (int)descs.size(), s.dev,
                                  s.size, s.host, current_stream());
  if (need) {
    s.ensure(need);
    need = multi_copy_launch(descs.data(), (int)descs.size(), s.dev, s.size,
                             s.host, current_stream());
    TORCH_CHECK(need == 0, "multi_copy_launch failed (", need, " bytes)");
  }
  scratch_commit(s, current_stream());
}

// Pack `tensors` back-to-back into `flat` (one fused kernel).
void multi_pack(torch::Tensor flat, std::vector<torch::Tensor> tensors) {
  std::vector<PackDesc> descs;
  build_descs(flat, tensors, /*pack=*/true, descs);
  run_multi(descs);
}

// Scatter `flat` back into `tensors`.
void multi_unpack(torch::Tensor flat, std::vector<torch::Tensor> tensors) {
  std::vector<PackDesc> descs;
  build_descs(flat, tensors, /*pack=*/false, descs);
  run_multi(descs);
}

// Fused SGD step over all parameters in one launch (fp32, torch.optim.SGD
// semantics).
void fused_sgd(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> momenta, double lr, double momentum,
               double weight_decay, bool nesterov) {
  TORCH_CHECK(params.size() == grads.size() &&
                  params.size() == momenta.size(),
              "list size mismatch");
  std::vector<SgdDesc> descs;
  descs.reserve(params.size());
  for (size_t i = 0; i < params.size(); i++) {
    auto &p = params[i], &g = grads[i], &m = momenta[i];
    TORCH_CHECK(p.is_cuda() && g.is_cuda() && m.is_cuda(),
                "fused_sgd requires CUDA tensors");
    TORCH_CHECK(p.scalar_type() == torch::kFloat32 &&
                    g.scalar_type() == torch::kFloat32 &&
                    m.scalar_type() == torch::kFloat32,
                "fused_sgd is fp32-only");
    TORCH_CHECK(p.is_contiguous() && g.is_contiguous() && m.is_contiguous(),
                "contiguous tensors required");
    TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel(),
                "numel mismatch");
    TORCH_CHECK(p.numel() <= (int64_t)UINT32_MAX,
                "tensor too large for 32-bit descriptor");
    descs.push_back({p.data_ptr<float>(), g.data_ptr<float>(),
                     m.data_ptr<float>(), (uint32_t)p.numel()});
  }
  auto& s = scratch();
  size_t need = multi_sgd_launch(descs.data(), (int)descs.size(), s.dev,
                                 s.size, s.host, (float)lr, (float)momentum,
                                 (float)weight_decay, nesterov ? 1 : 0,
                                 current_stream());
  if (need) {
    s.ensure(need);
    need = multi_sgd_launch(descs.data(), (int)descs.size(), s.dev, s.size,
                            s.host, (float)lr, (float)momentum,
                            (float)weight_decay, nesterov ? 1 : 0,
                            current_stream());
    TORCH_CHECK(need == 0, "multi_sgd_launch failed");
  }
  scratch_commit(s, current_stream());
}

}  // namespace
}  // namespace baguanet

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("copy_bytes", &baguanet::copy_bytes,
        "vectorized gfx950 copy kernel (dst, src)");
  m.def("multi_pack", &baguanet::multi_pack,
        "fused multi-tensor pack into flat buffer");
  m.def("multi_unpack", &baguanet::multi_unpack,
        "fused multi-tensor scatter from flat buffer");
  m.def("fused_sgd", &baguanet::fused_sgd,
        "fused multi-tensor SGD step (params, grads, momenta, lr, momentum, "
        "weight_decay, nesterov)");
}
