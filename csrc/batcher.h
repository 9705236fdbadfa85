// See batcher.cc. API surface mirrors the reference Batcher
// (src/moolib.cc:1887-1929): stack/cat/empty/size/get/__await__.
#pragma once

#include <torch/extension.h>

#include <optional>

#include "pybits.h"

namespace mrl {

// Register fn(list[Tensor] dsts, list[Tensor] srcs) as the fused
// multi-leaf copy (moolib_amd._kernels.batched_copy); None unregisters.
void setBatcherFusedCopy(py::object fn);

class Batcher {
 public:
  Batcher(int64_t size, py::object device, int64_t dim);

  void stack(py::object nest);
  void cat(py::object nest);
  bool empty();
  size_t size();
  py::object get();       // blocks (GIL released while waiting)
  PyFuture popFuture();   // for __await__

 private:
  at::Tensor makeTarget(const at::Tensor& src, bool insertDim);
  void completeBatch();

  int64_t size_;
  int64_t dim_;
  std::optional<at::Device> device_;
  py::object current_;  // partially-filled target nest (GIL)
  int64_t fill_ = 0;
  PyQueue ready_;
};

}  // namespace mrl
