// Python object <-> wire payload serialization with out-of-band tensors.
//
// Capability parity with the reference's pythonserialization.h: a tagged
// format for the common types (None/bool/int/float/str/bytes/tuple/list/
// dict/tensor/ndarray) with a pickle fallback for everything else. Tensors
// (and numpy arrays) are registered out-of-band so their storage rides the
// wire zero-copy as separate iovecs instead of being copied into the
// payload stream.
//
// All functions require the GIL.
#pragma once

#include <torch/csrc/utils/pybind.h>

#include "common.h"

namespace mrl {

namespace py = pybind11;

// ipcLocal: destination peer is on this machine — CUDA tensors serialize
// as hipIpc handles (moolib_amd.ipc.share) instead of being staged through
// the CPU; the receiver materializes a tensor aliasing the sender's HBM
// (zero copies). False (default) keeps the CPU-staging wire path.
void serializePy(py::handle obj, WireWriter& w, std::vector<at::Tensor>& tensors,
                 bool ipcLocal = false);
py::object deserializePy(WireReader& r, const std::vector<at::Tensor>& tensors);

// Convenience: serialize (args, kwargs) into payload+tensors and back.
std::string serializeCall(py::tuple args, py::dict kwargs, std::vector<at::Tensor>& tensors,
                          bool ipcLocal = false);
std::pair<py::tuple, py::dict> deserializeCall(std::string_view payload,
                                               const std::vector<at::Tensor>& tensors);

std::string serializeObject(py::handle obj, std::vector<at::Tensor>& tensors,
                            bool ipcLocal = false);
py::object deserializeObject(std::string_view payload, const std::vector<at::Tensor>& tensors);

}  // namespace mrl
