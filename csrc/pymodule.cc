// Python bindings for moolib_amd._core (placeholder; full bindings follow).
#include <torch/extension.h>

#include "rpc.h"
#include "services.h"

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "moolib_amd core runtime";
}
