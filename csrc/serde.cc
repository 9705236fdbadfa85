#include "serde.h"

#include <torch/csrc/autograd/python_variable.h>

namespace mrl {

namespace {

enum Tag : uint8_t {
  tNone = 0,
  tTrue = 1,
  tFalse = 2,
  tInt64 = 3,
  tFloat64 = 4,
  tStr = 5,
  tBytes = 6,
  tTuple = 7,
  tList = 8,
  tDict = 9,
  tTensor = 10,   // u32 index into out-of-band tensor list
  tNdarray = 11,  // u32 index (tensor view of the array's memory)
  tPickle = 12,   // pickled bytes (fallback)
  tDevTensor = 13,  // u32 index + u8 deviceIndex: tensor that lived on an accelerator
};

// These statics are intentionally leaked (raw new, never destroyed): a
// py::object destructor running after Py_Finalize aborts the process.
py::object& pickleDumps() {
  static py::object* f = new py::object(py::module_::import("pickle").attr("dumps"));
  return *f;
}
py::object& pickleLoads() {
  static py::object* f = new py::object(py::module_::import("pickle").attr("loads"));
  return *f;
}

bool isNumpyArray(py::handle h) {
  static py::object* ndarrayType = []() -> py::object* {
    try {
      return new py::object(py::module_::import("numpy").attr("ndarray"));
    } catch (...) {
      return new py::object();
    }
  }();
  return ndarrayType->ptr() && py::isinstance(h, *ndarrayType);
}

// moolib_amd.ipc.share — wraps a CUDA tensor so pickling it produces a
// hipIpc handle (torch multiprocessing reductions). Resolved lazily and
// cached; empty if the module is unavailable or MOOLIB_AMD_NO_IPC_RPC is
// set (the staging path then applies).
py::object& ipcShare() {
  static py::object* f = []() -> py::object* {
    if (std::getenv("MOOLIB_AMD_NO_IPC_RPC")) return new py::object();
    try {
      return new py::object(py::module_::import("moolib_amd.ipc").attr("share"));
    } catch (...) {
      PyErr_Clear();
      return new py::object();
    }
  }();
  return *f;
}

// moolib_amd.shm.share_if_memfd — returns a segment-identity wrapper for
// CPU tensors living in memfd segments (the fd+offset cross-process
// buffer identity of the reference's memfd allocator), None for ordinary
// tensors. Same kill switch as the hipIpc path.
py::object& memfdShare() {
  static py::object* f = []() -> py::object* {
    if (std::getenv("MOOLIB_AMD_NO_IPC_RPC")) return new py::object();
    try {
      return new py::object(py::module_::import("moolib_amd.shm").attr("share_if_memfd"));
    } catch (...) {
      PyErr_Clear();
      return new py::object();
    }
  }();
  return *f;
}

}  // namespace

void serializePy(py::handle obj, WireWriter& w, std::vector<at::Tensor>& tensors,
                 bool ipcLocal) {
  if (obj.is_none()) {
    w.u8(tNone);
  } else if (py::isinstance<py::bool_>(obj)) {
    w.u8(obj.cast<bool>() ? tTrue : tFalse);
  } else if (py::isinstance<py::int_>(obj)) {
    int overflow = 0;
    long long v = PyLong_AsLongLongAndOverflow(obj.ptr(), &overflow);
    if (overflow == 0 && !(v == -1 && PyErr_Occurred())) {
      w.u8(tInt64);
      w.i64(v);
    } else {
      PyErr_Clear();
      w.u8(tPickle);
      py::bytes b = pickleDumps()(py::reinterpret_borrow<py::object>(obj), 2);
      w.str(std::string_view(PyBytes_AS_STRING(b.ptr()), PyBytes_GET_SIZE(b.ptr())));
    }
  } else if (py::isinstance<py::float_>(obj)) {
    w.u8(tFloat64);
    w.f64(obj.cast<double>());
  } else if (py::isinstance<py::str>(obj)) {
    w.u8(tStr);
    Py_ssize_t len = 0;
    const char* s = PyUnicode_AsUTF8AndSize(obj.ptr(), &len);
    if (!s) throw py::error_already_set();
    w.str(std::string_view(s, len));
  } else if (py::isinstance<py::bytes>(obj)) {
    w.u8(tBytes);
    w.str(std::string_view(PyBytes_AS_STRING(obj.ptr()), PyBytes_GET_SIZE(obj.ptr())));
  } else if (THPVariable_Check(obj.ptr())) {
    at::Tensor t = THPVariable_Unpack(obj.ptr());
    if (!t.device().is_cpu() && ipcLocal && ipcShare()) {
      // Same-machine destination: ship a hipIpc handle, not the bytes —
      // the receiver materializes a tensor aliasing this process's HBM.
      // (The reference fatals on CUDA tensors over the wire,
      // src/rpc.cc:661-667; our cross-node path stages through the CPU.)
      try {
        py::object shared =
            ipcShare()(py::reinterpret_steal<py::object>(THPVariable_Wrap(t.detach())));
        w.u8(tPickle);
        py::bytes b = pickleDumps()(shared, 2);
        w.str(std::string_view(PyBytes_AS_STRING(b.ptr()), PyBytes_GET_SIZE(b.ptr())));
        return;
      } catch (...) {
        PyErr_Clear();  // e.g. unshareable storage — fall back to staging
      }
    }
    if (t.device().is_cpu() && ipcLocal && memfdShare()) {
      // memfd-backed tensor to a same-machine peer: ship the segment
      // identity (pid, fd, offset); the receiver re-opens /proc/<pid>/fd
      // and maps the same pages (reference memfd.cc capability).
      try {
        py::object shared =
            memfdShare()(py::reinterpret_borrow<py::object>(obj));
        if (!shared.is_none()) {
          w.u8(tPickle);
          py::bytes b = pickleDumps()(shared, 2);
          w.str(std::string_view(PyBytes_AS_STRING(b.ptr()), PyBytes_GET_SIZE(b.ptr())));
          return;
        }
      } catch (...) {
        PyErr_Clear();  // fall through to byte transfer
      }
    }
    if (t.device().is_cpu()) {
      w.u8(tTensor);
    } else {
      // Accelerator tensor: stage to CPU for the wire; remember the device
      // so the receiver-side layer may choose to move it back.
      w.u8(tDevTensor);
    }
    w.u32(static_cast<uint32_t>(tensors.size()));
    if (!t.device().is_cpu()) {
      w.u8(static_cast<uint8_t>(t.device().index() >= 0 ? t.device().index() : 0));
      t = t.to(at::kCPU);
    }
    tensors.push_back(t.is_contiguous() ? t : t.contiguous());
  } else if (isNumpyArray(obj)) {
    // Ride out-of-band as a tensor view when possible.
    try {
      static py::object* fromNumpy =
          new py::object(py::module_::import("torch").attr("from_numpy"));  // leaked, see above
      py::object ascontig =
          py::module_::import("numpy").attr("ascontiguousarray")(py::reinterpret_borrow<py::object>(obj));
      py::object t = (*fromNumpy)(ascontig);
      w.u8(tNdarray);
      w.u32(static_cast<uint32_t>(tensors.size()));
      tensors.push_back(THPVariable_Unpack(t.ptr()));
    } catch (...) {
      PyErr_Clear();
      w.u8(tPickle);
      py::bytes b = pickleDumps()(py::reinterpret_borrow<py::object>(obj), 2);
      w.str(std::string_view(PyBytes_AS_STRING(b.ptr()), PyBytes_GET_SIZE(b.ptr())));
    }
  } else if (py::isinstance<py::tuple>(obj)) {
    auto t = py::reinterpret_borrow<py::tuple>(obj);
    w.u8(tTuple);
    w.u32(static_cast<uint32_t>(t.size()));
    for (auto item : t) serializePy(item, w, tensors, ipcLocal);
  } else if (py::isinstance<py::list>(obj)) {
    auto l = py::reinterpret_borrow<py::list>(obj);
    w.u8(tList);
    w.u32(static_cast<uint32_t>(l.size()));
    for (auto item : l) serializePy(item, w, tensors, ipcLocal);
  } else if (py::isinstance<py::dict>(obj)) {
    auto d = py::reinterpret_borrow<py::dict>(obj);
    w.u8(tDict);
    w.u32(static_cast<uint32_t>(d.size()));
    for (auto item : d) {
      serializePy(item.first, w, tensors, ipcLocal);
      serializePy(item.second, w, tensors, ipcLocal);
    }
  } else {
    w.u8(tPickle);
    py::bytes b = pickleDumps()(py::reinterpret_borrow<py::object>(obj), 2);
    w.str(std::string_view(PyBytes_AS_STRING(b.ptr()), PyBytes_GET_SIZE(b.ptr())));
  }
}

py::object deserializePy(WireReader& r, const std::vector<at::Tensor>& tensors) {
  uint8_t tag = r.u8();
  switch (tag) {
    case tNone:
      return py::none();
    case tTrue:
      return py::bool_(true);
    case tFalse:
      return py::bool_(false);
    case tInt64:
      return py::int_(r.i64());
    case tFloat64:
      return py::float_(r.f64());
    case tStr: {
      auto s = r.str();
      return py::str(s.data(), s.size());
    }
    case tBytes: {
      auto s = r.str();
      return py::bytes(s.data(), s.size());
    }
    case tTensor: {
      uint32_t idx = r.u32();
      if (idx >= tensors.size()) throw RpcError("serde: tensor index out of range");
      return py::reinterpret_steal<py::object>(THPVariable_Wrap(tensors[idx]));
    }
    case tDevTensor: {
      uint32_t idx = r.u32();
      r.u8();  // original device index — receiver decides placement
      if (idx >= tensors.size()) throw RpcError("serde: tensor index out of range");
      return py::reinterpret_steal<py::object>(THPVariable_Wrap(tensors[idx]));
    }
    case tNdarray: {
      uint32_t idx = r.u32();
      if (idx >= tensors.size()) throw RpcError("serde: tensor index out of range");
      py::object t = py::reinterpret_steal<py::object>(THPVariable_Wrap(tensors[idx]));
      return t.attr("numpy")();
    }
    case tTuple: {
      uint32_t n = r.u32();
      py::tuple t(n);
      for (uint32_t i = 0; i < n; ++i) t[i] = deserializePy(r, tensors);
      return std::move(t);
    }
    case tList: {
      uint32_t n = r.u32();
      py::list l;
      for (uint32_t i = 0; i < n; ++i) l.append(deserializePy(r, tensors));
      return std::move(l);
    }
    case tDict: {
      uint32_t n = r.u32();
      py::dict d;
      for (uint32_t i = 0; i < n; ++i) {
        py::object k = deserializePy(r, tensors);
        py::object v = deserializePy(r, tensors);
        d[k] = v;
      }
      return std::move(d);
    }
    case tPickle: {
      auto s = r.str();
      return pickleLoads()(py::bytes(s.data(), s.size()));
    }
    default:
      throw RpcError("serde: unknown tag " + std::to_string(tag));
  }
}

std::string serializeCall(py::tuple args, py::dict kwargs, std::vector<at::Tensor>& tensors,
                          bool ipcLocal) {
  WireWriter w;
  serializePy(args, w, tensors, ipcLocal);
  serializePy(kwargs, w, tensors, ipcLocal);
  return std::move(w.out);
}

std::pair<py::tuple, py::dict> deserializeCall(std::string_view payload,
                                               const std::vector<at::Tensor>& tensors) {
  WireReader r(payload);
  py::object args = deserializePy(r, tensors);
  py::object kwargs = deserializePy(r, tensors);
  return {py::reinterpret_borrow<py::tuple>(args), py::reinterpret_borrow<py::dict>(kwargs)};
}

std::string serializeObject(py::handle obj, std::vector<at::Tensor>& tensors, bool ipcLocal) {
  WireWriter w;
  serializePy(obj, w, tensors, ipcLocal);
  return std::move(w.out);
}

py::object deserializeObject(std::string_view payload, const std::vector<at::Tensor>& tensors) {
  WireReader r(payload);
  return deserializePy(r, tensors);
}

}  // namespace mrl
