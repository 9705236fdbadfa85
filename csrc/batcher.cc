// Batcher: dynamic batching of nested dict/list/tuple-of-tensor structures.
//
// Capability parity with the reference's Batcher (src/moolib.cc:595-889):
// `stack` inserts a new batch dim at `dim` and fills preallocated target
// tensors on the configured device with copy_(non_blocking); `cat`
// concatenates along an existing dim with overflow carried into the next
// batch (one call can complete several batches). A completed batch moves to
// an awaitable ready-queue (get()/empty()/__await__ — same surface as the
// reference).
//
// MI355X note: targets live on the learner GPU; copies are issued
// non-blocking so H2D of actor rollouts overlaps the learner's compute
// stream. When moolib_amd._kernels is importable and a GPU is present,
// the package registers its batched_copy kernel here (setFusedCopy) and
// every stack()/cat() call moves ALL its leaves in one launch instead of
// one ramp-dominated runtime copy per leaf.
#include "batcher.h"

#include <torch/csrc/autograd/python_variable.h>

namespace mrl {

namespace {

// Structure-preserving map over a nested dict/list/tuple; f maps leaves.
py::object mapNest(py::handle n, const std::function<py::object(py::handle)>& f) {
  if (py::isinstance<py::dict>(n)) {
    py::dict out;
    for (auto item : py::reinterpret_borrow<py::dict>(n)) {
      out[item.first] = mapNest(item.second, f);
    }
    return std::move(out);
  }
  if (py::isinstance<py::list>(n)) {
    py::list out;
    for (auto item : py::reinterpret_borrow<py::list>(n)) out.append(mapNest(item, f));
    return std::move(out);
  }
  if (py::isinstance<py::tuple>(n)) {
    auto t = py::reinterpret_borrow<py::tuple>(n);
    py::tuple out(t.size());
    for (size_t i = 0; i < t.size(); ++i) out[i] = mapNest(t[i], f);
    return std::move(out);
  }
  return f(n);
}

// Parallel walk over two same-shaped nests.
void zipNest(py::handle a, py::handle b, const std::function<void(py::handle, py::handle)>& f) {
  if (py::isinstance<py::dict>(a)) {
    auto da = py::reinterpret_borrow<py::dict>(a);
    auto db = py::reinterpret_borrow<py::dict>(b);
    if (py::len(da) != py::len(db)) throw RpcError("batcher: nest structure mismatch");
    for (auto item : da) {
      if (!db.contains(item.first)) throw RpcError("batcher: nest key mismatch");
      zipNest(item.second, db[item.first], f);
    }
    return;
  }
  if (py::isinstance<py::list>(a) || py::isinstance<py::tuple>(a)) {
    auto sa = py::reinterpret_borrow<py::sequence>(a);
    auto sb = py::reinterpret_borrow<py::sequence>(b);
    if (sa.size() != sb.size()) throw RpcError("batcher: nest length mismatch");
    for (size_t i = 0; i < sa.size(); ++i) zipNest(sa[i], sb[i], f);
    return;
  }
  f(a, b);
}

bool isTensor(py::handle h) { return THPVariable_Check(h.ptr()); }

// Fused copy hook: fn(list[Tensor] dsts, list[Tensor] srcs) performing
// dst.copy_(src) for every pair (one kernel launch for the CUDA pairs).
// Leaked pointer: destroying a py::object at static-destruction time
// would race interpreter teardown.
py::object* g_fusedCopy = nullptr;

}  // namespace

void setBatcherFusedCopy(py::object fn) {
  if (g_fusedCopy == nullptr) g_fusedCopy = new py::object();
  *g_fusedCopy = fn.is_none() ? py::object() : std::move(fn);
}

Batcher::Batcher(int64_t size, py::object device, int64_t dim)
    : size_(size), dim_(dim) {
  if (size_ <= 0) throw RpcError("batcher: size must be positive");
  if (!device.is_none()) {
    if (py::isinstance<py::str>(device)) {
      device_ = at::Device(py::cast<std::string>(device));
    } else {
      device_ = py::cast<at::Device>(device);
    }
  }
}

at::Tensor Batcher::makeTarget(const at::Tensor& src, bool insertDim) {
  std::vector<int64_t> shape(src.sizes().begin(), src.sizes().end());
  if (insertDim) {
    if (dim_ > static_cast<int64_t>(shape.size())) throw RpcError("batcher: dim out of range");
    shape.insert(shape.begin() + dim_, size_);
  } else {
    if (dim_ >= static_cast<int64_t>(shape.size())) throw RpcError("batcher: dim out of range");
    shape[dim_] = size_;
  }
  auto opts = at::TensorOptions().dtype(src.scalar_type());
  opts = device_ ? opts.device(*device_) : opts.device(src.device());
  return at::empty(shape, opts);
}

void Batcher::completeBatch() {
  ready_.enqueue(current_);
  current_ = py::object();
  fill_ = 0;
}

void Batcher::stack(py::object nest) {
  if (!current_.ptr() || current_.is_none()) {
    current_ = mapNest(nest, [&](py::handle leaf) -> py::object {
      if (!isTensor(leaf)) return py::reinterpret_borrow<py::object>(leaf);
      at::Tensor src = THPVariable_Unpack(leaf.ptr());
      at::Tensor dst = makeTarget(src, /*insertDim=*/true);
      return py::reinterpret_steal<py::object>(THPVariable_Wrap(dst));
    });
    fill_ = 0;
  }
  bool fused = g_fusedCopy != nullptr && *g_fusedCopy;
  py::list fdsts, fsrcs;
  zipNest(current_, nest, [&](py::handle t, py::handle s) {
    if (!isTensor(t) || !isTensor(s)) return;
    at::Tensor dst = THPVariable_Unpack(t.ptr());
    at::Tensor src = THPVariable_Unpack(s.ptr());
    at::Tensor slot = dst.select(dim_, fill_);
    if (fused) {
      fdsts.append(py::reinterpret_steal<py::object>(THPVariable_Wrap(slot)));
      fsrcs.append(py::reinterpret_borrow<py::object>(s));
    } else {
      slot.copy_(src, /*non_blocking=*/true);
    }
  });
  if (fused && py::len(fdsts) > 0) (*g_fusedCopy)(fdsts, fsrcs);
  ++fill_;
  if (fill_ >= size_) completeBatch();
}

void Batcher::cat(py::object nest) {
  // Source batch length along dim_ (from the first tensor leaf).
  int64_t srcLen = -1;
  mapNest(nest, [&](py::handle leaf) -> py::object {
    if (srcLen < 0 && isTensor(leaf)) {
      srcLen = THPVariable_Unpack(leaf.ptr()).size(dim_);
    }
    return py::none();
  });
  if (srcLen < 0) throw RpcError("batcher: cat() needs at least one tensor leaf");
  int64_t srcOff = 0;
  while (srcOff < srcLen) {
    if (!current_.ptr() || current_.is_none()) {
      current_ = mapNest(nest, [&](py::handle leaf) -> py::object {
        if (!isTensor(leaf)) return py::reinterpret_borrow<py::object>(leaf);
        at::Tensor src = THPVariable_Unpack(leaf.ptr());
        at::Tensor dst = makeTarget(src, /*insertDim=*/false);
        return py::reinterpret_steal<py::object>(THPVariable_Wrap(dst));
      });
      fill_ = 0;
    }
    int64_t n = std::min(srcLen - srcOff, size_ - fill_);
    bool fused = g_fusedCopy != nullptr && *g_fusedCopy;
    py::list fdsts, fsrcs;
    zipNest(current_, nest, [&](py::handle t, py::handle s) {
      if (!isTensor(t) || !isTensor(s)) return;
      at::Tensor dst = THPVariable_Unpack(t.ptr());
      at::Tensor src = THPVariable_Unpack(s.ptr());
      at::Tensor dslot = dst.narrow(dim_, fill_, n);
      at::Tensor sslot = src.narrow(dim_, srcOff, n);
      if (fused) {
        fdsts.append(py::reinterpret_steal<py::object>(THPVariable_Wrap(dslot)));
        fsrcs.append(py::reinterpret_steal<py::object>(THPVariable_Wrap(sslot)));
      } else {
        dslot.copy_(sslot, /*non_blocking=*/true);
      }
    });
    if (fused && py::len(fdsts) > 0) (*g_fusedCopy)(fdsts, fsrcs);
    fill_ += n;
    srcOff += n;
    if (fill_ >= size_) completeBatch();
  }
}

bool Batcher::empty() { return ready_.size() == 0; }

size_t Batcher::size() { return ready_.size(); }

py::object Batcher::get() { return ready_.popFuture().resultNoTimeout(); }

PyFuture Batcher::popFuture() { return ready_.popFuture(); }

}  // namespace mrl
