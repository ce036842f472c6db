#include "python_kernel.h"

#include <pybind11/numpy.h>

#include "../memory.h"
#include "kernel.h"

namespace py = pybind11;

namespace sca {

namespace {

py::object element_to_py(const Element& e) {
  if (e.is_null || e.buffer == nullptr) return py::none();
  SCA_CHECK(!e.device.is_gpu(),
            "python kernels take CPU inputs (set device=CPU)");
  if (e.is_frame) {
    const FrameInfo& fi = e.frame_info;
    std::vector<ssize_t> shape;
    shape.push_back(fi.shape[0]);
    if (fi.shape[1]) shape.push_back(fi.shape[1]);
    if (fi.shape[2]) shape.push_back(fi.shape[2]);
    std::string fmt;
    switch (fi.type) {
      case FrameType::U8: fmt = py::format_descriptor<u8>::format(); break;
      case FrameType::U16: fmt = py::format_descriptor<u16>::format(); break;
      case FrameType::F32: fmt = py::format_descriptor<f32>::format(); break;
      case FrameType::F64: fmt = py::format_descriptor<f64>::format(); break;
    }
    size_t isz = frame_type_size(fi.type);
    std::vector<ssize_t> strides(shape.size());
    ssize_t s = isz;
    for (i64 i = (i64)shape.size() - 1; i >= 0; --i) {
      strides[i] = s;
      s *= shape[i];
    }
    // Zero-copy view; valid for the duration of the execute() call.
    return py::array(py::buffer_info(e.buffer, isz, fmt, shape.size(), shape,
                                     strides));
  }
  return py::bytes(reinterpret_cast<const char*>(e.buffer), e.size);
}

Element element_from_py(const py::handle& o, DeviceHandle dev) {
  Element e;
  if (o.is_none()) {
    e.is_null = true;
    return e;
  }
  if (py::isinstance<py::array>(o)) {
    auto arr = py::cast<py::array>(o);
    auto ac = py::array::ensure(arr, py::array::c_style | py::array::forcecast);
    FrameInfo fi;
    SCA_CHECK(ac.ndim() >= 1 && ac.ndim() <= 3,
              "python op frame output must be 1-3D");
    for (i32 i = 0; i < ac.ndim(); ++i) fi.shape[i] = (i32)ac.shape(i);
    auto dt = ac.dtype();
    if (dt.is(py::dtype::of<u8>())) fi.type = FrameType::U8;
    else if (dt.is(py::dtype::of<u16>())) fi.type = FrameType::U16;
    else if (dt.is(py::dtype::of<f32>())) fi.type = FrameType::F32;
    else if (dt.is(py::dtype::of<f64>())) fi.type = FrameType::F64;
    else throw ScannerError("python op output dtype must be u8/u16/f32/f64");
    e.is_frame = true;
    e.frame_info = fi;
    e.size = fi.size();
    e.buffer = new_buffer(dev, e.size);
    e.device = dev;
    std::memcpy(e.buffer, ac.data(), e.size);
    return e;
  }
  // bytes-like
  std::string s = py::cast<std::string>(py::cast<py::bytes>(o));
  e.size = s.size();
  e.buffer = new_buffer(dev, e.size == 0 ? 1 : e.size);
  e.device = dev;
  std::memcpy(e.buffer, s.data(), e.size);
  return e;
}

class PythonKernel : public BaseKernel {
 public:
  PythonKernel(const KernelConfig& cfg, py::object factory, size_t n_outputs)
      : BaseKernel(cfg), n_outputs_(n_outputs) {
    py::gil_scoped_acquire gil;
    py::bytes args(reinterpret_cast<const char*>(cfg.args.data()),
                   cfg.args.size());
    obj_ = factory(args);
  }
  ~PythonKernel() override {
    if (!Py_IsInitialized()) {
      obj_.release();  // interpreter gone; leak instead of crashing
      return;
    }
    py::gil_scoped_acquire gil;
    obj_ = py::object();
  }

  void new_stream(const std::vector<u8>& args) override {
    py::gil_scoped_acquire gil;
    if (py::hasattr(obj_, "new_stream")) {
      obj_.attr("new_stream")(
          py::bytes(reinterpret_cast<const char*>(args.data()), args.size()));
    }
  }

  void reset() override {
    py::gil_scoped_acquire gil;
    if (py::hasattr(obj_, "reset")) obj_.attr("reset")();
  }

  void fetch_resources(const std::vector<u8>& args) override {
    py::gil_scoped_acquire gil;
    if (py::hasattr(obj_, "fetch_resources")) {
      obj_.attr("fetch_resources")(
          py::bytes(reinterpret_cast<const char*>(args.data()), args.size()));
    }
  }

  void setup_with_resources(const std::vector<u8>& args) override {
    py::gil_scoped_acquire gil;
    if (py::hasattr(obj_, "setup_with_resources")) {
      obj_.attr("setup_with_resources")(
          py::bytes(reinterpret_cast<const char*>(args.data()), args.size()));
    }
  }

  void execute(const StenciledElements& input,
               BatchedElements& output) override {
    py::gil_scoped_acquire gil;
    py::list cols;
    for (auto& col : input) {
      py::list rows;
      for (auto& row : col) {
        py::list window;
        for (auto& e : row) window.append(element_to_py(e));
        rows.append(window);
      }
      cols.append(rows);
    }
    py::object result = obj_.attr("execute")(cols);
    auto out_cols = py::cast<py::list>(result);
    SCA_CHECK(out_cols.size() == n_outputs_,
              "python op returned wrong number of output columns");
    for (size_t c = 0; c < n_outputs_; ++c) {
      auto rows = py::cast<py::list>(out_cols[c]);
      for (auto row : rows) {
        output[c].push_back(element_from_py(row, config_.device));
      }
    }
  }

 private:
  py::object obj_;
  size_t n_outputs_;
};

}  // namespace

void register_python_op_binding(
    const std::string& name, py::object factory,
    const std::vector<std::pair<std::string, int>>& input_columns,
    const std::vector<std::pair<std::string, int>>& output_columns,
    int device_type, int batch, std::vector<int> stencil, bool bounded_state,
    int warmup, bool unbounded_state) {
  OpInfo o;
  o.name = name;
  for (auto& c : input_columns)
    o.input_columns.push_back({c.first, (ColumnType)c.second});
  for (auto& c : output_columns)
    o.output_columns.push_back({c.first, (ColumnType)c.second});
  o.stencil.clear();
  for (int s : stencil) o.stencil.push_back(s);
  if (o.stencil.empty()) o.stencil = {0};
  o.has_bounded_state = bounded_state;
  o.warmup = warmup;
  o.has_unbounded_state = unbounded_state;
  op_registry().add(o);

  size_t n_out = output_columns.size();
  // Keep the factory alive for process lifetime. The registry is a static;
  // its destruction runs after Py_Finalize, so the py::object must never
  // be destroyed — leak it deliberately.
  std::shared_ptr<py::object> holder(new py::object(std::move(factory)),
                                     [](py::object* p) {
                                       if (Py_IsInitialized()) {
                                         py::gil_scoped_acquire g;
                                         delete p;
                                       } else {
                                         p->release();
                                       }
                                     });
  KernelFactory f;
  f.op_name = name;
  f.device_type = (DeviceType)device_type;
  f.preferred_batch = batch > 0 ? batch : 1;
  f.make = [holder, n_out](const KernelConfig& cfg)
      -> std::unique_ptr<BaseKernel> {
    // executor threads call this without the GIL; copying the factory
    // py::object inc_refs it
    py::gil_scoped_acquire gil;
    return std::make_unique<PythonKernel>(cfg, *holder, n_out);
  };
  kernel_registry().add(f);
}

}  // namespace sca
