// Kernel SDK: base classes C++ ops implement, plus the registries that map
// op names to metadata and (op, device) to kernel factories.
// Capability parity: scanner/api/kernel.h:117-476, op.h, kernel_factory.h,
// engine/{op,kernel}_registry.h. MI355X difference: a GPU KernelConfig
// carries the pipeline instance's HIP stream — kernels launch onto it and
// never synchronize the device themselves.
#pragma once

#include <functional>
#include <map>
#include <memory>

#include "../element.h"
#include "../profiler.h"

namespace sca {

struct KernelConfig {
  DeviceHandle device;
  // Where this kernel's outputs live. Defaults to `device`; kernels
  // registered with an explicit output device (reference:
  // KernelBuilder.output_device(), kernel.h:412-475) may differ — e.g. a
  // GPU detector emitting small CPU-side box lists.
  DeviceHandle output_device;
  std::vector<std::string> input_columns;
  std::vector<std::string> output_columns;
  std::vector<u8> args;         // op-instance args (msgpack from Python)
  i32 node_id = 0;
  i32 max_batch = 1;
  void* hip_stream = nullptr;   // hipStream_t of the owning pipeline instance
  Profiler* profiler = nullptr;
};

// The single internal execution interface. Input is per-column, per-row,
// per-stencil-offset; output is per-column, per-row. Subclass adapters
// below present the reference's four user-facing shapes.
class BaseKernel {
 public:
  explicit BaseKernel(const KernelConfig& config) : config_(config) {}
  virtual ~BaseKernel() = default;

  // Called when the kernel starts a new output stream (job). args are the
  // per-stream op args (may be empty).
  virtual void new_stream(const std::vector<u8>& args) {}
  // Reset stateful kernels at slice-group / discontinuity boundaries.
  virtual void reset() {}
  // Fetch-once resources (weights etc.) — worker 0 only, then barrier.
  virtual void fetch_resources(const std::vector<u8>& args) {}
  virtual void setup_with_resources(const std::vector<u8>& args) {}

  virtual void execute(const StenciledElements& input_columns,
                       BatchedElements& output_columns) = 0;

  // Preferred batch size (from factory registration).
  const KernelConfig& config() const { return config_; }

 protected:
  KernelConfig config_;
};

// One row at a time, no stencil: execute(row_inputs, row_outputs).
class Kernel : public BaseKernel {
 public:
  using BaseKernel::BaseKernel;
  virtual void execute_row(const ElementVector& input_columns,
                           ElementVector& output_columns) = 0;
  void execute(const StenciledElements& input, BatchedElements& output) override;
};

// Whole batch, no stencil: input[col][row], output[col][row].
class BatchedKernel : public BaseKernel {
 public:
  using BaseKernel::BaseKernel;
  virtual void execute_batch(const BatchedElements& input_columns,
                             BatchedElements& output_columns) = 0;
  void execute(const StenciledElements& input, BatchedElements& output) override;
};

// One row with stencil window: input[col][stencil_off].
class StenciledKernel : public BaseKernel {
 public:
  using BaseKernel::BaseKernel;
  virtual void execute_stencil(const BatchedElements& input_columns,
                               ElementVector& output_columns) = 0;
  void execute(const StenciledElements& input, BatchedElements& output) override;
};

// Full shape (the internal interface) — subclass BaseKernel directly.
using StenciledBatchedKernel = BaseKernel;

// ---------------- registries ----------------

struct OpColumnDef {
  std::string name;
  ColumnType type = ColumnType::Bytes;
};

struct OpInfo {
  std::string name;
  std::vector<OpColumnDef> input_columns;
  std::vector<OpColumnDef> output_columns;
  bool variadic_inputs = false;
  std::vector<i32> stencil = {0};
  bool has_bounded_state = false;
  i32 warmup = 0;
  bool has_unbounded_state = false;
  // builtin ops (Input/Output/Sample/Space/Slice/Unslice) are executed by
  // the engine itself, not by a kernel
  bool is_builtin = false;
};

struct KernelFactory {
  std::string op_name;
  DeviceType device_type = DeviceType::CPU;
  // -1 = outputs on the kernel device (default); else a DeviceType the
  // kernel's outputs live on (reference .output_device()).
  i32 output_device_type = -1;
  i32 preferred_batch = 1;
  // number of devices this kernel wants (reference .num_devices());
  i32 num_devices = 1;
  std::function<std::unique_ptr<BaseKernel>(const KernelConfig&)> make;
};

class OpRegistry {
 public:
  void add(OpInfo info);
  bool has(const std::string& name) const;
  const OpInfo& get(const std::string& name) const;
  std::vector<std::string> names() const;

 private:
  std::map<std::string, OpInfo> ops_;
};

class KernelRegistry {
 public:
  void add(KernelFactory f);
  bool has(const std::string& op, DeviceType d) const;
  const KernelFactory& get(const std::string& op, DeviceType d) const;

 private:
  std::map<std::pair<std::string, DeviceType>, KernelFactory> factories_;
};

OpRegistry& op_registry();
KernelRegistry& kernel_registry();

// Static-registration helpers (REGISTER_OP / REGISTER_KERNEL analogue).
struct OpRegistrar {
  explicit OpRegistrar(OpInfo info);
};
struct KernelRegistrar {
  explicit KernelRegistrar(KernelFactory f);
};

#define SCA_REGISTER_OP(var, ...) \
  static ::sca::OpRegistrar op_registrar_##var(__VA_ARGS__)
#define SCA_REGISTER_KERNEL(var, ...) \
  static ::sca::KernelRegistrar kernel_registrar_##var(__VA_ARGS__)

// Register all built-in C++ ops/kernels (called once from module init; we
// avoid static-initializer ordering issues by explicit registration).
void register_stdlib_ops();
void register_resnet50_op();
void register_optflow_gpu();
void register_pose_op();
void register_color_gpu();
void register_image_encoder_op();
void register_detector_op();

}  // namespace sca
