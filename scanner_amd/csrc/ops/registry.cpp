#include "kernel.h"

namespace sca {

void Kernel::execute(const StenciledElements& input, BatchedElements& output) {
  size_t ncols = input.size();
  size_t nrows = ncols ? input[0].size() : 0;
  for (size_t r = 0; r < nrows; ++r) {
    ElementVector row_in(ncols);
    for (size_t c = 0; c < ncols; ++c) {
      SCA_CHECK(input[c][r].size() == 1, "plain Kernel got stencil input");
      row_in[c] = input[c][r][0];
    }
    ElementVector row_out(output.size());
    execute_row(row_in, row_out);
    for (size_t c = 0; c < output.size(); ++c) output[c].push_back(row_out[c]);
  }
}

void BatchedKernel::execute(const StenciledElements& input,
                            BatchedElements& output) {
  size_t ncols = input.size();
  size_t nrows = ncols ? input[0].size() : 0;
  BatchedElements in(ncols);
  for (size_t c = 0; c < ncols; ++c) {
    in[c].reserve(nrows);
    for (size_t r = 0; r < nrows; ++r) {
      SCA_CHECK(input[c][r].size() == 1, "BatchedKernel got stencil input");
      in[c].push_back(input[c][r][0]);
    }
  }
  execute_batch(in, output);
}

void StenciledKernel::execute(const StenciledElements& input,
                              BatchedElements& output) {
  size_t ncols = input.size();
  size_t nrows = ncols ? input[0].size() : 0;
  for (size_t r = 0; r < nrows; ++r) {
    BatchedElements row_in(ncols);
    for (size_t c = 0; c < ncols; ++c) row_in[c] = input[c][r];
    ElementVector row_out(output.size());
    execute_stencil(row_in, row_out);
    for (size_t c = 0; c < output.size(); ++c) output[c].push_back(row_out[c]);
  }
}

void OpRegistry::add(OpInfo info) {
  ops_[info.name] = std::move(info);
}
bool OpRegistry::has(const std::string& name) const { return ops_.count(name); }
const OpInfo& OpRegistry::get(const std::string& name) const {
  auto it = ops_.find(name);
  SCA_CHECK(it != ops_.end(), "unknown op '" + name + "'");
  return it->second;
}
std::vector<std::string> OpRegistry::names() const {
  std::vector<std::string> out;
  for (auto& kv : ops_) out.push_back(kv.first);
  return out;
}

void KernelRegistry::add(KernelFactory f) {
  factories_[{f.op_name, f.device_type}] = std::move(f);
}
bool KernelRegistry::has(const std::string& op, DeviceType d) const {
  return factories_.count({op, d});
}
const KernelFactory& KernelRegistry::get(const std::string& op,
                                         DeviceType d) const {
  auto it = factories_.find({op, d});
  SCA_CHECK(it != factories_.end(),
            "no kernel for op '" + op + "' on device type " +
                std::to_string((i32)d));
  return it->second;
}

OpRegistry& op_registry() {
  static OpRegistry r;
  return r;
}
KernelRegistry& kernel_registry() {
  static KernelRegistry r;
  return r;
}

OpRegistrar::OpRegistrar(OpInfo info) { op_registry().add(std::move(info)); }
KernelRegistrar::KernelRegistrar(KernelFactory f) {
  kernel_registry().add(std::move(f));
}

}  // namespace sca
