// Python-op bridge (parity role: scanner/engine/python_kernel.cpp +
// @register_python_op). Design difference vs the reference: the reference
// spawns one Python child process per kernel instance and pipes cloudpickled
// frames to dodge the GIL; here Python kernels run in-process — executor
// threads hold the GIL only for the Python call itself, and the hot path
// (C++/HIP ops) never touches Python. A subprocess pool can be added behind
// the same registration call if Python-bound pipelines need it.
#pragma once

#include <pybind11/pybind11.h>

namespace sca {

// Registers OpInfo + a KernelFactory whose kernels call back into `factory`
// (a Python callable returning an object with methods new_stream(bytes),
// reset(), execute(cols) where cols is
// [input_col][row][stencil] of (numpy|bytes|None) and the return is
// [output_col][row] of (numpy|bytes|None)).
void register_python_op_binding(
    const std::string& name, pybind11::object factory,
    const std::vector<std::pair<std::string, int>>& input_columns,
    const std::vector<std::pair<std::string, int>>& output_columns,
    int device_type, int batch, std::vector<int> stencil, bool bounded_state,
    int warmup, bool unbounded_state);

void register_gpu_ops();

}  // namespace sca
