"""User C++ op plugin: build with tools/build_op.py, load with
Client.load_op, run in a graph (parity: the reference's user-op .so
workflow — REGISTER_OP static registrars + build_flags.py + load_op)."""
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

PLUGIN_SRC = r"""
// Example user op: per-pixel invert of u8 frames.
#include "csrc/memory.h"
#include "csrc/ops/kernel.h"

namespace {

using namespace sca;

class InvertKernel : public BatchedKernel {
 public:
  using BatchedKernel::BatchedKernel;
  void execute_batch(const BatchedElements& in, BatchedElements& out) override {
    for (const Element& f : in[0]) {
      Element e;
      e.is_frame = true;
      e.frame_info = f.frame_info;
      e.size = f.size;
      e.device = config_.device;
      e.buffer = new_buffer(config_.device, e.size);
      for (size_t i = 0; i < f.size; ++i) e.buffer[i] = 255 - f.buffer[i];
      out[0].push_back(e);
    }
  }
};

OpInfo invert_info() {
  OpInfo o;
  o.name = "Invert";
  o.input_columns = {{"frame", ColumnType::Video}};
  o.output_columns = {{"frame", ColumnType::Video}};
  return o;
}

KernelFactory invert_factory() {
  KernelFactory f;
  f.op_name = "Invert";
  f.device_type = DeviceType::CPU;
  f.preferred_batch = 4;
  f.make = [](const KernelConfig& c) -> std::unique_ptr<BaseKernel> {
    return std::make_unique<InvertKernel>(c);
  };
  return f;
}

SCA_REGISTER_OP(invert, invert_info());
SCA_REGISTER_KERNEL(invert, invert_factory());

}  // namespace
"""


@pytest.mark.timeout(300)
def test_cpp_op_plugin(sc, tmp_path):
    import scanner_amd as sp
    from conftest import make_video

    src = tmp_path / "invert_op.cpp"
    src.write_text(PLUGIN_SRC)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "build_op.py"),
         str(src), "-o", str(tmp_path / "invert_op.so")],
        capture_output=True, timeout=240)
    assert out.returncode == 0, out.stderr.decode()

    sc.load_op(str(tmp_path / "invert_op.so"))
    from scanner_amd import _core
    assert "Invert" in _core.registered_ops()

    frames = make_video(n=6, h=24, w=32)
    video = sp.NamedVideoStream(sc, "inv", frames=frames, codec="raw")
    frame = sc.io.Input([video])
    inv = sc.ops.Invert(frame=frame)
    o = sp.NamedStream(sc, "inv_out")
    sc.run(sc.io.Output(inv, [o]), sp.PerfParams.manual(4, 8),
           cache_mode=sp.CacheMode.Overwrite)
    got = np.stack(list(sp.NamedVideoStream(sc, "inv_out").load()))
    np.testing.assert_array_equal(got, 255 - frames)
