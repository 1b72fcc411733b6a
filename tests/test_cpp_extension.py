"""Custom-op framework: JIT-compile a C++ op through
paddle.utils.cpp_extension.load (reference: utils/cpp_extension/).
Pure-C++ op so the test runs on the CPU box; hipcc-compiled ops use the
same path with PYTORCH_ROCM_ARCH=gfx950 pinned."""
import os
import tempfile

import torch


def test_load_cpp_custom_op(tmp_path):
    src = tmp_path / "my_op.cpp"
    src.write_text(r'''
#include <torch/extension.h>
torch::Tensor double_add(torch::Tensor a, torch::Tensor b) {
  return 2 * (a + b);
}
PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) { m.def("double_add", &double_add); }
''')
    from paddle_amd.utils import cpp_extension as ce
    mod = ce.load(name="pa_test_double_add", sources=[str(src)],
                  build_directory=str(tmp_path), verbose=False)
    out = mod.double_add(torch.ones(3), torch.ones(3))
    assert torch.allclose(out, torch.full((3,), 4.0))
    assert os.environ.get("PYTORCH_ROCM_ARCH") == "gfx950"
