"""Build the MI355X (gfx950) HIP kernel extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

produces ``tf_yarn_amd/ops/_C.*.so`` next to its Python wrappers so the
built library travels with the source tree.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from pybind11.setup_helpers import Pybind11Extension  # noqa: E402
from setuptools import find_packages, setup  # noqa: E402
from torch.utils.cpp_extension import (BuildExtension,  # noqa: E402
                                       CUDAExtension)

CSRC = os.path.join("tf_yarn_amd", "ops", "csrc")

# torch-free (imports without loading libtorch; the control plane must be
# usable in any task process)
kv_ext = Pybind11Extension(
    "tf_yarn_amd._kv_native",
    sources=[os.path.join("tf_yarn_amd", "csrc_kv", "kv_server.cpp")],
    cxx_std=17,
    extra_compile_args=["-O2", "-pthread"],
)

ext = CUDAExtension(
    name="tf_yarn_amd.ops._C",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "fused_optimizers.hip"),
        os.path.join(CSRC, "embedding.hip"),
        os.path.join(CSRC, "binned_scatter.hip"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "wgrad.hip"),
        os.path.join(CSRC, "wgrad128.hip"),
        os.path.join(CSRC, "wgrad256.hip"),
        os.path.join(CSRC, "gemm_bt.hip"),
        os.path.join(CSRC, "lt_linear.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "--offload-arch=gfx950"],
    },
    libraries=["hipblaslt"],
)

setup(
    name="tf_yarn_amd",
    version="0.1.0",
    description=("MI355X-native distributed-training launcher with "
                 "criteo/tf-yarn's capabilities"),
    packages=find_packages(include=["tf_yarn_amd", "tf_yarn_amd.*"]),
    package_data={"tf_yarn_amd": ["default.log.conf"]},
    python_requires=">=3.9",
    ext_modules=[ext, kv_ext],
    cmdclass={"build_ext": BuildExtension},
)
