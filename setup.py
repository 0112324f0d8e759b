"""Build the msbn._C extension in-tree for gfx950 (MI355X).

    python setup.py build_ext --inplace

The built .so lands at msbn/_C*.so (git-ignored; it travels to the GPU box via
the gpurun snapshot).  hipcc cross-compiles gfx950 without a GPU present.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from setuptools import setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "msbn", "ops", "csrc")

ext = CUDAExtension(
    name="msbn._C",
    sources=[
        os.path.join(CSRC, "module.cpp"),
        os.path.join(CSRC, "bn_kernels.hip"),
    ],
    include_dirs=[CSRC],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

exts = [ext]
if os.environ.get("MSBN_BUILD_NONT", "0") == "1":
    # A/B variant with nontemporal accesses disabled (cached loads/stores),
    # for same-box kernel comparisons: tools/kernel_bench.py --impl nont
    exts.append(CUDAExtension(
        name="msbn._C_nont",
        # dedicated translation units: sharing bn_kernels.hip would collide
        # on the object path and silently reuse the product (NT) kernels
        sources=[
            os.path.join(CSRC, "module_nont_ab.cpp"),
            os.path.join(CSRC, "bn_kernels_nont_ab.hip"),
        ],
        include_dirs=[CSRC],
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17", "-DMSBN_DISABLE_NT"],
            "nvcc": ["-O3", "-std=c++17", "-DMSBN_DISABLE_NT"],
        },
    ))

setup(
    name="msbn",
    version="0.1.0",
    description="MI355X-native SyncBatchNorm + DDP training framework",
    packages=[
        "msbn", "msbn.nn", "msbn.ops", "msbn.parallel", "msbn.data",
        "msbn.models", "msbn.utils",
    ],
    ext_modules=exts,
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
