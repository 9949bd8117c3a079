"""Build the lambdipy_amd gfx950 kernel extension in-tree.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The extension is AOT-compiled for gfx950 only (no JIT at cold start, no
multi-arch fat binaries): the built .so ships inside recipe wheels / the
repo tree and travels to the GPU box with the source snapshot.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "lambdipy_amd", "ops", "csrc")

sources = [
    os.path.join(CSRC, f)
    for f in sorted(os.listdir(CSRC))
    if f.endswith((".cpp", ".hip")) and not f.endswith(("_hip.cpp", "_hip.hip"))
    and f != "standalone.hip"  # --genco code object, built by the builder
    and f != "nn_capi.cpp"  # liblpx_nn C API, built by builder.ensure_nn_lib
]

setup(
    name="lambdipy-amd",
    version="0.1.0",
    description="MI355X-native scientific-Python packager + runtime",
    packages=[
        "lambdipy_amd",
        "lambdipy_amd.resolver",
        "lambdipy_amd.recipes",
        "lambdipy_amd.builder",
        "lambdipy_amd.packager",
        "lambdipy_amd.ops",
        "lambdipy_amd.runtime",
        "lambdipy_amd.models",
        "lambdipy_amd.parallel",
        "lambdipy_amd.utils",
    ],
    package_data={"lambdipy_amd.recipes": ["data/*.json"]},
    ext_modules=[
        CUDAExtension(
            name="lambdipy_amd.ops._lpx_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
    entry_points={"console_scripts": ["lpx = lambdipy_amd.cli:main"]},
)
