"""Build the dtmx HIP extension in-tree for gfx950 (MI355X):

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces dtmx/_C.*.so next to the package sources so the snapshot that
travels to a GPU box carries the binary.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "dtmx", "csrc")

# Exclude hipify-generated "*_hip.hip" twins: torch's build hipifies each
# .hip source into a sibling <name>_hip.hip; globbing those back in would
# feed every kernel through the compiler twice (duplicate symbols / stale
# binaries on incremental builds).
sources = [
    os.path.join("dtmx", "csrc", f)
    for f in sorted(os.listdir(CSRC))
    if f.endswith((".cpp", ".hip")) and not f.endswith("_hip.hip")
]

setup(
    name="dtmx",
    version="0.1.0",
    packages=["dtmx"],
    ext_modules=[
        CUDAExtension(
            name="dtmx._C",
            sources=sources,
            # libjpeg (v9, /opt/conda ships the only dev copy in this image)
            # powers the data pipeline's decode stage (recordio.cpp); rpath
            # pins the matching runtime .so.9. Torch/stdc++ sonames are
            # already loaded by `import torch` before this extension, so the
            # extra rpath entry cannot redirect them.
            include_dirs=["/opt/conda/include"],
            library_dirs=["/opt/conda/lib"],
            runtime_library_dirs=["/opt/conda/lib"],
            libraries=["jpeg"],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
