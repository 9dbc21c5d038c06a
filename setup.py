"""In-tree build of the tnn_amd HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting tnn_amd/_hip*.so travels with the source tree (it is
git-ignored but NOT gpurun-ignored, so it ships to GPU boxes).
"""

import os
import glob

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "tnn_amd", "csrc")

# torch's hipify writes *_hip.hip shadow copies next to the originals;
# exclude them or a rebuild would compile every kernel twice.
sources = sorted(p for p in glob.glob(os.path.join(CSRC, "*.hip"))
                 if not p.endswith("_hip.hip")) + [
    os.path.join(CSRC, "bindings.cpp"),
    os.path.join(CSRC, "imagecodec.cpp"),  # host-only (stb_image analog)
]

cxx_flags = ["-O3", "-std=c++17"]
hip_flags = ["-O3", "-std=c++17", "--offload-arch=gfx950"]
if os.environ.get("TNN_DEBUG"):
    # debug builds get host sanitizers (reference cmake/CompilerFlags.cmake
    # -fsanitize=address,undefined analog); device-side: run under
    # AMD_SERIALIZE_KERNEL=3 HIP_LAUNCH_BLOCKING=1
    san = ["-g", "-fsanitize=address,undefined", "-fno-omit-frame-pointer"]
    cxx_flags += san
    hip_flags += ["-g"]

setup(
    name="tnn_amd_hip",
    ext_modules=[
        CUDAExtension(
            name="tnn_amd._hip",
            sources=sources,
            extra_compile_args={"cxx": cxx_flags, "nvcc": hip_flags},
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
