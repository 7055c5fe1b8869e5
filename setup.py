"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the sources (pdnlp_amd/ops/) so it travels with
the repo snapshot to the GPU box; it is git-ignored (history stays
source-only).
"""

import glob
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                    "pdnlp_amd", "ops", "csrc")
# hipify writes X_hip.hip copies next to X.hip — exclude them from the glob
# (they are build artifacts, also git-ignored)
sources = sorted(glob.glob(os.path.join(CSRC, "*.cpp"))
                 + [f for f in glob.glob(os.path.join(CSRC, "*.hip"))
                    if not f.endswith("_hip.hip")])

ext_modules = []
if sources:
    ext_modules.append(cpp_extension.CUDAExtension(
        name="pdnlp_amd.ops._hip_ext",
        sources=sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
        },
    ))

setup(
    name="pdnlp_amd",
    version="0.1.0",
    packages=["pdnlp_amd", "pdnlp_amd.ops", "pdnlp_amd.data",
              "pdnlp_amd.models", "pdnlp_amd.parallel", "pdnlp_amd.engine",
              "pdnlp_amd.amp", "pdnlp_amd.utils"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
