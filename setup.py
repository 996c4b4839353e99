"""In-tree build of the ring_attention_amd CDNA4 (gfx950) HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The .so lands inside the package (ring_attention_amd/_ring_attn_hip*.so) so it
travels with the repo snapshot to GPU boxes — no JIT cache dependence.
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "ring_attention_amd", "csrc")

ext = CUDAExtension(
    name="ring_attention_amd._ring_attn_hip",
    sources=[
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "attn_fwd.hip"),
        os.path.join(CSRC, "attn_fwd_v2.hip"),
        os.path.join(CSRC, "attn_fwd_fp8.hip"),
        os.path.join(CSRC, "attn_bwd.hip"),
        os.path.join(CSRC, "decode.hip"),
        os.path.join(CSRC, "rotary.hip"),
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="ring_attention_amd",
    version="0.1.0",
    packages=["ring_attention_amd", "ring_attention_amd.ops",
              "ring_attention_amd.models", "ring_attention_amd.parallel",
              "ring_attention_amd.utils"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
