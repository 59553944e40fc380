"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _hip_ops .so lands inside mi355x_autoscaler/ so it travels
with the repo snapshot to GPU boxes (no JIT cache dependence).
"""

import os

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))


def _invalidate_stale_objects():
    """distutils tracks NO header dependencies for hipcc-compiled .hip
    sources: editing common.h / pipeline_device.h relinks stale objects
    silently (which once shipped kernels with two different MAXSEL
    widths in one .so). Delete any object older than the newest header
    so header edits force recompilation."""
    import glob

    hdrs = glob.glob(os.path.join(ROOT, "mi355x_autoscaler", "ops", "**", "*.h"),
                     recursive=True)
    if not hdrs:
        return
    newest = max(os.path.getmtime(h) for h in hdrs)
    for o in glob.glob(os.path.join(ROOT, "build", "**", "*.o"), recursive=True):
        if os.path.getmtime(o) < newest:
            os.remove(o)


_invalidate_stale_objects()

ext = CUDAExtension(
    name="mi355x_autoscaler._hip_ops",
    sources=[
        "mi355x_autoscaler/ops/csrc/bindings.cpp",
        "mi355x_autoscaler/ops/csrc/snapshot_codec.cpp",
        "mi355x_autoscaler/ops/csrc/orchestrator.cpp",
        "mi355x_autoscaler/ops/hip/classify.hip",
        "mi355x_autoscaler/ops/hip/node_free.hip",
        "mi355x_autoscaler/ops/hip/fit_predicates.hip",
        "mi355x_autoscaler/ops/hip/binpack.hip",
        "mi355x_autoscaler/ops/hip/drain_score.hip",
        "mi355x_autoscaler/ops/hip/segsum_mfma.hip",
        "mi355x_autoscaler/ops/hip/segment_hash.hip",
        "mi355x_autoscaler/ops/hip/mega_decide.hip",
        "mi355x_autoscaler/ops/hip/snapshot_diff.hip",
    ],
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"]
        + (["-fsanitize=address", "-fno-omit-frame-pointer"]
           if os.environ.get("MI355X_AS_ASAN") == "1" else []),
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"]
        + (["-fsanitize=address", "-fno-omit-frame-pointer"]
           if os.environ.get("MI355X_AS_ASAN") == "1" else []),
    },
)
# MI355X_AS_ASAN=1: host-side AddressSanitizer build of the extension
# (SURVEY §5 race-detection plan); run pytest with
# LD_PRELOAD=$(gcc -print-file-name=libasan.so).

setup(
    name="mi355x-autoscaler",
    version="0.1.0",
    description=("MI355X-native Kubernetes GPU-node autoscaler for "
                 "ACS-Engine agent pools (CDNA4 HIP scheduling simulation, "
                 "RCCL-sharded controller)"),
    # every subpackage (ops, snapshot, ctl, ...) must ship in a wheel
    # install, not just the source-tree build_ext --inplace path
    packages=find_packages(include=["mi355x_autoscaler", "mi355x_autoscaler.*"]),
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
