"""Build the native extensions in-tree:

  python setup.py build_ext --inplace

* metaflow_amd/ops/_mfx_hip.so  — gfx950 HIP kernel library (torch ext)
* metaflow_amd/ops/_mfx_cas.so  — C++ CAS engine (threaded SHA-256 + IO)
* metaflow_amd/ops/_mfx_io.so   — standalone pread pool (NO torch link)

gfx950-only by design (PYTORCH_ROCM_ARCH=gfx950); the .so files travel with
the repo snapshot to the GPU box.
"""

import os

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", str(os.cpu_count() or 8))

from setuptools import Extension, setup  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension, CppExtension  # noqa: E402

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "metaflow_amd", "ops", "csrc")

# torch's hipcc path emits no depfiles, so edits to #included .hip files
# never trigger a rebuild of the single TU. Track deps ourselves: if any
# csrc file is newer than the built object, drop the object + the hipified
# intermediate so ninja recompiles.
def _force_rebuild_if_stale():
    import glob

    objs = glob.glob(os.path.join(ROOT, "build", "temp*", "metaflow_amd",
                                  "ops", "csrc", "mfx_hip*.o"))
    objs += glob.glob(os.path.join(ROOT, "build", "temp*", "metaflow_amd",
                                   "ops", "csrc", "gemm_lt*.o"))
    if not objs:
        return
    newest_src = max(
        os.path.getmtime(p)
        for p in glob.glob(os.path.join(CSRC, "*"))
        if not p.endswith("_hip.hip"))
    for obj in objs:
        if os.path.getmtime(obj) < newest_src:
            os.unlink(obj)
            for gen_name in ("mfx_hip_hip.hip", "gemm_lt_hip.hip"):
                gen = os.path.join(CSRC, gen_name)
                if os.path.exists(gen):
                    os.unlink(gen)
            # force re-hipify by bumping the source mtimes
            os.utime(os.path.join(CSRC, "mfx_hip.hip"))
            os.utime(os.path.join(CSRC, "gemm_lt.hip"))


_force_rebuild_if_stale()

ext_modules = [
    CUDAExtension(
        name="metaflow_amd.ops._mfx_hip",
        sources=[os.path.join(CSRC, "mfx_hip.hip")],
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17"],
        },
    ),
    CUDAExtension(
        name="metaflow_amd.ops._mfx_gemm",
        sources=[os.path.join(CSRC, "gemm_lt.hip")],
        libraries=["hipblaslt"],
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17"],
        },
    ),
    CppExtension(
        name="metaflow_amd.ops._mfx_cas",
        sources=[os.path.join(CSRC, "cas_engine.cpp")],
        extra_compile_args={"cxx": ["-O3", "-std=c++17", "-pthread"]},
    ),
    # plain setuptools Extension ON PURPOSE: _mfx_io must not link
    # torch so torch-less task subprocesses can import it in ~10 ms
    # (ops/cas_native.py explains the 10x regression that rule avoids)
    Extension(
        name="metaflow_amd.ops._mfx_io",
        sources=[os.path.join(CSRC, "io_engine.cpp")],
        extra_compile_args=["-O3", "-std=c++17", "-pthread"],
        extra_link_args=["-pthread"],
    ),
]

setup(
    name="metaflow_amd",
    version="0.1.0",
    packages=["metaflow_amd"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension},
)
