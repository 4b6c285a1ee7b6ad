"""In-tree build of the horizonml_amd gfx950 HIP extension.

`python setup.py build_ext --inplace` produces
``horizonml_amd/ops/_C.cpython-*.so``:
  * every ``ops/csrc/*.hip`` is compiled by hipcc with
    ``--offload-arch=gfx950`` (cross-compiles fine on GPU-less hosts),
  * ``bind.cpp`` is compiled/linked through torch's CppExtension so the ABI
    matches the installed PyTorch-ROCm.
No hipify, no CUDA paths — the kernels are native CDNA4 HIP.
"""
import os
import subprocess

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "horizonml_amd", "ops", "csrc")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

HIP_SOURCES = ["conv_kernels.hip", "elem_kernels.hip"]
HIPCC_FLAGS = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
               "-ffast-math"]


def compile_hip_objects():
    objs = []
    for src in HIP_SOURCES:
        src_path = os.path.join(CSRC, src)
        obj_path = src_path.replace(".hip", ".o")
        if (not os.path.exists(obj_path)
                or os.path.getmtime(obj_path) < os.path.getmtime(src_path)
                or os.path.getmtime(obj_path) < os.path.getmtime(
                    os.path.join(CSRC, "common.h"))):
            cmd = [os.path.join(ROCM, "bin", "hipcc"), "-c", src_path,
                   "-o", obj_path] + HIPCC_FLAGS
            print("[hipcc]", " ".join(cmd), flush=True)
            subprocess.check_call(cmd)
        objs.append(obj_path)
    return objs


def main():
    objs = compile_hip_objects()
    ext = CppExtension(
        "horizonml_amd.ops._C",
        [os.path.join(CSRC, "bind.cpp")],
        include_dirs=[os.path.join(ROCM, "include")],
        library_dirs=[os.path.join(ROCM, "lib")],
        libraries=["amdhip64"],
        extra_objects=objs,
        extra_compile_args=["-O2"],
    )
    setup(
        name="horizonml_amd",
        version="0.1.0",
        packages=["horizonml_amd"],
        ext_modules=[ext],
        cmdclass={"build_ext": BuildExtension},
    )


if __name__ == "__main__":
    main()
