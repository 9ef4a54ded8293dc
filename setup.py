"""Build the srtb_amd native extension (in-tree, gfx950).

Two stages:
  1. hipcc --offload-arch=gfx950 compiles the pure-HIP kernel/engine TUs
     (csrc/kernels/*.hip, csrc/engine/engine.cpp) — no torch headers, fast.
  2. the torch extension TU (csrc/bind/module.cpp) is built by
     torch.utils.cpp_extension and linked against the stage-1 objects and
     hipfft.

Usage: python setup.py build_ext --inplace
The resulting srtb_amd/_C*.so travels with the repo snapshot to GPU boxes.
"""

import os
import subprocess
import sys

from setuptools import setup

ROOT = os.path.dirname(os.path.abspath(__file__))
OBJ_DIR = os.path.join(ROOT, "build", "hip_obj")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
HIPCC = os.path.join(ROCM, "bin", "hipcc")
GPU_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

HIP_SOURCES = [
    "csrc/kernels/unpack.hip",
    "csrc/kernels/spectrum.hip",
    "csrc/kernels/display.hip",
    "csrc/kernels/fft.hip",
    "csrc/engine/engine.cpp",
]

HIPCC_FLAGS = [
    f"--offload-arch={GPU_ARCH}",
    "-O3",
    "-std=c++17",
    "-fPIC",
    "-ffp-contract=fast",
]


def build_hip_objects():
    os.makedirs(OBJ_DIR, exist_ok=True)
    objects = []
    for src in HIP_SOURCES:
        src_path = os.path.join(ROOT, src)
        obj = os.path.join(OBJ_DIR, os.path.basename(src).rsplit(".", 1)[0] + ".o")
        objects.append(obj)
        deps = [src_path,
                os.path.join(ROOT, "csrc/include/srtb_kernels.h"),
                os.path.join(ROOT, "csrc/kernels/common.h"),
                os.path.join(ROOT, "csrc/engine/engine.h"),
                os.path.join(ROOT, "csrc/fft/fft_plans.h"),
                os.path.join(ROOT, "csrc/fft/native_fft.h")]
        if os.path.exists(obj) and all(
                os.path.getmtime(obj) >= os.path.getmtime(d) for d in deps
                if os.path.exists(d)):
            continue
        cmd = [HIPCC, *HIPCC_FLAGS, "-x", "hip", "-c", src_path, "-o", obj,
               f"-I{ROOT}/csrc/include"]
        print("+", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)
    return objects


APP_BINARIES = {
    "srtb-backend": "csrc/app/srtb_backend.cpp",
    "srtb-correlator": "csrc/app/correlator.cpp",
    "srtb-baseband-receiver": "csrc/app/baseband_receiver.cpp",
    "srtb-pipe-test": "csrc/app/pipe_test.cpp",
}


def build_apps(objects):
    """Link the native executables (reference src/*.cpp equivalents)."""
    bindir = os.path.join(ROOT, "bin")
    os.makedirs(bindir, exist_ok=True)
    for name, src in APP_BINARIES.items():
        out = os.path.join(bindir, name)
        src_path = os.path.join(ROOT, src)
        deps = [src_path] + objects + [
            os.path.join(ROOT, p) for p in (
                "csrc/app/config.h", "csrc/app/runtime.h", "csrc/app/comm.h",
                "csrc/app/pipe.h",
                "csrc/app/udp_receiver.h", "csrc/app/writers.h",
                "csrc/engine/engine.h", "csrc/fft/native_fft.h")]
        if os.path.exists(out) and all(
                os.path.getmtime(out) >= os.path.getmtime(d) for d in deps
                if os.path.exists(d)):
            continue
        cmd = [HIPCC, *HIPCC_FLAGS, "-x", "hip", src_path, "-x", "none", *objects,
               f"-I{ROOT}/csrc/include", f"-L{ROCM}/lib", "-lhipfft",
               "-lroctx64", "-lrccl", "-o", out]
        print("+", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)


def main():
    objects = build_hip_objects()
    build_apps(objects)

    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    ext = CUDAExtension(
        name="srtb_amd._C",
        sources=["csrc/bind/module.cpp"],
        extra_objects=objects,
        include_dirs=[os.path.join(ROOT, "csrc/include")],
        libraries=["hipfft", "roctx64"],
        extra_compile_args={"cxx": ["-O2", "-std=c++17"], "nvcc": ["-O2"]},
    )

    setup(
        name="srtb_amd",
        version="0.1.0",
        packages=["srtb_amd"],
        ext_modules=[ext],
        cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
    )


if __name__ == "__main__":
    main()
