"""In-tree build of the MI355X HIP extension.

`python setup.py build_ext --inplace` produces
mmlspark_amd/ops/_hip_ops*.so next to the package sources so the .so travels
with the repo snapshot to GPU boxes.  Kernels (*.hip) are compiled directly
with hipcc --offload-arch=gfx950 (no hipify, no CUDA path); the torch
binding (ext.cpp) is built as a normal torch CppExtension and linked against
the kernel objects + libamdhip64.
"""
import os
import subprocess

from setuptools import setup
from torch.utils import cpp_extension

ROOT = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(ROOT, "mmlspark_amd", "ops", "hip")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(";")[0]

KERNEL_SOURCES = ["gbdt_kernels.hip", "vw_kernels.hip"]


def compile_hip_kernels():
    objs = []
    for src in KERNEL_SOURCES:
        src_path = os.path.join(HIP_DIR, src)
        obj_path = src_path.replace(".hip", ".o")
        if (not os.path.exists(obj_path)
                or os.path.getmtime(obj_path) < os.path.getmtime(src_path)):
            cmd = [os.path.join(ROCM, "bin", "hipcc"),
                   f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
                   "-c", src_path, "-o", obj_path]
            print("[hipcc]", " ".join(cmd), flush=True)
            subprocess.check_call(cmd)
        objs.append(obj_path)
    return objs


extra_objects = compile_hip_kernels()

# setuptools does not track extra_objects as dependencies: if only a .hip
# kernel changed, the freshly compiled .o would NOT be relinked into the .so.
# Force a relink by touching the binding sources when any kernel .o is newer.
for _binding in ("ext.cpp", "gbdt_grower.cpp"):
    _bp = os.path.join(HIP_DIR, _binding)
    if any(os.path.getmtime(o) > os.path.getmtime(_bp) for o in extra_objects):
        os.utime(_bp, None)

setup(
    name="mmlspark_amd_hip_ops",
    ext_modules=[
        cpp_extension.CppExtension(
            name="mmlspark_amd.ops._hip_ops",
            sources=[os.path.join(HIP_DIR, "ext.cpp")],
            extra_objects=extra_objects,
            extra_compile_args=["-O2", "-D__HIP_PLATFORM_AMD__=1",
                                f"-I{ROCM}/include"],
            extra_link_args=[f"-L{ROCM}/lib", "-lamdhip64"],
        ),
        cpp_extension.CppExtension(
            name="mmlspark_amd.ops._hip_grower",
            sources=[os.path.join(HIP_DIR, "gbdt_grower.cpp")],
            extra_objects=[o for o in extra_objects if "gbdt" in o],
            extra_compile_args=["-O2", "-D__HIP_PLATFORM_AMD__=1",
                                f"-I{ROCM}/include"],
            extra_link_args=[f"-L{ROCM}/lib", "-lamdhip64"],
        ),
        cpp_extension.CppExtension(
            name="mmlspark_amd.io_http._jpeg_native",
            sources=[os.path.join(ROOT, "mmlspark_amd", "io_http",
                                  "jpeg_native.cpp")],
            extra_compile_args=["-O3"],
        ),
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(
        use_ninja=False)},
)
