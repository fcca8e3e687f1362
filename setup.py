"""In-tree build of the native components.

  python setup.py build_ext --inplace     (or: python setup.py build_hip)

Builds:
  gofr_amd/_gofr_hip.so  — gfx950 CDNA4 kernels (hipcc, no GPU needed)
  gofr_amd/_core.*.so    — C++ epoll ingress (pybind11)
"""

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.abspath(__file__))


def build_hip():
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    src = os.path.join(REPO, "gofr_amd", "native", "hip", "gofr_kernels.hip")
    out = os.path.join(REPO, "gofr_amd", "_gofr_hip.so")
    cmd = [hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
           "-shared", src, "-o", out]
    print(" ".join(cmd))
    subprocess.run(cmd, check=True)


def build_core():
    import pybind11
    ext = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    src = os.path.join(REPO, "gofr_amd", "native", "core",
                       "epoll_server.cpp")
    out = os.path.join(REPO, "gofr_amd", "_core" + ext)
    cmd = ["g++", "-O2", "-shared", "-fPIC", "-std=c++17",
           f"-I{pybind11.get_include()}",
           f"-I{sysconfig.get_path('include')}",
           src, "-o", out]
    print(" ".join(cmd))
    subprocess.run(cmd, check=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] in ("build_ext", "build_hip",
                                             "build"):
        build_hip()
        build_core()
    else:
        print(__doc__)
