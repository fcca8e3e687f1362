"""In-tree build of the native components.

  python setup.py build_ext --inplace     (or: python setup.py build_hip)

Builds (through __graft_entry__.build(), the single source of truth —
binaries carry the content hash of their source, asserted by
tests/test_build_hash.py):
  gofr_amd/_gofr_hip.so  — gfx950 CDNA4 kernels (hipcc, no GPU needed)
  gofr_amd/_core.*.so    — C++ epoll ingress (pybind11)
"""

import sys

if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] in ("build_ext", "build_hip",
                                             "build"):
        import __graft_entry__
        __graft_entry__.build()
    else:
        print(__doc__)
