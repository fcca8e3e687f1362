"""Forced-verifiable builds: the loaded native
binaries must carry the content hash of the source committed in the
tree. A stale .so (source edited, binary not rebuilt) or a foreign
binary fails these tests instead of silently serving old code."""

import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import __graft_entry__ as entry  # noqa: E402


def test_hip_so_matches_committed_source():
    if not os.path.exists(entry.HIP_SO):
        pytest.skip("HIP extension not built")
    from gofr_amd.ops import HipOps
    hip = HipOps()
    want = entry.src_hash(
        entry.HIP_SRC,
        "--offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared")
    loaded = hip.lib.gofr_src_hash().decode()
    assert loaded == want, (
        f"stale _gofr_hip.so (loaded {loaded[:12]}, source {want[:12]}): "
        "run __graft_entry__.build()")


def test_core_so_matches_committed_source():
    import sysconfig
    ext = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    core_so = os.path.join(REPO, "gofr_amd", "_core" + ext)
    if not os.path.exists(core_so):
        pytest.skip("_core extension not built")
    import gofr_amd._core as _core
    core_src = os.path.join(REPO, "gofr_amd", "native", "core",
                            "epoll_server.cpp")
    want = entry.src_hash(core_src, "-O2 -shared -fPIC -std=c++17")
    assert getattr(_core, "src_hash", "missing") == want, (
        "stale _core extension: run __graft_entry__.build()")


def test_build_rebuilds_on_content_change(tmp_path):
    """_needs_build is driven by content hash, not mtime."""
    so = tmp_path / "x.so"
    so.write_bytes(b"bin")
    assert entry._needs_build(str(so), "h1")  # no stamp yet
    (tmp_path / "x.so.srchash").write_text("h1")
    assert not entry._needs_build(str(so), "h1")
    assert entry._needs_build(str(so), "h2")  # content changed
