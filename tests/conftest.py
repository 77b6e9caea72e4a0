import os
import subprocess
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# FUSE tests run the daemon IN the pytest process (convenient for asserts).
# glibc posix_spawn suspends the vfork'ing thread WITH the GIL held while the
# child execs; the child's exec closes inherited CLOEXEC fds on the fuse
# mount, which sends FLUSH to our daemon — whose channel thread then can't
# take the GIL: a deadlock triangle.  Plain fork avoids the suspension.
# (Production deployments run the daemon as its own process: cv-fuse.)
subprocess._USE_POSIX_SPAWN = False
subprocess._USE_VFORK = False   # fork_exec also vforks (bpo-35823)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X via gpurun)")


def pytest_collection_modifyitems(config, items):
    try:
        from curvine_amd.native import gpu_available
        has_gpu = gpu_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
