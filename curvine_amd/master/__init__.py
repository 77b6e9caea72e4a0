from curvine_amd.master.server import Master  # noqa: F401
