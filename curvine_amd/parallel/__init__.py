from curvine_amd.parallel.distributor import BlockDistributor  # noqa: F401
