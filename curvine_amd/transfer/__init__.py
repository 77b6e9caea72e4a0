from curvine_amd.transfer.service import TransferService  # noqa: F401
