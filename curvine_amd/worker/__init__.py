from curvine_amd.worker.server import Worker  # noqa: F401
from curvine_amd.worker.block_store import BlockStore  # noqa: F401
