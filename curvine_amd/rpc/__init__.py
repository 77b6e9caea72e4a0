from curvine_amd.rpc.codes import RpcCode  # noqa: F401
from curvine_amd.rpc.message import Message, Status, MAX_DATA_SIZE  # noqa: F401
from curvine_amd.rpc.server import RpcServer, HandlerService  # noqa: F401
from curvine_amd.rpc.client import RpcClient, ClientFactory, ClusterConnector  # noqa: F401
