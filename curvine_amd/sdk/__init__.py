from curvine_amd.sdk.fsspec_fs import CurvineFileSystemSpec  # noqa: F401
from curvine_amd.sdk.torch_io import read_into_tensor, CurvineTensorReader  # noqa: F401
