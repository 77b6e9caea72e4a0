from curvine_amd.client.filesystem import CurvineFileSystem  # noqa: F401
