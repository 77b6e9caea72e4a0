from curvine_amd.unified.unified_fs import UnifiedFileSystem  # noqa: F401
