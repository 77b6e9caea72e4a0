from curvine_amd.web.server import WebServer  # noqa: F401
