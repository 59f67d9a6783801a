from .server import McpServer, run_stdio_server

__all__ = ["McpServer", "run_stdio_server"]
