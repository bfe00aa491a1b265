from .client import MCPStdioClient
from .manager import MCPManager

__all__ = ["MCPStdioClient", "MCPManager"]
