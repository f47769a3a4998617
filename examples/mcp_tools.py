"""MCP tool server (JSON-RPC 2.0 over POST /mcp)."""

from beta9_amd.integrations import MCPServer

srv = MCPServer(name="calc-tools", cpu=1000, memory=1024)


@srv.tool
def add(a: int, b: int) -> int:
    """Add two numbers."""
    return a + b


@srv.tool
def word_count(text: str) -> dict:
    """Count words in a text."""
    return {"words": len(text.split())}


if __name__ == "__main__":
    print(srv.deploy("calc-tools"))
