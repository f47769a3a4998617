"""First-party serving integrations (reference:
sdk/src/beta9/abstractions/integrations/{vllm.py,fastmcp.py}).

The reference wraps third-party engines (vLLM, FastMCP) as preconfigured
ASGI stubs. Offline-first MI355X equivalents:

- ``llm``: an OpenAI-compatible completions server backed by
  transformers-on-ROCm (bf16, HIP) with iteration-level batched decoding.
- ``mcp``: a dependency-free MCP (JSON-RPC 2.0) tool server.
"""

from .llm import LLM, LLMArgs  # noqa: F401
from .mcp import MCPServer  # noqa: F401
