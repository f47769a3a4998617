"""OpenAI-compatible LLM endpoint on MI355X (synthetic offline model).

    b9 gateway &          # terminal 1
    python examples/llm_server.py   # deploys; then:
    curl -H "authorization: Bearer $TOKEN" \
         -X POST http://127.0.0.1:1994/asgi/id/<stub>/v1/completions \
         -d '{"prompt": "hello", "max_tokens": 16}'
"""

from beta9_amd.integrations import LLM, LLMArgs

llm = LLM(
    args=LLMArgs(model="synthetic-llama", hidden_size=1024, num_hidden_layers=8,
                 max_num_seqs=32),
    gpu="MI355X",
    cpu=4000,
    memory=16384,
)

if __name__ == "__main__":
    print(llm.deploy("my-llm"))
