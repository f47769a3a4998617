"""OpenAI-compatible LLM serving integration.

The reference's ``integrations/vllm.py`` deploys a vLLM OpenAI server as an
ASGI stub (reference: sdk/src/beta9/abstractions/integrations/vllm.py:297-477
— VLLMArgs + VLLM(ASGI) building the app and deploy/serve plumbing). The
MI355X-native equivalent serves the same API surface (/v1/models,
/v1/completions, /v1/chat/completions, /health) from a transformers
CausalLM running in bf16 on ROCm, with iteration-level batched greedy/
sampling decode: pending requests join the active batch at step-group
boundaries and finished sequences are masked out, so one forward pass per
step serves every in-flight request (the continuous-batching idea sized
for a single 288 GB MI355X).

Offline-first: ``model`` may be a local HF directory (loaded with
from_pretrained) or a synthetic spec — when the path does not exist a
random-init LlamaConfig model of the requested dims is built and a
byte-level tokenizer is used, so everything runs hermetically with no
network or weight files.
"""

import asyncio
import os
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional


@dataclass
class LLMArgs:
    """Engine knobs (reference vllm.py:VLLMArgs subset that is meaningful
    for the in-process engine)."""

    model: str = "synthetic-llama"
    dtype: str = "bfloat16"
    max_model_len: int = 2048
    max_num_seqs: int = 32  # decode batch width
    tensor_parallel_size: int = 1
    seed: int = 0
    # synthetic model dims (used when `model` is not a local directory)
    hidden_size: int = 1024
    intermediate_size: int = 2816
    num_hidden_layers: int = 8
    num_attention_heads: int = 16
    vocab_size: int = 512


class ByteTokenizer:
    """Dependency-free byte-level tokenizer for synthetic/offline mode."""

    bos_token_id = 256
    eos_token_id = 257
    vocab_size = 258

    def encode(self, text: str) -> List[int]:
        return [self.bos_token_id] + list(text.encode("utf-8", "replace"))

    def decode(self, ids: List[int]) -> str:
        return bytes(i for i in ids if i < 256).decode("utf-8", "replace")


class _Request:
    def __init__(self, prompt_ids: List[int], max_tokens: int, temperature: float):
        self.prompt_ids = prompt_ids
        self.max_tokens = max_tokens
        self.temperature = temperature
        self.out_ids: List[int] = []
        self.done = asyncio.Event()
        self.created = time.time()


class LLMEngine:
    """Iteration-batched decode engine over a transformers CausalLM."""

    def __init__(self, args: LLMArgs):
        self.args = args
        self._queue: asyncio.Queue = asyncio.Queue()
        self._task: Optional[asyncio.Task] = None
        self._build()

    def _build(self) -> None:
        import torch

        torch.manual_seed(self.args.seed)
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        self.dtype = (
            torch.bfloat16 if self.args.dtype == "bfloat16" and self.device == "cuda" else torch.float32
        )
        if os.path.isdir(self.args.model):
            from transformers import AutoModelForCausalLM, AutoTokenizer

            self.tokenizer = AutoTokenizer.from_pretrained(self.args.model)
            self.model = AutoModelForCausalLM.from_pretrained(
                self.args.model, torch_dtype=self.dtype
            )
        else:
            from transformers import LlamaConfig, LlamaForCausalLM

            self.tokenizer = ByteTokenizer()
            cfg = LlamaConfig(
                vocab_size=max(self.args.vocab_size, ByteTokenizer.vocab_size),
                hidden_size=self.args.hidden_size,
                intermediate_size=self.args.intermediate_size,
                num_hidden_layers=self.args.num_hidden_layers,
                num_attention_heads=self.args.num_attention_heads,
                max_position_embeddings=self.args.max_model_len,
            )
            self.model = LlamaForCausalLM(cfg).to(self.dtype)
        self.model = self.model.to(self.device).eval()
        self.eos_id = getattr(self.tokenizer, "eos_token_id", None)

    async def start(self) -> None:
        if self._task is None:
            self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None

    async def generate(self, prompt: str, max_tokens: int = 16, temperature: float = 0.0) -> str:
        await self.start()  # idempotent: first request spins the decode loop
        ids = self.tokenizer.encode(prompt)[-self.args.max_model_len // 2 :]
        req = _Request(ids, max_tokens, temperature)
        await self._queue.put(req)
        await req.done.wait()
        return self.tokenizer.decode(req.out_ids)

    async def _loop(self) -> None:
        while True:
            batch = [await self._queue.get()]
            # join everything already pending, up to the batch width
            while len(batch) < self.args.max_num_seqs and not self._queue.empty():
                batch.append(self._queue.get_nowait())
            await asyncio.get_event_loop().run_in_executor(None, self._decode_group, batch)
            for r in batch:
                r.done.set()

    def _decode_group(self, batch: List[_Request]) -> None:
        """One step-group: batched greedy/sampling decode with KV cache;
        finished sequences are masked out of the sampling, every in-flight
        request shares each forward pass."""
        import torch

        with torch.inference_mode():
            maxlen = max(len(r.prompt_ids) for r in batch)
            pad = 0
            input_ids = torch.full((len(batch), maxlen), pad, dtype=torch.long)
            attn = torch.zeros((len(batch), maxlen), dtype=torch.long)
            for i, r in enumerate(batch):
                input_ids[i, maxlen - len(r.prompt_ids) :] = torch.tensor(r.prompt_ids)
                attn[i, maxlen - len(r.prompt_ids) :] = 1
            input_ids = input_ids.to(self.device)
            attn = attn.to(self.device)
            past = None
            steps = max(r.max_tokens for r in batch)
            alive = [True] * len(batch)
            cur = input_ids
            for _ in range(steps):
                out = self.model(
                    input_ids=cur, attention_mask=attn, past_key_values=past, use_cache=True
                )
                past = out.past_key_values
                logits = out.logits[:, -1, :].float()
                nxt = []
                for i, r in enumerate(batch):
                    if not alive[i]:
                        nxt.append(pad)
                        continue
                    if r.temperature > 0:
                        probs = torch.softmax(logits[i] / r.temperature, dim=-1)
                        tok = int(torch.multinomial(probs, 1).item())
                    else:
                        tok = int(logits[i].argmax().item())
                    r.out_ids.append(tok)
                    if (self.eos_id is not None and tok == self.eos_id) or len(
                        r.out_ids
                    ) >= r.max_tokens:
                        alive[i] = False
                    nxt.append(tok)
                if not any(alive):
                    break
                cur = torch.tensor(nxt, dtype=torch.long, device=self.device).unsqueeze(1)
                attn = torch.cat(
                    [attn, torch.ones((len(batch), 1), dtype=torch.long, device=self.device)], dim=1
                )


def build_openai_app(engine: LLMEngine):
    """FastAPI app exposing the OpenAI-compatible surface the reference's
    vLLM stub serves (vllm.py:391-430 init_app_state routes)."""
    from fastapi import FastAPI, Request

    app = FastAPI()

    @app.get("/health")
    async def health():
        return {"ok": True, "model": engine.args.model}

    @app.get("/v1/models")
    async def models():
        return {
            "object": "list",
            "data": [{"id": engine.args.model, "object": "model", "owned_by": "b9"}],
        }

    @app.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        text = await engine.generate(
            body.get("prompt", ""),
            max_tokens=int(body.get("max_tokens", 16)),
            temperature=float(body.get("temperature", 0.0)),
        )
        return {
            "id": f"cmpl-{uuid.uuid4().hex[:24]}",
            "object": "text_completion",
            "model": engine.args.model,
            "choices": [{"index": 0, "text": text, "finish_reason": "length"}],
        }

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        body = await request.json()
        prompt = "\n".join(
            f"{m.get('role', 'user')}: {m.get('content', '')}" for m in body.get("messages", [])
        )
        text = await engine.generate(
            prompt + "\nassistant:",
            max_tokens=int(body.get("max_tokens", 16)),
            temperature=float(body.get("temperature", 0.0)),
        )
        return {
            "id": f"chatcmpl-{uuid.uuid4().hex[:24]}",
            "object": "chat.completion",
            "model": engine.args.model,
            "choices": [
                {
                    "index": 0,
                    "message": {"role": "assistant", "content": text},
                    "finish_reason": "stop",
                }
            ],
        }

    return app


class LLM:
    """Deployable LLM endpoint (reference vllm.py:297 class VLLM(ASGI)):
    wraps the OpenAI app as an asgi stub via the SDK.

    Usage::

        from beta9_amd.integrations import LLM, LLMArgs
        llm = LLM(args=LLMArgs(model="synthetic-llama"), gpu="MI355X")
        llm.deploy("my-llm")   # serves /v1/completions etc.
    """

    def __init__(self, args: Optional[LLMArgs] = None, **endpoint_kwargs):
        self.args = args or LLMArgs()
        self.endpoint_kwargs = endpoint_kwargs

    def app(self):
        """Build the ASGI app in-process (used by the runner handler)."""
        engine = LLMEngine(self.args)
        return build_openai_app(engine)

    def _abstraction(self):
        from ..sdk import Endpoint

        ep = Endpoint(func=None, **self.endpoint_kwargs)
        ep.stub_kind = "asgi"
        return ep

    def deploy(self, name: str) -> Dict[str, Any]:
        ab = self._abstraction()
        ab.func = self.app
        return ab.deploy(name)

    def serve(self, timeout: float = 600.0) -> str:
        ab = self._abstraction()
        ab.func = self.app
        return ab.serve(timeout=timeout)
