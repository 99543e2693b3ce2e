"""Serve integration: OpenAI-ish completions app over LLMEngine
(reference: ray.serve.llm build_openai_app, llm/_internal/serve/)."""
from __future__ import annotations

from typing import Optional

from ray_amd import serve

from .engine import ContinuousBatchingEngine, LLMConfig, LLMEngine


@serve.deployment
class LLMServer:
    """`batching="continuous"` shares a batched KV cache across the
    replica's concurrent requests (reference: the vLLM engine loop
    behind serve.llm); default is the single-request hipGraph engine."""

    def __init__(self, config: Optional[dict] = None):
        config = dict(config or {})
        self._batching = config.pop("batching", "none")
        cfg = LLMConfig(**config)
        if self._batching == "continuous":
            self.engine = ContinuousBatchingEngine(cfg)
            self._futures = {}
            self._pump_task = None
        else:
            self.engine = LLMEngine(cfg)

    async def _pump(self):
        import asyncio

        loop = asyncio.get_running_loop()
        while self.engine.has_work():
            finished = await loop.run_in_executor(None, self.engine.step)
            for f in finished:
                fut = self._futures.pop(f["id"], None)
                if fut is not None and not fut.done():
                    fut.set_result(f["token_ids"])
            await asyncio.sleep(0)
        self._pump_task = None

    async def _generate_cb(self, prompt_ids, max_new_tokens, temperature):
        import asyncio

        rid = self.engine.submit(prompt_ids, max_new_tokens, temperature)
        fut = asyncio.get_running_loop().create_future()
        self._futures[rid] = fut
        if self._pump_task is None or self._pump_task.done():
            self._pump_task = asyncio.ensure_future(self._pump())
        token_ids = await fut
        return {"token_ids": token_ids, "decode_tok_s": 0.0}

    async def generate(self, prompt_ids, max_new_tokens: int = 32,
                       temperature: float = 0.0):
        if self._batching == "continuous":
            return await self._generate_cb(
                prompt_ids, max_new_tokens, temperature
            )
        return self.engine.generate(prompt_ids, max_new_tokens, temperature)

    def stats(self):
        return dict(self.engine.stats)

    async def __call__(self, request):
        body = await request.json()
        prompt_ids = body.get("prompt_ids") or [
            hash(w) % 50000 for w in str(body.get("prompt", "")).split()
        ]
        r = await self.generate(
            prompt_ids,
            int(body.get("max_tokens", 32)),
            float(body.get("temperature", 0.0)),
        )
        return {
            "id": "cmpl-ray-amd",
            "object": "text_completion",
            "choices": [
                {"index": 0, "token_ids": r["token_ids"],
                 "finish_reason": "length"}
            ],
            "usage": {
                "prompt_tokens": len(prompt_ids),
                "completion_tokens": len(r["token_ids"]),
            },
            "decode_tok_s": r["decode_tok_s"],
        }


def build_llm_deployment(config: Optional[dict] = None, *,
                         num_replicas: int = 1,
                         ray_actor_options: Optional[dict] = None):
    opts = dict(num_replicas=num_replicas)
    if ray_actor_options:
        opts["ray_actor_options"] = ray_actor_options
    elif config and config.get("use_gpu"):
        opts["ray_actor_options"] = {"num_gpus": 1}
    config = {k: v for k, v in (config or {}).items() if k != "use_gpu"}
    return LLMServer.options(**opts).bind(config)


def build_openai_app(config: Optional[dict] = None, **kwargs):
    return build_llm_deployment(config, **kwargs)
