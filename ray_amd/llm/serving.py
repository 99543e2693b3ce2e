"""Serve integration: OpenAI-ish completions app over LLMEngine
(reference: ray.serve.llm build_openai_app, llm/_internal/serve/)."""
from __future__ import annotations

from typing import Optional

from ray_amd import serve

from .engine import LLMConfig, LLMEngine


@serve.deployment
class LLMServer:
    def __init__(self, config: Optional[dict] = None):
        cfg = LLMConfig(**(config or {}))
        self.engine = LLMEngine(cfg)

    def generate(self, prompt_ids, max_new_tokens: int = 32,
                 temperature: float = 0.0):
        return self.engine.generate(prompt_ids, max_new_tokens, temperature)

    def stats(self):
        return dict(self.engine.stats)

    async def __call__(self, request):
        body = await request.json()
        prompt_ids = body.get("prompt_ids") or [
            hash(w) % 50000 for w in str(body.get("prompt", "")).split()
        ]
        r = self.engine.generate(
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
