"""ray_amd.llm — LLM serving on Serve (reference: python/ray/llm/,
build_openai_app + vLLM engine integration). MI355X-native: the engine
is our own bf16 Llama with hipGraph-captured decode
(models/llama.py GraphedDecoder) — no external inference engine.
"""
from .engine import (  # noqa: F401
    ContinuousBatchingEngine,
    LLMConfig,
    LLMEngine,
)
from .serving import LLMServer, build_llm_deployment, build_openai_app  # noqa: F401
