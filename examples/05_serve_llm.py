"""Serve an LLM with hipGraph-captured decode (tiny random-init model;
swap model_id for llama3-8b on an MI355X)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray_amd as ray
from ray_amd import serve
from ray_amd.llm import build_llm_deployment

ray.init()
app = build_llm_deployment(
    {"model_id": "llama-tiny", "max_seq_len": 128, "use_hip_graph": False}
)
handle = serve.run(app, name="llm", http=False)
out = handle.generate.remote([1, 2, 3, 4], max_new_tokens=16).result()
print("generated token ids:", out["token_ids"])
print(f"decode: {out['decode_tok_s']:.0f} tok/s")
serve.shutdown()
ray.shutdown()
