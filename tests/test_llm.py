"""LLM engine/serving tests. CPU: eager decode; GPU: hipGraph decode
equivalence + Serve deployment (north-star config 5)."""
import pytest
import torch

import ray_amd as ray
from ray_amd.llm import LLMConfig, LLMEngine


def test_engine_generate_cpu():
    eng = LLMEngine(LLMConfig(model_id="llama-tiny", max_seq_len=128,
                              use_hip_graph=False))
    r = eng.generate([1, 2, 3, 4], max_new_tokens=8)
    assert len(r["token_ids"]) == 8
    assert all(0 <= t < 512 for t in r["token_ids"])
    # greedy decode is deterministic
    r2 = eng.generate([1, 2, 3, 4], max_new_tokens=8)
    assert r["token_ids"] == r2["token_ids"]


def test_engine_temperature_sampling_cpu():
    eng = LLMEngine(LLMConfig(model_id="llama-tiny", max_seq_len=64,
                              use_hip_graph=False))
    torch.manual_seed(0)
    r = eng.generate([5, 6], max_new_tokens=16, temperature=1.0)
    assert len(r["token_ids"]) == 16


def test_llm_serve_deployment_cpu(ray_start_regular):
    from ray_amd import serve
    from ray_amd.llm import build_llm_deployment

    app = build_llm_deployment(
        {"model_id": "llama-tiny", "max_seq_len": 64, "use_hip_graph": False}
    )
    h = serve.run(app, name="llm", http=False)
    out = h.generate.remote([1, 2, 3], 4).result(timeout_s=120)
    assert len(out["token_ids"]) == 4
    serve.shutdown()


@pytest.mark.gpu
def test_hipgraph_decode_matches_eager():
    """Graph-captured decode must produce identical greedy tokens to the
    eager decode path."""
    assert torch.cuda.is_available()
    eager = LLMEngine(LLMConfig(model_id="llama-tiny", max_seq_len=128,
                                use_hip_graph=False))
    graphed = LLMEngine(LLMConfig(model_id="llama-tiny", max_seq_len=128,
                                  use_hip_graph=True))
    assert graphed.decoder is not None and graphed.decoder.graph is not None
    prompt = [7, 11, 13, 17, 19]
    r_e = eager.generate(prompt, max_new_tokens=16)
    r_g = graphed.generate(prompt, max_new_tokens=16)
    assert r_e["token_ids"] == r_g["token_ids"], (
        r_e["token_ids"], r_g["token_ids"]
    )


@pytest.mark.gpu
def test_hipgraph_decode_speedup():
    eager = LLMEngine(LLMConfig(model_id="llama-tiny", max_seq_len=256,
                                use_hip_graph=False))
    graphed = LLMEngine(LLMConfig(model_id="llama-tiny", max_seq_len=256,
                                  use_hip_graph=True))
    prompt = list(range(8))
    eager.generate(prompt, 16)
    graphed.generate(prompt, 16)
    r_e = eager.generate(prompt, 128)
    r_g = graphed.generate(prompt, 128)
    print(f"eager {r_e['decode_tok_s']:.0f} tok/s vs graph {r_g['decode_tok_s']:.0f} tok/s")
    # launch-bound tiny model: graph replay must be significantly faster
    assert r_g["decode_tok_s"] > r_e["decode_tok_s"] * 1.5


def test_continuous_batching_matches_sequential():
    """Continuous batching (shared batched KV cache, mid-flight
    admission) produces the same greedy tokens as one-at-a-time
    decoding."""
    from ray_amd.llm import ContinuousBatchingEngine, LLMConfig, LLMEngine

    cfg = LLMConfig(model_id="llama-tiny", max_seq_len=96,
                    max_batch_size=2, use_hip_graph=False)
    seq_engine = LLMEngine(cfg)
    prompts = [
        list(range(5, 17)),          # 12 tokens
        list(range(100, 104)),       # 4 tokens
        list(range(40, 61)),         # 21 tokens
    ]
    budgets = [8, 14, 6]
    expected = [
        seq_engine.generate(p, max_new_tokens=n)["token_ids"]
        for p, n in zip(prompts, budgets)
    ]

    cb = ContinuousBatchingEngine(cfg)  # same seed -> same weights
    ids = [cb.submit(p, max_new_tokens=n)
           for p, n in zip(prompts, budgets)]
    results = cb.run_until_complete()
    assert set(results) == set(ids)
    for rid, exp in zip(ids, expected):
        assert results[rid] == exp, rid
    # 3 requests through 2 slots: queuing + mid-flight admission happened
    assert cb.stats["requests"] == 3
    assert cb.stats["decode_steps"] < sum(budgets)  # batched, not serial


def test_llm_server_continuous_batching(ray_start_regular):
    """Concurrent requests to one replica share the batched decode loop."""
    from ray_amd import serve
    from ray_amd.llm.serving import build_llm_deployment

    app = build_llm_deployment(
        {"model_id": "llama-tiny", "max_seq_len": 96,
         "max_batch_size": 2, "batching": "continuous"}
    )
    h = serve.run(app, name="llm_cb", http=False)
    try:
        resps = [
            h.generate.remote(list(range(i + 3, i + 10)), 6)
            for i in range(4)
        ]
        outs = [r.result(timeout_s=120) for r in resps]
        assert all(len(o["token_ids"]) == 6 for o in outs)
        st = h.stats.remote().result(timeout_s=30)
        assert st["requests"] == 4
    finally:
        serve.shutdown()


@pytest.mark.gpu
def test_batched_decoder_gpu_logits():
    """BatchedDecoder (per-slot positions, shared KV cache) matches the
    plain KVCache decode path logit-for-logit under teacher forcing —
    greedy ROLLOUTS would amplify bf16 tie-breaks, logits don't."""
    import torch

    from ray_amd.models.llama import (
        CONFIGS,
        BatchedDecoder,
        KVCache,
        LlamaModel,
    )

    dev = torch.device("cuda:0")
    cfg = CONFIGS["llama-tiny"]
    torch.manual_seed(0)
    m = LlamaModel(cfg, dtype=torch.bfloat16).to(dev).eval()
    m.cosT = m.cosT.to(dev)
    m.sinT = m.sinT.to(dev)
    hd = cfg.hidden_size // cfg.num_heads

    prompts = [list(range(5, 15)), list(range(50, 58))]
    forced = [list(range(300, 308)), list(range(400, 408))]

    # reference: per-sequence KVCache decode
    ref_logits = []
    with torch.no_grad():
        for p, f in zip(prompts, forced):
            caches = [KVCache(1, 96, cfg.num_kv_heads, hd, dev,
                              torch.bfloat16)
                      for _ in range(cfg.num_layers)]
            toks = torch.tensor([p], device=dev)
            m(toks, kv_caches=caches, pos0=0)
            seq_logits = []
            pos = len(p)
            for t in f:
                lg = m(torch.tensor([[t]], device=dev),
                       kv_caches=caches, pos0=pos)[:, -1]
                seq_logits.append(lg[0].float())
                pos += 1
            ref_logits.append(seq_logits)

    # batched decoder: both sequences decode together at their own pos
    bd = BatchedDecoder(m, 2, 96, dev)
    with torch.no_grad():
        for slot, p in enumerate(prompts):
            bd.prefill_slot(slot, torch.tensor(p, device=dev))
        pos = torch.tensor([len(p) for p in prompts], device=dev)
        for step in range(8):
            for slot in range(2):
                bd.set_slot_len(slot, int(pos[slot]) + 1)
            toks = torch.tensor([forced[0][step], forced[1][step]],
                                device=dev)
            logits = bd.decode(toks, pos)
            for slot in range(2):
                ref = ref_logits[slot][step]
                got = logits[slot].float()
                rel = (got - ref).norm() / (ref.norm() + 1e-6)
                assert rel < 0.05, (slot, step, float(rel))
            pos += 1
