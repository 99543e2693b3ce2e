"""LLM engine: prefill + hipGraph decode over the ray_amd Llama.

There is no network/checkpoint access in this environment, so models
are random-init of the named architecture (same as bench.py); the
serving data path (prefill, KV cache, graph-captured decode loop,
sampling) is the real one.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from ray_amd.models.llama import (
    CONFIGS,
    BatchedDecoder,
    GraphedDecoder,
    KVCache,
    LlamaModel,
)


@dataclass
class LLMConfig:
    model_id: str = "llama-tiny"
    max_seq_len: int = 512
    max_batch_size: int = 1
    dtype: str = "bfloat16"
    use_hip_graph: bool = True
    temperature: float = 0.0


class LLMEngine:
    def __init__(self, config: LLMConfig):
        self.config = config
        cfg = CONFIGS[config.model_id]
        self.model_cfg = cfg
        self.device = (
            torch.device("cuda", 0)
            if torch.cuda.is_available()
            else torch.device("cpu")
        )
        dtype = (
            torch.bfloat16 if self.device.type == "cuda" else torch.float32
        )
        torch.manual_seed(0)
        self.model = LlamaModel(cfg, dtype=dtype).to(self.device).eval()
        self.model.cosT = self.model.cosT.to(self.device)
        self.model.sinT = self.model.sinT.to(self.device)
        self.max_T = min(config.max_seq_len, cfg.max_seq_len)
        self.decoder: Optional[GraphedDecoder] = None
        if self.device.type == "cuda" and config.use_hip_graph:
            self.decoder = GraphedDecoder(self.model, 1, self.max_T, self.device)
            self.decoder.capture()
        self.stats = {"requests": 0, "tokens_generated": 0, "decode_s": 0.0}

    def _sample(self, logits: torch.Tensor, temperature: float) -> int:
        if temperature <= 0:
            return int(logits.argmax(-1).item())
        probs = torch.softmax(logits.float() / temperature, dim=-1)
        return int(torch.multinomial(probs, 1).item())

    @torch.no_grad()
    def generate(self, prompt_ids: List[int], max_new_tokens: int = 32,
                 temperature: Optional[float] = None) -> dict:
        temperature = (
            self.config.temperature if temperature is None else temperature
        )
        cfg = self.model_cfg
        prompt_ids = [t % cfg.vocab_size for t in prompt_ids][: self.max_T - max_new_tokens - 1]
        toks = torch.tensor([prompt_ids], device=self.device)
        t0 = time.perf_counter()
        if self.decoder is not None:
            # prefill into the decoder's caches token-position-wise via
            # eager path, then graph decode
            caches = [
                _ViewCache(self.decoder.cache_k[i], self.decoder.cache_v[i])
                for i in range(cfg.num_layers)
            ]
            logits = self.model(toks, kv_caches=caches, pos0=0)[:, -1]
        else:
            caches = [
                KVCache(1, self.max_T, cfg.num_kv_heads,
                        cfg.hidden_size // cfg.num_heads, self.device,
                        self.model.dtype)
                for _ in range(cfg.num_layers)
            ]
            logits = self.model(toks, kv_caches=caches, pos0=0)[:, -1]
        prefill_s = time.perf_counter() - t0
        out = []
        pos = len(prompt_ids)
        t0 = time.perf_counter()
        for _ in range(max_new_tokens):
            nxt = self._sample(logits[0], temperature)
            out.append(nxt)
            if pos >= self.max_T - 1:
                break
            tok = torch.tensor([nxt], device=self.device)
            if self.decoder is not None:
                logits = self.decoder.decode(tok, pos)
            else:
                logits = self.model(
                    tok.view(1, 1), kv_caches=caches, pos0=pos
                )[:, -1]
            pos += 1
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        decode_s = time.perf_counter() - t0
        self.stats["requests"] += 1
        self.stats["tokens_generated"] += len(out)
        self.stats["decode_s"] += decode_s
        return {
            "token_ids": out,
            "prefill_s": prefill_s,
            "decode_s": decode_s,
            "decode_tok_s": len(out) / decode_s if decode_s > 0 else 0.0,
        }


class _ViewCache:
    """Adapter: lets the eager prefill path fill the GraphedDecoder's
    static cache buffers."""

    def __init__(self, k_buf, v_buf):
        self.k_buf = k_buf
        self.v_buf = v_buf
        self.len = 0

    def update(self, k, v, pos0):
        T = k.size(2)
        self.k_buf[:, :, pos0 : pos0 + T] = k.to(self.k_buf.dtype)
        self.v_buf[:, :, pos0 : pos0 + T] = v.to(self.v_buf.dtype)
        self.len = pos0 + T
        return self.k_buf[:, :, : self.len], self.v_buf[:, :, : self.len]


class ContinuousBatchingEngine:
    """Continuous batching (the vLLM-style loop the reference's serve
    LLM rides on): up to max_batch_size sequences decode together in a
    shared batched KV cache; finished sequences retire and queued
    requests are admitted mid-flight, so decode throughput does not
    drain between requests."""

    def __init__(self, config: LLMConfig):
        self.config = config
        cfg = CONFIGS[config.model_id]
        self.model_cfg = cfg
        self.device = (
            torch.device("cuda", 0)
            if torch.cuda.is_available()
            else torch.device("cpu")
        )
        dtype = (
            torch.bfloat16 if self.device.type == "cuda" else torch.float32
        )
        torch.manual_seed(0)
        self.model = LlamaModel(cfg, dtype=dtype).to(self.device).eval()
        self.model.cosT = self.model.cosT.to(self.device)
        self.model.sinT = self.model.sinT.to(self.device)
        self.max_T = min(config.max_seq_len, cfg.max_seq_len)
        self.B = max(1, config.max_batch_size)
        self.decoder = BatchedDecoder(self.model, self.B, self.max_T,
                                      self.device)
        self._queue: List[dict] = []          # pending requests
        self._slots: List[Optional[dict]] = [None] * self.B
        self._finished: List[dict] = []
        self._next_id = 0
        self._tok = torch.zeros(self.B, dtype=torch.long,
                                device=self.device)
        self._pos = torch.zeros(self.B, dtype=torch.long,
                                device=self.device)
        self.stats = {"requests": 0, "tokens_generated": 0,
                      "decode_steps": 0}

    def submit(self, prompt_ids: List[int], max_new_tokens: int = 32,
               temperature: Optional[float] = None) -> int:
        rid = self._next_id
        self._next_id += 1
        cfg = self.model_cfg
        prompt = [t % cfg.vocab_size for t in prompt_ids][
            : self.max_T - max_new_tokens - 1
        ]
        self._queue.append({
            "id": rid, "prompt": prompt,
            "max_new_tokens": max_new_tokens,
            "temperature": (self.config.temperature
                            if temperature is None else temperature),
        })
        self.stats["requests"] += 1
        return rid

    def _sample(self, logits: torch.Tensor, temperature: float) -> int:
        if temperature <= 0:
            return int(logits.argmax(-1).item())
        probs = torch.softmax(logits.float() / temperature, dim=-1)
        return int(torch.multinomial(probs, 1).item())

    def _admit(self):
        for slot in range(self.B):
            if self._slots[slot] is not None or not self._queue:
                continue
            req = self._queue.pop(0)
            toks = torch.tensor(req["prompt"], device=self.device)
            with torch.no_grad():
                logits = self.decoder.prefill_slot(slot, toks)
            first = self._sample(logits, req["temperature"])
            req.update({
                "out": [first],
                "pos": len(req["prompt"]),  # next decode position
                "slot": slot,
            })
            self._slots[slot] = req

    def _retire(self, slot: int):
        req = self._slots[slot]
        self._slots[slot] = None
        self._finished.append({
            "id": req["id"], "token_ids": req["out"],
        })
        self.stats["tokens_generated"] += len(req["out"])

    @torch.no_grad()
    def step(self) -> List[dict]:
        """Admit + one batched decode iteration; returns newly finished
        requests."""
        self._admit()
        # retire sequences that are already complete post-prefill
        for slot in range(self.B):
            r = self._slots[slot]
            if r is not None and (
                len(r["out"]) >= r["max_new_tokens"]
                or r["pos"] >= self.max_T - 1
            ):
                self._retire(slot)
        self._admit()
        active = [s for s in range(self.B) if self._slots[s] is not None]
        if not active:
            out, self._finished = self._finished, []
            return out
        for s in active:
            r = self._slots[s]
            self._tok[s] = r["out"][-1]
            self._pos[s] = r["pos"]
            self.decoder.set_slot_len(s, r["pos"] + 1)  # attend new tok
        logits = self.decoder.decode(self._tok, self._pos)
        for s in active:
            r = self._slots[s]
            nxt = self._sample(logits[s], r["temperature"])
            r["out"].append(nxt)
            r["pos"] += 1
            if (len(r["out"]) >= r["max_new_tokens"]
                    or r["pos"] >= self.max_T - 1):
                self._retire(s)
        self.stats["decode_steps"] += 1
        out, self._finished = self._finished, []
        return out

    def has_work(self) -> bool:
        return bool(self._queue) or any(
            s is not None for s in self._slots
        )

    def run_until_complete(self) -> dict:
        """Drain everything; returns {request_id: token_ids}."""
        results = {}
        while self.has_work():
            for f in self.step():
                results[f["id"]] = f["token_ids"]
        return results
