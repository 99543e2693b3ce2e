"""Serving decode throughput: llama3-8b bf16, hipGraph-captured decode
(GraphedDecoder) vs eager, across batch sizes.

    python tools/decode_bench.py [--model llama3-8b] [--tokens 128]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from ray_amd.models.llama import CONFIGS, GraphedDecoder, LlamaModel


def bench(dec, cfg, B, n_tokens, prefill=64):
    dev = dec.device
    tok = torch.randint(0, cfg.vocab_size, (B,), device=dev)
    for p in range(prefill, prefill + 8):  # warmup
        logits = dec.decode(tok, p)
        tok = logits.argmax(-1)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for p in range(prefill + 8, prefill + 8 + n_tokens):
        logits = dec.decode(tok, p)
        tok = logits.argmax(-1)
    torch.cuda.synchronize()
    return time.perf_counter() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--tokens", type=int, default=128)
    ap.add_argument("--batches", default="1,8,32")
    args = ap.parse_args()

    dev = torch.device("cuda", 0)
    cfg = CONFIGS[args.model]
    torch.manual_seed(0)
    model = LlamaModel(cfg, dtype=torch.bfloat16).to(dev).eval()
    model.cosT = model.cosT.to(dev)
    model.sinT = model.sinT.to(dev)

    for B in [int(b) for b in args.batches.split(",")]:
        dec = GraphedDecoder(model, B, 512, dev)
        dt_eager = bench(dec, cfg, B, max(args.tokens // 4, 16))
        n_eager = max(args.tokens // 4, 16)
        dec.capture()
        dt = bench(dec, cfg, B, args.tokens)
        tok_s = B * args.tokens / dt
        print(f"batch {B:3d}: hipGraph {tok_s:9.1f} tok/s "
              f"({dt / args.tokens * 1e3:6.2f} ms/step)   "
              f"eager {B * n_eager / dt_eager:9.1f} tok/s "
              f"({dt_eager / n_eager * 1e3:6.2f} ms/step)")


if __name__ == "__main__":
    main()
