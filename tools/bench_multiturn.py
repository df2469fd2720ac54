#!/usr/bin/env python3
"""Multi-turn serving benchmark — the reference's headline scenario
(reference keps/74-mooncake-integration benchmark: first turn vs multi-turn
with KV reuse).  Turn 2 prompts extend turn 1's
conversation, so the prefix cache serves the shared history and only the
new tokens prefill.  Prints one JSON line with both turns' p50 TTFT and
output tok/s, with and without the prefix cache.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def run_turns(enable_prefix, args, model_cfg):
    from rbg_amd.engine.config import EngineConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    cfg = EngineConfig(
        model=model_cfg, device=args.device,
        max_batch_size=args.batch,
        max_seq_len=args.seq_len + 2 * args.out_len + args.turn2_len + 64,
        kv_pool_tokens=2 * args.batch * (args.seq_len + 2 * args.out_len +
                                         args.turn2_len + 64) + 4096,
        enable_prefix_cache=enable_prefix,
        enforce_eager=args.device != "cuda")
    eng = LLMEngine(cfg)
    torch.manual_seed(7)
    convs = [torch.randint(0, model_cfg.vocab_size,
                           (args.seq_len,)).tolist()
             for _ in range(args.batch)]
    results = {}
    for turn in (1, 2):
        eng.stats.ttfts.clear()
        t0 = time.monotonic()
        seqs = eng.generate(
            convs, SamplingParams(max_new_tokens=args.out_len))
        if args.device == "cuda":
            torch.cuda.synchronize()
        wall = time.monotonic() - t0
        ttfts = sorted(eng.stats.ttfts)
        out_tokens = sum(len(s.output_tokens) for s in seqs)
        results[f"turn{turn}"] = {
            "p50_ttft_ms": round(ttfts[len(ttfts) // 2] * 1000, 1),
            "output_tok_s": round(out_tokens / wall, 1),
            "wall_s": round(wall, 2),
        }
        # extend each conversation: history + model output + new user turn
        torch.manual_seed(100 + turn)
        convs = [c + s.output_tokens +
                 torch.randint(0, model_cfg.vocab_size,
                               (args.turn2_len,)).tolist()
                 for c, s in zip(convs, seqs)]
    if eng.runner.cache.prefix is not None:
        results["prefix_stats"] = eng.runner.cache.prefix.stats()
    return results


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--turn2-len", type=int, default=64)
    ap.add_argument("--out-len", type=int, default=128)
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--device", default="cuda")
    args = ap.parse_args()
    from rbg_amd.engine.config import ModelConfig
    model_cfg = ModelConfig.preset(args.model)
    report = {
        "config": {"model": model_cfg.name, "batch": args.batch,
                   "seq_len": args.seq_len, "out_len": args.out_len},
        "with_prefix_cache": run_turns(True, args, model_cfg),
        "without_prefix_cache": run_turns(False, args, model_cfg),
        "reference_baseline": {
            "first_turn": {"tok_s": 1300.41, "mean_ttft_ms": 4808.37},
            "multi_turn_mooncake": {"tok_s": 1935.69, "mean_ttft_ms": 94.50},
        },
    }
    print(json.dumps(report, indent=1), flush=True)


if __name__ == "__main__":
    main()
