#!/usr/bin/env python3
"""70B TP=1 smoke on one MI355X: init in HBM, constrained generation,
throughput stats. (TP=8 over xGMI uses the same code path — the driver's
multi-GPU tier exercises the collectives; numerics are covered by the
TP=2 gloo test.)"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from runbookai_amd.agent.llm_parser import PROMPT_SCHEMAS
from runbookai_amd.engine.engine import LLMEngine


def main() -> None:
    model = sys.argv[1] if len(sys.argv) > 1 else "llama3-70b"
    t0 = time.time()
    eng = LLMEngine(model=model, device="cuda:0", tp=1, kv_blocks=512, background=False)
    torch.cuda.synchronize()
    init_s = time.time() - t0
    free, total = torch.cuda.mem_get_info()
    print(f"{model} init {init_s:.1f}s, mem used {(total - free) / 2**30:.0f} GiB "
          f"of {total / 2**30:.0f}")
    ids = eng.tokenizer.encode_chat(
        "You are Runbook.", "triage: checkout latency spike and redis timeouts")
    t0 = time.time()
    req = eng.generate(ids, max_new_tokens=120, schema=PROMPT_SCHEMAS["triage"])
    dt = time.time() - t0
    stats = eng.throughput_stats()
    out = {
        "model": model,
        "init_s": round(init_s, 1),
        "mem_gib": round((total - free) / 2**30, 1),
        "gen_bytes": len(req.out_ids),
        "gen_s": round(dt, 2),
        "decode_tok_per_s": round(stats.get("decode_tok_per_s", 0.0), 1),
        "prefill_tok_per_s": round(stats.get("prefill_tok_per_s", 0.0), 1),
        "sample": eng.tokenizer.decode(req.out_ids)[:80],
    }
    print(json.dumps(out, indent=1))
    os.makedirs("gpurun_out", exist_ok=True)
    with open(f"gpurun_out/smoke_{model}.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
