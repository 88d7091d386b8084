#!/usr/bin/env python3
"""Train the small investigation policy on synthetic oracle traces and
export it as a servable HF checkpoint (+ tokenizer.json).

Pipeline (all offline, no network):
  1. generate_traces(): oracle-driven runs of the REAL orchestrator over
     archetype-sampled simulated incidents -> (prompt, JSON answer) pairs.
  2. Train a ByteLevel BPE tokenizer (vocab 4096) on the corpus with the
     Llama-3 chat special tokens the engine's BpeTokenizer expects.
  3. Next-token CE on [chat-encoded prompt][response][<|eot_id|>] with
     loss masked to the response+eot span (TrainableLlama, policy-small).
  4. export_trained() -> HF safetensors + config.json + tokenizer.json,
     servable via LLMEngine(checkpoint=DIR) / bench.py --checkpoint DIR.
  5. --eval: run the held-out eval fixture cases (redis/gateway/kafka/tls)
     through the served checkpoint + reference scorer -> pass_rate.

Usage:  python scripts/train_policy.py --cases 400 --steps 1200 \
            --out gpurun_out/policy_ckpt --eval
"""
from __future__ import annotations

import argparse
import json
import math
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

SYSTEM = "You are Runbook, an SRE agent. Respond with ONLY the requested JSON."
SPECIALS = ["<|begin_of_text|>", "<|start_header_id|>", "<|end_header_id|>",
            "<|eot_id|>"]


def build_corpus(n_cases: int, seed: int):
    from runbookai_amd.evals.trace_gen import generate_traces

    t0 = time.time()
    recs = generate_traces(n_cases, seed=seed)
    print(f"[data] {len(recs)} trace pairs from {n_cases} cases "
          f"in {time.time() - t0:.1f}s", flush=True)
    return recs


def train_tokenizer(recs, out_dir: str, vocab_size: int = 4096) -> str:
    import tokenizers

    corpus = [SYSTEM]
    for r in recs:
        corpus.append(r["body"])
        corpus.append(r["response"])
    tok = tokenizers.ByteLevelBPETokenizer()
    tok.train_from_iterator(corpus, vocab_size=vocab_size - len(SPECIALS),
                            min_frequency=2, special_tokens=SPECIALS)
    os.makedirs(out_dir, exist_ok=True)
    path = os.path.join(out_dir, "tokenizer.json")
    tok.save(path)
    return path


def encode_samples(recs, tok, max_len: int):
    """-> list of (ids, loss_start). loss covers response + eot."""
    eot = tok.eot_id
    samples = []
    drop = 0
    for r in recs:
        prompt_ids = tok.encode_chat(SYSTEM, r["body"])
        resp_ids = tok.encode(r["response"]) + [eot]
        ids = prompt_ids + resp_ids
        if len(ids) > max_len:
            drop += 1
            continue
        samples.append((ids, len(prompt_ids)))
    if drop:
        print(f"[data] dropped {drop} over-length samples (> {max_len})")
    return samples


def batches(samples, batch_tokens: int, rng: random.Random):
    """Length-bucketed batches, padded, with loss masks."""
    order = sorted(range(len(samples)), key=lambda i: len(samples[i][0]))
    buckets = []
    cur = []
    cur_max = 0
    for i in order:
        L = len(samples[i][0])
        if cur and (len(cur) + 1) * max(cur_max, L) > batch_tokens:
            buckets.append(cur)
            cur, cur_max = [], 0
        cur.append(i)
        cur_max = max(cur_max, L)
    if cur:
        buckets.append(cur)
    rng.shuffle(buckets)
    for b in buckets:
        L = max(len(samples[i][0]) for i in b)
        ids = torch.zeros(len(b), L, dtype=torch.long)
        mask = torch.zeros(len(b), L, dtype=torch.bool)
        for row, i in enumerate(b):
            s, ls = samples[i]
            ids[row, :len(s)] = torch.tensor(s)
            mask[row, ls:len(s)] = True   # loss on response+eot positions
        yield ids, mask


def run_training(args):
    from runbookai_amd.engine.bpe_tokenizer import BpeTokenizer
    from runbookai_amd.engine.train import CONFIGS, TrainableLlama, export_trained

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    recs = build_corpus(args.cases, args.seed)
    rng = random.Random(args.seed)
    rng.shuffle(recs)
    n_val = max(8, len(recs) // 20)
    val_recs, train_recs = recs[:n_val], recs[n_val:]

    tok_path = train_tokenizer(train_recs, args.out, args.vocab)
    tok = BpeTokenizer.from_file(tok_path)
    print(f"[tok] vocab {len(tok.vocab)} eot {tok.eot_id}")

    cfg = CONFIGS[args.config]
    cfg.vocab_size = max(4096, ((len(tok.vocab) + 63) // 64) * 64)
    # fp32 master weights; bf16 compute via autocast in loss_of
    model = TrainableLlama(cfg).to(dev)
    n_params = sum(p.numel() for p in model.parameters())
    print(f"[model] {args.config}: {n_params/1e6:.1f}M params on {dev}")

    train_s = encode_samples(train_recs, tok, args.max_len)
    val_s = encode_samples(val_recs, tok, args.max_len)
    print(f"[data] {len(train_s)} train / {len(val_s)} val samples; "
          f"{sum(len(s[0]) for s in train_s)/1e6:.2f}M train tokens")

    opt = torch.optim.AdamW(model.parameters(), lr=args.lr, weight_decay=0.01,
                            betas=(0.9, 0.95))
    step = 0
    t0 = time.time()
    losses = []

    def loss_of(ids, mask):
        ids = ids.to(dev)
        mask = mask.to(dev)
        with torch.autocast(device_type="cuda", dtype=torch.bfloat16,
                            enabled=(dev == "cuda")):
            logits = model(ids[:, :-1])
        tgt = ids[:, 1:]
        m = mask[:, 1:]
        lt = logits.float().reshape(-1, logits.shape[-1])
        loss = torch.nn.functional.cross_entropy(
            lt[m.reshape(-1)], tgt.reshape(-1)[m.reshape(-1)])
        return loss

    while step < args.steps:
        for ids, mask in batches(train_s, args.batch_tokens, rng):
            if step >= args.steps:
                break
            lr = args.lr * 0.5 * (1 + math.cos(math.pi * step / args.steps))
            for g in opt.param_groups:
                g["lr"] = lr
            loss = loss_of(ids, mask)
            opt.zero_grad(set_to_none=True)
            loss.backward()
            torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
            opt.step()
            losses.append(float(loss))
            step += 1
            if step % 100 == 0 or step == args.steps:
                with torch.no_grad():
                    vl = [float(loss_of(i, m))
                          for i, m in list(batches(val_s, args.batch_tokens,
                                                   random.Random(0)))[:4]]
                print(f"[train] step {step}/{args.steps} "
                      f"loss {sum(losses[-100:])/len(losses[-100:]):.3f} "
                      f"val {sum(vl)/len(vl):.3f} "
                      f"({time.time()-t0:.0f}s)", flush=True)

    model = model.float().cpu()
    export_trained(model, args.out)
    print(f"[export] checkpoint at {args.out}")
    return args.out


def eval_checkpoint(ckpt: str, concurrency: int = 4,
                    include_generated: bool = False) -> dict:
    """Held-out eval: the REAL fixture cases through the served checkpoint.
    include_generated adds the converter-generated suites (rcaeval /
    rootly / tracerca — reported separately as out-of-distribution)."""
    from concurrent.futures import ThreadPoolExecutor

    from runbookai_amd.agent.orchestrator import InvestigationOrchestrator
    from runbookai_amd.engine.client import LocalEngineClient
    from runbookai_amd.engine.engine import LLMEngine
    from runbookai_amd.evals.benchmark import load_fixtures
    from runbookai_amd.evals.scoring import score_investigation_result
    from runbookai_amd.knowledge.retriever.default import create_retriever
    from runbookai_amd.providers.simulation import (
        _SCENARIOS,
        SimScenario,
        set_thread_scenario,
    )
    from runbookai_amd.tools.registry import ToolRegistry

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    engine = LLMEngine(device=dev, checkpoint=ckpt,
                       kv_blocks=2048 if dev == "cuda" else 256)
    retriever = create_retriever(in_memory=True)
    retriever.sync()
    fix_dir = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examples", "evals")
    cases = []
    seen = set()
    files = ["investigation-fixtures.sample.json",
             "investigation-fixtures.extended.json"]
    gen_files = ["rcaeval-fixtures.generated.json",
                 "rootly-logs-fixtures.generated.json",
                 "tracerca-fixtures.generated.json"]
    gen_ids = set()
    if include_generated:
        files += gen_files
    for fn in files:
        for c in load_fixtures(os.path.join(fix_dir, fn))["cases"]:
            if c["id"] not in seen:
                seen.add(c["id"])
                cases.append(c)
                if fn in gen_files:
                    gen_ids.add(c["id"])

    def run(case):
        scen = (_SCENARIOS[case["id"]]() if case["id"] in _SCENARIOS
                else SimScenario.from_fixture(case))
        set_thread_scenario(scen)
        client = LocalEngineClient(engine, max_tokens=512)
        orch = InvestigationOrchestrator(
            llm=client, tool_executor=ToolRegistry(knowledge_retriever=retriever),
            knowledge_retriever=retriever, max_iterations=5)
        result = orch.investigate(case["query"], incident_id=case.get("incidentId"))
        score = score_investigation_result(result.to_dict(), case.get("expected", {}))
        return {"id": case["id"], "score": round(score["overall"], 3),
                "passed": score["overall"] >= 0.7,
                "rootCause": (result.root_cause or "")[:120],
                "hypotheses": [h.get("statement", "")[:90]
                               for h in (result.hypotheses or [])[:3]]}

    t0 = time.time()
    with ThreadPoolExecutor(max_workers=concurrency) as pool:
        results = list(pool.map(run, cases))
    elapsed = time.time() - t0
    core = [r for r in results if r["id"] not in gen_ids]
    gen = [r for r in results if r["id"] in gen_ids]
    report = {
        "checkpoint": ckpt,
        "cases": results,
        "pass_rate": sum(r["passed"] for r in core) / max(1, len(core)),
        "mean_score": round(sum(r["score"] for r in core) / max(1, len(core)), 3),
        "wall_s": round(elapsed, 1),
        "grammar_constrained": engine.supports_bpe_grammar,
        "device": dev,
    }
    if gen:
        report["generated_suites"] = {
            "pass_rate": sum(r["passed"] for r in gen) / len(gen),
            "mean_score": round(sum(r["score"] for r in gen) / len(gen), 3),
            "n": len(gen),
        }
    engine.shutdown()
    return report


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--cases", type=int, default=400)
    p.add_argument("--steps", type=int, default=1200)
    p.add_argument("--seed", type=int, default=17)
    p.add_argument("--vocab", type=int, default=4096)
    p.add_argument("--config", default="policy-small")
    p.add_argument("--max-len", type=int, default=1536)
    p.add_argument("--batch-tokens", type=int, default=65536)
    p.add_argument("--lr", type=float, default=6e-4)
    p.add_argument("--out", default="gpurun_out/policy_ckpt")
    p.add_argument("--eval", action="store_true")
    p.add_argument("--eval-only", action="store_true")
    p.add_argument("--eval-generated", action="store_true",
                   help="also evaluate the converter-generated suites")
    args = p.parse_args()

    if not args.eval_only:
        run_training(args)
    if args.eval or args.eval_only:
        report = eval_checkpoint(args.out,
                                 include_generated=args.eval_generated)
        print(json.dumps(report, indent=1))
        with open(os.path.join(args.out, "eval_report.json"), "w") as f:
            json.dump(report, f, indent=1)


if __name__ == "__main__":
    main()
