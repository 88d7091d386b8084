#!/usr/bin/env python3
"""Flagship benchmark: simulated `runbook investigate` on the local MI355X
engine (BASELINE config 3/4 shape).

One STEP = one complete structured investigation (TRIAGE→…→REMEDIATE)
driven end-to-end through the Llama-3-8B (bf16, random-init) engine with
grammar-constrained decoding against the simulated incident set, scored
with the reference scorer. Data is synthetic (simulated telemetry +
fixture-derived incidents); weights are random-init (no network for
checkpoints) — constrained decoding keeps every phase schema-valid so the
full agent/tool/LLM path executes identically to a trained checkpoint.

Scaling is WEAK data-parallelism: each rank runs its own TP=1 engine
replica and `--steps` investigations (`--concurrency` of them in flight
sharing the continuous-batching engine); the whole-job value aggregates
over ranks.

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  torchrun --nnodes=1 --nproc-per-node N bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=2,
                        help="timed investigations per rank")
    parser.add_argument("--warmup", type=int, default=1)
    parser.add_argument("--concurrency", type=int, default=32,
                        help="investigations in flight per rank (continuous "
                             "batching; BASELINE config 4 names 32)")
    parser.add_argument("--model", default=None, help="tiny | llama3-8b | llama3-70b")
    parser.add_argument("--tp", type=int, default=1)
    parser.add_argument("--max-tokens", type=int, default=768)
    parser.add_argument("--no-prefix-cache", action="store_true",
                        help="disable shared-prompt KV reuse (A/B)")
    parser.add_argument("--checkpoint", default=None,
                        help="serve a trained HF checkpoint directory instead "
                             "of random-init weights (accuracy axis)")
    args = parser.parse_args()

    import torch

    from runbookai_amd.parallel.dist import barrier, destroy, init_distributed

    rank, world = init_distributed()
    has_gpu = torch.cuda.is_available()
    device = f"cuda:{rank % torch.cuda.device_count()}" if has_gpu else "cpu"
    model_name = args.model or ("llama3-8b" if has_gpu else "tiny")

    from runbookai_amd.engine.client import LocalEngineClient
    from runbookai_amd.engine.engine import LLMEngine
    from runbookai_amd.evals.benchmark import load_fixtures
    from runbookai_amd.evals.scoring import score_investigation_result
    from runbookai_amd.agent.orchestrator import InvestigationOrchestrator
    from runbookai_amd.providers.simulation import (
        _SCENARIOS,
        SimScenario,
        set_scenario,
        set_thread_scenario,
    )
    from runbookai_amd.tools.registry import ToolRegistry

    # measured loop covers EVERY simulated scenario (redis / gateway-5xx /
    # kafka-disk / tls-expiry), not just the redis pair — sample + extended
    # fixture files, deduped by case id
    fix_dir = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                           "examples", "evals")
    fixtures = load_fixtures(os.path.join(fix_dir, "investigation-fixtures.sample.json"))
    extended = load_fixtures(os.path.join(fix_dir, "investigation-fixtures.extended.json"))
    seen_ids = set()
    cases = []
    for case in fixtures["cases"] + extended["cases"]:
        if case.get("id") not in seen_ids:
            seen_ids.add(case.get("id"))
            cases.append(case)

    # knowledge base: runbooks synced through the (GPU when available)
    # embedder + vector store, searched during triage/remediation
    from runbookai_amd.knowledge.indexer.embedder import create_embedder
    from runbookai_amd.knowledge.retriever.default import create_retriever

    retriever = create_retriever(in_memory=True, embedder=create_embedder())
    retriever.sync()

    # 6144 blocks x 16 = 98k KV tokens (~13 GB at 8B): the prefix pool no
    # longer churns under LRU pressure mid-run, which was the main source
    # of the round-1 +-35% run-to-run spread (cached tokens 190k-344k)
    kv_blocks = 6144 if model_name != "tiny" else 512
    # weak-scaling benchmark = DATA parallel replicas (tp=1 per rank);
    # --tp > 1 shards ONE model across all ranks instead (70B config):
    # rank 0 schedules + samples, other ranks follow broadcast steps
    tp = args.tp if args.tp > 1 else 1
    if tp > 1:
        assert world == tp, f"--tp {tp} needs torchrun with {tp} ranks"
    engine = LLMEngine(model=model_name, device=device, tp=tp,
                       kv_blocks=kv_blocks, background=(tp == 1 or rank == 0),
                       prefix_cache=not args.no_prefix_cache,
                       checkpoint=args.checkpoint)
    if tp > 1 and rank != 0:
        from runbookai_amd.parallel.tp_serving import run_follower_loop

        run_follower_loop(engine.model)   # returns on rank 0's stop payload
        barrier()
        destroy()
        return
    # process-global default (other threads, e.g. knowledge sync paths);
    # each investigation pins ITS case's scenario on its worker thread
    set_scenario(SimScenario.redis_exhaustion())
    scenarios = {case["id"]: (_SCENARIOS[case["id"]]()
                              if case["id"] in _SCENARIOS
                              else SimScenario.from_fixture(case))
                 for case in cases}

    def run_investigation(i: int) -> dict:
        case = cases[i % len(cases)]
        set_thread_scenario(scenarios[case["id"]])
        client = LocalEngineClient(engine, max_tokens=args.max_tokens)
        registry = ToolRegistry(knowledge_retriever=retriever)
        orch = InvestigationOrchestrator(
            llm=client, tool_executor=registry, knowledge_retriever=retriever,
            max_iterations=int(case.get("execute", {}).get("maxIterations", 6)),
        )
        t_start = time.time()
        result = orch.investigate(case["query"], incident_id=case.get("incidentId"))
        latency = time.time() - t_start
        score = score_investigation_result(result.to_dict(), case.get("expected", {}))
        return {"score": score["overall"],
                "passed": score["overall"] >= fixtures.get("passThreshold", 0.7),
                "success": result.success, "error": result.error,
                "latency_s": latency,
                "phases": len(result.phases_visited)}

    def run_batch(n: int) -> list[dict]:
        with ThreadPoolExecutor(max_workers=args.concurrency) as pool:
            return list(pool.map(run_investigation, range(n)))

    # warmup (untimed). The FIRST pass runs one investigation per distinct
    # case SEQUENTIALLY: every shared prompt prefix lands in the KV prefix
    # pool deterministically (no admission race on the cold wave), so the
    # timed region's cache-hit volume stops breathing across runs.
    if args.warmup > 0:
        for i in range(len(cases)):
            run_investigation(i)
        if args.warmup > 1:
            run_batch(args.warmup)

    # timed region: barrier + device sync on both sides. Under TP the
    # follower ranks are inside their broadcast loop (they execute every
    # timed step with rank 0), so rank 0 times alone and the barrier moves
    # to after the stop payload.
    tp_mode = tp > 1
    if not tp_mode:
        barrier()
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.time()
    results = run_batch(args.steps)
    if has_gpu:
        torch.cuda.synchronize()
    if not tp_mode:
        barrier()
    elapsed = time.time() - t0
    if tp_mode:
        from runbookai_amd.parallel.tp_serving import broadcast_stop

        engine.shutdown()
        broadcast_stop()
        barrier()

    # MAX over ranks (DP mode only; under TP rank 0 is the sole timer and
    # follower ranks have already exited past the final barrier)
    if world > 1 and not tp_mode:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    stats = engine.throughput_stats()
    # surface investigation/engine errors on stderr — a run full of failed
    # investigations is NOT a valid fast run
    errors = sorted({r["error"] for r in results if r.get("error")})
    if errors or stats.get("step_errors"):
        print(f"[bench] {len([r for r in results if r.get('error')])} of "
              f"{len(results)} investigations errored; engine step_errors="
              f"{stats.get('step_errors', 0)} last={stats.get('last_error', '')!r} "
              f"samples={errors[:3]}", file=sys.stderr)
    ms_per_step = elapsed * 1000.0 / args.steps
    total_investigations = args.steps if tp_mode else world * args.steps
    inv_per_hour = total_investigations / elapsed * 3600.0
    pass_rate = sum(1 for r in results if r["passed"]) / max(1, len(results))
    lat = sorted(r["latency_s"] for r in results)
    p50 = lat[len(lat) // 2] if lat else 0.0
    p95 = lat[min(len(lat) - 1, int(len(lat) * 0.95))] if lat else 0.0

    # accuracy tier (hermetic): the reference-scorer offline gate over the
    # fixtures' mockResults — semantic accuracy needs trained weights, so
    # the random-init pass_rate below is reported next to this gate
    from runbookai_amd.evals.benchmark import run_benchmark

    offline_pass_rate = run_benchmark(fixtures, offline=True)["passRate"]

    if rank == 0:
        line = {
            "metric": "investigations_per_hour",
            "value": round(inv_per_hour, 3),
            "unit": "investigations/hour",
            "n_gpus": world if has_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 1),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": ("synthetic (simulated incident set, trained policy "
                     "checkpoint, grammar-constrained decoding)" if args.checkpoint
                     else "synthetic (simulated incident set, random-init "
                          "weights, grammar-constrained decoding)"),
            "config": {
                "model": model_name,
                "checkpoint": args.checkpoint or "random-init",
                "global_batch": args.concurrency * world,
                "seq_len": engine.cfg.max_seq_len,
                "parallelism": f"tp{world}" if tp_mode else f"dp{world}",
                "scenarios": sorted(scenarios),
                "pass_rate": pass_rate,
                "offline_gate_pass_rate": offline_pass_rate,
                "latency_p50_s": round(p50, 2),
                "latency_p95_s": round(p95, 2),
                "decode_tok_per_s": round(stats.get("decode_tok_per_s", 0.0), 1),
                "prefill_tok_per_s": round(stats.get("prefill_tok_per_s", 0.0), 1),
                "llm_calls_total": stats.get("requests", 0),
                "engine_prefill_s": round(stats.get("prefill_time", 0.0), 1),
                "engine_decode_s": round(stats.get("decode_time", 0.0), 1),
                "engine_steps": stats.get("steps", 0),
                "decode_tokens": stats.get("decode_tokens", 0),
                "chunk_tokens": stats.get("chunk_tokens", 0),
                "cached_prefix_tokens": stats.get("cached_prefix_tokens", 0),
                # phase attribution: *_gpu from CUDA events (device-side),
                # launch/pre/mask/wait/advance are host wall — round-1's
                # un-synced stamps booked GPU decode into t_sample
                "t_decode_pre": round(stats.get("decode_pre_time", 0.0), 1),
                "t_decode_launch": round(stats.get("decode_launch_time", 0.0), 1),
                "t_decode_gpu": round(stats.get("decode_gpu_time", 0.0), 1),
                "t_chunk_pre": round(stats.get("chunk_pre_time", 0.0), 1),
                "t_chunk_launch": round(stats.get("chunk_launch_time", 0.0), 1),
                "t_chunk_gpu": round(stats.get("chunk_gpu_time", 0.0), 1),
                "t_sample": round(stats.get("sample_time", 0.0), 1),
                "t_sample_gpu": round(stats.get("sample_gpu_time", 0.0), 1),
                "t_sample_mask": round(stats.get("sample_mask_time", 0.0), 1),
                "t_sample_wait": round(stats.get("sample_wait_time", 0.0), 1),
                "t_sample_advance": round(stats.get("sample_advance_time", 0.0), 1),
                "chunk_steps": stats.get("chunk_steps", 0),
            },
        }
        print(json.dumps(line), flush=True)
    engine.shutdown()
    destroy()


if __name__ == "__main__":
    main()
