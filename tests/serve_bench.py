"""Serving throughput/latency benchmark (InferenceService engine, 1 GPU).

Usage: python3 tests/serve_bench.py [model] [n_requests] [max_new] [prompt_len]
Prints one JSON line: decode tokens/s out, request latency p50/p99, TTFT p50.
"""
import json
import os
import sys
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.ops import tunable as _t
_t.enable()

from kubeflow_amd.runtime.serving import InferenceEngine  # noqa: E402


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    n_req = int(sys.argv[2]) if len(sys.argv) > 2 else 16
    max_new = int(sys.argv[3]) if len(sys.argv) > 3 else 64
    # prompt_len: an int, "mix" = cycle 128/1024/4096 (the mixed-load
    # p99 inter-token measurement), or "same1024" = every request sends
    # the SAME 1024-token prompt (prefix-cache hit path: TTFT after the
    # first request is a row-copy + 1-token prefill)
    plen_arg = sys.argv[4] if len(sys.argv) > 4 else "128"
    mixed = plen_arg == "mix"
    same = plen_arg.startswith("same")
    lens = [128, 1024, 4096] if mixed else         [int(plen_arg[4:])] if same else [int(plen_arg)]
    plen = plen_arg if (mixed or same) else int(plen_arg)

    eng = InferenceEngine(model, max_slots=32, smax=4096,
                          max_batch=32).start()
    # warm-up (captures the decode graphs for the buckets used)
    eng.generate(list(range(1, lens[0] + 1)), max_new_tokens=8, timeout=300)

    results = []
    def run(i):
        L = lens[i % len(lens)]
        off = 1 if same else 1 + i
        results.append(eng.generate(list(range(off, L + off)),
                                    max_new_tokens=max_new, timeout=600))

    threads = [threading.Thread(target=run, args=(i,)) for i in range(n_req)]
    t0 = time.time()
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    el = time.time() - t0

    prefix_hits = eng.stats.get("prefix_hits", 0)
    toks = sum(len(r.generated) for r in results)
    lat = sorted(1000 * (r.finished_at - r.submitted)
                 for r in results if r.finished_at)
    ttft = sorted(1000 * (r.first_token_at - r.submitted)
                  for r in results if r.first_token_at)
    # inter-token gaps across all streams: a long prefill stalling active
    # decodes shows up here as p99 spikes
    gaps = sorted(
        1000 * (b - a)
        for r in results
        for a, b in zip(r.token_times, r.token_times[1:]))
    out = {
        "metric": "inferenceservice_tokens_per_s_out",
        "model": model,
        "value": round(toks / el, 1),
        "n_requests": n_req,
        "max_new_tokens": max_new,
        "prompt_len": plen, "prefix_hits": prefix_hits,
        "latency_p50_ms": round(lat[len(lat) // 2], 1),
        "latency_p99_ms": round(lat[min(len(lat) - 1, int(len(lat) * 0.99))], 1),
        "ttft_p50_ms": round(ttft[len(ttft) // 2], 1),
        "itl_p50_ms": round(gaps[len(gaps) // 2], 2) if gaps else None,
        "itl_p99_ms": round(gaps[min(len(gaps) - 1,
                                     int(len(gaps) * 0.99))], 2)
                       if gaps else None,
        "itl_max_ms": round(gaps[-1], 2) if gaps else None,
        "graph_replays": eng.stats.get("graph_replays", 0),
        "errors": sum(1 for r in results if r.error),
    }
    eng.stop()
    print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
