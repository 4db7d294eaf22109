"""Decode-step anatomy: where does the serving inter-token time go?

Builds the engine (no HTTP / no engine thread), prefills B requests, then
times (a) the captured graph replay alone, (b) the full _decode_step
(bookkeeping + H2D staging + replay + sampling sync), (c) the eager
decode forward. The (b)-(a) gap is host-side overhead the graph cannot
hide; (a) is the pure GPU decode step to compare against the weight-read
floor (~16 GB / 8 TB/s = 2 ms for llama3-8b).

Usage: python scripts/decode_profile.py [model] [batch] [prompt_len] [steps]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.ops import tunable as _t
_t.enable()

import torch  # noqa: E402

from kubeflow_amd.runtime.serving import InferenceEngine, Request  # noqa


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    B = int(sys.argv[2]) if len(sys.argv) > 2 else 16
    plen = int(sys.argv[3]) if len(sys.argv) > 3 else 128
    steps = int(sys.argv[4]) if len(sys.argv) > 4 else 50

    eng = InferenceEngine(model, max_slots=max(34, B + 2), smax=4096,
                      max_batch=32)
    for i in range(B):
        r = Request(rid=f"r{i}", prompt=list(range(1, plen + 1)),
                    max_new_tokens=steps * 4 + 64)
        eng.pending.put(r)
    for _ in range(B * 8):
        if len(eng.active) >= B:
            break
        eng._admit()
        while eng._chunking is not None:
            eng._advance_chunk()
    assert len(eng.active) == B, len(eng.active)

    # warm + capture the bucket
    for _ in range(3):
        eng._decode_step()
        eng.active = sorted(eng.active, key=lambda r: r.rid)

    dev = eng.device
    torch.cuda.synchronize()

    # (a) pure graph replay
    bucket = 1
    while bucket < B:
        bucket *= 2
    graph, static = eng._graph_for(min(bucket, eng.max_batch))
    t0 = time.perf_counter()
    for _ in range(steps):
        graph.replay()
    torch.cuda.synchronize()
    replay_ms = (time.perf_counter() - t0) / steps * 1e3

    # (b) full decode step
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng._decode_step()
    torch.cuda.synchronize()
    full_ms = (time.perf_counter() - t0) / steps * 1e3

    # (c) eager decode forward (same tensors as the graph statics)
    eng.use_graphs = False
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        eng._decode_step()
    torch.cuda.synchronize()
    eager_ms = (time.perf_counter() - t0) / steps * 1e3
    eng.use_graphs = True

    print(f"model={model} B={B} plen={plen} steps={steps} "
          f"skinny={os.environ.get('KF_SKINNY', 'small')}")
    print(f"graph_replay_ms={replay_ms:.3f}  (pure GPU decode step)")
    print(f"decode_step_ms={full_ms:.3f}  (replay + host bookkeeping "
          f"+ sample sync)")
    print(f"host_overhead_ms={full_ms - replay_ms:.3f}")
    print(f"eager_ms={eager_ms:.3f}  (launch-bound, no graph)")
    print(f"tok_per_s_batch={B / full_ms * 1e3:.0f}")


if __name__ == "__main__":
    main()
