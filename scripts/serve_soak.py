"""Serving soak: hammer the engine with mixed prompt lengths, repeated
prompts (prefix-cache churn + eviction), streaming consumers and small
max_new, all concurrently — a race hunt, not a throughput bench.

Usage: python scripts/serve_soak.py [model] [n_requests]
"""
import os
import random
import sys
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.ops import tunable as _t
_t.enable()

from kubeflow_amd.runtime.serving import InferenceEngine  # noqa: E402


def main():
    model = sys.argv[1] if len(sys.argv) > 1 else "llama3-8b"
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 120
    rng = random.Random(7)
    eng = InferenceEngine(model, max_slots=24, smax=4096,
                          max_batch=16).start()
    errors = []
    done = [0]
    lock = threading.Lock()

    base = [list(range(3, 1027)), list(range(5, 133)),
            list(range(9, 2058))]  # repeated -> prefix hits + eviction

    def run(i):
        if i % 4 == 0:
            prompt = base[i % len(base)]
        else:
            L = rng.choice([64, 128, 700, 1024, 2500])
            prompt = [rng.randrange(1, 30000) for _ in range(L)]
        mn = rng.choice([4, 16, 48])
        try:
            if i % 5 == 0:
                toks = list(eng.generate_stream(prompt, max_new_tokens=mn,
                                                timeout=300))
                assert len(toks) == mn, (len(toks), mn)
            else:
                r = eng.generate(prompt, max_new_tokens=mn, timeout=300,
                                 temperature=0.7 if i % 3 == 0 else 0.0)
                assert not r.error, r.error
                assert len(r.generated) == mn
        except Exception as e:
            with lock:
                errors.append(f"req {i}: {type(e).__name__}: {e}")
        with lock:
            done[0] += 1

    threads = []
    for i in range(n):
        t = threading.Thread(target=run, args=(i,))
        t.start()
        threads.append(t)
        if len(threads) > 40:
            threads.pop(0).join()
    for t in threads:
        t.join()
    eng.stop()
    print(f"soak: {done[0]}/{n} done, {len(errors)} errors, "
          f"prefix_hits={eng.stats.get('prefix_hits', 0)}, "
          f"completed={eng.stats['completed']}")
    for e in errors[:5]:
        print(" ", e)
    sys.exit(1 if errors else 0)


if __name__ == "__main__":
    main()
