"""Object-store throughput microbench (control-plane performance evidence).

Usage: python tests/store_bench.py [n_objects]
Measures create/update/get/list/watch-fanout/patch-contention rates for the
in-process store — the analog of kube-apiserver+etcd throughput in the
reference's stack.
"""
import sys
import threading
import time

from kubeflow_amd.api import ObjectStore, new_object


def main(n=5000):
    s = ObjectStore()
    t0 = time.perf_counter()
    for i in range(n):
        s.create(new_object("PyTorchJob", f"job-{i}", "default",
                            spec={"i": i}))
    t_create = time.perf_counter() - t0

    objs = [s.get("PyTorchJob", f"job-{i}", "default") for i in range(n)]
    t0 = time.perf_counter()
    for o in objs:
        o["status"]["x"] = 1
        s.update(o, check_version=False)
    t_update = time.perf_counter() - t0

    t0 = time.perf_counter()
    for i in range(n):
        s.get("PyTorchJob", f"job-{i}", "default")
    t_get = time.perf_counter() - t0

    t0 = time.perf_counter()
    for _ in range(20):
        s.list("PyTorchJob", "default")
    t_list = (time.perf_counter() - t0) / 20

    # watch fanout: 8 watchers, measure event delivery latency
    lat = []

    def watcher(_ev):
        lat.append(time.perf_counter())

    for _ in range(8):
        s.watch(watcher, kind="Notebook")
    t0 = time.perf_counter()
    s.create(new_object("Notebook", "nb-fanout", "default"))
    time.sleep(0.2)
    fanout_ms = (max(lat) - t0) * 1e3 if lat else float("nan")

    # patch contention: 8 threads x 50 optimistic patches on one object
    s.create(new_object("PyTorchJob", "contended", "default"))

    def patcher():
        for _ in range(50):
            s.patch("PyTorchJob", "contended", "default",
                    {"status": {"n": time.time()}})

    ths = [threading.Thread(target=patcher) for _ in range(8)]
    t0 = time.perf_counter()
    for t in ths:
        t.start()
    for t in ths:
        t.join()
    t_patch = time.perf_counter() - t0
    final = s.get("PyTorchJob", "contended", "default")

    print(f"objects: {n}")
    print(f"create: {n / t_create:,.0f}/s")
    print(f"update: {n / t_update:,.0f}/s")
    print(f"get:    {n / t_get:,.0f}/s")
    print(f"list({n}): {t_list * 1e3:.1f} ms")
    print(f"watch fanout (8 watchers): {fanout_ms:.2f} ms")
    print(f"contended patch (8 thr x 50): {400 / t_patch:,.0f}/s, "
          f"final resourceVersion "
          f"{final['metadata']['resourceVersion']}")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 5000)
