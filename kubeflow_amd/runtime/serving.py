"""InferenceService engine — continuous batching on one MI355X, no Triton.

The runtime half of the KServe InferenceService the reference integrates
with (SURVEY.md §2.12 "InferenceService: predictor spec, dynamic batching").

MI355X-first design: with 288 GB of HBM3E, the KV cache is a single
preallocated region of fixed-stride slots ([SLOTS, SMAX, Hkv, D] per layer)
rather than a paged pool — slot granularity removes page tables and keeps
each sequence's K/V rows contiguous for the decode kernel's streaming reads.
(Llama-3-8B: 16 GB weights + 64 slots × 8192 tokens × 128 KB/token ≈ 80 GB —
a third of one GPU.)

Scheduling: continuous batching. Each engine iteration admits queued
prefills (chunked into the running batch) and then runs ONE batched decode
step for every active sequence via the kf_attn_decode kernel. Prefill uses
the training flash-attention kernel over the padded prompt.
"""
from __future__ import annotations

import os
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from kubeflow_amd import ops
from kubeflow_amd.models import build_model
from kubeflow_amd.models.llama import LlamaModel


class KVCache:
    """Per-layer slot cache: k/v [SLOTS, SMAX, Hkv, D] bf16."""

    def __init__(self, n_layers: int, slots: int, smax: int, hkv: int,
                 d: int, device, dtype=torch.bfloat16):
        self.k = [torch.zeros(slots, smax, hkv, d, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.v = [torch.zeros(slots, smax, hkv, d, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.slots = slots
        self.smax = smax
        self._free = list(range(slots))
        self._lock = threading.Lock()

    def alloc(self) -> Optional[int]:
        with self._lock:
            return self._free.pop() if self._free else None

    def has_free(self) -> bool:
        with self._lock:
            return bool(self._free)

    def free(self, slot: int):
        with self._lock:
            self._free.append(slot)

    def copy_rows(self, src: int, dst: int, n: int):
        """Copy the first n cache rows of slot src into slot dst (all
        layers) — the prefix-cache hit path. Device-side copies only."""
        for l in range(len(self.k)):
            self.k[l][dst, :n].copy_(self.k[l][src, :n])
            self.v[l][dst, :n].copy_(self.v[l][src, :n])


@dataclass
class Request:
    rid: str
    prompt: List[int]
    max_new_tokens: int = 32
    temperature: float = 0.0
    submitted: float = field(default_factory=time.time)
    # filled by the engine:
    slot: int = -1
    pos: int = 0
    generated: List[int] = field(default_factory=list)
    token_times: List[float] = field(default_factory=list)  # per-token stamps
    done: threading.Event = field(default_factory=threading.Event)
    # streaming consumers: tokens are pushed as produced; None = finished
    stream: "Optional[queue.Queue]" = None
    scores: Optional[List[float]] = None  # classifier probabilities
    first_token_at: Optional[float] = None
    finished_at: Optional[float] = None
    error: str = ""


class InferenceEngine:
    def __init__(self, model_name: str, device=None, max_slots: int = 32,
                 smax: int = 4096, max_batch: int = 32,
                 storage_uri: str | None = None,
                 quant: str | None = None):
        self.device = device or (torch.device("cuda", 0)
                                 if torch.cuda.is_available()
                                 else torch.device("cpu"))
        dtype = (torch.bfloat16 if self.device.type == "cuda"
                 else torch.float32)
        self.model: LlamaModel = build_model(model_name, device=self.device,
                                             dtype=dtype)
        # KServe storageUri analog: load trained weights from a checkpoint
        # directory (resolved to a filesystem path by the controller);
        # without it the engine serves random-init weights (dev mode).
        self.loaded_step: int | None = None
        if storage_uri:
            from kubeflow_amd.runtime import checkpoint as _ckpt
            self.loaded_step = _ckpt.load_model_weights(self.model,
                                                        storage_uri)
        self.model.eval()
        cfg = self.model.cfg
        # classifier models (BERT family): no KV cache / graphs — each
        # request is one batched forward; same-length prompts batch, the
        # rest pad to a 128 multiple with masked_attention(kv_len)
        self.classify = hasattr(cfg, "n_classes")
        if self.classify:
            self.max_batch = max_batch
            self.cache = None
            self.quant = False
            self._qw = None
            self.use_graphs = False
            self._graphs = {}
            self._pad_slot = None
            self._chunking = None
            self.pending = queue.Queue()
            self.active = []
            self.prefix_slots = 0
            self._prefix = {}
            self._stop = False
            self._thread = None
            self.stats = {"requests": 0, "completed": 0, "tokens_out": 0,
                          "prefill_tokens": 0, "graph_replays": 0}
            return
        # KF_SERVE_QUANT=fp8: per-output-row OCP e4m3 weight-only quant
        # for the DECODE linears (W8A16 — halves the weight traffic the
        # decode step is bound by; prefill keeps the bf16 weights).
        # Applied after storageUri load so trained weights quantize.
        if quant is None:
            quant = os.environ.get("KF_SERVE_QUANT", "off")
        if quant not in ("off", "fp8"):
            raise ValueError(f"unsupported quantization {quant!r} "
                             "(supported: fp8)")
        self.quant = quant == "fp8"
        self._qw = None
        if self.quant and not getattr(cfg, "n_experts", 0):
            self._qw = [
                {n: ops.quantize_fp8_rows(getattr(layer, n).weight)
                 for n in ("wqkv", "wo", "w13", "w2")}
                for layer in self.model.layers
            ]
            # lm_head stays bf16 under "small" routing (sampling
            # sensitivity); KF_SKINNY=all opts it into fp8 too
            self._q_lm = ops.quantize_fp8_rows(self.model.lm_head.weight)
        elif self.quant:
            self.quant = False  # MoE decode is eager/dense — not routed
        smax = min(smax, cfg.max_seq_len)
        self.cache = KVCache(cfg.n_layers, max_slots, smax,
                             cfg.n_kv_heads, cfg.head_dim, self.device, dtype)
        self.max_batch = max_batch
        self.pending: "queue.Queue[Request]" = queue.Queue()
        self.active: List[Request] = []
        self._chunking = None  # in-progress chunked prefill state
        # prefix cache: finished sequences donate their slot (keyed by
        # their full token history) so requests sharing a prompt prefix
        # copy rows instead of recomputing them (KF_PREFIX_CACHE slots,
        # 0 disables). LRU; evicted on demand when admission needs slots.
        self.prefix_slots = int(os.environ.get("KF_PREFIX_CACHE", "4"))
        self._prefix = {}   # tokens tuple -> slot (insertion = LRU order)
        self._stop = False
        self._thread: Optional[threading.Thread] = None
        self.stats = {"requests": 0, "completed": 0, "tokens_out": 0,
                      "prefill_tokens": 0, "graph_replays": 0}
        # hipGraph decode: capture the whole batched decode step per batch
        # bucket (the ~10 kernels x 32 layers of launch overhead dominate
        # small-batch decode latency otherwise). One cache slot is reserved
        # as the padding target for bucket rows beyond the live batch.
        # MoE decode goes through MoEMLP.decode_dense (all experts on all
        # tokens, dense gate mask) — fixed shapes, so it captures.
        self.use_graphs = (self.device.type == "cuda"
                           and os.environ.get("KF_SERVE_GRAPH", "1") == "1")
        self._graphs = {}
        self._pad_slot = self.cache.alloc() if self.use_graphs else None

    # ------------------------------------------------------------- public
    def start(self, precapture: bool | None = None):
        """Start the engine loop. By default (KF_SERVE_PRECAPTURE=1) every
        power-of-two decode-batch bucket's hipGraph is captured up front:
        lazy capture showed up as 100s-of-ms inter-token spikes the first
        time each bucket appeared mid-traffic (profiles/r02_serve_mixed.md
        itl_max)."""
        if precapture is None:
            precapture = os.environ.get("KF_SERVE_PRECAPTURE", "1") == "1"
        if precapture and self.use_graphs:
            b = 1
            while True:
                self._graph_for(b)
                if b >= self.max_batch:
                    break
                b = min(b * 2, self.max_batch)
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="inference-engine")
        self._thread.start()
        return self

    def stop(self):
        self._stop = True
        if self._thread:
            self._thread.join(timeout=10)

    def submit(self, req: Request) -> Request:
        self.stats["requests"] += 1
        self.pending.put(req)
        return req

    def generate(self, prompt: List[int], max_new_tokens: int = 32,
                 timeout: float = 120.0, temperature: float = 0.0) -> Request:
        req = Request(rid=f"r{time.monotonic_ns()}", prompt=list(prompt),
                      max_new_tokens=max_new_tokens, temperature=temperature)
        self.submit(req)
        if not req.done.wait(timeout):
            req.error = req.error or "timeout"
        return req

    def generate_stream(self, prompt: List[int], max_new_tokens: int = 32,
                        temperature: float = 0.0, timeout: float = 120.0):
        """Submit and yield tokens as they are produced (generator)."""
        req = Request(rid=f"r{time.monotonic_ns()}", prompt=list(prompt),
                      max_new_tokens=max_new_tokens,
                      temperature=temperature, stream=queue.Queue())
        self.submit(req)
        while True:
            try:
                tok = req.stream.get(timeout=timeout)
            except queue.Empty:
                req.error = req.error or "timeout"
                return
            if tok is None:
                return
            yield tok

    # -------------------------------------------------------------- engine
    @torch.no_grad()
    def _loop(self):
        if self.classify:
            while not self._stop:
                if not self._classify_batch():
                    time.sleep(0.002)
            return
        while not self._stop:
            self._admit()
            if not self.active:
                if self._chunking is None:
                    time.sleep(0.002)  # queued-but-unadmittable / idle
                continue
            self._decode_step()

    # prompts longer than this prefill in CHUNK-token slices interleaved
    # with decode steps (rectangular-causal flash kernel
    # kf_attn_fwd4_rect), bounding active streams' inter-token stalls to
    # one chunk instead of the whole prompt. Measured A/B at mixed
    # 128/1k/4k load (profiles/r02_serve_mixed.md): inter-token p99
    # 196 -> 68 ms for ~-25% aggregate tok/s — an SLO/throughput knob.
    # KF_PREFILL_CHUNK overrides; 0 disables chunking.
    PREFILL_CHUNK = int(os.environ.get("KF_PREFILL_CHUNK", "1024")) or (1 << 30)

    @torch.no_grad()
    def _classify_batch(self) -> int:
        """Drain up to max_batch same-length classify requests and run
        one padded forward (masked_attention over the real length)."""
        taken, back = [], []
        while len(taken) < self.max_batch:
            try:
                r = self.pending.get_nowait()
            except queue.Empty:
                break
            if not taken or len(r.prompt) == len(taken[0].prompt):
                taken.append(r)
            else:
                back.append(r)
        for r in back:
            self.pending.put(r)
        if not taken:
            return 0
        try:
            S = len(taken[0].prompt)
            Sp = (S + 127) // 128 * 128
            toks = torch.zeros(len(taken), Sp, dtype=torch.int64,
                               device=self.device)
            for i, r in enumerate(taken):
                toks[i, :S] = torch.tensor(r.prompt, dtype=torch.int64)
            logits = self.model(toks, kv_len=S if Sp > S else None)
            probs = torch.softmax(logits.float(), dim=-1).cpu()
            now = time.time()
            for i, r in enumerate(taken):
                r.scores = [round(float(p), 6) for p in probs[i]]
                r.generated = [int(probs[i].argmax())]
                r.first_token_at = r.finished_at = now
                r.token_times.append(now)
                self.stats["tokens_out"] += 1
                self.stats["completed"] += 1
                if r.stream is not None:
                    r.stream.put(r.generated[0])
                    r.stream.put(None)
                r.done.set()
        except Exception as e:  # pragma: no cover
            import traceback
            traceback.print_exc()
            for r in taken:
                r.error = f"{type(e).__name__}: {e}"
                r.finished_at = time.time()
                if r.stream is not None:
                    r.stream.put(None)
                r.done.set()
        return len(taken)

    def _admit(self) -> int:
        # Fairness: one prefill unit (a full short prompt, or ONE chunk of
        # a long one) per loop iteration while sequences decode; with
        # nothing active, drain freely.
        if self._chunking is not None:
            self._advance_chunk()
            return 0
        if not self.active:
            # ramp-up: batch same-length queued prompts into ONE prefill
            # forward — B=1 prefill is GEMM-efficiency bound (M = seq
            # len), so grouping multiplies M at no latency cost while
            # nothing is decoding. (While decodes run, prefills stay
            # single so inter-token stalls stay bounded.)
            group = self._take_prefill_group()
            if group:
                try:
                    self._prefill_many(group)
                    self.active.extend(group)
                except Exception as e:  # pragma: no cover
                    import traceback
                    traceback.print_exc()
                    for r in group:
                        r.error = f"{type(e).__name__}: {e}"
                        self.cache.free(r.slot)
                        r.finished_at = time.time()
                        if r.stream is not None:
                            r.stream.put(None)
                        r.done.set()
                return len(group)
        limit = 1 if self.active else self.max_batch
        n = 0
        while len(self.active) < self.max_batch and n < limit:
            if not self.cache.has_free() and not self._prefix:
                break  # no KV slot: leave requests queued — decode frees
                # slots as sequences finish (an unconditional loop here
                # re-took requeued requests forever and starved decode)
            try:
                req = self.pending.get_nowait()
            except queue.Empty:
                break
            self._start_request(req)
            n += 1
            if self._chunking is not None:
                break  # the long prompt continues next iteration
        return n

    PREFILL_GROUP = int(os.environ.get("KF_PREFILL_GROUP", "4"))

    def _take_prefill_group(self):
        """Pop up to PREFILL_GROUP pending requests whose EFFECTIVE prompt
        length matches the head request's; requeue mismatches (slight
        reorder — a scheduler decision, not a protocol one). Returns []
        when batching does not apply (it needs >=2 same-length prompts,
        free slots, and an inline-size prompt)."""
        if self.PREFILL_GROUP < 2:
            return []

        def eff_len(r):
            return len(r.prompt[-max(1, self.cache.smax
                                     - r.max_new_tokens - 1):])

        taken, back = [], []
        while len(taken) < self.PREFILL_GROUP:
            try:
                r = self.pending.get_nowait()
            except queue.Empty:
                break
            if not taken:
                taken.append(r)
                continue
            if eff_len(r) == eff_len(taken[0]):
                taken.append(r)
            else:
                back.append(r)
        for r in back:
            self.pending.put(r)
        if len(taken) < 2 or eff_len(taken[0]) > self.PREFILL_CHUNK:
            for r in taken:
                self.pending.put(r)
            return []
        group = []
        for r in taken:
            slot = self._alloc_slot()
            if slot is None:
                self.pending.put(r)
                continue
            r.slot = slot
            group.append(r)
        return group

    def _alloc_slot(self):
        """cache.alloc, evicting the oldest prefix-cache donation if the
        pool is exhausted."""
        slot = self.cache.alloc()
        if slot is None and self._prefix:
            oldest = next(iter(self._prefix))
            self.cache.free(self._prefix.pop(oldest))
            slot = self.cache.alloc()
        return slot

    def _finish_slot(self, req: Request):
        """Free a finished request's slot — or donate it to the prefix
        cache when there is room and the history is worth caching."""
        if (self.prefix_slots > 0 and len(req.prompt) >= 16
                and len(self._prefix) < self.prefix_slots):
            key = tuple(req.prompt[-max(1, self.cache.smax
                                        - req.max_new_tokens - 1):]
                        ) + tuple(req.generated)
            if key not in self._prefix:
                self._prefix[key] = req.slot
                return
        self.cache.free(req.slot)

    def _prefix_hit(self, prompt):
        """Longest cached history sharing a >=16-token prefix with
        `prompt` (capped so >=1 prompt token remains to prefill)."""
        best_len, best_key = 0, None
        limit = len(prompt) - 1
        for key in self._prefix:
            n = 0
            for a, b in zip(key, prompt):
                if a != b or n >= limit:
                    break
                n += 1
            if n > best_len:
                best_len, best_key = n, key
        if best_len >= 16:
            return best_key, best_len
        return None, 0

    def _start_request(self, req: Request):
        slot = self._alloc_slot()
        if slot is None:
            # no slot free: push back and decode on (slots free as seqs end)
            self.pending.put(req)
            return
        req.slot = slot
        keep = max(1, self.cache.smax - req.max_new_tokens - 1)
        prompt = req.prompt[-keep:]
        key, hit = self._prefix_hit(prompt)
        if hit:
            self.cache.copy_rows(self._prefix[key], slot, hit)
            # re-insert as most recently used
            self._prefix[key] = self._prefix.pop(key)
            self.stats["prefix_hits"] = self.stats.get("prefix_hits", 0) + 1
            self.stats["prefix_tokens_reused"] = (
                self.stats.get("prefix_tokens_reused", 0) + hit)
            # prefill only the tail via the rectangular-causal machinery
            self._chunking = {"req": req, "prompt": prompt, "pos": hit,
                              "start": hit}
            self._advance_chunk()
            return
        if self.active and len(prompt) > self.PREFILL_CHUNK:
            # chunked prefill: first chunk now, rest interleaved with decode
            self._chunking = {"req": req, "prompt": prompt, "pos": 0}
            self._advance_chunk()
            return
        try:
            self._prefill(req)
            self.active.append(req)
        except Exception as e:  # pragma: no cover
            import traceback
            traceback.print_exc()
            req.error = f"{type(e).__name__}: {e}"
            self.cache.free(slot)
            req.finished_at = time.time()
            if req.stream is not None:
                req.stream.put(None)
            req.done.set()

    @torch.no_grad()
    def _advance_chunk(self):
        st = self._chunking
        req, prompt, pos = st["req"], st["prompt"], st["pos"]
        end = min(pos + self.PREFILL_CHUNK, len(prompt))
        try:
            x = self._prefill_chunk(req, prompt[pos:end], pos)
            st["pos"] = end
            if end == len(prompt):  # final chunk: sample the first token
                x = self.model.final_norm(x[:, -1:])
                logits = torch.nn.functional.linear(
                    x, self.model.lm_head.weight)
                tok = self._sample(logits[0, -1], req.temperature)
                req.pos = len(prompt)
                req.generated.append(tok)
                req.first_token_at = time.time()
                req.token_times.append(req.first_token_at)
                if req.stream is not None:
                    req.stream.put(tok)
                self.stats["prefill_tokens"] += (len(prompt)
                                                  - st.get("start", 0))
                self.active.append(req)
                self._chunking = None
        except Exception as e:  # pragma: no cover
            import traceback
            traceback.print_exc()
            req.error = f"{type(e).__name__}: {e}"
            self.cache.free(req.slot)
            req.finished_at = time.time()
            if req.stream is not None:
                req.stream.put(None)
            req.done.set()
            self._chunking = None

    def _prefill_chunk(self, req: Request, chunk, pos: int):
        """One CHUNK-token slice through all layers: cache rows
        [pos, pos+C) fill and the chunk attends the whole prefix via the
        rectangular-causal kernel."""
        cfg = self.model.cfg
        C = len(chunk)
        import torch.nn.functional as F
        tokens = torch.tensor([chunk], dtype=torch.int64, device=self.device)
        x = self.model.embed(tokens)
        cos, sin = self.model.rope_cos, self.model.rope_sin
        for li, layer in enumerate(self.model.layers):
            qkv = F.linear(layer.attn_norm(x), layer.wqkv.weight)
            q, k, v = qkv.split([cfg.n_heads * cfg.head_dim,
                                 cfg.n_kv_heads * cfg.head_dim,
                                 cfg.n_kv_heads * cfg.head_dim], dim=-1)
            q = q.view(1, C, cfg.n_heads, cfg.head_dim)
            k = k.view(1, C, cfg.n_kv_heads, cfg.head_dim)
            v = v.view(1, C, cfg.n_kv_heads, cfg.head_dim)
            q, k = ops.rope(q, k, cos, sin, pos)
            self.cache.k[li][req.slot, pos:pos + C] = k[0]
            self.cache.v[li][req.slot, pos:pos + C] = v[0]
            kv_k = self.cache.k[li][req.slot:req.slot + 1, :pos + C]
            kv_v = self.cache.v[li][req.slot:req.slot + 1, :pos + C]
            o = ops.flash_attention_rect(q, kv_k, kv_v, q_offset=pos,
                                         scale=cfg.head_dim ** -0.5)
            o = layer.wo(o.reshape(1, C, cfg.n_heads * cfg.head_dim))
            x = x + o
            if layer.moe is not None:
                x = x + layer.moe(layer.mlp_norm(x))
            else:
                g, u = F.linear(layer.mlp_norm(x), layer.w13.weight).split(
                    [cfg.ffn_dim, cfg.ffn_dim], dim=-1)
                x = x + layer.w2(F.silu(g) * u)
        return x

    @torch.no_grad()
    def _prefill(self, req: Request):
        self._prefill_many([req])

    @torch.no_grad()
    def _prefill_many(self, reqs):
        """Run same-length prompts through the model as one batch, filling
        each request's cache slot and producing its first token."""
        keep = [max(1, self.cache.smax - r.max_new_tokens - 1) for r in reqs]
        prompts = [r.prompt[-k:] for r, k in zip(reqs, keep)]
        S = len(prompts[0])
        tokens = torch.tensor(prompts, dtype=torch.int64, device=self.device)
        x = self.model.embed(tokens)
        cos, sin = self.model.rope_cos, self.model.rope_sin
        slots = [r.slot for r in reqs]
        for li, layer in enumerate(self.model.layers):
            x = self._layer_prefill(layer, li, x, cos, sin, slots, S)
        x = self.model.final_norm(x[:, -1:])
        logits = torch.nn.functional.linear(x, self.model.lm_head.weight)
        now = time.time()
        for i, r in enumerate(reqs):
            tok = self._sample(logits[i, -1], r.temperature)
            r.pos = S
            r.generated.append(tok)
            r.first_token_at = now
            r.token_times.append(now)
            if r.stream is not None:
                r.stream.put(tok)
            self.stats["prefill_tokens"] += S

    def _layer_prefill(self, layer, li, x, cos, sin, slots, S):
        cfg = self.model.cfg
        B = len(slots)
        import torch.nn.functional as F
        qkv = F.linear(layer.attn_norm(x), layer.wqkv.weight)
        q, k, v = qkv.split([cfg.n_heads * cfg.head_dim,
                             cfg.n_kv_heads * cfg.head_dim,
                             cfg.n_kv_heads * cfg.head_dim], dim=-1)
        q = q.view(B, S, cfg.n_heads, cfg.head_dim)
        k = k.view(B, S, cfg.n_kv_heads, cfg.head_dim)
        v = v.view(B, S, cfg.n_kv_heads, cfg.head_dim)
        q, k = ops.rope(q, k, cos, sin, 0)
        for gi, slot in enumerate(slots):
            self.cache.k[li][slot, :S] = k[gi]
            self.cache.v[li][slot, :S] = v[gi]
        o = ops.flash_attention(q, k, v, causal=True)
        o = layer.wo(o.reshape(B, S, cfg.n_heads * cfg.head_dim))
        x = x + o
        if layer.moe is not None:  # MoE predictor (dense experts, EP=1)
            return x + layer.moe(layer.mlp_norm(x))
        g, u = torch.nn.functional.linear(
            layer.mlp_norm(x), layer.w13.weight).split(
                [cfg.ffn_dim, cfg.ffn_dim], dim=-1)
        return x + layer.w2(torch.nn.functional.silu(g) * u)

    @torch.no_grad()
    def _decode_forward(self, tokens, positions, slots, lens):
        """Batched single-token forward over the KV cache -> next tokens.
        Pure function of the given (static, for graph capture) tensors."""
        cfg = self.model.cfg
        import torch.nn.functional as F
        N = tokens.shape[0]
        # decode GEMMs: KF_SKINNY routes the GEMV-shaped linears through
        # the weight-streaming kernel (ops.skinny_linear) — "small" (the
        # shapes it wins at M<=16: qkv N6144/K4096, wo N4096/K4096 —
        # profiles/r02_skinny_gemm.md), "all", or "off" (hipBLASLt
        # everywhere). The K14336 / N28672 / lm_head shapes stay on
        # hipBLASLt under "small": measured losses there.
        mode = os.environ.get("KF_SKINNY", "small")
        sel = {"small": {"qkv", "wo", "w13", "w2"}, "off": set(),
               "all": {"qkv", "wo", "w13", "w2", "lm"}}.get(
                   mode, set(mode.split(",")))
        qw = self._qw if self.quant else None

        def _lin(name, wname):
            if qw is not None and name in sel:
                return lambda li, t, w, rms=None: ops.skinny_linear_q8(
                    t, *qw[li][wname], rms=rms)
            if name in sel:
                return lambda li, t, w, rms=None: ops.skinny_linear(
                    t, w, rms=rms)
            return (lambda li, t, w, rms=None:
                    F.linear(rms[0](t) if rms else t, w))
        lin_qkv = _lin("qkv", "wqkv")
        lin_w13 = _lin("w13", "w13")
        if "lm" in sel and qw is not None:
            lin_lm = lambda t, w: ops.skinny_linear_q8(t, *self._q_lm)
        elif "lm" in sel:
            lin_lm = ops.skinny_linear
        else:
            lin_lm = F.linear

        def _lin_res(name, wname, fuse_swiglu=False):
            if qw is not None and name in sel:
                return lambda li, t, w, r: ops.skinny_linear_q8(
                    t, *qw[li][wname], residual=r,
                    fuse_swiglu=fuse_swiglu)
            if name in sel:
                return lambda li, t, w, r: ops.skinny_linear(
                    t, w, residual=r, fuse_swiglu=fuse_swiglu)
            return lambda li, t, w, r: F.linear(t, w) + r
        lin_wo = _lin_res("wo", "wo")
        lin_w2 = _lin_res("w2", "w2")
        x = self.model.embed(tokens)  # [N,1,H]
        cos, sin = self.model.rope_cos, self.model.rope_sin
        for li, layer in enumerate(self.model.layers):
            # NOTE: fusing the RMSNorms into the GEMM prologue (rms=...)
            # measured WORSE (replay 4.1 -> 5.8 ms): the per-block rstd
            # pass stalls the weight stream behind 16 LLC row reads per
            # block. Separate norm kernels stay (r02_decode_anatomy.md).
            qkv = lin_qkv(li, layer.attn_norm(x), layer.wqkv.weight)
            # fused RoPE + cache scatter straight off the QKV projection
            q = ops.decode_rope_store(qkv, self.cache.k[li],
                                      self.cache.v[li], cos, sin, slots,
                                      positions, cfg.n_heads,
                                      cfg.n_kv_heads)
            o = ops.attention_decode(q, self.cache.k[li],
                                     self.cache.v[li], slots, lens)
            x = lin_wo(li, o.reshape(N, 1, cfg.n_heads * cfg.head_dim),
                       layer.wo.weight, x)  # residual fused in epilogue
            if layer.moe is not None:
                x = x + layer.moe.decode_dense(layer.mlp_norm(x))
            else:
                # NOTE: fusing swiglu into the w2 fragment loader
                # (fuse_swiglu=True) measured WORSE (replay 4.3 -> 5.0,
                # quant 3.5 -> 4.6): the silu exp chain lands on the
                # MFMA dependency path and breaks the load pipeline.
                # The standalone swiglu kernel stays.
                y = ops.swiglu(lin_w13(li, layer.mlp_norm(x),
                                       layer.w13.weight))
                x = lin_w2(li, y, layer.w2.weight, x)
        x = self.model.final_norm(x)
        logits = lin_lm(x, self.model.lm_head.weight)  # [N,1,V]
        return logits[:, -1]  # [N,V] (sampling happens outside the graph)

    def _graph_for(self, bucket: int):
        """Capture (once) and return the decode graph for a batch bucket."""
        if bucket in self._graphs:
            return self._graphs[bucket]
        dev = self.device
        static = {
            "tokens": torch.zeros(bucket, 1, dtype=torch.int64, device=dev),
            "positions": torch.zeros(bucket, dtype=torch.int64, device=dev),
            "slots": torch.full((bucket,), self._pad_slot, dtype=torch.int32,
                                device=dev),
            "lens": torch.ones(bucket, dtype=torch.int32, device=dev),
        }
        # warm-up on a side stream, then capture
        s = torch.cuda.Stream(device=dev)
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                out = self._decode_forward(**static)
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            static["logits"] = self._decode_forward(
                static["tokens"], static["positions"], static["slots"],
                static["lens"])
        self._graphs[bucket] = (graph, static)
        return self._graphs[bucket]

    @torch.no_grad()
    def _decode_step(self):
        """One token for every active sequence, batched (graph replay on
        GPU; eager on CPU)."""
        acts = self.active
        N = len(acts)
        tokens = torch.tensor([[r.generated[-1]] for r in acts],
                              dtype=torch.int64, device=self.device)
        positions = torch.tensor([r.pos for r in acts], dtype=torch.int64,
                                 device=self.device)
        slots = torch.tensor([r.slot for r in acts], dtype=torch.int32,
                             device=self.device)
        lens = torch.tensor([r.pos + 1 for r in acts], dtype=torch.int32,
                            device=self.device)
        if self.use_graphs:
            bucket = 1
            while bucket < N:
                bucket *= 2
            bucket = min(bucket, self.max_batch)
            graph, static = self._graph_for(bucket)
            static["tokens"][:N].copy_(tokens)
            static["positions"][:N].copy_(positions)
            static["slots"][:N].copy_(slots)
            static["lens"][:N].copy_(lens)
            if bucket > N:  # park padding rows on the reserved slot
                static["tokens"][N:].zero_()
                static["positions"][N:].zero_()
                static["slots"][N:].fill_(self._pad_slot)
                static["lens"][N:].fill_(1)
            graph.replay()
            self.stats["graph_replays"] += 1
            logits = static["logits"][:N]
            toks = self._sample_batch(logits, acts)
        else:
            logits = self._decode_forward(tokens, positions, slots, lens)
            toks = self._sample_batch(logits, acts)
        still = []
        now = time.time()
        for i, r in enumerate(acts):
            r.pos += 1
            r.generated.append(int(toks[i]))
            r.token_times.append(now)
            if r.stream is not None:
                r.stream.put(int(toks[i]))
            self.stats["tokens_out"] += 1
            if (len(r.generated) >= r.max_new_tokens
                    or r.pos + 1 >= self.cache.smax):
                r.finished_at = now
                self._finish_slot(r)
                self.stats["completed"] += 1
                if r.stream is not None:
                    r.stream.put(None)
                r.done.set()
            else:
                still.append(r)
        self.active = still

    @staticmethod
    def _sample(logits: torch.Tensor, temperature: float) -> int:
        if temperature <= 0:
            return int(logits.argmax().item())
        probs = torch.softmax(logits.float() / temperature, dim=-1)
        return int(torch.multinomial(probs, 1).item())

    @staticmethod
    def _sample_batch(logits: torch.Tensor, acts) -> list:
        """Per-request temperature sampling on [N,V] logits (outside the
        captured graph so RNG stays ordinary)."""
        temps = [r.temperature for r in acts]
        if all(t <= 0 for t in temps):
            return logits.argmax(-1).tolist()
        t = torch.tensor([max(tt, 1e-6) if tt > 0 else 1.0 for tt in temps],
                         device=logits.device).unsqueeze(1)
        probs = torch.softmax(logits.float() / t, dim=-1)
        sampled = torch.multinomial(probs, 1).squeeze(1)
        greedy = logits.argmax(-1)
        pick = torch.tensor([tt > 0 for tt in temps], device=logits.device)
        return torch.where(pick, sampled, greedy).tolist()
