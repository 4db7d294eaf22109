"""GPU numerics tests: every CDNA4 HIP kernel vs the fp32 torch reference.

Per the project test strategy (SURVEY.md §4 rebuild mapping): "numerics
tests for a HIP kernel compare it against a plain PyTorch fp32 reference of
the same op". Tolerances account for bf16 storage (rel ~1e-2).
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from kubeflow_amd import ops
from kubeflow_amd.ops import reference as R


def _relerr(a, b):
    a, b = a.float(), b.float()
    return ((a - b).norm() / (b.norm() + 1e-12)).item()


@pytest.fixture(autouse=True)
def _require_native():
    from kubeflow_amd.ops import _backend
    _backend.require()  # fail loudly if .so missing on the GPU box


# ---------------------------------------------------------------- RMSNorm

@pytest.mark.parametrize("rows,cols", [(128, 4096), (1024, 768), (8, 256)])
def test_rmsnorm_fwd(rows, cols, gpu_device):
    torch.manual_seed(0)
    x = torch.randn(rows, cols, device=gpu_device, dtype=torch.bfloat16)
    w = torch.randn(cols, device=gpu_device, dtype=torch.bfloat16)
    y = ops.rms_norm(x, w)
    ref = R.rms_norm(x.float().cpu(), w.float().cpu()).to(torch.bfloat16)
    assert _relerr(y.cpu(), ref) < 2e-2


def test_rmsnorm_bwd(gpu_device):
    torch.manual_seed(0)
    rows, cols = 512, 1024
    x = torch.randn(rows, cols, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(cols, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.rms_norm(x, w)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().float().cpu().requires_grad_(True)
    yr = R.rms_norm(xr, wr)
    yr.backward(dy.float().cpu())

    assert _relerr(x.grad.cpu(), xr.grad) < 3e-2
    assert _relerr(w.grad.cpu(), wr.grad) < 3e-2


# ------------------------------------------------------------------- RoPE

def test_rope_fwd_bwd(gpu_device):
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 2, 128, 4, 2, 128
    q = torch.randn(B, S, Hq, D, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    cos, sin = ops.rope_cos_sin(S, D, device=gpu_device)
    q2, k2 = ops.rope(q, k, cos, sin)

    qr = q.detach().float().cpu().requires_grad_(True)
    kr = k.detach().float().cpu().requires_grad_(True)
    ccpu, scpu = ops.rope_cos_sin(S, D)
    q2r = R.rope_apply(qr, ccpu, scpu)
    k2r = R.rope_apply(kr, ccpu, scpu)
    assert _relerr(q2.cpu(), q2r) < 2e-2
    assert _relerr(k2.cpu(), k2r) < 2e-2

    dq2 = torch.randn_like(q2)
    dk2 = torch.randn_like(k2)
    qg = torch.autograd.grad(
        (q2r * dq2.float().cpu()).sum() + (k2r * dk2.float().cpu()).sum(),
        [qr, kr])
    ((q2 * dq2).sum() + (k2 * dk2).sum()).backward()
    assert _relerr(q.grad.cpu(), qg[0]) < 2e-2
    assert _relerr(k.grad.cpu(), qg[1]) < 2e-2


# -------------------------------------------------------- Flash attention

@pytest.mark.parametrize("B,S,Hq,Hkv,causal", [
    (2, 128, 4, 1, True),
    (1, 256, 8, 2, True),
    (2, 128, 4, 4, False),
    (1, 512, 32, 8, True),   # llama-3 8B head config
])
def test_attention_fwd(B, S, Hq, Hkv, causal, gpu_device):
    torch.manual_seed(0)
    D = 128
    q = torch.randn(B, S, Hq, D, device=gpu_device, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=gpu_device, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=gpu_device, dtype=torch.bfloat16)
    o = ops.flash_attention(q, k, v, causal=causal)
    ref = R.sdpa(q.float().cpu().transpose(1, 2),
                 k.float().cpu().transpose(1, 2),
                 v.float().cpu().transpose(1, 2),
                 causal=causal).transpose(1, 2)
    assert _relerr(o.cpu(), ref) < 2e-2


def test_attention_bwd(gpu_device):
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 1, 256, 4, 2, 128
    q = torch.randn(B, S, Hq, D, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hkv, D, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    o = ops.flash_attention(q, k, v, causal=True)
    do = torch.randn_like(o)
    o.backward(do)

    qr = q.detach().float().cpu().transpose(1, 2).requires_grad_(True)
    kr = k.detach().float().cpu().transpose(1, 2).requires_grad_(True)
    vr = v.detach().float().cpu().transpose(1, 2).requires_grad_(True)
    orf = R.sdpa(qr, kr, vr, causal=True, scale=D ** -0.5)
    orf.backward(do.float().cpu().transpose(1, 2))

    assert _relerr(q.grad.cpu(), qr.grad.transpose(1, 2)) < 4e-2
    assert _relerr(k.grad.cpu(), kr.grad.transpose(1, 2)) < 4e-2
    assert _relerr(v.grad.cpu(), vr.grad.transpose(1, 2)) < 4e-2


# ---------------------------------------------------------- Cross entropy

@pytest.mark.parametrize("T,V", [(64, 1000), (256, 128256)])
def test_cross_entropy(T, V, gpu_device):
    torch.manual_seed(0)
    logits = torch.randn(T, V, device=gpu_device, dtype=torch.bfloat16,
                         requires_grad=True)
    targets = torch.randint(0, V, (T,), device=gpu_device)
    targets[::7] = -100
    loss = ops.cross_entropy(logits, targets)
    loss.backward()

    lr = logits.detach().float().cpu().requires_grad_(True)
    ref = R.softmax_cross_entropy(lr, targets.cpu())
    ref.backward()
    assert abs(loss.item() - ref.item()) / abs(ref.item()) < 1e-2
    assert _relerr(logits.grad.cpu(), lr.grad) < 2e-2


# ------------------------------------------------------------ Fused AdamW

def test_fused_adamw_matches_reference(gpu_device):
    torch.manual_seed(0)
    n = 4096 + 64
    p32 = torch.randn(n, device=gpu_device)
    p16 = p32.to(torch.bfloat16)
    m = torch.zeros_like(p32)
    v = torch.zeros_like(p32)
    mask = (torch.rand(n, device=gpu_device) > 0.5).float()

    p32r = p32.cpu().clone()
    mr = torch.zeros(n)
    vr = torch.zeros(n)

    for step in range(1, 4):
        g16 = torch.randn(n, device=gpu_device).to(torch.bfloat16)
        ops.fused_adamw(p16, p32, g16, m, v, mask, 1e-2, 0.9, 0.95, 1e-8,
                        0.1, step)
        # reference with same masked decay
        g32 = g16.float().cpu()
        mr.mul_(0.9).add_(g32, alpha=0.1)
        vr.mul_(0.95).addcmul_(g32, g32, value=0.05)
        bc1 = 1 - 0.9 ** step
        bc2 = 1 - 0.95 ** step
        denom = (vr / bc2).sqrt().add_(1e-8)
        p32r.mul_(1 - 1e-2 * 0.1 * mask.cpu())
        p32r.addcdiv_(mr / bc1, denom, value=-1e-2)
    assert _relerr(p32.cpu(), p32r) < 1e-4
    assert _relerr(p16.float().cpu(), p32r) < 1e-2


# ------------------------------------------------------- End-to-end train

def test_llama_tiny_train_step(gpu_device):
    from kubeflow_amd.models import build_model
    from kubeflow_amd.runtime import Trainer, TrainConfig
    torch.manual_seed(0)
    model = build_model("llama-tiny", device=gpu_device)
    tr = Trainer(model, TrainConfig(lr=1e-3, warmup_steps=2))
    tokens = torch.randint(0, model.cfg.vocab_size, (2, 128), device=gpu_device)
    targets = torch.randint(0, model.cfg.vocab_size, (2, 128), device=gpu_device)
    losses = [float(tr.step(tokens, targets)) for _ in range(10)]
    assert all(l == l for l in losses), losses
    assert losses[-1] < losses[0], losses


# -------------------------------------------------------------- LayerNorm

@pytest.mark.parametrize("rows,cols", [(512, 768), (64, 256)])
def test_layernorm_fwd_bwd(rows, cols, gpu_device):
    torch.manual_seed(0)
    x = torch.randn(rows, cols, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(cols, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.randn(cols, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.layer_norm(x, w, b, eps=1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().cpu().requires_grad_(True)
    wr = w.detach().float().cpu().requires_grad_(True)
    br = b.detach().float().cpu().requires_grad_(True)
    yr = R.layer_norm(xr, wr, br, eps=1e-6)
    yr.backward(dy.float().cpu())
    assert _relerr(y.cpu(), yr) < 2e-2
    assert _relerr(x.grad.cpu(), xr.grad) < 5e-2
    assert _relerr(w.grad.cpu(), wr.grad) < 3e-2
    assert _relerr(b.grad.cpu(), br.grad) < 3e-2


# ------------------------------------------------------- Decode attention

def test_attention_decode_matches_sdpa(gpu_device):
    torch.manual_seed(0)
    N, Hq, Hkv, D, SMAX = 4, 8, 2, 128, 256
    kcache = torch.randn(8, SMAX, Hkv, D, device=gpu_device,
                         dtype=torch.bfloat16)
    vcache = torch.randn(8, SMAX, Hkv, D, device=gpu_device,
                         dtype=torch.bfloat16)
    q = torch.randn(N, Hq, D, device=gpu_device, dtype=torch.bfloat16)
    slots = torch.tensor([5, 0, 3, 7], dtype=torch.int32, device=gpu_device)
    lens = torch.tensor([200, 1, 77, 256], dtype=torch.int32,
                        device=gpu_device)
    out = ops.attention_decode(q, kcache, vcache, slots, lens)
    # reference per sequence
    for i in range(N):
        L, s = int(lens[i]), int(slots[i])
        ref = R.sdpa(q[i].view(1, 1, Hq, D).float().cpu().transpose(1, 2),
                     kcache[s, :L].unsqueeze(0).float().cpu().transpose(1, 2),
                     vcache[s, :L].unsqueeze(0).float().cpu().transpose(1, 2),
                     causal=False, scale=D ** -0.5)
        assert _relerr(out[i].cpu(), ref.transpose(1, 2).reshape(Hq, D)) < 2e-2, i


def test_serving_engine_gpu_decode_consistency(gpu_device):
    """Engine incremental decode must match full forward on GPU (llama-tiny,
    all HIP kernels on the path)."""
    from kubeflow_amd.runtime.serving import InferenceEngine, Request
    torch.manual_seed(0)
    eng = InferenceEngine("llama-tiny", device=gpu_device, max_slots=2,
                          smax=256, max_batch=2)
    prompt = [3, 14, 15, 9, 2, 6, 1, 2]
    req = Request(rid="t", prompt=list(prompt), max_new_tokens=3)
    req.slot = eng.cache.alloc()
    eng._prefill(req)
    eng.active = [req]
    eng._decode_step()
    eng.active = [req]
    with torch.no_grad():
        full = eng.model(torch.tensor([prompt], device=gpu_device))
        t1 = int(full[0, -1].argmax())
        full2 = eng.model(torch.tensor([prompt + [t1]], device=gpu_device))
        t2 = int(full2[0, -1].argmax())
    assert req.generated[0] == t1
    assert req.generated[1] == t2


# ------------------------------------------------------------------ SwiGLU

def test_swiglu_fwd_bwd(gpu_device):
    torch.manual_seed(0)
    T, F = 256, 512
    h = torch.randn(T, 2 * F, device=gpu_device, dtype=torch.bfloat16,
                    requires_grad=True)
    y = ops.swiglu(h)
    dy = torch.randn_like(y)
    y.backward(dy)

    hr = h.detach().float().cpu().requires_grad_(True)
    g, u = hr.chunk(2, dim=-1)
    yr = torch.nn.functional.silu(g) * u
    yr.backward(dy.float().cpu())
    assert _relerr(y.cpu(), yr) < 2e-2
    assert _relerr(h.grad.cpu(), hr.grad) < 3e-2


# ------------------------------------------------- Fused QKV attention path

def test_fused_qkv_attention_matches_composed(gpu_device):
    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 2, 256, 4, 2, 128
    ts = (Hq + 2 * Hkv) * D
    qkv = torch.randn(B, S, ts, device=gpu_device, dtype=torch.bfloat16)
    cos, sin = ops.rope_cos_sin(S, D, device=gpu_device)

    # composed reference on fp32 CPU
    qr, kr, vr = qkv.float().cpu().split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    qr = qr.reshape(B, S, Hq, D)
    kr = kr.reshape(B, S, Hkv, D)
    vr = vr.reshape(B, S, Hkv, D)
    ccpu, scpu = ops.rope_cos_sin(S, D)
    qr2 = R.rope_apply(qr, ccpu, scpu)
    kr2 = R.rope_apply(kr, ccpu, scpu)
    oref = R.sdpa(qr2.transpose(1, 2), kr2.transpose(1, 2),
                  vr.transpose(1, 2), causal=True).transpose(1, 2)

    qkv_in = qkv.clone().requires_grad_(True)
    o = ops.fused_qkv_attention(qkv_in, cos, sin, Hq, Hkv, D)
    assert _relerr(o.cpu(), oref.reshape(B, S, Hq * D)) < 2e-2

    # backward vs autograd through the composed fp32 path
    do = torch.randn_like(o)
    o.backward(do)
    qkv_ref = qkv.float().cpu().requires_grad_(True)
    qc, kc, vc = qkv_ref.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q2 = R.rope_apply(qc.reshape(B, S, Hq, D), ccpu, scpu)
    k2 = R.rope_apply(kc.reshape(B, S, Hkv, D), ccpu, scpu)
    ocomp = R.sdpa(q2.transpose(1, 2), k2.transpose(1, 2),
                   vc.reshape(B, S, Hkv, D).transpose(1, 2),
                   causal=True).transpose(1, 2).reshape(B, S, Hq * D)
    ocomp.backward(do.float().cpu())
    assert _relerr(qkv_in.grad.cpu(), qkv_ref.grad) < 4e-2


def test_llama_tiny_fused_vs_cpu_loss(gpu_device):
    """GPU fused-path loss should track the CPU reference-path loss."""
    from kubeflow_amd.models import build_model
    torch.manual_seed(0)
    mg = build_model("llama-tiny", device=gpu_device)
    mc = build_model("llama-tiny", dtype=torch.float32)
    # same weights
    with torch.no_grad():
        for pc, pg in zip(mc.parameters(), mg.parameters()):
            pg.copy_(pc.to(pg.dtype))
    tokens = torch.randint(0, mg.cfg.vocab_size, (2, 128))
    targets = torch.randint(0, mg.cfg.vocab_size, (2, 128))
    lg = mg(tokens.to(gpu_device), targets.to(gpu_device))
    lc = mc(tokens, targets)
    assert abs(lg.item() - lc.item()) / abs(lc.item()) < 2e-2, \
        (lg.item(), lc.item())


def test_serving_graph_matches_eager(gpu_device, monkeypatch):
    """hipGraph decode replay must produce the same greedy tokens as eager."""
    from kubeflow_amd.runtime.serving import InferenceEngine
    monkeypatch.setenv("KF_SERVE_GRAPH", "1")
    torch.manual_seed(0)
    eng_g = InferenceEngine("llama-tiny", device=gpu_device, max_slots=4,
                            smax=256, max_batch=4)
    assert eng_g.use_graphs
    monkeypatch.setenv("KF_SERVE_GRAPH", "0")
    torch.manual_seed(0)
    eng_e = InferenceEngine("llama-tiny", device=gpu_device, max_slots=4,
                            smax=256, max_batch=4)
    with torch.no_grad():
        for pg, pe in zip(eng_g.model.parameters(), eng_e.model.parameters()):
            pe.copy_(pg)
    eng_g.start()
    eng_e.start()
    try:
        rg = eng_g.generate([5, 9, 2, 7], max_new_tokens=12, timeout=120)
        re_ = eng_e.generate([5, 9, 2, 7], max_new_tokens=12, timeout=120)
        assert rg.error == "" and re_.error == ""
        assert rg.generated == re_.generated, (rg.generated, re_.generated)
        assert eng_g.stats["graph_replays"] > 0
    finally:
        eng_g.stop()
        eng_e.stop()


@pytest.mark.gpu
def test_attention_rect_gpu_parity():
    """kf_attn_fwd4_rect (chunked prefill) vs the fp32 reference at a
    non-aligned Skv with a q offset."""
    torch.manual_seed(11)
    dev = torch.device("cuda", 0)
    B, Hq, Hkv, D = 1, 8, 2, 128
    off, C = 640, 512           # chunk of 512 at offset 640
    Skv = off + C               # 1152 (non-64-multiple + 64*18 = fine)
    # kv buffer padded to the next 64-multiple capacity (cache-slab shape)
    cap = (Skv + 63) // 64 * 64
    q = torch.randn(B, C, Hq, D, device=dev, dtype=torch.bfloat16)
    kbuf = torch.randn(B, cap, Hkv, D, device=dev, dtype=torch.bfloat16)
    vbuf = torch.randn(B, cap, Hkv, D, device=dev, dtype=torch.bfloat16)
    from kubeflow_amd import ops
    o = ops.flash_attention_rect(q, kbuf[:, :Skv], vbuf[:, :Skv],
                                 q_offset=off)
    # reference on CPU in fp32 (rect mask via sdpa tril(Sk-Sq))
    from kubeflow_amd.ops import reference as R
    want = R.sdpa(q.float().cpu().transpose(1, 2),
                  kbuf[:, :Skv].float().cpu().transpose(1, 2),
                  vbuf[:, :Skv].float().cpu().transpose(1, 2),
                  causal=True).transpose(1, 2)
    err = ((o.cpu().float() - want).norm() / want.norm()).item()
    assert err < 2e-2, err

    # odd Skv (not a 64-multiple): padding rows must stay masked
    Skv2 = off + C - 37
    o2 = ops.flash_attention_rect(q[:, :256], kbuf[:, :Skv2],
                                  vbuf[:, :Skv2], q_offset=Skv2 - 256)
    want2 = R.sdpa(q[:, :256].float().cpu().transpose(1, 2),
                   kbuf[:, :Skv2].float().cpu().transpose(1, 2),
                   vbuf[:, :Skv2].float().cpu().transpose(1, 2),
                   causal=True).transpose(1, 2)
    err2 = ((o2.cpu().float() - want2).norm() / want2.norm()).item()
    assert err2 < 2e-2, err2


@pytest.mark.gpu
def test_skinny_gemm_parity():
    """Decode-batch linear kernel vs F.linear at the serving shapes."""
    torch.manual_seed(2)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    for M, N, K in ((16, 6144, 4096), (1, 4096, 4096), (5, 28672, 4096),
                    (16, 4096, 14336), (16, 128256, 4096),
                    (32, 6144, 4096), (24, 4096, 14336)):
        x = torch.randn(M, 1, K, device=dev, dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        got = ops.skinny_linear(x, w)
        want = torch.nn.functional.linear(x, w)
        err = ((got.float() - want.float()).norm() /
               want.float().norm().clamp(min=1e-6)).item()
        assert err < 2e-2, (M, N, K, err)


@pytest.mark.gpu
def test_kv_store_parity():
    """Fused decode cache scatter vs advanced indexing."""
    torch.manual_seed(3)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    SLOTS, SMAX, Hkv, D, N = 8, 64, 8, 128, 5
    ck = torch.randn(SLOTS, SMAX, Hkv, D, device=dev, dtype=torch.bfloat16)
    cv = torch.randn_like(ck)
    ck2, cv2 = ck.clone(), cv.clone()
    k = torch.randn(N, 1, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn_like(k)
    slots = torch.tensor([7, 0, 3, 3, 5], dtype=torch.int32, device=dev)
    positions = torch.tensor([0, 63, 10, 11, 32], dtype=torch.int64,
                             device=dev)
    ops.kv_store(ck, cv, k, v, slots, positions)
    ck2[slots.long(), positions] = k[:, 0]
    cv2[slots.long(), positions] = v[:, 0]
    assert torch.equal(ck, ck2)
    assert torch.equal(cv, cv2)


@pytest.mark.gpu
def test_decode_rope_store_parity():
    """Fused RoPE+scatter vs split/rope/kv_store reference composition."""
    torch.manual_seed(4)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    SLOTS, SMAX, Hq, Hkv, D, N = 6, 64, 8, 2, 128, 5
    ck = torch.zeros(SLOTS, SMAX, Hkv, D, device=dev, dtype=torch.bfloat16)
    cv = torch.zeros_like(ck)
    ck2, cv2 = ck.clone(), cv.clone()
    qkv = torch.randn(N, 1, (Hq + 2 * Hkv) * D, device=dev,
                      dtype=torch.bfloat16)
    cos, sin = ops.rope_cos_sin(SMAX, D, device=dev)
    slots = torch.tensor([5, 0, 2, 2, 4], dtype=torch.int32, device=dev)
    positions = torch.tensor([0, 63, 10, 11, 32], dtype=torch.int64,
                             device=dev)
    q = ops.decode_rope_store(qkv, ck, cv, cos, sin, slots, positions,
                              Hq, Hkv)
    # reference composition (native rope kernel + kv_store)
    qq, kk, vv = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    qq = qq.view(N, 1, Hq, D)
    kk = kk.view(N, 1, Hkv, D)
    vv = vv.view(N, 1, Hkv, D)
    qr, kr = ops.rope(qq, kk, cos, sin, positions=positions)
    ops.kv_store(ck2, cv2, kr, vv, slots, positions)
    assert torch.equal(q, qr.reshape(N, Hq, D))
    assert torch.equal(ck, ck2)
    assert torch.equal(cv, cv2)


@pytest.mark.gpu
def test_skinny_residual_parity():
    torch.manual_seed(5)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    M, N, K = 16, 4096, 4096
    x = torch.randn(M, 1, K, device=dev, dtype=torch.bfloat16) * 0.5
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
    r = torch.randn(M, 1, N, device=dev, dtype=torch.bfloat16)
    got = ops.skinny_linear(x, w, residual=r)
    want = torch.nn.functional.linear(x.float(), w.float()) + r.float()
    err = ((got.float() - want).norm() / want.norm()).item()
    assert err < 2e-2, err


@pytest.mark.gpu
def test_skinny_q8_parity():
    """W8A16 kernel vs F.linear on the dequantized weights (same math up
    to bf16 rounding), with and without residual, M 16 and 32."""
    torch.manual_seed(6)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    for M, N, K in ((16, 6144, 4096), (32, 4096, 14336), (3, 4096, 4096)):
        x = torch.randn(M, 1, K, device=dev, dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        w8, sc = ops.quantize_fp8_rows(w)
        r = torch.randn(M, 1, N, device=dev, dtype=torch.bfloat16)
        wd = ops.dequantize_fp8_rows(w8, sc, torch.float32)
        got = ops.skinny_linear_q8(x, w8, sc, residual=r)
        want = torch.nn.functional.linear(x.float(), wd) + r.float()
        err = ((got.float() - want).norm() / want.norm()).item()
        assert err < 2e-2, (M, N, K, err)


@pytest.mark.gpu
def test_quantized_decode_logits_close(monkeypatch):
    """W8A16 decode logits stay close to the bf16 engine's (same weights,
    cosine > 0.99) — the quant path changes precision, not semantics."""
    import torch.nn.functional as TF

    from kubeflow_amd.runtime.serving import InferenceEngine, Request

    dev = torch.device("cuda", 0)
    prompt = [3, 14, 15, 9, 2, 6, 1, 2]

    def run(quant):
        if quant:
            monkeypatch.setenv("KF_SERVE_QUANT", "fp8")
        else:
            monkeypatch.delenv("KF_SERVE_QUANT", raising=False)
        torch.manual_seed(0)
        eng = InferenceEngine("llama-tiny", device=dev, max_slots=2,
                              smax=256, max_batch=2)
        req = Request(rid="t", prompt=list(prompt), max_new_tokens=4)
        req.slot = eng.cache.alloc()
        eng._prefill(req)
        tokens = torch.tensor([[req.generated[-1]]], device=dev)
        positions = torch.tensor([req.pos], device=dev)
        slots = torch.tensor([req.slot], dtype=torch.int32, device=dev)
        lens = torch.tensor([req.pos + 1], dtype=torch.int32, device=dev)
        logits = eng._decode_forward(tokens, positions, slots, lens)
        return logits.float().cpu()

    lq = run(True)
    lb = run(False)
    cos = TF.cosine_similarity(lq.flatten(), lb.flatten(), dim=0).item()
    assert cos > 0.99, cos


@pytest.mark.gpu
def test_masked_attention_gpu_parity():
    """masked_attention (rect kernel, q_offset=kv_len) vs fp32 sdpa over
    only the real kv rows — the padded-classifier case, D=128."""
    torch.manual_seed(7)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    B, S, Hq, Hkv, D = 2, 128, 6, 6, 128
    kv_len = 37
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16) * 0.5
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16) * 0.5
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    got = ops.masked_attention(q, k, v, kv_len)
    ref = ops.reference.sdpa(
        q.float().transpose(1, 2), k[:, :kv_len].float().transpose(1, 2),
        v[:, :kv_len].float().transpose(1, 2), causal=False,
    ).transpose(1, 2)
    err = ((got.float() - ref).norm() / ref.norm()).item()
    assert err < 2e-2, err


@pytest.mark.gpu
def test_classifier_engine_gpu_matches_forward():
    """BERT-hd128 classify on GPU (native masked_attention path) ==
    direct full forward at a 128-multiple length."""
    from kubeflow_amd.runtime.serving import InferenceEngine

    torch.manual_seed(10)
    dev = torch.device("cuda", 0)
    eng = InferenceEngine("bert-base-hd128", device=dev, max_batch=4)
    eng.start()
    try:
        prompt = list(range(2, 53))  # odd length -> padded masked path
        r = eng.generate(prompt, timeout=120)
        assert not r.error, r.error
        with torch.no_grad():
            # the unpadded direct GPU forward is impossible (the flash
            # kernel needs S % 128 == 0), so check the class PROBABILITIES
            # against a CPU fp32 clone of the same weights — argmax on
            # random-init logits is too close to compare across dtypes
            import copy
            cpu = copy.deepcopy(eng.model).float().cpu()
            want = torch.softmax(cpu(torch.tensor([prompt]))[0].float(),
                                 dim=-1)
        got = torch.tensor(r.scores)
        assert torch.allclose(got, want, atol=0.05), (got, want)
    finally:
        eng.stop()


@pytest.mark.gpu
def test_skinny_fused_rms_parity():
    """skinny_linear(rms=...) == F.linear(rms_norm(x)) for bf16/fp8."""
    torch.manual_seed(11)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops
    from kubeflow_amd.models.llama import RMSNorm

    for M, N, K in ((16, 4096, 4096), (32, 4096, 14336)):
        x = torch.randn(M, 1, K, device=dev, dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        norm = RMSNorm(K, 1e-5).to(device=dev, dtype=torch.bfloat16)
        with torch.no_grad():
            norm.weight.normal_(1.0, 0.1)
        g32 = norm.weight.detach().float()
        r = torch.randn(M, 1, N, device=dev, dtype=torch.bfloat16)
        got = ops.skinny_linear(x, w, residual=r, rms=(norm, g32))
        xn = ops.reference.rms_norm(x.float(), g32, 1e-5)
        want = torch.nn.functional.linear(xn, w.float()) + r.float()
        err = ((got.float() - want).norm() / want.norm()).item()
        assert err < 3e-2, (M, N, K, err)
        # fp8 path
        w8, sc = ops.quantize_fp8_rows(w)
        got8 = ops.skinny_linear_q8(x, w8, sc, residual=r, rms=(norm, g32))
        wd = ops.dequantize_fp8_rows(w8, sc, torch.float32)
        want8 = torch.nn.functional.linear(xn, wd) + r.float()
        err8 = ((got8.float() - want8).norm() / want8.norm()).item()
        assert err8 < 3e-2, (M, N, K, err8)


@pytest.mark.gpu
def test_skinny_fused_swiglu_parity():
    """skinny_linear(fuse_swiglu=True) on the raw [M, 2K] w13 output ==
    F.linear(swiglu(h)) for bf16 and fp8 weights."""
    torch.manual_seed(12)
    dev = torch.device("cuda", 0)
    from kubeflow_amd import ops

    for M, FFN, H in ((16, 14336, 4096), (32, 1024, 4096)):
        h = torch.randn(M, 1, 2 * FFN, device=dev,
                        dtype=torch.bfloat16) * 0.5
        w2 = torch.randn(H, FFN, device=dev, dtype=torch.bfloat16) * 0.02
        r = torch.randn(M, 1, H, device=dev, dtype=torch.bfloat16)
        got = ops.skinny_linear(h, w2, residual=r, fuse_swiglu=True)
        hf = h.float()
        g, u = hf.split(FFN, dim=-1)
        y = torch.nn.functional.silu(g) * u
        want = torch.nn.functional.linear(y, w2.float()) + r.float()
        err = ((got.float() - want).norm() / want.norm()).item()
        assert err < 3e-2, (M, FFN, err)
        w8, sc = ops.quantize_fp8_rows(w2)
        got8 = ops.skinny_linear_q8(h, w8, sc, residual=r,
                                    fuse_swiglu=True)
        wd = ops.dequantize_fp8_rows(w8, sc, torch.float32)
        want8 = torch.nn.functional.linear(y, wd) + r.float()
        err8 = ((got8.float() - want8).norm() / want8.norm()).item()
        assert err8 < 3e-2, (M, FFN, err8)
