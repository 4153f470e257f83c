"""GPU tests (MI355X): HIP pack/unpack kernel numerics vs fp32 torch
reference, RCCL single-rank collectives, adasum kernels, optimizer path.

These run on the 1-GPU gpurun box; multi-rank semantics are covered by the
CPU multi-process suite (same controller/negotiation code) and the driver's
8-GPU scaling bench.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@pytest.fixture(scope="module")
def hvd():
    import horovod_amd.torch as hvd
    hvd.init()
    torch.cuda.set_device(0)
    yield hvd
    # process-level shutdown handled by atexit


@requires_gpu
def test_allreduce_single_gpu(hvd):
    t = torch.randn(1000, device="cuda")
    out = hvd.allreduce(t, average=False, name="g1")
    assert torch.allclose(out, t)
    from horovod_amd import _core
    assert _core.rccl_used(), "native RCCL path must run on GPU"


@requires_gpu
def test_fused_pack_unpack_numerics(hvd):
    """Grouped allreduce forces the fusion-buffer path: batched_copy_k pack +
    RCCL + unpack.  Single rank => output must equal input exactly."""
    torch.manual_seed(0)
    ts = [torch.randn(n, device="cuda") for n in (1, 17, 1024, 100000, 3)]
    outs = hvd.grouped_allreduce(ts, average=False, name="fuse")
    for t, o in zip(ts, outs):
        assert torch.equal(t, o), (t - o).abs().max()


@requires_gpu
def test_fused_scale_numerics(hvd):
    torch.manual_seed(1)
    ts = [torch.randn(513, device="cuda") for _ in range(3)]
    outs = hvd.grouped_allreduce(ts, average=False, name="fscale",
                                 prescale_factor=2.0, postscale_factor=0.25)
    for t, o in zip(ts, outs):
        assert torch.allclose(o, t * 0.5, rtol=1e-6, atol=1e-6), \
            (o - t * 0.5).abs().max()


@requires_gpu
def test_wire_compression_numerics(hvd):
    """fp32 tensors compressed to bf16/fp16 on the wire inside the pack
    kernel; single rank round-trips through the wire dtype."""
    from horovod_amd.torch.compression import Compression
    torch.manual_seed(2)
    t = torch.randn(4096, device="cuda")
    for comp, tol in ((Compression.fp16, 1e-3), (Compression.bf16, 1e-2)):
        out = hvd.allreduce(t, average=False, compression=comp,
                            name=f"wc{tol}")
        assert torch.allclose(out, t, rtol=tol, atol=tol), \
            (out - t).abs().max()


@requires_gpu
def test_allreduce_dtypes_gpu(hvd):
    for dtype in [torch.float32, torch.float16, torch.bfloat16, torch.int32,
                  torch.int64, torch.uint8]:
        t = (torch.arange(100, device="cuda") % 17).to(dtype)
        out = hvd.allreduce(t, average=False, name=f"gdt{dtype}")
        assert torch.equal(out, t), dtype


@requires_gpu
def test_broadcast_allgather_alltoall_reducescatter_gpu(hvd):
    t = torch.randn(64, 8, device="cuda")
    assert torch.allclose(hvd.broadcast(t, root_rank=0, name="gbc"), t)
    assert torch.allclose(hvd.allgather(t, name="gag"), t)
    out, rs = hvd.alltoall(t, splits=[64], name="ga2a")
    assert torch.allclose(out, t)
    rs_out = hvd.reducescatter(t, op=hvd.Sum, name="grs")
    assert torch.allclose(rs_out, t)


@requires_gpu
def test_adasum_gpu_single(hvd):
    # n=1: no combine; output == input through pack/unpack
    t = torch.randn(2048, device="cuda")
    out = hvd.allreduce(t, op=hvd.Adasum, name="gada")
    assert torch.allclose(out, t, atol=1e-6)


@requires_gpu
def test_adasum_repeat_determinism_gpu(hvd):
    """Repeated Adasum gives bit-identical results (determinism only; the
    numeric golden comparison is test_adasum_kernels_golden_gpu below)."""
    from horovod_amd import _core  # noqa: F401  (ensures lib loaded)
    # exercise via a 1-rank process-set trick is not possible; instead test
    # the CPU golden against the GPU tree by simulating: pack two halves as
    # separate "ranks" is core-internal.  Covered indirectly; here we check
    # determinism of repeated adasum.
    t = torch.randn(512, device="cuda")
    o1 = hvd.allreduce(t, op=hvd.Adasum, name="det1")
    o2 = hvd.allreduce(t, op=hvd.Adasum, name="det2")
    assert torch.equal(o1, o2)


@requires_gpu
def test_optimizer_step_gpu(hvd):
    from horovod_amd.models import resnet50
    torch.manual_seed(3)
    model = resnet50().cuda().to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt,
                                   named_parameters=model.named_parameters())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    data = torch.randn(8, 3, 224, 224, device="cuda").to(
        memory_format=torch.channels_last)
    target = torch.randint(0, 1000, (8,), device="cuda")
    for _ in range(2):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(data), target)
        loss.backward()
        opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@requires_gpu
def test_native_extension_loaded():
    """Fail loudly if the in-tree native extension is not what's loaded."""
    import horovod_amd._core as core
    assert "/horovod_amd/_core.so" in core.__file__, core.__file__


@requires_gpu
def test_fused_sgd_matches_torch(hvd):
    """FusedSGD kernel vs torch.optim.SGD over several steps."""
    from horovod_amd.ops import FusedSGD
    torch.manual_seed(9)
    ref_params = [torch.randn(n, device="cuda", requires_grad=False)
                  for n in (1000, 37, 4096)]
    fused_params = [p.clone() for p in ref_params]
    ref_opt = torch.optim.SGD(ref_params, lr=0.1, momentum=0.9,
                              weight_decay=0.01)
    fused_opt = FusedSGD(fused_params, lr=0.1, momentum=0.9,
                         weight_decay=0.01)
    for step in range(5):
        grads = [torch.randn_like(p) for p in ref_params]
        for p, g in zip(ref_params, grads):
            p.grad = g
        for p, g in zip(fused_params, grads):
            p.grad = g.clone()
        ref_opt.step()
        fused_opt.step()
    torch.cuda.synchronize()
    for rp, fp in zip(ref_params, fused_params):
        assert torch.allclose(rp, fp, rtol=1e-5, atol=1e-6), \
            (rp - fp).abs().max()


@requires_gpu
def test_channels_last_dense_layout(hvd):
    """Dense-but-permuted (channels_last) tensors: raw memory-order
    collectives must round-trip exactly (the layout every ResNet conv
    grad/param has under channels_last training)."""
    t = torch.randn(8, 3, 4, 4, device="cuda").to(
        memory_format=torch.channels_last)
    out = hvd.allreduce(t, average=False, name="cl1")
    assert torch.equal(out, t)
    t2 = t.clone()
    hvd.broadcast_(t2, root_rank=0, name="cl2")
    assert torch.equal(t2, t)
    outs = hvd.grouped_allreduce([t, t.clone()], average=False, name="cl3")
    assert torch.equal(outs[0], t) and torch.equal(outs[1], t)


@requires_gpu
def test_adasum_kernels_golden_gpu(hvd):
    """Drive adasum_dots_k / adasum_scaledadd_k directly and compare against
    the fp64 torch formula (per-tensor coefficient isolation)."""
    from horovod_amd import _core
    torch.manual_seed(11)
    for dtype, tol in ((torch.float32, 1e-6), (torch.bfloat16, 5e-2),
                       (torch.float64, 1e-12)):
        a = [torch.randn(n, device="cuda").to(dtype) for n in (17, 1024, 65537)]
        b = [torch.randn(n, device="cuda").to(dtype) for n in (17, 1024, 65537)]
        expected = []
        for x, y in zip(a, b):
            xd, yd = x.double(), y.double()
            dot = xd.dot(yd)
            na, nb = xd.dot(xd), yd.dot(yd)
            ac = 1 - dot / (2 * na) if na > 0 else 1.0
            bc = 1 - dot / (2 * nb) if nb > 0 else 1.0
            expected.append((ac * xd + bc * yd).to(dtype))
        _core.adasum_combine_(a, b)
        torch.cuda.synchronize()
        for got, exp in zip(a, expected):
            assert torch.allclose(got.float(), exp.float(), rtol=tol,
                                  atol=tol), \
                (dtype, (got.float() - exp.float()).abs().max())


@requires_gpu
def test_fused_bn_relu_matches_torch(hvd):
    """FusedBNReLU / FusedBNAddReLU vs eager BN+ReLU (fwd, bwd, running
    stats), fp32 and bf16, NHWC."""
    from horovod_amd.ops import FusedBNAddReLU, FusedBNReLU
    for dtype, tol in ((torch.float32, 2e-4), (torch.bfloat16, 5e-2)):
        torch.manual_seed(6)
        N, C, H, W = 8, 64, 14, 14
        x = (torch.randn(N, C, H, W, device="cuda")
             .to(memory_format=torch.channels_last).to(dtype)
             .requires_grad_(True))
        xr = x.detach().clone().requires_grad_(True)

        fused = FusedBNReLU(C).cuda().train()
        ref = torch.nn.BatchNorm2d(C).cuda().train()
        ref.load_state_dict(fused.state_dict())

        y = fused(x)
        yr = torch.nn.functional.relu(ref(xr))
        assert torch.allclose(y.float(), yr.float(), rtol=tol, atol=tol), \
            (dtype, (y.float() - yr.float()).abs().max())
        assert torch.allclose(fused.running_mean, ref.running_mean,
                              atol=1e-4)
        assert torch.allclose(fused.running_var, ref.running_var, atol=1e-3)

        g = torch.randn_like(y)
        y.backward(g)
        yr.backward(g)
        assert torch.allclose(x.grad.float(), xr.grad.float(), rtol=tol,
                              atol=tol), \
            (dtype, (x.grad.float() - xr.grad.float()).abs().max())
        assert torch.allclose(fused.weight.grad, ref.weight.grad, rtol=1e-2,
                              atol=1e-2)
        assert torch.allclose(fused.bias.grad, ref.bias.grad, rtol=1e-2,
                              atol=1e-2)

        # Add variant
        a = (torch.randn(N, C, H, W, device="cuda")
             .to(memory_format=torch.channels_last).to(dtype)
             .requires_grad_(True))
        r = (torch.randn(N, C, H, W, device="cuda")
             .to(memory_format=torch.channels_last).to(dtype)
             .requires_grad_(True))
        ar = a.detach().clone().requires_grad_(True)
        rr = r.detach().clone().requires_grad_(True)
        fadd = FusedBNAddReLU(C).cuda().train()
        radd = torch.nn.BatchNorm2d(C).cuda().train()
        radd.load_state_dict(fadd.state_dict())
        ya = fadd(a, r)
        yra = torch.nn.functional.relu(radd(ar) + rr)
        assert torch.allclose(ya.float(), yra.float(), rtol=tol, atol=tol)
        ga = torch.randn_like(ya)
        ya.backward(ga)
        yra.backward(ga)
        # boundary-mask caveat: our single-rounded z = bn(x)+res can flip the
        # relu mask vs torch's doubly-rounded bf16 z on near-zero elements, so
        # low precision admits a tiny fraction of pointwise mismatches
        def mostly_close(u, v):
            diff = (u.float() - v.float()).abs()
            ok = diff <= tol + tol * v.float().abs()
            frac = ok.float().mean().item()
            return frac > (0.999 if dtype != torch.float32 else 0.9999999)
        assert mostly_close(a.grad, ar.grad), \
            (dtype, (a.grad.float() - ar.grad.float()).abs().max())
        assert mostly_close(r.grad, rr.grad)


@requires_gpu
def test_fused_bn_resnet_step(hvd):
    from horovod_amd.models import resnet50
    torch.manual_seed(3)
    model = resnet50(fused_bn=True).cuda().to(
        memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    data = torch.randn(8, 3, 224, 224, device="cuda").to(
        memory_format=torch.channels_last)
    target = torch.randint(0, 1000, (8,), device="cuda")
    for _ in range(2):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = torch.nn.functional.cross_entropy(model(data), target)
        loss.backward()
        opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


@requires_gpu
def test_integer_average_gpu(hvd):
    t = torch.tensor([3, -3, 7], dtype=torch.int64, device="cuda")
    out = hvd.allreduce(t, average=True, name="giavg")
    assert out.tolist() == [3, -3, 7], out  # n=1: sum==self, divisor 1
    out2 = hvd.allreduce(torch.arange(64, device="cuda", dtype=torch.int32),
                         average=True, name="giavg2")
    assert torch.equal(out2, torch.arange(64, device="cuda",
                                          dtype=torch.int32))


@requires_gpu
def test_timeline_gpu_ops(hvd, tmp_path):
    """Timeline + roctx around real RCCL ops must not disturb results."""
    tl = str(tmp_path / "gtl.json")
    hvd.start_timeline(tl)
    t = torch.randn(4096, device="cuda")
    for i in range(5):
        out = hvd.allreduce(t, average=False, name=f"gtl{i}")
    hvd.stop_timeline()
    assert torch.allclose(out, t)
    import json
    data = json.load(open(tl))
    names = {str(e.get("name")) for e in data}
    assert "ALLREDUCE" in names and "NEGOTIATE" in names, names


@requires_gpu
def test_mixed_cpu_gpu_ops(hvd):
    """Interleaved CPU-tensor and GPU-tensor collectives in one session
    (separate fusion classes; both paths live)."""
    for i in range(5):
        c = hvd.allreduce(torch.ones(64) * i, average=False, name=f"mcpu{i}")
        g = hvd.allreduce(torch.ones(64, device="cuda") * i, average=False,
                          name=f"mgpu{i}")
        assert c.sum().item() == 64.0 * i
        assert g.sum().item() == 64.0 * i


@requires_gpu
def test_fused_bn_wide_channels(hvd):
    """Banked-reduction path at wide C (round-2: the flush is spread over
    kBnBanks partial rows; verify numerics at C=2048 where the old single
    accumulator was most contended)."""
    from horovod_amd.ops import FusedBNReLU
    torch.manual_seed(8)
    N, C, H, W = 8, 2048, 7, 7
    x = (torch.randn(N, C, H, W, device="cuda")
         .to(memory_format=torch.channels_last).requires_grad_(True))
    xr = x.detach().clone().requires_grad_(True)
    fused = FusedBNReLU(C).cuda().train()
    ref = torch.nn.BatchNorm2d(C).cuda().train()
    ref.load_state_dict(fused.state_dict())
    y = fused(x)
    yr = torch.nn.functional.relu(ref(xr))
    assert torch.allclose(y, yr, rtol=2e-4, atol=2e-4), \
        (y - yr).abs().max()
    g = torch.randn_like(y)
    y.backward(g)
    yr.backward(g)
    assert torch.allclose(x.grad, xr.grad, rtol=2e-4, atol=2e-4)
    assert torch.allclose(fused.weight.grad, ref.weight.grad, rtol=1e-2,
                          atol=1e-2)
    assert torch.allclose(fused.running_var, ref.running_var, atol=1e-3)


@requires_gpu
def test_per_tensor_ready_events_gpu():
    """The HOROVOD_PER_TENSOR_READY_EVENTS=1 opt-in (fine-grained producer
    ordering for non-default-stream users) still completes and matches."""
    from tests.parallel_util import run_workers
    run_workers(1, """
        torch.cuda.set_device(0)
        d = torch.device("cuda")
        s = torch.cuda.Stream()
        with torch.cuda.stream(s):
            t = torch.ones(4096, device=d) * 3
        out = hvd.allreduce(t, average=False, name="pte")
        assert torch.equal(out, t)
        ts = [torch.randn(257, device=d) for _ in range(4)]
        outs = hvd.grouped_allreduce(ts, average=False, name="pteg")
        for a, b in zip(ts, outs):
            assert torch.equal(a, b)
    """, extra_env={"HOROVOD_PER_TENSOR_READY_EVENTS": "1"}, timeout=240)


@requires_gpu
def test_fused_adamw_matches_torch(hvd):
    """FusedAdamW vs torch.optim.AdamW: identical trajectories over several
    steps (fp32), including decoupled weight decay and bias correction."""
    from horovod_amd.ops import FusedAdamW
    torch.manual_seed(5)
    shapes = [(33,), (128, 64), (7, 3, 3), (1024,)]
    ps_f = [torch.randn(s, device="cuda").requires_grad_(True)
            for s in shapes]
    ps_r = [p.detach().clone().requires_grad_(True) for p in ps_f]
    of = FusedAdamW(ps_f, lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                    weight_decay=0.05)
    orr = torch.optim.AdamW(ps_r, lr=1e-2, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.05)
    for step in range(6):
        g = torch.Generator(device="cuda").manual_seed(100 + step)
        grads = [torch.randn(s, device="cuda", generator=g) for s in shapes]
        for p, q, gr in zip(ps_f, ps_r, grads):
            p.grad = gr.clone()
            q.grad = gr.clone()
        of.step()
        orr.step()
        for i, (p, q) in enumerate(zip(ps_f, ps_r)):
            assert torch.allclose(p, q, rtol=1e-5, atol=1e-6), \
                (step, i, (p - q).abs().max().item())
    # state_dict round trip stays compatible
    sd = of.state_dict()
    of2 = FusedAdamW(ps_f, lr=1e-2)
    of2.load_state_dict(sd)
