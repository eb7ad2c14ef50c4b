"""Property-based checks (hypothesis) of the op layer's CPU reference paths —
the same code the GPU kernels are numerically tested against, fuzzed across
shapes/flags so the parity oracle itself is trustworthy."""
import torch
import torch.nn as nn
import torch.nn.functional as F
from hypothesis import given, settings, strategies as st

settings.register_profile("ci", max_examples=25, deadline=None)
settings.load_profile("ci")


@given(n=st.integers(1, 4), c=st.integers(1, 8), hw=st.sampled_from([4, 7, 8]),
       k=st.integers(1, 8), r=st.sampled_from([1, 3]),
       stride=st.sampled_from([1, 2]))
def test_conv2d_wrapper_matches_torch(n, c, hw, k, r, stride):
    from mi355x_ddp.ops import conv2d
    pad = r // 2
    g = torch.Generator().manual_seed(n * 1000 + c * 100 + hw + k + r + stride)
    x = torch.randn(n, c, hw, hw, generator=g)
    w = torch.randn(k, c, r, r, generator=g)
    got = conv2d(x, w, (stride, stride), (pad, pad))
    ref = F.conv2d(x, w, None, stride, pad)
    assert torch.allclose(got, ref, atol=1e-5)


@given(n=st.integers(1, 4), c=st.integers(1, 12), hw=st.sampled_from([2, 5, 8]),
       relu=st.booleans(), training=st.booleans(), residual=st.booleans())
def test_fused_bn_matches_eager_composition(n, c, hw, relu, training, residual):
    from mi355x_ddp.ops import bn_relu, bn_add_relu
    g = torch.Generator().manual_seed(n + c * 10 + hw * 100 + relu + 2 * training)
    x = torch.randn(n, c, hw, hw, generator=g)
    bn = nn.BatchNorm2d(c)
    with torch.no_grad():
        bn.weight.normal_(1.0, 0.1, generator=g)
        bn.bias.normal_(0.0, 0.1, generator=g)
        bn.running_mean.normal_(0, 0.5, generator=g)
        bn.running_var.uniform_(0.5, 1.5, generator=g)
    ref_bn = nn.BatchNorm2d(c)
    ref_bn.load_state_dict(bn.state_dict())
    bn.train(training)
    ref_bn.train(training)

    if residual:
        res = torch.randn(n, c, hw, hw, generator=g)
        got = bn_add_relu(x, res, bn)
        ref = torch.relu(ref_bn(x) + res)
    else:
        got = bn_relu(x, bn, relu=relu)
        ref = ref_bn(x)
        if relu:
            ref = torch.relu(ref)
    assert torch.allclose(got, ref, atol=1e-4, rtol=1e-4), \
        (got - ref).abs().max()
    assert torch.allclose(bn.running_mean, ref_bn.running_mean, atol=1e-5)
    assert torch.allclose(bn.running_var, ref_bn.running_var, atol=1e-5)


@given(n=st.integers(1, 16), c=st.sampled_from([2, 10, 100]))
def test_xent_matches_cross_entropy(n, c):
    from mi355x_ddp.ops.xent import SoftmaxCrossEntropy
    g = torch.Generator().manual_seed(n * 31 + c)
    logits = torch.randn(n, c, generator=g, requires_grad=True)
    target = torch.randint(0, c, (n,), generator=g)
    loss = SoftmaxCrossEntropy()(logits, target)
    loss.backward()
    ref_logits = logits.detach().clone().requires_grad_(True)
    ref = F.cross_entropy(ref_logits, target)
    ref.backward()
    assert torch.allclose(loss, ref, atol=1e-5)
    assert torch.allclose(logits.grad, ref_logits.grad, atol=1e-5)


@given(nt=st.integers(1, 5), momentum=st.sampled_from([0.0, 0.9]),
       wd=st.sampled_from([0.0, 1e-2]), steps=st.integers(1, 3))
def test_fused_sgd_matches_torch_sgd_fuzz(nt, momentum, wd, steps):
    from mi355x_ddp.ops import FusedSGD
    g = torch.Generator().manual_seed(nt * 7 + int(momentum * 10) + steps)
    shapes = [(3, 5), (7,), (2, 3, 4), (1,), (6, 2)][:nt]
    ps = [torch.randn(*s, generator=g, requires_grad=True) for s in shapes]
    qs = [p.detach().clone().requires_grad_(True) for p in ps]
    opt = FusedSGD(ps, lr=0.1, momentum=momentum, weight_decay=wd)
    ref = torch.optim.SGD(qs, lr=0.1, momentum=momentum, weight_decay=wd)
    for i in range(steps):
        gen = torch.Generator().manual_seed(100 + i)
        grads = [torch.randn(*s, generator=gen) for s in shapes]
        for p, q, gr in zip(ps, qs, grads):
            p.grad = gr.clone()
            q.grad = gr.clone()
        opt.step()
        ref.step()
    for p, q in zip(ps, qs):
        assert torch.allclose(p, q, atol=1e-6)


@given(nt=st.integers(1, 4), momentum=st.sampled_from([0.0, 0.9]),
       wd=st.sampled_from([0.0, 1e-2]), steps=st.integers(1, 3))
def test_fused_sgd_o2_master_matches_fp32_sgd_fuzz(nt, momentum, wd, steps):
    """O2 (bf16 params + fp32 masters): the MASTER trajectory equals plain
    fp32 SGD fed the same (bf16-rounded) gradients; the bf16 params are the
    rounded masters."""
    from mi355x_ddp.ops import FusedSGD
    g = torch.Generator().manual_seed(nt * 13 + int(momentum * 10) + steps)
    shapes = [(4, 6), (9,), (2, 2, 3), (5,)][:nt]
    w0 = [torch.randn(*s, generator=g) for s in shapes]
    ps = [w.bfloat16() for w in w0]
    opt = FusedSGD(ps, lr=0.1, momentum=momentum, weight_decay=wd)
    refs = [p.float().requires_grad_(True) for p in ps]
    ref = torch.optim.SGD(refs, lr=0.1, momentum=momentum, weight_decay=wd)
    for i in range(steps):
        gen = torch.Generator().manual_seed(300 + i)
        grads = [torch.randn(*s, generator=gen).bfloat16() for s in shapes]
        for p, q, gr in zip(ps, refs, grads):
            p.grad = gr.clone()
            q.grad = gr.float()
        opt.step()
        ref.step()
    for p, q in zip(ps, refs):
        assert torch.allclose(opt.state[p]["master"], q.detach(), atol=1e-6)
        assert torch.equal(p, q.detach().bfloat16())


@given(seed=st.integers(0, 50), cap_kb=st.sampled_from([1, 8, 64, 1024]))
def test_flat_ddp_bucket_tiling_fuzz(seed, cap_kb):
    """Bucket invariants hold for arbitrary models: buckets tile the flat
    buffer contiguously/disjointly/completely and every param maps into
    exactly the bucket containing its span."""
    import torch.nn as nn
    from mi355x_ddp.parallel import FlatDDP
    g = torch.Generator().manual_seed(seed)
    layers = []
    dims = [int(torch.randint(1, 40, (1,), generator=g))
            for _ in range(int(torch.randint(2, 6, (1,), generator=g)))]
    prev = 7
    for d in dims:
        layers += [nn.Linear(prev, d, bias=bool(d % 2))]
        prev = d
    model = FlatDDP(nn.Sequential(*layers), bucket_cap_mb=cap_kb / 1024.0,
                    static_grads=True)
    numel = sum(p.numel() for p in model._params)
    assert model.flat_grads.numel() == numel
    prev_end = 0
    for (s_, e_, ps) in model._buckets:
        assert s_ == prev_end and e_ > s_
        prev_end = e_
    assert prev_end == numel
    for p in model._params:
        off, view = model._views[p]
        assert p.grad is view and view.shape == p.shape
        s_, e_, ps = model._buckets[model._param_bucket[p]]
        assert s_ <= off and off + p.numel() <= e_ and p in ps
