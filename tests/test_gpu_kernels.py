"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32 oracle
(nats_amd.ops.eager). All marked gpu; run with `pytest -m gpu` on MI355X."""

import numpy
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from nats_amd.ops import _hip_ext
    m = _hip_ext()
    assert m is not None, "HIP extension must build/load on a GPU box"
    return m


def test_mfma_fragment_layout(ext):
    """Transpose-detecting self-test of the MFMA lane maps (asymmetric B)."""
    torch.manual_seed(0)
    M, K, N = 32, 64, 48
    A = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    B = torch.randn(K, N, device="cuda").to(torch.bfloat16)
    C = ext.mfma_gemm_bt(A.contiguous(), B.t().contiguous())
    ref = (A.float() @ B.float())
    torch.testing.assert_close(C, ref, rtol=2e-2, atol=2e-2)


def _gru_inputs(T=13, B=5, H=48, E=20, seed=0, with_mask=True):
    g = torch.Generator().manual_seed(seed)
    xg = torch.randn(T, B, 2 * H, generator=g)
    xc = torch.randn(T, B, H, generator=g)
    U = 0.3 * torch.randn(H, 2 * H, generator=g)
    Ux = 0.3 * torch.randn(H, H, generator=g)
    mask = None
    if with_mask:
        lens = torch.randint(2, T + 1, (B,), generator=g)
        mask = (torch.arange(T).unsqueeze(1) < lens.unsqueeze(0)).float()
    return xg, xc, U, Ux, mask


@pytest.mark.parametrize("with_mask", [False, True])
@pytest.mark.parametrize("shape", [(13, 5, 48), (40, 20, 96), (7, 32, 16)])
def test_gru_scan_forward(ext, shape, with_mask):
    from nats_amd.ops import eager
    from nats_amd.ops.gru import gru_scan_hip
    T, B, H = shape
    xg, xc, U, Ux, mask = _gru_inputs(T, B, H, with_mask=with_mask)
    ref = eager.gru_scan(xg, xc, mask, U, Ux)
    out = gru_scan_hip(xg.cuda(), xc.cuda(),
                       mask.cuda() if mask is not None else None,
                       U.cuda(), Ux.cuda())
    torch.testing.assert_close(out.cpu(), ref, rtol=3e-2, atol=3e-2)


def test_gru_scan_backward(ext):
    from nats_amd.ops import eager
    from nats_amd.ops.gru import gru_scan_hip
    T, B, H = 11, 6, 32
    xg, xc, U, Ux, mask = _gru_inputs(T, B, H, seed=3)

    # fp32 eager oracle
    ref_in = [t.clone().requires_grad_(True) for t in (xg, xc, U, Ux)]
    h_ref = eager.gru_scan(ref_in[0], ref_in[1], mask, ref_in[2], ref_in[3])
    loss_w = torch.randn_like(h_ref)
    (h_ref * loss_w).sum().backward()

    hip_in = [t.clone().cuda().requires_grad_(True) for t in (xg, xc, U, Ux)]
    h_hip = gru_scan_hip(hip_in[0], hip_in[1], mask.cuda(), hip_in[2],
                         hip_in[3])
    (h_hip * loss_w.cuda()).sum().backward()

    for r, h, name in zip(ref_in, hip_in, ["xg", "xc", "U", "Ux"]):
        a, b = r.grad, h.grad.cpu().float()
        denom = a.abs().max().clamp_min(1e-3)
        rel = (a - b).abs().max() / denom
        assert rel < 0.06, (name, float(rel))


def test_softmax_ce(ext):
    from nats_amd.ops import eager
    from nats_amd.ops.softmax_ce import softmax_xent_hip
    torch.manual_seed(1)
    N, V = 37, 3001
    logits = (5 * torch.randn(N, V)).to(torch.bfloat16)
    targets = torch.randint(0, V, (N,))
    ref_l = logits.float().clone().requires_grad_(True)
    ref = eager.softmax_xent(ref_l, targets)
    dn = torch.randn(N)
    (ref * dn).sum().backward()

    hip_l = logits.cuda().clone().requires_grad_(True)
    out = softmax_xent_hip(hip_l, targets.cuda())
    torch.testing.assert_close(out.cpu(), ref.detach(), rtol=2e-2, atol=2e-2)
    (out * dn.cuda()).sum().backward()
    a, b = ref_l.grad, hip_l.grad.cpu().float()
    assert (a - b).abs().max() < 2e-2


def test_model_forward_gpu_vs_cpu():
    """Full model: GPU (HIP gru + fused CE under autocast) vs CPU fp32."""
    from nats_amd.data.prepare import prepare_data
    from nats_amd.models.distraction import NatsModel, default_options
    opts = default_options(dim_word=16, dim=32, dim_att=8, n_words=120)
    model = NatsModel(opts, seed=9)
    rng = numpy.random.RandomState(0)
    xs = [list(rng.randint(2, 120, size=rng.randint(4, 12))) for _ in range(6)]
    ys = [list(rng.randint(2, 120, size=rng.randint(3, 8))) for _ in range(6)]
    arrs = [torch.from_numpy(a) for a in prepare_data(xs, ys)]
    cost_cpu = model(*arrs)

    gmodel = NatsModel(opts, params=model.get_params()).cuda()
    with torch.autocast("cuda", dtype=torch.bfloat16):
        cost_gpu = gmodel(*[a.cuda() for a in arrs])
    rel = ((cost_cpu - cost_gpu.float().cpu()).abs() / cost_cpu.abs()).max()
    assert rel < 0.05, float(rel)


def test_model_train_step_gpu():
    """One full train step (fwd+bwd+clip+fused adadelta) on the HIP path:
    finite cost, finite grads, params move."""
    from nats_amd.data.synthetic import synthetic_batch
    from nats_amd.engine.optim import build_optimizer
    from nats_amd.models.distraction import NatsModel, default_options
    opts = default_options(dim_word=24, dim=48, dim_att=12, n_words=500)
    model = NatsModel(opts, seed=2).cuda()
    opt = build_optimizer("adadelta", list(model.P.items()), clip_c=1.0)
    rng = numpy.random.RandomState(3)
    x, xm, y, ym = [torch.from_numpy(a).cuda()
                    for a in synthetic_batch(rng, 8, 30, 10, 500)]
    before = model.P["encoder_U"].detach().clone()
    costs = []
    for _ in range(3):
        opt.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            cost = model(x, xm, y, ym).mean()
        cost.backward()
        opt.step()
        costs.append(float(cost))
    assert all(numpy.isfinite(c) for c in costs), costs
    assert not torch.equal(before, model.P["encoder_U"].detach())


def test_native_extension_is_used_on_gpu():
    """Fail-loud contract: GPU tensors must NOT silently run eager."""
    import nats_amd.ops as ops
    assert ops._hip_ext() is not None
    import os
    assert not os.environ.get("NATS_AMD_FORCE_EAGER")
