"""GPU numerics tests: HIP kernels vs plain PyTorch fp32 references.

All marked @pytest.mark.gpu — run on a real MI355X via gpurun.
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from gordo_amd.model.torch_nets import FeedForwardNet
    from gordo_amd.ops import ext
    from gordo_amd.ops.hourglass_engine import (
        WMAX,
        FusedHourglassForward,
        FusedHourglassTrainer,
        _PackedParams,
        pad_rows_bf16,
    )
    from gordo_amd.runtime.trainer import TrainConfig
else:  # collection on CPU hosts
    pytest.skip("no GPU", allow_module_level=True)

DEV = torch.device("cuda:0")


def make_net(widths=(50, 38, 25, 38, 50), seed=0):
    torch.manual_seed(seed)
    dims = list(widths[1:-1])
    funcs = ["tanh"] * len(dims)
    return FeedForwardNet(widths[0], widths[-1], dims, funcs, "linear")


def test_extension_loads():
    E = ext()
    assert E is not None
    assert E.WMAX == 64


def test_forward_matches_torch():
    net = make_net()
    rng = np.random.default_rng(0)
    X = rng.normal(size=(257, 50)).astype(np.float32)  # non-multiple of 64
    ref = net(torch.as_tensor(X)).detach().numpy()

    fwd = FusedHourglassForward(net, DEV)
    out = fwd.forward(torch.as_tensor(X)).cpu().numpy()
    assert out.shape == ref.shape
    # bf16 compute tolerance (weights + activations rounded to bf16)
    np.testing.assert_allclose(out, ref, atol=3e-2, rtol=3e-2)


def test_forward_asymmetric_single_layer():
    """Transpose-detecting check (guide §5.4 rule 16): a 1-layer linear net
    with asymmetric weights catches any operand/output transpose."""
    net = FeedForwardNet(50, 40, [], [], "linear")
    with torch.no_grad():
        w = torch.arange(40 * 50, dtype=torch.float32).reshape(40, 50) / (40 * 50)
        net.net[0].weight.copy_(w)
        net.net[0].bias.copy_(torch.linspace(-1, 1, 40))
    X = np.random.default_rng(1).normal(size=(64, 50)).astype(np.float32)
    ref = net(torch.as_tensor(X)).detach().numpy()
    out = FusedHourglassForward(net, DEV).forward(torch.as_tensor(X)).cpu().numpy()
    np.testing.assert_allclose(out, ref, atol=3e-2, rtol=3e-2)


def test_anomaly_fused_matches_reference():
    net = make_net()
    rng = np.random.default_rng(2)
    X = rng.normal(size=(200, 50)).astype(np.float32)
    inv_scale = (1.0 / (rng.uniform(0.5, 2.0, size=50))).astype(np.float32)

    fwd = FusedHourglassForward(net, DEV)
    yhat, err, total = fwd.anomaly(torch.as_tensor(X), inv_scale)
    yhat, err, total = yhat.cpu().numpy(), err.cpu().numpy(), total.cpu().numpy()

    # reference (fp32 torch, bf16-rounded input like the kernel sees)
    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    ref_y = net(Xb).detach().numpy()
    ref_err = np.abs(ref_y - Xb.numpy()) * inv_scale
    ref_total = np.linalg.norm(ref_err, axis=1)
    np.testing.assert_allclose(yhat, ref_y, atol=3e-2, rtol=3e-2)
    np.testing.assert_allclose(err, ref_err, atol=4e-2, rtol=5e-2)
    np.testing.assert_allclose(total, ref_total, atol=0.1, rtol=5e-2)


def test_train_step_gradients_match_autograd():
    """One fused step vs torch autograd on identical data/weights: dW, db and
    the Adam-updated weights must agree to bf16 tolerances."""
    net = make_net(seed=3)
    ref_net = make_net(seed=3)  # identical init
    rng = np.random.default_rng(3)
    N = 256
    X = rng.normal(size=(N, 50)).astype(np.float32)

    # drive the kernels directly for determinism (identity perm)
    E = ext()
    p = _PackedParams(net, DEV)
    L, F_out = p.L, p.widths[-1]
    Xd = pad_rows_bf16(X, DEV)
    B = N
    MT = int(E.MT)
    n_tiles = (B + MT - 1) // MT
    dw = torch.zeros(n_tiles, L, WMAX * WMAX, dtype=torch.bfloat16, device=DEV)
    db = torch.zeros(n_tiles, L, WMAX, dtype=torch.bfloat16, device=DEV)
    loss = torch.zeros(1, dtype=torch.float32, device=DEV)
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)
    # kbits=0 -> identity gather (row m = sample m)
    E.hg_train_step(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                    loss, B, F_out, p.act_ids, gscale, True, None, WMAX, 0, 0)
    torch.cuda.synchronize()

    # --- torch reference on the same bf16-rounded input ------------------
    Xref = Xd[:, :50].float().cpu()
    out = ref_net(Xref)
    loss_ref = torch.nn.functional.mse_loss(out, Xref)
    loss_ref.backward()

    kernel_loss = float(loss.item()) / (B * F_out)
    ref_val = float(loss_ref.detach())
    assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 5e-2

    linears = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)]
    dw_sum = dw.sum(dim=0, dtype=torch.float32).view(L, WMAX, WMAX).cpu()
    db_sum = db.sum(dim=0, dtype=torch.float32).view(L, WMAX).cpu()
    for i, lin in enumerate(linears):
        o, ninp = lin.weight.shape
        got_w = dw_sum[i, :ninp, :o].T  # kernel stores [in][out]
        ref_w = lin.weight.grad.detach()
        scale = ref_w.abs().max().item() or 1.0
        assert (got_w - ref_w).abs().max().item() < 0.05 * scale + 1e-4, f"layer {i} dW"
        got_b = db_sum[i, :o]
        ref_b = lin.bias.grad.detach()
        bscale = ref_b.abs().max().item() or 1.0
        assert (got_b - ref_b).abs().max().item() < 0.05 * bscale + 1e-4, f"layer {i} db"
        # pad region produced zero grads
        assert dw_sum[i, ninp:, :].abs().max().item() == 0.0
        assert dw_sum[i, :, o:].abs().max().item() == 0.0

    # --- Adam step: fused update must match a manual fp32 Adam -----------
    lr, b1, b2, eps = 1e-3, 0.9, 0.999, 1e-8
    master_before = p.master.clone()
    m_opt = torch.zeros_like(p.master)
    v_opt = torch.zeros_like(p.master)
    E.hg_adam(p.master, m_opt, v_opt, dw, db, p.w_bf16, p.wt_bf16, p.bias,
              counters, n_tiles, L, lr, b1, b2, eps, WMAX)
    torch.cuda.synchronize()

    # manual reference on the kernel's own summed grads (layout-adjusted)
    g = torch.zeros_like(master_before)
    PW = L * WMAX * WMAX
    g[:PW] = dw.sum(dim=0, dtype=torch.float32).view(L, WMAX, WMAX).transpose(1, 2).reshape(-1)
    g[PW:] = db.sum(dim=0, dtype=torch.float32).reshape(-1)
    m_ref = (1 - b1) * g
    v_ref = (1 - b2) * g * g
    upd = master_before - lr * (m_ref / (1 - b1)) / ((v_ref / (1 - b2)).sqrt() + eps)
    diff = (p.master - upd).abs().max().item()
    assert diff < 1e-5, f"adam mismatch {diff}"
    # bf16 working copies match the new master
    w_now = p.master[:PW].view(L, WMAX, WMAX)
    assert torch.allclose(p.w_bf16.float(), w_now.to(torch.bfloat16).float())
    assert torch.allclose(p.wt_bf16.float(),
                          w_now.transpose(1, 2).contiguous().to(torch.bfloat16).float())


def test_fused_trainer_learns():
    net = make_net(seed=4)
    rng = np.random.default_rng(4)
    # low-rank structured data: an AE can actually learn it
    Z = rng.normal(size=(4096, 8)).astype(np.float32)
    P = rng.normal(size=(8, 50)).astype(np.float32)
    X = np.tanh(Z @ P)

    cfg = TrainConfig(epochs=20, batch_size=1024, lr=3e-3, seed=0)
    tr = FusedHourglassTrainer(net, DEV, cfg)
    hist = tr.fit(X, X)
    assert hist.engine.startswith("fused_hip")
    assert hist.loss[-1] < hist.loss[0] * 0.7, hist.loss
    # module weights were synced back: torch forward agrees with training
    out = net(torch.as_tensor(X[:64])).detach().numpy()
    assert np.isfinite(out).all()


def test_estimator_uses_fused_engine():
    from gordo_amd.model.models import TorchAutoEncoder

    X = np.random.default_rng(5).normal(size=(2048, 50)).astype(np.float32)
    model = TorchAutoEncoder(kind="feedforward_hourglass", encoding_layers=2,
                             epochs=3, batch_size=512)
    model.fit(X)
    assert model.history["engine"].startswith("fused_hip"), model.history
    out = model.predict(X)
    assert out.shape == X.shape


def test_feistel_shuffle_is_bijective():
    """The device-side minibatch shuffle must be a true permutation of [0, N)
    for every epoch key (a broken shuffle silently corrupts training)."""
    from gordo_amd.ops.hourglass_engine import feistel_kbits

    E = ext()
    for n in (7, 64, 1000, 131072):
        kbits = feistel_kbits(n)
        for key in (0, 1, 0xDEADBEEF):
            out = torch.zeros(n, dtype=torch.int32, device=DEV)
            E.hg_permute_fill(out, kbits, key)
            torch.cuda.synchronize()
            vals = out.cpu().numpy()
            assert sorted(vals) == list(range(n)), (n, key)
    # different keys give different orders (the point of the epoch key)
    a = torch.zeros(1000, dtype=torch.int32, device=DEV)
    b = torch.zeros(1000, dtype=torch.int32, device=DEV)
    E.hg_permute_fill(a, feistel_kbits(1000), 1)
    E.hg_permute_fill(b, feistel_kbits(1000), 2)
    torch.cuda.synchronize()
    assert not torch.equal(a, b)


def test_train_step_w1_gradients_match_autograd():
    """The wave-local (16-row, zero-barrier) W=64 train step vs torch
    autograd AND vs the 4-wave kernel's slab sums on identical data."""
    net = make_net(seed=13)
    ref_net = make_net(seed=13)
    rng = np.random.default_rng(13)
    N = 270  # deliberately not a multiple of 16: last tile is ragged
    X = rng.normal(size=(N, 50)).astype(np.float32)

    E = ext()
    if not hasattr(E, "hg_train_step_w1"):
        pytest.skip("extension built without the wave-local step")
    p = _PackedParams(net, DEV)
    L, F_out = p.L, p.widths[-1]
    Xd = pad_rows_bf16(X, DEV)
    B = N
    MT = int(E.MT_W1)
    n_tiles = (B + MT - 1) // MT
    dw = torch.zeros(n_tiles, L, WMAX * WMAX, dtype=torch.bfloat16, device=DEV)
    db = torch.zeros(n_tiles, L, WMAX, dtype=torch.bfloat16, device=DEV)
    loss = torch.zeros(1, dtype=torch.float32, device=DEV)
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)
    E.hg_train_step_w1(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                       loss, B, F_out, p.act_ids, gscale, True, None, WMAX, 0, 0)
    torch.cuda.synchronize()

    Xref = Xd[:, :50].float().cpu()
    out = ref_net(Xref)
    loss_ref = torch.nn.functional.mse_loss(out, Xref)
    loss_ref.backward()
    kernel_loss = float(loss.item()) / (B * F_out)
    ref_val = float(loss_ref.detach())
    assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 5e-2

    linears = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)]
    dw_sum = dw.sum(dim=0, dtype=torch.float32).view(L, WMAX, WMAX).cpu()
    db_sum = db.sum(dim=0, dtype=torch.float32).view(L, WMAX).cpu()
    for i, lin in enumerate(linears):
        o, ninp = lin.weight.shape
        got_w = dw_sum[i, :ninp, :o].T
        ref_w = lin.weight.grad.detach()
        scale = ref_w.abs().max().item() or 1.0
        assert (got_w - ref_w).abs().max().item() < 0.05 * scale + 1e-4, f"layer {i} dW"
        got_b = db_sum[i, :o]
        ref_b = lin.bias.grad.detach()
        bscale = ref_b.abs().max().item() or 1.0
        assert (got_b - ref_b).abs().max().item() < 0.05 * bscale + 1e-4, f"layer {i} db"
        assert dw_sum[i, ninp:, :].abs().max().item() == 0.0
        assert dw_sum[i, :, o:].abs().max().item() == 0.0

    # and against the 4-wave kernel's own slab sums (tight: same math path)
    MT4 = int(E.MT)
    nt4 = (B + MT4 - 1) // MT4
    dw4 = torch.zeros(nt4, L, WMAX * WMAX, dtype=torch.bfloat16, device=DEV)
    db4 = torch.zeros(nt4, L, WMAX, dtype=torch.bfloat16, device=DEV)
    loss4 = torch.zeros(1, dtype=torch.float32, device=DEV)
    E.hg_train_step(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw4, db4,
                    loss4, B, F_out, p.act_ids, gscale, True, None, WMAX, 0, 0)
    torch.cuda.synchronize()
    assert abs(float(loss.item()) - float(loss4.item())) / max(float(loss4.item()), 1e-9) < 1e-3
    s1 = dw.sum(dim=0, dtype=torch.float32)
    s4 = dw4.sum(dim=0, dtype=torch.float32)
    ref = s4.abs().max().item() or 1.0
    assert (s1 - s4).abs().max().item() < 0.01 * ref + 1e-5


def test_train_step_w1_dx_out_matches():
    """dx_out (the LSTM-head contract) from the wave-local step matches the
    4-wave kernel bit-for-tolerance."""
    net = make_net(seed=14)
    rng = np.random.default_rng(14)
    N = 128
    X = rng.normal(size=(N, 50)).astype(np.float32)
    E = ext()
    if not hasattr(E, "hg_train_step_w1"):
        pytest.skip("extension built without the wave-local step")
    p = _PackedParams(net, DEV)
    L, F_out = p.L, p.widths[-1]
    Xd = pad_rows_bf16(X, DEV)
    B = N
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)

    outs = []
    for fn, mt in ((E.hg_train_step_w1, int(E.MT_W1)), (E.hg_train_step, int(E.MT))):
        nt = (B + mt - 1) // mt
        dw = torch.zeros(nt, L, WMAX * WMAX, dtype=torch.bfloat16, device=DEV)
        db = torch.zeros(nt, L, WMAX, dtype=torch.bfloat16, device=DEV)
        loss = torch.zeros(1, dtype=torch.float32, device=DEV)
        dx = torch.zeros(B, WMAX, dtype=torch.bfloat16, device=DEV)
        fn(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
           loss, B, F_out, p.act_ids, gscale, True, dx, WMAX, 0, 0)
        torch.cuda.synchronize()
        outs.append(dx.float().cpu())
    scale = outs[1].abs().max().item() or 1.0
    assert (outs[0] - outs[1]).abs().max().item() < 0.02 * scale + 1e-6
