"""GPU coverage tests: fused kernels across the architecture envelope
(activations, layer counts, odd shapes) and end-to-end training quality
vs the eager fp32 reference."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from gordo_amd.model.torch_nets import FeedForwardNet
    from gordo_amd.ops.hourglass_engine import FusedHourglassForward, FusedHourglassTrainer
    from gordo_amd.runtime.trainer import TrainConfig, Trainer
else:
    pytest.skip("no GPU", allow_module_level=True)

DEV = torch.device("cuda:0")


@pytest.mark.parametrize(
    "widths,funcs,out_func",
    [
        ((50, 38, 25, 38, 50), ["tanh"] * 3, "linear"),
        ((50, 38, 25, 38, 50), ["relu"] * 3, "linear"),
        ((50, 38, 25, 38, 50), ["sigmoid"] * 3, "sigmoid"),
        ((20, 10, 20), ["relu"], "tanh"),  # mixed funcs
        ((64, 64, 64, 64, 64, 64), ["tanh"] * 4, "linear"),  # max width, 5 layers
        ((7, 3, 7), ["tanh"], "linear"),  # tiny odd widths
        ((33, 24, 17), ["relu"], "linear"),  # single hidden, asymmetric in/out
    ],
)
def test_fused_forward_envelope(widths, funcs, out_func):
    torch.manual_seed(0)
    dims = list(widths[1:-1])
    net = FeedForwardNet(widths[0], widths[-1], dims, list(funcs), out_func)
    X = np.random.default_rng(1).normal(size=(197, widths[0])).astype(np.float32)
    ref = net(torch.as_tensor(X)).detach().numpy()
    out = FusedHourglassForward(net, DEV).forward(torch.as_tensor(X)).cpu().numpy()
    np.testing.assert_allclose(out, ref, atol=4e-2, rtol=4e-2)


@pytest.mark.parametrize("funcs", [["tanh"], ["relu"], ["sigmoid"]])
def test_fused_training_per_activation(funcs):
    """One training run per activation: grads flow and loss decreases."""
    torch.manual_seed(1)
    net = FeedForwardNet(30, 30, [16], funcs, "linear")
    rng = np.random.default_rng(2)
    Z = rng.normal(size=(2048, 6)).astype(np.float32)
    P = rng.normal(size=(6, 30)).astype(np.float32)
    X = np.tanh(Z @ P) * 0.5
    cfg = TrainConfig(epochs=15, batch_size=512, lr=3e-3, seed=0)
    hist = FusedHourglassTrainer(net, DEV, cfg).fit(X, X)
    assert hist.engine.startswith("fused_hip")
    assert hist.loss[-1] < hist.loss[0] * 0.8, (funcs, hist.loss)


def test_fused_vs_eager_training_quality():
    """The fused bf16 trainer must reach the same reconstruction quality as
    eager fp32 torch on the same structured data (explained variance)."""
    from sklearn.metrics import explained_variance_score

    rng = np.random.default_rng(3)
    Z = rng.normal(size=(8192, 8)).astype(np.float32)
    P = rng.normal(size=(8, 50)).astype(np.float32)
    X = np.tanh(Z @ P)

    def run(device, use_fused):
        torch.manual_seed(7)
        net = FeedForwardNet(50, 50, [38, 25, 38], ["tanh"] * 3, "linear")
        cfg = TrainConfig(epochs=40, batch_size=1024, lr=3e-3, seed=7,
                          device=device)
        if use_fused:
            hist = FusedHourglassTrainer(net, DEV, cfg).fit(X, X)
            assert hist.engine.startswith("fused_hip")
        else:
            Trainer(cfg)._fit_eager(net, X, X, 0, 0)
        with torch.no_grad():
            out = net.cpu()(torch.as_tensor(X)).numpy()
        return explained_variance_score(X, out)

    ev_fused = run("cuda:0", True)
    ev_eager = run("cuda:0", False)
    assert ev_fused > 0.5, ev_fused
    assert ev_fused > ev_eager - 0.1, (ev_fused, ev_eager)


# ---------------------------------------------------------------------------
# wide variant (W=128): tag sets above 64 use the 128-padded kernels
# ---------------------------------------------------------------------------
def test_fused_forward_wide():
    torch.manual_seed(2)
    net = FeedForwardNet(128, 128, [96, 64, 96], ["tanh"] * 3, "linear")
    X = np.random.default_rng(3).normal(size=(211, 128)).astype(np.float32)
    ref = net(torch.as_tensor(X)).detach().numpy()
    out = FusedHourglassForward(net, DEV).forward(torch.as_tensor(X)).cpu().numpy()
    np.testing.assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_fused_training_wide_matches_autograd():
    """Wide (W=128) grads vs fp32 autograd on one step."""
    from gordo_amd.ops import ext
    from gordo_amd.ops.hourglass_engine import _PackedParams, pad_rows_bf16

    torch.manual_seed(3)
    net = FeedForwardNet(100, 100, [72], ["tanh"], "linear")
    ref_net = FeedForwardNet(100, 100, [72], ["tanh"], "linear")
    ref_net.load_state_dict(net.state_dict())
    X = np.random.default_rng(4).normal(size=(256, 100)).astype(np.float32)

    E = ext()
    p = _PackedParams(net, DEV)
    assert p.wmax == 128
    WM, L, F_out = p.wmax, p.L, 100
    B = 256
    MT = int(E.MT)
    n_tiles = (B + MT - 1) // MT
    Xd = pad_rows_bf16(X, DEV, WM)
    dw = torch.zeros(n_tiles, L, WM * WM, dtype=torch.bfloat16, device=DEV)
    db = torch.zeros(n_tiles, L, WM, dtype=torch.bfloat16, device=DEV)
    loss = torch.zeros(1, dtype=torch.float32, device=DEV)
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)
    E.hg_train_step(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                    loss, B, F_out, p.act_ids, gscale, True, None, WM, 0, 0)
    torch.cuda.synchronize()

    Xref = Xd[:, :100].float().cpu()
    out = ref_net(Xref)
    loss_ref = torch.nn.functional.mse_loss(out, Xref)
    loss_ref.backward()
    kernel_loss = float(loss.item()) / (B * F_out)
    ref_val = float(loss_ref.detach())
    assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 5e-2

    linears = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)]
    dw_sum = dw.sum(dim=0, dtype=torch.float32).view(L, WM, WM).cpu()
    db_sum = db.sum(dim=0, dtype=torch.float32).view(L, WM).cpu()
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
        assert dw_sum[i, ninp:, :].abs().max().item() == 0.0
        assert dw_sum[i, :, o:].abs().max().item() == 0.0


def test_fused_trainer_wide_learns():
    rng = np.random.default_rng(5)
    Z = rng.normal(size=(4096, 12)).astype(np.float32)
    P = rng.normal(size=(12, 100)).astype(np.float32)
    X = np.tanh(Z @ P)
    torch.manual_seed(5)
    net = FeedForwardNet(100, 100, [72, 48, 72], ["tanh"] * 3, "linear")
    cfg = TrainConfig(epochs=20, batch_size=1024, lr=3e-3, seed=0)
    hist = FusedHourglassTrainer(net, DEV, cfg).fit(X, X)
    assert hist.engine.startswith("fused_hip")
    assert hist.loss[-1] < hist.loss[0] * 0.7, hist.loss


def test_fused_training_y_not_x():
    """Fused trainer with a distinct regression target (y != X), vs autograd."""
    from gordo_amd.ops import ext
    from gordo_amd.ops.hourglass_engine import WMAX, _PackedParams, pad_rows_bf16

    torch.manual_seed(9)
    net = FeedForwardNet(30, 10, [20], ["tanh"], "linear")
    ref_net = FeedForwardNet(30, 10, [20], ["tanh"], "linear")
    ref_net.load_state_dict(net.state_dict())
    rng = np.random.default_rng(9)
    X = rng.normal(size=(256, 30)).astype(np.float32)
    Y = rng.normal(size=(256, 10)).astype(np.float32)

    E = ext()
    p = _PackedParams(net, DEV)
    B, L, F_out = 256, p.L, 10
    MT = int(E.MT)
    n_tiles = (B + MT - 1) // MT
    Xd = pad_rows_bf16(X, DEV)
    Yd = pad_rows_bf16(Y, DEV)
    dw = torch.zeros(n_tiles, L, WMAX * WMAX, dtype=torch.bfloat16, device=DEV)
    db = torch.zeros(n_tiles, L, WMAX, dtype=torch.bfloat16, device=DEV)
    loss = torch.zeros(1, dtype=torch.float32, device=DEV)
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)
    E.hg_train_step(Xd, Yd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                    loss, B, F_out, p.act_ids, gscale, False, None, WMAX, 0, 0)
    torch.cuda.synchronize()

    Xr = Xd[:, :30].float().cpu()
    Yr = Yd[:, :10].float().cpu()
    out = ref_net(Xr)
    loss_ref = torch.nn.functional.mse_loss(out, Yr)
    loss_ref.backward()
    kernel_loss = float(loss.item()) / (B * F_out)
    ref_val = float(loss_ref.detach())
    assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 5e-2
    lin0 = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)][0]
    got = dw.sum(0, dtype=torch.float32).view(L, WMAX, WMAX)[0, :30, :20].T.cpu()
    ref = lin0.weight.grad.detach()
    s = ref.abs().max().item() or 1.0
    assert (got - ref).abs().max().item() < 0.05 * s + 1e-4


def test_fused_training_is_deterministic():
    """Two fits with the same seed must produce bitwise-identical weights:
    every kernel writes to a fixed slot (no atomics on the weight path) and
    the shuffle is a deterministic keyed permutation."""
    rng = np.random.default_rng(11)
    X = rng.normal(size=(4096, 50)).astype(np.float32)

    def run():
        torch.manual_seed(3)
        net = FeedForwardNet(50, 50, [38, 25, 38], ["tanh"] * 3, "linear")
        cfg = TrainConfig(epochs=5, batch_size=1024, lr=3e-3, seed=3)
        FusedHourglassTrainer(net, DEV, cfg).fit(X, X)
        return torch.cat([p.detach().cpu().reshape(-1) for p in net.parameters()])

    w1, w2 = run(), run()
    assert torch.equal(w1, w2), (w1 - w2).abs().max()


def test_lstm_fused_vs_eager_quality():
    """Fused bf16 LSTM training reaches eager-fp32 reconstruction quality."""
    from sklearn.metrics import explained_variance_score

    from gordo_amd.model.torch_nets import LSTMNet
    from gordo_amd.ops.lstm_engine import FusedLSTMTrainer
    from gordo_amd.runtime.trainer import Trainer

    rng = np.random.default_rng(12)
    t = np.arange(6000)
    X = np.stack(
        [np.sin(t / (6 + 2 * i)) * 0.5 + 0.02 * rng.standard_normal(len(t)) for i in range(16)],
        axis=1,
    ).astype(np.float32)
    T = 24

    def ev_of(net):
        from gordo_amd.ops.lstm_engine import FusedLSTMForward

        out = FusedLSTMForward(net, DEV).predict(X[:2000])
        tgt = X[T - 1 : 2000]
        return explained_variance_score(tgt, out)

    torch.manual_seed(5)
    net_f = LSTMNet(16, 16, [24], [None], "linear", lookback_window=T)
    cfg = TrainConfig(epochs=12, batch_size=1024, lr=3e-3, seed=5)
    FusedLSTMTrainer(net_f, DEV, cfg).fit(X, X, lookback=T)
    ev_fused = ev_of(net_f)

    torch.manual_seed(5)
    net_e = LSTMNet(16, 16, [24], [None], "linear", lookback_window=T)
    cfg_e = TrainConfig(epochs=12, batch_size=1024, lr=3e-3, seed=5,
                        device="cuda:0", dtype="fp32")
    Trainer(cfg_e)._fit_eager(net_e, X, X, T, 0)
    ev_eager = ev_of(net_e)

    assert ev_fused > 0.45, (ev_fused, ev_eager)
    assert ev_fused > ev_eager - 0.15, (ev_fused, ev_eager)


def test_lstm_beyond_fused_envelope_gemmloop_gpu():
    """>256-wide LSTM (300 tags — past ALL three fused width tiers): the
    estimator must run the hipGraph GEMM-loop engine on GPU and still learn.
    (200-tag models moved onto the fused W=256 tier in round 2 and are
    covered by test_gpu_lstm.py's w256 tests.)"""
    from gordo_amd.model.models import TorchLSTMAutoEncoder

    rng = np.random.default_rng(31)
    t = np.arange(3000)
    X = np.stack(
        [np.sin(t / (8 + i % 13)) + 0.05 * rng.standard_normal(len(t)) for i in range(300)],
        axis=1,
    ).astype(np.float32)
    model = TorchLSTMAutoEncoder(
        kind="lstm_symmetric", dims=[280], lookback_window=12, epochs=3,
        batch_size=512, lr=3e-3,
    )
    model.fit(X)
    assert model.history["engine"].startswith("wide_lstm_gemmloop"), model.history
    assert model.history["loss"][-1] < model.history["loss"][0]
    out = model.predict(X[:100])
    assert out.shape == (100 - 12 + 1, 300)
    assert np.isfinite(out).all()


def test_fused_checkpoint_resume_bit_exact(tmp_path):
    """Fused trainer checkpoint: the device-side step/epoch counters + Feistel
    key are part of the state, so a resumed run replays the identical
    minibatch sequence and trajectory."""
    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer

    torch.manual_seed(21)
    module = feedforward_hourglass(50, 50, encoding_layers=2, compression_factor=0.5)
    X = np.random.default_rng(21).uniform(0, 1, size=(4096, 50)).astype(np.float32)
    dp = DPContext(device=DEV)
    cfg = TrainConfig(batch_size=512, lr=1e-3, dtype="bf16")
    trainer = HourglassStepTrainer(module, X, 512, cfg, dp=dp)
    assert trainer.fused
    # odd step count: the checkpoint lands on ODD counter parity, so this
    # also proves load_state_dict's parity normalization (restore into slot
    # 0, parity 0) is trajectory-exact
    for _ in range(5):
        trainer.step()
    trainer.synchronize()
    ckpt = tmp_path / "fused.pt"
    trainer.save_checkpoint(str(ckpt))
    for _ in range(4):
        trainer.step()
    trainer.synchronize()
    vec1 = trainer.params_vector().cpu().numpy()
    trainer.load_checkpoint(str(ckpt))
    for _ in range(4):
        trainer.step()
    trainer.synchronize()
    vec2 = trainer.params_vector().cpu().numpy()
    np.testing.assert_array_equal(vec1, vec2)


def test_lstm_checkpoint_restores_params(tmp_path):
    from gordo_amd.model.factories.lstm_autoencoder import lstm_symmetric
    from gordo_amd.ops.lstm_engine import LSTMStepTrainer

    torch.manual_seed(22)
    module = lstm_symmetric(20, 20, dims=[32], lookback_window=12)
    X = np.random.default_rng(22).uniform(0, 1, size=(2048, 20)).astype(np.float32)
    cfg = TrainConfig(batch_size=512, lr=1e-3, dtype="bf16")
    tr = LSTMStepTrainer(module, X, 512, cfg)
    for _ in range(3):
        tr.step()
    tr.synchronize()
    ckpt = tmp_path / "lstm.pt"
    tr.save_checkpoint(str(ckpt))
    ref = [t.detach().cpu().clone() for t in tr.p.masters()]
    for _ in range(3):
        tr.step()
    tr.synchronize()
    tr.load_checkpoint(str(ckpt))
    for got, want in zip(tr.p.masters(), ref):
        np.testing.assert_array_equal(got.detach().cpu().numpy(), want.numpy())
    tr.step()  # resumes cleanly
    tr.synchronize()


# ---------------------------------------------------------------------------
# x-wide variant (W=256): tag sets above 128 use the 256-padded kernels
# ---------------------------------------------------------------------------
def test_fused_forward_x256():
    torch.manual_seed(11)
    net = FeedForwardNet(200, 200, [150, 100, 150], ["tanh"] * 3, "linear")
    X = np.random.default_rng(11).normal(size=(211, 200)).astype(np.float32)
    ref = net(torch.as_tensor(X)).detach().numpy()
    out = FusedHourglassForward(net, DEV).forward(torch.as_tensor(X)).cpu().numpy()
    np.testing.assert_allclose(out, ref, atol=6e-2, rtol=6e-2)


def test_fused_training_x256_matches_autograd():
    """W=256 grads vs fp32 autograd on one step (200-tag hourglass layer)."""
    from gordo_amd.ops import ext
    from gordo_amd.ops.hourglass_engine import _PackedParams, pad_rows_bf16

    torch.manual_seed(12)
    net = FeedForwardNet(200, 200, [144], ["tanh"], "linear")
    ref_net = FeedForwardNet(200, 200, [144], ["tanh"], "linear")
    ref_net.load_state_dict(net.state_dict())
    X = np.random.default_rng(12).normal(size=(256, 200)).astype(np.float32)

    E = ext()
    p = _PackedParams(net, DEV)
    assert p.wmax == 256
    WM, L, F_out = p.wmax, p.L, 200
    B = 256
    MT = int(E.MT)
    n_tiles = (B + MT - 1) // MT
    Xd = pad_rows_bf16(X, DEV, WM)
    dw = torch.zeros(n_tiles, L, WM * WM, dtype=torch.bfloat16, device=DEV)
    db = torch.zeros(n_tiles, L, WM, dtype=torch.bfloat16, device=DEV)
    loss = torch.zeros(1, dtype=torch.float32, device=DEV)
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)
    E.hg_train_step(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                    loss, B, F_out, p.act_ids, gscale, True, None, WM, 0, 0)
    torch.cuda.synchronize()

    Xref = Xd[:, :200].float().cpu()
    out = ref_net(Xref)
    loss_ref = torch.nn.functional.mse_loss(out, Xref)
    loss_ref.backward()
    kernel_loss = float(loss.item()) / (B * F_out)
    ref_val = float(loss_ref.detach())
    assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 5e-2

    linears = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)]
    dw_sum = dw.sum(dim=0, dtype=torch.float32).view(L, WM, WM).cpu()
    db_sum = db.sum(dim=0, dtype=torch.float32).view(L, WM).cpu()
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
        assert dw_sum[i, ninp:, :].abs().max().item() == 0.0
        assert dw_sum[i, :, o:].abs().max().item() == 0.0


def test_fused_trainer_x256_learns():
    rng = np.random.default_rng(13)
    Z = rng.normal(size=(4096, 16)).astype(np.float32)
    P = rng.normal(size=(16, 200)).astype(np.float32)
    X = np.tanh(Z @ P)
    torch.manual_seed(13)
    net = FeedForwardNet(200, 200, [150, 100, 150], ["tanh"] * 3, "linear")
    cfg = TrainConfig(epochs=20, batch_size=1024, lr=3e-3, seed=0)
    hist = FusedHourglassTrainer(net, DEV, cfg).fit(X, X)
    assert hist.engine.startswith("fused_hip")
    assert hist.loss[-1] < hist.loss[0] * 0.7, hist.loss


def test_fused_anomaly_epilogue_fuzz():
    """hg_forward's anomaly epilogue (scaled |err| + row-L2 total) vs numpy,
    random widths incl. the scale/shift fusion."""
    from gordo_amd.ops.hourglass_engine import FusedHourglassForward

    rng = np.random.default_rng(55)
    for trial in range(6):
        F = int(rng.integers(2, 257))
        L = int(rng.integers(1, 4))
        dims = [int(rng.integers(2, 257)) for _ in range(L - 1)]
        B = int(rng.integers(1, 150))
        torch.manual_seed(3000 + trial)
        net = FeedForwardNet(F, F, dims, ["tanh"] * (L - 1), "linear")
        fwd = FusedHourglassForward(net, DEV)
        scale = rng.uniform(0.5, 2.0, F).astype(np.float32)
        shift = rng.uniform(-0.5, 0.5, F).astype(np.float32)
        inv = rng.uniform(0.1, 3.0, F).astype(np.float32)
        fwd.set_input_scaler(scale, shift)
        X = rng.normal(size=(B, F)).astype(np.float32)
        yhat, err, total = fwd.anomaly(torch.as_tensor(X), inv)
        yhat, err, total = yhat.cpu().numpy(), err.cpu().numpy(), total.cpu().numpy()

        xs = X * scale + shift
        ref_y = net(torch.as_tensor(xs)).detach().numpy()
        msg = f"trial {trial}: F={F} dims={dims} B={B}"
        np.testing.assert_allclose(yhat, ref_y, atol=7e-2, rtol=7e-2, err_msg=msg)
        # epilogue math vs the kernel's own yhat (isolates the epilogue)
        xs_b = torch.as_tensor(xs).to(torch.bfloat16).float().numpy()
        ref_err = np.abs(yhat - xs_b) * inv
        np.testing.assert_allclose(err, ref_err, atol=2e-2, rtol=2e-2, err_msg=msg)
        np.testing.assert_allclose(
            total, np.sqrt((ref_err**2).sum(axis=1)), atol=5e-2, rtol=2e-2, err_msg=msg
        )


def test_fused_training_grads_fuzz():
    """Randomized-shape one-step grad parity vs fp32 autograd (all width
    tiers, random layer stacks/activations, y != X half the time)."""
    from gordo_amd.ops import ext
    from gordo_amd.ops.hourglass_engine import _PackedParams, pad_rows_bf16

    E = ext()
    MT = int(E.MT)
    rng = np.random.default_rng(321)
    acts = ["tanh", "relu", "sigmoid", "linear"]
    for trial in range(8):
        # trials 0..5: widths up to 256 with 1..4 layers; 6..7: deep (5..8
        # layers) narrow stacks — the raised W=64/128 layer limit (r2)
        if trial < 6:
            L = int(rng.integers(1, 5))
            widths = [int(rng.integers(3, 257)) for _ in range(L + 1)]
        else:
            L = int(rng.integers(5, 9))
            widths = [int(rng.integers(3, 129)) for _ in range(L + 1)]
        funcs = [str(rng.choice(acts))] * (L - 1)
        torch.manual_seed(500 + trial)
        net = FeedForwardNet(widths[0], widths[-1], widths[1:-1], funcs, "tanh")
        ref_net = FeedForwardNet(widths[0], widths[-1], widths[1:-1], funcs, "tanh")
        ref_net.load_state_dict(net.state_dict())
        B = int(rng.integers(1, 5)) * MT
        y_is_x = bool(rng.integers(0, 2)) and widths[0] == widths[-1]
        X = rng.normal(size=(B, widths[0])).astype(np.float32)
        Y = X if y_is_x else rng.normal(size=(B, widths[-1])).astype(np.float32)

        p = _PackedParams(net, DEV)
        WM, F_out = p.wmax, widths[-1]
        n_tiles = (B + MT - 1) // MT
        Xd = pad_rows_bf16(X, DEV, WM)
        Yd = Xd if y_is_x else pad_rows_bf16(Y, DEV, WM)
        dw = torch.zeros(n_tiles, L, WM * WM, dtype=torch.bfloat16, device=DEV)
        db = torch.zeros(n_tiles, L, WM, dtype=torch.bfloat16, device=DEV)
        loss = torch.zeros(1, dtype=torch.float32, device=DEV)
        counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
        gscale = 2.0 / (B * F_out)
        E.hg_train_step(Xd, Yd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                        loss, B, F_out, p.act_ids, gscale, y_is_x, None, WM, 0, 0,
                        [int(x) for x in widths])
        torch.cuda.synchronize()

        Xr = Xd[:, : widths[0]].float().cpu()
        Yr = Xr if y_is_x else Yd[:, :F_out].float().cpu()
        loss_ref = torch.nn.functional.mse_loss(ref_net(Xr), Yr)
        loss_ref.backward()
        msg = f"trial {trial}: widths={widths} funcs={funcs} B={B} y_is_x={y_is_x}"
        kernel_loss = float(loss.item()) / (B * F_out)
        ref_val = float(loss_ref.detach())
        assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 6e-2, msg

        linears = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)]
        dw_sum = dw.sum(dim=0, dtype=torch.float32).view(L, WM, WM).cpu()
        db_sum = db.sum(dim=0, dtype=torch.float32).view(L, WM).cpu()
        for i, lin in enumerate(linears):
            o, ninp = lin.weight.shape
            got_w = dw_sum[i, :ninp, :o].T
            ref_w = lin.weight.grad.detach()
            s = ref_w.abs().max().item() or 1.0
            assert (got_w - ref_w).abs().max().item() < 0.06 * s + 2e-4, f"{msg} layer {i} dW"
            got_b = db_sum[i, :o]
            ref_b = lin.bias.grad.detach()
            s = ref_b.abs().max().item() or 1.0
            assert (got_b - ref_b).abs().max().item() < 0.06 * s + 2e-4, f"{msg} layer {i} db"
            assert dw_sum[i, ninp:, :].abs().max().item() == 0.0, msg
            assert dw_sum[i, :, o:].abs().max().item() == 0.0, msg


def test_fused_forward_fuzz():
    """Seeded randomized shape fuzz of the fused forward vs fp32 torch:
    widths 2..256, 1..5 layers, mixed activations, awkward batch sizes."""
    rng = np.random.default_rng(123)
    acts = ["tanh", "relu", "sigmoid", "linear"]
    for trial in range(28):
        if trial < 25:
            L = int(rng.integers(1, 6))
            widths = [int(rng.integers(2, 257)) for _ in range(L + 1)]
        else:  # deep narrow stacks (6..8 layers, <=128 wide) — r2 limit
            L = int(rng.integers(6, 9))
            widths = [int(rng.integers(2, 129)) for _ in range(L + 1)]
        funcs = [str(rng.choice(acts)) for _ in range(L - 1)]
        out_func = str(rng.choice(acts))
        B = int(rng.integers(1, 200))
        torch.manual_seed(trial)
        net = FeedForwardNet(widths[0], widths[-1], widths[1:-1], funcs, out_func)
        X = rng.normal(size=(B, widths[0])).astype(np.float32)
        ref = net(torch.as_tensor(X)).detach().numpy()
        out = FusedHourglassForward(net, DEV).forward(torch.as_tensor(X)).cpu().numpy()
        np.testing.assert_allclose(
            out, ref, atol=6e-2, rtol=6e-2,
            err_msg=f"trial {trial}: widths={widths} funcs={funcs}+{out_func} B={B}",
        )


def test_default_hourglass_l7_fused_grads_match_autograd():
    """The FACTORY-DEFAULT hourglass (encoding_layers=3 -> 7 layers with the
    duplicated-bottleneck composition) must run the fused path at W=64
    (GA_MAXL raised to 8 in round 2 — it silently fell back to eager
    before). One-step grads vs fp32 autograd."""
    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.ops import ext
    from gordo_amd.ops.hourglass_engine import (
        _PackedParams,
        pad_rows_bf16,
        supported,
    )

    torch.manual_seed(9)
    net = feedforward_hourglass(50, 50)  # defaults: 3 encoding layers
    assert len(net.layer_widths) == 8  # 50-42-33-25-25-33-42-50
    assert supported(net) is None
    ref_net = feedforward_hourglass(50, 50)
    ref_net.load_state_dict(net.state_dict())
    X = np.random.default_rng(9).normal(size=(256, 50)).astype(np.float32)

    E = ext()
    p = _PackedParams(net, DEV)
    assert p.wmax == 64 and p.L == 7
    WM, L, F_out = p.wmax, p.L, 50
    B = 256
    MT = int(E.MT)
    n_tiles = (B + MT - 1) // MT
    Xd = pad_rows_bf16(X, DEV, WM)
    dw = torch.zeros(n_tiles, L, WM * WM, dtype=torch.bfloat16, device=DEV)
    db = torch.zeros(n_tiles, L, WM, dtype=torch.bfloat16, device=DEV)
    loss = torch.zeros(1, dtype=torch.float32, device=DEV)
    counters = torch.tensor([0, 1, 0], dtype=torch.int32, device=DEV)
    gscale = 2.0 / (B * F_out)
    E.hg_train_step(Xd, Xd, counters, p.w_bf16, p.wt_bf16, p.bias, dw, db,
                    loss, B, F_out, p.act_ids, gscale, True, None, WM, 0, 0,
                    [int(w) for w in p.widths])
    torch.cuda.synchronize()

    Xref = Xd[:, :50].float().cpu()
    loss_ref = torch.nn.functional.mse_loss(ref_net(Xref), Xref)
    loss_ref.backward()
    kernel_loss = float(loss.item()) / (B * F_out)
    ref_val = float(loss_ref.detach())
    assert abs(kernel_loss - ref_val) / max(ref_val, 1e-9) < 5e-2

    linears = [m for m in ref_net.net if isinstance(m, torch.nn.Linear)]
    assert len(linears) == 7
    dw_sum = dw.sum(dim=0, dtype=torch.float32).view(L, WM, WM).cpu()
    db_sum = db.sum(dim=0, dtype=torch.float32).view(L, WM).cpu()
    for i, lin in enumerate(linears):
        o, ninp = lin.weight.shape
        got_w = dw_sum[i, :ninp, :o].T
        ref_w = lin.weight.grad.detach()
        scale = ref_w.abs().max().item() or 1.0
        assert (got_w - ref_w).abs().max().item() < 0.06 * scale + 2e-4, f"layer {i} dW"
        got_b = db_sum[i, :o]
        ref_b = lin.bias.grad.detach()
        bscale = ref_b.abs().max().item() or 1.0
        assert (got_b - ref_b).abs().max().item() < 0.06 * bscale + 2e-4, f"layer {i} db"
        assert dw_sum[i, ninp:, :].abs().max().item() == 0.0


def test_default_hourglass_l7_trains_and_serves_fused(tmp_path):
    """End-to-end: default-kind estimator trains on the fused engine AND the
    server runtime builds the fused anomaly forward for the L=7 model."""
    import pandas as pd

    from gordo_amd.model.anomaly.diff import DiffBasedAnomalyDetector
    from gordo_amd.model.models import TorchAutoEncoder
    from gordo_amd.serializer import dump
    from gordo_amd.server.model_io import ModelRuntime
    from sklearn.pipeline import Pipeline
    from sklearn.preprocessing import MinMaxScaler

    det = DiffBasedAnomalyDetector(
        base_estimator=Pipeline(
            [
                ("scaler", MinMaxScaler()),
                ("model", TorchAutoEncoder(
                    kind="feedforward_hourglass", epochs=3, batch_size=4096, seed=0
                )),  # DEFAULT encoding_layers=3 -> L=7
            ]
        )
    )
    X = np.random.default_rng(5).uniform(0, 1, size=(20000, 50)).astype(np.float32)
    det.fit(X)
    est = det.base_estimator.steps[-1][1]
    assert est.history["engine"].startswith("fused_hip"), est.history["engine"]

    d = tmp_path / "m"
    dump(det, str(d), metadata={"dataset": {"tag_list": [{"name": f"t{i}"} for i in range(50)]}})
    rt = ModelRuntime(str(d))
    assert rt._fused is not None, "fused serving path must cover the default model"
    req = pd.DataFrame(X[:500], columns=[f"t{i}" for i in range(50)])
    frame = rt.anomaly(req, None)
    assert np.isfinite(frame["total-anomaly-score"].values).all()


def test_ping_pong_counters_track_steps():
    """The fused hg_reduce_adam bumps [batch, adam-t, epoch] across two
    ping-pong counter slots; after n steps the LIVE slot (the one the next
    step reads) must hold exactly n steps' worth of state."""
    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer

    torch.manual_seed(3)
    module = feedforward_hourglass(50, 50, encoding_layers=2, compression_factor=0.5)
    X = np.random.default_rng(3).uniform(0, 1, size=(4096, 50)).astype(np.float32)
    cfg = TrainConfig(batch_size=1024, lr=1e-3, dtype="bf16")
    trainer = HourglassStepTrainer(module, X, 1024, cfg, dp=DPContext(device=DEV))
    assert trainer.fused
    n_batches = trainer.n_batches  # 4096/1024 = 4

    for n in (1, 2, 3, 7, 10):  # odd and even parities, epoch wraps
        while int(trainer._cnt[trainer._parity][1].item()) < n:
            trainer.step()
        trainer.synchronize()
        live = trainer._cnt[trainer._parity].cpu().tolist()
        assert live == [n % n_batches, n, n // n_batches], (n, live)

    # steps() (pair/quad graphs) advances the same counter trajectory
    trainer.steps(9)
    trainer.synchronize()
    live = trainer._cnt[trainer._parity].cpu().tolist()
    assert live == [19 % n_batches, 19, 19 // n_batches], live
