"""GPU numerics tests for the fused LSTM cell kernels (K4/K5/K6) vs plain
PyTorch fp32 references. All @pytest.mark.gpu."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from gordo_amd.model.torch_nets import LSTMNet
    from gordo_amd.ops.lstm_engine import (
        H4,
        WMAX,
        FusedLSTMForward,
        FusedLSTMTrainer,
        _LSTMPackedParams,
    )
    from gordo_amd.runtime.trainer import TrainConfig
else:
    pytest.skip("no GPU", allow_module_level=True)

DEV = torch.device("cuda:0")


def make_lstm(n_features=20, dims=(32,), T=12, seed=0, out=None):
    torch.manual_seed(seed)
    return LSTMNet(
        n_features, out or n_features, list(dims), [None] * len(dims),
        out_func="linear", lookback_window=T,
    )


def ref_forward(net, X, T):
    """fp32 torch forward over all rolling windows of bf16-rounded X."""
    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    n = len(Xb) - T + 1
    w = torch.stack([Xb[i : i + T] for i in range(n)])
    with torch.no_grad():
        return net(w).numpy()


def test_lstm_fused_forward_matches_torch():
    T = 12
    net = make_lstm(T=T, seed=1)
    X = np.random.default_rng(0).normal(size=(200, 20)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    np.testing.assert_allclose(out, ref, atol=4e-2, rtol=4e-2)


def test_lstm_fused_forward_stacked_long():
    """2-layer stack, lookback 144 (the headline config's shape)."""
    T = 144
    net = make_lstm(n_features=50, dims=(48, 24), T=T, seed=2)
    X = np.random.default_rng(1).normal(size=(400, 50)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    # longer bf16 recurrence accumulates more rounding
    np.testing.assert_allclose(out, ref, atol=8e-2, rtol=8e-2)


def test_lstm_grads_match_autograd():
    """One fused step's wcat/bias/head grads vs fp32 torch autograd."""
    T, B, F = 12, 256, 20
    net = make_lstm(n_features=F, dims=(32,), T=T, seed=3)
    ref_net = make_lstm(n_features=F, dims=(32,), T=T, seed=3)
    X = np.random.default_rng(2).normal(size=(B + T + 2, F)).astype(np.float32)

    cfg = TrainConfig(epochs=1, batch_size=B)
    tr = FusedLSTMTrainer(net, DEV, cfg)
    p = _LSTMPackedParams(net, DEV)
    tr.p = p
    from gordo_amd.ops.hourglass_engine import pad_rows_bf16

    Xd = pad_rows_bf16(X, DEV, p.W)
    st = tr._alloc_step_state(Xd, Xd, B, T, 1, B, 0, F)
    win = torch.arange(B, dtype=torch.int32, device=DEV)
    tr._compute_grads(st, win)
    torch.cuda.synchronize()

    # fp32 reference on the same bf16-rounded inputs
    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    wins = torch.stack([Xb[i : i + T] for i in range(B)])
    tgt = torch.stack([Xb[i + T - 1] for i in range(B)])
    out = ref_net(wins)
    loss = torch.nn.functional.mse_loss(out, tgt)
    loss.backward()

    # loss agreement
    kern_loss = float(st["loss_acc"].item()) / (B * F)
    assert abs(kern_loss - float(loss)) / max(float(loss), 1e-9) < 6e-2

    cell = ref_net.cells[0]
    H, I = cell.hidden_size, cell.input_size
    W = p.Wl[0]  # per-layer tier (r2): a 20->32 cell runs the W=32 kernels
    gw = p.wcat_master[0].grad.cpu()
    gb = p.bias_master[0].grad.cpu()
    for g in range(4):
        ref_ih = cell.weight_ih_l0.grad[g * H : (g + 1) * H]
        ref_hh = cell.weight_hh_l0.grad[g * H : (g + 1) * H]
        # torch's two bias vectors get IDENTICAL grads (both add to the same
        # pre-activation); the fused combined bias's grad equals either one.
        ref_b = cell.bias_ih_l0.grad[g * H : (g + 1) * H]
        got_ih = gw[g * W : g * W + H, :I]
        got_hh = gw[g * W : g * W + H, W : W + H]
        got_b = gb[g * W : g * W + H]
        s = ref_ih.abs().max().item() or 1.0
        assert (got_ih - ref_ih).abs().max().item() < 0.08 * s + 2e-4, f"gate {g} dW_ih"
        s = ref_hh.abs().max().item() or 1.0
        assert (got_hh - ref_hh).abs().max().item() < 0.08 * s + 2e-4, f"gate {g} dW_hh"
        s = ref_b.abs().max().item() or 1.0
        assert (got_b - ref_b).abs().max().item() < 0.08 * s + 2e-4, f"gate {g} db"
        # padded rows/cols carry zero grad
        if H < W:
            assert gw[g * W + H : (g + 1) * W].abs().max().item() == 0.0
    # head grads
    ref_hw = ref_net.head.weight.grad
    got_hw = p.head_wm.grad.cpu()[: ref_hw.shape[0], : ref_hw.shape[1]]
    s = ref_hw.abs().max().item() or 1.0
    assert (got_hw - ref_hw).abs().max().item() < 0.08 * s + 2e-4, "head dW"


def test_lstm_estimator_learns_fused():
    from gordo_amd.model.models import TorchLSTMAutoEncoder

    rng = np.random.default_rng(4)
    t = np.arange(3000)
    base = np.stack(
        [np.sin(t / (8 + i)) + 0.05 * rng.standard_normal(len(t)) for i in range(20)],
        axis=1,
    ).astype(np.float32)
    model = TorchLSTMAutoEncoder(
        kind="lstm_hourglass", lookback_window=16, epochs=5, batch_size=512, lr=3e-3
    )
    model.fit(base)
    assert model.history["engine"] == "fused_hip_lstm", model.history
    assert model.history["loss"][-1] < model.history["loss"][0], model.history["loss"]
    out = model.predict(base[:200])
    assert out.shape == (200 - 16 + 1, 20)
    assert np.isfinite(out).all()


def test_lstm_forecast_fused():
    from gordo_amd.model.models import TorchLSTMForecast

    X = np.random.default_rng(5).normal(size=(1500, 10)).astype(np.float32)
    model = TorchLSTMForecast(
        kind="lstm_symmetric", dims=[32], lookback_window=12, epochs=2, batch_size=256
    )
    model.fit(X)
    assert model.history["engine"] == "fused_hip_lstm", model.history
    score = model.score(X)
    assert np.isfinite(score)


def test_lstm_grads_fuzz():
    """Randomized-shape one-step grad parity vs fp32 autograd (both width
    tiers, random stacks/lookbacks); also asserts padded wcat rows stay
    zero-gradient."""
    from gordo_amd.ops.hourglass_engine import pad_rows_bf16
    from gordo_amd.runtime.trainer import TrainConfig

    rng = np.random.default_rng(88)
    for trial in range(6):
        # trials 0..3 sample the 64/128 tiers; 4..5 force the W=256 band
        lo, hi = (3, 129) if trial < 4 else (129, 257)
        F = int(rng.integers(lo, hi))
        L = int(rng.integers(1, 3))
        dims = [int(rng.integers(max(4, lo // 2), hi)) for _ in range(L)]
        T = int(rng.integers(3, 17))
        B = 256
        torch.manual_seed(2000 + trial)
        net = LSTMNet(F, F, dims, [None] * L, out_func="linear", lookback_window=T)
        ref_net = LSTMNet(F, F, dims, [None] * L, out_func="linear", lookback_window=T)
        ref_net.load_state_dict(net.state_dict())
        X = rng.normal(size=(B + T + 2, F)).astype(np.float32)
        msg = f"trial {trial}: F={F} dims={dims} T={T}"

        cfg = TrainConfig(epochs=1, batch_size=B)
        tr = FusedLSTMTrainer(net, DEV, cfg)
        p = _LSTMPackedParams(net, DEV)
        tr.p = p
        Xd = pad_rows_bf16(X, DEV, p.W)
        st = tr._alloc_step_state(Xd, Xd, B, T, 1, B, 0, F)
        win = torch.arange(B, dtype=torch.int32, device=DEV)
        tr._compute_grads(st, win)
        torch.cuda.synchronize()

        Xb = torch.as_tensor(X).to(torch.bfloat16).float()
        wins = torch.stack([Xb[i : i + T] for i in range(B)])
        tgt = torch.stack([Xb[i + T - 1] for i in range(B)])
        loss = torch.nn.functional.mse_loss(ref_net(wins), tgt)
        loss.backward()

        kern_loss = float(st["loss_acc"].item()) / (B * F)
        assert abs(kern_loss - float(loss.detach())) / max(float(loss), 1e-9) < 6e-2, msg

        for li, cell in enumerate(ref_net.cells):
            H, I = cell.hidden_size, cell.input_size
            W = p.Wl[li]  # per-layer tier (r2 mixed-width stacks)
            gw = p.wcat_master[li].grad.cpu()
            gb = p.bias_master[li].grad.cpu()
            for g in range(4):
                ref_ih = cell.weight_ih_l0.grad[g * H : (g + 1) * H]
                ref_hh = cell.weight_hh_l0.grad[g * H : (g + 1) * H]
                ref_b = cell.bias_ih_l0.grad[g * H : (g + 1) * H]
                got_ih = gw[g * W : g * W + H, :I]
                got_hh = gw[g * W : g * W + H, W : W + H]
                got_b = gb[g * W : g * W + H]
                s = ref_ih.abs().max().item() or 1.0
                assert (got_ih - ref_ih).abs().max().item() < 0.09 * s + 3e-4, f"{msg} l{li} g{g} dW_ih"
                s = ref_hh.abs().max().item() or 1.0
                assert (got_hh - ref_hh).abs().max().item() < 0.09 * s + 3e-4, f"{msg} l{li} g{g} dW_hh"
                s = ref_b.abs().max().item() or 1.0
                assert (got_b - ref_b).abs().max().item() < 0.09 * s + 3e-4, f"{msg} l{li} g{g} db"
                if H < W:
                    assert gw[g * W + H : (g + 1) * W].abs().max().item() == 0.0, msg
                if I < W:
                    assert gw[g * W : g * W + H, I:W].abs().max().item() == 0.0, msg
        ref_hw = ref_net.head.weight.grad
        got_hw = p.head_wm.grad.cpu()[: ref_hw.shape[0], : ref_hw.shape[1]]
        s = ref_hw.abs().max().item() or 1.0
        assert (got_hw - ref_hw).abs().max().item() < 0.09 * s + 3e-4, f"{msg} head dW"


def test_lstm_fused_forward_fuzz():
    """Seeded randomized shape fuzz of the fused LSTM forward vs fp32 torch:
    features/hidden 2..256 (all three width tiers), stacks 1..3, lookback
    2..24, awkward row counts — exercises the padding/packing envelope."""
    rng = np.random.default_rng(77)
    for trial in range(12):
        # trials 0..9 sample the 64/128 tiers; 10..11 force the W=256 band
        lo, hi = (2, 129) if trial < 10 else (129, 257)
        F = int(rng.integers(lo, hi))
        L = int(rng.integers(1, 4))
        dims = [int(rng.integers(max(2, lo), hi)) for _ in range(L)]
        T = int(rng.integers(2, 25))
        n = int(rng.integers(T + 1, T + 120))
        torch.manual_seed(1000 + trial)
        net = LSTMNet(F, F, dims, [None] * L, out_func="linear", lookback_window=T)
        X = rng.normal(size=(n, F)).astype(np.float32)
        ref = ref_forward(net, X, T)
        out = FusedLSTMForward(net, DEV).predict(X)
        assert out.shape == ref.shape
        np.testing.assert_allclose(
            out, ref, atol=8e-2, rtol=8e-2,
            err_msg=f"trial {trial}: F={F} dims={dims} T={T} n={n}",
        )


# ---------------------------------------------------------------------------
# wide variant (W=128): feature/hidden widths 65..128
# ---------------------------------------------------------------------------
def test_wide_lstm_fused_forward_matches_torch():
    T = 16
    net = make_lstm(n_features=100, dims=(96,), T=T, seed=7)
    from gordo_amd.ops.lstm_engine import lstm_width

    assert lstm_width(net) == 128
    X = np.random.default_rng(7).normal(size=(300, 100)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    np.testing.assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_wide_lstm_grads_match_autograd():
    T, B, F = 12, 256, 100
    net = make_lstm(n_features=F, dims=(96,), T=T, seed=8)
    ref_net = make_lstm(n_features=F, dims=(96,), T=T, seed=8)
    X = np.random.default_rng(8).normal(size=(B + T + 2, F)).astype(np.float32)

    from gordo_amd.ops.hourglass_engine import pad_rows_bf16
    from gordo_amd.runtime.trainer import TrainConfig

    cfg = TrainConfig(epochs=1, batch_size=B)
    tr = FusedLSTMTrainer(net, DEV, cfg)
    pp = _LSTMPackedParams(net, DEV)
    assert pp.W == 128
    tr.p = pp
    Xd = pad_rows_bf16(X, DEV, pp.W)
    st = tr._alloc_step_state(Xd, Xd, B, T, 1, B, 0, F)
    win = torch.arange(B, dtype=torch.int32, device=DEV)
    tr._compute_grads(st, win)
    torch.cuda.synchronize()

    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    wins = torch.stack([Xb[i : i + T] for i in range(B)])
    tgt = torch.stack([Xb[i + T - 1] for i in range(B)])
    loss = torch.nn.functional.mse_loss(ref_net(wins), tgt)
    loss.backward()

    kern_loss = float(st["loss_acc"].item()) / (B * F)
    assert abs(kern_loss - float(loss.detach())) / max(float(loss), 1e-9) < 6e-2

    cell = ref_net.cells[0]
    H, I = cell.hidden_size, cell.input_size
    W = pp.W
    gw = pp.wcat_master[0].grad.cpu()
    gb = pp.bias_master[0].grad.cpu()
    for g in range(4):
        ref_ih = cell.weight_ih_l0.grad[g * H : (g + 1) * H]
        ref_hh = cell.weight_hh_l0.grad[g * H : (g + 1) * H]
        ref_b = cell.bias_ih_l0.grad[g * H : (g + 1) * H]
        got_ih = gw[g * W : g * W + H, :I]
        got_hh = gw[g * W : g * W + H, W : W + H]
        got_b = gb[g * W : g * W + H]
        s = ref_ih.abs().max().item() or 1.0
        assert (got_ih - ref_ih).abs().max().item() < 0.08 * s + 2e-4, f"gate {g} dW_ih"
        s = ref_hh.abs().max().item() or 1.0
        assert (got_hh - ref_hh).abs().max().item() < 0.08 * s + 2e-4, f"gate {g} dW_hh"
        s = ref_b.abs().max().item() or 1.0
        assert (got_b - ref_b).abs().max().item() < 0.08 * s + 2e-4, f"gate {g} db"
        assert gw[g * W + H : (g + 1) * W].abs().max().item() == 0.0
    ref_hw = ref_net.head.weight.grad
    got_hw = pp.head_wm.grad.cpu()[: ref_hw.shape[0], : ref_hw.shape[1]]
    s = ref_hw.abs().max().item() or 1.0
    assert (got_hw - ref_hw).abs().max().item() < 0.08 * s + 2e-4, "head dW"


def test_wide_lstm_estimator_learns_fused():
    from gordo_amd.model.models import TorchLSTMAutoEncoder

    rng = np.random.default_rng(9)
    t = np.arange(4000)
    X = np.stack(
        [np.sin(t / (8 + i % 11)) + 0.05 * rng.standard_normal(len(t)) for i in range(100)],
        axis=1,
    ).astype(np.float32)
    model = TorchLSTMAutoEncoder(
        kind="lstm_symmetric", dims=[96], lookback_window=16, epochs=4,
        batch_size=1024, lr=3e-3,
    )
    model.fit(X)
    assert model.history["engine"] == "fused_hip_lstm", model.history
    assert model.history["loss"][-1] < model.history["loss"][0]
    out = model.predict(X[:200])
    assert out.shape == (200 - 16 + 1, 100)
    assert np.isfinite(out).all()


# ---------------------------------------------------------------------------
# W=256 variant: feature/hidden widths 129..256 (the 200-tag asset class,
# VERDICT r1 item #3). Column-half-split cells in lstm_wide256_kernels.hip.
# ---------------------------------------------------------------------------
def test_w256_lstm_fused_forward_matches_torch():
    T = 16
    net = make_lstm(n_features=200, dims=(150,), T=T, seed=10)
    from gordo_amd.ops.lstm_engine import lstm_width

    assert lstm_width(net) == 256
    X = np.random.default_rng(10).normal(size=(300, 200)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    np.testing.assert_allclose(out, ref, atol=5e-2, rtol=5e-2)


def test_w256_lstm_fused_forward_stacked_long():
    """2-layer 200-tag stack, lookback 144 — h_seq-input path (win=None)."""
    T = 144
    net = make_lstm(n_features=200, dims=(150, 100), T=T, seed=11)
    X = np.random.default_rng(11).normal(size=(400, 200)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    np.testing.assert_allclose(out, ref, atol=8e-2, rtol=8e-2)


def _w256_grad_check(dims, seed, F=200, T=12, B=256):
    net = make_lstm(n_features=F, dims=dims, T=T, seed=seed)
    ref_net = make_lstm(n_features=F, dims=dims, T=T, seed=seed)
    X = np.random.default_rng(seed).normal(size=(B + T + 2, F)).astype(np.float32)

    from gordo_amd.ops.hourglass_engine import pad_rows_bf16

    cfg = TrainConfig(epochs=1, batch_size=B)
    tr = FusedLSTMTrainer(net, DEV, cfg)
    pp = _LSTMPackedParams(net, DEV)
    assert pp.W == 256
    tr.p = pp
    Xd = pad_rows_bf16(X, DEV, pp.W)
    st = tr._alloc_step_state(Xd, Xd, B, T, 1, B, 0, F)
    win = torch.arange(B, dtype=torch.int32, device=DEV)
    tr._compute_grads(st, win)
    torch.cuda.synchronize()

    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    wins = torch.stack([Xb[i : i + T] for i in range(B)])
    tgt = torch.stack([Xb[i + T - 1] for i in range(B)])
    loss = torch.nn.functional.mse_loss(ref_net(wins), tgt)
    loss.backward()

    kern_loss = float(st["loss_acc"].item()) / (B * F)
    assert abs(kern_loss - float(loss.detach())) / max(float(loss), 1e-9) < 6e-2

    W = pp.W
    for l, cell in enumerate(ref_net.cells):
        H, I = cell.hidden_size, cell.input_size
        gw = pp.wcat_master[l].grad.cpu()
        gb = pp.bias_master[l].grad.cpu()
        for g in range(4):
            ref_ih = cell.weight_ih_l0.grad[g * H : (g + 1) * H]
            ref_hh = cell.weight_hh_l0.grad[g * H : (g + 1) * H]
            ref_b = cell.bias_ih_l0.grad[g * H : (g + 1) * H]
            got_ih = gw[g * W : g * W + H, :I]
            got_hh = gw[g * W : g * W + H, W : W + H]
            got_b = gb[g * W : g * W + H]
            s = ref_ih.abs().max().item() or 1.0
            assert (got_ih - ref_ih).abs().max().item() < 0.08 * s + 2e-4, f"layer {l} gate {g} dW_ih"
            s = ref_hh.abs().max().item() or 1.0
            assert (got_hh - ref_hh).abs().max().item() < 0.08 * s + 2e-4, f"layer {l} gate {g} dW_hh"
            s = ref_b.abs().max().item() or 1.0
            assert (got_b - ref_b).abs().max().item() < 0.08 * s + 2e-4, f"layer {l} gate {g} db"
            assert gw[g * W + H : (g + 1) * W].abs().max().item() == 0.0
    ref_hw = ref_net.head.weight.grad
    got_hw = pp.head_wm.grad.cpu()[: ref_hw.shape[0], : ref_hw.shape[1]]
    s = ref_hw.abs().max().item() or 1.0
    assert (got_hw - ref_hw).abs().max().item() < 0.08 * s + 2e-4, "head dW"


def test_w256_lstm_grads_match_autograd():
    """Single 200->150 cell: rec-only backward path."""
    _w256_grad_check((150,), seed=12)


def test_w256_lstm_grads_match_autograd_stacked():
    """200->150->130 stack: exercises the fused-dx waves of
    lstm_bwd_wide256 (lower layer's dh comes from the upper layer's dx)
    and the h_seq-input forward path."""
    _w256_grad_check((150, 130), seed=13)


def test_w256_lstm_estimator_learns_fused():
    from gordo_amd.model.models import TorchLSTMAutoEncoder

    rng = np.random.default_rng(14)
    t = np.arange(4000)
    X = np.stack(
        [np.sin(t / (8 + i % 13)) + 0.05 * rng.standard_normal(len(t)) for i in range(200)],
        axis=1,
    ).astype(np.float32)
    model = TorchLSTMAutoEncoder(
        kind="lstm_symmetric", dims=[150], lookback_window=16, epochs=4,
        batch_size=1024, lr=3e-3,
    )
    model.fit(X)
    assert model.history["engine"] == "fused_hip_lstm", model.history
    assert model.history["loss"][-1] < model.history["loss"][0]
    out = model.predict(X[:200])
    assert out.shape == (200 - 16 + 1, 200)
    assert np.isfinite(out).all()


def test_mixed_tier_stack_grads_match_autograd():
    """Per-layer width tiers (r2): a 200->150->100->40 stack runs cells at
    W=256/256/128/64 with zero-padded boundary copies; grads must still match
    fp32 autograd layer by layer."""
    from gordo_amd.ops.lstm_engine import _tier

    T, B, F = 10, 128, 200
    dims = (150, 100, 40)
    net = make_lstm(n_features=F, dims=dims, T=T, seed=21)
    assert [_tier(max(c.input_size, c.hidden_size)) for c in net.cells] == [256, 256, 128]
    ref_net = make_lstm(n_features=F, dims=dims, T=T, seed=21)
    ref_net.load_state_dict(net.state_dict())
    X = np.random.default_rng(21).normal(size=(B + T + 2, F)).astype(np.float32)

    from gordo_amd.ops.hourglass_engine import pad_rows_bf16

    cfg = TrainConfig(epochs=1, batch_size=B)
    tr = FusedLSTMTrainer(net, DEV, cfg)
    pp = _LSTMPackedParams(net, DEV)
    assert pp.Wl == [256, 256, 128] and pp.W_head == 256  # F_out=200
    tr.p = pp
    Xd = pad_rows_bf16(X, DEV, pp.W)
    st = tr._alloc_step_state(Xd, Xd, B, T, 1, B, 0, F)
    win = torch.arange(B, dtype=torch.int32, device=DEV)
    tr._compute_grads(st, win)
    torch.cuda.synchronize()

    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    wins = torch.stack([Xb[i : i + T] for i in range(B)])
    tgt = torch.stack([Xb[i + T - 1] for i in range(B)])
    loss = torch.nn.functional.mse_loss(ref_net(wins), tgt)
    loss.backward()

    kern_loss = float(st["loss_acc"].item()) / (B * F)
    assert abs(kern_loss - float(loss.detach())) / max(float(loss), 1e-9) < 6e-2

    for l, cell in enumerate(ref_net.cells):
        H, I = cell.hidden_size, cell.input_size
        W = pp.Wl[l]
        gw = pp.wcat_master[l].grad.cpu()
        gb = pp.bias_master[l].grad.cpu()
        for g in range(4):
            ref_ih = cell.weight_ih_l0.grad[g * H : (g + 1) * H]
            ref_hh = cell.weight_hh_l0.grad[g * H : (g + 1) * H]
            ref_b = cell.bias_ih_l0.grad[g * H : (g + 1) * H]
            got_ih = gw[g * W : g * W + H, :I]
            got_hh = gw[g * W : g * W + H, W : W + H]
            got_b = gb[g * W : g * W + H]
            s = ref_ih.abs().max().item() or 1.0
            assert (got_ih - ref_ih).abs().max().item() < 0.08 * s + 2e-4, f"l{l} g{g} dW_ih"
            s = ref_hh.abs().max().item() or 1.0
            assert (got_hh - ref_hh).abs().max().item() < 0.08 * s + 2e-4, f"l{l} g{g} dW_hh"
            s = ref_b.abs().max().item() or 1.0
            assert (got_b - ref_b).abs().max().item() < 0.08 * s + 2e-4, f"l{l} g{g} db"
            assert gw[g * W + H : (g + 1) * W].abs().max().item() == 0.0
    ref_hw = ref_net.head.weight.grad
    got_hw = pp.head_wm.grad.cpu()[: ref_hw.shape[0], : ref_hw.shape[1]]
    s = ref_hw.abs().max().item() or 1.0
    assert (got_hw - ref_hw).abs().max().item() < 0.08 * s + 2e-4, "head dW"


def test_mixed_tier_forward_matches_torch():
    T = 12
    net = make_lstm(n_features=150, dims=(100, 60), T=T, seed=22)
    from gordo_amd.ops.lstm_engine import _tier

    assert [_tier(max(c.input_size, c.hidden_size)) for c in net.cells] == [256, 128]
    X = np.random.default_rng(22).normal(size=(260, 150)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    np.testing.assert_allclose(out, ref, atol=6e-2, rtol=6e-2)


# ---------------------------------------------------------------------------
# W=32 narrow tier (real widths <= 32): register-resident weights at half the
# lane width; the headline 50-tag stack's 25->25 cell runs here.
# ---------------------------------------------------------------------------
def test_h32_lstm_fused_forward_matches_torch():
    from gordo_amd.ops.lstm_engine import _tier

    T = 16
    net = make_lstm(n_features=25, dims=(25, 20), T=T, seed=30)
    assert [_tier(max(c.input_size, c.hidden_size)) for c in net.cells] == [32, 32]
    X = np.random.default_rng(30).normal(size=(300, 25)).astype(np.float32)
    ref = ref_forward(net, X, T)
    out = FusedLSTMForward(net, DEV).predict(X)
    assert out.shape == ref.shape
    np.testing.assert_allclose(out, ref, atol=4e-2, rtol=4e-2)


def test_h32_lstm_grads_match_autograd():
    """Full-narrow stack: 25->25->20, all cells at W=32 (head at 64)."""
    from gordo_amd.ops.hourglass_engine import pad_rows_bf16

    T, B, F = 12, 256, 25
    net = make_lstm(n_features=F, dims=(25, 20), T=T, seed=31)
    ref_net = make_lstm(n_features=F, dims=(25, 20), T=T, seed=31)
    ref_net.load_state_dict(net.state_dict())
    X = np.random.default_rng(31).normal(size=(B + T + 2, F)).astype(np.float32)

    cfg = TrainConfig(epochs=1, batch_size=B)
    tr = FusedLSTMTrainer(net, DEV, cfg)
    pp = _LSTMPackedParams(net, DEV)
    assert pp.Wl == [32, 32] and pp.W_head == 64
    tr.p = pp
    Xd = pad_rows_bf16(X, DEV, pp.W)
    st = tr._alloc_step_state(Xd, Xd, B, T, 1, B, 0, F)
    win = torch.arange(B, dtype=torch.int32, device=DEV)
    tr._compute_grads(st, win)
    torch.cuda.synchronize()

    Xb = torch.as_tensor(X).to(torch.bfloat16).float()
    wins = torch.stack([Xb[i : i + T] for i in range(B)])
    tgt = torch.stack([Xb[i + T - 1] for i in range(B)])
    loss = torch.nn.functional.mse_loss(ref_net(wins), tgt)
    loss.backward()

    kern_loss = float(st["loss_acc"].item()) / (B * F)
    assert abs(kern_loss - float(loss.detach())) / max(float(loss), 1e-9) < 6e-2

    for l, cell in enumerate(ref_net.cells):
        H, I = cell.hidden_size, cell.input_size
        W = pp.Wl[l]
        gw = pp.wcat_master[l].grad.cpu()
        gb = pp.bias_master[l].grad.cpu()
        for g in range(4):
            ref_ih = cell.weight_ih_l0.grad[g * H : (g + 1) * H]
            ref_hh = cell.weight_hh_l0.grad[g * H : (g + 1) * H]
            ref_b = cell.bias_ih_l0.grad[g * H : (g + 1) * H]
            got_ih = gw[g * W : g * W + H, :I]
            got_hh = gw[g * W : g * W + H, W : W + H]
            got_b = gb[g * W : g * W + H]
            s = ref_ih.abs().max().item() or 1.0
            assert (got_ih - ref_ih).abs().max().item() < 0.08 * s + 2e-4, f"l{l} g{g} dW_ih"
            s = ref_hh.abs().max().item() or 1.0
            assert (got_hh - ref_hh).abs().max().item() < 0.08 * s + 2e-4, f"l{l} g{g} dW_hh"
            s = ref_b.abs().max().item() or 1.0
            assert (got_b - ref_b).abs().max().item() < 0.08 * s + 2e-4, f"l{l} g{g} db"
            if H < W:
                assert gw[g * W + H : (g + 1) * W].abs().max().item() == 0.0
    ref_hw = ref_net.head.weight.grad
    got_hw = pp.head_wm.grad.cpu()[: ref_hw.shape[0], : ref_hw.shape[1]]
    s = ref_hw.abs().max().item() or 1.0
    assert (got_hw - ref_hw).abs().max().item() < 0.08 * s + 2e-4, "head dW"


def test_headline_stack_uses_narrow_tier_and_learns():
    """The BASELINE config #3 stack (50 tags, dims 38-25-25-38) now runs its
    25->25 cell at W=32; the estimator must train fused and learn."""
    from gordo_amd.model.models import TorchLSTMAutoEncoder
    from gordo_amd.model.factories.lstm_autoencoder import lstm_hourglass
    from gordo_amd.ops.lstm_engine import _tier

    m = lstm_hourglass(50, 50, lookback_window=16, encoding_layers=2, compression_factor=0.5)
    assert [_tier(max(c.input_size, c.hidden_size)) for c in m.cells] == [64, 64, 32, 64]

    rng = np.random.default_rng(32)
    t = np.arange(3000)
    X = np.stack(
        [np.sin(t / (8 + i % 9)) + 0.05 * rng.standard_normal(len(t)) for i in range(50)],
        axis=1,
    ).astype(np.float32)
    model = TorchLSTMAutoEncoder(
        kind="lstm_hourglass", lookback_window=16, encoding_layers=2,
        compression_factor=0.5, epochs=4, batch_size=1024, lr=3e-3,
    )
    model.fit(X)
    assert model.history["engine"] == "fused_hip_lstm", model.history
    assert model.history["loss"][-1] < model.history["loss"][0]
    out = model.predict(X[:200])
    assert out.shape == (200 - 16 + 1, 50)
    assert np.isfinite(out).all()
