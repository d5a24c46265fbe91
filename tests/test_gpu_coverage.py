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
