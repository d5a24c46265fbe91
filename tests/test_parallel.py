"""Multi-process DP tests on CPU (gloo, world_size=2) — covers the comm code
path that runs over RCCL on the GPU node."""
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _dp_worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    torch.manual_seed(100 + rank)  # deliberately different init per rank

    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    dp = DPContext(backend="gloo", device=torch.device("cpu"))
    module = feedforward_hourglass(10, 10, encoding_layers=2, compression_factor=0.5)
    X = np.random.default_rng(rank).normal(size=(512, 10)).astype(np.float32)
    cfg = TrainConfig(batch_size=128, lr=1e-3, dtype="fp32")
    trainer = HourglassStepTrainer(module, X, 128, cfg, dp=dp)
    for _ in range(5):
        trainer.step()
    dp.barrier()
    vec = trainer.params_vector().cpu().numpy()
    q.put((rank, vec))
    import torch.distributed as dist

    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_dp_ranks_stay_in_sync():
    """After broadcast-init + per-step all-reduced grads, both ranks hold
    identical parameters."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29501 + os.getpid() % 500
    procs = [
        ctx.Process(target=_dp_worker, args=(r, 2, port, q)) for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, vec = q.get()
        results[rank] = vec
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    np.testing.assert_allclose(results[0], results[1], rtol=1e-6, atol=1e-7)


def test_dp_context_single_process():
    from gordo_amd.parallel.ddp import DPContext

    dp = DPContext(device=torch.device("cpu"))
    assert not dp.enabled
    t = torch.ones(4)
    dp.all_reduce_mean(t)  # no-op
    assert t.sum().item() == 4.0


def test_checkpoint_resume_bit_exact(tmp_path):
    """save_checkpoint -> more steps -> load_checkpoint -> same steps replays
    the identical trajectory (params + Adam moments + sampler state)."""
    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    torch.manual_seed(7)
    module = feedforward_hourglass(10, 10, encoding_layers=2, compression_factor=0.5)
    X = np.random.default_rng(7).normal(size=(512, 10)).astype(np.float32)
    dp = DPContext(device=torch.device("cpu"))
    trainer = HourglassStepTrainer(module, X, 128, TrainConfig(batch_size=128, lr=1e-3, dtype="fp32"), dp=dp)
    for _ in range(3):
        trainer.step()
    ckpt = tmp_path / "step3.pt"
    trainer.save_checkpoint(str(ckpt))
    for _ in range(3):
        trainer.step()
    vec1 = trainer.params_vector().numpy()
    trainer.load_checkpoint(str(ckpt))
    for _ in range(3):
        trainer.step()
    vec2 = trainer.params_vector().numpy()
    np.testing.assert_array_equal(vec1, vec2)


def test_checkpoint_engine_mismatch_rejected(tmp_path):
    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    module = feedforward_hourglass(10, 10, encoding_layers=2, compression_factor=0.5)
    X = np.zeros((256, 10), dtype=np.float32)
    dp = DPContext(device=torch.device("cpu"))
    trainer = HourglassStepTrainer(module, X, 128, TrainConfig(batch_size=128, dtype="fp32"), dp=dp)
    with pytest.raises(ValueError, match="mismatch"):
        trainer.load_state_dict({"fused": True})
