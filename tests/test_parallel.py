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


# ---------------------------------------------------------------------------
# DP semantics: dp2 at the same GLOBAL batch == a single-rank run
# ---------------------------------------------------------------------------

def _semantics_worker(rank: int, world: int, port: int, shards, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)

    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    class SequentialTrainer(HourglassStepTrainer):
        # deterministic full-shard batch each step (see _next_indices hook)
        def _next_indices(self):
            import torch as _t

            return _t.arange(len(self.Xt))

    dp = DPContext(backend="gloo", device=torch.device("cpu"))
    torch.manual_seed(7)  # identical init everywhere (matches single-rank run)
    module = feedforward_hourglass(8, 8, encoding_layers=2, compression_factor=0.5)
    X = shards[rank]
    cfg = TrainConfig(batch_size=len(X), lr=1e-3, dtype="fp32")
    trainer = SequentialTrainer(module, X, len(X), cfg, dp=dp)
    for _ in range(4):
        trainer.step()
    dp.barrier()
    q.put((rank, trainer.params_vector().cpu().numpy()))
    import torch.distributed as dist

    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_dp2_equals_single_rank_at_same_global_batch():
    """Two ranks each holding half the global batch, gradients averaged by the
    flat-bucket all-reduce, must follow the same trajectory as ONE rank
    training on the full batch: mean(grad(shard0), grad(shard1)) ==
    grad(shard0 ∪ shard1) for equal shard sizes + MSE mean loss."""
    rng = np.random.default_rng(11)
    X_full = rng.normal(size=(256, 8)).astype(np.float32)
    shards = [X_full[:128], X_full[128:]]

    # --- single-rank run on the full batch, deterministic sequential batches
    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    class SequentialTrainer(HourglassStepTrainer):
        def _next_indices(self):
            return torch.arange(len(self.Xt))

    torch.manual_seed(7)
    module = feedforward_hourglass(8, 8, encoding_layers=2, compression_factor=0.5)
    dp = DPContext(device=torch.device("cpu"))
    trainer = SequentialTrainer(
        module, X_full, len(X_full), TrainConfig(batch_size=len(X_full), lr=1e-3, dtype="fp32"), dp=dp
    )
    for _ in range(4):
        trainer.step()
    single = trainer.params_vector().cpu().numpy()

    # --- dp2 run, same global batch split across ranks
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29551 + os.getpid() % 400
    procs = [
        ctx.Process(target=_semantics_worker, args=(r, 2, port, shards, q))
        for r in range(2)
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

    np.testing.assert_allclose(results[0], results[1], rtol=0, atol=0)
    np.testing.assert_allclose(results[0], single, rtol=1e-5, atol=1e-6)


# ---------------------------------------------------------------------------
# Rank failure: surviving ranks abort via the watchdog timeout (no deadlock),
# and a restarted world resumes bit-exact from the checkpoint
# ---------------------------------------------------------------------------

def _failure_worker(rank: int, world: int, port: int, ckpt_dir: str, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["GORDO_DIST_TIMEOUT"] = "10"  # fast watchdog for the test

    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    dp = DPContext(backend="gloo", device=torch.device("cpu"))
    torch.manual_seed(7)
    module = feedforward_hourglass(8, 8, encoding_layers=2, compression_factor=0.5)
    X = np.random.default_rng(rank).normal(size=(256, 8)).astype(np.float32)
    trainer = HourglassStepTrainer(
        module, X, 64, TrainConfig(batch_size=64, lr=1e-3, dtype="fp32"), dp=dp
    )
    for _ in range(3):
        trainer.step()
    dp.barrier()
    trainer.save_checkpoint(os.path.join(ckpt_dir, f"rank{rank}.pt"))
    dp.barrier()
    if rank == 1:
        os._exit(17)  # simulated rank death mid-training
    try:
        for _ in range(3):
            trainer.step()  # rank 0's next all_reduce has no peer
        q.put(("no-abort", None))
    except RuntimeError as e:
        q.put(("aborted", str(e)[:200]))


def _resume_worker(rank: int, world: int, port: int, ckpt_dir: str, steps: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)

    from gordo_amd.model.factories import feedforward_hourglass
    from gordo_amd.parallel.ddp import DPContext, HourglassStepTrainer
    from gordo_amd.runtime.trainer import TrainConfig

    dp = DPContext(backend="gloo", device=torch.device("cpu"))
    torch.manual_seed(7)
    module = feedforward_hourglass(8, 8, encoding_layers=2, compression_factor=0.5)
    X = np.random.default_rng(rank).normal(size=(256, 8)).astype(np.float32)
    trainer = HourglassStepTrainer(
        module, X, 64, TrainConfig(batch_size=64, lr=1e-3, dtype="fp32"), dp=dp
    )
    ckpt = os.path.join(ckpt_dir, f"rank{rank}.pt")
    if os.path.exists(ckpt):
        trainer.load_checkpoint(ckpt)
    for _ in range(steps):
        trainer.step()
    dp.barrier()
    q.put((rank, trainer.params_vector().cpu().numpy()))
    import torch.distributed as dist

    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_rank_failure_aborts_then_resumes_from_checkpoint(tmp_path):
    """Rank 1 dies after the step-3 checkpoint; rank 0's next collective must
    RAISE within the GORDO_DIST_TIMEOUT watchdog (abort, not deadlock). A
    fresh 2-rank world restarted from the checkpoints then reaches exactly
    the state of an uninterrupted 6-step run (bit-exact resume)."""
    ctx = mp.get_context("spawn")

    # --- phase 1: failure run (checkpoints at step 3, rank 1 dies)
    q = ctx.SimpleQueue()
    port = 29951 + os.getpid() % 400
    procs = [
        ctx.Process(target=_failure_worker, args=(r, 2, port, str(tmp_path), q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    status, _msg = q.get()
    for p in procs:
        p.join(120)
    assert status == "aborted", "surviving rank must raise, not hang"
    assert procs[1].exitcode == 17
    assert (tmp_path / "rank0.pt").exists() and (tmp_path / "rank1.pt").exists()

    # --- phase 2: restart world from checkpoints, run remaining 3 steps
    def run_world(steps, ckpt_dir, port):
        q = ctx.SimpleQueue()
        procs = [
            ctx.Process(
                target=_resume_worker, args=(r, 2, port, ckpt_dir, steps, q)
            )
            for r in range(2)
        ]
        for p in procs:
            p.start()
        out = {}
        for _ in range(2):
            rank, vec = q.get()
            out[rank] = vec
        for p in procs:
            p.join(60)
            assert p.exitcode == 0
        return out

    resumed = run_world(3, str(tmp_path), port + 7)
    # --- uninterrupted 6-step run for comparison
    clean = run_world(6, str(tmp_path / "nonexistent"), port + 13)

    np.testing.assert_array_equal(resumed[0], resumed[1])
    np.testing.assert_array_equal(resumed[0], clean[0])


# ---------------------------------------------------------------------------
# 8-rank launch wiring dry-run: the exact driver command shape
# ---------------------------------------------------------------------------

@pytest.mark.timeout(300)
def test_bench_8rank_launch_dry_run():
    """Rehearses the driver's 8-GPU launch on CPU/gloo: torch.distributed.run
    with --nproc-per-node 8 over bench.py --gpus 8 --dry-run must rendezvous,
    step, reduce max-over-ranks and print one JSON line with n_gpus=8."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    port = 29100 + os.getpid() % 400
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
        "--master-port", str(port), os.path.join(repo, "bench.py"),
        "--gpus", "8", "--steps", "2", "--warmup", "1", "--dry-run",
    ]
    proc = subprocess.run(
        cmd, capture_output=True, text=True, timeout=240, cwd=repo
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, proc.stdout
    out = json.loads(lines[0])
    assert out["n_gpus"] == 8
    assert out["dry_run"] is True
    assert out["config"]["parallelism"] == "dp8"
    assert np.isfinite(out["value"])
