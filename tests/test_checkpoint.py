"""Checkpoint snapshotter tests (CPU path; GPU side-stream path covered
by test_gpu_ops/test_gpu_train)."""
import torch

from skypilot_amd.checkpoint.snapshotter import Snapshotter
from skypilot_amd.train.trainer import TrainConfig, Trainer


def _trainer():
    cfg = TrainConfig(model="llama-debug", micro_batch=1, seq_len=64,
                      device="cpu", seed=7)
    return Trainer(cfg)


def test_snapshot_and_resume(tmp_path):
    tr = _trainer()
    for _ in range(3):
        tr.train_step()
    snap = Snapshotter(tr, str(tmp_path))
    snap.save(blocking=True)
    ref_master = [m.clone() for m in tr.opt.master]
    ref_params = [p.detach().clone() for p in tr.opt.params]

    tr2 = _trainer()
    snap2 = Snapshotter(tr2, str(tmp_path))
    resumed = snap2.try_resume()
    assert resumed == 3
    assert tr2.opt.step_count == tr.opt.step_count
    for a, b in zip(tr2.opt.master, ref_master):
        assert torch.equal(a, b)
    for a, b in zip(tr2.opt.params, ref_params):
        assert torch.equal(a.detach(), b)
    # Training continues from the restored state.
    tr2.train_step()
    assert tr2.step_count == 4


def test_resume_absent_returns_none(tmp_path):
    tr = _trainer()
    snap = Snapshotter(tr, str(tmp_path))
    assert snap.try_resume() is None


def test_async_snapshot_overlap(tmp_path):
    """snapshot_async + later commit must capture the state at snapshot
    time even if training continues in between (CPU: copies are sync,
    but the API contract is exercised)."""
    tr = _trainer()
    tr.train_step()
    snap = Snapshotter(tr, str(tmp_path))
    snap.snapshot_async()
    step_at_snap = tr.opt.step_count
    tr.train_step()  # mutates optimizer state after the snapshot
    snap.commit(blocking=True)
    tr2 = _trainer()
    snap2 = Snapshotter(tr2, str(tmp_path))
    assert snap2.try_resume() == step_at_snap
