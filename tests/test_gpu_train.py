"""GPU training-path tests: checkpoint snapshotter (pinned side-stream),
swiglu in model, train.run entrypoint resume."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_snapshot_resume_gpu(tmp_path):
    from skypilot_amd.checkpoint.snapshotter import Snapshotter
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-smoke", micro_batch=2, seq_len=256,
                      seed=3)
    tr = Trainer(cfg)
    for _ in range(3):
        tr.train_step()
    ref_master0 = tr.opt.master[0].detach().cpu().clone()  # state @ step 3
    snap = Snapshotter(tr, str(tmp_path))
    snap.snapshot_async()
    # keep training while the D2H copies drain on the side stream
    tr.train_step()
    snap.commit(blocking=True)

    tr2 = Trainer(cfg)
    snap2 = Snapshotter(tr2, str(tmp_path))
    assert snap2.try_resume() == 3
    # master state must match the state at snapshot time (step 3), not 4.
    assert tr2.opt.step_count == 3
    assert torch.equal(tr2.opt.master[0].cpu(), ref_master0)


def test_train_loss_decreases_with_all_kernels():
    from skypilot_amd.train.trainer import TrainConfig, Trainer
    cfg = TrainConfig(model="llama-smoke", micro_batch=2, seq_len=512)
    tr = Trainer(cfg)
    tok, tgt = tr.synthetic_batch()
    losses = [tr.train_step((tok, tgt)) for _ in range(10)]
    assert losses[-1] < losses[0], losses  # monotone-ish decrease on random data
