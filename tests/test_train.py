"""Train-facade tests: fit/evaluate contract, callbacks, autolog."""
import torch

from ddlw_amd.core import tracking
from ddlw_amd.models import build_small_cnn
from ddlw_amd.train import (
    EarlyStopping,
    History,
    LearningRateWarmupCallback,
    Model,
    ModelCheckpoint,
    ReduceLROnPlateau,
    autolog,
)


def _toy_data(n=32, bs=8, num_classes=3, seed=0):
    g = torch.Generator().manual_seed(seed)
    xs = torch.randn(n, 3, 16, 16, generator=g)
    ys = torch.randint(0, num_classes, (n,), generator=g)
    return [(xs[i : i + bs], ys[i : i + bs]) for i in range(0, n, bs)]


def test_fit_history_contract(ddlw_home):
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("Adam", learning_rate=1e-3)
    hist = m.fit(_toy_data(), epochs=2, validation_data=_toy_data(seed=1), verbose=0)
    assert isinstance(hist, History)
    for key in ("loss", "accuracy", "val_loss", "val_accuracy"):
        assert len(hist.history[key]) == 2


def test_evaluate_returns_loss_and_metrics(ddlw_home):
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.1)
    out = m.evaluate(_toy_data(), steps=2)
    assert len(out) == 2  # [loss, accuracy]
    d = m.evaluate(_toy_data(), steps=2, return_dict=True)
    assert set(d) == {"loss", "accuracy"}


def test_early_stopping_stops(ddlw_home):
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.0)
    # lr=0 -> no real improvement. min_delta must dominate the tiny val_loss
    # drift from BatchNorm running-stat updates (which continue at lr=0 and
    # can micro-"improve" for many epochs — was a rare flake at 1e-9).
    hist = m.fit(
        _toy_data(),
        epochs=10,
        validation_data=_toy_data(seed=1),
        callbacks=[EarlyStopping(monitor="val_loss", min_delta=0.5, patience=1)],
        verbose=0,
    )
    assert len(hist.history["loss"]) < 10


def test_reduce_lr_on_plateau(ddlw_home):
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=1.0)
    cb = ReduceLROnPlateau(monitor="val_loss", factor=0.5, patience=0)
    cb.set_model(m)
    cb.on_epoch_end(0, {"val_loss": 1.0})
    cb.on_epoch_end(1, {"val_loss": 1.0})  # plateau -> reduce
    assert m.optimizer.param_groups[0]["lr"] == 0.5
    cb.on_epoch_end(2, {"val_loss": 0.5})  # improvement -> no change
    assert m.optimizer.param_groups[0]["lr"] == 0.5


def test_lr_warmup_ramps(ddlw_home):
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.8)
    lrs = []

    class Probe(LearningRateWarmupCallback):
        def on_batch_begin(self, batch, logs=None):
            super().on_batch_begin(batch, logs)
            lrs.append(self.model.optimizer.param_groups[0]["lr"])

    m.fit(_toy_data(), steps_per_epoch=4, epochs=2, callbacks=[Probe(warmup_epochs=2)], verbose=0)
    assert lrs[-1] >= lrs[0]
    assert abs(lrs[-1] - 0.8) < 1e-6


def test_checkpoint_naming(ddlw_home, tmp_path):
    ckpt = tmp_path / "trial" / "checkpoint-{epoch}.ckpt"
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.1)
    m.fit(_toy_data(), epochs=2, callbacks=[ModelCheckpoint(str(ckpt))], verbose=0)
    # reference layout: checkpoint-{epoch}.ckpt, weights only (P2/02:206-211)
    assert (tmp_path / "trial" / "checkpoint-1.ckpt").exists()
    assert (tmp_path / "trial" / "checkpoint-2.ckpt").exists()
    sd = torch.load(tmp_path / "trial" / "checkpoint-2.ckpt", weights_only=True)
    assert "classifier.weight" in sd


def test_timeline_trace(ddlw_home, tmp_path, monkeypatch):
    """DDLW_TIMELINE -> chrome-trace JSON with step/data spans (the Horovod
    Timeline equivalent)."""
    import json

    path = tmp_path / "timeline.json"
    monkeypatch.setenv("DDLW_TIMELINE", str(path))
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.1)
    m.fit(_toy_data(), epochs=1, verbose=0)
    trace = json.loads(path.read_text())
    names = [e["name"] for e in trace["traceEvents"]]
    assert any(n.startswith("train_step") for n in names)
    assert any(n.startswith("data") for n in names)


def test_autolog(ddlw_home):
    tracking.set_experiment("autolog")
    autolog(True)
    try:
        with tracking.start_run() as run:
            m = Model(build_small_cnn(16, 16, num_classes=3)).compile("Adam", learning_rate=1e-3)
            m.fit(_toy_data(), epochs=1, verbose=0)
        r = tracking.get_run(run.run_id)
        assert "loss" in r.metrics()
        assert r.params()["optimizer_name"] == "Adam"
        assert (r.dir / "artifacts" / "model" / "state_dict.pt").exists()
    finally:
        autolog(False)


def test_checkpoint_resume(ddlw_home, tmp_path):
    """resume from a checkpoint-{epoch}.ckpt file and continue
    training (the broadcast callback covers rank parity in the DP case)."""
    m1 = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.05)
    ckpt = tmp_path / "ck" / "checkpoint-{epoch}.ckpt"
    m1.fit(_toy_data(), epochs=2, callbacks=[ModelCheckpoint(str(ckpt))], verbose=0)

    # fresh model resumes from epoch-2 weights
    m2 = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.05)
    sd = torch.load(tmp_path / "ck" / "checkpoint-2.ckpt", weights_only=True)
    m2.module.load_state_dict(sd)
    for (k1, v1), (k2, v2) in zip(m1.module.state_dict().items(), m2.module.state_dict().items()):
        assert torch.allclose(v1, v2), k1
    hist = m2.fit(_toy_data(), epochs=1, verbose=0)
    assert len(hist.history["loss"]) == 1


def test_fused_optimizer_desc_invalidated_on_state_load():
    """load_state_dict replaces state tensors; the chunk-descriptor cache
    must be invalidated or the kernel would write through stale pointers."""
    import torch

    from ddlw_amd.ops.optim import FusedAdam, FusedSGD

    for cls in (FusedSGD, FusedAdam):
        p = torch.nn.Parameter(torch.randn(32))
        opt = cls([p])
        p.grad = torch.randn(32)
        opt.step()
        opt._desc[0] = (("sentinel",), None, 0)  # simulate a built cache
        opt.load_state_dict(opt.state_dict())
        assert opt._desc == {}, cls.__name__


def test_fused_optimizer_fresh_instance_resume_bf16():
    """resuming a fused optimizer on a FRESH instance must
    (a) keep the fp32 master/momentum/m/v state fp32 for bf16 params (torch's
    default load casts them to the param dtype) and (b) restore Adam's step
    count so bias correction doesn't restart at t=1. Checked by comparing a
    save/load-interrupted run against an uninterrupted one."""
    import torch

    from ddlw_amd.ops.optim import FusedAdam, FusedSGD

    def make(cls, seed):
        torch.manual_seed(seed)
        p32 = torch.nn.Parameter(torch.randn(17))
        p16 = torch.nn.Parameter(torch.randn(23).to(torch.bfloat16))
        return p32, p16, cls([p32, p16], lr=0.05)

    for cls in (FusedSGD, FusedAdam):
        torch.manual_seed(7)
        grads = [(torch.randn(17), torch.randn(23)) for _ in range(6)]

        # uninterrupted run
        a32, a16, opt_a = make(cls, 0)
        for g32, g16 in grads:
            a32.grad = g32.clone()
            a16.grad = g16.to(torch.bfloat16)
            opt_a.step()

        # interrupted at step 3: save, rebuild fresh, load, continue
        b32, b16, opt_b = make(cls, 0)
        for g32, g16 in grads[:3]:
            b32.grad = g32.clone()
            b16.grad = g16.to(torch.bfloat16)
            opt_b.step()
        sd = opt_b.state_dict()
        c32, c16, opt_c = make(cls, 0)
        with torch.no_grad():
            c32.copy_(b32)
            c16.copy_(b16)
        opt_c.load_state_dict(sd)
        # state dtype preserved (the fused kernels read these as fp32)
        for st in opt_c.state.values():
            for k, v in st.items():
                if isinstance(v, torch.Tensor):
                    assert v.dtype == torch.float32, k
        if cls is FusedAdam:
            assert opt_c._step_t == 3
        for g32, g16 in grads[3:]:
            c32.grad = g32.clone()
            c16.grad = g16.to(torch.bfloat16)
            opt_c.step()

        assert torch.allclose(a32, c32, atol=1e-6), cls.__name__
        assert torch.equal(a16, c16), cls.__name__


def test_fit_rejects_infinite_loader_without_steps(ddlw_home):
    """an infinite stream + steps_per_epoch=None must
    raise, not spin forever."""
    import pyarrow as pa
    import pytest

    from ddlw_amd.data import make_converter, make_synthetic_dataset
    from ddlw_amd.train import Model

    contents, labels = make_synthetic_dataset(8, 16, 16, num_classes=3, jpeg=True)
    tbl = pa.table({"content": pa.array(contents, pa.binary()), "label_idx": labels})
    conv = make_converter(tbl, row_group_rows=8)
    m = Model(build_small_cnn(16, 16, num_classes=3)).compile("SGD", learning_rate=0.05)
    with conv.make_torch_dataset(batch_size=4, num_epochs=None, img_height=16, img_width=16) as ds:
        with pytest.raises(ValueError, match="infinite"):
            m.fit(ds, epochs=1, verbose=0)
    conv.delete()
