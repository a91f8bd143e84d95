# -*- coding: utf-8 -*-
"""Stoke facade CPU-path tests: accumulation semantics, loss tracking, IO."""

import os

import pytest
import torch
import torch.nn as nn

from stoke import ClipGradConfig, ClipGradNormConfig, Stoke, StokeOptimizer
from stoke import io_ops


def tiny_model():
    torch.manual_seed(0)
    return nn.Sequential(nn.Linear(8, 16), nn.Tanh(), nn.Linear(16, 4))


def mk_stoke(grad_accum=1, grad_clip=None, loss=None, **kw):
    return Stoke(
        model=tiny_model(),
        optimizer=StokeOptimizer(
            optimizer=torch.optim.SGD, optimizer_kwargs={"lr": 0.05}
        ),
        loss=loss or nn.CrossEntropyLoss(),
        batch_size_per_device=4,
        grad_accum_steps=grad_accum,
        grad_clip=grad_clip,
        verbose=False,
        **kw,
    )


def data(n=4):
    torch.manual_seed(1)
    return torch.randn(n, 8), torch.randint(0, 4, (n,))


def run_steps(s, n):
    x, y = data()
    for _ in range(n):
        out = s.model(x)
        loss = s.loss(out, y)
        s.backward(loss)
        s.step()


@pytest.mark.parametrize("accum,backwards,expected_steps", [
    (1, 4, 4),      # step fires on every backward
    (2, 4, 2),      # fires on the 2nd and 4th backward
    (4, 8, 2),      # fires on the 4th and 8th
    (4, 5, 1),      # counter resets after step; 5th is a fresh accum window
])
def test_grad_accum_counter_semantics(accum, backwards, expected_steps):
    """Table-driven check of the reference modulo (stoke.py:326-344)."""
    s = mk_stoke(grad_accum=accum)
    run_steps(s, backwards)
    assert s._optimizer_steps == expected_steps
    assert s._backward_steps == backwards


def test_pre_accum_check():
    s = mk_stoke(grad_accum=4)
    x, y = data()
    fired_pre = []
    for i in range(4):
        out = s.model(x)
        loss = s.loss(out, y)
        fired_pre.append(s._check_pre_accum())
        s.backward(loss)
        s.step()
    # pre-accum (checked BEFORE backward) is true exactly on the iteration
    # whose backward will trigger the step — the pre-backward print hook point
    assert fired_pre == [False, False, False, True]


def test_loss_scaled_by_grad_accum():
    s = mk_stoke(grad_accum=2)
    x, y = data()
    out = s.model(x)
    raw = nn.CrossEntropyLoss()(out, y)
    scaled = s.loss(out, y)
    assert torch.allclose(scaled * 2, raw, rtol=1e-6)
    # but tracked losses are unscaled
    assert abs(s.step_loss - raw.item()) < 1e-6


def test_training_converges():
    s = Stoke(
        model=tiny_model(),
        optimizer=StokeOptimizer(
            optimizer=torch.optim.Adam, optimizer_kwargs={"lr": 0.02}
        ),
        loss=nn.CrossEntropyLoss(),
        batch_size_per_device=4,
        verbose=False,
    )
    x, y = data(32)
    first = None
    for _ in range(100):
        out = s.model(x)
        loss = s.loss(out, y)
        if first is None:
            first = loss.item()
        s.backward(loss)
        s.step()
    assert s.step_loss < first * 0.5


def test_ema_loss_math():
    s = mk_stoke()
    # First value seeds the EMA; after that ema = w*v + (1-w)*prev  (w=0.1)
    s._handle_ema_loss(2.0)
    assert s.ema_loss == 2.0
    s._handle_ema_loss(1.0)
    assert abs(s.ema_loss - (0.1 * 1.0 + 0.9 * 2.0)) < 1e-9
    s.reset_ema()
    assert s.ema_loss == 0.0 and s._rolling_loss_steps == 0


def test_multi_loss():
    losses = [nn.CrossEntropyLoss(), nn.CrossEntropyLoss(label_smoothing=0.1)]
    s = mk_stoke(loss=losses)
    x, y = data()
    out = s.model(x)
    lvals = s.loss(out, y)
    assert isinstance(lvals, list) and len(lvals) == 2
    s.backward(lvals)
    s.step()
    assert s._optimizer_steps == 1
    assert isinstance(s.step_loss, list)
    assert isinstance(s.ema_loss, list)


def test_agg_loss_and_reset():
    s = mk_stoke(grad_accum=2)
    x, y = data()
    out = s.model(x)
    l1 = s.loss(out, y)
    s.backward(l1)
    s.step()  # not a boundary: agg keeps accumulating
    assert s._agg_loss > 0
    out = s.model(x)
    l2 = s.loss(out, y)
    s.backward(l2)
    s.step()  # boundary: reset
    assert s._agg_loss == 0.0 and s._grad_accum_counter == 0


def test_grad_clip_value():
    s = mk_stoke(grad_clip=ClipGradConfig(clip_value=1e-4))
    x, y = data()
    out = s.model(x)
    loss = s.loss(out, y)
    s.backward(loss)
    # capture grads before step clears them
    s._runner.clip_grad(s.grad_clip, s.model_access, s.optimizer)
    for p in s.model_access.parameters():
        assert p.grad.abs().max() <= 1e-4 + 1e-9


def test_grad_clip_norm():
    s = mk_stoke(grad_clip=ClipGradNormConfig(max_norm=1e-3, norm_type=2.0))
    x, y = data()
    out = s.model(x)
    loss = s.loss(out, y)
    s.backward(loss)
    s._runner.clip_grad(s.grad_clip, s.model_access, s.optimizer)
    total = torch.sqrt(
        sum(p.grad.pow(2).sum() for p in s.model_access.parameters())
    )
    assert total <= 1e-3 * 1.01


def test_tag_format():
    assert io_ops.make_tag("abc", 17) == "stoke-abc-backward-step-17"
    assert (
        io_ops.make_full_save_path("/tmp/x", "abc", 17, "pt")
        == "/tmp/x/stoke-abc-backward-step-17.pt"
    )


def test_save_load_roundtrip(tmp_path):
    s = mk_stoke()
    run_steps(s, 3)
    path, tag = s.save(str(tmp_path), name="ckpt", extras={"epoch": 7})
    assert tag == "stoke-ckpt-backward-step-3.pt"
    payload = torch.load(os.path.join(path, tag), weights_only=False)
    for key in [
        "backward_step", "grad_accum_step", "optimizer_step", "stoke_status",
        "model_state_dict", "optimizer_state_dict", "scaler_state_dict", "extras",
    ]:
        assert key in payload
    assert payload["backward_step"] == 3
    # Fresh instance loads and resumes counters + weights
    s2 = mk_stoke()
    extras = s2.load(path, tag)
    assert extras == {"epoch": 7}
    assert s2._backward_steps == 3 and s2._optimizer_steps == 3
    for p1, p2 in zip(s.model_access.parameters(), s2.model_access.parameters()):
        assert torch.equal(p1, p2)
    # Training continues identically after resume
    x, y = data()
    out1, out2 = s.model(x), s2.model(x)
    assert torch.equal(out1, out2)


def test_type_checks():
    with pytest.raises(TypeError):
        Stoke(model="nope", optimizer={}, loss=nn.MSELoss(),
              batch_size_per_device=1)
    with pytest.raises(TypeError):
        Stoke(model=tiny_model(), optimizer="nope", loss=nn.MSELoss(),
              batch_size_per_device=1)
    with pytest.raises(TypeError):
        Stoke(
            model=tiny_model(),
            optimizer=StokeOptimizer(optimizer=torch.optim.SGD,
                                     optimizer_kwargs={"lr": 0.1}),
            loss="nope",
            batch_size_per_device=1,
        )


def test_dataloader_shim():
    s = mk_stoke()
    ds = torch.utils.data.TensorDataset(torch.randn(32, 8),
                                        torch.randint(0, 4, (32,)))
    dl = s.DataLoader(ds, shuffle=True)
    batches = list(dl)
    assert len(batches) == 8  # 32 / batch_size 4
    assert batches[0][0].shape == (4, 8)


def test_properties_surface():
    s = mk_stoke()
    assert s.rank == "cpu" and s.world_size == 1
    assert s.num_model_parameters == sum(
        p.numel() for p in s.model_access.parameters()
    )
    assert s.scaler is None and s.fp16 is None
    assert s.is_ddp is False and s.oss is False
    assert s.batch_size == 4
    s.print_num_model_parameters()
    s.dump_model_parameter_info()
    s.print_ema_loss()


def test_flops_profiler_counts():
    import torch as _t

    from benchmarks.models import resnet18
    from stoke.utils import FlopsProfiler

    m = resnet18(num_classes=10, small_input=True)
    fp = FlopsProfiler(m)
    fp.start_profile()
    m(_t.randn(2, 3, 32, 32))
    fp.stop_profile()
    # ResNet-18 CIFAR-shape is ~0.56 GMACs/sample forward
    per_sample = fp.get_total_flops() / 2
    assert 0.8e9 < per_sample < 1.5e9
    assert abs(fp.get_total_params() - sum(p.numel() for p in m.parameters())) < 1e4
    text = fp.print_model_profile(detailed=True)
    assert "GFLOPs" in text


def test_flops_profiler_per_step_reset():
    """ADVICE round-1 fix: with profile_step > 1 the printed profile covers
    exactly ONE optimizer step's forwards, not everything since init."""
    import torch.nn as nn
    from stoke.utils import FlopsProfiler

    model = nn.Linear(8, 4)
    fp = FlopsProfiler(model)
    fp.start_profile()
    x = torch.randn(2, 8)
    model(x)
    model(x)
    inflated = fp.get_total_flops()
    fp.reset_flops()           # the facade calls this at step boundary
    model(x)
    per_step = fp.get_total_flops()
    fp.stop_profile()
    assert inflated == 2 * per_step
    assert per_step == 2.0 * 2 * 4 * 8  # one forward: 2*N*out*in
