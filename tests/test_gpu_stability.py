"""Sustained-stepping stability tests (GPU): thousands of back-to-back
optimizer steps with exact device-step-counter accounting — the
invariant that held over 11M-step soaks (profiles/r02_gen_kernel_stats.md)
asserted in-suite at a size the round-end runner can afford."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    return "cuda:0"


def _soak(clf, Xbf, y, epochs, batch):
    loss = clf.train_epochs(Xbf, y, epochs=epochs, batch_size=batch, lr=1e-3)
    assert loss == loss, "NaN loss"
    assert torch.isfinite(clf.master).all(), "non-finite params"
    return loss


def test_spec_sustained_steps_exact_counter(dev):
    from unionml_amd.ops.tabular import TabularMLP

    clf = TabularMLP(device=dev, seed=0)
    n, batch, epochs = 4096, 512, 250
    X = torch.rand(n, 64, device=dev) * 16
    y = torch.randint(0, 10, (n,), dtype=torch.int32, device=dev)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    _soak(clf, Xbf, y, epochs, batch)
    expected = epochs * (n // batch)
    assert int(clf.t_dev.item()) == expected, (
        f"step counter drifted: {int(clf.t_dev.item())} != {expected} "
        "(lost/duplicated Adam steps across launches)"
    )


def test_gen_sustained_steps_exact_counter(dev):
    from unionml_amd.ops.tabular import TabularMLP

    clf = TabularMLP(in_features=784, hidden=128, classes=10, device=dev, seed=0)
    n, batch, epochs = 4096, 512, 250
    X = torch.rand(n, 784, device=dev) * 16
    y = torch.randint(0, 10, (n,), dtype=torch.int32, device=dev)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    _soak(clf, Xbf, y, epochs, batch)
    expected = epochs * (n // batch)
    assert int(clf.t_dev.item()) == expected


def test_mixed_engines_counter_consistent(dev):
    """Interleaving the fused-Adam and reduce+Adam modes (the spec
    kernel's shared epoch counter — advisor finding r01) over many
    alternations keeps the counter exact and params finite."""
    from unionml_amd.ops.tabular import ADAM_BETA1, ADAM_BETA2, ADAM_EPS, TabularMLP

    clf = TabularMLP(device=dev, seed=3)
    B = 512
    X = torch.rand(B, 64, device=dev) * 16
    y = torch.randint(0, 10, (B,), dtype=torch.int32, device=dev)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    g = clf.g
    clf._ensure_slabs((B + 127) // 128)
    loss_out = clf.grads[g.nparam : g.nparam + 1]
    for i in range(400):
        if i % 2 == 0:
            clf._fused_adam_step(Xbf, y, 1.0 / B, 1e-3, loss_out)
        else:
            clf._step_reduce(Xbf, y, 1.0 / B, 1e-3)
            clf._adam(1e-3)
    torch.cuda.synchronize()
    assert int(clf.t_dev.item()) == 400
    assert torch.isfinite(clf.master).all()
