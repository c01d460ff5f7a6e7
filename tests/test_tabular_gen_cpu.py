"""CPU tests for the generalized tabular geometry: Geometry math,
parametric reference ops, and TabularMLP at non-digits shapes (the same
class that runs the gen HIP kernels on a GPU runs the parametric torch
reference here — one code path, two substrates)."""

import pickle

import pytest
import torch

from unionml_amd.ops import reference as ref
from unionml_amd.ops.reference import Geometry
from unionml_amd.ops.tabular import TabularMLP


def test_geometry_padding_math():
    g = Geometry(784, 128, 10)
    assert (g.inp, g.hid, g.cpad) == (800, 128, 16)
    assert g.off_b1 == 800 * 128
    assert g.nparam == 800 * 128 + 128 + 128 * 16 + 16
    assert not g.is_specialized

    g2 = Geometry(64, 32, 10)
    assert g2.is_specialized
    assert (g2.off_b1, g2.off_w2, g2.off_b2, g2.nparam) == (2048, 2080, 2592, 2608)

    g3 = Geometry(100, 50, 7)
    assert (g3.inp, g3.hid) == (128, 64)
    g4 = Geometry(33, 250, 3)
    assert (g4.inp, g4.hid) == (64, 256)


def test_geometry_rejects_unsupported():
    with pytest.raises(ValueError, match="classes"):
        Geometry(64, 32, 33)
    with pytest.raises(ValueError, match="hidden"):
        Geometry(64, 300, 10)
    with pytest.raises(ValueError):
        Geometry(0, 32, 10)


def test_geometry_wide_class_head():
    """17..32 classes use the two-tile classifier head (cpad 32)."""
    g = Geometry(64, 32, 26)
    assert g.cpad == 32
    assert g.off_b2 == g.off_w2 + 32 * 32
    assert g.nparam == 64 * 32 + 32 + 32 * 32 + 32
    assert Geometry(64, 32, 16).cpad == 16
    assert Geometry(64, 32, 17).cpad == 32


def test_parametric_matches_legacy_at_digits_shape():
    """mlp_step_g at the digits geometry must agree with the original
    hardcoded reference (the specialized-kernel oracle) closely — the
    only difference is one extra bf16 round of H before layer 2."""
    g = Geometry(64, 32, 10)
    torch.manual_seed(0)
    B = 128
    clf = TabularMLP(device="cpu", seed=3)
    Xbf = (torch.randn(B, 64) * 1.2).bfloat16()
    y = torch.randint(0, 10, (B,), dtype=torch.int32)

    grads_a = torch.zeros(g.nparam + 1)
    ref.mlp_step(Xbf, y, clf.W1bf, clf.W2bf, clf.master, grads_a, 1.0 / B)
    grads_b = torch.zeros(g.nparam + 1)
    ref.mlp_step_g(g, Xbf, y, clf.W1bf, clf.W2bf, clf.master, grads_b, 1.0 / B)

    scale = grads_a[: g.nparam].abs().max()
    err = (grads_a[: g.nparam] - grads_b[: g.nparam]).abs().max()
    assert err < max(2e-2 * float(scale), 2e-4)
    assert abs(grads_a[g.nparam] - grads_b[g.nparam]) < 2e-3


@pytest.mark.parametrize("shape", [(784, 128, 10), (100, 50, 7), (33, 250, 3)],
                         ids=lambda s: "x".join(map(str, s)))
def test_cpu_train_predict_odd_shapes(shape):
    inf, hid, cls = shape
    clf = TabularMLP(in_features=inf, hidden=hid, classes=cls, device="cpu", seed=0)
    g = torch.Generator().manual_seed(1)
    n = 512
    centers = torch.randn(cls, inf, generator=g) * 3.0
    y = torch.randint(0, cls, (n,), generator=g, dtype=torch.int32)
    X = centers[y.long()] + torch.randn(n, inf, generator=g) * 0.5
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    assert Xbf.shape == (n, clf.g.inp)
    if clf.g.inp > inf:
        assert (Xbf[:, inf:] == 0).all(), "staging must zero-pad"
    loss = clf.train_epochs(Xbf, y, epochs=8, batch_size=128, lr=5e-3)
    assert loss == loss  # not NaN
    preds = clf.predict(X)
    acc = (preds == y).float().mean().item()
    assert acc > 0.9, f"{shape}: accuracy {acc}, loss {loss}"
    # padded params stayed exactly zero through training
    W1, b1, W2, b2 = ref.unpack_master_g(clf.g, clf.master)
    assert (W1[inf:, :] == 0).all() and (W1[:, hid:] == 0).all()
    assert (W2[hid:, :] == 0).all() and (W2[:, cls:] == 0).all()
    assert (b1[hid:] == 0).all() and (b2[cls:] == 0).all()


def test_state_dict_roundtrip_and_geometry_guard():
    clf = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu", seed=0)
    X = torch.randn(64, 100)
    y = torch.randint(0, 7, (64,), dtype=torch.int32)
    clf.fit_standardizer(X)
    clf.train_epochs(clf.stage(X), y, epochs=2, batch_size=32, lr=1e-3)

    state = clf.state_dict()
    assert state["W1"].shape == (100, 50)
    assert state["W2"].shape == (50, 7)

    rebuilt = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu", seed=9)
    rebuilt.load_state_dict(state)
    assert torch.equal(clf.predict(X), rebuilt.predict(X))

    wrong = TabularMLP(in_features=64, hidden=32, classes=10, device="cpu")
    with pytest.raises(ValueError, match="geometry"):
        wrong.load_state_dict(state)


def test_pickle_roundtrip_gen_shape():
    clf = TabularMLP(in_features=200, hidden=96, classes=5, device="cpu", seed=0)
    X = torch.randn(32, 200)
    clf.fit_standardizer(X)
    blob = pickle.dumps(clf)
    clone = pickle.loads(blob)
    assert clone.g == clf.g
    assert torch.equal(clone.predict(X), clf.predict(X))


def test_legacy_pickle_without_geometry_loads():
    """Round-1 pickles predate the Geometry field; they must rehydrate
    as the digits shape."""
    clf = TabularMLP(device="cpu", seed=0)
    state = clf.__getstate__()
    state.pop("g")
    state.pop("use_spec")
    clone = TabularMLP.__new__(TabularMLP)
    clone.__setstate__(state)
    assert clone.g.is_specialized
    X = torch.randn(8, 64)
    assert clone.predict(X).shape == (8,)


@pytest.mark.parametrize("shape", [(64, 32, 26), (100, 60, 20)],
                         ids=lambda s: "x".join(map(str, s)))
def test_cpu_train_predict_wide_classes(shape):
    inf, hid, cls = shape
    clf = TabularMLP(in_features=inf, hidden=hid, classes=cls, device="cpu", seed=0)
    assert clf.g.cpad == 32
    g = torch.Generator().manual_seed(2)
    n = 512
    centers = torch.randn(cls, inf, generator=g) * 3.0
    y = torch.randint(0, cls, (n,), generator=g, dtype=torch.int32)
    X = centers[y.long()] + torch.randn(n, inf, generator=g) * 0.5
    clf.fit_standardizer(X)
    loss = clf.train_epochs(clf.stage(X), y, epochs=10, batch_size=128, lr=5e-3)
    assert loss == loss
    preds = clf.predict(X)
    acc = (preds == y).float().mean().item()
    assert acc > 0.9, f"{shape}: accuracy {acc}, loss {loss}"


def test_checkpoint_resumes_training_exactly():
    """Full checkpoints carry Adam state: 2+2 epochs through a
    state_dict roundtrip must be bit-identical to 4 straight epochs
    (same deterministic batch order, same moments, same step count)."""
    torch.manual_seed(11)
    X = torch.randn(128, 100) * 3
    y = torch.randint(0, 7, (128,), dtype=torch.int32)

    a = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu", seed=0)
    a.fit_standardizer(X)
    a.train_epochs(a.stage(X), y, epochs=4, batch_size=32, lr=1e-3)

    b = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu", seed=0)
    b.fit_standardizer(X)
    b.train_epochs(b.stage(X), y, epochs=2, batch_size=32, lr=1e-3)
    state = b.state_dict()

    c = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu", seed=99)
    c.load_state_dict(state)  # restores weights, standardizer, m/v/t
    c.train_epochs(c.stage(X), y, epochs=2, batch_size=32, lr=1e-3)

    assert int(c.t_dev.item()) == int(a.t_dev.item()) == 16
    assert torch.equal(a.master, c.master), (
        (a.master - c.master).abs().max()
    )


def test_weights_only_checkpoint_resets_optimizer():
    """Checkpoints from before optimizer-state support (no adam_* keys)
    still load — with fresh moments and t=0."""
    clf = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu", seed=0)
    X = torch.randn(64, 100)
    y = torch.randint(0, 7, (64,), dtype=torch.int32)
    clf.fit_standardizer(X)
    clf.train_epochs(clf.stage(X), y, epochs=2, batch_size=32, lr=1e-3)
    state = clf.state_dict()
    for k in ("adam_m", "adam_v", "adam_t"):
        state.pop(k)

    fresh = TabularMLP(in_features=100, hidden=50, classes=7, device="cpu")
    fresh.load_state_dict(state)
    assert int(fresh.t_dev.item()) == 0
    assert fresh.m.abs().max() == 0 and fresh.v.abs().max() == 0
    assert torch.equal(fresh.predict(X), clf.predict(X))
