"""CPU tests of the tabular engine (reference path) and the flagship
digits app end-to-end."""

import numpy as np
import pytest
import torch

from unionml_amd.ops import reference as ref
from unionml_amd.ops.tabular import TabularMLP


def test_standardize_reference():
    X = torch.randn(500, 64) * 3 + 1
    mean, invstd = ref.standardize_fit(X)
    Xs = ref.standardize_apply(X, mean, invstd).float()
    assert abs(Xs.mean().item()) < 0.01
    assert abs(Xs.std().item() - 1.0) < 0.05


def test_reference_grads_match_autograd():
    """The fused-step reference must equal a plain autograd MLP (fp32,
    modulo the bf16 round-trips the kernel's LDS staging implies)."""
    torch.manual_seed(0)
    B = 64
    X = torch.randn(B, 64).bfloat16()
    y = torch.randint(0, 10, (B,), dtype=torch.int32)

    clf = TabularMLP(device="cpu", seed=3)
    grads = torch.zeros(ref.NPARAM + 1)
    ref.mlp_step(X, y, clf.W1bf, clf.W2bf, clf.master, grads, 1.0 / B)

    # autograd oracle on the same bf16-rounded weights
    W1 = clf.W1bf.float().requires_grad_()
    b1 = clf.master[ref.OFF_B1 : ref.OFF_B1 + 32].clone().requires_grad_()
    W2 = clf.W2bf.float().requires_grad_()
    b2 = clf.master[ref.OFF_B2 : ref.OFF_B2 + 16].clone().requires_grad_()
    H = torch.relu(X.float() @ W1 + b1)
    logits = (H @ W2 + b2)[:, :10]
    loss = torch.nn.functional.cross_entropy(logits, y.long())
    loss.backward()

    assert abs(grads[ref.NPARAM].item() - loss.item()) < 1e-3
    dW1 = grads[ref.OFF_W1 : ref.OFF_W1 + 64 * 32].view(64, 32)
    # bf16 staging of intermediates costs ~1e-2 relative
    assert torch.allclose(dW1, W1.grad, rtol=5e-2, atol=5e-4), (
        (dW1 - W1.grad).abs().max()
    )
    db2 = grads[ref.OFF_B2 : ref.OFF_B2 + 16][:10]
    assert torch.allclose(db2, b2.grad[:10], rtol=5e-2, atol=5e-4)


def test_adam_reference_matches_torch_adam():
    torch.manual_seed(1)
    clf = TabularMLP(device="cpu", seed=1)
    g = torch.randn(ref.NPARAM + 1) * 0.01
    p_torch = clf.master.clone().requires_grad_()
    opt = torch.optim.Adam([p_torch], lr=1e-3, betas=(0.9, 0.999), eps=1e-8)
    m = torch.zeros(ref.NPARAM)
    v = torch.zeros(ref.NPARAM)
    master = clf.master.clone()
    mirror = master.bfloat16()
    for t in range(1, 4):
        p_torch.grad = g[: ref.NPARAM].clone()
        opt.step()
        ref.adam_step(master, mirror, g, m, v, t, 1e-3)
    assert torch.allclose(master, p_torch.detach(), rtol=1e-5, atol=1e-7)


def test_tabular_mlp_learns_digits_cpu():
    from sklearn.datasets import load_digits

    digits = load_digits()
    X = torch.tensor(digits.data, dtype=torch.float32)
    y = torch.tensor(digits.target, dtype=torch.int32)
    clf = TabularMLP(device="cpu", seed=0)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    loss = clf.train_epochs(Xbf, y, epochs=15, batch_size=512, lr=3e-3)
    preds = clf.predict(X)
    acc = (preds == y).float().mean().item()
    assert acc > 0.85, f"accuracy {acc}, loss {loss}"


def test_mlp_app_end_to_end_cpu():
    from unionml_amd.models.mlp import model

    model.artifact = None
    _, metrics = model.train(
        trainer_kwargs={"epochs": 15, "lr": 3e-3}, n=600
    )
    assert metrics["train"] > 0.7
    frame_preds = model.predict(n=20, synthetic=True)
    assert len(frame_preds) == 20
    feats = [{f"p{i}": float(i % 16) for i in range(64)}]
    preds = model.predict(features=feats)
    assert len(preds) == 1 and 0 <= preds[0] <= 9


def test_mlp_app_save_load_roundtrip(tmp_path):
    from unionml_amd.models.mlp import model

    model.artifact = None
    model.train(trainer_kwargs={"epochs": 5}, n=300)
    path = tmp_path / "digits.pt"
    model.save(str(path))
    feats = [{f"p{i}": float((i * 7) % 16) for i in range(64)}]
    before = model.predict(features=feats)
    model.artifact = None
    model.load(str(path))
    assert model.predict(features=feats) == before


def test_tabular_mlp_pickle_roundtrip():
    """TabularMLP must pickle CPU-portably (backend process boundary)."""
    import pickle

    import torch

    from unionml_amd.ops.tabular import TabularMLP

    clf = TabularMLP(device="cpu", seed=4)
    clf.custom_attr = "kept"
    blob = pickle.dumps(clf)
    back = pickle.loads(blob)
    assert back.custom_attr == "kept"
    for k, v in clf.state_dict().items():
        assert torch.equal(v, back.state_dict()[k]), k
    assert back.slabs is None and back._graph is None


def test_missing_extension_is_loud_on_gpu(monkeypatch):
    """On a machine that claims a GPU, a missing/unloadable HIP
    extension must raise KernelExtensionNotBuilt — the HIP path never
    silently falls back to eager torch there (driver contract: GPU
    tests passing on a silent fallback read as 'native code not
    loaded')."""
    import torch

    import unionml_amd.ops as ops
    from unionml_amd.exceptions import KernelExtensionNotBuilt

    monkeypatch.setattr(ops, "_ext", None)
    monkeypatch.setattr(ops, "_load_error", ImportError("no .so for this arch"))
    monkeypatch.setattr(ops, "_try_load", lambda: None)
    monkeypatch.setattr(torch.cuda, "is_available", lambda: True)

    with pytest.raises(KernelExtensionNotBuilt, match="build_ext"):
        ops.hip_ext()  # required=None resolves to GPU-present -> required

    with pytest.raises(KernelExtensionNotBuilt):
        ops.hip_ext(required=True)

    # explicit CPU-style call still returns None quietly
    assert ops.hip_ext(required=False) is None
