"""GPU numerics tests: every CDNA4 HIP kernel vs the plain PyTorch fp32
reference (unionml_amd/ops/reference.py). Asymmetric random operands
throughout (guide §3: symmetric inputs mask transposed layouts)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@pytest.fixture(scope="module")
def ext():
    from unionml_amd.ops import hip_ext

    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    return hip_ext(required=True)  # must be the real extension on a GPU box


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda:0")


def test_standardize_fit_apply(ext, dev):
    from unionml_amd.ops import reference as ref

    torch.manual_seed(0)
    X = (torch.randn(3000, 64) * torch.linspace(0.1, 5, 64) + torch.randn(64)).to(dev)
    mean = torch.empty(64, device=dev)
    invstd = torch.empty(64, device=dev)
    ext.standardize_fit(X, mean, invstd, 1e-5)
    mean_ref, invstd_ref = ref.standardize_fit(X.cpu())
    assert torch.allclose(mean.cpu(), mean_ref, rtol=1e-5, atol=1e-5)
    assert torch.allclose(invstd.cpu(), invstd_ref, rtol=1e-4, atol=1e-5)

    out = torch.empty(3000, 64, dtype=torch.bfloat16, device=dev)
    ext.standardize_apply(X, mean, invstd, out)
    out_ref = ref.standardize_apply(X.cpu(), mean_ref, invstd_ref)
    assert torch.allclose(out.cpu().float(), out_ref.float(), rtol=1e-2, atol=1e-2)


@pytest.mark.parametrize("B", [64, 128, 200, 512, 1000])
def test_mlp_step_vs_reference(ext, dev, B):
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import TabularMLP

    torch.manual_seed(B)
    clf = TabularMLP(device=dev, seed=2)
    Xbf = (torch.randn(B, 64) * 1.3 + 0.2).bfloat16().to(dev)
    y = torch.randint(0, 10, (B,), dtype=torch.int32, device=dev)

    grads = torch.zeros(ref.NPARAM + 1, device=dev)
    ext.mlp_step(Xbf, y, clf.W1bf, clf.W2bf, clf.master, grads, 1.0 / B)
    torch.cuda.synchronize()

    grads_ref = torch.zeros(ref.NPARAM + 1)
    ref.mlp_step(
        Xbf.cpu(), y.cpu(), clf.W1bf.cpu(), clf.W2bf.cpu(), clf.master.cpu(),
        grads_ref, 1.0 / B,
    )
    g, gr = grads.cpu(), grads_ref
    assert abs(g[ref.NPARAM] - gr[ref.NPARAM]) < 2e-3, "loss mismatch"
    scale = gr[: ref.NPARAM].abs().max()
    err = (g[: ref.NPARAM] - gr[: ref.NPARAM]).abs().max()
    assert err < max(2e-2 * float(scale), 2e-4), f"grad err {err} vs scale {scale}"


def test_mlp_predict_vs_reference(ext, dev):
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import TabularMLP

    torch.manual_seed(7)
    clf = TabularMLP(device=dev, seed=5)
    X = (torch.rand(777, 64) * 16).to(dev)
    ext.standardize_fit(X, clf.mean, clf.invstd, 1e-5)

    preds = torch.empty(777, dtype=torch.int32, device=dev)
    probs = torch.empty(777, 10, device=dev)
    ext.mlp_predict(X, clf.mean, clf.invstd, clf.W1bf, clf.W2bf, clf.master, preds, probs)
    torch.cuda.synchronize()

    preds_ref, probs_ref = ref.mlp_predict(
        X.cpu(), clf.mean.cpu(), clf.invstd.cpu(), clf.W1bf.cpu(), clf.W2bf.cpu(),
        clf.master.cpu(), return_probs=True,
    )
    agree = (preds.cpu() == preds_ref).float().mean().item()
    assert agree > 0.99, f"argmax agreement {agree}"
    assert torch.allclose(probs.cpu(), probs_ref, rtol=5e-2, atol=5e-3)


def test_adam_step_vs_reference(ext, dev):
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import TabularMLP

    torch.manual_seed(9)
    clf = TabularMLP(device=dev, seed=9)
    g = (torch.randn(ref.NPARAM + 1) * 0.01).to(dev)

    master_ref = clf.master.cpu().clone()
    mirror_ref = master_ref.bfloat16()
    m_ref = torch.zeros(ref.NPARAM)
    v_ref = torch.zeros(ref.NPARAM)
    for t in range(1, 4):
        ext.adam_step(clf.master, clf.bfmirror, g, clf.m, clf.v, clf.t_dev,
                      1e-3, 0.9, 0.999, 1e-8)
        ref.adam_step(master_ref, mirror_ref, g.cpu(), m_ref, v_ref, t, 1e-3)
    torch.cuda.synchronize()
    assert int(clf.t_dev.item()) == 3
    assert torch.allclose(clf.master.cpu(), master_ref, rtol=1e-4, atol=1e-6)


def test_train_steps_kernel_matches_stepwise(ext, dev):
    """The persistent multi-step kernel must reproduce the per-step
    (mlp_step + adam_step) sequence exactly (same math, same order)."""
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import ADAM_BETA1, ADAM_BETA2, ADAM_EPS, TabularMLP

    torch.manual_seed(11)
    N, B, n_steps = 512, 128, 9
    Xbf = (torch.randn(N, 64) * 1.1).bfloat16().to(dev)
    y = torch.randint(0, 10, (N,), dtype=torch.int32, device=dev)

    # per-step path
    a = TabularMLP(device=dev, seed=4)
    for s in range(n_steps):
        off = (s % (N // B)) * B
        a.grads.zero_()
        ext.mlp_step(Xbf[off : off + B], y[off : off + B], a.W1bf, a.W2bf,
                     a.master, a.grads, 1.0 / B)
        ext.adam_step(a.master, a.bfmirror, a.grads, a.m, a.v, a.t_dev,
                      1e-3, ADAM_BETA1, ADAM_BETA2, ADAM_EPS)
    torch.cuda.synchronize()

    # persistent kernel
    b = TabularMLP(device=dev, seed=4)
    loss_out = b.grads[ref.NPARAM : ref.NPARAM + 1]
    ok = ext.mlp_train_steps(Xbf, y, B, n_steps, b.master, b.bfmirror,
                             b.m, b.v, b.t_dev, loss_out,
                             1e-3, ADAM_BETA1, ADAM_BETA2, ADAM_EPS)
    assert ok
    torch.cuda.synchronize()

    assert int(b.t_dev.item()) == n_steps
    err = (a.master - b.master).abs().max().item()
    assert err < 1e-5, f"master mismatch {err}"
    err_m = (a.m - b.m).abs().max().item()
    assert err_m < 1e-6, f"moment mismatch {err_m}"


def test_step_fused_matches_stepwise(ext, dev):
    """The fully-fused step (slab reduction + Adam in-kernel via the G16
    ticket hand-off) must reproduce the 3-kernel sequence."""
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import ADAM_BETA1, ADAM_BETA2, ADAM_EPS, TabularMLP

    torch.manual_seed(21)
    B, n_steps = 640, 7   # 5 workgroups, odd batch tail of 0
    Xbf = (torch.randn(B, 64) * 1.2).bfloat16().to(dev)
    y = torch.randint(0, 10, (B,), dtype=torch.int32, device=dev)

    a = TabularMLP(device=dev, seed=6)
    for _ in range(n_steps):
        a.grads.zero_()
        ext.mlp_step(Xbf, y, a.W1bf, a.W2bf, a.master, a.grads, 1.0 / B)
        ext.adam_step(a.master, a.bfmirror, a.grads, a.m, a.v, a.t_dev,
                      1e-3, ADAM_BETA1, ADAM_BETA2, ADAM_EPS)
    torch.cuda.synchronize()

    b = TabularMLP(device=dev, seed=6)
    b._ensure_slabs((B + 127) // 128)
    loss_out = b.grads[ref.NPARAM : ref.NPARAM + 1]
    for _ in range(n_steps):
        ok = ext.mlp_step_fused(Xbf, y, b.W1bf, b.W2bf, b.master, b.bfmirror,
                                b.m, b.v, b.t_dev, b.slabs, b.counter, loss_out,
                                1.0 / B, 1e-3, ADAM_BETA1, ADAM_BETA2, ADAM_EPS)
        assert ok
    torch.cuda.synchronize()

    assert int(b.t_dev.item()) == n_steps
    # slab reduction order differs from atomic order -> tiny fp32 drift only
    err = (a.master - b.master).abs().max().item()
    assert err < 1e-4, f"master mismatch {err}"
    a_loss = a.grads[ref.NPARAM].item()
    assert abs(loss_out.item() - a_loss) < 1e-3


def test_train_digits_gpu_accuracy(ext, dev):
    from sklearn.datasets import load_digits

    from unionml_amd.ops.tabular import TabularMLP

    digits = load_digits()
    X = torch.tensor(digits.data, dtype=torch.float32)
    y = torch.tensor(digits.target, dtype=torch.int32)
    clf = TabularMLP(device=dev, seed=0)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    loss = clf.train_epochs(Xbf, y, epochs=30, batch_size=512, lr=3e-3, use_graph=False)
    preds = clf.predict(X)
    acc = (preds.cpu() == y).float().mean().item()
    assert acc > 0.9, f"accuracy {acc}, loss {loss}"


def test_train_with_hipgraph_matches_eager(ext, dev):
    from sklearn.datasets import load_digits

    from unionml_amd.ops.tabular import TabularMLP

    digits = load_digits()
    X = torch.tensor(digits.data, dtype=torch.float32)
    y = torch.tensor(digits.target, dtype=torch.int32)

    results = {}
    for use_graph in (False, True):
        clf = TabularMLP(device=dev, seed=0)
        clf.fit_standardizer(X)
        Xbf = clf.stage(X)
        loss = clf.train_epochs(Xbf, y, epochs=12, batch_size=512, lr=3e-3, use_graph=use_graph)
        acc = (clf.predict(X).cpu() == y).float().mean().item()
        results[use_graph] = (loss, acc)
    # identical kernels, identical data order -> losses must agree closely
    assert abs(results[True][0] - results[False][0]) < 5e-2, results
    assert results[True][1] > 0.85


def test_graphed_serving_runner(ext, dev):
    from sklearn.datasets import load_digits

    from unionml_amd.ops.tabular import TabularMLP
    from unionml_amd.serving.graph_runner import TabularGraphRunner

    digits = load_digits()
    X = torch.tensor(digits.data, dtype=torch.float32)
    y = torch.tensor(digits.target, dtype=torch.int32)
    clf = TabularMLP(device=dev, seed=0)
    clf.fit_standardizer(X)
    Xbf = clf.stage(X)
    clf.train_epochs(Xbf, y, epochs=10, batch_size=512, lr=3e-3, use_graph=False)

    runner = TabularGraphRunner(clf, max_batch_size=64)
    for n in (1, 3, 17, 64, 130):
        out = runner(X[:n].numpy())
        direct = clf.predict(X[:n]).cpu().numpy()
        assert (out == direct).mean() > 0.99, f"bucketed replay mismatch at n={n}"


def test_mlp_app_end_to_end_gpu(ext, dev):
    from unionml_amd.models.mlp import model

    model.artifact = None
    _, metrics = model.train(trainer_kwargs={"epochs": 20, "lr": 3e-3})
    assert metrics["test"] > 0.85, metrics
    feats = [{f"p{i}": float(i % 16) for i in range(64)}]
    preds = model.predict(features=feats)
    assert len(preds) == 1


def test_step_fused_reduce_only_matches_mlp_step(ext, dev):
    """grads_out mode: same summed grads as the atomic-accumulation
    mlp_step kernel, with Adam state untouched."""
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import TabularMLP

    torch.manual_seed(31)
    B = 384  # 3 workgroups
    Xbf = (torch.randn(B, 64) * 1.3 - 0.1).bfloat16().to(dev)
    y = torch.randint(0, 10, (B,), dtype=torch.int32, device=dev)

    a = TabularMLP(device=dev, seed=8)
    a.grads.zero_()
    ext.mlp_step(Xbf, y, a.W1bf, a.W2bf, a.master, a.grads, 1.0 / B)

    b = TabularMLP(device=dev, seed=8)
    b._ensure_slabs((B + 127) // 128)
    master0 = b.master.clone()
    loss_out = torch.zeros(1, device=dev)
    for _ in range(2):  # twice: launch-epoch counter must advance per call
        ok = ext.mlp_step_fused(
            Xbf, y, b.W1bf, b.W2bf, b.master, b.bfmirror, b.m, b.v, b.t_dev,
            b.slabs, b.counter, loss_out, 1.0 / B, 1e-3, 0.9, 0.999, 1e-8,
            grads_out=b.grads,
        )
        assert ok
    torch.cuda.synchronize()

    # Adam state untouched in reduce-only mode
    assert int(b.t_dev.item()) == 0
    assert torch.equal(b.master, master0)
    assert b.m.abs().max().item() == 0.0
    # summed grads (incl. loss at NPARAM) match the atomic path
    err = (a.grads[: ref.NPARAM + 1] - b.grads[: ref.NPARAM + 1]).abs().max().item()
    scale = a.grads.abs().max().item()
    assert err < max(1e-5 * scale, 1e-6), (err, scale)


def test_wimg_path_matches_plain(ext, dev):
    """The packed weight-image path (wimg=) must train identically to
    the transposed-gather path, across fused and stepwise flows."""
    from unionml_amd.ops import reference as ref
    from unionml_amd.ops.tabular import ADAM_BETA1, ADAM_BETA2, ADAM_EPS, TabularMLP

    torch.manual_seed(41)
    B, n_steps = 512, 9
    Xbf = (torch.randn(B, 64) * 1.2 + 0.1).bfloat16().to(dev)
    y = torch.randint(0, 10, (B,), dtype=torch.int32, device=dev)

    def train(use_wimg):
        c = TabularMLP(device=dev, seed=12)
        c._ensure_slabs((B + 127) // 128)
        loss_out = c.grads[ref.NPARAM : ref.NPARAM + 1]
        for _ in range(n_steps):
            ok = ext.mlp_step_fused(
                Xbf, y, c.W1bf, c.W2bf, c.master, c.bfmirror, c.m, c.v,
                c.t_dev, c.slabs, c.counter, loss_out, 1.0 / B,
                1e-3, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                wimg=c.wimg if use_wimg else None,
            )
            assert ok
        torch.cuda.synchronize()
        return c

    a, b = train(False), train(True)
    err = (a.master - b.master).abs().max().item()
    assert err < 1e-5, f"wimg path diverged: {err}"
    # wimg stays in sync with bfmirror after kernel updates
    rebuilt = TabularMLP(device=dev, seed=12)
    rebuilt.master.copy_(b.master)
    rebuilt.bfmirror.copy_(b.master.bfloat16())
    rebuilt._build_wimg()
    assert torch.equal(rebuilt.wimg, b.wimg), "wimg out of sync with master"
