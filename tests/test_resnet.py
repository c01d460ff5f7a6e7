"""ResNet-18 app tests (BASELINE config 4 family) — CPU-sized configs;
the full 224×224 perf path is benchmarks/bench_resnet.py on GPU."""

import numpy as np
import pytest
import torch

TINY = {"num_classes": 4, "width": 8}


def tiny_reader_kwargs():
    return dict(n=24, image_size=32, num_classes=4, seed=1)


def test_resnet18_forward_shapes():
    from unionml_amd.models.resnet import ResNet18

    net = ResNet18(**TINY)
    out = net(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, 4)
    # default config is the real ResNet-18: 11.7M params at 1000 classes
    full = ResNet18(num_classes=1000)
    n_params = sum(p.numel() for p in full.parameters())
    assert 11.0e6 < n_params < 12.5e6, n_params


def test_resnet_app_train_predict_save_load(tmp_path):
    from unionml_amd.models.resnet import model

    model.artifact = None
    net, metrics = model.train(
        hyperparameters=TINY,
        trainer_kwargs={"epochs": 1, "batch_size": 8, "lr": 1e-3, "amp": False},
        **tiny_reader_kwargs(),
    )
    assert set(metrics) == {"train", "test"}
    imgs = np.random.RandomState(0).rand(3, 3, 32, 32).astype(np.float32)
    preds = model.predict(features=imgs)
    assert preds.shape == (3,)
    assert set(np.unique(preds)) <= set(range(4))

    path = tmp_path / "resnet.pt"
    model.save(path)
    loaded = model._loader(str(path))
    for a, b in zip(net.state_dict().values(), loaded.state_dict().values()):
        assert torch.equal(a.cpu(), b.cpu())


def test_resnet_single_image_feature_loader():
    from unionml_amd.models.resnet import dataset

    one = np.zeros((3, 32, 32), dtype=np.float32)
    out = dataset.get_features(one)
    assert out.shape == (1, 3, 32, 32)


@pytest.mark.timeout(300)
def test_resnet_dp2_trains_on_cpu():
    """dp=2 over gloo: spawn 2 ranks, shard rows, bucketed all-reduce;
    must produce a valid artifact (multi-process CPU proof of the
    generic DP path)."""
    from unionml_amd.models.resnet import model

    model.artifact = None
    net, metrics = model.train(
        dp=2,
        hyperparameters=TINY,
        trainer_kwargs={"epochs": 1, "batch_size": 8, "lr": 1e-3, "amp": False},
        **tiny_reader_kwargs(),
    )
    assert set(metrics) == {"train", "test"}
    assert sum(p.numel() for p in net.parameters()) > 0


def test_module_graph_runner_cpu_fallback():
    """On CPU the ModuleGraphRunner runs a plain no-grad forward and must
    match the predictor exactly."""
    from unionml_amd.models.resnet import ResNet18
    from unionml_amd.serving.graph_runner import ModuleGraphRunner

    torch.manual_seed(3)
    net = ResNet18(**TINY)
    runner = ModuleGraphRunner(net, max_batch_size=8, postprocess="argmax")
    x = np.random.RandomState(1).rand(5, 3, 32, 32).astype(np.float32)
    out = runner(x)
    with torch.no_grad():
        direct = net.eval()(torch.from_numpy(x)).argmax(dim=1).numpy()
    assert (out == direct).all()

    logits_runner = ModuleGraphRunner(net, max_batch_size=8, postprocess=None)
    logits = logits_runner(x)
    with torch.no_grad():
        ref = net.eval()(torch.from_numpy(x)).numpy()
    assert np.allclose(logits, ref, atol=1e-6)
