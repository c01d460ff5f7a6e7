"""MNIST-shape app tests: the generalized hot path exercised through
the full decorator API (reference tutorial parity:
/root/reference/docs/source/tutorials/mnist.md). CPU here; GPU-marked
end-to-end at the bottom."""

import numpy as np
import pandas as pd
import pytest
import torch


@pytest.fixture()
def mnist_model():
    from unionml_amd.models.mnist import model

    model.artifact = None
    return model


def test_mnist_train_predict_cpu(mnist_model):
    clf, metrics = mnist_model.train(
        trainer_kwargs={"epochs": 6, "lr": 5e-3}, n=1200
    )
    assert clf.g.in_features == 784 and clf.g.hid == 128
    assert metrics["train"] > 0.9, metrics
    assert metrics["test"] > 0.85, metrics

    # ndarray feature form (the tutorial's gradio-style loader)
    raw = np.random.RandomState(0).rand(3, 784).astype(np.float32) * 255
    preds = mnist_model.predict(features=raw)
    assert len(preds) == 3 and all(0 <= p <= 9 for p in preds)
    # flat single image
    preds1 = mnist_model.predict(features=raw[0])
    assert len(preds1) == 1

    # records form still works through the default loader
    rec = [{f"pixel{i + 1}": float(v) for i, v in enumerate(raw[0])}]
    preds2 = mnist_model.predict(features=rec)
    assert preds2 == preds1


def test_mnist_save_load_roundtrip(mnist_model, tmp_path):
    mnist_model.train(trainer_kwargs={"epochs": 3}, n=600)
    path = tmp_path / "mnist.pt"
    mnist_model.save(path)

    from unionml_amd.models.mnist import model as fresh

    artifact_obj = fresh.load(path)
    assert artifact_obj.g.in_features == 784
    raw = np.random.RandomState(1).rand(2, 784).astype(np.float32)
    preds = fresh.predict(features=raw)
    assert len(preds) == 2


def test_mnist_reader_npz_path(tmp_path):
    from unionml_amd.models.mnist import reader

    X = np.random.RandomState(2).rand(50, 784).astype(np.float32)
    y = np.random.RandomState(3).randint(0, 10, 50)
    path = tmp_path / "data.npz"
    np.savez(path, X=X, y=y)
    frame = reader(n=30, path=str(path))
    assert frame.shape == (30, 785)
    assert list(frame["class"]) == list(y[:30])


@pytest.mark.gpu
def test_mnist_app_end_to_end_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    from unionml_amd.models.mnist import model

    model.artifact = None
    clf, metrics = model.train(trainer_kwargs={"epochs": 15, "lr": 3e-3}, n=4000)
    assert clf.use_hip and not clf.use_spec
    assert metrics["test"] > 0.9, metrics
    raw = np.random.RandomState(0).rand(5, 784).astype(np.float32) * 255
    preds = model.predict(features=raw)
    assert len(preds) == 5


@pytest.mark.gpu
def test_mnist_graphed_serving_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    from unionml_amd.models.mnist import model
    from unionml_amd.serving.graph_runner import TabularGraphRunner

    model.artifact = None
    clf, _ = model.train(trainer_kwargs={"epochs": 5}, n=2000)
    runner = TabularGraphRunner(clf, max_batch_size=64)
    X = np.random.RandomState(4).rand(130, 784).astype(np.float32) * 255
    out = runner(X)
    direct = clf.predict(torch.from_numpy(X)).cpu().numpy()
    assert (out == direct).mean() > 0.99
