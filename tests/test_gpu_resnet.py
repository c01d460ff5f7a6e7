"""GPU tests for the ResNet-18 DP flagship app (bf16, channels-last,
MIOpen convs; reducer path is covered world-2 on CPU in test_parallel)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def needs_gpu():
    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")


def test_resnet_gpu_train_step_and_predict():
    from unionml_amd.models.resnet import model

    model.artifact = None
    net, metrics = model.train(
        trainer_kwargs={"epochs": 1, "batch_size": 32, "lr": 1e-3},
        n=96,
        image_size=224,
        num_classes=1000,
        seed=3,
    )
    assert next(net.parameters()).is_cuda
    assert np.isfinite(metrics["train"]) and np.isfinite(metrics["test"])
    imgs = np.random.RandomState(0).rand(4, 3, 224, 224).astype(np.float32)
    preds = model.predict(features=imgs)
    assert preds.shape == (4,)


def test_resnet_gpu_bf16_loss_finite():
    import torch.nn.functional as F

    from unionml_amd.models.resnet import ResNet18

    torch.manual_seed(0)
    net = ResNet18(num_classes=100).cuda().to(memory_format=torch.channels_last)
    x = torch.rand(16, 3, 224, 224, device="cuda").to(memory_format=torch.channels_last)
    y = torch.randint(0, 100, (16,), device="cuda")
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loss = F.cross_entropy(net(x), y)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    g = torch.cat([p.grad.reshape(-1) for p in net.parameters() if p.grad is not None])
    assert torch.isfinite(g).all().item()


def test_module_graph_runner_matches_eager_gpu():
    """Bucketed hipGraph replay of the ResNet forward must agree with the
    eager bf16-autocast forward."""
    from unionml_amd.models.resnet import ResNet18
    from unionml_amd.serving.graph_runner import ModuleGraphRunner

    torch.manual_seed(1)
    net = ResNet18(num_classes=50).cuda().to(memory_format=torch.channels_last).eval()
    runner = ModuleGraphRunner(
        net, max_batch_size=16, channels_last=True, postprocess=None
    )
    x = np.random.RandomState(2).rand(21, 3, 224, 224).astype(np.float32)
    out = runner(x)  # buckets 16 + 8 (padded), two replays
    with torch.no_grad(), torch.autocast("cuda", dtype=torch.bfloat16):
        xb = torch.from_numpy(x).cuda().to(memory_format=torch.channels_last)
        ref = net(xb).float().cpu().numpy()
    assert out.shape == ref.shape
    # bf16 forward, same kernels -> near-identical
    assert np.allclose(out, ref, rtol=2e-2, atol=2e-2), np.abs(out - ref).max()

    argmax_runner = ModuleGraphRunner(
        net, max_batch_size=16, channels_last=True, postprocess="argmax"
    )
    # same net, same buckets, deterministic kernels -> argmax of the
    # logits replay and the captured-argmax replay agree
    preds = argmax_runner(x)
    agree = (preds == out.argmax(axis=1)).mean()
    assert agree > 0.95, agree
