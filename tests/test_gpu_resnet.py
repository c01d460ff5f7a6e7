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
