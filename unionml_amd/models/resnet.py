"""ResNet-18 app on synthetic 224×224 images — the DP flagship
(BASELINE.md config 4).

Proves the decorator API + DP path generalize past the fused tabular
kernels (SURVEY.md §7 stage 5): the same ``@dataset``/``@model``
surface, a torch ResNet-18 (defined here; PyTorch-ROCm/MIOpen is the
conv substrate), bf16 autocast, channels-last layout for MFMA-friendly
implicit GEMMs, and gradient all-reduce through the framework's own
bucketed RCCL reducer (unionml_amd/parallel/ddp.py) overlapped with
backward.

The model architecture mirrors the capability the reference exercises in
its quickdraw CNN template (templates/quickdraw/.../model.py:39-54) —
a from-scratch torch CNN classifier trained through the decorator API.
"""

from typing import Dict, Tuple

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from unionml_amd import Dataset, Model
from unionml_amd.parallel import maybe_wrap
from unionml_amd.serving.graph_runner import graphed


class BasicBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, stride=1, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(out_ch)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_ch),
            )

    def forward(self, x):
        identity = x if self.down is None else self.down(x)
        out = F.relu(self.bn1(self.conv1(x)), inplace=True)
        out = self.bn2(self.conv2(out))
        return F.relu(out + identity, inplace=True)


class ResNet18(nn.Module):
    """ResNet-18 classifier (64-128-256-512, two BasicBlocks per stage)."""

    def __init__(self, num_classes: int = 1000, in_ch: int = 3, width: int = 64):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(in_ch, width, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(width),
            nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1),
        )
        chans = [width, width * 2, width * 4, width * 8]
        stages = []
        in_c = width
        for i, c in enumerate(chans):
            stride = 1 if i == 0 else 2
            stages += [BasicBlock(in_c, c, stride), BasicBlock(c, c)]
            in_c = c
        self.stages = nn.Sequential(*stages)
        self.head = nn.Linear(chans[-1], num_classes)

    def forward(self, x):
        x = self.stages(self.stem(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.head(x)


# ----------------------------------------------------------------------
# the app
# ----------------------------------------------------------------------

dataset = Dataset(name="synthetic_images", test_size=0.1, shuffle=False)


@dataset.reader
def reader(
    n: int = 512, image_size: int = 224, num_classes: int = 1000, seed: int = 0
) -> Dict[str, np.ndarray]:
    """Synthetic image batch (no network for datasets)."""
    rng = np.random.RandomState(seed)
    return {
        "images": rng.rand(n, 3, image_size, image_size).astype(np.float32),
        "labels": rng.randint(0, num_classes, size=n).astype(np.int64),
    }


@dataset.loader
def loader(data: Dict[str, np.ndarray]) -> Dict[str, np.ndarray]:
    return data


@dataset.splitter
def splitter(
    data: Dict[str, np.ndarray], *, test_size: float, shuffle: bool, random_state: int
) -> Tuple[Dict[str, np.ndarray], Dict[str, np.ndarray]]:
    n = len(data["labels"])
    n_test = max(1, int(round(n * test_size)))
    idx = np.arange(n)
    if shuffle:
        idx = np.random.RandomState(random_state).permutation(n)
    tr, te = idx[n_test:], idx[:n_test]
    pick = lambda i: {k: v[i] for k, v in data.items()}  # noqa: E731
    return pick(tr), pick(te)


@dataset.parser
def parser(
    data: Dict[str, np.ndarray], features, targets
) -> Tuple[np.ndarray, np.ndarray]:
    return data["images"], data["labels"]


@dataset.feature_loader
def feature_loader(features) -> np.ndarray:
    arr = np.asarray(features, dtype=np.float32)
    if arr.ndim == 3:
        arr = arr[None]
    return arr


model = Model(name="resnet18_synth", dataset=dataset)


@model.init
def init(hyperparameters: dict) -> ResNet18:
    return ResNet18(**(hyperparameters or {"num_classes": 1000}))


def _device() -> torch.device:
    if torch.cuda.is_available():
        from unionml_amd.parallel import get_rank

        return torch.device("cuda", get_rank() % torch.cuda.device_count())
    return torch.device("cpu")


@model.trainer
def trainer(
    net: ResNet18,
    images: np.ndarray,
    labels: np.ndarray,
    *,
    epochs: int = 1,
    batch_size: int = 64,
    lr: float = 1e-3,
    amp: bool = True,
) -> ResNet18:
    """bf16-autocast train loop; under DP the framework shards rows per
    rank and this body's reducer all-reduces bucket-flat gradients on
    RCCL, overlapped with backward."""
    device = _device()
    use_amp = amp and device.type == "cuda"
    net = net.to(device)
    if device.type == "cuda":
        net = net.to(memory_format=torch.channels_last)
        torch.backends.cudnn.benchmark = True  # MIOpen conv autotuning
    reducer = maybe_wrap(net)
    opt = torch.optim.Adam(net.parameters(), lr=lr, foreach=True)
    X = torch.from_numpy(np.ascontiguousarray(images))
    y = torch.from_numpy(np.ascontiguousarray(labels))
    net.train()
    for _ in range(epochs):
        for off in range(0, len(y), batch_size):
            xb = X[off : off + batch_size].to(device, non_blocking=True)
            yb = y[off : off + batch_size].to(device, non_blocking=True)
            if device.type == "cuda":
                xb = xb.to(memory_format=torch.channels_last)
            with torch.autocast(device.type, dtype=torch.bfloat16, enabled=use_amp):
                loss = F.cross_entropy(net(xb), yb)
            loss.backward()
            if reducer is not None:
                reducer.finalize()
            opt.step()
            if reducer is not None:
                reducer.zero_grad()
            else:
                opt.zero_grad(set_to_none=True)
    if reducer is not None:
        reducer.detach()
    return net


def _graph_runner(net, max_batch):
    from unionml_amd.serving.graph_runner import ModuleGraphRunner

    return ModuleGraphRunner(
        net, max_batch_size=max_batch, channels_last=True, postprocess="argmax"
    )


@model.predictor
@graphed(_graph_runner)
def predictor(net: ResNet18, images: np.ndarray) -> np.ndarray:
    device = next(net.parameters()).device
    net.eval()
    out = []
    with torch.no_grad():
        for off in range(0, len(images), 256):
            xb = torch.from_numpy(np.ascontiguousarray(images[off : off + 256])).to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16, enabled=device.type == "cuda"):
                out.append(net(xb).argmax(dim=1).cpu())
    return torch.cat(out).numpy()


@model.evaluator
def evaluator(net: ResNet18, images: np.ndarray, labels: np.ndarray) -> float:
    return float((predictor(net, images) == labels).mean())


@model.saver
def saver(net: ResNet18, hyperparameters, file, **kwargs):
    torch.save(
        {"state_dict": net.state_dict(), "hyperparameters": hyperparameters}, file
    )
    return file


@model.loader
def loader_fn(file, **kwargs) -> ResNet18:
    payload = torch.load(file, map_location="cpu", weights_only=False)
    hp = payload.get("hyperparameters") or {}
    if hasattr(hp, "__dict__") and not isinstance(hp, dict):
        hp = dict(hp.__dict__)
    net = ResNet18(**(hp or {"num_classes": 1000}))
    net.load_state_dict(payload["state_dict"])
    return net
