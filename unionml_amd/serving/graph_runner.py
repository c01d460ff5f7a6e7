"""hipGraph-captured inference for serving (SURVEY.md §2c row
'hipGraph inference step').

Graphs are shape-static, so inference is captured once per batch-size
bucket {1, 2, 4, ..., max_batch}; a request batch is padded up to its
bucket, the bucket's graph is replayed, and the padded tail discarded.
Input rows land in a pinned staging buffer -> async H2D into the
graph's static input tensor -> replay -> D2H of the predictions.
"""

from typing import Callable, Dict

import numpy as np
import torch

from unionml_amd._logging import logger
from unionml_amd.serving.batcher import bucket_for


class TabularGraphRunner:
    """Bucketed hipGraph replay around TabularMLP's fused predict kernel."""

    def __init__(self, mlp, max_batch_size: int = 64, precapture: bool = True):
        from unionml_amd.ops.tabular import TabularMLP

        assert isinstance(mlp, TabularMLP)
        self.mlp = mlp
        self.max_batch = max_batch_size
        self.use_graphs = mlp.device.type == "cuda"
        self._buckets: Dict[int, tuple] = {}
        if self.use_graphs:
            self._pinned = torch.empty(
                max_batch_size, mlp.g.in_features, dtype=torch.float32, pin_memory=True
            )
            if precapture:
                # capture every bucket up front so a first request of a
                # new size never pays ~100ms of graph capture (observed
                # as p99 spikes under concurrent load)
                b = 1
                while b < max_batch_size:
                    self._get_bucket(b)
                    b <<= 1
                self._get_bucket(max_batch_size)

    def _get_bucket(self, b: int):
        from unionml_amd.ops import hip_ext

        entry = self._buckets.get(b)
        if entry is None:
            mlp = self.mlp
            x = torch.zeros(b, mlp.g.in_features, dtype=torch.float32, device=mlp.device)
            preds = torch.zeros(b, dtype=torch.int32, device=mlp.device)
            ext = hip_ext(required=True)

            def fwd():
                if mlp.use_spec:
                    ext.mlp_predict(
                        x, mlp.mean, mlp.invstd, mlp.W1bf, mlp.W2bf,
                        mlp.master, preds, None,
                    )
                else:
                    ext.mlp_predict_gen(
                        x, mlp.g.inp, mlp.g.hid, mlp.g.classes, mlp.mean,
                        mlp.invstd, mlp.wimg, mlp.master, preds, None,
                    )

            fwd()  # warmup then capture
            torch.cuda.synchronize(mlp.device)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                fwd()
            entry = (g, x, preds)
            self._buckets[b] = entry
            logger.info("captured inference hipGraph for bucket %d", b)
        return entry

    def __call__(self, features) -> np.ndarray:
        if hasattr(features, "to_numpy"):
            features = features.to_numpy()
        if isinstance(features, np.ndarray):
            features = torch.from_numpy(np.ascontiguousarray(features, dtype=np.float32))
        n = features.shape[0]
        if not self.use_graphs:
            return self.mlp.predict(features).cpu().numpy()
        out = np.empty(n, dtype=np.int32)
        off = 0
        while off < n:
            chunk = min(n - off, self.max_batch)
            b = bucket_for(chunk, self.max_batch)
            g, x_static, preds_static = self._get_bucket(b)
            host = self._pinned[:b]
            host[:chunk].copy_(features[off : off + chunk])
            if chunk < b:
                host[chunk:].zero_()
            x_static.copy_(host, non_blocking=True)
            g.replay()
            out[off : off + chunk] = preds_static[:chunk].cpu().numpy()
            off += chunk
        return out


class ModuleGraphRunner:
    """Bucketed hipGraph replay around an arbitrary ``nn.Module`` forward.

    Generalizes :class:`TabularGraphRunner` to any torch model: the
    module's (bf16-autocast, eval-mode) forward — plus an optional
    captured postprocess such as argmax — is recorded once per
    power-of-two batch bucket; requests are padded to their bucket and
    replayed. Rows stage through a pinned host buffer when the input
    arrives from the CPU. On CPU devices it falls back to a plain
    no-grad forward (one code path, two substrates).
    """

    def __init__(
        self,
        module,
        max_batch_size: int = 64,
        row_shape: tuple = None,
        autocast_dtype=torch.bfloat16,
        channels_last: bool = False,
        postprocess: str = "argmax",
        precapture: bool = False,
    ):
        self.module = module.eval()
        try:
            self.device = next(module.parameters()).device
        except StopIteration:
            self.device = torch.device("cpu")
        self.max_batch = max_batch_size
        self.use_graphs = self.device.type == "cuda"
        self.row_shape = tuple(row_shape) if row_shape else None
        self.autocast_dtype = autocast_dtype
        self.channels_last = channels_last
        self.postprocess = postprocess
        self._buckets: Dict[int, tuple] = {}
        self._pinned = None
        if self.use_graphs and precapture and self.row_shape:
            b = 1
            while b < max_batch_size:
                self._get_bucket(b)
                b <<= 1
            self._get_bucket(max_batch_size)

    def _apply_post(self, out):
        return out.argmax(dim=1) if self.postprocess == "argmax" else out

    def _forward(self, x):
        with torch.no_grad(), torch.autocast(
            self.device.type, dtype=self.autocast_dtype, enabled=self.use_graphs
        ):
            return self._apply_post(self.module(x))

    def _get_bucket(self, b: int):
        entry = self._buckets.get(b)
        if entry is None:
            x = torch.zeros(b, *self.row_shape, dtype=torch.float32, device=self.device)
            if self.channels_last and x.dim() == 4:
                x = x.to(memory_format=torch.channels_last)
            # warm up (MIOpen find, workspace allocs) outside the capture
            for _ in range(2):
                self._forward(x)
            torch.cuda.synchronize(self.device)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                out = self._forward(x)
            entry = (g, x, out)
            self._buckets[b] = entry
            logger.info("captured module hipGraph for bucket %d", b)
        return entry

    def __call__(self, features):
        if hasattr(features, "to_numpy"):
            features = features.to_numpy()
        if isinstance(features, np.ndarray):
            features = torch.from_numpy(np.ascontiguousarray(features, dtype=np.float32))
        if self.row_shape is None:
            self.row_shape = tuple(features.shape[1:])
        n = features.shape[0]
        if not self.use_graphs:
            return self._forward(features.to(self.device)).cpu().numpy()
        if self._pinned is None:
            self._pinned = torch.empty(
                self.max_batch, *self.row_shape, dtype=torch.float32, pin_memory=True
            )
        host_src = features if not features.is_cuda else None
        outs = []
        off = 0
        while off < n:
            chunk = min(n - off, self.max_batch)
            b = bucket_for(chunk, self.max_batch)
            g, x_static, out_static = self._get_bucket(b)
            if host_src is not None:
                host = self._pinned[:b]
                host[:chunk].copy_(host_src[off : off + chunk])
                if chunk < b:
                    host[chunk:].zero_()
                x_static.copy_(host, non_blocking=True)
            else:
                x_static[:chunk].copy_(features[off : off + chunk])
                if chunk < b:
                    x_static[chunk:].zero_()
            g.replay()
            outs.append(out_static[:chunk].cpu())
            off += chunk
        res = torch.cat(outs)
        if res.dtype == torch.bfloat16:
            res = res.float()
        return res.numpy()


def graphed(factory: Callable):
    """Decorator attaching a graph-runner factory to a predictor fn:

        @model.predictor
        @graphed(lambda model_obj, max_batch: TabularGraphRunner(model_obj, max_batch))
        def predictor(mlp, features) -> List[int]: ...
    """

    def wrap(fn):
        fn.__unionml_graphed__ = factory
        return fn

    return wrap
