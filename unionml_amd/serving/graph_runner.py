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
                max_batch_size, 64, dtype=torch.float32, pin_memory=True
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
            x = torch.zeros(b, 64, dtype=torch.float32, device=self.mlp.device)
            preds = torch.zeros(b, dtype=torch.int32, device=self.mlp.device)
            ext = hip_ext(required=True)
            # warmup then capture
            ext.mlp_predict(
                x, self.mlp.mean, self.mlp.invstd, self.mlp.W1bf, self.mlp.W2bf,
                self.mlp.master, preds, None,
            )
            torch.cuda.synchronize(self.mlp.device)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                ext.mlp_predict(
                    x, self.mlp.mean, self.mlp.invstd, self.mlp.W1bf, self.mlp.W2bf,
                    self.mlp.master, preds, None,
                )
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
