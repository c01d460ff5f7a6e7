"""TabularMLP — the framework's default tabular hot path, for ANY
(in_features, hidden, classes) geometry.

A one-hidden-layer classifier (Linear -> ReLU -> Linear -> Softmax)
whose training step (standardize, fused forward, cross-entropy
backward, fused Adam) runs as hand-written CDNA4 HIP kernels on MFMA:

* the hand-tuned specialized kernels for the 64->32->10 digits shape
  (unionml_amd/ops/hip/tabular_kernels.hip — the headline benchmark
  path, with the persistent and packed-weight-image engines), and
* the generalized templated kernels for every other shape
  (unionml_amd/ops/hip/tabular_gen.hip — input width streamed through
  LDS in k-tiles, hidden width dispatched over compiled instantiations
  {32,64,128,256}, odd sizes zero-padded exactly).

The per-epoch minibatch loop is captured ONCE into a hipGraph and
replayed per epoch — zero launch overhead in steady state. On CPU the
same class runs the pure-torch reference path
(unionml_amd/ops/reference.py): one user-visible code path, two
substrates.

Data parallel: under an active torch.distributed process group each
rank trains its row shard and the flat gradient buffer is all-reduced
on RCCL between the reduce-only step and the Adam kernel.

Reference behavior mirrored (no code ported — the reference is pure
Python): tests/integration/pytorch_app/quickstart.py:14-70.
"""

import math
from typing import Dict, Optional

import torch

from unionml_amd._logging import logger
from unionml_amd.ops import hip_ext
from unionml_amd.ops import reference as ref
from unionml_amd.ops.reference import CPAD, HID, IN, Geometry

ADAM_BETA1, ADAM_BETA2, ADAM_EPS = 0.9, 0.999, 1e-8


class TabularMLP:
    """Device-resident parameter/optimizer state + fused train/predict."""

    def __init__(
        self,
        in_features: int = 64,
        hidden: int = 32,
        classes: int = 10,
        device: Optional[str] = None,
        seed: int = 0,
    ):
        self.g = Geometry(in_features, hidden, classes)
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.use_hip = self.device.type == "cuda"
        if self.use_hip:
            hip_ext(required=True)  # loud failure if the gfx950 ext is missing
        # hand-tuned specialized kernels for the digits shape; generalized
        # templated kernels for everything else
        self.use_spec = self.g.is_specialized

        g = self.g
        gen_t = torch.Generator().manual_seed(seed)
        master = torch.zeros(g.nparam, dtype=torch.float32)
        W1, b1, W2, b2 = ref.unpack_master_g(g, master)
        W1[: g.in_features, : g.hidden] = (
            torch.randn(g.in_features, g.hidden, generator=gen_t)
            * math.sqrt(2.0 / g.in_features)
        )
        W2[: g.hidden, : g.classes] = (
            torch.randn(g.hidden, g.classes, generator=gen_t) * math.sqrt(2.0 / g.hidden)
        )

        self.master = master.to(self.device)
        self.bfmirror = self.master.bfloat16()
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)
        self.t_dev = torch.zeros(1, dtype=torch.int32, device=self.device)
        self.grads = torch.zeros(g.nparam + 1, dtype=torch.float32, device=self.device)
        self.mean = torch.zeros(g.in_features, dtype=torch.float32, device=self.device)
        self.invstd = torch.ones(g.in_features, dtype=torch.float32, device=self.device)
        self.slabs = None       # per-WG partial-grad slabs for the fused step
        self.counter = None     # G16 epoch counter (shared by both step modes)
        self._graph = None
        self._graph_key = None
        # packed weight images (kernel-layout staging) so the training
        # prologue is a straight vectorized copy; kept in sync by the
        # Adam kernels (wimg=) and rebuilt on any host-side weight
        # mutation. Specialized and generalized kernels use different
        # packings (LDS-stride padded vs dense row-major).
        self.wimg = (
            torch.zeros(self._wimg_len(), dtype=torch.bfloat16, device=self.device)
            if self.use_hip
            else None
        )
        self._build_wimg()

    _XS, _WS = 72, 40  # specialized-kernel LDS row strides (tabular_kernels.hip)

    def _wimg_len(self) -> int:
        return 4224 if self.use_spec else self.g.wimg_n

    #: per hidden width: (default rows/WG, small-batch rows/WG). The
    #: small variant keeps >= ~16 workgroups in flight at modest batch
    #: sizes — the fwd/bwd phase is bandwidth-bound and achievable
    #: bandwidth scales with resident CUs.
    _GEN_RTS = {32: (128, 64), 64: (128, 64), 128: (64, 32), 256: (32, 32)}

    def _rows_per_wg(self, batch: Optional[int] = None) -> int:
        if self.use_spec:
            return 128
        big, small = self._GEN_RTS[self.g.hid]
        if self.g.cpad > 16:
            return big  # 32-class head compiles only at the default rows/WG
        # smaller rows/WG doubles the grid (the bandwidth-bound fwd/bwd
        # scales with resident CUs: measured -12% at B=2048, -11% at
        # B=4096 on the MNIST shape) — but every extra WG re-reads the
        # whole W1 image and adds a full-nparam gradient slab, which
        # turns pathological past ~128 WGs (B=8192: 335 vs 65 us).
        if batch is not None and (batch + small - 1) // small <= 128:
            return small
        return big

    def _build_wimg(self):
        if self.wimg is None:
            return
        g = self.g
        W1, _, W2, _ = ref.unpack_master_g(g, self.master.detach().cpu())
        if self.use_spec:
            XS, WS = self._XS, self._WS
            img = torch.zeros(4224, dtype=torch.bfloat16)
            img[: HID * XS].view(HID, XS)[:, :IN] = W1.t().bfloat16()      # W1T[h][k]
            img[HID * XS : HID * XS + HID * WS].view(HID, WS)[:, :CPAD] = W2.bfloat16()
            img[HID * XS + HID * WS :].view(CPAD, WS)[:, :HID] = W2.t().bfloat16()
        else:
            img = torch.zeros(g.wimg_n, dtype=torch.bfloat16)
            o1 = g.hid * g.inp
            o2 = o1 + g.hid * 32
            img[:o1].view(g.hid, g.inp).copy_(W1.t().bfloat16())           # W1T[h][k]
            img[o1:o2].view(g.hid, 32)[:, : g.cpad] = W2.bfloat16()        # W2s K-pad
            img[o2:].view(g.cpad, g.hid).copy_(W2.t().bfloat16())          # W2T[c][h]
        self.wimg.copy_(img.to(self.device))

    def _ensure_slabs(self, n_wg: int):
        if self.slabs is None or self.slabs.shape[0] < n_wg:
            self.slabs = torch.zeros(
                n_wg, self.g.slab_stride, dtype=torch.float32, device=self.device
            )
            self.counter = torch.zeros(1, dtype=torch.uint32, device=self.device)

    # -- views ----------------------------------------------------------------

    @property
    def W1bf(self) -> torch.Tensor:
        g = self.g
        return self.bfmirror[g.off_w1 : g.off_w1 + g.inp * g.hid].view(g.inp, g.hid)

    @property
    def W2bf(self) -> torch.Tensor:
        g = self.g
        return self.bfmirror[g.off_w2 : g.off_w2 + g.hid * g.cpad].view(g.hid, g.cpad)

    # -- standardizer ----------------------------------------------------------

    def fit_standardizer(self, X: torch.Tensor, eps: float = 1e-5):
        X = X.to(self.device, torch.float32).contiguous()
        if self.use_hip:
            hip_ext().standardize_fit(X, self.mean, self.invstd, eps)
        else:
            mean, invstd = ref.standardize_fit(X, eps)
            self.mean.copy_(mean)
            self.invstd.copy_(invstd)

    def stage(self, X: torch.Tensor) -> torch.Tensor:
        """Standardize fp32 features into a device-resident bf16 matrix,
        zero-padded to the kernel's input width."""
        g = self.g
        X = X.to(self.device, torch.float32).contiguous()
        if self.use_hip:
            if g.inp == g.in_features:
                out = torch.empty(X.shape[0], g.inp, dtype=torch.bfloat16, device=self.device)
            else:
                out = torch.zeros(X.shape[0], g.inp, dtype=torch.bfloat16, device=self.device)
            hip_ext().standardize_apply(X, self.mean, self.invstd, out)
            return out
        out_raw = ref.standardize_apply(X, self.mean, self.invstd)
        if g.inp == g.in_features:
            return out_raw
        out = torch.zeros(X.shape[0], g.inp, dtype=torch.bfloat16)
        out[:, : g.in_features] = out_raw
        return out

    # -- training --------------------------------------------------------------

    def _step_reduce(self, Xbf: torch.Tensor, y: torch.Tensor, invBtot: float, lr: float):
        """Fused fwd/bwd + in-kernel slab reduction writing the summed
        grads into self.grads (reduce-only mode: the DP pre-collective
        kernel — no Adam)."""
        g = self.g
        rpw = self._rows_per_wg(Xbf.shape[0])
        n_wg = (Xbf.shape[0] + rpw - 1) // rpw
        self._ensure_slabs(n_wg)
        loss_out = self.grads[g.nparam : g.nparam + 1]
        ext = hip_ext()
        if self.use_spec:
            ok = ext.mlp_step_fused(
                Xbf, y, self.W1bf, self.W2bf, self.master, self.bfmirror,
                self.m, self.v, self.t_dev, self.slabs, self.counter, loss_out,
                invBtot, lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                grads_out=self.grads, wimg=self.wimg,
            )
            assert ok, "fused step slab capacity exceeded"
        else:
            ok = ext.mlp_step_gen(
                Xbf, y, g.hid, g.classes, rpw, self.wimg, self.master,
                self.slabs, invBtot,
            )
            assert ok, "fused step slab capacity exceeded"
            ext.reduce_adam_gen(
                self.slabs, n_wg, g.inp, g.hid, g.cpad, self.master,
                self.bfmirror, self.m, self.v, self.t_dev, self.counter,
                loss_out, lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                wimg=self.wimg, grads_out=self.grads,
            )

    def _adam(self, lr: float):
        g = self.g
        ext = hip_ext()
        if self.use_spec:
            ext.adam_step(
                self.master, self.bfmirror, self.grads, self.m, self.v, self.t_dev,
                lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS, wimg=self.wimg,
            )
        else:
            ext.adam_step_gen(
                self.master, self.bfmirror, self.grads, self.m, self.v, self.t_dev,
                g.inp, g.hid, g.cpad, lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                wimg=self.wimg,
            )

    def _step(self, Xbf: torch.Tensor, y: torch.Tensor, invBtot: float, lr: float,
              allreduce: bool):
        g = self.g
        if self.use_hip:
            self._step_reduce(Xbf, y, invBtot, lr)
        else:
            self.grads.zero_()
            ref.mlp_step_g(g, Xbf, y, self.W1bf, self.W2bf, self.master, self.grads, invBtot)
        if allreduce:
            import torch.distributed as dist

            dist.all_reduce(self.grads)
        if self.use_hip:
            self._adam(lr)
        else:
            self.t_dev += 1
            ref.adam_step_g(
                g, self.master, self.bfmirror, self.grads, self.m, self.v,
                int(self.t_dev.item()), lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
            )
        return self.grads[g.nparam]

    def _fused_adam_step(self, Xbf, y, invBtot, lr, loss_out):
        """One optimizer step: a single fused launch for the specialized
        shape; for generalized shapes the slab-producing step kernel
        followed by the wide-grid reduce+Adam kernel (the kernel
        boundary is the inter-WG barrier — full-chip bandwidth for the
        large-nparam reduction instead of a handshake inside a tiny
        grid)."""
        g = self.g
        ext = hip_ext()
        if self.use_spec:
            ok = ext.mlp_step_fused(
                Xbf, y, self.W1bf, self.W2bf, self.master, self.bfmirror,
                self.m, self.v, self.t_dev, self.slabs, self.counter, loss_out,
                invBtot, lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS, wimg=self.wimg,
            )
            assert ok, "fused step slab capacity exceeded"
        else:
            rpw = self._rows_per_wg(Xbf.shape[0])
            n_wg = (Xbf.shape[0] + rpw - 1) // rpw
            self._ensure_slabs(n_wg)
            ok = ext.mlp_step_gen(
                Xbf, y, g.hid, g.classes, rpw, self.wimg, self.master,
                self.slabs, invBtot,
            )
            assert ok, "fused step slab capacity exceeded"
            ext.reduce_adam_gen(
                self.slabs, n_wg, g.inp, g.hid, g.cpad, self.master,
                self.bfmirror, self.m, self.v, self.t_dev, self.counter,
                loss_out, lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS, wimg=self.wimg,
            )

    def _train_epochs_fused(self, Xbf, y, batches, *, epochs, lr, use_graph) -> float:
        g = self.g
        self._ensure_slabs(
            max(
                (bs + self._rows_per_wg(bs) - 1) // self._rows_per_wg(bs)
                for _, bs in batches
            )
        )
        loss_out = self.grads[g.nparam : g.nparam + 1]

        def run_epoch():
            for off, bs in batches:
                self._fused_adam_step(
                    Xbf[off : off + bs], y[off : off + bs], 1.0 / bs, lr, loss_out
                )

        run_epoch()   # warmup epoch (eager)
        remaining = epochs - 1
        key = ("fused", len(batches), lr, Xbf.data_ptr(), y.data_ptr())
        if use_graph and self._graph_key != key:
            try:
                gph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(gph):
                    run_epoch()
                self._graph, self._graph_key = gph, key
            except RuntimeError as exc:
                logger.warning("hipGraph capture failed (%s); eager stepping", exc)
                self._graph, self._graph_key = None, None
        if use_graph and self._graph is not None:
            for _ in range(remaining):
                self._graph.replay()
        else:
            for _ in range(remaining):
                run_epoch()
        torch.cuda.synchronize(self.device)
        return float(loss_out.item())

    def train_epochs(
        self,
        Xbf: torch.Tensor,
        y: torch.Tensor,
        *,
        epochs: int = 10,
        batch_size: int = 512,
        lr: float = 1e-3,
        use_graph: bool = True,
        world_size: int = 1,
        engine: str = "auto",
    ) -> float:
        """Minibatch Adam training over the staged (bf16) features.

        Returns the last step's loss. ``world_size > 1`` means this rank
        holds a shard and gradients all-reduce over RCCL each step.
        """
        g = self.g
        y = y.to(self.device, torch.int32).contiguous()
        n = Xbf.shape[0]
        batches = [
            (off, min(batch_size, n - off)) for off in range(0, n, batch_size)
        ]
        allreduce = world_size > 1

        # single-GPU flagship: the fully-fused step kernel (fwd+bwd +
        # cross-WG slab reduction + Adam in ONE launch), with the epoch's
        # minibatch loop captured into a hipGraph.
        if self.use_hip and not allreduce:
            if (
                self.use_spec
                and engine == "persistent"
                and batch_size % 128 == 0
                and n % batch_size == 0
            ):
                loss_out = self.grads[g.nparam : g.nparam + 1]
                n_steps = epochs * (n // batch_size)
                ok = hip_ext().mlp_train_steps(
                    Xbf, y, batch_size, n_steps, self.master, self.bfmirror,
                    self.m, self.v, self.t_dev, loss_out,
                    lr, ADAM_BETA1, ADAM_BETA2, ADAM_EPS,
                )
                if ok:
                    torch.cuda.synchronize(self.device)
                    # the persistent kernel updates master/bfmirror but not
                    # the packed weight images; refresh them so a later
                    # fused/adam step (wimg=) doesn't load stale weights
                    self._build_wimg()
                    return float(loss_out.item())
            return self._train_epochs_fused(
                Xbf, y, batches, epochs=epochs, lr=lr, use_graph=use_graph
            )

        def run_epoch():
            last = None
            for off, bs in batches:
                last = self._step(
                    Xbf[off : off + bs],
                    y[off : off + bs],
                    1.0 / (bs * world_size),
                    lr,
                    allreduce,
                )
            return last

        if not (self.use_hip and use_graph):
            loss = None
            for _ in range(epochs):
                loss = run_epoch()
            # with allreduce the stored loss is already the global mean
            return float(loss.item()) if loss is not None else float("nan")

        # hipGraph path: one eager warmup epoch (also warms the RCCL
        # communicator / memory pool), then capture one epoch (capture
        # records without executing) and replay the remaining epochs.
        run_epoch()
        remaining = epochs - 1
        key = (n, batch_size, world_size, lr, Xbf.data_ptr(), y.data_ptr())
        if self._graph_key != key:
            try:
                gph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(gph):
                    run_epoch()
                self._graph, self._graph_key = gph, key
            except RuntimeError as exc:  # e.g. RCCL capture unsupported
                logger.warning("hipGraph capture failed (%s); eager stepping", exc)
                self._graph, self._graph_key = None, None
        if self._graph is not None:
            for _ in range(remaining):
                self._graph.replay()
        else:
            for _ in range(remaining):
                run_epoch()
        torch.cuda.synchronize(self.device)
        return float(self.grads[g.nparam].item())

    # -- inference -------------------------------------------------------------

    def predict(self, X: torch.Tensor, return_probs: bool = False):
        g = self.g
        X = X.to(self.device, torch.float32).contiguous()
        if self.use_hip:
            preds = torch.empty(X.shape[0], dtype=torch.int32, device=self.device)
            probs = (
                torch.empty(X.shape[0], g.classes, dtype=torch.float32, device=self.device)
                if return_probs
                else None
            )
            if self.use_spec:
                hip_ext().mlp_predict(
                    X, self.mean, self.invstd, self.W1bf, self.W2bf, self.master,
                    preds, probs,
                )
            else:
                hip_ext().mlp_predict_gen(
                    X, g.inp, g.hid, g.classes, self.mean, self.invstd, self.wimg,
                    self.master, preds, probs,
                )
            return (preds, probs) if return_probs else preds
        return ref.mlp_predict_g(
            g, X, self.mean, self.invstd, self.W1bf, self.W2bf, self.master, return_probs
        )

    # -- persistence -----------------------------------------------------------

    def __getstate__(self):
        """Pickle CPU-portably: device caches (slabs, graphs) are
        rebuilt lazily; tensors travel as CPU copies. This is what lets
        a GPU-trained classifier cross the backend's process boundary
        and load on any machine."""
        d = dict(self.__dict__)
        d["slabs"] = d["counter"] = None
        d["_graph"] = d["_graph_key"] = None
        d["wimg"] = None
        d["device"] = str(self.device)
        for k, v in list(d.items()):
            if torch.is_tensor(v):
                d[k] = v.detach().cpu()
        return d

    def __setstate__(self, state):
        device = torch.device(state.pop("device"))
        if device.type == "cuda" and not torch.cuda.is_available():
            device = torch.device("cpu")
        state.setdefault("g", Geometry(64, 32, 10))       # pre-geometry pickles
        state.setdefault("use_spec", state["g"].is_specialized)
        self.__dict__.update(state)
        self.device = device
        self.use_hip = device.type == "cuda"
        if self.use_hip:
            hip_ext(required=True)
        for k, v in list(self.__dict__.items()):
            if torch.is_tensor(v):
                setattr(self, k, v.to(device))
        if self.use_hip and self.wimg is None:
            self.wimg = torch.zeros(self._wimg_len(), dtype=torch.bfloat16, device=device)
        self._build_wimg()

    def state_dict(self) -> Dict[str, torch.Tensor]:
        """Logical (unpadded) weights + standardizer + Adam state.

        Includes the optimizer moments and step counter so a reloaded
        model RESUMES training exactly (bit-identical step sequence),
        not just serves inference. The moments are stored in the padded
        master layout — safe because load_state_dict enforces an exact
        geometry match, and the padded lanes are exact zeros by the
        Adam-from-zero-init invariant.
        """
        g = self.g
        W1, b1, W2, b2 = ref.unpack_master_g(g, self.master.cpu())
        return {
            "W1": W1[: g.in_features, : g.hidden].clone(),
            "b1": b1[: g.hidden].clone(),
            "W2": W2[: g.hidden, : g.classes].clone(),
            "b2": b2[: g.classes].clone(),
            "mean": self.mean.cpu().clone(),
            "invstd": self.invstd.cpu().clone(),
            "geometry": torch.tensor([g.in_features, g.hidden, g.classes]),
            "adam_m": self.m.cpu().clone(),
            "adam_v": self.v.cpu().clone(),
            "adam_t": self.t_dev.cpu().clone(),
        }

    def load_state_dict(self, state: Dict[str, torch.Tensor]):
        g = self.g
        if "geometry" in state:
            want = tuple(int(x) for x in state["geometry"])
            have = (g.in_features, g.hidden, g.classes)
            if want != have:
                raise ValueError(f"state geometry {want} != model geometry {have}")
        master = torch.zeros(g.nparam, dtype=torch.float32)
        W1, b1, W2, b2 = ref.unpack_master_g(g, master)
        W1[: g.in_features, : g.hidden] = state["W1"]
        b1[: g.hidden] = state["b1"]
        W2[: g.hidden, : g.classes] = state["W2"]
        b2[: g.classes] = state["b2"]
        self.master.copy_(master.to(self.device))
        self.bfmirror.copy_(self.master.bfloat16())
        self.mean.copy_(state["mean"].to(self.device))
        self.invstd.copy_(state["invstd"].to(self.device))
        if "adam_m" in state:  # full checkpoint: resume training exactly
            self.m.copy_(state["adam_m"].to(self.device))
            self.v.copy_(state["adam_v"].to(self.device))
            self.t_dev.copy_(state["adam_t"].to(self.device))
        else:  # weights-only (pre-resume checkpoints): fresh optimizer
            self.m.zero_()
            self.v.zero_()
            self.t_dev.zero_()
        self._graph = self._graph_key = None
        self._build_wimg()
