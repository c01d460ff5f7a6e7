"""Pure-PyTorch fp32 reference implementations of the CDNA4 kernels.

These are (a) the numerics oracle the GPU tests compare the HIP kernels
against, and (b) the CPU execution path, so the same user-facing API
runs everywhere. Layouts match the kernels exactly (flat master with
W2/b2 padded to 16 columns — tabular_kernels.hip OFF_* constants).
"""

from typing import Optional, Tuple

import torch

IN, HID, CLS, CPAD = 64, 32, 10, 16
OFF_W1, OFF_B1, OFF_W2, OFF_B2 = 0, 2048, 2080, 2592
NPARAM = 2608

# hidden widths with a compiled gen-kernel instantiation (tabular_gen.hip)
SUPPORTED_HID = (32, 64, 128, 256)
MAX_CLASSES = 32


class Geometry:
    """Padded kernel geometry for an arbitrary (in_features, hidden,
    classes) tabular MLP.

    The HIP kernels run on a zero-padded image of the model: input width
    padded to a multiple of 32 (MFMA K granularity), hidden padded up to
    a compiled instantiation width, classes padded to one or two 16-wide
    MFMA tiles. Zero padding is mathematically exact under Adam-from-zero-init
    (zero init + zero gradient -> zero update), so the padded model's
    real sub-block evolves identically to the unpadded math.
    """

    def __init__(self, in_features: int, hidden: int, classes: int):
        if in_features < 1 or hidden < 1 or classes < 2:
            raise ValueError(f"bad geometry ({in_features}, {hidden}, {classes})")
        if classes > MAX_CLASSES:
            raise ValueError(
                f"classes={classes} unsupported: the fused classifier head spans "
                f"at most two MFMA tiles ({MAX_CLASSES} classes max)"
            )
        if hidden > SUPPORTED_HID[-1]:
            raise ValueError(
                f"hidden={hidden} exceeds the largest compiled width {SUPPORTED_HID[-1]}"
            )
        self.in_features = in_features
        self.hidden = hidden
        self.classes = classes
        self.inp = (in_features + 31) // 32 * 32
        self.hid = next(h for h in SUPPORTED_HID if h >= hidden)
        self.cpad = 16 if classes <= 16 else 32
        self.off_w1 = 0
        self.off_b1 = self.inp * self.hid
        self.off_w2 = self.off_b1 + self.hid
        self.off_b2 = self.off_w2 + self.hid * self.cpad
        self.nparam = self.off_b2 + self.cpad
        self.slab_stride = (self.nparam + 2 + 15) // 16 * 16
        # packed weight images: W1T [hid][inp] + W2s [hid][32] + W2T [cpad][hid]
        self.wimg_n = self.hid * self.inp + self.hid * 32 + self.cpad * self.hid
        # the hand-tuned specialized kernel path covers exactly this shape
        self.is_specialized = (in_features, hidden, classes) == (64, 32, 10)

    def __repr__(self):
        return (
            f"Geometry({self.in_features}->{self.hidden}->{self.classes}, "
            f"padded {self.inp}x{self.hid}x{self.cpad}, nparam={self.nparam})"
        )

    def __eq__(self, other):
        return isinstance(other, Geometry) and (
            (self.in_features, self.hidden, self.classes)
            == (other.in_features, other.hidden, other.classes)
        )


DIGITS = Geometry(64, 32, 10)
assert (DIGITS.off_b1, DIGITS.off_w2, DIGITS.off_b2, DIGITS.nparam) == (
    OFF_B1,
    OFF_W2,
    OFF_B2,
    NPARAM,
), "digits geometry must match the specialized kernel's layout"


def standardize_fit(X: torch.Tensor, eps: float = 1e-5) -> Tuple[torch.Tensor, torch.Tensor]:
    """Population column mean / 1/sqrt(var+eps) (fp64 accumulation like
    the kernel)."""
    Xd = X.double()
    mean = Xd.mean(dim=0)
    var = (Xd * Xd).mean(dim=0) - mean * mean
    invstd = 1.0 / torch.sqrt(torch.clamp(var, min=0) + eps)
    return mean.float(), invstd.float()


def standardize_apply(X: torch.Tensor, mean: torch.Tensor, invstd: torch.Tensor) -> torch.Tensor:
    return ((X - mean) * invstd).bfloat16()


def unpack_master(master: torch.Tensor):
    W1 = master[OFF_W1 : OFF_W1 + IN * HID].view(IN, HID)
    b1 = master[OFF_B1 : OFF_B1 + HID]
    W2 = master[OFF_W2 : OFF_W2 + HID * CPAD].view(HID, CPAD)
    b2 = master[OFF_B2 : OFF_B2 + CPAD]
    return W1, b1, W2, b2


def mlp_step(
    Xbf: torch.Tensor,
    y: torch.Tensor,
    W1bf: torch.Tensor,
    W2bf: torch.Tensor,
    master: torch.Tensor,
    grads: torch.Tensor,
    invBtot: float,
) -> None:
    """Fused fwd+bwd reference: accumulates grads (+loss at [NPARAM])
    into ``grads`` exactly like the kernel (which atomically adds onto a
    pre-zeroed buffer)."""
    _, b1, _, b2 = unpack_master(master)
    X = Xbf.float()
    W1 = W1bf.float()
    W2 = W2bf.float()
    Hpre = X @ W1 + b1
    H = torch.relu(Hpre)
    logits = H @ W2 + b2
    logits[:, CLS:] = -1e30
    m = logits.max(dim=1, keepdim=True).values
    e = torch.exp(logits - m)
    e[:, CLS:] = 0
    s = e.sum(dim=1, keepdim=True)
    p = e / s
    B = X.shape[0]
    onehot = torch.zeros_like(p)
    onehot[torch.arange(B, device=X.device), y.long()] = 1.0
    dlogits = (p - onehot) * invBtot
    loss = float(-(torch.log(p[torch.arange(B, device=X.device), y.long()])).sum() * invBtot)

    # bwd — bf16 round-trips where the kernel stores intermediates in LDS bf16
    Hbf = H.bfloat16().float()
    dlogits_bf = dlogits.bfloat16().float()
    dH = dlogits_bf @ W2.T
    dH = dH * (Hbf > 0)
    dH_bf = dH.bfloat16().float()

    dW2 = Hbf.T @ dlogits_bf
    db2 = dlogits_bf.sum(dim=0)
    dW1 = X.T @ dH_bf
    db1 = dH_bf.sum(dim=0)

    grads[OFF_W1 : OFF_W1 + IN * HID] += dW1.reshape(-1)
    grads[OFF_B1 : OFF_B1 + HID] += db1
    grads[OFF_W2 : OFF_W2 + HID * CPAD] += dW2.reshape(-1)
    grads[OFF_B2 : OFF_B2 + CPAD] += db2
    grads[NPARAM] += loss


def mlp_predict(
    X: torch.Tensor,
    mean: torch.Tensor,
    invstd: torch.Tensor,
    W1bf: torch.Tensor,
    W2bf: torch.Tensor,
    master: torch.Tensor,
    return_probs: bool = False,
):
    _, b1, _, b2 = unpack_master(master)
    Xs = ((X - mean) * invstd).bfloat16().float()
    H = torch.relu(Xs @ W1bf.float() + b1)
    logits = H.bfloat16().float() @ W2bf.float() + b2
    logits = logits[:, :CLS]
    preds = logits.argmax(dim=1).int()
    if return_probs:
        return preds, torch.softmax(logits, dim=1)
    return preds


# ---------------------------------------------------------------------------
# geometry-parametric variants (oracle + CPU path for the generalized
# kernels in tabular_gen.hip; layouts match the padded Geometry exactly)
# ---------------------------------------------------------------------------


def unpack_master_g(g: Geometry, master: torch.Tensor):
    W1 = master[g.off_w1 : g.off_w1 + g.inp * g.hid].view(g.inp, g.hid)
    b1 = master[g.off_b1 : g.off_b1 + g.hid]
    W2 = master[g.off_w2 : g.off_w2 + g.hid * g.cpad].view(g.hid, g.cpad)
    b2 = master[g.off_b2 : g.off_b2 + g.cpad]
    return W1, b1, W2, b2


def mlp_step_g(
    g: Geometry,
    Xbf: torch.Tensor,  # [B][inp] staged bf16 (zero-padded)
    y: torch.Tensor,
    W1bf: torch.Tensor,  # [inp][hid] bf16 mirror
    W2bf: torch.Tensor,  # [hid][cpad] bf16 mirror
    master: torch.Tensor,
    grads: torch.Tensor,
    invBtot: float,
) -> None:
    """Parametric fused fwd+bwd reference: accumulates grads (+loss at
    [g.nparam]) into ``grads`` exactly like mlp_step_gen_kernel."""
    _, b1, _, b2 = unpack_master_g(g, master)
    X = Xbf.float()
    W1 = W1bf.float()
    W2 = W2bf.float()
    Hpre = X @ W1 + b1
    H = torch.relu(Hpre)
    logits = H.bfloat16().float() @ W2 + b2
    logits[:, g.classes :] = -1e30
    m = logits.max(dim=1, keepdim=True).values
    e = torch.exp(logits - m)
    e[:, g.classes :] = 0
    s = e.sum(dim=1, keepdim=True)
    p = e / s
    B = X.shape[0]
    rows = torch.arange(B, device=X.device)
    onehot = torch.zeros_like(p)
    onehot[rows, y.long()] = 1.0
    dlogits = (p - onehot) * invBtot
    loss = float(-(torch.log(p[rows, y.long()])).sum() * invBtot)

    Hbf = H.bfloat16().float()
    dlogits_bf = dlogits.bfloat16().float()
    dH = dlogits_bf @ W2.T
    dH = dH * (Hbf > 0)
    dH_bf = dH.bfloat16().float()

    dW2 = Hbf.T @ dlogits_bf
    db2 = dlogits_bf.sum(dim=0)
    dW1 = X.T @ dH_bf
    db1 = dH_bf.sum(dim=0)

    grads[g.off_w1 : g.off_w1 + g.inp * g.hid] += dW1.reshape(-1)
    grads[g.off_b1 : g.off_b1 + g.hid] += db1
    grads[g.off_w2 : g.off_w2 + g.hid * g.cpad] += dW2.reshape(-1)
    grads[g.off_b2 : g.off_b2 + g.cpad] += db2
    grads[g.nparam] += loss


def mlp_predict_g(
    g: Geometry,
    X: torch.Tensor,  # [B][in_features] raw fp32
    mean: torch.Tensor,
    invstd: torch.Tensor,
    W1bf: torch.Tensor,
    W2bf: torch.Tensor,
    master: torch.Tensor,
    return_probs: bool = False,
):
    _, b1, _, b2 = unpack_master_g(g, master)
    Xs = ((X - mean) * invstd).bfloat16().float()
    if g.inp > g.in_features:  # zero-pad to the staged width
        pad = torch.zeros(X.shape[0], g.inp - g.in_features, device=X.device)
        Xs = torch.cat([Xs, pad], dim=1)
    H = torch.relu(Xs @ W1bf.float() + b1)
    logits = H.bfloat16().float() @ W2bf.float() + b2
    logits = logits[:, : g.classes]
    preds = logits.argmax(dim=1).int()
    if return_probs:
        return preds, torch.softmax(logits, dim=1)
    return preds


def adam_step_g(
    g: Geometry,
    master: torch.Tensor,
    bfmirror: torch.Tensor,
    grads: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    t: int,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
) -> None:
    gr = grads[: g.nparam]
    m.mul_(beta1).add_(gr, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gr, gr, value=1 - beta2)
    corr1 = 1.0 / (1.0 - beta1**t)
    corr2 = 1.0 / (1.0 - beta2**t)
    master[: g.nparam] -= lr * (m * corr1) / (torch.sqrt(v * corr2) + eps)
    bfmirror[: g.nparam] = master[: g.nparam].bfloat16()


def adam_step(
    master: torch.Tensor,
    bfmirror: torch.Tensor,
    grads: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    t: int,
    lr: float,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
) -> None:
    g = grads[:NPARAM]
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    corr1 = 1.0 / (1.0 - beta1**t)
    corr2 = 1.0 / (1.0 - beta2**t)
    master[:NPARAM] -= lr * (m * corr1) / (torch.sqrt(v * corr2) + eps)
    bfmirror[:NPARAM] = master[:NPARAM].bfloat16()
