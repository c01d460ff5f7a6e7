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
