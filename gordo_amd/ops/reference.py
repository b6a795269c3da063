"""
CPU/fp32 reference implementations of every device op.

These are the numerics oracles: each HIP kernel in ``csrc/`` is tested
against the same-named function here (tests/test_ops_gpu.py), and they
are the execution path on hosts without a GPU (the CPU test lane).

Layouts (G = models in the pack, B = rows, In/Out = features):
    X   [G, B, In]
    W   [G, In, Out]
    b   [G, Out]
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

ACT_LINEAR, ACT_TANH, ACT_RELU, ACT_SIGMOID = 0, 1, 2, 3

_ACT_NAMES = {
    "linear": ACT_LINEAR,
    None: ACT_LINEAR,
    "tanh": ACT_TANH,
    "relu": ACT_RELU,
    "sigmoid": ACT_SIGMOID,
}


def act_code(name) -> int:
    if isinstance(name, int):
        return name
    try:
        return _ACT_NAMES[name]
    except KeyError:
        raise ValueError(f"Unsupported activation {name!r}") from None


def apply_act(z: torch.Tensor, act: int) -> torch.Tensor:
    if act == ACT_LINEAR:
        return z
    if act == ACT_TANH:
        return torch.tanh(z)
    if act == ACT_RELU:
        return torch.relu(z)
    if act == ACT_SIGMOID:
        return torch.sigmoid(z)
    raise ValueError(f"Unknown activation code {act}")


def act_grad_from_output(y: torch.Tensor, act: int) -> torch.Tensor:
    """d act / d z expressed from the activation OUTPUT y (tanh: 1-y²,
    sigmoid: y(1-y), relu: 1[y>0]) — what the fused kernel computes."""
    if act == ACT_LINEAR:
        return torch.ones_like(y)
    if act == ACT_TANH:
        return 1.0 - y * y
    if act == ACT_RELU:
        return (y > 0).to(y.dtype)
    if act == ACT_SIGMOID:
        return y * (1.0 - y)
    raise ValueError(f"Unknown activation code {act}")


def grouped_linear_fwd(
    X: torch.Tensor, W: torch.Tensor, b: torch.Tensor, act: int
) -> torch.Tensor:
    """Y = act(X @ W + b) per group."""
    z = torch.baddbmm(b.unsqueeze(1), X, W)
    return apply_act(z, act)


def act_l1_bwd(
    dA: torch.Tensor, Y: torch.Tensor, act: int, l1: float
) -> torch.Tensor:
    """dZ = (dA + l1*sign(Y)) * act'(z), with act' from output Y.

    The l1 term is the gradient of an L1 activity regularizer on the
    activation output (reference: keras l1(1e-4) activity_regularizer,
    feedforward_autoencoder.py:81)."""
    g = dA if l1 == 0.0 else dA + l1 * torch.sign(Y)
    return g * act_grad_from_output(Y, act)


def grouped_linear_bwd_data(dZ: torch.Tensor, W: torch.Tensor) -> torch.Tensor:
    """dX = dZ @ W^T per group."""
    return torch.bmm(dZ, W.transpose(1, 2))


def grouped_linear_wgrad(
    X: torch.Tensor, dZ: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """dW = X^T @ dZ;  db = sum_rows dZ."""
    dW = torch.bmm(X.transpose(1, 2), dZ)
    db = dZ.sum(dim=1)
    return dW, db


def grouped_gemm_acc(A: torch.Tensor, B: torch.Tensor, C: torch.Tensor):
    """C += A @ B per group, in place (the LSTM recurrent-gate GEMM
    accumulating onto the precomputed x-side gates)."""
    C.baddbmm_(A, B)
    return C


def mse_bwd(
    Y: torch.Tensor, T: torch.Tensor, real_n: int = -1
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-model MSE loss and its gradient.

    loss[g] = mean((Y-T)^2) over (B, F); dY = 2 (Y-T) / (B*F).
    ``real_n`` overrides the divisor when the feature dim carries zero
    padding (pad diffs are exactly 0; only the mean's count changes).
    """
    diff = Y - T
    n = Y.shape[1] * Y.shape[2] if real_n <= 0 else real_n
    loss = (diff * diff).sum(dim=(1, 2)) / n
    dY = diff * (2.0 / n)
    return loss, dY


def adam_step(
    p: torch.Tensor,
    g: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    step: int,
    p_lp: Optional[torch.Tensor] = None,
):
    """Fused Adam over a flat fp32 parameter buffer (in-place); also
    refreshes the low-precision (bf16) mirror when given.

    Matches Keras Adam semantics: bias-corrected m̂/v̂,
    update = lr * m̂ / (sqrt(v̂) + eps).
    """
    m.mul_(beta1).add_(g, alpha=1.0 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    denom = (v / bc2).sqrt_().add_(eps)
    p.addcdiv_(m, denom, value=-lr / bc1)
    if p_lp is not None:
        p_lp.copy_(p)


def lstm_pointwise_fwd(
    gates: torch.Tensor, c_prev: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """
    LSTM cell pointwise math (Keras gate order i, f, c(g), o).

    gates: [G, B, 4H] pre-activation (x@Wx + h@Wh + b).
    Returns (h, c, gact) where gact = activated gates [G, B, 4H]
    saved for the backward pass.
    """
    H = gates.shape[-1] // 4
    i = torch.sigmoid(gates[..., 0 * H : 1 * H])
    f = torch.sigmoid(gates[..., 1 * H : 2 * H])
    g = torch.tanh(gates[..., 2 * H : 3 * H])
    o = torch.sigmoid(gates[..., 3 * H : 4 * H])
    c = f * c_prev + i * g
    h = o * torch.tanh(c)
    gact = torch.cat([i, f, g, o], dim=-1)
    return h, c, gact


def lstm_pointwise_bwd(
    dh: torch.Tensor,
    dc_next: torch.Tensor,
    gact: torch.Tensor,
    c: torch.Tensor,
    c_prev: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """
    Backward of the pointwise cell math.

    Inputs: dh (grad wrt h_t), dc_next (grad wrt c_t from t+1),
    gact = [i,f,g,o] activated, c = c_t, c_prev = c_{t-1}.
    Returns (dgates_pre [G,B,4H], dc_prev [G,B,H]).
    """
    H = dh.shape[-1]
    i, f, g, o = (
        gact[..., 0 * H : 1 * H],
        gact[..., 1 * H : 2 * H],
        gact[..., 2 * H : 3 * H],
        gact[..., 3 * H : 4 * H],
    )
    tanh_c = torch.tanh(c)
    do = dh * tanh_c
    dc = dc_next + dh * o * (1.0 - tanh_c * tanh_c)
    di = dc * g
    df = dc * c_prev
    dg = dc * i
    dc_prev = dc * f
    dgates = torch.cat(
        [
            di * i * (1.0 - i),
            df * f * (1.0 - f),
            dg * (1.0 - g * g),
            do * o * (1.0 - o),
        ],
        dim=-1,
    )
    return dgates, dc_prev
