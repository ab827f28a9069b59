"""Population (many-model) functional models over the MFMA popconv kernels.

The whole rank-local client population trains as ONE computation: per-layer
activations live in the padded population layout

    [C, ch, B, Hp, Wp]   (Hp = H+2, Wp = W+2, 1-pixel zero pads)

and every 3x3 conv is the direct shifted-tap MFMA kernel (csrc/popconv.hip)
— no MIOpen, no im2col, per-client weight grads native.  BatchNorm
(batch-stats), ReLU, residual adds, pooling and the FC head are plain
autograd tensor ops on the padded layout (pads stay exactly zero).

``pop_resnet18`` mirrors blades_amd.models.resnet.ResNet(BasicBlock,
[2,2,2,2], norm="batch-local") parameter-for-parameter: the same ParamSpec
slab drives either implementation, and tests assert gradient parity against
the per-client loop engine.

CPU fallback: the autograd Functions compute with torch.nn.grad per client
so the full population path is testable without a GPU.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn.functional as F

Tensor = torch.Tensor

_EXT = None
_EXT_ERR: Optional[str] = None


def _ext():
    global _EXT, _EXT_ERR
    if _EXT is None and _EXT_ERR is None:
        try:
            from blades_amd import _hip_popconv
            _EXT = _hip_popconv
        except ImportError as e:  # pragma: no cover
            _EXT_ERR = str(e)
    return _EXT


def popconv_available() -> bool:
    return _ext() is not None


# ---------------------------------------------------------------- helpers

def to_pop_layout(x: Tensor) -> Tensor:
    """[C, B, ch, H, W] -> padded population layout [C, ch, B, H+2, W+2]."""
    x = x.permute(0, 2, 1, 3, 4)
    return F.pad(x, (1, 1, 1, 1))


def zero_pads_(x: Tensor) -> Tensor:
    """Zero the 1-pixel spatial border in place (no autograd tracking)."""
    with torch.no_grad():
        x[..., 0, :] = 0
        x[..., -1, :] = 0
        x[..., :, 0] = 0
        x[..., :, -1] = 0
    return x


def interior(x: Tensor) -> Tensor:
    return x[..., 1:-1, 1:-1]


def _cpu_conv(x5: Tensor, w5: Tensor) -> Tensor:
    """Reference per-client conv on the padded layout (CPU fallback)."""
    C, ci, B, Hp, Wp = x5.shape
    co = w5.shape[1]
    out = x5.new_zeros(C, co, B, Hp, Wp)
    for c in range(C):
        xc = x5[c].permute(1, 0, 2, 3)  # [B, ci, Hp, Wp]
        yc = F.conv2d(xc[..., 1:-1, 1:-1], w5[c], padding=1)
        out[c, :, :, 1:-1, 1:-1] = yc.permute(1, 0, 2, 3)
    return out


class _PopConv3x3(torch.autograd.Function):
    """Y = popconv(X, W) on the padded population layout.

    X: [C, ci, B, Hp, Wp] (pads zero), W: [C, co, ci, 3, 3] (may be a
    stride-0 expand over C for shared weights).  Y has zeroed pads.
    """

    @staticmethod
    def forward(ctx, X: Tensor, W: Tensor) -> Tensor:
        C, ci, B, Hp, Wp = X.shape
        co = W.shape[1]
        ctx.save_for_backward(X, W)
        if X.is_cuda:
            ext = _ext()
            if ext is None:
                raise RuntimeError(
                    f"popconv extension missing on GPU: {_EXT_ERR}")
            Xf = X.reshape(C, ci, B * Hp * Wp)
            Y = ext.popconv_fwd(Xf, W, B, Hp, Wp).view(C, co, B, Hp, Wp)
            return zero_pads_(Y)
        return _cpu_conv(X, W)

    @staticmethod
    def backward(ctx, dY: Tensor):
        X, W = ctx.saved_tensors
        C, ci, B, Hp, Wp = X.shape
        co = W.shape[1]
        dY = zero_pads_(dY.contiguous())
        if X.is_cuda:
            ext = _ext()
            Np = B * Hp * Wp
            dYf = dY.reshape(C, co, Np)
            Xf = X.reshape(C, ci, Np)
            dX = None
            if ctx.needs_input_grad[0]:
                # dX = conv(dY, W transposed in (co,ci), taps flipped).
                # Shared weights arrive as a stride-0 expand — transform the
                # single base copy, never materialize C copies.
                if W.stride(0) == 0:
                    Wt = (W[0].permute(1, 0, 2, 3).flip(2, 3).contiguous()
                          .unsqueeze(0).expand(C, ci, co, 3, 3))
                else:
                    Wt = W.permute(0, 2, 1, 3, 4).flip(3, 4).contiguous()
                dX = ext.popconv_fwd(dYf, Wt, B, Hp, Wp).view(X.shape)
                zero_pads_(dX)
            dW = None
            if ctx.needs_input_grad[1]:
                # dW[c,:,:,t] = dY_c @ shift(X_c, Δt)ᵀ — nine shifted
                # strided-batched GEMMs (dY pads are zero, so truncating the
                # contraction range by |Δ| is exact).  rocBLAS runs these at
                # 56-113 TF (benchmarks/microbench_bmm.py); the custom
                # popconv_dw kernel remains for shapes where its launch
                # geometry wins.
                dW = Xf.new_empty(C, co, ci, 3, 3)
                for t in range(9):
                    dyy, dxx = t // 3 - 1, t % 3 - 1
                    delta = dyy * Wp + dxx
                    a = max(0, -delta)
                    b = Np - max(0, delta)
                    tap = torch.bmm(dYf[:, :, a:b],
                                    Xf[:, :, a + delta:b + delta].transpose(1, 2))
                    dW[:, :, :, t // 3, t % 3].copy_(tap)
            return dX, dW
        # CPU fallback via torch.nn.grad per client
        dX = torch.zeros_like(X) if ctx.needs_input_grad[0] else None
        dW = torch.zeros_like(W) if ctx.needs_input_grad[1] else None
        for c in range(C):
            xc = X[c, :, :, 1:-1, 1:-1].permute(1, 0, 2, 3)
            gc = dY[c, :, :, 1:-1, 1:-1].permute(1, 0, 2, 3)
            if dX is not None:
                gx = torch.nn.grad.conv2d_input(xc.shape, W[c], gc, padding=1)
                dX[c, :, :, 1:-1, 1:-1] = gx.permute(1, 0, 2, 3)
            if dW is not None:
                dW[c] = torch.nn.grad.conv2d_weight(xc, W[c].shape, gc,
                                                    padding=1)
        return dX, dW


def pop_conv3x3(X: Tensor, W: Tensor, stride: int = 1) -> Tensor:
    """3x3 population conv, stride 1 or 2 (stride 2 = full conv + subsample,
    re-padded)."""
    Y = _PopConv3x3.apply(X, W)
    if stride == 1:
        return Y
    Ys = Y[..., 1:-1:stride, 1:-1:stride]
    return F.pad(Ys, (1, 1, 1, 1))


def pop_conv1x1(X: Tensor, W: Tensor, stride: int = 1) -> Tensor:
    """1x1 population conv as a batched matmul (library GEMM)."""
    C, ci, B, Hp, Wp = X.shape
    co = W.shape[1]
    xi = X[..., 1:-1:stride, 1:-1:stride]
    h, w = xi.shape[-2:]
    y = torch.matmul(W.reshape(C, co, ci), xi.reshape(C, ci, -1))
    return F.pad(y.view(C, co, B, h, w), (1, 1, 1, 1))


def pop_batchnorm(X: Tensor, weight: Tensor, bias: Tensor,
                  eps: float = 1e-5) -> Tensor:
    """Per-(client, channel) batch-stat normalization on the interior;
    output pads are rebuilt as zeros (F.pad keeps autograd clean)."""
    xi = interior(X)  # [C, ch, B, H, W]
    mu = xi.mean(dim=(2, 3, 4), keepdim=True)
    var = xi.var(dim=(2, 3, 4), unbiased=False, keepdim=True)
    xn = (xi - mu) * torch.rsqrt(var + eps)
    xn = xn * weight.unsqueeze(-1).unsqueeze(-1).unsqueeze(-1) \
        + bias.unsqueeze(-1).unsqueeze(-1).unsqueeze(-1)
    return F.pad(xn, (1, 1, 1, 1))


def _basic_block(x: Tensor, p: Dict[str, Tensor], prefix: str,
                 stride: int, has_ds: bool) -> Tensor:
    out = pop_conv3x3(x, p[f"{prefix}.conv1.weight"], stride)
    out = pop_batchnorm(out, p[f"{prefix}.bn1.weight"], p[f"{prefix}.bn1.bias"])
    out = F.relu(out)
    out = pop_conv3x3(out, p[f"{prefix}.conv2.weight"], 1)
    out = pop_batchnorm(out, p[f"{prefix}.bn2.weight"], p[f"{prefix}.bn2.bias"])
    if has_ds:
        sc = pop_conv1x1(x, p[f"{prefix}.shortcut.0.weight"], stride)
        sc = pop_batchnorm(sc, p[f"{prefix}.shortcut.1.weight"],
                           p[f"{prefix}.shortcut.1.bias"])
    else:
        sc = x
    return F.relu(out + sc)


def pop_resnet18(params: Dict[str, Tensor], x: Tensor) -> Tensor:
    """Population forward of models.resnet.resnet18(norm='batch-local').

    ``params``: name -> [C, *shape] batched parameters (ParamSpec views or
    stride-0 expands of shared θ).  ``x``: [C, B, 3, H, W].  Returns logits
    [C, B, num_classes].
    """
    h = to_pop_layout(x)  # [C, 3, B, H+2, W+2]
    h = pop_conv3x3(h, params["conv1.weight"], 1)
    h = F.relu(pop_batchnorm(h, params["bn1.weight"], params["bn1.bias"]))
    plan = [  # (layer, blocks, stride of first block, downsample in first)
        ("layer1", 2, 1, False),
        ("layer2", 2, 2, True),
        ("layer3", 2, 2, True),
        ("layer4", 2, 2, True),
    ]
    for name, nblocks, stride, ds in plan:
        for i in range(nblocks):
            h = _basic_block(h, params, f"{name}.{i}",
                             stride if i == 0 else 1, ds if i == 0 else False)
    feat = interior(h).mean(dim=(3, 4))          # [C, co, B]
    feat = feat.permute(0, 2, 1)                 # [C, B, co]
    W = params["fc.weight"]                      # [C, 10, co]
    b = params["fc.bias"]                        # [C, 10]
    return torch.matmul(feat, W.transpose(1, 2)) + b.unsqueeze(1)


#: model-class name -> population forward
POPULATION_FORWARDS = {
    "resnet18": pop_resnet18,
}


def population_forward_for(model: torch.nn.Module):
    """Return the population forward fn for a supported model, else None."""
    from blades_amd.models.resnet import ResNet, BasicBlock

    if isinstance(model, ResNet):
        blocks = [len(model.layer1), len(model.layer2), len(model.layer3),
                  len(model.layer4)]
        if blocks == [2, 2, 2, 2] and isinstance(model.layer1[0], BasicBlock):
            # requires batch-stats norm (the engine converts BN anyway)
            return pop_resnet18
    return None
