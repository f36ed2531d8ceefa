"""RMSNorm / LayerNorm modules backed by the CDNA4 HIP kernels.

Kernel parity targets: reference rms_norm.cu / layer_norm.cu (inference) and
normalize_kernels.cu (training fwd+bwd). CPU falls back to a numerically
matching torch implementation (fp32 math internally, like the kernels).
"""

import torch

from ._loader import get_ext


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = get_ext()
        x = x.contiguous()
        if ext is not None and x.is_cuda:
            y, invrms, _ = ext.norm_fwd(x, weight.contiguous(), None, eps, False)
        else:
            xf = x.float()
            invrms = torch.rsqrt(xf.pow(2).mean(-1) + eps)
            y = (xf * invrms.unsqueeze(-1) * weight.float()).to(x.dtype)
            invrms = invrms.reshape(-1)
        ctx.save_for_backward(x, weight, invrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, invrms = ctx.saved_tensors
        ext = get_ext()
        dy = dy.contiguous()
        if ext is not None and x.is_cuda:
            dx, dw, _ = ext.norm_bwd(dy, x, weight.contiguous(), invrms, None,
                                     False)
        else:
            H = x.shape[-1]
            xf = x.float().reshape(-1, H)
            dyf = dy.float().reshape(-1, H)
            inv = invrms.unsqueeze(-1)
            xhat = xf * inv
            dyw = dyf * weight.float()
            dot = (dyw * xhat).mean(-1, keepdim=True)
            dx = ((dyw - xhat * dot) * inv).reshape(x.shape).to(x.dtype)
            dw = (dyf * xhat).sum(0)
        return dx, dw.to(weight.dtype), None


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = get_ext()
        x = x.contiguous()
        if ext is not None and x.is_cuda:
            y, invrms, mean = ext.norm_fwd(
                x, weight.contiguous(),
                bias.contiguous() if bias is not None else None, eps, True)
        else:
            xf = x.float()
            mean = xf.mean(-1)
            var = xf.var(-1, unbiased=False)
            invrms = torch.rsqrt(var + eps)
            y = ((xf - mean.unsqueeze(-1)) * invrms.unsqueeze(-1) *
                 weight.float())
            if bias is not None:
                y = y + bias.float()
            y = y.to(x.dtype)
            invrms, mean = invrms.reshape(-1), mean.reshape(-1)
        ctx.save_for_backward(x, weight, invrms, mean)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, invrms, mean = ctx.saved_tensors
        ext = get_ext()
        dy = dy.contiguous()
        if ext is not None and x.is_cuda:
            dx, dw, db = ext.norm_bwd(dy, x, weight.contiguous(), invrms, mean,
                                      True)
        else:
            H = x.shape[-1]
            xf = x.float().reshape(-1, H)
            dyf = dy.float().reshape(-1, H)
            inv = invrms.unsqueeze(-1)
            xhat = (xf - mean.unsqueeze(-1)) * inv
            dyw = dyf * weight.float()
            dot = (dyw * xhat).mean(-1, keepdim=True)
            dsum = dyw.mean(-1, keepdim=True)
            dx = ((dyw - dsum - xhat * dot) * inv).reshape(x.shape).to(x.dtype)
            dw = (dyf * xhat).sum(0)
            db = dyf.sum(0)
        grad_bias = db.to(weight.dtype) if ctx.has_bias else None
        return dx, dw.to(weight.dtype), grad_bias, None


def rms_norm(x, weight, eps=1e-6):
    return _RMSNormFn.apply(x, weight, eps)


def layer_norm(x, weight, bias=None, eps=1e-5):
    return _LayerNormFn.apply(x, weight, bias, eps)


class RMSNorm(torch.nn.Module):
    def __init__(self, hidden_size, eps=1e-6, dtype=None, device=None):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.ones(hidden_size, dtype=dtype, device=device))
        self.eps = eps

    def reset_parameters(self):
        torch.nn.init.ones_(self.weight)

    def forward(self, x):
        return _RMSNormFn.apply(x, self.weight, self.eps)

    def extra_repr(self):
        return f"{self.weight.numel()}, eps={self.eps}"


class FusedLayerNorm(torch.nn.Module):
    def __init__(self, hidden_size, eps=1e-5, bias=True, dtype=None, device=None):
        super().__init__()
        self.weight = torch.nn.Parameter(
            torch.ones(hidden_size, dtype=dtype, device=device))
        self.bias = torch.nn.Parameter(
            torch.zeros(hidden_size, dtype=dtype, device=device)) if bias else None
        self.eps = eps

    def reset_parameters(self):
        torch.nn.init.ones_(self.weight)
        if self.bias is not None:
            torch.nn.init.zeros_(self.bias)

    def forward(self, x):
        return _LayerNormFn.apply(x, self.weight, self.bias, self.eps)
