"""Block-wise Hessian eigenvalue estimation via power iteration
(reference: deepspeed/runtime/eigenvalue.py :153 — used by MoQ to schedule
quantization precision by layer sensitivity)."""

import torch

from ..utils.logging import logger


class Eigenvalue:
    def __init__(self, verbose=False, max_iter=100, tol=1e-2, stability=1e-6,
                 gas_boundary_resolution=1, layer_name="", layer_num=0):
        self.verbose = verbose
        self.max_iter = max_iter
        self.tol = tol
        self.stability = stability
        self.gas_boundary_resolution = gas_boundary_resolution
        self.layer_name = layer_name
        self.layer_num = layer_num

    def nan_to_num(self, x):
        return torch.nan_to_num(x, nan=0.0, posinf=1.0, neginf=-1.0)

    def normalize(self, vs):
        norm_sq = sum(v.float().norm() ** 2 for v in vs)
        norm = norm_sq.sqrt() + self.stability
        return [self.nan_to_num(v / norm) for v in vs]

    def compute_eigenvalue(self, module, device=None, scale=1.0):
        """Top Hessian eigenvalue per block via Hv power iteration on the
        existing autograd graph (loss.backward(create_graph=True) must have
        run). Returns one eigenvalue per block_{i} submodule, scaled."""
        block_eigenvalue = []
        for block in module.modules():
            if not getattr(block, "_deepspeed_eigenvalue_block", False):
                continue
            params = [p for p in block.parameters()
                      if p.requires_grad and p.grad is not None and
                      p.grad.grad_fn is not None]
            if not params:
                block_eigenvalue.append(0.0)
                continue
            grads = [p.grad for p in params]
            vs = self.normalize([torch.randn_like(p) for p in params])
            eigenvalue = 0.0
            for i in range(self.max_iter):
                Hv = torch.autograd.grad(grads, params, grad_outputs=vs,
                                         retain_graph=True,
                                         only_inputs=True, allow_unused=True)
                Hv = [self.nan_to_num(h if h is not None else
                                      torch.zeros_like(v))
                      for h, v in zip(Hv, vs)]
                new_ev = float(sum((h * v).sum() for h, v in zip(Hv, vs)))
                vs = self.normalize(Hv)
                if abs(new_ev - eigenvalue) < self.tol * max(abs(new_ev), 1e-9):
                    eigenvalue = new_ev
                    break
                eigenvalue = new_ev
            if self.verbose:
                logger.info(f"eigenvalue {eigenvalue:.4e}")
            block_eigenvalue.append(eigenvalue * scale)
        return block_eigenvalue
