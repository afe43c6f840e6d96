"""Epilogue-fused GPT MLP: fc1 + GELU + fc2 with hipBLASLt epilogues.

Forward runs fc1 as one hipBLASLt GEMM with the GELU_AUX_BIAS epilogue
(bias + tanh-GELU applied in the epilogue, pre-GELU saved for backward);
backward fuses dGELU and the fc1-bias gradient into fc2's data-grad GEMM
(DGELU_BGRAD epilogue). Deletes the standalone bias+gelu memory passes
(~2x [tokens, ffn] bf16 read+write per block per direction) while
keeping library GEMM throughput — the hand-written MFMA tile loses
end-to-end here (see docs/KERNELS.md "Where library calls remain").

GELU flavor is the tanh approximation (what the hipBLASLt epilogue
implements); the CPU fallback matches it.

RETIRED on this stack (round-2 measurement): ROCm 7.2's hipBLASLt ships
NO algorithm for HIPBLASLT_EPILOGUE_GELU_AUX_BIAS on gfx950 bf16 — the
heuristic returns empty at every probed shape (4096^3, 8192x10240x2560,
32768x10240x2560; gpurun_out/r2_profile epilogue probe). The default
path (library GEMM + aten tanh-GELU, whose backward is a single fused
gelu_backward kernel) therefore stays; METIS_FC1_EPILOGUE=1 raises at
the first hipBLASLt call on current ROCm and is kept only so the route
lights up automatically if a future hipBLASLt adds the algorithms.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from metis_amd import ops as _ops


def _gelu_tanh_grad(pre: torch.Tensor) -> torch.Tensor:
    x = pre.float()
    k = 0.7978845608028654  # sqrt(2/pi)
    inner = k * (x + 0.044715 * x ** 3)
    t = torch.tanh(inner)
    return (0.5 * (1 + t)
            + 0.5 * x * (1 - t * t) * k * (1 + 3 * 0.044715 * x * x)).to(pre.dtype)


class _FusedMLP(torch.autograd.Function):
    """y_partial = gelu_tanh(x @ w1^T + b1) @ w2^T  (fc2 bias and the TP
    g-collective stay outside, matching RowParallelLinear's order)."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2):
        if x.is_cuda:
            ext = _ops.require_extension()
            h, pre = ext.lt_fc1_forward(x, w1, b1)
        else:
            pre = F.linear(x, w1, b1)
            h = F.gelu(pre, approximate="tanh")
        y = h @ w2.t()
        ctx.save_for_backward(x, w1, w2, pre, h)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w1, w2, pre, h = ctx.saved_tensors
        dy = dy.contiguous()
        if dy.is_cuda:
            ext = _ops.require_extension()
            dpre, db1 = ext.lt_matmul_dgelu_bgrad(dy, w2, pre)
        else:
            dpre = (dy @ w2) * _gelu_tanh_grad(pre)
            db1 = dpre.sum(dim=0)
        dw2 = dy.t() @ h
        dx = dpre @ w1
        dw1 = dpre.t() @ x
        return dx, dw1, db1, dw2


def fused_mlp(x: torch.Tensor, w1: torch.Tensor, b1: torch.Tensor,
              w2: torch.Tensor) -> torch.Tensor:
    """x [T, K] -> [T, H]; w1 [F, K] (column-parallel shard), w2 [H, F]
    (row-parallel shard)."""
    return _FusedMLP.apply(x, w1, b1, w2)
