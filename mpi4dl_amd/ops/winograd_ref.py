"""Winograd F(2x2, 3x3) reference implementation (CPU oracle).

Round-2 plan (docs/ROUND2_PLAN.md): MIOpen's 3x3 conv runs at ~318
effective TF on MI355X — almost certainly Winograd's 2.25x FLOP
reduction over direct conv. The round-2 HIP kernel brings the same
transform onto MFMA:

    U = G g G^T          (per (k, c) filter -> 4x4, precomputed on host)
    V = B^T d B          (per input 4x4 tile, stride 2 -> 2x2 outputs)
    M[i][j] = sum_c U[k,c,i,j] * V[c,t,i,j]   (16 independent GEMMs
                                               of shape K x C @ C x T)
    Y = A^T M A          (2x2 output tile)

The 16 element-wise GEMMs are the MFMA-shaped work: K x T output per
(i, j), contracted over C — exactly the bk0mk1 layout conv_fwd_v2
already stages, with C as the k-dimension. Numerical note: bf16 inputs
with fp32 transforms/accumulation keep the error within direct-conv
bf16 tolerance for the filter norms seen in these models (tested).

This module is the NUMERICAL ORACLE for that kernel (tests compare the
HIP output against it), not a production path — production CPU/GPU
dispatch stays on torch/MIOpen/gemscore.

Transform matrices (Lavin & Gray, "Fast Algorithms for Convolutional
Neural Networks", arXiv:1509.09308):

    B^T = [[1,  0, -1,  0],      G = [[1,    0,   0 ],
           [0,  1,  1,  0],           [1/2,  1/2, 1/2],
           [0, -1,  1,  0],           [1/2, -1/2, 1/2],
           [0,  1,  0, -1]]           [0,    0,   1 ]]

    A^T = [[1, 1,  1,  0],
           [0, 1, -1, -1]]
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

BT = torch.tensor(
    [[1.0, 0.0, -1.0, 0.0],
     [0.0, 1.0, 1.0, 0.0],
     [0.0, -1.0, 1.0, 0.0],
     [0.0, 1.0, 0.0, -1.0]]
)
G = torch.tensor(
    [[1.0, 0.0, 0.0],
     [0.5, 0.5, 0.5],
     [0.5, -0.5, 0.5],
     [0.0, 0.0, 1.0]]
)
AT = torch.tensor(
    [[1.0, 1.0, 1.0, 0.0],
     [0.0, 1.0, -1.0, -1.0]]
)


def filter_transform(w: torch.Tensor) -> torch.Tensor:
    """[K, C, 3, 3] -> U [K, C, 4, 4] = G w G^T (host-side, once per step
    in training; the HIP kernel reads U like it reads permuted weights)."""
    assert w.shape[-2:] == (3, 3), w.shape
    g = G.to(device=w.device,
             dtype=w.dtype if w.dtype.is_floating_point else torch.float32)
    return torch.einsum("ir,kcrs,js->kcij", g, w.float(), g)


def winograd_conv2d_ref(
    x: torch.Tensor,
    w: torch.Tensor,
    bias: torch.Tensor = None,
    padding: int = 1,
) -> torch.Tensor:
    """3x3 stride-1 convolution via F(2x2, 3x3), fp32 accumulation.

    Matches F.conv2d(x, w, bias, stride=1, padding=padding) for any
    H, W (odd sizes are tiled by padding up to even and cropping).
    """
    assert w.shape[-2:] == (3, 3), "F(2x2,3x3) is for 3x3 kernels"
    N, C, H, W = x.shape
    K = w.shape[0]
    OH, OW = H + 2 * padding - 2, W + 2 * padding - 2
    # tile grid over outputs, 2x2 per tile
    TH, TW = (OH + 1) // 2, (OW + 1) // 2
    # input span needed: 2*T + 2 in each dim, from -padding
    xin = F.pad(
        x.float(),
        (padding, 2 * TW + 2 - W - padding, padding, 2 * TH + 2 - H - padding),
    )
    # gather 4x4 input tiles at stride 2: d [N, C, TH, TW, 4, 4]
    d = xin.unfold(2, 4, 2).unfold(3, 4, 2)
    bt = BT.to(d.dtype)
    at = AT.to(d.dtype)
    V = torch.einsum("ir,ncturs,js->nctuij", bt, d, bt)
    U = filter_transform(w)
    M = torch.einsum("kcij,nctuij->nktuij", U, V)
    Y = torch.einsum("pi,nktuij,qj->nktupq", at, M, at)
    # [N, K, TH, TW, 2, 2] -> [N, K, 2*TH, 2*TW] -> crop
    out = Y.permute(0, 1, 2, 4, 3, 5).reshape(N, K, 2 * TH, 2 * TW)
    out = out[:, :, :OH, :OW]
    if bias is not None:
        out = out + bias.float().view(1, -1, 1, 1)
    return out


def winograd_bmm_conv2d(
    x: torch.Tensor,
    U: torch.Tensor,
    bias: torch.Tensor = None,
    padding: int = 1,
    out_dtype: torch.dtype = None,
) -> torch.Tensor:
    """GPU-capable F(2x2,3x3) conv: transforms in torch, the 16 GEMMs as
    ONE batched matmul (rides hipBLASLt on ROCm — a plain library GEMM).

    U is the precomputed filter transform (filter_transform(w), shape
    [K, C, 4, 4]) — amortised across steps in inference / recomputed
    once per step in training. The round-2 plan fuses the input/output
    transforms into HIP kernels around the same bmm; this version is the
    drop-in correctness/perf staging point (and the A/B baseline).

    Accumulation: matmul in x.dtype (bf16 on GPU -> MFMA with fp32
    accumulate inside hipBLASLt); transforms in fp32.
    """
    N, C, H, W = x.shape
    K = U.shape[0]
    OH, OW = H + 2 * padding - 2, W + 2 * padding - 2
    TH, TW = (OH + 1) // 2, (OW + 1) // 2
    xin = F.pad(
        x, (padding, 2 * TW + 2 - W - padding, padding, 2 * TH + 2 - H - padding)
    )
    d = xin.unfold(2, 4, 2).unfold(3, 4, 2).float()  # [N,C,TH,TW,4,4]
    bt = BT.to(d.device)
    V = torch.einsum("ir,ncturs,js->ijnctu", bt, d, bt)  # [4,4,N,C,TH,TW]
    T = N * TH * TW
    Vm = V.reshape(16, N, C, TH * TW).permute(0, 2, 1, 3).reshape(16, C, T)
    Um = U.reshape(K, C, 16).permute(2, 0, 1)  # [16, K, C]
    cd = x.dtype if x.is_cuda else torch.float32
    M = torch.bmm(Um.to(cd), Vm.to(cd)).float()  # [16, K, T]
    M = M.reshape(4, 4, K, N, TH, TW)
    at = AT.to(M.device)
    Y = torch.einsum("pi,ijkntu,qj->kntupq", at, M, at)
    out = Y.permute(1, 0, 2, 4, 3, 5).reshape(N, K, 2 * TH, 2 * TW)[:, :, :OH, :OW]
    if bias is not None:
        out = out + bias.float().view(1, -1, 1, 1)
    return out.to(out_dtype) if out_dtype is not None else out
