"""Similarity matrices, top-k stats, and image-complexity metrics.

Capability parity: /root/reference/diff_retrieval.py:388-483 (dense GEMM
similarity + stats), :393-400,643-662 (chunked patch-wise einsum),
:113-121 (tv_loss), :498-528 (GLCM entropy / JPEG size complexity,
skimage/cv2-free here). GEMMs go through rocBLAS (torch.mm) — the
BASELINE-sanctioned library path; everything runs on CPU too (config 1).
"""
from __future__ import annotations

import io
from typing import Dict, Optional, Tuple

import numpy as np
import torch
import torch.nn.functional as F


def l2_normalize(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
    return F.normalize(x, dim=dim)


def sim_matrix(a: torch.Tensor, b: torch.Tensor,
               chunk: int = 8192) -> torch.Tensor:
    """a [N, D] @ b [M, D]^T -> [N, M] in fp32, row-chunked for memory."""
    a = a.float()
    b = b.float()
    if a.shape[0] <= chunk:
        return a @ b.t()
    out = torch.empty(a.shape[0], b.shape[0], device=a.device)
    for i in range(0, a.shape[0], chunk):
        out[i:i + chunk] = a[i:i + chunk] @ b.t()
    return out


def einsum_in_chunks(feat_a: torch.Tensor, feat_b: torch.Tensor,
                     chunk: int = 64, stype: str = "cross") -> torch.Tensor:
    """Patch-wise 'splitloss' similarity; feats are [N, C, P] per-patch
    descriptors. stype='cross' takes the max over all patch PAIRS
    (reference's einsum_in_chunks intent, diff_retrieval.py:643-662,
    'ncp,mdp->nmcd' in its layout); stype='' matches the reference's live
    splitloss path (diff_retrieval.py:397-400): same-patch dot, max over
    patches."""
    sims = []
    for i in range(0, feat_a.shape[0], chunk):
        a = feat_a[i:i + chunk].float()
        if stype == "cross":
            s = torch.einsum("ncp,mcq->nmpq", a, feat_b.float()).amax(dim=(2, 3))
        else:
            s = torch.einsum("ncp,mcp->nmp", a, feat_b.float()).amax(dim=2)
        sims.append(s)
    return torch.cat(sims, dim=0)


def topk_stats(sim_gen_train: torch.Tensor, sim_train_train: Optional[torch.Tensor] = None,
               threshold: float = 0.5) -> Dict[str, float]:
    """Reference metric schema (diff_retrieval.py:417-483 / wandb keys):
    top-1 gen→train similarity stats + train-train background (top-2,
    self-match dropped) + fraction over the copy threshold."""
    top1 = sim_gen_train.max(dim=1).values.float()
    out = {
        "sim_mean": top1.mean().item(),
        "sim_std": top1.std().item(),
        "sim_75pc": top1.quantile(0.75).item(),
        "sim_90pc": top1.quantile(0.90).item(),
        "sim_95pc": top1.quantile(0.95).item(),
        "sim_gt_05pc": (top1 > threshold).float().mean().item(),
    }
    if sim_train_train is not None:
        # top-2 because top-1 is the self-match (diff_retrieval.py:419);
        # 'bg_*' = train-train background, reference wandb key names
        # (diff_retrieval.py:456-468)
        t2 = sim_train_train.topk(2, dim=1).values[:, 1].float()
        out.update({
            "bg_mean": t2.mean().item(),
            "bg_std": t2.std().item(),
            "bg_75pc": t2.quantile(0.75).item(),
            "bg_90pc": t2.quantile(0.90).item(),
            "bg_95pc": t2.quantile(0.95).item(),
        })
    return out


def top_matches(sim: torch.Tensor, k: int = 10) -> Tuple[torch.Tensor, torch.Tensor]:
    """argsort top-k train matches per generation (gallery plots)."""
    vals, idx = sim.topk(k, dim=1)
    return vals, idx


def similarity_histogram(top1: torch.Tensor, bins: int = 50,
                         range_=(0.0, 1.0)) -> Tuple[np.ndarray, np.ndarray]:
    return np.histogram(top1.cpu().numpy(), bins=bins, range=range_)


# ------------------------------------------------------------- complexity
def tv_loss(img: torch.Tensor) -> torch.Tensor:
    """Total variation (reference diff_retrieval.py:113-121)."""
    if img.dim() == 3:
        img = img[None]
    bs, _, h, w = img.shape
    tv_h = (img[:, :, 1:, :] - img[:, :, :-1, :]).pow(2).sum()
    tv_w = (img[:, :, :, 1:] - img[:, :, :, :-1]).pow(2).sum()
    return (tv_h + tv_w) / (bs * img.shape[1] * h * w)


def glcm_entropy(img: np.ndarray, distance: int = 1, levels: int = 256) -> float:
    """Gray-level co-occurrence entropy (reference uses
    skimage.feature.graycomatrix at diff_retrieval.py:~500; skimage is not
    installed here so the co-occurrence histogram is computed directly,
    horizontal offset, symmetric+normed)."""
    if img.ndim == 3:
        img = (0.299 * img[..., 0] + 0.587 * img[..., 1] + 0.114 * img[..., 2])
    img = img.astype(np.int64)
    if img.max() > levels - 1:
        img = img * (levels - 1) // max(1, img.max())
    a = img[:, :-distance].ravel()
    b = img[:, distance:].ravel()
    idx = a * levels + b
    counts = np.bincount(idx, minlength=levels * levels).astype(np.float64)
    counts += np.bincount(b * levels + a, minlength=levels * levels)  # symmetric
    p = counts / counts.sum()
    nz = p[p > 0]
    return float(-(nz * np.log2(nz)).sum())


def jpeg_size(pil_img, quality: int = 95) -> int:
    """JPEG-encoded byte size (compression complexity proxy; reference
    uses cv2.imencode)."""
    buf = io.BytesIO()
    pil_img.save(buf, format="JPEG", quality=quality)
    return buf.tell()


def pearson(x: np.ndarray, y: np.ndarray) -> float:
    return pearson_with_p(x, y)[0]


def pearson_with_p(x: np.ndarray, y: np.ndarray) -> Tuple[float, float]:
    from scipy.stats import pearsonr
    r, p = pearsonr(np.asarray(x, dtype=np.float64), np.asarray(y, dtype=np.float64))
    return float(r), float(p)
