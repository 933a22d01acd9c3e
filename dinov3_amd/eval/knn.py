"""Feature extraction + k-NN evaluation.

The reference leaves do_test unimplemented (train.py:315-316) while its
configs name knn/linear benchmarks; this implements the im1k-knn protocol the
headline metric quotes (82.2% for the ViT-L recipe): cosine-similarity k-NN
over L2-normalized cls features with temperature-weighted voting.
"""

from __future__ import annotations

import logging
from typing import Tuple

import torch

from .. import parallel

logger = logging.getLogger("dinov3")


@torch.no_grad()
def extract_features(model, data_loader, device=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Run the backbone over a loader of (image, label); returns (features, labels)."""
    device = device or parallel.device()
    model.eval()
    feats, labels = [], []
    for images, targets in data_loader:
        images = images.to(device, non_blocking=True)
        if next(model.parameters()).dtype == torch.bfloat16:
            images = images.bfloat16()
        out = model(images)
        if isinstance(out, dict):
            out = out["x_norm_clstoken"]
        feats.append(out.float().cpu())
        labels.append(torch.as_tensor(targets))
    return torch.cat(feats), torch.cat(labels)


@torch.no_grad()
def evaluate_knn(train_features: torch.Tensor, train_labels: torch.Tensor,
                 test_features: torch.Tensor, test_labels: torch.Tensor,
                 k: int = 20, temperature: float = 0.07, num_classes: int | None = None,
                 chunk: int = 1024) -> float:
    """Temperature-weighted cosine k-NN top-1 accuracy (DINO protocol)."""
    num_classes = num_classes or int(max(train_labels.max(), test_labels.max()).item()) + 1
    train_features = torch.nn.functional.normalize(train_features, dim=1)
    test_features = torch.nn.functional.normalize(test_features, dim=1)
    correct = 0
    total = test_features.shape[0]
    k = min(k, train_features.shape[0])
    for i in range(0, total, chunk):
        tf = test_features[i: i + chunk]
        sim = tf @ train_features.T  # [c, Ntrain]
        topk_sim, topk_idx = sim.topk(k, dim=1)
        topk_labels = train_labels[topk_idx]  # [c, k]
        weights = (topk_sim / temperature).exp()
        votes = torch.zeros(tf.shape[0], num_classes)
        votes.scatter_add_(1, topk_labels, weights)
        pred = votes.argmax(dim=1)
        correct += (pred == test_labels[i: i + chunk]).sum().item()
    acc = correct / max(total, 1)
    logger.info("knn top-1: %.4f (k=%d, T=%.2f)", acc, k, temperature)
    return acc
