"""Linear probe on frozen features (the im1k-linear protocol of the headline
metric, 83.3% for the ViT-L recipe)."""

from __future__ import annotations

import logging

import torch

logger = logging.getLogger("dinov3")


def evaluate_linear_probe(train_features: torch.Tensor, train_labels: torch.Tensor,
                          test_features: torch.Tensor, test_labels: torch.Tensor,
                          num_classes: int | None = None, epochs: int = 10,
                          lr: float = 0.01, batch_size: int = 1024,
                          device: str = "cpu") -> float:
    """SGD logistic regression on frozen features; returns top-1 accuracy."""
    num_classes = num_classes or int(max(train_labels.max(), test_labels.max()).item()) + 1
    dim = train_features.shape[1]
    clf = torch.nn.Linear(dim, num_classes).to(device)
    opt = torch.optim.SGD(clf.parameters(), lr=lr, momentum=0.9, weight_decay=0.0)
    sched = torch.optim.lr_scheduler.CosineAnnealingLR(
        opt, T_max=max(1, epochs * (len(train_features) // batch_size + 1)))
    tf = train_features.float().to(device)
    tl = train_labels.long().to(device)
    for epoch in range(epochs):
        perm = torch.randperm(tf.shape[0], device=device)
        for i in range(0, tf.shape[0], batch_size):
            idx = perm[i: i + batch_size]
            loss = torch.nn.functional.cross_entropy(clf(tf[idx]), tl[idx])
            opt.zero_grad()
            loss.backward()
            opt.step()
            sched.step()
    with torch.no_grad():
        pred = clf(test_features.float().to(device)).argmax(dim=1).cpu()
    acc = (pred == test_labels.long()).float().mean().item()
    logger.info("linear probe top-1: %.4f", acc)
    return acc
