"""Label-smoothing / soft-target cross-entropy
(reference timm/loss/cross_entropy.py:6-36)."""

import torch
import torch.nn as nn
import torch.nn.functional as F


class LabelSmoothingCrossEntropy(nn.Module):
    def __init__(self, smoothing=0.1):
        super().__init__()
        assert smoothing < 1.0
        self.smoothing = smoothing
        self.confidence = 1. - smoothing

    def forward(self, x, target):
        logprobs = F.log_softmax(x, dim=-1)
        nll_loss = -logprobs.gather(dim=-1, index=target.unsqueeze(1)).squeeze(1)
        smooth_loss = -logprobs.mean(dim=-1)
        loss = self.confidence * nll_loss + self.smoothing * smooth_loss
        return loss.mean()


class SoftTargetCrossEntropy(nn.Module):
    def forward(self, x, target):
        loss = torch.sum(-target * F.log_softmax(x, dim=-1), dim=-1)
        return loss.mean()


class FusedCrossEntropy(nn.Module):
    """Plain cross-entropy routed through the fused softmax-xent HIP kernel
    (csrc/softmax_xent.hip) on GPU; eager log_softmax+nll on CPU."""

    def forward(self, x, target):
        from ... import ops
        return ops.cross_entropy(x, target)
