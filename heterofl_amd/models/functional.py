"""Functional pieces shared by the model family.

masked_cross_entropy implements the label-masked classification loss
(reference: src/models/resnet.py:152-157, src/models/conv.py:66-71): logits of
classes absent from the client's label set are zeroed before cross-entropy.
The fused HIP kernel version lives in heterofl_amd.ops; this is the semantic
definition (and the CPU path).
"""
import torch
import torch.nn.functional as F


def label_mask_fill(score, label_split, num_classes):
    """Zero logits for classes not in label_split.  score: (N, C) or (N, C, S)."""
    if label_split is None:
        return score
    if score.dim() == 3:
        # LM head: (N, vocab, S) masked along vocab
        mask = torch.zeros((num_classes, 1), device=score.device)
        mask[label_split] = 1
    else:
        mask = torch.zeros(num_classes, device=score.device)
        mask[label_split] = 1
    return score.masked_fill(mask == 0, 0)


def masked_cross_entropy(out, label, label_split, num_classes):
    """Returns (score, loss) with the reference's mask-then-CE order."""
    score = label_mask_fill(out, label_split, num_classes)
    loss = F.cross_entropy(score, label)
    return score, loss
