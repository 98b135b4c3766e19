"""Segmentation losses (parity: reference metrics/loss.py:1-23)."""


def dice_loss_binary(outputs, target, beta=1.0, weights=None, eps=1e-5):
    """Weighted soft-dice loss in beta-F-score form for binary segmentation.

    outputs: probabilities in [0,1], same shape as target.
    """
    outputs = outputs.reshape(-1).float()
    target = target.reshape(-1).float()
    if weights is not None:
        w = weights.reshape(-1).float()
        outputs = outputs * w
        target = target * w
    b2 = beta * beta
    intersection = (outputs * target).sum()
    num = (1 + b2) * intersection
    den = b2 * target.sum() + outputs.sum()
    return 1.0 - (num + eps) / (den + eps)
