from .dino_clstoken_loss import DINOLoss
from .gram_loss import GramLoss
from .ibot_patch_loss import iBOTPatchLoss
from .koleo_loss import KoLeoLoss, KoLeoLossDistributed

__all__ = ["DINOLoss", "iBOTPatchLoss", "KoLeoLoss", "KoLeoLossDistributed", "GramLoss"]
