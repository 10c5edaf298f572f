"""HIP/CDNA4 op layer: dispatch between gfx950 kernels (ROCm) and the
pure-PyTorch CPU reference implementations. See SURVEY.md §2.4 for the
kernel inventory (K1-K9)."""
from . import extension
from .lstm import lstm_forward
from .pool import concat_pool
from .dropout import variational_dropout
from .crossentropy import tied_decoder_ce, TiedDecoderCE
from .adam import FusedAdamW

__all__ = [
    "extension", "lstm_forward", "concat_pool", "variational_dropout",
    "tied_decoder_ce", "TiedDecoderCE", "FusedAdamW",
]
