from .bleu import bleu, smoothed_bleu4
from .calc_code_bleu import calc_code_bleu

__all__ = ["bleu", "smoothed_bleu4", "calc_code_bleu"]
