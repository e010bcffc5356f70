from .progen import ProGen, ProGenBase, TransformedProGen

__all__ = ["ProGen", "ProGenBase", "TransformedProGen"]
