"""progen_amd — MI355X-native ProGen protein language model framework.

Public API parity with the reference (reference: progen_transformer/__init__.py:1):

    from progen_amd import ProGen
"""

from .models.progen import ProGen, ProGenBase, TransformedProGen
from .config import ProGenConfig

__all__ = ["ProGen", "ProGenBase", "TransformedProGen", "ProGenConfig"]
__version__ = "0.1.0"
