"""adanet_amd version.

Mirrors the reference's version module (reference: adanet/version.py:3).
"""

__version__ = "0.9.0"
