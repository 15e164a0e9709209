"""Module-API compatibility shim: ``from model import SAGEDeterministic``
works exactly as with the reference repo (reference model.py:10)."""
from pertgnn.models import SAGEDeterministic, TransformerConv  # noqa: F401
