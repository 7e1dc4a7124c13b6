"""PyG compatibility layer.

The reference library emits ``torch_geometric.data.Data``/``HeteroData``
batches and leaves modeling to PyG (reference README.md:279-303).  This image
has no PyG, so glt_amd ships a lightweight, API-compatible ``Data``/
``HeteroData`` (attribute-bag semantics, ``.to()``, edge stores) and uses the
real PyG classes transparently when torch_geometric is importable.
"""
try:  # pragma: no cover - exercised only when PyG exists
    from torch_geometric.data import Data, HeteroData  # type: ignore

    HAS_PYG = True
except Exception:  # ModuleNotFoundError and friends
    from .data import Data, HeteroData

    HAS_PYG = False

__all__ = ["Data", "HeteroData", "HAS_PYG"]
