"""Offline graph partitioning (filled in as the distributed runtime lands)."""
__all__ = []
