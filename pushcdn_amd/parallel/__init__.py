"""Distributed mesh communication (RCCL over xGMI)."""

from .mesh import RcclMesh  # noqa: F401
