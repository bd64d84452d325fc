"""Compat shim: the reference's Overview notebook names the future type as
``ray._raylet.ObjectRef`` (Overview_of_Ray.ipynb cell-34 prints it); our
futures come from the mi355 runtime."""
from mi355.runtime.core import ObjectRef

__all__ = ["ObjectRef"]
