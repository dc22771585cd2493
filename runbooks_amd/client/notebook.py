"""Model/Server/Dataset -> Notebook conversion for the dev loop.

Parity: reference internal/client/notebook.go:12-90.
"""
from __future__ import annotations

from ..api.types import Dataset, Model, Notebook, Server


def pod_for_notebook(nb: Notebook) -> tuple[str, str]:
    """(namespace, pod name) of the notebook pod."""
    return nb.namespace, f"{nb.name}-notebook"


def notebook_for_object(obj) -> Notebook:
    if isinstance(obj, Notebook):
        return obj
    if isinstance(obj, Model):
        nb = Notebook(name=f"{obj.name}-model", namespace=obj.namespace,
                      image=obj.image, env=dict(obj.env),
                      params=dict(obj.params), model=obj.model,
                      dataset=obj.dataset, resources=obj.resources)
    elif isinstance(obj, Server):
        nb = Notebook(name=f"{obj.name}-server", namespace=obj.namespace,
                      image=obj.image, env=dict(obj.env),
                      params=dict(obj.params), model=obj.model,
                      resources=obj.resources)
    elif isinstance(obj, Dataset):
        nb = Notebook(name=f"{obj.name}-dataset", namespace=obj.namespace,
                      image=obj.image, env=dict(obj.env),
                      params=dict(obj.params), resources=obj.resources)
    else:
        raise TypeError(f"unknown object type: {type(obj)}")
    nb.build = obj.build
    return nb
