"""Client surface the reconcilers program against.

Mirrors the controller-runtime ``client.Client`` verbs the reference uses
(get/list/create/update/status-update/delete + watch).  The default
implementation wraps :class:`InMemoryStore`; a remote implementation speaking
to a real kube-apiserver can provide the same surface without touching the
controllers.  Tests inject per-verb overrides the same way the reference's
``MyClient`` mock does (suite_test.go:244-294) — see tests/conftest.py.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Type, TypeVar

from ..api.v1alpha1.types import K8sObject
from .store import InMemoryStore, WatchEvent  # noqa: F401

T = TypeVar("T", bound=K8sObject)


class Client:
    def __init__(self, store: InMemoryStore):
        self.store = store

    def create(self, obj: T) -> T:
        return self.store.create(obj)

    def get(self, cls: Type[T], name: str) -> T:
        return self.store.get(cls.KIND, name)

    def try_get(self, cls: Type[T], name: str) -> Optional[T]:
        from .errors import NotFoundError

        try:
            return self.store.get(cls.KIND, name)
        except NotFoundError:
            return None

    def list(
        self,
        cls: Type[T],
        labels: Optional[Dict[str, str]] = None,
        copy: bool = True,
    ) -> List[T]:
        """``copy=False`` = read-only snapshot (no per-object deep copy);
        see InMemoryStore.list. Remote clients ignore the flag (wire
        transfer already isolates)."""
        return self.store.list(cls.KIND, labels, copy=copy)

    def update(self, obj: T) -> T:
        return self.store.update(obj)

    def update_status(self, obj: T) -> T:
        return self.store.update_status(obj)

    def delete(self, obj_or_cls, name: Optional[str] = None) -> None:
        if name is None:
            self.store.delete(obj_or_cls.kind, obj_or_cls.metadata.name)
        else:
            self.store.delete(obj_or_cls.KIND, name)

    def watch(self, kinds: Optional[List[str]] = None):
        return self.store.watch(kinds)
