"""Shared-memory object store: the Ray plasma-store equivalent.

The reference parks dataset shards and the prediction model in Ray's C++
object store (``ray.put``/``ray.get``, reference matrix.py:467-487,
main.py:1790). On a single 8xMI355X node the equivalent is POSIX shared
memory: ``put`` writes a numpy array (zero-copy mmap) or a pickled object
into a ``multiprocessing.shared_memory`` segment; actors in other
processes ``get`` it without a second copy of the bytes over a pipe.

Refs are plain picklable descriptors, so they travel through actor RPC
exactly like Ray ObjectRefs travel through remote calls.
"""

import pickle
import uuid
from dataclasses import dataclass
from multiprocessing import shared_memory
from typing import Any, List, Optional

import numpy as np


@dataclass
class ObjectRef:
    shm_name: str
    kind: str  # "ndarray" | "pickle"
    dtype: Optional[str] = None
    shape: Optional[tuple] = None
    nbytes: int = 0

    def __hash__(self):
        return hash(self.shm_name)


class ShmStore:
    """Process-local handle table; segments are system-global by name."""

    def __init__(self):
        self._owned: List[shared_memory.SharedMemory] = []
        self._attached = {}

    def put(self, obj: Any) -> ObjectRef:
        if isinstance(obj, np.ndarray) and obj.dtype != object:
            arr = np.ascontiguousarray(obj)
            nbytes = max(arr.nbytes, 1)
            shm = shared_memory.SharedMemory(
                create=True, size=nbytes, name=f"rxgb_{uuid.uuid4().hex[:16]}"
            )
            if arr.nbytes:
                dst = np.ndarray(arr.shape, dtype=arr.dtype, buffer=shm.buf)
                dst[...] = arr
            self._owned.append(shm)
            return ObjectRef(
                shm_name=shm.name,
                kind="ndarray",
                dtype=str(arr.dtype),
                shape=tuple(arr.shape),
                nbytes=arr.nbytes,
            )
        payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
        shm = shared_memory.SharedMemory(
            create=True, size=max(len(payload), 1), name=f"rxgb_{uuid.uuid4().hex[:16]}"
        )
        shm.buf[: len(payload)] = payload
        self._owned.append(shm)
        return ObjectRef(shm_name=shm.name, kind="pickle", nbytes=len(payload))

    def get(self, ref: ObjectRef, copy: bool = False) -> Any:
        if ref.shm_name in self._attached:
            shm = self._attached[ref.shm_name]
        else:
            shm = shared_memory.SharedMemory(name=ref.shm_name)
            self._attached[ref.shm_name] = shm
        if ref.kind == "ndarray":
            arr = np.ndarray(ref.shape, dtype=np.dtype(ref.dtype), buffer=shm.buf)
            if copy:
                return arr.copy()
            arr.flags.writeable = False
            return arr
        return pickle.loads(bytes(shm.buf[: ref.nbytes]))

    def detach(self):
        for shm in self._attached.values():
            try:
                shm.close()
            except Exception:
                pass
        self._attached = {}

    def free(self, ref: ObjectRef):
        """Unlink one owned segment (matrix unload_data path)."""
        for i, shm in enumerate(self._owned):
            if shm.name == ref.shm_name:
                try:
                    shm.close()
                    shm.unlink()
                except Exception:
                    pass
                del self._owned[i]
                return

    def shutdown(self):
        """Owner-side cleanup: unlink all segments this store created."""
        self.detach()
        for shm in self._owned:
            try:
                shm.close()
                shm.unlink()
            except Exception:
                pass
        self._owned = []


_GLOBAL_STORE: Optional[ShmStore] = None


def get_store() -> ShmStore:
    global _GLOBAL_STORE
    if _GLOBAL_STORE is None:
        _GLOBAL_STORE = ShmStore()
        import atexit

        atexit.register(_GLOBAL_STORE.shutdown)
    return _GLOBAL_STORE


def put(obj) -> ObjectRef:
    return get_store().put(obj)


def get(ref: ObjectRef, copy: bool = False):
    if isinstance(ref, ObjectRef):
        return get_store().get(ref, copy=copy)
    return ref
