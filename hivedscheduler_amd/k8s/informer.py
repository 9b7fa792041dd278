"""List+watch informer (client-go SharedInformer equivalent).

Maintains a local cache of objects, dispatches add/update/delete callbacks,
re-lists on watch expiry (410 Gone) and reconnects with backoff. The
scheduler's recovery depends on the initial LIST delivering every bound pod
before scheduling starts (reference scheduler.go:196-216).
"""
from __future__ import annotations

import logging
import threading
import time
from typing import Callable, Dict, Optional

log = logging.getLogger("hivedscheduler.informer")


class Informer:
    def __init__(
        self,
        client,
        path: str,
        on_add: Callable[[dict], None],
        on_update: Callable[[dict, dict], None],
        on_delete: Callable[[dict], None],
        relist_backoff_s: float = 5.0,
    ):
        self.client = client
        self.path = path
        self.on_add = on_add
        self.on_update = on_update
        self.on_delete = on_delete
        self.relist_backoff_s = relist_backoff_s
        self.cache: Dict[str, dict] = {}  # uid -> object
        self.synced = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    @staticmethod
    def _uid(obj: dict) -> str:
        meta = obj.get("metadata", {})
        return meta.get("uid") or f"{meta.get('namespace', '')}/{meta.get('name', '')}"

    def start(self) -> "Informer":
        self._thread = threading.Thread(target=self._run, name=f"informer{self.path}",
                                        daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()

    def wait_for_cache_sync(self, timeout_s: float = 60.0) -> bool:
        return self.synced.wait(timeout_s)

    def _relist(self) -> str:
        data = self.client.list(self.path)
        rv = data.get("metadata", {}).get("resourceVersion", "0")
        seen = set()
        for obj in data.get("items", []):
            obj.setdefault("apiVersion", data.get("apiVersion"))
            uid = self._uid(obj)
            seen.add(uid)
            old = self.cache.get(uid)
            self.cache[uid] = obj
            if old is None:
                self.on_add(obj)
            else:
                self.on_update(old, obj)
        for uid in list(self.cache):
            if uid not in seen:
                gone = self.cache.pop(uid)
                self.on_delete(gone)
        self.synced.set()
        return rv

    def _run(self) -> None:
        while not self._stop.is_set():
            try:
                rv = self._relist()
                for event in self.client.watch(self.path, rv):
                    if self._stop.is_set():
                        return
                    etype = event.get("type")
                    obj = event.get("object") or {}
                    if etype == "BOOKMARK":
                        rv = obj.get("metadata", {}).get("resourceVersion", rv)
                        continue
                    if etype == "ERROR":
                        log.warning("watch error on %s: %s; relisting", self.path, obj)
                        break
                    uid = self._uid(obj)
                    if etype == "ADDED":
                        old = self.cache.get(uid)
                        self.cache[uid] = obj
                        if old is None:
                            self.on_add(obj)
                        else:
                            self.on_update(old, obj)
                    elif etype == "MODIFIED":
                        old = self.cache.get(uid)
                        self.cache[uid] = obj
                        if old is not None:
                            self.on_update(old, obj)
                        else:
                            self.on_add(obj)
                    elif etype == "DELETED":
                        self.cache.pop(uid, None)
                        self.on_delete(obj)
                    rv = obj.get("metadata", {}).get("resourceVersion", rv)
            except Exception as e:
                if self._stop.is_set():
                    return
                log.warning("informer %s failed: %s; retrying in %.1fs", self.path, e,
                            self.relist_backoff_s)
                time.sleep(self.relist_backoff_s)
