"""Service discovery helper (mirrors reference persia/service.py).

In the collapsed single-node architecture "embedding worker services" are the
trainer ranks themselves; this returns the TCP batch-queue endpoints used by
separate data-loader processes, from the ``EMBEDDING_WORKER_SERVICE`` env."""
import os
from typing import List


def get_embedding_worker_services() -> List[str]:
    services = os.environ.get("EMBEDDING_WORKER_SERVICE", "")
    return [s for s in services.split(",") if s]
