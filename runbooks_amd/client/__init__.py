"""Client library for the CLI / kubectl plugins.

Parity: reference internal/client/ — tarball build + md5 (upload.go:38-68,
209-281), server-side apply + the signed-URL upload handshake
(upload.go:110-192), WaitReady polling (client.go:114-135), Notebook
conversion (notebook.go:20-86), file sync driven by the in-pod nbwatch
agent (sync.go:28-135), and port-forwarding (port_forward.go:21-46).
"""
from . import sync
from .notebook import notebook_for_object, pod_for_notebook
from .sync import port_forward, sync_files_from_notebook
from .upload import (
    Tarball,
    clear_image,
    prepare_image_tarball,
    set_upload_container_spec,
    upload,
    wait_ready,
)

__all__ = [
    "Tarball", "prepare_image_tarball", "set_upload_container_spec",
    "clear_image", "upload", "wait_ready",
    "notebook_for_object", "pod_for_notebook",
    "sync_files_from_notebook", "port_forward", "sync",
]
