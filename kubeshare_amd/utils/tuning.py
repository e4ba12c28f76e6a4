"""MIOpen tuning wiring: point MIOPEN_USER_DB_PATH at the tuned gfx950
perf-db shipped in-tree (produced once with MIOPEN_FIND_ENFORCE=SEARCH
on an MI355X; ResNet50 bf16/channels-last step 40.0 -> 33.4 ms, +16% —
profiles/README.md). Must run BEFORE the first convolution; harmless on
CPU-only boxes."""
from __future__ import annotations

import os
import shutil

_UDB_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "ops", "miopen_udb")


def apply_miopen_tuning(writable_copy: bool = True) -> str | None:
    """Set MIOPEN_USER_DB_PATH to the shipped tuned db (a per-process
    /tmp copy by default: MIOpen also WRITES to this path during finds,
    and concurrent pods sharing one file tree would contend on locks).
    Returns the path used, or None if no db is shipped / already set."""
    if os.environ.get("MIOPEN_USER_DB_PATH"):
        return os.environ["MIOPEN_USER_DB_PATH"]
    if not os.path.isdir(_UDB_DIR) or not os.listdir(_UDB_DIR):
        return None
    path = _UDB_DIR
    if writable_copy:
        path = f"/tmp/kubeshare-miopen-{os.getpid()}"
        os.makedirs(path, exist_ok=True)
        for name in os.listdir(_UDB_DIR):
            shutil.copy(os.path.join(_UDB_DIR, name),
                        os.path.join(path, name))
    os.environ["MIOPEN_USER_DB_PATH"] = path
    return path
