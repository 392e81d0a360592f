"""Version watch / auto-update hook.

The reference polls GitHub for a ``__version__`` bump in
template/__init__.py and reclones + pm2-restarts on mismatch
(/root/reference/run_miner.sh:10-11,129-155,233-268 and
hivetrain/utils/auto_update.py:6-64). This environment has no network, so
the offline equivalent watches a version *source* (a file, or the
installed package's ``__version__``) and reports/acts on change; the
restart itself is done by scripts/supervise.sh (DTA_VERSION_WATCH) or by
the callback passed here.
"""

from __future__ import annotations

import logging
import os
import re
import subprocess
import time
from typing import Callable, Optional

log = logging.getLogger(__name__)

_VERSION_RE = re.compile(r"__version__\s*=\s*['\"]([^'\"]+)['\"]")


def read_version(path: str) -> Optional[str]:
    """Extract __version__ from a python file (the reference regex,
    mlflow_utils.py:72-82), or the raw content of a plain version file."""
    if not os.path.exists(path):
        return None
    with open(path) as f:
        text = f.read()
    m = _VERSION_RE.search(text)
    return m.group(1) if m else text.strip() or None


def git_head(repo_dir: str) -> Optional[str]:
    """Current commit SHA — the reference's auto_update compares remote
    SHAs (utils/auto_update.py:6-30)."""
    try:
        out = subprocess.run(["git", "rev-parse", "HEAD"], cwd=repo_dir,
                             capture_output=True, text=True, timeout=10)
        return out.stdout.strip() or None
    except (subprocess.SubprocessError, FileNotFoundError):
        return None


def watch(version_path: str, on_change: Callable[[str, str], None],
          interval_s: float = 60.0, max_iters: Optional[int] = None) -> None:
    """Poll the version source; call on_change(old, new) on a bump.
    ``max_iters`` bounds the loop for tests (None = forever)."""
    last = read_version(version_path)
    i = 0
    while max_iters is None or i < max_iters:
        time.sleep(interval_s)
        cur = read_version(version_path)
        if cur is not None and last is not None and cur != last:
            log.info("version change detected: %s -> %s", last, cur)
            on_change(last, cur)
        if cur is not None:
            last = cur
        i += 1
