# -*- coding: utf-8 -*-
"""Git-derived version info (in-house versioneer equivalent).

The reference vendors versioneer (~70k LoC of boilerplate) to stamp
``__version__`` from git tags (``stoke/_version.py`` + ``versioneer.py``,
used at ``stoke/__init__.py:45-47``).  This 40-line equivalent keeps the
same ``get_versions()`` contract: a dict with "version", "full-revisionid",
"dirty" and "error" keys, derived from ``git describe`` when the package
runs from a checkout, falling back to the static release version.
"""

import os
import subprocess

_FALLBACK = "0.2.0"


def get_versions() -> dict:
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    info = {"version": _FALLBACK, "full-revisionid": None, "dirty": None,
            "error": None, "date": None}
    if not os.path.isdir(os.path.join(root, ".git")):
        info["error"] = "not a git checkout"
        return info
    def _git(*args):
        return subprocess.run(
            ["git", *args], cwd=root, capture_output=True, text=True,
            timeout=5,
        ).stdout.strip()
    try:
        rev = _git("rev-parse", "HEAD")
        if rev:
            info["full-revisionid"] = rev
        describe = _git("describe", "--tags", "--dirty", "--always")
        if describe:
            info["dirty"] = describe.endswith("-dirty")
            tag = describe[:-6] if info["dirty"] else describe
            # "v1.2.3-4-gabcdef" -> "1.2.3+4.gabcdef"; bare hash -> fallback+hash
            if "-g" in tag:
                base, n, g = tag.rsplit("-", 2)
                info["version"] = f"{base.lstrip('v')}+{n}.{g}"
            elif tag == rev[: len(tag)]:
                info["version"] = f"{_FALLBACK}+g{tag}"
            else:
                info["version"] = tag.lstrip("v")
            if info["dirty"]:
                info["version"] += ".dirty"
        date = _git("show", "-s", "--format=%ci", "HEAD")
        if date:
            info["date"] = date
    except (OSError, subprocess.TimeoutExpired) as exc:  # pragma: no cover
        info["error"] = str(exc)
    return info
