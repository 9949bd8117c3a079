"""requirements.txt parsing.

Reproduces the reference resolver's front end (reads
requirements.txt, produces the pinned dependency set; graph resolution is
delegated rather than reimplemented as a SAT solver). We parse PEP 508-style
requirement lines with ``packaging`` and ignore pip-specific directives that
make no sense offline (index URLs, hashes are accepted but unused).
"""

from __future__ import annotations

import os
import re
from dataclasses import dataclass
from typing import List, Optional

from packaging.requirements import InvalidRequirement
from packaging.requirements import Requirement as _PkgRequirement
from packaging.utils import canonicalize_name


@dataclass
class Requirement:
    """One requirement line: name + version specifier (+ extras/markers)."""

    name: str  # canonicalized (PEP 503) project name
    raw: str  # the original line
    specifier: str = ""  # e.g. "==1.26.4", ">=2,<3", or "" (any)
    extras: tuple = ()
    marker: Optional[str] = None

    @property
    def pinned_version(self) -> Optional[str]:
        """The exact version if the specifier is a single ``==`` pin."""
        m = re.fullmatch(r"==\s*([A-Za-z0-9_.!+*]+)", self.specifier.strip())
        if m and "*" not in m.group(1):
            return m.group(1)
        return None


# pip directives we skip (offline build: no index, no VCS, no URLs)
_SKIP_PREFIXES = (
    "-i",
    "--index-url",
    "--extra-index-url",
    "--no-index",
    "-f",
    "--find-links",
    "--trusted-host",
    "-e",
    "--editable",
    "--no-binary",
    "--only-binary",
    "--prefer-binary",
    "--require-hashes",
    "--pre",
)


def parse_requirements(text: str, *, origin: str = "<string>") -> List[Requirement]:
    """Parse requirements.txt content into Requirement objects.

    Handles comments, line continuations, ``-r`` includes are NOT followed here
    (see parse_requirements_file), and pip option lines are skipped.
    """
    reqs: List[Requirement] = []
    # join line continuations
    logical: List[str] = []
    pending = ""
    for line in text.splitlines():
        line = line.rstrip()
        if line.endswith("\\"):
            pending += line[:-1]
            continue
        logical.append(pending + line)
        pending = ""
    if pending:
        logical.append(pending)

    for lineno, line in enumerate(logical, 1):
        # strip comments (a ' #' or leading '#')
        line = re.sub(r"(^|\s)#.*$", "", line).strip()
        if not line:
            continue
        if line.startswith(("-r ", "--requirement ", "-c ", "--constraint ")):
            # includes handled by parse_requirements_file
            reqs.append(Requirement(name="", raw=line))
            continue
        if any(line == p or line.startswith(p + " ") or line.startswith(p + "=") for p in _SKIP_PREFIXES):
            continue
        try:
            pr = _PkgRequirement(line)
        except InvalidRequirement as e:
            raise ValueError(f"{origin}:{lineno}: invalid requirement {line!r}: {e}") from None
        reqs.append(
            Requirement(
                name=canonicalize_name(pr.name),
                raw=line,
                specifier=str(pr.specifier),
                extras=tuple(sorted(pr.extras)),
                marker=str(pr.marker) if pr.marker else None,
            )
        )
    return reqs


def parse_pipfile_lock(path: str, *, dev: bool = False) -> List[Requirement]:
    """Parse a Pipfile.lock (the reference reads
    Pipfile(.lock) as an alternative front end). The lock file is already
    fully pinned, so entries map 1:1 onto Requirement objects; the
    ``develop`` section is excluded unless dev=True (build layers ship
    runtime deps only, like the reference)."""
    import json

    with open(path) as f:
        data = json.load(f)
    reqs: List[Requirement] = []
    sections = ["default"] + (["develop"] if dev else [])
    for section in sections:
        for name, spec in (data.get(section) or {}).items():
            if not isinstance(spec, dict):
                spec = {"version": spec}
            ver = spec.get("version") or ""
            if ver == "*":
                ver = ""
            reqs.append(Requirement(
                name=canonicalize_name(name),
                raw=f"{name}{ver}",
                specifier=ver,
                extras=tuple(sorted(spec.get("extras") or ())),
                marker=spec.get("markers"),
            ))
    return reqs


def parse_pipfile(path: str, *, dev: bool = False) -> List[Requirement]:
    """Parse a Pipfile (TOML). Entries are either version strings
    (``numpy = "*"``) or tables (``{version = ">=1.0", markers = "..."}``)."""
    import tomli

    with open(path, "rb") as f:
        data = tomli.load(f)
    reqs: List[Requirement] = []
    sections = ["packages"] + (["dev-packages"] if dev else [])
    for section in sections:
        for name, spec in (data.get(section) or {}).items():
            if isinstance(spec, str):
                spec = {"version": spec}
            ver = spec.get("version") or ""
            if ver == "*":
                ver = ""
            reqs.append(Requirement(
                name=canonicalize_name(name),
                raw=f"{name}{ver}",
                specifier=ver,
                extras=tuple(sorted(spec.get("extras") or ())),
                marker=spec.get("markers"),
            ))
    return reqs


def parse_project_file(path: str) -> List[Requirement]:
    """Front-door dispatch on the project file type: requirements.txt,
    Pipfile.lock (preferred over Pipfile when both exist, like the
    reference's pipenv flow) or Pipfile."""
    base = os.path.basename(path)
    if base == "Pipfile.lock":
        return parse_pipfile_lock(path)
    if base == "Pipfile":
        lock = os.path.join(os.path.dirname(path) or ".", "Pipfile.lock")
        if os.path.exists(lock):
            return parse_pipfile_lock(lock)
        return parse_pipfile(path)
    return parse_requirements_file(path)


def parse_requirements_file(path: str, _seen: Optional[set] = None) -> List[Requirement]:
    """Parse a requirements file, following ``-r`` includes relative to it."""
    _seen = _seen if _seen is not None else set()
    ap = os.path.abspath(path)
    if ap in _seen:
        return []
    _seen.add(ap)
    with open(ap, "r") as f:
        text = f.read()
    out: List[Requirement] = []
    for r in parse_requirements(text, origin=path):
        if r.name == "" and r.raw.startswith(("-r ", "--requirement ")):
            inc = r.raw.split(None, 1)[1].strip()
            out.extend(parse_requirements_file(os.path.join(os.path.dirname(ap), inc), _seen))
        elif r.name == "":
            continue  # constraints: not supported offline, skipped
        else:
            out.append(r)
    return out
