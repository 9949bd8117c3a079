from .requirements import (Requirement, parse_pipfile, parse_pipfile_lock,
                           parse_project_file, parse_requirements,
                           parse_requirements_file)
from .resolve import ResolvedDep, Resolution, resolve

__all__ = [
    "Requirement",
    "parse_requirements",
    "parse_requirements_file",
    "parse_pipfile",
    "parse_pipfile_lock",
    "parse_project_file",
    "ResolvedDep",
    "Resolution",
    "resolve",
]
