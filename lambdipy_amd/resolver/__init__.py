from .requirements import Requirement, parse_requirements, parse_requirements_file
from .resolve import ResolvedDep, Resolution, resolve

__all__ = [
    "Requirement",
    "parse_requirements",
    "parse_requirements_file",
    "ResolvedDep",
    "Resolution",
    "resolve",
]
