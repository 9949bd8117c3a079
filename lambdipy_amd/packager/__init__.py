from .assemble import BuildReport, assemble_layer
from .prune import prune_tree
from .strip import strip_native_libs
from .zipper import dir_size_bytes, make_zip, zip_report

__all__ = [
    "BuildReport",
    "assemble_layer",
    "prune_tree",
    "strip_native_libs",
    "dir_size_bytes",
    "make_zip",
    "zip_report",
]
