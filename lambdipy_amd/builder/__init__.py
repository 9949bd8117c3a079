from .wheel_builder import build_recipe_artifacts, ops_extension_path

__all__ = ["build_recipe_artifacts", "ops_extension_path"]
