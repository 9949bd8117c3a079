import pytest

from lambdipy_amd.recipes import BuildRecipe, RecipeError, get_registry


def test_registry_loads_core_set():
    reg = get_registry()
    assert {"requests", "numpy", "scipy", "scikit-learn", "torch"} <= set(reg.names())


def test_version_matching():
    reg = get_registry()
    assert reg.match("numpy", "1.26.4") is not None
    assert reg.match("numpy", "0.9") is None
    assert reg.match("nonexistent", "1.0") is None


def test_native_recipes_have_kernels_and_shims():
    reg = get_registry()
    for name in ("numpy", "scipy", "scikit-learn", "torch"):
        r = reg.get(name)
        assert r.kind == "native"
        assert r.kernels, name
        assert r.shim, name


def test_recipe_schema_validation():
    with pytest.raises(RecipeError):
        BuildRecipe.from_dict({"recipe_format": 1, "name": "x", "versions": ">=1"})
    with pytest.raises(RecipeError):
        BuildRecipe.from_dict(
            {"recipe_format": 99, "name": "x", "versions": ">=1", "kind": "pure"}
        )
    with pytest.raises(RecipeError):
        BuildRecipe.from_dict(
            {"recipe_format": 1, "name": "x", "versions": ">=1", "kind": "weird"}
        )
    # native WITH a shim but without kernels (shim needs ops to route to);
    # shimless native recipes (strip/prune-only, e.g. pandas) are legal
    with pytest.raises(RecipeError):
        BuildRecipe.from_dict(
            {"recipe_format": 1, "name": "x", "versions": ">=1",
             "kind": "native", "shim": "lpx_x"}
        )
    BuildRecipe.from_dict(
        {"recipe_format": 1, "name": "x", "versions": ">=1", "kind": "native"}
    )


def test_provides_defaults_to_name():
    r = BuildRecipe.from_dict(
        {"recipe_format": 1, "name": "Foo_Bar", "versions": ">=1", "kind": "pure"}
    )
    assert r.name == "foo-bar"
    assert r.provides == ["foo-bar"]


def test_provides_alias_matching():
    """A pin on a provides alias (psycopg2-binary, PIL) resolves to the
    canonical recipe — SURVEY.md §2.1 R3 provides/excludes info."""
    from lambdipy_amd.recipes import RecipeRegistry

    reg = RecipeRegistry.load()
    assert reg.match("psycopg2-binary", "2.9.9").name == "psycopg2"
    assert reg.match("PIL", "10.0.0").name == "pillow"
    assert reg.match("Pillow", "10.4.0").name == "pillow"
    # non-alias misses stay None
    assert reg.match("definitely-not-a-recipe", "1.0") is None
