import os
import textwrap

import pytest

from lambdipy_amd.resolver import (
    Requirement,
    parse_requirements,
    parse_requirements_file,
    resolve,
)
from lambdipy_amd.resolver.resolve import ResolutionError


def test_parse_basic():
    reqs = parse_requirements("requests==2.31.0\nnumpy>=1.24,<3\n# comment\n\nclick\n")
    names = [r.name for r in reqs]
    assert names == ["requests", "numpy", "click"]
    assert reqs[0].pinned_version == "2.31.0"
    assert reqs[1].pinned_version is None
    assert reqs[2].specifier == ""


def test_parse_extras_markers_case():
    reqs = parse_requirements('Requests[socks]==2.31.0 ; python_version >= "3.6"\n')
    assert reqs[0].name == "requests"  # canonicalized
    assert reqs[0].extras == ("socks",)
    assert "python_version" in reqs[0].marker


def test_parse_skips_pip_options():
    reqs = parse_requirements("--index-url https://x\n-f wheels/\nrequests\n")
    assert [r.name for r in reqs] == ["requests"]


def test_parse_continuation_and_inline_comment():
    reqs = parse_requirements("requests \\\n==2.31.0  # pinned\n")
    assert reqs[0].pinned_version == "2.31.0"


def test_parse_invalid_raises():
    with pytest.raises(ValueError):
        parse_requirements("requests === === bad\n")


def test_parse_file_includes(tmp_path):
    (tmp_path / "base.txt").write_text("click\n")
    (tmp_path / "reqs.txt").write_text("-r base.txt\nrequests\n")
    reqs = parse_requirements_file(str(tmp_path / "reqs.txt"))
    assert [r.name for r in reqs] == ["click", "requests"]


def test_resolve_requests_from_wheelhouse():
    res = resolve(parse_requirements("requests\n"))
    d = res.get("requests")
    assert d is not None and d.source == "wheelhouse"
    assert d.recipe == "requests"
    # transitive closure pulled urllib3/idna/certifi/charset-normalizer
    names = {x.name for x in res.deps}
    assert {"urllib3", "idna", "certifi"} <= names


def test_resolve_installed_fallback():
    # numpy has no wheel in the wheelhouse but is installed in the image
    res = resolve(parse_requirements("numpy\n"), follow_transitive=False)
    d = res.get("numpy")
    assert d is not None
    assert d.recipe == "numpy"


def test_resolve_unsatisfiable():
    with pytest.raises(ResolutionError):
        resolve(parse_requirements("requests==0.0.999\n"), follow_transitive=False)


def test_recipe_split():
    res = resolve(parse_requirements("requests\nclick\n"))
    hit_names = {d.name for d in res.recipe_hits}
    assert "requests" in hit_names
    assert all(d.name != "click" for d in res.recipe_hits)
