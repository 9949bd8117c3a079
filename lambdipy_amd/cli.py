"""lpx — the lambdipy-shaped CLI.

Commands mirror the reference's surface (build/publish — SURVEY.md §2.1 R1)
plus the MI355X runtime half (run/bench). Config surface, like the
reference's, is three tiers: CLI flags, env (LPX_*), recipe JSON (N3).
"""

from __future__ import annotations

import json
import os
import shutil
import sys

import click

from .utils.log import get_logger

log = get_logger("cli")

DEFAULT_RELEASE_DIR = os.environ.get(
    "LPX_RELEASES", os.path.expanduser("~/.lpx/releases")
)


def _install_handlers(build_dir: str, handlers):
    """Copy handler modules + fan-out support into the layer."""
    pkg = os.path.dirname(os.path.abspath(__file__))
    hsrc = os.path.join(pkg, "runtime", "handlers")
    dest = os.path.join(build_dir, "lpx_handlers")
    os.makedirs(dest, exist_ok=True)
    with open(os.path.join(dest, "__init__.py"), "w") as f:
        f.write("# lambdipy_amd layer handlers\n")
    from .runtime.coldstart import HANDLERS

    for h in handlers:
        mod = HANDLERS.get(h, h)
        src = os.path.join(hsrc, mod + ".py")
        if not os.path.exists(src):
            raise click.ClickException(f"unknown handler {h!r}")
        shutil.copy2(src, os.path.join(dest, mod + ".py"))
    # fan-out + torch-free runtime support ship as top-level modules
    shutil.copy2(os.path.join(pkg, "parallel", "fanout.py"),
                 os.path.join(build_dir, "lpx_fanout.py"))
    shutil.copy2(os.path.join(pkg, "shims", "lpx_rt.py"),
                 os.path.join(build_dir, "lpx_rt.py"))
    shutil.copy2(os.path.join(pkg, "shims", "lpx_nn.py"),
                 os.path.join(build_dir, "lpx_nn.py"))
    # NN lite handlers dlopen liblpx_nn (AOT gfx950): ship it unless the
    # layer's lpx_ops wheel already carries it
    needs_nn = any(h in ("matmul_lite", "fftspmv_lite", "resnet50_lite")
                   for h in handlers)
    in_ops = os.path.exists(os.path.join(build_dir, "lpx_ops",
                                         "liblpx_nn_gfx950.so"))
    if needs_nn and not in_ops:
        from .builder.wheel_builder import ensure_nn_lib

        lib = ensure_nn_lib()
        shutil.copy2(lib, os.path.join(build_dir, os.path.basename(lib)))
    # rf_lite hipModuleLoads the standalone .hsaco — same rule (a handlers-
    # only layer must be self-contained; real recipe layers get it via the
    # lpx_ops wheel)
    if any(h in ("rf_lite", "rf") for h in handlers) and not os.path.exists(
            os.path.join(build_dir, "lpx_ops", "standalone_gfx950.hsaco")):
        from .builder.wheel_builder import ensure_standalone_hsaco

        hs = ensure_standalone_hsaco()
        shutil.copy2(hs, os.path.join(build_dir, os.path.basename(hs)))


@click.group()
def main():
    """MI355X-native scientific-Python packager + runtime."""


@main.command()
@click.option("-r", "--requirements", default=None,
              help="project file (requirements.txt, Pipfile or Pipfile.lock; "
                   "default: auto-detect in cwd)")
@click.option("-o", "--output", default="build", help="layer output directory")
@click.option("-z", "--zip", "zip_path", default=None,
              help="also produce a zip artifact at this path")
@click.option("--handler", "handlers", multiple=True,
              help="handler(s) to ship in the layer (rf, matmul, fftspmv, requests, resnet50)")
@click.option("--no-transitive", is_flag=True, help="skip transitive resolution")
@click.option("--force-build", is_flag=True,
              help="rebuild native artifacts even if cached (the reference's "
                   "force-docker-build switch: here it drops the in-tree "
                   "object cache and the lpx wheel-cache entries first)")
@click.option("--python", "py_version", default=None,
              help="target python version, e.g. 3.10 (this offline image has "
                   "exactly one interpreter; other versions error loudly)")
@click.option("--json", "as_json", is_flag=True, help="print the report as JSON")
def build(requirements, output, zip_path, handlers, no_transitive, force_build,
          py_version, as_json):
    """Resolve requirements, build recipe wheels, assemble the layer."""
    from .builder import build_recipe_artifacts
    from .packager import assemble_layer
    from .resolver import parse_project_file, resolve

    if py_version is not None:
        have = f"{sys.version_info.major}.{sys.version_info.minor}"
        if py_version not in (have, f"cp{have.replace('.', '')}"):
            raise click.ClickException(
                f"python {py_version} requested but only {have} is available "
                "offline (the reference selects a docker build image; this "
                "image ships a single interpreter)")
    if force_build:
        import glob as _glob
        import subprocess as _sp

        from .builder.wheel_builder import REPO_ROOT, ensure_nn_lib, \
            ensure_standalone_hsaco
        from .builder.wheelgen import WHEEL_CACHE

        removed = 0
        for pat in ("lpx_ops-*.whl", "lpx_shim_*-*.whl", "lpx_handlers-*.whl"):
            for p in _glob.glob(os.path.join(WHEEL_CACHE, pat)):
                os.unlink(p)
                removed += 1
        log.info("force-build: dropped %d cached wheels; recompiling "
                 "gfx950 artifacts...", removed)
        # full recompile now (setuptools --force ignores object mtimes);
        # the nn lib / hsaco mtime caches are invalidated by deletion and
        # rebuilt synchronously below, so the tree is never left bare
        _sp.run([sys.executable, "setup.py", "build_ext", "--inplace",
                 "--force"],
                cwd=REPO_ROOT,
                env=dict(os.environ, PYTORCH_ROCM_ARCH="gfx950"),
                check=True)
        ops_dir = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                               "ops")
        for a in ("liblpx_nn_gfx950.so", "standalone_gfx950.hsaco"):
            if os.path.exists(os.path.join(ops_dir, a)):
                os.unlink(os.path.join(ops_dir, a))
        ensure_nn_lib()
        ensure_standalone_hsaco()

    if requirements is None:
        for cand in ("requirements.txt", "Pipfile.lock", "Pipfile"):
            if os.path.exists(cand):
                requirements = cand
                break
        else:
            raise click.ClickException(
                "no requirements.txt / Pipfile(.lock) found (use -r)")
    elif not os.path.exists(requirements):
        raise click.ClickException(f"{requirements!r} does not exist")
    reqs = parse_project_file(requirements)
    res = resolve(reqs, follow_transitive=not no_transitive)
    log.info("resolved %d deps (%d recipe hits)", len(res.deps), len(res.recipe_hits))
    report = assemble_layer(
        res, output, zip_path=zip_path, wheel_builder=build_recipe_artifacts
    )
    if handlers:
        handlers = list(handlers)
        # every torch handler gets its torch-free fast-path twin shipped
        # alongside (lpx_rt/lpx_nn + AOT artifacts, ~2 MB total) so
        # `lpx run/bench --handler <x>_lite` works on any such layer
        for h in ("rf", "matmul", "fftspmv", "resnet50"):
            if h in handlers and h + "_lite" not in handlers:
                handlers.append(h + "_lite")
        _install_handlers(output, handlers)
        if zip_path:  # re-zip with handlers included
            from .packager import make_zip, zip_report
            make_zip(output, zip_path)
            report.sizes = zip_report(output, zip_path)
    d = report.to_dict()
    if as_json:
        click.echo(json.dumps(d, indent=1))
    else:
        click.echo(f"layer: {output}  ({d.get('unzipped_mb')} MB unzipped"
                   + (f", {d.get('zipped_mb')} MB zipped" if zip_path else "") + ")")
        for dep in d["deps"]:
            click.echo(f"  - {dep['name']}=={dep.get('version','?')} [{dep.get('action')}]")


@main.command()
@click.argument("zip_path", required=False, type=click.Path(exists=True))
@click.option("--recipe", "recipe_name", default=None,
              help="publish a recipe's per-version ROCm wheels instead of a layer")
@click.option("--name", default=None, help="release name (default: filename)")
@click.option("--release-dir", default=DEFAULT_RELEASE_DIR, show_default=True)
def publish(zip_path, recipe_name, name, release_dir):
    """Store artifacts in the local keyed store.

    Offline analog of the reference's GitHub-Releases upload (SURVEY.md
    §3.2): assets are keyed by (name, version, python tag, gfx arch) with a
    sha256 sidecar that fetch verifies. Two asset kinds, mirroring the
    reference's publish path: finished layer zips (positional arg) and
    per-recipe build artifacts (--recipe emits + publishes the recipe's
    ROCm wheels, the analog of recipe x version release assets)."""
    from .store import publish_file

    if recipe_name:
        from .builder.wheel_builder import emit_ops_wheel, emit_shim_wheel
        from .builder.wheelgen import WHEEL_CACHE
        from .recipes import get_registry

        recipe = get_registry().get(recipe_name)
        if recipe is None:
            raise click.ClickException(f"unknown recipe {recipe_name!r}")
        wheels = [emit_ops_wheel(WHEEL_CACHE)]
        if recipe.shim:
            wheels.append(emit_shim_wheel(recipe, WHEEL_CACHE))
        for w in wheels:
            a = publish_file(w, release_dir)
            click.echo(f"published {a.name} {a.version} "
                       f"(py={a.python or '-'}, arch={a.arch or '-'}) "
                       f"-> {a.path}")
        return
    if not zip_path:
        raise click.ClickException("pass a layer zip or --recipe NAME")
    a = publish_file(zip_path, release_dir, name=name)
    click.echo(f"published {a.path}")


@main.command()
@click.argument("layer", type=click.Path(exists=True))
@click.option("--handler", default="rf", show_default=True)
@click.option("--gpus", default=1, show_default=True)
@click.option("--timeout", default=600.0, show_default=True)
@click.option("--json", "as_json", is_flag=True)
def run(layer, handler, gpus, timeout, as_json):
    """Cold-start a handler from a layer (zip or dir) and time to first
    inference."""
    from .runtime import cold_start, prewarm_host

    prewarm_host()
    res = cold_start(layer, handler=handler, gpus=gpus, timeout_s=timeout)
    d = res.to_dict()
    if as_json:
        click.echo(json.dumps(d, indent=1))
    else:
        click.echo(f"cold-start->first-inference: {d['coldstart_ms']} ms "
                   f"({handler}, {gpus} GPU)")


@main.command()
@click.argument("layer", type=click.Path(exists=True))
@click.option("--handler", default="rf", show_default=True)
@click.option("--gpus", default="1", help="comma list, e.g. 1,2,4,8")
@click.option("--repeats", default=1, show_default=True)
@click.option("--out", default=None, help="append JSON lines to this file")
@click.option("--rocprof", "rocprof_dir", default=None,
              help="wrap the g=1 run in rocprofv3 --kernel-trace --stats, "
                   "writing to this directory (MFMA/LDS evidence)")
def bench(layer, handler, gpus, repeats, out, rocprof_dir):
    """Cold-start scaling sweep: layer MB + cold-start ms per GPU count."""
    import shutil as _sh

    from .runtime import cold_start, prewarm_host

    prewarm_host()

    rows = []
    for g in [int(x) for x in gpus.split(",")]:
        wrap = None
        if rocprof_dir and g == 1 and _sh.which("rocprofv3"):
            wrap = ["rocprofv3", "--kernel-trace", "--stats", "-d", rocprof_dir,
                    "-o", f"{handler}_g1", "--"]
        best = None
        for _ in range(repeats):
            r = cold_start(layer, handler=handler, gpus=g, wrap_cmd=wrap)
            if best is None or r.total_ms < best.total_ms:
                best = r
        rows.append(best.to_dict())
        click.echo(json.dumps(rows[-1]))
    if out:
        with open(out, "a") as f:
            for row in rows:
                f.write(json.dumps(row) + "\n")


@main.command()
@click.argument("name")
@click.option("-o", "--output", default=None, help="destination path")
@click.option("--version", default=None, help="exact version to match")
@click.option("--python", "py_tag", default=None, help="python tag, e.g. cp310")
@click.option("--arch", default=None, help="gfx arch, e.g. gfx950")
@click.option("--release-dir", default=DEFAULT_RELEASE_DIR, show_default=True)
def fetch(name, output, version, py_tag, arch, release_dir):
    """Fetch a published artifact from the local keyed store.

    Offline analog of the reference's prebuilt GitHub-Releases download
    (SURVEY.md §2.1 R4): matches on (name, version, python tag, gfx arch),
    verifies the stored sha256 against the file, and copies it out."""
    from .store import StoreError, fetch_asset

    try:
        dest = fetch_asset(name, output, release_dir, version=version,
                           python=py_tag, arch=arch)
    except StoreError as e:
        raise click.ClickException(str(e))
    click.echo(f"fetched {dest}")


@main.command("strip-archs")
@click.argument("paths", nargs=-1, type=click.Path(exists=True))
@click.option("--arch", "archs", multiple=True, default=("gfx950",),
              show_default=True, help="gfx arch(s) to keep")
@click.option("-o", "--output", default=None,
              help="output path (single input only; default: in place)")
@click.option("--json", "as_json", is_flag=True)
def strip_archs(paths, archs, output, as_json):
    """Strip non-target GPU code from HIP shared libraries.

    ROCm DSOs carry clang-offload fat binaries for ~14 gfx architectures;
    for MI355X-only layers everything but gfx950 is dead weight (e.g.
    libtorch_hip.so: 251 MB of fatbin -> ~26 MB). Rewrites the ELF in
    place: rebundles each fat-binary entry, repoints the registration
    wrappers/relocs, and compacts the file."""
    from .packager.fatbin import strip_fatbin

    if output and len(paths) != 1:
        raise click.ClickException("-o requires exactly one input")
    rows = []
    for p in paths:
        st = strip_fatbin(p, output if output else None, archs=tuple(archs))
        rows.append(st)
        if not as_json:
            click.echo(f"{p}: {st['old_size']/1e6:.1f} -> {st['new_size']/1e6:.1f} MB "
                       f"({st['stripped']}/{st['bundles']} bundles stripped)")
    if as_json:
        click.echo(json.dumps(rows, indent=1))


@main.command("stub-deps")
@click.argument("paths", nargs=-1, type=click.Path(exists=True))
@click.option("-o", "--output", default=None,
              help="output path (single input only; default: in place)")
def stub_deps(paths, output):
    """Replace shared libraries with symbol stubs (same SONAME).

    For never-called DT_NEEDED dependencies in a deployment layer (e.g.
    libmagma.so 1.3 GB -> 4 MB): functions trap if called, data symbols
    become zeroed storage, version nodes are replicated."""
    from .packager.stubs import DEFAULT_BENIGN, make_stub

    if output and len(paths) != 1:
        raise click.ClickException("-o requires exactly one input")
    for p in paths:
        base = os.path.basename(p)
        st = make_stub(p, output if output else p,
                       benign=DEFAULT_BENIGN.get(base, ()))
        click.echo(f"{p}: {st['old_size']/1e6:.1f} MB -> {st['new_size']/1e3:.0f} KB "
                   f"({st['symbols']} symbols, versions={st['versions'] or '-'})")


@main.command()
def recipes():
    """List the recipe registry."""
    from .recipes import get_registry

    reg = get_registry()
    for name in reg.names():
        r = reg.get(name)
        click.echo(f"{name:15s} {r.kind:7s} versions={r.versions} kernels={','.join(r.kernels) or '-'}")


if __name__ == "__main__":
    main()
