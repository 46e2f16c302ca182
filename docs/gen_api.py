"""Generate the API reference (docs/api/*.md) from the package's
docstrings — stdlib-only (the image has no sphinx/pdoc).

Usage: python docs/gen_api.py
"""

import importlib
import inspect
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

MODULES = [
    "gossipy_amd",
    "gossipy_amd.core",
    "gossipy_amd.simul",
    "gossipy_amd.node",
    "gossipy_amd.flow_control",
    "gossipy_amd.utils",
    "gossipy_amd.model",
    "gossipy_amd.model.nn",
    "gossipy_amd.model.handler",
    "gossipy_amd.model.sampling",
    "gossipy_amd.data",
    "gossipy_amd.data.handler",
    "gossipy_amd.engine",
    "gossipy_amd.engine.arena",
    "gossipy_amd.engine.models",
    "gossipy_amd.engine.schedule",
    "gossipy_amd.engine.runner",
    "gossipy_amd.engine.backend",
    "gossipy_amd.engine.metrics",
    "gossipy_amd.engine.rng",
    "gossipy_amd.ops",
    "gossipy_amd.ops.build",
]


def _sig(obj):
    try:
        return str(inspect.signature(obj))
    except (ValueError, TypeError):
        return "(...)"


def _doc(obj, indent=""):
    d = inspect.getdoc(obj)
    if not d:
        return ""
    return "\n".join(indent + line for line in d.splitlines()) + "\n"


def render_module(name: str) -> str:
    mod = importlib.import_module(name)
    out = [f"# `{name}`\n"]
    md = inspect.getdoc(mod)
    if md:
        out.append(md + "\n")
    public = getattr(mod, "__all__", None)
    members = inspect.getmembers(mod)
    for mname, obj in members:
        if public is not None and mname not in public:
            continue
        if public is None and mname.startswith("_"):
            continue
        if inspect.ismodule(obj):
            continue
        owner = getattr(obj, "__module__", None)
        if owner is not None and not str(owner).startswith("gossipy_amd"):
            continue
        if inspect.isclass(obj):
            out.append(f"## class `{mname}{_sig(obj)}`\n")
            out.append(_doc(obj))
            for aname, attr in inspect.getmembers(obj):
                if aname.startswith("_") or not (
                    inspect.isfunction(attr) or isinstance(
                        attr, (classmethod, staticmethod, property)
                    )
                ):
                    continue
                if isinstance(attr, property):
                    out.append(f"### property `{mname}.{aname}`\n")
                    out.append(_doc(attr.fget) if attr.fget else "")
                    continue
                fn = attr.__func__ if isinstance(
                    attr, (classmethod, staticmethod)
                ) else attr
                if not _doc(fn):
                    continue
                out.append(f"### `{mname}.{aname}{_sig(fn)}`\n")
                out.append(_doc(fn))
        elif inspect.isfunction(obj):
            out.append(f"## `{mname}{_sig(obj)}`\n")
            out.append(_doc(obj))
    return "\n".join(out)


def main():
    api_dir = os.path.join(ROOT, "docs", "api")
    os.makedirs(api_dir, exist_ok=True)
    index = ["# gossipy_amd API reference\n",
             "Generated from docstrings by `docs/gen_api.py`.\n"]
    for name in MODULES:
        fname = name.replace(".", "_") + ".md"
        with open(os.path.join(api_dir, fname), "w") as f:
            f.write(render_module(name))
        index.append(f"- [`{name}`](api/{fname})")
        print("wrote", fname)
    with open(os.path.join(ROOT, "docs", "API.md"), "w") as f:
        f.write("\n".join(index) + "\n")


if __name__ == "__main__":
    main()
