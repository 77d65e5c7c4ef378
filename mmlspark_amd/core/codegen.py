"""Binding/codegen layer — language surfaces generated from the Param registry.

The reference reflects over Scala Params to emit complete PySpark and
sparklyr wrappers (core/.../codegen/CodeGen.scala:196, Wrappable.scala:92,393).
This framework IS Python, so the analogous artifacts generated from the same
single source of truth (the stage registry + Param declarations) are:
  * .pyi type stubs with typed constructors and set/get accessors,
  * markdown API docs per stage,
  * an R wrapper sketch (reticulate-based) mirroring RWrappable.
"""
from __future__ import annotations

import os
from .param import Param
from .registry import all_stages


def _pytype(p: Param) -> str:
    d = p.default
    if isinstance(d, bool):
        return "bool"
    if isinstance(d, int):
        return "int"
    if isinstance(d, float):
        return "float"
    if isinstance(d, str):
        return "str"
    if isinstance(d, (list, tuple)):
        return "list"
    return "object"


def _camel(name: str) -> str:
    return name[0].upper() + name[1:]


def generate_stubs(out_dir: str) -> int:
    """Write one .pyi per module with typed stage classes; returns stage count."""
    import mmlspark_amd
    mmlspark_amd._register_all()
    by_module = {}
    for name, cls in sorted(all_stages().items()):
        by_module.setdefault(cls.__module__, []).append((name, cls))
    os.makedirs(out_dir, exist_ok=True)
    count = 0
    for mod, stages in sorted(by_module.items()):
        lines = ["from typing import Any, Optional", ""]
        for name, cls in stages:
            params = cls.params()
            lines.append(f"class {name}:")
            args = ", ".join(
                [f"{p.name}: {_pytype(p)} = ..." for p in params.values()])
            lines.append(f"    def __init__(self, *, {args}) -> None: ...")
            for p in params.values():
                t = _pytype(p)
                lines.append(f"    def set{_camel(p.name)}(self, value: {t})"
                             f" -> '{name}': ...")
                lines.append(f"    def get{_camel(p.name)}(self) -> {t}: ...")
            for m in ("fit", "transform", "save", "load"):
                if hasattr(cls, m):
                    lines.append(f"    def {m}(self, *args: Any, **kwargs: Any)"
                                 " -> Any: ...")
            lines.append("")
            count += 1
        fname = mod.replace(".", "_") + ".pyi"
        with open(os.path.join(out_dir, fname), "w") as f:
            f.write("\n".join(lines))
    return count


def generate_docs(out_dir: str) -> int:
    """Markdown API reference per stage from Param docs."""
    import mmlspark_amd
    mmlspark_amd._register_all()
    os.makedirs(out_dir, exist_ok=True)
    count = 0
    index = ["# API reference", ""]
    for name, cls in sorted(all_stages().items()):
        lines = [f"# {name}", "", (cls.__doc__ or "").strip(), "", "## Params",
                 "", "| name | default | doc |", "|---|---|---|"]
        for p in cls.params().values():
            lines.append(f"| `{p.name}` | `{p.default!r}` | {p.doc} |")
        with open(os.path.join(out_dir, f"{name}.md"), "w") as f:
            f.write("\n".join(lines) + "\n")
        index.append(f"- [{name}]({name}.md)")
        count += 1
    with open(os.path.join(out_dir, "index.md"), "w") as f:
        f.write("\n".join(index) + "\n")
    return count


def generate_r_wrappers(out_path: str) -> int:
    """sparklyr-style R functions over reticulate (RWrappable analog)."""
    import mmlspark_amd
    mmlspark_amd._register_all()
    lines = ["# Auto-generated R bindings (reticulate)",
             "library(reticulate)",
             "mmlspark_amd <- import(\"mmlspark_amd\")", ""]
    count = 0
    for name, cls in sorted(all_stages().items()):
        params = cls.params()
        arglist = ", ".join(f"{p.name} = NULL" for p in params.values())
        setters = "\n".join(
            f"  if (!is.null({p.name})) stage$set(\"{p.name}\", {p.name})"
            for p in params.values())
        mod = cls.__module__.split(".", 1)[1].replace(".", "$")
        lines.append(
            f"ml_{_snake(name)} <- function({arglist}) {{\n"
            f"  stage <- mmlspark_amd${mod}${name}()\n{setters}\n  stage\n}}\n")
        count += 1
    os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
    with open(out_path, "w") as f:
        f.write("\n".join(lines))
    return count


def _snake(name: str) -> str:
    out = []
    for i, c in enumerate(name):
        if c.isupper() and i and (
                not name[i - 1].isupper()
                or (i + 1 < len(name) and name[i + 1].islower())):
            out.append("_")
        out.append(c.lower())
    return "".join(out)
