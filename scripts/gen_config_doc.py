#!/usr/bin/env python3
"""Generate docs/CONFIG.md from the config dataclasses (single source
of truth: scaffold/config.py).  tests/unit/test_config_doc.py asserts
the committed file matches, so the reference can never drift."""

import dataclasses
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scaffold import config as C  # noqa: E402


def _fmt_default(v):
    if dataclasses.is_dataclass(v):
        return "(section)"
    if isinstance(v, str):
        return f'`"{v}"`'
    return f"`{v}`"


def render() -> str:
    lines = [
        "# Config reference",
        "",
        "Generated from `scaffold/config.py` by `scripts/gen_config_doc.py`",
        "(do not edit by hand; a unit test keeps this in sync).  Every key",
        "is overridable from the CLI as `section.key=value`, or set in a",
        "JSON/TOML config file passed with `--config`.",
        "",
    ]
    root = C.TrainConfig
    sections = []
    scalars = []
    for f in dataclasses.fields(root):
        if dataclasses.is_dataclass(f.type) or (
                f.default_factory is not dataclasses.MISSING):
            sections.append(f)
        else:
            scalars.append(f)

    lines += ["## Top level", "", "| key | default | type |", "|---|---|---|"]
    for f in scalars:
        lines.append(f"| `{f.name}` | {_fmt_default(f.default)} | "
                     f"`{getattr(f.type, '__name__', f.type)}` |")
    lines.append("")

    for f in sections:
        cls = (f.default_factory if f.default_factory is not dataclasses.MISSING
               else f.type)
        inst = cls()
        lines += [f"## `{f.name}.*` ({cls.__name__})", "",
                  "| key | default |", "|---|---|"]
        for sf in dataclasses.fields(inst):
            lines.append(
                f"| `{f.name}.{sf.name}` | "
                f"{_fmt_default(getattr(inst, sf.name))} |")
        lines.append("")
    return "\n".join(lines) + "\n"


if __name__ == "__main__":
    out = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "docs", "CONFIG.md")
    with open(out, "w") as fh:
        fh.write(render())
    print(f"wrote {out}")
