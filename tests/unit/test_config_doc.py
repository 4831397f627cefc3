"""docs/CONFIG.md is generated from the config dataclasses; this keeps
the committed reference in sync with scaffold/config.py."""

import os


def test_config_doc_in_sync():
    import importlib.util

    repo = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    spec = importlib.util.spec_from_file_location(
        "gen_config_doc", os.path.join(repo, "scripts", "gen_config_doc.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    with open(os.path.join(repo, "docs", "CONFIG.md")) as fh:
        assert fh.read() == mod.render(), (
            "docs/CONFIG.md is stale: run scripts/gen_config_doc.py")
