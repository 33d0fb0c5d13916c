"""Driver-contract hooks: __graft_entry__.build()/smoke() must exist and
stay importable (the round driver calls build() here on CPU and smoke()
on the MI355X box)."""

import importlib.util
import os


def test_graft_entry_hooks_exist():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    spec = importlib.util.spec_from_file_location(
        "graft_entry_check", os.path.join(repo, "__graft_entry__.py")
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    assert callable(mod.build)
    assert callable(mod.smoke)
