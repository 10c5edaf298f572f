"""Every module imports cleanly (catches syntax/import rot in CLIs and
rarely-exercised modules)."""
import importlib
import pkgutil

import pytest

import code_intelligence_amd as pkg

MODULES = sorted(
    m.name for m in pkgutil.walk_packages(pkg.__path__, pkg.__name__ + "."))


@pytest.mark.parametrize("name", MODULES)
def test_module_imports(name):
    if name.endswith("__main__"):
        pytest.skip("entry module")
    importlib.import_module(name)
