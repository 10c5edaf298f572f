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


def test_extension_require_fails_loudly(monkeypatch):
    """GPU ops must refuse to run without the native .so (no silent eager
    fallback) — the require() contract."""
    from code_intelligence_amd.ops import extension as ext
    monkeypatch.setattr(ext, "_ext", None)
    monkeypatch.setattr(ext, "_tried", False)
    monkeypatch.setattr(ext, "_find_so", lambda: None)
    with pytest.raises(RuntimeError, match="HIP extension"):
        ext.require()


def test_extension_loads_in_tree(monkeypatch):
    from code_intelligence_amd.ops import extension as ext
    lib = ext.load(required=False)
    if lib is None:
        pytest.skip("extension not built in this checkout")
    assert "code_intelligence_amd" in lib.__file__
