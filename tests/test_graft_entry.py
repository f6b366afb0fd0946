"""Guard the driver contract entry points in __graft_entry__.py."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_build_entry_point():
    """build() must compile/refresh both native extensions and import the
    package (the driver's does-it-build check)."""
    out = subprocess.run(
        [sys.executable, "-c",
         "import sys; sys.path.insert(0, %r); "
         "import __graft_entry__ as g; g.build()" % REPO],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "build OK" in out.stdout
    import sysconfig
    ext = sysconfig.get_config_var("EXT_SUFFIX")
    pkg = os.path.join(REPO, "kata_xpu_device_plugin_amd")
    assert os.path.exists(os.path.join(pkg, f"_native{ext}"))
    assert os.path.exists(os.path.join(pkg, f"_gpuprobe{ext}"))


def test_smoke_is_gpu_only():
    """smoke() requires a GPU; on CPU it must fail loudly (RuntimeError
    from the probe layer), never silently pass."""
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "graft_entry", os.path.join(REPO, "__graft_entry__.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    import pytest
    with pytest.raises(Exception):
        mod.smoke()
