"""In-tree build of the native extensions.

* ``kata_xpu_device_plugin_amd._native`` — C++ fast paths (g++; runs on any
  Linux node, no GPU/ROCm needed).
* ``kata_xpu_device_plugin_amd._gpuprobe`` — HIP gfx950 health probes,
  compiled by hipcc (cross-compiles fine on GPU-less builders). Built by
  ``build_hip()`` below / __graft_entry__.build(), not by setuptools, so a
  missing ROCm toolchain never breaks the control-plane build.

Usage: python setup.py build_ext --inplace && python -c "from setup import build_hip; build_hip()"
"""
import os
import subprocess
import sys

import pybind11

ROOT = os.path.dirname(os.path.abspath(__file__))
PKG = "kata_xpu_device_plugin_amd"


def build_hip(arch: str = "gfx950", verbose: bool = True) -> str:
    """Compile probes/xpu_probe.hip → kata_xpu_device_plugin_amd/_gpuprobe.so."""
    import sysconfig

    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    src = os.path.join(ROOT, "probes", "xpu_probe.hip")
    ext = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    out = os.path.join(ROOT, PKG, f"_gpuprobe{ext}")
    if os.path.exists(out) and os.path.getmtime(out) > os.path.getmtime(src):
        return out
    cmd = [
        hipcc, f"--offload-arch={arch}", "-O3", "-std=c++17",
        "-fPIC", "-shared",
        "-x", "hip", src,
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        "-o", out,
    ]
    if verbose:
        print("+", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return out


if __name__ == "__main__" or "setuptools" in sys.modules:
    from setuptools import Extension, setup

    if __name__ == "__main__":
        setup(
            name="kata-xpu-device-plugin-amd",
            version="0.1.0",
            packages=[
                PKG, f"{PKG}.discovery", f"{PKG}.topology", f"{PKG}.cdi",
                f"{PKG}.plugin", f"{PKG}.health", f"{PKG}.utils",
                f"{PKG}.testing", f"{PKG}.tools",
            ],
            # pinned pci.ids snapshot: deterministic naming on air-gapped
            # nodes (first entry of config.pci_ids_paths)
            package_data={PKG: ["data/pci.ids"]},
            include_package_data=True,
            entry_points={
                "console_scripts": [
                    "kata-xpu-device-plugin-amd=kata_xpu_device_plugin_amd.plugin.manager:main",
                    "kxdp-topo=kata_xpu_device_plugin_amd.tools.topo:main",
                    "kxdp-burnin=kata_xpu_device_plugin_amd.tools.burnin:main",
                    "kxdp-bind=kata_xpu_device_plugin_amd.tools.bind:main",
                    "kxdp-sriov=kata_xpu_device_plugin_amd.tools.sriov:main",
                    "kxdp-validate=kata_xpu_device_plugin_amd.tools.validate:main",
                    "kxdp-ident=kata_xpu_device_plugin_amd.tools.ident:main",
                    "kxdp-doctor=kata_xpu_device_plugin_amd.tools.doctor:main",
                    "kxdp-assignments=kata_xpu_device_plugin_amd.tools.assignments:main",
                    "kxdp-resourceslice=kata_xpu_device_plugin_amd.tools.resourceslice:main",
                    "kxdp-partition=kata_xpu_device_plugin_amd.tools.partition:main",
                ],
            },
            ext_modules=[
                Extension(
                    f"{PKG}._native",
                    sources=["native/xpu_native.cpp"],
                    include_dirs=[pybind11.get_include()],
                    extra_compile_args=["-O2", "-std=c++17"],
                    language="c++",
                ),
            ],
        )
