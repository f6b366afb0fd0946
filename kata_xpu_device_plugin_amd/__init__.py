"""kata-xpu-device-plugin-amd — MI355X-native Kubernetes xPU device plugin.

A brand-new, MI355X-first implementation of a Kubernetes DevicePlugin
(v1beta1) + CDI stack for passing AMD Instinct GPUs (vfio-pci bound, whole
GPU or SR-IOV VF) into Kata Containers VMs.

Capability parity target: Apokleos/kata-xpu-device-plugin (reference design
documented in SURVEY.md; reference entry point `cmd/main.go:5-7`,
orchestrator `pkg/device_plugin/device_plugin.go:44-53`). This package is a
clean-room AMD-native design, not a port:

* discovery filters PCI vendor 0x1002 (not 10de) with a curated Instinct
  device table + pci.ids fallback (reference: `device_plugin.go:126-180`,
  `device_plugin.go:208-259`),
* device health comes from inotify on /dev/vfio plus amd-smi / KFD
  (reference used fsnotify only, `generic_device_plugin.go:389-457`),
* `GetPreferredAllocation` is implemented for real with xGMI-hive-aware
  placement (reference stubs it: `generic_device_plugin.go:378-386`),
* CDI devices are one-per-IOMMU-group (the reference's one-per-function
  layout duplicates /dev/vfio device nodes for multi-function groups,
  `device_plugin.go:55-80` — deliberately not reproduced),
* hot paths (sysfs scan, allocation scoring) have native C++ implementations
  (the reference runtime is native Go; ours is C++ via pybind11).
"""

__version__ = "0.1.0"

from .config import Config  # noqa: F401
