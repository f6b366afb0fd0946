from .sysfs import (  # noqa: F401
    PCIFunction,
    XPUDevice,
    NodeInventory,
    scan_node,
)
from .naming import device_model_name, resource_name  # noqa: F401
