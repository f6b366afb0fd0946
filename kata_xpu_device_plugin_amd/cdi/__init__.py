from .spec import (  # noqa: F401
    ANNOTATION_ATTACH_PCI,
    ANNOTATION_BDF,
    CDISpec,
    build_spec,
    parse_qualified_name,
    qualified_name,
    write_spec,
)
