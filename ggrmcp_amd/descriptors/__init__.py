"""FileDescriptorSet loading (reference pkg/descriptors)."""

from .loader import DescriptorLoader, build_pool, extract_comments  # noqa: F401
