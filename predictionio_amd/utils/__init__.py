"""Shared utilities namespace.

Most cross-cutting helpers live next to their primary consumers instead:
class resolution (`controller.base.resolve_class`), storage env parsing
(`data.storage`), distributed helpers (`parallel.dist`), kernel build
(`ops.build`). This package is the reserved home for future helpers
that genuinely span layers.
"""
