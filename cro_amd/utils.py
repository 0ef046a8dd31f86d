"""Small shared helpers (internal/utils/stringutils.go parity)."""

from __future__ import annotations

import uuid


def generate_composable_resource_name(resource_type: str) -> str:
    """``{type}-{uuid4}`` lowercased (stringutils.go:26-33)."""
    return f"{resource_type}-{uuid.uuid4()}".lower()
