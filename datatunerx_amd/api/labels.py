"""Label + event-reason constants (pkg/util/label/label.go:1-35 and
pkg/events/events.go:3-6 parity)."""

from __future__ import annotations

from typing import Dict

GROUP_LABEL = "finetune.datatunerx.io/group"
INSTANCE_LABEL = "finetune.datatunerx.io/instance"
PART_OF_LABEL = "finetune.datatunerx.io/part-of"
MANAGED_BY_LABEL = "finetune.datatunerx.io/managed-by"

MANAGER_NAME = "datatunerx-amd"

# event reasons (pkg/events/events.go)
REASON_CREATED = "Created"
REASON_FAILED = "Failed"


def generate_instance_label(instance: str,
                            extra: Dict[str, str] | None = None
                            ) -> Dict[str, str]:
    """label.go:15-21 GenerateInstanceLabel + merge."""
    out = {INSTANCE_LABEL: instance, MANAGED_BY_LABEL: MANAGER_NAME}
    out.update(extra or {})
    return out
