"""Component rolling updates (hypervisor/worker/client pods).

Reference: internal/component/ (interface :16, ManageUpdate :31):
hash-based change detection of the pool's component pod templates and a
batch rolling-update engine (batch percent + interval) that recreates
out-of-date pods a few at a time.
"""
from .rollout import ComponentRollout, template_hash

__all__ = ["ComponentRollout", "template_hash"]
