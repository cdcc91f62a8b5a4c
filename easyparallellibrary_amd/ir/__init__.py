from easyparallellibrary_amd.ir.plan import Plan, TaskGraph

__all__ = ["Plan", "TaskGraph"]
