from easyparallellibrary_amd.profiler.cost_model import (module_flops,
                                                         profile_flops,
                                                         profile_memory)
from easyparallellibrary_amd.profiler.hooks import (FlopsProfiler,
                                                    MemoryProfiler,
                                                    StepTimer)

__all__ = ["module_flops", "profile_flops", "profile_memory",
           "MemoryProfiler", "FlopsProfiler", "StepTimer"]
