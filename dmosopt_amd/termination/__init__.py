from dmosopt_amd.termination.basic import (  # noqa: F401
    ConstraintViolationToleranceTermination,
    MaximumGenerationTermination,
    MultiObjectiveToleranceTermination,
    ParameterToleranceTermination,
    SlidingWindowTermination,
    Termination,
    TerminationCollection,
    WindowedCriterion,
    Windows,
)
from dmosopt_amd.termination.adaptive import (  # noqa: F401
    AdaptiveWindowTermination,
    CompositeAdaptiveTermination,
    MultiScaleStagnationTermination,
    PerObjectiveConvergence,
    ResourceAwareTermination,
    create_adaptive_termination,
)
from dmosopt_amd.termination.hv_progress import (  # noqa: F401
    HypervolumeProgressTermination,
)
