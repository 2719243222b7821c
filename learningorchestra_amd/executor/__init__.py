from .parameters import Parameters  # noqa: F401
from .scheduler import JobScheduler  # noqa: F401
from .execution import Execution, ReflectiveRuntime  # noqa: F401
