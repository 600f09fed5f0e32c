from .builder import RSPBuilder  # noqa: F401
from .engine import (  # noqa: F401
    CrossWindowReasoningMode, OperationMode, QueryExecutionMode, RSPEngine,
)
from .r2r import R2ROperator, SimpleR2R  # noqa: F401
from .r2s import Relation2StreamOperator, StreamOperator  # noqa: F401
from .s2r import (  # noqa: F401
    CSPARQLWindow, ContentContainer, Report, ReportStrategy, Tick,
)
