from .parser import PollStreamParser, format_record, replay  # noqa: F401
from .state import ACTIVE, INACTIVE, Flow, FlowMeta, FlowTable  # noqa: F401
